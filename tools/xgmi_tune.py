"""Measure RCCL all-reduce latency/bandwidth per bucket size → bucket_mb pick.

The reducer's bucket size was chosen in round 1 from a rationale (8 MB ⇒ ~3
in-flight all-reduces for ResNet-18's ~23 MB of bf16 grads); this probe
replaces the rationale with measurement: for each
candidate bucket size it times a ring of back-to-back all-reduces the way
the reducer issues them, and prints per-size GB/s (bus bandwidth, i.e.
2*(N-1)/N * bytes / t) plus the smallest size within 10% of peak — large
enough to saturate the 7-link xGMI fan-out, small enough to overlap.

Run on an 8-GPU node:
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 tools/xgmi_tune.py
Also runs at any world size (including 1, where it measures the local
memcpy-bound path and is only a smoke check).
"""

from __future__ import annotations

import json
import os
import time

import torch
import torch.distributed as dist


def main():
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    use_gpu = torch.cuda.is_available()
    if world > 1:
        if use_gpu:
            torch.cuda.set_device(local_rank)
        dist.init_process_group(backend="nccl" if use_gpu else "gloo")
    dev = torch.device("cuda", local_rank) if use_gpu else torch.device("cpu")

    sizes_mb = [1, 2, 4, 8, 16, 32, 64, 128]
    iters = 20
    results = []
    for mb in sizes_mb:
        n = (mb << 20) // 2  # bf16 elements
        buf = torch.randn(n, dtype=torch.bfloat16 if use_gpu else torch.float32,
                          device=dev)
        # warmup
        for _ in range(5):
            if world > 1:
                dist.all_reduce(buf)
        if use_gpu:
            torch.cuda.synchronize()
        if world > 1:
            dist.barrier()
        t0 = time.perf_counter()
        for _ in range(iters):
            if world > 1:
                dist.all_reduce(buf)
        if use_gpu:
            torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / iters
        bytes_ = buf.numel() * buf.element_size()
        bus_gb = 2 * (world - 1) / max(1, world) * bytes_ / dt / 1e9 if world > 1 else 0.0
        results.append({"mb": mb, "ms": dt * 1e3, "bus_GBps": round(bus_gb, 1)})
        if rank == 0:
            print(f"bucket {mb:4d} MB: {dt*1e3:8.3f} ms  bus {bus_gb:8.1f} GB/s",
                  flush=True)

    if rank == 0 and world > 1:
        peak = max(r["bus_GBps"] for r in results)
        pick = next(r for r in results if r["bus_GBps"] >= 0.9 * peak)
        print(json.dumps({"world": world, "results": results,
                          "recommended_bucket_mb": pick["mb"]}))
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
