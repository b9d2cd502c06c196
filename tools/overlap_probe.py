"""Comm/compute overlap probe, runnable at N=1.

Workload mode (default): run a few flagship DDP steps through the native
reducer under a ws=1 RCCL process group, so a kernel trace shows the RCCL
all-reduce kernels on their comm stream against the backward's compute
kernels.  Profile it with:

    rocprofv3 --kernel-trace --output-format csv -d gpurun_out/ovl -- \
        python -m torch.distributed.run --nnodes=1 --nproc-per-node 1 \
        --master-addr 127.0.0.1 tools/overlap_probe.py

Analyze mode: ``python tools/overlap_probe.py --analyze <kernel_trace.csv>``
reports, per stream, kernel time and how much of the RCCL streams' busy time
intersects compute-stream busy time (the overlap fraction the reducer design
claims).
"""

from __future__ import annotations

import argparse
import csv
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def run_workload(steps: int):
    import torch
    import torch.distributed as dist

    from pytorch_ddp_template_amd.models import build_model
    from pytorch_ddp_template_amd.ops import CrossEntropyLoss
    from pytorch_ddp_template_amd.optim import SGD
    from pytorch_ddp_template_amd.parallel import DistributedModel

    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    torch.cuda.set_device(local_rank)
    dist.init_process_group(backend="nccl")
    dev = torch.device("cuda", local_rank)
    torch.manual_seed(0)
    model = build_model("resnet18").to(torch.bfloat16).to(dev)
    opt = SGD(model.parameters(), lr=0.1, momentum=0.9, master_weights=True)
    model = DistributedModel(model, bucket_bytes=8 << 20)
    crit = CrossEntropyLoss()
    x = torch.randn(2048, 32, 32, 3, dtype=torch.bfloat16, device=dev)
    y = torch.randint(0, 10, (2048,), device=dev)
    for _ in range(steps):
        loss = crit(model(x), y)
        loss.backward()
        model.finish_gradient_sync()
        opt.step()
        model.zero_grad()
    torch.cuda.synchronize()
    print(f"overlap probe: {steps} steps done, loss={float(loss.detach()):.3f}")
    dist.destroy_process_group()


def _intervals_union(iv):
    iv.sort()
    out = []
    for s, e in iv:
        if out and s <= out[-1][1]:
            out[-1][1] = max(out[-1][1], e)
        else:
            out.append([s, e])
    return out


def _intersect_len(a, b):
    i = j = 0
    total = 0
    while i < len(a) and j < len(b):
        s = max(a[i][0], b[j][0])
        e = min(a[i][1], b[j][1])
        if e > s:
            total += e - s
        if a[i][1] < b[j][1]:
            i += 1
        else:
            j += 1
    return total


def analyze(path: str):
    rows = list(csv.DictReader(open(path)))
    if not rows:
        print("empty trace")
        return
    cols = rows[0].keys()
    name_c = next(c for c in cols if c.lower() in ("kernel_name", "name"))
    start_c = next(c for c in cols if "start" in c.lower())
    end_c = next(c for c in cols if "end" in c.lower())
    stream_c = next(
        (c for c in cols if "stream" in c.lower() or "queue" in c.lower()), None
    )
    by_stream: dict[str, list] = {}
    rccl_streams, compute_streams = set(), set()
    for r in rows:
        sid = r[stream_c] if stream_c else "0"
        s, e = int(r[start_c]), int(r[end_c])
        by_stream.setdefault(sid, []).append([s, e])
        if "rccl" in r[name_c].lower() or "nccl" in r[name_c].lower():
            rccl_streams.add(sid)
        else:
            compute_streams.add(sid)
    compute_streams -= rccl_streams
    comp_iv = _intervals_union(
        [iv for sid in compute_streams for iv in by_stream[sid]]
    )
    print(f"streams: {len(by_stream)} (rccl: {sorted(rccl_streams)})")
    for sid in sorted(by_stream):
        iv = _intervals_union(list(by_stream[sid]))
        busy = sum(e - s for s, e in iv)
        kind = "RCCL" if sid in rccl_streams else "compute"
        line = f"stream {sid} [{kind}]: busy {busy/1e6:.2f} ms"
        if sid in rccl_streams:
            ovl = _intersect_len(iv, comp_iv)
            line += f", overlapped with compute {ovl/1e6:.2f} ms ({100*ovl/max(1,busy):.0f}%)"
        print(line)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--analyze", type=str, default=None,
                   help="kernel_trace.csv from rocprofv3 --kernel-trace")
    p.add_argument("--steps", type=int, default=5)
    args = p.parse_args()
    if args.analyze:
        analyze(args.analyze)
    else:
        run_workload(args.steps)


if __name__ == "__main__":
    main()
