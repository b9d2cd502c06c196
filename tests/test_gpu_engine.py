"""End-to-end engine runs on the GPU — the reference-parity CLI path
(setup -> train -> checkpoint -> cleanup) driving the HIP kernels, in bf16
and in fp16 with the native loss scaler (the reference's --fp16 slot,
reference ddp.py:165-181, rebuilt without apex)."""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu


def _run(tmp_path, extra):
    from pytorch_ddp_template_amd.ddp import build_parser, main

    argv = [
        "--model", "resnet18", "--dataset", "cifar", "--dataset_size", "512",
        "--per_gpu_train_batch_size", "64", "--max_steps", "3",
        "--logging_steps", "1", "--save_steps", "2",
        "--output_dir", str(tmp_path / "out"), "--num_workers", "0",
        "--no_tensorboard", "--no_progress_bar", "--seed", "3",
    ] + extra
    main(argv)
    return tmp_path / "out"


def test_engine_bf16_e2e(tmp_path):
    out = _run(tmp_path, ["--bf16"])
    ck = out / "checkpoint-2"
    assert (ck / "model.bin").exists()
    assert (ck / "optimizer.pt").exists()
    sd = torch.load(ck / "model.bin", map_location="cpu", weights_only=True)
    assert any(v.dtype == torch.bfloat16 for v in sd.values())


def test_engine_fp16_scaler_e2e(tmp_path):
    out = _run(tmp_path, ["--fp16", "--loss_scale", "0"])  # dynamic scale
    assert (out / "checkpoint-2" / "model.bin").exists()


def test_engine_hip_graph_e2e(tmp_path):
    """--hip_graph: the ws=1 train loop's captured step (GraphStep) must
    train through warmup + capture + replays, keep a sane loss, and still
    checkpoint.  Weights must actually CHANGE across replays (a frozen
    capture — e.g. stale plan pointers — would leave them fixed)."""
    from pytorch_ddp_template_amd.models import build_model

    out = _run(tmp_path, ["--bf16", "--hip_graph", "--max_steps", "8",
                          "--save_steps", "8", "--learning_rate", "0.05",
                          "--momentum", "0.9"])
    ck = out / "checkpoint-8"
    assert (ck / "model.bin").exists()
    sd = torch.load(ck / "model.bin", map_location="cpu", weights_only=True)
    init = build_model("resnet18", 10).to(torch.bfloat16).state_dict()
    moved = sum(
        1 for k in sd
        if k in init and sd[k].shape == init[k].shape
        and not torch.equal(sd[k], init[k])
    )
    assert moved > 10, f"only {moved} tensors changed — frozen capture?"
    for v in sd.values():
        assert torch.isfinite(v.float()).all()


def test_engine_resume_steps_after_load(tmp_path):
    """Resume must TRAIN after loading (regression: torch's state-dtype
    cast crashed the fused SGD on the first post-resume step with bf16
    params — the old roundtrip test loaded but never stepped)."""
    out = _run(tmp_path, ["--bf16", "--max_steps", "4", "--save_steps", "2",
                          "--momentum", "0.9"])
    assert (out / "checkpoint-2").exists()
    # second run resumes from step 2 and must complete steps 3..4
    _run(tmp_path, ["--bf16", "--max_steps", "4", "--save_steps", "0",
                    "--momentum", "0.9", "--global-step", "2"])
