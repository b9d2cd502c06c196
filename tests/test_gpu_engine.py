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
