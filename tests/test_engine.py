"""Engine tests: flags, setup, train loop, checkpoint layout + resume,
evaluate — the reference API surface (SURVEY.md §7 checklist)."""

import os

import pytest
import torch

from pytorch_ddp_template_amd.ddp import (
    build_parser,
    cleanup,
    evaluate,
    find_latest_checkpoint,
    load_checkpoint,
    save_model,
    setup,
    train,
)
from pytorch_ddp_template_amd.models import build_model


def make_args(tmp_path, extra=()):
    args = build_parser().parse_args(
        [
            "--no_cuda",
            "--output_dir",
            str(tmp_path / "outputs"),
            "--dataset_size",
            "256",
            "--max_steps",
            "4",
            "--logging_steps",
            "2",
            "--save_steps",
            "2",
            "--no_tensorboard",
            *extra,
        ]
    )
    args.dataset = args.dataset or args.model
    return args


def test_reference_flags_present_with_defaults():
    # flag names + defaults 1:1 with reference ddp.py:293-308
    args = build_parser().parse_args([])
    assert args.global_step == 0
    assert args.output_dir == "outputs"
    assert args.seed == 42
    assert args.gradient_accumulation_steps == 1
    assert args.per_gpu_train_batch_size == 32
    assert args.max_steps == 0
    assert args.logging_steps == 100
    assert args.save_steps == 1000
    assert args.num_train_epochs == 10
    assert args.warmup_steps == 100
    assert args.max_grad_norm == 1000.0
    assert args.local_rank == -1
    assert args.fp16 is False
    assert args.loss_scale == 0
    assert args.fp16_opt_level == "O2"


def test_setup_single_process(tmp_path):
    args = make_args(tmp_path)
    setup(args)
    assert args.device.type == "cpu"
    assert args.world_size == 1
    assert args.train_batch_size == 32
    assert args.node_rank == 0


def test_setup_reads_local_rank_env(tmp_path, monkeypatch):
    # env LOCAL_RANK overrides the flag (reference ddp.py:85) — but -1 keeps
    # single-process mode
    monkeypatch.setenv("LOCAL_RANK", "-1")
    args = make_args(tmp_path, ["--local_rank", "7"])
    setup(args)
    assert args.local_rank == -1


def test_train_checkpoints_and_max_steps(tmp_path):
    args = make_args(tmp_path)
    setup(args)
    model = build_model("foo")
    gs, avg = train(args, model)
    # exact stop at max_steps (reference overran by one — we fixed it)
    assert gs == 4
    # checkpoint layout: outputs/checkpoint-{step}/{model.bin,
    # training_args.bin, optimizer.pt, scheduler.pt} (reference ddp.py:256-277)
    for step in (2, 4):
        d = os.path.join(args.output_dir, f"checkpoint-{step}")
        for f in ("model.bin", "training_args.bin", "optimizer.pt",
                  "scheduler.pt"):
            assert os.path.exists(os.path.join(d, f)), (d, f)
    cleanup(args)


def test_save_model_rejects_file_path(tmp_path):
    p = tmp_path / "somefile"
    p.write_text("x")
    with pytest.raises(ValueError):
        save_model(build_model("foo"), str(p))


def test_resume_roundtrip(tmp_path):
    args = make_args(tmp_path)
    setup(args)
    model = build_model("foo")
    train(args, model)
    latest = find_latest_checkpoint(args.output_dir)
    assert latest.endswith("checkpoint-4")

    # fresh model+opt; resume restores weights, optimizer, scheduler, step
    model2 = build_model("foo")
    from pytorch_ddp_template_amd.optim import (
        SGD,
        get_linear_schedule_with_warmup,
    )

    opt = SGD(model2.parameters(), lr=1e-3)
    sched = get_linear_schedule_with_warmup(opt, 10, 100)
    step = load_checkpoint(args, model2, opt, sched)
    assert step == 4
    sd1 = model.state_dict()
    sd2 = model2.state_dict()
    for k in sd1:
        torch.testing.assert_close(sd1[k], sd2[k], rtol=0, atol=0)
    assert sched.last_epoch == 4  # scheduler state restored


def test_resume_from_global_step_flag(tmp_path):
    args = make_args(tmp_path)
    setup(args)
    train(args, build_model("foo"))
    # the reference parsed --global-step but never used it; we wire it
    args2 = make_args(tmp_path)
    args2.global_step = 2
    args2.max_steps = 6
    setup(args2)
    model = build_model("foo")
    gs, _ = train(args2, model)
    assert gs == 6


def test_evaluate_returns_metrics(tmp_path):
    args = make_args(tmp_path)
    setup(args)
    model = build_model("foo")
    res = evaluate(args, model, max_batches=2)
    assert "eval_loss" in res and res["eval_loss"] >= 0


def test_gradient_accumulation_runs(tmp_path):
    args = make_args(tmp_path, ["--gradient_accumulation_steps", "2"])
    setup(args)
    gs, _ = train(args, build_model("foo"))
    assert gs == 4


def test_fp16_loss_scaler_path(tmp_path):
    # native fp16 path (the reference's --fp16 crashed with a NameError,
    # ddp.py:172); CPU half matmuls are slow but a 2-step run verifies the
    # scaler plumbing end-to-end
    args = make_args(tmp_path, ["--fp16", "--per_gpu_train_batch_size", "8"])
    args.max_steps = 2
    setup(args)
    gs, avg = train(args, build_model("foo"))
    assert gs == 2
    assert avg == avg  # not NaN


def test_loss_scaler_dynamic():
    import torch as _t

    from pytorch_ddp_template_amd.ddp import LossScaler

    s = LossScaler(0.0)
    assert s.scale == 65536.0
    assert s.step_ok(_t.tensor(1.0))
    before = s.scale
    assert not s.step_ok(_t.tensor(float("inf")))
    assert s.scale == before / 2
    s2 = LossScaler(128.0)
    assert s2.scale == 128.0
    assert not s2.step_ok(_t.tensor(float("nan")))
    assert s2.scale == 128.0  # static scale never changes


def test_mid_epoch_resume_exact(tmp_path):
    """Resume from checkpoint-2 (saved mid-epoch) continues on batch 2, not
    batch 0: the resumed run's final weights match the unbroken run exactly.
    Round 1 restored step/weights/optimizer but replayed data (VERDICT)."""
    args = make_args(tmp_path)
    setup(args)
    model = build_model("foo")
    train(args, model)  # unbroken 4-step run; checkpoints at 2 and 4
    sd_unbroken = {
        k: v.clone()
        for k, v in torch.load(
            os.path.join(args.output_dir, "checkpoint-4", "model.bin"),
            map_location="cpu",
            weights_only=True,
        ).items()
    }

    args2 = make_args(tmp_path)
    args2.global_step = 2  # resume from checkpoint-2
    setup(args2)
    model2 = build_model("foo")
    gs, _ = train(args2, model2)
    assert gs == 4
    sd_resumed = model2.state_dict()
    for k in sd_unbroken:
        torch.testing.assert_close(
            sd_resumed[k], sd_unbroken[k], rtol=0, atol=0
        ), k


def test_training_state_saved(tmp_path):
    args = make_args(tmp_path)
    setup(args)
    train(args, build_model("foo"))
    st = torch.load(
        os.path.join(args.output_dir, "checkpoint-2", "training_state.pt"),
        map_location="cpu",
        weights_only=False,
    )
    assert st["epoch"] == 0
    assert st["batches_in_epoch"] == 2
    assert "torch" in st["rng"]


def test_evaluate_partial_batch_denominator(tmp_path):
    """Accuracy denominator counts actual samples (round 1 used
    n_batches * batch_size, overcounting on a partial final batch)."""
    args = make_args(tmp_path)
    args.model = args.dataset = "resnet18"
    args.dataset_size = 40  # batch 32 -> one full + one partial batch
    setup(args)
    model = build_model("resnet18")
    res = evaluate(args, model)
    assert "eval_acc" in res
    assert 0.0 <= res["eval_acc"] <= 1.0


def test_hip_graph_flag_ignored_on_cpu(tmp_path):
    """--hip_graph on CPU must fall back to the eager path (warn, not
    crash)."""
    from pytorch_ddp_template_amd.ddp import main

    main([
        "--model", "foo", "--dataset_size", "64",
        "--per_gpu_train_batch_size", "16", "--max_steps", "2",
        "--no_cuda", "--hip_graph", "--output_dir", str(tmp_path / "o"),
        "--num_workers", "0", "--no_tensorboard", "--no_progress_bar",
        "--logging_steps", "1", "--save_steps", "0",
    ])

def test_fp16_scaler_scale_survives_checkpoint(tmp_path):
    """Dynamic loss scale persists through save/resume (a resumed fp16 run
    otherwise re-converges its scale from the default via overflow skips)."""
    import torch

    from pytorch_ddp_template_amd.ddp import (
        LossScaler, load_training_state, save_checkpoint,
    )

    import argparse

    A = argparse.Namespace(output_dir=str(tmp_path), n_gpu=0)

    m = torch.nn.Linear(4, 4)
    opt = torch.optim.SGD(m.parameters(), lr=0.1)
    sch = torch.optim.lr_scheduler.LambdaLR(opt, lambda s: 1.0)
    sc = LossScaler(0)
    sc.scale = 1024.0
    save_checkpoint(A, m, opt, sch, 7, scaler=sc)
    sc2 = LossScaler(0)
    load_training_state(A, str(tmp_path / "checkpoint-7"), scaler=sc2)
    assert sc2.scale == 1024.0
