"""Per-rank training entry point — the reference's ddp.py rebuilt MI355X-first.

API fidelity (SURVEY.md §7 checklist):
* flag names + defaults match reference ddp.py:293-308 (plus new, clearly
  additive flags for the benchmark model zoo),
* env contract: ``LOCAL_RANK`` overrides ``--local_rank`` (reference
  ddp.py:85); launcher sets RANK/WORLD_SIZE/MASTER_ADDR/MASTER_PORT,
* entry points ``setup`` / ``train`` / ``evaluate`` / ``cleanup`` /
  ``save_model`` (reference ddp.py:80,126,123,118,64),
* checkpoint layout ``<output_dir>/checkpoint-{global_step}/{model.bin,
  training_args.bin, optimizer.pt, scheduler.pt}`` (reference ddp.py:256-277),
* sampler ``set_epoch`` per epoch (ddp.py:213-214), main-rank-only
  TensorBoard/checkpoint gating (ddp.py:246,255),
* logging convention ``log.info(msg, dict(k=v))`` (reference utils.py:18-21).

Deliberate fixes over the reference (documented quirks):
* ``--global-step`` is wired to checkpoint resume (it was parsed-but-dead at
  reference ddp.py:293; resume never existed — SURVEY.md §5.4),
* the ``--fp16`` path is native bf16/fp16 with fp32 master weights instead
  of the broken apex branch (reference ddp.py:172 called an undefined
  ``FusedSGD`` → NameError),
* the ``max_steps`` off-by-one (reference ddp.py:280 used ``>`` so it ran one
  extra step) is fixed to stop exactly at ``max_steps``,
* loss readback is periodic (logging boundaries only) instead of the
  per-step ``loss.item()`` device sync (reference ddp.py:232,234),
* single-process multi-GPU DataParallel (reference ddp.py:189-191) is
  documented out: one process per GPU is the only multi-GPU mode.
"""

from __future__ import annotations

import argparse
import glob
import os
import random
import re

import numpy as np
import torch
from torch.utils.data import DataLoader

from .data import CudaPrefetcher, ShardedSampler, build_dataset
from .models import build_model
from .optim import SGD, clip_grad_norm_, get_linear_schedule_with_warmup
from .ops import scale_grads_
from .ops import CrossEntropyLoss, MSELoss
from .parallel import DistributedModel
from .utils import (
    getLoggerWithRank,
    is_main_process,
    redirect_warnings_to_logger,
)

try:  # reference ddp.py:36-39 — tensorboard with tensorboardX fallback
    from torch.utils.tensorboard import SummaryWriter
except Exception:  # pragma: no cover
    try:
        from tensorboardX import SummaryWriter
    except Exception:
        SummaryWriter = None

logger = None


def set_seed(args):
    """Seed random/numpy/torch (+ all GPUs) — reference ddp.py:44-49.

    Same seed on every rank; shard divergence comes from the sampler.
    """
    random.seed(args.seed)
    np.random.seed(args.seed)
    torch.manual_seed(args.seed)
    if args.n_gpu > 0:
        torch.cuda.manual_seed_all(args.seed)


def setup(args):
    """Device/backend selection + process group init — reference ddp.py:80-115.

    Mutates args in place: device, n_gpu, node_rank (holds the GLOBAL rank —
    the reference's misnamed field, kept for API parity, ddp.py:104),
    world_size, train_batch_size.
    """
    global logger
    # launcher env overrides the flag (reference ddp.py:85)
    args.local_rank = int(os.environ.get("LOCAL_RANK", args.local_rank))
    logger = getLoggerWithRank(
        __name__, int(os.environ.get("RANK", 0)), args.local_rank
    )
    redirect_warnings_to_logger(logger)

    if args.local_rank == -1:
        # single-process path (reference ddp.py:90-98). DataParallel is
        # documented out: with several visible GPUs we still use one.
        use_gpu = torch.cuda.is_available() and not args.no_cuda
        args.device = torch.device("cuda:0" if use_gpu else "cpu")
        args.n_gpu = 1 if use_gpu else 0
        args.node_rank = 0
        args.world_size = 1
    else:
        # one process per device; gloo on CPU (--no_cuda or no GPU) keeps the
        # whole distributed path runnable hardware-free (BASELINE config 1)
        use_gpu = torch.cuda.is_available() and not args.no_cuda
        if use_gpu:
            torch.cuda.set_device(args.local_rank)
            args.device = torch.device("cuda", args.local_rank)
            backend = "nccl"  # = RCCL over xGMI on ROCm
        else:
            # CPU plumbing rung (BASELINE.json config 1): gloo
            args.device = torch.device("cpu")
            backend = "gloo"
        torch.distributed.init_process_group(backend=backend)
        args.node_rank = torch.distributed.get_rank()  # reference ddp.py:104
        args.world_size = torch.distributed.get_world_size()
        args.n_gpu = 1  # forced (reference ddp.py:108)
    args.train_batch_size = args.per_gpu_train_batch_size * max(1, args.n_gpu)
    set_seed(args)
    logger.info(
        "Runtime setup complete.",
        dict(
            device=str(args.device),
            n_gpu=args.n_gpu,
            world_size=args.world_size,
            local_rank=args.local_rank,
        ),
    )


def cleanup(args):
    """destroy_process_group iff distributed — reference ddp.py:118-121."""
    if args.local_rank != -1 and torch.distributed.is_initialized():
        torch.distributed.destroy_process_group()


def save_model(model, output_dir):
    """state_dict -> <dir>/model.bin — reference ddp.py:64-77."""
    if os.path.isfile(output_dir):
        raise ValueError(
            f"output_dir ({output_dir}) should be a directory, not a file"
        )
    os.makedirs(output_dir, exist_ok=True)
    model_to_save = model.module if hasattr(model, "module") else model
    torch.save(model_to_save.state_dict(), os.path.join(output_dir, "model.bin"))


def save_checkpoint(
    args, model, optimizer, scheduler, global_step, epoch=0,
    batches_in_epoch=0, scaler=None
):
    """Full checkpoint in the reference layout (ddp.py:255-277), plus
    ``training_state.pt`` (epoch / in-epoch position / RNG states) so a
    resume continues mid-epoch without replaying data — the reference never
    loaded at all."""
    output_dir = os.path.join(args.output_dir, f"checkpoint-{global_step}")
    save_model(model, output_dir)
    torch.save(args, os.path.join(output_dir, "training_args.bin"))
    torch.save(optimizer.state_dict(), os.path.join(output_dir, "optimizer.pt"))
    torch.save(scheduler.state_dict(), os.path.join(output_dir, "scheduler.pt"))
    state = {
        "epoch": epoch,
        "batches_in_epoch": batches_in_epoch,
        "rng": {
            "torch": torch.get_rng_state(),
            "numpy": np.random.get_state(),
            "random": random.getstate(),
        },
    }
    if scaler is not None:
        # fp16 dynamic loss scale: resume at the calibrated scale instead
        # of re-converging from the default through overflow skips
        state["scaler_scale"] = float(scaler.scale)
    if torch.cuda.is_available() and args.n_gpu > 0:
        state["rng"]["cuda"] = torch.cuda.get_rng_state_all()
    torch.save(state, os.path.join(output_dir, "training_state.pt"))
    logger.info("Saved checkpoint.", dict(dir=output_dir, step=global_step))


def find_latest_checkpoint(output_dir):
    ckpts = []
    for d in glob.glob(os.path.join(output_dir, "checkpoint-*")):
        m = re.match(r".*checkpoint-(\d+)$", d)
        if m and os.path.isdir(d):
            ckpts.append((int(m.group(1)), d))
    return max(ckpts)[1] if ckpts else None


def load_checkpoint(args, model, optimizer=None, scheduler=None, path=None):
    """Resume — the subsystem the reference never implemented.

    Returns the global_step encoded in the checkpoint directory name.
    """
    if path is None:
        path = find_latest_checkpoint(args.output_dir)
    if path is None:
        return 0
    target = model.module if hasattr(model, "module") else model
    state = torch.load(
        os.path.join(path, "model.bin"), map_location="cpu", weights_only=True
    )
    target.load_state_dict(state)
    if optimizer is not None and os.path.exists(os.path.join(path, "optimizer.pt")):
        optimizer.load_state_dict(
            torch.load(
                os.path.join(path, "optimizer.pt"),
                map_location="cpu",
                weights_only=True,
            )
        )
    if scheduler is not None and os.path.exists(os.path.join(path, "scheduler.pt")):
        scheduler.load_state_dict(
            torch.load(
                os.path.join(path, "scheduler.pt"),
                map_location="cpu",
                weights_only=True,
            )
        )
    m = re.match(r".*checkpoint-(\d+)$", path)
    step = int(m.group(1)) if m else 0
    logger.info("Resumed from checkpoint.", dict(dir=path, step=step))
    return step


def load_training_state(args, path, scaler=None):
    """Restore epoch / in-epoch position / RNG states (and the fp16 loss
    scale) saved by save_checkpoint.  Returns (epoch, batches_in_epoch);
    (0, 0) when the checkpoint predates training_state.pt."""
    f = os.path.join(path, "training_state.pt")
    if not os.path.exists(f):
        return 0, 0
    state = torch.load(f, map_location="cpu", weights_only=False)
    if scaler is not None and "scaler_scale" in state:
        scaler.scale = float(state["scaler_scale"])
    rng = state.get("rng", {})
    if "torch" in rng:
        torch.set_rng_state(rng["torch"])
    if "numpy" in rng:
        np.random.set_state(rng["numpy"])
    if "random" in rng:
        random.setstate(rng["random"])
    if "cuda" in rng and torch.cuda.is_available() and args.n_gpu > 0:
        try:
            torch.cuda.set_rng_state_all(rng["cuda"])
        except Exception:  # device count mismatch across restarts
            pass
    return state.get("epoch", 0), state.get("batches_in_epoch", 0)


class LossScaler:
    """fp16 loss scaling: static (``--loss_scale N``) or dynamic (N == 0,
    the reference's apex-style default — ddp.py:174-180 intent, natively).

    On GPU the overflow check is fully device-side: the fused SGD kernel
    skips the update when the (pre-unscale) grad norm is non-finite and
    ticks a device counter; the host reads that counter once per
    ``sync_interval`` optimizer steps to adjust the scale — NOT once per
    accumulation step (the round-1 per-step ``isfinite().all()`` host sync
    was the exact stall pattern the bf16 path avoids)."""

    def __init__(self, static_scale: float = 0.0, sync_interval: int = 50):
        self.dynamic = static_scale == 0
        self.scale = static_scale if static_scale > 0 else 65536.0
        self.growth_interval = 2000
        self._good_steps = 0
        self.sync_interval = sync_interval
        self._since_sync = 0
        self._skip_dev = None

    def skip_counter(self, device):
        if self._skip_dev is None or self._skip_dev.device != torch.device(device):
            self._skip_dev = torch.zeros(
                (), dtype=torch.float32, device=device
            )
        return self._skip_dev

    def tick(self):
        """Called once per optimizer step (GPU path); syncs the device skip
        counter only every sync_interval steps (one readback)."""
        self._since_sync += 1
        self._good_steps += 1
        if self._since_sync < self.sync_interval:
            return
        self._since_sync = 0
        skips = float(self._skip_dev) if self._skip_dev is not None else 0.0
        if skips > 0:
            if self.dynamic:
                self.scale = max(1.0, self.scale / (2.0 ** min(skips, 16.0)))
            self._good_steps = 0
            if self._skip_dev is not None:
                self._skip_dev.zero_()
        elif self.dynamic and self._good_steps >= self.growth_interval:
            self.scale *= 2.0
            self._good_steps = 0

    def step_ok(self, total_norm) -> bool:
        """Synchronous variant (CPU path / tests): one host sync per call."""
        finite = bool(torch.isfinite(total_norm).all())
        if not self.dynamic:
            return finite
        if finite:
            self._good_steps += 1
            if self._good_steps >= self.growth_interval:
                self.scale *= 2.0
                self._good_steps = 0
            return True
        self.scale = max(1.0, self.scale / 2.0)
        self._good_steps = 0
        return False


class GraphStep:
    """hipGraph-captured training step for the ws=1 CLI engine.

    Same recipe the flagship bench uses (bench.py --graph): three eager
    warmup iterations on a side stream (so AccumulateGrad stream metadata
    matches the capture stream), then one capture of
    fwd+loss+bwd+clip+fused-SGD+in-place-zero with pointer-stable
    grads/optimizer state, replayed per step with a D2D input refresh and
    the LR read from a device scalar updated from the LIVE schedule.
    Removes the per-kernel launch tail; no work is skipped."""

    WARMUP = 3

    def __init__(self, model, criterion, optimizer, max_grad_norm):
        self.model, self.criterion = model, criterion
        self.optimizer, self.max_grad_norm = optimizer, max_grad_norm
        self.side = torch.cuda.Stream()
        self.graph = None
        self.loss_st = None
        self.x_st = self.y_st = None
        self.lr_dev = None
        self._warm = 0

    def _inner(self):
        out = self.model(self.x_st)
        loss = self.criterion(out, self.y_st)
        loss.backward()
        clip_grad_norm_(list(self.model.parameters()), self.max_grad_norm)
        self.optimizer.step(lr_tensor=self.lr_dev)
        self.model.zero_grad(set_to_none=False)
        return loss

    @property
    def active(self):
        return self.graph is not None

    def step(self, x, y):
        """Run one full optimizer step (captured once ready); returns the
        loss tensor (device scalar, no host sync)."""
        if self.x_st is None:
            self.x_st, self.y_st = x.clone(), y.clone()
            self.lr_dev = torch.zeros((), dtype=torch.float32, device=x.device)
        else:
            self.x_st.copy_(x)
            self.y_st.copy_(y)
        self.lr_dev.fill_(self.optimizer.param_groups[0]["lr"])
        if self.graph is None:
            if self._warm < self.WARMUP:
                self.side.wait_stream(torch.cuda.current_stream())
                with torch.cuda.stream(self.side):
                    loss = self._inner()
                torch.cuda.current_stream().wait_stream(self.side)
                self._warm += 1
                return loss.detach()
            torch.cuda.synchronize()
            self.graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self.graph, stream=self.side):
                self.loss_st = self._inner()
            # capture RECORDS the work without executing it: replay now so
            # this batch still trains
        self.graph.replay()
        return self.loss_st.detach()


def _criterion_for(args):
    if args.model in ("foo", "foomodel", "mlp"):
        return MSELoss()  # reference ddp.py:164
    return CrossEntropyLoss()


def _cast_model(args, model):
    if args.bf16 or (args.fp16 and args.fp16_opt_level in ("O2", "O3")):
        dtype = torch.bfloat16 if args.bf16 else torch.float16
        model = model.to(dtype)
        # keep norm/scalar params fp32? O2-style: norms stay low precision
        # with fp32 master in the optimizer (simple + numerically fine for
        # BN thanks to fp32 stat accumulation in the kernels).
    return model


def evaluate(args, model, dataset=None, max_batches=None):
    """Evaluation loop (the reference left this as a stub, ddp.py:123-124).

    Distributed-correct: each rank evaluates its deterministic shard
    (ShardedSampler, shuffle off, drop_last so no padded duplicates skew the
    metrics), per-rank [loss_sum, n_samples, n_correct] are all-reduced, and
    accuracy uses the TRUE sample count as denominator (round 1 divided by
    n_batches * batch_size, wrong on a partial final batch).
    """
    if dataset is None:
        dataset = build_dataset(args.dataset, size=min(args.dataset_size, 10000))
    distributed = args.local_rank != -1 and torch.distributed.is_initialized()
    sampler = (
        ShardedSampler(
            dataset, num_replicas=args.world_size, rank=args.node_rank,
            shuffle=False, drop_last=True,
        )
        if distributed
        else None
    )
    loader = DataLoader(
        dataset,
        sampler=sampler,
        batch_size=args.train_batch_size,
        pin_memory=args.device.type == "cuda",
    )
    criterion = _criterion_for(args)
    model.eval()
    # [loss_sum (sample-weighted), n_samples, n_correct] — one all-reduce
    totals = torch.zeros(3, dtype=torch.float64)
    with torch.no_grad():
        for i, (x, y) in enumerate(loader):
            if max_batches is not None and i >= max_batches:
                break
            x = x.to(args.device)
            y = y.to(args.device)
            if x.dtype.is_floating_point:
                x = x.to(next(model.parameters()).dtype)
            out = model(x)
            loss = criterion(out.float() if out.dtype != torch.float32 else out, y)
            bs = x.shape[0]
            totals[0] += float(loss) * bs
            totals[1] += bs
            if y.dtype == torch.long:
                totals[2] += int((out.argmax(-1) == y).sum())
    if distributed:
        t = totals.to(args.device) if args.device.type == "cuda" else totals
        torch.distributed.all_reduce(t)
        totals = t.cpu()
    model.train()
    n_samples = max(1.0, float(totals[1]))
    result = {"eval_loss": float(totals[0]) / n_samples}
    if float(totals[2]) > 0:
        result["eval_acc"] = float(totals[2]) / n_samples
    logger.info("Evaluation complete.", dict(**result))
    return result


def train(args, model):
    """The train-loop engine — reference ddp.py:126-288, rebuilt.

    Keeps the reference's structure (sampler/set_epoch, accumulation,
    clipping, warmup-linear schedule, periodic TB scalars, checkpoint-{step}
    layout) with the MI355X-native substitutions: our C++ bucket reducer in
    place of torch DDP, fused HIP optimizer/clip kernels, periodic (not
    per-step) loss readback.
    """
    tb_writer = None
    if is_main_process() and not args.no_tensorboard and SummaryWriter is not None:
        tb_writer = SummaryWriter()  # default runs/ dir (reference ddp.py:128-129)

    model = _cast_model(args, model)
    model.to(args.device)

    dataset = build_dataset(args.dataset, size=args.dataset_size)
    if args.local_rank != -1:
        train_sampler = ShardedSampler(
            dataset, num_replicas=args.world_size, rank=args.node_rank,
            seed=args.seed,
        )
    else:
        # epoch-seeded shuffle even single-process (world 1 shard = the whole
        # set): unlike the reference's RandomSampler (global-RNG order,
        # ddp.py:144-145) the data order is a pure function of (seed, epoch),
        # which is what makes mid-epoch resume replay-free and exact.
        train_sampler = ShardedSampler(
            dataset, num_replicas=1, rank=0, seed=args.seed
        )
    if getattr(dataset, "batched_indexing", False):
        # vectorized batch fetch: the BatchSampler hands the dataset a LIST
        # of indices and __getitem__ gathers the whole batch in one indexing
        # op — no per-sample fetch, no torch.stack collation.  At the MI355X
        # operating point (8192 samples/GPU/step) the per-sample path is
        # host-bound; this one keeps the DataLoader off the critical path.
        from torch.utils.data import BatchSampler

        loader = DataLoader(
            dataset,
            sampler=BatchSampler(
                train_sampler, args.train_batch_size, drop_last=True
            ),
            batch_size=None,
            pin_memory=args.device.type == "cuda",
            num_workers=args.num_workers,
        )
    else:
        loader = DataLoader(
            dataset,
            sampler=train_sampler,
            batch_size=args.train_batch_size,
            pin_memory=args.device.type == "cuda",
            num_workers=args.num_workers,
            drop_last=True,
        )

    if args.max_steps > 0:
        t_total = args.max_steps
        args.num_train_epochs = (
            args.max_steps // max(1, len(loader) // args.gradient_accumulation_steps)
            + 1
        )
    else:
        t_total = (
            len(loader) // args.gradient_accumulation_steps * args.num_train_epochs
        )

    criterion = _criterion_for(args)
    use_master = args.bf16 or args.fp16
    optimizer = SGD(
        model.parameters(),
        lr=args.learning_rate,
        momentum=args.momentum,
        weight_decay=args.weight_decay,
        master_weights=use_master,
    )
    scheduler = get_linear_schedule_with_warmup(
        optimizer, num_warmup_steps=args.warmup_steps, num_training_steps=t_total
    )

    if args.local_rank != -1:
        model = DistributedModel(
            model,
            bucket_bytes=args.bucket_mb << 20,
            find_unused_parameters=args.find_unused_parameters,
        )

    # Native fp16 loss scaling (replaces the reference's broken apex branch,
    # ddp.py:165-181: static scale when --loss_scale > 0, else dynamic).
    scaler = (
        LossScaler(args.loss_scale) if (args.fp16 and not args.bf16) else None
    )

    global_step = 0
    resume_epoch, resume_batches = 0, 0
    if args.resume_from or args.global_step:
        path = args.resume_from
        if path is None and args.global_step:
            path = os.path.join(args.output_dir, f"checkpoint-{args.global_step}")
            if not os.path.isdir(path):
                path = None  # fall back to latest
        # optimizer/scheduler state (incl. last_epoch) comes from the files
        global_step = load_checkpoint(args, model, optimizer, scheduler, path)
        if path is None:
            path = find_latest_checkpoint(args.output_dir)
        if path is not None and global_step > 0:
            # mid-epoch position + RNG states (true resume, not data replay)
            resume_epoch, resume_batches = load_training_state(
                args, path, scaler=scaler
            )

    logger.info(
        "***** Running training *****",
        dict(
            num_examples=len(dataset),
            epochs=args.num_train_epochs,
            per_gpu_batch=args.per_gpu_train_batch_size,
            total_batch=args.train_batch_size
            * args.gradient_accumulation_steps
            * max(1, args.world_size),
            accumulation=args.gradient_accumulation_steps,
            total_steps=t_total,
        ),
    )


    # hipGraph-captured step (opt-in): ws=1, GPU, no accumulation, no fp16
    # scaler (the distributed reducer's in-backward RCCL launches and the
    # scaler's control flow stay eager)
    graph_step = None
    if (
        args.hip_graph
        and args.device.type == "cuda"
        and args.local_rank == -1
        and args.gradient_accumulation_steps == 1
        and scaler is None
    ):
        graph_step = GraphStep(model, criterion, optimizer, args.max_grad_norm)
    elif args.hip_graph and is_main_process():
        logger.warning(
            "--hip_graph ignored (needs single-process CUDA, accumulation=1,"
            " no fp16 scaler)"
        )

    model.train()
    model.zero_grad()
    device_loss = torch.zeros((), device=args.device)
    logged_loss = 0.0
    steps_since_log = 0
    # per-step wall timers (SURVEY.md §5.1 — the reference had none): window
    # throughput measured between logging boundaries, no extra device syncs
    import time as _time

    window_t0 = _time.perf_counter()
    done = False
    tr_loss_total = 0.0
    log_count = 0

    # epoch/step progress bars on the main rank only (reference ddp.py:212,
    # 215; disabled elsewhere).  The loss postfix updates at logging
    # boundaries, not per step — the reference's per-step set_postfix forced
    # a device sync every iteration (ddp.py:232).
    from tqdm import tqdm, trange

    show_bars = is_main_process() and not args.no_progress_bar
    epoch_iter = trange(
        int(args.num_train_epochs), desc="Epoch", disable=not show_bars
    )
    # Double-buffered H2D prefetch on a dedicated copy stream (GPU): the
    # next batch's pinned-host copy overlaps this batch's fwd/bwd.  CPU and
    # --no_prefetch fall back to the reference's in-loop .to(device).
    model_dtype = next(model.parameters()).dtype
    use_prefetch = args.prefetch and args.device.type == "cuda"

    for epoch in epoch_iter:
        if epoch < resume_epoch:
            continue  # scheduler/optimizer state already restored
        if isinstance(train_sampler, ShardedSampler):
            train_sampler.set_epoch(epoch)  # reference ddp.py:213-214
        skip_batches = resume_batches if epoch == resume_epoch else 0
        batches = (
            CudaPrefetcher(loader, args.device, model_dtype)
            if use_prefetch
            else loader
        )
        step_iter = tqdm(
            batches, desc="Iteration", total=len(loader), disable=not show_bars
        )
        for step, (x, y) in enumerate(step_iter):
            if step < skip_batches:
                continue  # fast-forward to the mid-epoch resume point
            if not use_prefetch:
                x = x.to(args.device, non_blocking=True)
                y = y.to(args.device, non_blocking=True)
                if x.dtype.is_floating_point:
                    x = x.to(model_dtype)
                if y.dtype.is_floating_point:
                    y = y.to(model_dtype)

            accum_boundary = (step + 1) % args.gradient_accumulation_steps == 0
            sync_ctx = (
                model.no_sync()
                if (not accum_boundary and isinstance(model, DistributedModel))
                else _nullcontext()
            )
            if graph_step is not None:
                # captured step: fwd+bwd+clip+opt+zero in one replay
                device_loss += graph_step.step(x, y)
                steps_since_log += 1
                scheduler.step()
                global_step += 1
            else:
              with sync_ctx:
                outputs = model(x)
                loss = criterion(outputs, y)
                if args.gradient_accumulation_steps > 1:
                    loss = loss / args.gradient_accumulation_steps
                if scaler is not None:
                    (loss * scaler.scale).backward()
                else:
                    loss.backward()
              device_loss += loss.detach()
              steps_since_log += 1

            if accum_boundary and graph_step is None:
                if isinstance(model, DistributedModel):
                    model.finish_gradient_sync()
                if scaler is not None:
                    grads = [
                        p.grad for p in model.parameters() if p.grad is not None
                    ]
                    scale_grads_(grads, 1.0 / scaler.scale)
                total_norm = clip_grad_norm_(
                    [p for p in model.parameters()], args.max_grad_norm
                )
                if scaler is not None and args.device.type == "cuda":
                    # async overflow handling: the fused SGD kernel skips the
                    # update ON DEVICE when total_norm is non-finite; the
                    # host syncs the skip counter once per sync_interval
                    optimizer.step(
                        guard=total_norm,
                        skip_count=scaler.skip_counter(args.device),
                    )
                    scaler.tick()
                elif scaler is not None:
                    if not scaler.step_ok(total_norm):
                        # inf/nan grads: skip the step, shrink the scale
                        model.zero_grad()
                        continue
                    optimizer.step()
                else:
                    optimizer.step()
                scheduler.step()
                model.zero_grad()
                global_step += 1

            if accum_boundary:

                if (
                    is_main_process()
                    and args.logging_steps > 0
                    and global_step % args.logging_steps == 0
                ):
                    # ONE readback per logging window (not per step)
                    now = _time.perf_counter()
                    win_steps = steps_since_log
                    ms_per_step = (now - window_t0) / max(1, win_steps) * 1e3
                    sps = (
                        args.train_batch_size * max(1, args.world_size)
                        * win_steps / max(1e-9, now - window_t0)
                    )
                    window_t0 = now
                    window = float(device_loss) / max(1, steps_since_log)
                    tr_loss_total += float(device_loss)
                    log_count += steps_since_log
                    device_loss.zero_()
                    steps_since_log = 0
                    logged_loss = window
                    if tb_writer is not None:
                        tb_writer.add_scalar(
                            "lr", scheduler.get_last_lr()[0], global_step
                        )
                        tb_writer.add_scalar("loss", window, global_step)
                        tb_writer.add_scalar(
                            "samples_per_sec", sps, global_step
                        )
                    logger.info(
                        "train step",
                        dict(step=global_step, loss=window,
                             lr=scheduler.get_last_lr()[0],
                             ms_per_step=round(ms_per_step, 2),
                             samples_per_sec=round(sps, 1)),
                    )
                    if show_bars:
                        step_iter.set_postfix(loss=window)

                if (
                    args.eval_steps > 0
                    and global_step % args.eval_steps == 0
                ):
                    # all ranks participate (sharded eval + all-reduce)
                    result = evaluate(args, model, max_batches=args.eval_max_batches)
                    if tb_writer is not None:
                        for k, v in result.items():
                            tb_writer.add_scalar(k, v, global_step)

                if (
                    is_main_process()
                    and args.save_steps > 0
                    and global_step % args.save_steps == 0
                ):
                    save_checkpoint(
                        args, model, optimizer, scheduler, global_step,
                        epoch=epoch, batches_in_epoch=step + 1, scaler=scaler,
                    )

                if args.max_steps > 0 and global_step >= args.max_steps:
                    done = True  # exact stop (reference's ddp.py:280 overran by one)
                    break
        if done:
            break

    # final average loss (reference ddp.py:287-288)
    tr_loss_total += float(device_loss)
    log_count += steps_since_log
    avg = tr_loss_total / max(1, log_count)
    logger.info("Training complete.", dict(global_step=global_step, avg_loss=avg))
    if tb_writer is not None:
        tb_writer.close()
    return global_step, avg


from contextlib import nullcontext as _nullcontext  # noqa: E402


def build_parser():
    parser = argparse.ArgumentParser()
    # ---- reference flag set, names + defaults 1:1 (ddp.py:293-308) ----
    parser.add_argument("--global-step", dest="global_step", type=int, default=0,
                        help="resume from checkpoint-<global_step> (the reference "
                             "parsed this but never used it)")
    parser.add_argument("--no_cuda", action="store_true")
    parser.add_argument("--output_dir", type=str, default="outputs")
    parser.add_argument("--seed", type=int, default=42)
    parser.add_argument("--gradient_accumulation_steps", type=int, default=1)
    parser.add_argument("--per_gpu_train_batch_size", type=int, default=32)
    parser.add_argument("--max_steps", type=int, default=0)
    parser.add_argument("--logging_steps", type=int, default=100)
    parser.add_argument("--save_steps", type=int, default=1000)
    parser.add_argument("--num_train_epochs", type=int, default=10)
    parser.add_argument("--warmup_steps", type=int, default=100)
    parser.add_argument("--max_grad_norm", type=float, default=1000.0)
    parser.add_argument("--local_rank", type=int, default=-1)
    parser.add_argument("--fp16", action="store_true")
    parser.add_argument("--loss_scale", type=float, default=0)
    parser.add_argument("--fp16_opt_level", type=str, default="O2")
    # ---- additive flags (model zoo / MI355X path) ----
    parser.add_argument("--model", type=str, default="foo",
                        help="foo | resnet18 | resnet18-imagenet | resnet50 | vit-b16")
    parser.add_argument("--dataset", type=str, default=None)
    parser.add_argument("--dataset_size", type=int, default=100000)
    parser.add_argument("--learning_rate", "--lr", type=float, default=1e-3)
    parser.add_argument("--momentum", type=float, default=0.0)
    parser.add_argument("--weight_decay", type=float, default=0.0)
    parser.add_argument("--bf16", action="store_true",
                        help="bf16 weights/activations with fp32 master weights")
    parser.add_argument("--resume_from", type=str, default=None)
    parser.add_argument("--bucket_mb", type=int, default=50,
                        help="gradient bucket size (MiB) for the xGMI reducer")
    parser.add_argument("--find_unused_parameters", action="store_true")
    parser.add_argument("--num_workers", type=int, default=0)
    parser.add_argument("--no_tensorboard", action="store_true")
    parser.add_argument("--no_progress_bar", action="store_true")
    parser.add_argument("--eval_steps", type=int, default=0,
                        help="run evaluate() every N optimizer steps (0 = off; "
                             "the reference's evaluate was an empty stub)")
    parser.add_argument("--eval_max_batches", type=int, default=None)
    parser.add_argument("--hip_graph", action="store_true",
                        help="capture the ws=1 training step in a hipGraph "
                             "(fwd+bwd+clip+fused SGD replayed per step; "
                             "live LR via device scalar)")
    parser.add_argument("--prefetch", dest="prefetch", action="store_true",
                        default=True,
                        help="double-buffered pinned-host H2D prefetch on a "
                             "dedicated copy stream (GPU only)")
    parser.add_argument("--no_prefetch", dest="prefetch", action="store_false")
    return parser


def main(argv=None):
    args = build_parser().parse_args(argv)
    if args.dataset is None:
        args.dataset = args.model
    setup(args)
    model = build_model(args.model)
    train(args, model)
    cleanup(args)
    logger.info("Process exited.")


if __name__ == "__main__":
    main()
