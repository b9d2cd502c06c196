"""Optimizers + LR schedule.

* ``SGD`` — fused multi-tensor SGD/momentum (reference: torch.optim.SGD at
  ddp.py:183, stepped at ddp.py:240).  On ROCm devices the whole step is a
  single multi-tensor HIP kernel sweep (one launch per chunk list, not one
  per parameter); supports bf16 params with fp32 master weights (the
  reference's apex-O2 slot, ddp.py:165-181, reincarnated natively).
* ``clip_grad_norm_`` — global L2 norm + scale (reference ddp.py:238-239)
  via the multi-tensor l2norm/scale kernels.
* ``get_linear_schedule_with_warmup`` — linear warmup then linear decay to 0
  (reference ddp.py:52-61), same LambdaLR shape.
"""

from __future__ import annotations

import torch
from torch.optim.lr_scheduler import LambdaLR

from .ops import grad_l2_norm, scale_grads_
from .ops.native import native, use_native


def get_linear_schedule_with_warmup(optimizer, num_warmup_steps, num_training_steps, last_epoch=-1):
    """Reference ddp.py:52-61."""

    def lr_lambda(current_step: int):
        if current_step < num_warmup_steps:
            return float(current_step) / float(max(1, num_warmup_steps))
        return max(
            0.0,
            float(num_training_steps - current_step)
            / float(max(1, num_training_steps - num_warmup_steps)),
        )

    return LambdaLR(optimizer, lr_lambda, last_epoch)


def clip_grad_norm_(parameters, max_norm: float) -> torch.Tensor:
    """Global-L2 gradient clipping (reference ddp.py:238-239)."""
    if isinstance(parameters, torch.Tensor):
        parameters = [parameters]
    grads = [p.grad for p in parameters if p.grad is not None]
    total_norm = grad_l2_norm(grads)
    if max_norm > 0:
        # host-side compare against a device scalar would sync; compute the
        # clip coefficient on device and always scale (coef clamped to 1).
        if grads and grads[0].is_cuda:
            coef = (max_norm / (total_norm + 1e-6)).clamp(max=1.0)
            native().scale_by_tensor_(list(grads), coef)
        else:
            coef = float(max_norm / (float(total_norm) + 1e-6))
            if coef < 1.0:
                scale_grads_(grads, coef)
    return total_norm


class SGD(torch.optim.Optimizer):
    """Fused multi-tensor SGD with momentum / weight decay / nesterov.

    With ``master_weights=True`` and bf16 parameters, keeps an fp32 master
    copy per parameter; the fused kernel updates master in fp32 and writes
    the bf16 working copy in the same pass.
    """

    def __init__(
        self,
        params,
        lr: float,
        momentum: float = 0.0,
        weight_decay: float = 0.0,
        dampening: float = 0.0,
        nesterov: bool = False,
        master_weights: bool = False,
    ):
        defaults = dict(
            lr=lr,
            momentum=momentum,
            weight_decay=weight_decay,
            dampening=dampening,
            nesterov=nesterov,
            master_weights=master_weights,
        )
        super().__init__(params, defaults)

    def load_state_dict(self, state_dict):
        """torch's Optimizer.load_state_dict casts floating-point state to
        each param's dtype — with bf16 params that silently rounds the fp32
        master weights and momentum through bf16 (destroying the master
        precision) and then crashes the fused kernel, which requires fp32
        state.  Restore the fp32 slots from the ORIGINAL (on-disk) tensors
        after the structural load."""
        super().load_state_dict(state_dict)
        id_map = {}
        for g_saved, g in zip(state_dict["param_groups"], self.param_groups):
            for pid, p in zip(g_saved["params"], g["params"]):
                id_map[pid] = p
        for pid, st_saved in state_dict["state"].items():
            p = id_map.get(pid)
            if p is None:
                continue
            st = self.state[p]
            for k in ("momentum_buffer", "master"):
                if k in st_saved and isinstance(st_saved[k], torch.Tensor):
                    st[k] = st_saved[k].to(
                        device=p.device, dtype=torch.float32
                    )

    @torch.no_grad()
    def step(self, closure=None, guard=None, skip_count=None, lr_tensor=None):
        """``guard`` (f32 device scalar, e.g. the pre-clip grad norm): when
        non-finite the fused kernel skips the whole update ON DEVICE and
        ticks ``skip_count`` — no host sync (the fp16 loss-scaler path).
        ``lr_tensor`` (f32 device scalar): the kernel reads lr from the
        device, so a hipGraph-captured step follows the live schedule."""
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        self._guard, self._skip_count = guard, skip_count
        self._lr_tensor = lr_tensor
        for group in self.param_groups:
            params, grads, moms, masters = [], [], [], []
            momentum = group["momentum"]
            use_master = group["master_weights"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if momentum != 0 and "momentum_buffer" not in state:
                    state["momentum_buffer"] = torch.zeros_like(
                        p, dtype=torch.float32
                    )
                if use_master and p.dtype != torch.float32 and "master" not in state:
                    state["master"] = p.detach().float().clone()
                params.append(p)
                grads.append(p.grad)
                moms.append(state["momentum_buffer"] if momentum != 0 else None)
                masters.append(state.get("master") if use_master else None)
            if not params:
                continue
            self._fused_step(group, params, grads, moms, masters)
        self._guard = self._skip_count = self._lr_tensor = None
        return loss

    def _fused_step(self, group, params, grads, moms, masters):
        lr = group["lr"]
        momentum = group["momentum"]
        wd = group["weight_decay"]
        damp = group["dampening"]
        nesterov = group["nesterov"]
        if use_native(*params):
            native().sgd_step(
                params,
                grads,
                [m if m is not None else torch.Tensor() for m in moms],
                [m if m is not None else torch.Tensor() for m in masters],
                lr,
                momentum,
                wd,
                damp,
                nesterov,
                getattr(self, "_guard", None),
                getattr(self, "_skip_count", None),
                getattr(self, "_lr_tensor", None),
            )
            return
        # CPU reference path (also the numerics reference for the kernel)
        lrt = getattr(self, "_lr_tensor", None)
        if lrt is not None:
            lr = float(lrt)
        guard = getattr(self, "_guard", None)
        if guard is not None and not bool(torch.isfinite(guard).all()):
            sc = getattr(self, "_skip_count", None)
            if sc is not None:
                sc.add_(1)
            return
        for p, g, m, mw in zip(params, grads, moms, masters):
            work = mw if mw is not None else p
            gf = g.float()
            if wd != 0:
                gf = gf.add(work.float(), alpha=wd)
            if momentum != 0:
                m.mul_(momentum).add_(gf, alpha=1 - damp)
                gf = gf.add(m, alpha=momentum) if nesterov else m
            if mw is not None:
                mw.add_(gf, alpha=-lr)
                p.copy_(mw.to(p.dtype))
            else:
                p.add_(gf.to(p.dtype), alpha=-lr)
