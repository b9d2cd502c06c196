"""DistributedModel — the framework's DistributedDataParallel equivalent.

Wraps an nn.Module for one-process-per-GPU data parallelism over RCCL/xGMI
(gloo on CPU for the hardware-free plumbing rung).  The gradient-bucket
machinery is the native C++ reducer (``parallel/csrc/reducer.cpp``); this
wrapper only:

* has the C++ reducer register its own per-parameter hooks directly on the
  AccumulateGrad autograd nodes (``Reducer.install_hooks``) — grad-ready →
  bucket countdown → collective launch all happens inside the autograd
  engine thread with no Python re-entry (gradients live as views into the
  reducer's flat buckets, so there is also no copy between autograd and the
  collective),
* optionally walks the autograd graph after forward to find parameters that
  did not participate (``find_unused_parameters`` — reference ddp.py:195
  passes True) and marks them ready,
* provides ``no_sync()`` for gradient accumulation and ``finish_gradient_sync``
  for the engine to call before clipping/stepping.

Reference parity: wrap-time rank-0 broadcast of params+buffers
(ddp.py:194), bucketed all-reduce overlapped with backward (triggered from
loss.backward(), ddp.py:231), SUM + divide-by-world averaging.
"""

from __future__ import annotations

import contextlib

import torch
from torch import nn

from ..utils.dist import get_world_size


def _load_core():
    try:
        from pytorch_ddp_template_amd import _ddp_core  # built in-tree

        return _ddp_core
    except Exception as e:  # noqa: BLE001
        raise RuntimeError(
            "pytorch_ddp_template_amd._ddp_core (the native C++ reducer) is "
            "not built. Run `python setup.py build_ext --inplace` — "
            f"import error: {e!r}"
        ) from e


class DistributedModel(nn.Module):
    def __init__(
        self,
        module: nn.Module,
        process_group=None,
        first_bucket_bytes: int = 1 << 20,
        bucket_bytes: int = 50 << 20,
        find_unused_parameters: bool = False,
        broadcast_params: bool = True,
    ):
        super().__init__()
        self.module = module
        self.find_unused_parameters = find_unused_parameters
        if process_group is None:
            import torch.distributed as dist

            if not (dist.is_available() and dist.is_initialized()):
                raise RuntimeError(
                    "DistributedModel requires an initialized process group"
                )
            process_group = dist.group.WORLD
        self.process_group = process_group

        core = _load_core()
        self._params = [p for p in module.parameters() if p.requires_grad]
        self.reducer = core.Reducer(
            self._params, process_group, first_bucket_bytes, bucket_bytes
        )
        if broadcast_params and get_world_size() > 1:
            buffers = [b for b in module.buffers() if b.is_floating_point()]
            int_buffers = [b for b in module.buffers() if not b.is_floating_point()]
            self.reducer.broadcast_state(buffers + int_buffers)

        # C++-side hooks on the AccumulateGrad nodes (no Python per-grad hops)
        self.reducer.install_hooks()

    def forward(self, *args, **kwargs):
        out = self.module(*args, **kwargs)
        # Only walk when a backward can actually follow: under no_grad()
        # (evaluation) every param would look unused and the walk would
        # launch collectives from an eval pass.
        if (
            self.find_unused_parameters
            and self.reducer.sync()
            and torch.is_grad_enabled()
        ):
            self._mark_unused(out)
        return out

    def _mark_unused(self, outputs):
        # Walk the autograd graph from the outputs to find which parameters
        # will receive gradients; everything else is marked ready now so its
        # bucket can still fly (same semantics as torch's reducer for
        # find_unused_parameters=True — reference ddp.py:195).
        tensors = []

        def collect(o):
            if isinstance(o, torch.Tensor):
                if o.grad_fn is not None:
                    tensors.append(o)
            elif isinstance(o, (list, tuple)):
                for x in o:
                    collect(x)
            elif isinstance(o, dict):
                for x in o.values():
                    collect(x)

        collect(outputs)
        seen_accum = set()
        stack = [t.grad_fn for t in tensors]
        visited = set()
        while stack:
            fn = stack.pop()
            if fn is None or fn in visited:
                continue
            visited.add(fn)
            if hasattr(fn, "variable"):
                seen_accum.add(id(fn.variable))
            for next_fn, _ in fn.next_functions:
                stack.append(next_fn)
        unused = [
            i for i, p in enumerate(self._params) if id(p) not in seen_accum
        ]
        if unused:
            self.reducer.mark_unused(unused)

    @contextlib.contextmanager
    def no_sync(self):
        """Skip gradient synchronization (gradient accumulation micro-batches)."""
        self.reducer.set_sync(False)
        try:
            yield
        finally:
            self.reducer.set_sync(True)

    def finish_gradient_sync(self):
        """Wait for all bucket all-reduces and average; call before clip/step."""
        self.reducer.finalize()

    def zero_grad(self, set_to_none: bool = False):  # noqa: ARG002
        # Grads are views into the reducer's flat buckets; zero the buckets.
        # (set_to_none would detach the views — deliberately ignored.)
        self.reducer.zero_grads()

    # convenience passthroughs
    def state_dict(self, *a, **k):
        return self.module.state_dict(*a, **k)

    def load_state_dict(self, *a, **k):
        return self.module.load_state_dict(*a, **k)
