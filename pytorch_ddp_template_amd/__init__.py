"""pytorch_ddp_template_amd — a brand-new MI355X-native DDP training framework.

Capabilities of howardlau1999/pytorch-ddp-template, built
MI355X-first: HIP/CDNA4 (gfx950) MFMA kernels for the compute path, a native
C++ bucketed gradient reducer doing RCCL all-reduce over xGMI, one process
per GPU via a torchrun-style launcher, NHWC models, synthetic datasets, and
the reference's launch/train-loop API + checkpoint layout.
"""

__version__ = "0.1.0"

from . import data, models, ops, optim, parallel, utils  # noqa: F401
