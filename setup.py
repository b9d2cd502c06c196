"""In-tree build of the native extensions.

Two extensions, both built IN-TREE (the .so files land inside the package so
they travel with the repo snapshot to GPU boxes):

* ``pytorch_ddp_template_amd._ddp_core`` — the C++ bucketed gradient reducer
  (no HIP; builds anywhere; collectives via c10d ProcessGroup → RCCL/gloo).
* ``pytorch_ddp_template_amd._hip_ops`` — the CDNA4 (gfx950) HIP kernels.
  Cross-compiled with PYTORCH_ROCM_ARCH=gfx950 (no GPU needed to build).

    python setup.py build_ext --inplace
"""

import os
import sys
from pathlib import Path

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CppExtension  # noqa: E402

ROOT = Path(__file__).parent
PKG = ROOT / "pytorch_ddp_template_amd"

ext_modules = [
    CppExtension(
        name="pytorch_ddp_template_amd._ddp_core",
        sources=[str(PKG / "parallel" / "csrc" / "reducer.cpp")],
        extra_compile_args=["-O3", "-std=c++17"],
    )
]

hip_sources = sorted(str(p) for p in (PKG / "ops" / "csrc").glob("*.cpp"))
hip_sources += sorted(str(p) for p in (PKG / "ops" / "csrc").glob("*.hip"))
if hip_sources:
    try:
        from torch.utils.cpp_extension import ROCM_HOME

        have_hip = ROCM_HOME is not None
    except Exception:
        have_hip = False
    if have_hip:
        from torch.utils.cpp_extension import CUDAExtension

        ext_modules.append(
            CUDAExtension(
                name="pytorch_ddp_template_amd._hip_ops",
                sources=hip_sources,
                extra_compile_args={
                    "cxx": ["-O3", "-std=c++17"],
                    "nvcc": ["-O3", "-std=c++17"],
                },
            )
        )
    else:
        print("WARNING: ROCm not found; skipping _hip_ops", file=sys.stderr)

setup(
    name="pytorch_ddp_template_amd",
    version="0.1.0",
    packages=[
        "pytorch_ddp_template_amd",
        "pytorch_ddp_template_amd.models",
        "pytorch_ddp_template_amd.ops",
        "pytorch_ddp_template_amd.parallel",
        "pytorch_ddp_template_amd.data",
        "pytorch_ddp_template_amd.utils",
    ],
    ext_modules=ext_modules,
    cmdclass={"build_ext": BuildExtension},
)
