"""In-tree build of rocps native extensions.

    python setup.py build_ext --inplace

Extensions:
- ``rocps._bundle``  — TF-V2 tensor-bundle codec (C++, CPU).
- ``rocps._hipops``  — HIP/CDNA4 kernels for gfx950 (MFMA/LDS), built
  with hipcc via torch.utils.cpp_extension (PYTORCH_ROCM_ARCH=gfx950).

Built ``.so`` files land inside ``rocps/`` so they travel with the
repo snapshot to GPU machines (no site-packages install, no JIT cache).
"""

import os
import sys

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
os.environ.setdefault("MAX_JOBS", str(min(16, os.cpu_count() or 8)))

# ROCPS_SANITIZE=address|thread builds the CPU extensions (bundle codec,
# control plane) under the sanitizer — SURVEY §5 race-detection plan.
_SAN = os.environ.get("ROCPS_SANITIZE")
_SAN_FLAGS = [f"-fsanitize={_SAN}", "-fno-omit-frame-pointer"] if _SAN else []

from torch.utils.cpp_extension import BuildExtension, CppExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))

ext_modules = [
    CppExtension(
        name="rocps._bundle",
        sources=["csrc/bundle/bundle.cc"],
        extra_compile_args=["-O2", "-std=c++17"] + _SAN_FLAGS,
        extra_link_args=_SAN_FLAGS,
    ),
    CppExtension(
        name="rocps._ccp",
        sources=["csrc/control/control_plane.cc"],
        extra_compile_args=["-O2", "-std=c++17"] + _SAN_FLAGS,
        extra_link_args=_SAN_FLAGS,
    ),
    CppExtension(
        name="rocps._cpuapply",
        sources=["csrc/cpu/cpu_apply.cc"],
        # -ffp-contract=off: bit-parity with torch's unfused CPU ops;
        # -mavx2 baseline (safe on the GPU pool's hosts), OMP across
        # the shard
        extra_compile_args=["-O3", "-std=c++17", "-mavx2", "-mfma", "-fopenmp",
                            "-ffp-contract=off"] + _SAN_FLAGS,
        extra_link_args=["-fopenmp"] + _SAN_FLAGS,
    ),
]

_hip_sources = [
    "csrc/kernels/ext.cc",
    "csrc/kernels/fused_optimizers.hip",
    "csrc/kernels/softmax_xent.hip",
    "csrc/kernels/wgrad.hip",
    "csrc/kernels/bn_relu.hip",
    "csrc/kernels/gelu.hip",
    "csrc/kernels/gemm_bf16.hip",
    "csrc/kernels/gemm_bf16_256.hip",
    "csrc/kernels/maxpool.hip",
    "csrc/kernels/bias_act.hip",
    "csrc/kernels/layernorm.hip",
]
_hip_sources = [s for s in _hip_sources if os.path.exists(os.path.join(ROOT, s))]
if _hip_sources:
    import torch
    if torch.version.hip is not None:
        from torch.utils.cpp_extension import CUDAExtension
        ext_modules.append(
            CUDAExtension(
                name="rocps._hipops",
                sources=_hip_sources,
                extra_compile_args={
                    "cxx": ["-O2", "-std=c++17"],
                    "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950",
                             "-fgpu-flush-denormals-to-zero"],
                },
            ))
    else:  # pragma: no cover — this repo targets ROCm builds only
        print("WARNING: torch has no HIP — skipping rocps._hipops",
              file=sys.stderr)

setup(
    name="rocps",
    version="0.1.0",
    description="MI355X-native parameter-server data-parallel training runtime",
    packages=["rocps", "rocps.train", "rocps.engine", "rocps.checkpoint",
              "rocps.models", "rocps.ops", "rocps.obs", "rocps.cli"],
    ext_modules=ext_modules,
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
