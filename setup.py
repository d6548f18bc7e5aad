"""Build the native extensions in-tree (gfx950 / MI355X).

- torchbeast_amd.runtime._tbruntime : C++ actor-learner runtime (queues,
  dynamic batcher, actor pool, unix-socket env server, nest).
- torchbeast_amd.ops._tbops         : hand-written CDNA4 HIP kernels
  (V-trace, fused losses, LSTM unroll, RMSProp, sampling, conv trunk).

Usage: PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
"""

import glob
import os

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CppExtension, CUDAExtension

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

ROOT = os.path.dirname(os.path.abspath(__file__))

# Sanitizer flavors for the C++ runtime (race/heap checking of the queue and
# actor-pool concurrency): TBAMD_SANITIZE=thread|address.
_sanitize = os.environ.get("TBAMD_SANITIZE")
_san_flags = [f"-fsanitize={_sanitize}", "-fno-omit-frame-pointer"] if _sanitize else []

ext_modules = [
    # CUDAExtension (ROCm) so the runtime can drive HIP streams/ATen-GPU
    # ops from C++ (GIL-free inference engine).
    CUDAExtension(
        name="torchbeast_amd.runtime._tbruntime",
        sources=[
            "torchbeast_amd/runtime/csrc/module.cc",
            "torchbeast_amd/ops/hip/atari_trunk.hip",
            "torchbeast_amd/ops/hip/conv_mfma.hip",
            "torchbeast_amd/runtime/csrc/runtime_kernels.hip",
        ],
        extra_compile_args={
            "cxx": ["-O3", "-std=c++17", "-pthread"] + _san_flags,
            "nvcc": ["-O3", "-std=c++17"],
        },
        extra_link_args=_san_flags,
    )
]

# torch's hipify writes *_hip.hip shadow copies next to the sources; they
# define the same symbols, so they must never enter the source list.
hip_sources = sorted(
    p for p in glob.glob("torchbeast_amd/ops/hip/*.hip") if not p.endswith("_hip.hip")
)
if hip_sources:
    ext_modules.append(
        CUDAExtension(
            name="torchbeast_amd.ops._tbops",
            sources=hip_sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    )

setup(
    name="torchbeast_amd",
    version="0.1.0",
    description="MI355X-native IMPALA actor-learner framework",
    packages=[
        "torchbeast_amd",
        "torchbeast_amd.core",
        "torchbeast_amd.models",
        "torchbeast_amd.envs",
        "torchbeast_amd.ops",
        "torchbeast_amd.runtime",
        "torchbeast_amd.nest",
        "torchbeast_amd.parallel",
        "nest",
    ],
    ext_modules=ext_modules,
    cmdclass={"build_ext": BuildExtension},
)
