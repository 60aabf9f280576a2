"""Build the in-tree HIP extensions for gfx950.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The resulting mi355x_scale/ops/_C*.so is git-ignored but travels with the
repo snapshot to GPU boxes.
"""
import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup
from torch.utils.cpp_extension import (BuildExtension, CppExtension,
                                       CUDAExtension)

SRC = [
    "mi355x_scale/ops/csrc/bindings.cpp",
    "mi355x_scale/ops/csrc/preprocess.hip",
    "mi355x_scale/ops/csrc/groupfit.hip",
    "mi355x_scale/ops/csrc/mfma_project.hip",
    "mi355x_scale/ops/csrc/fused_bn.hip",
    "mi355x_scale/ops/csrc/adam.hip",
    "mi355x_scale/ops/csrc/maxpool.hip",
    "mi355x_scale/ops/csrc/arma_gen.hip",
    "mi355x_scale/ops/csrc/stemconv.hip",
    "mi355x_scale/ops/csrc/conv3x3wrw.hip",
]

ASAN = os.environ.get("MI355X_ASAN") == "1"

if ASAN:
    # device AddressSanitizer build of the kernel extension
    # (gfx950:xnack+ + asanrtl.bc device runtime). Built as a separate
    # module; tools/gpu_sanitize.py drives every kernel through it once
    # per round on a GPU box under HSA_XNACK=1.
    asan_flags = ["-g", "-O1", "-fsanitize=address",
                  "-shared-libsan", "--offload-arch=gfx950:xnack+"]
    ext_modules = [
        CUDAExtension(
            name="mi355x_scale.ops._C_asan",
            sources=SRC,
            extra_compile_args={"cxx": ["-g", "-O1", "-fsanitize=address"],
                                "nvcc": asan_flags},
            # the final link runs through g++ (no -shared-libsan there);
            # -fsanitize=address pulls the asan runtime it knows
            extra_link_args=["-fsanitize=address"],
        ),
    ]
else:
    ext_modules = None  # filled below

setup(
    name="mi355x_scale",
    version="0.1.0",
    packages=["mi355x_scale"],
    ext_modules=ext_modules if ASAN else [
        CUDAExtension(
            name="mi355x_scale.ops._C",
            sources=SRC,
            extra_compile_args={
                "cxx": ["-O3"],
                "nvcc": ["-O3", "--offload-arch=gfx950"],
            },
        ),
        # hand-written xGMI p2p primitives (IPC buffers + multi-source
        # reduce) behind parallel/p2p_allreduce.py
        CUDAExtension(
            name="mi355x_scale.parallel._p2p",
            sources=["mi355x_scale/parallel/csrc/p2p_bindings.cpp",
                     "mi355x_scale/parallel/csrc/p2p.hip"],
            extra_compile_args={
                "cxx": ["-O3"],
                "nvcc": ["-O3", "--offload-arch=gfx950"],
            },
        ),
        # host-side C++ group-gather engine (no HIP): multithreaded key
        # factorize + panel scatter
        CppExtension(
            name="mi355x_scale.groupby._gather",
            sources=["mi355x_scale/groupby/csrc/gather.cpp"],
            extra_compile_args={"cxx": ["-O3"]},
        ),
    ],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
