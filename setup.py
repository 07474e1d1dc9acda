"""In-tree build of the room_amd._C HIP extension for gfx950 (MI355X only).

Build: PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
The built .so lives inside room_amd/ so it travels to GPU boxes with the repo
snapshot (JIT caches under ~/.cache do not).
"""
import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup
# On ROCm builds torch.utils.cpp_extension exposes the HIP toolchain through
# the CUDAExtension class name (its "nvcc" args feed hipcc); the sources are
# plain HIP written for gfx950 — the pipeline's mechanical hipify pass is an
# identity transform over them (its *_hip.hip copies are gitignored).
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

SOURCES = [
    "room_amd/ops/csrc/bindings.cpp",
    "room_amd/ops/csrc/norm_rope.hip",
    "room_amd/ops/csrc/paged_attn.hip",
    "room_amd/ops/csrc/moe.hip",
    "room_amd/ops/csrc/gemv.hip",
    "room_amd/ops/csrc/fused_decode.hip",
    "room_amd/ops/csrc/flash_prefill.hip",
    "room_amd/ops/csrc/sampling.hip",
    "room_amd/ops/csrc/vector_store.hip",
]

setup(
    name="room_amd_C",
    ext_modules=[
        CUDAExtension(
            name="room_amd._C",
            sources=SOURCES,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
