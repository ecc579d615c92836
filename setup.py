"""In-tree build of the gfx950 HIP extension.

Usage: PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
The built _dmx_C.so lands inside detectmateservice_amd/ops/ so it travels
with the repo snapshot to GPU boxes (JIT caches under ~/.cache do not).
"""
import os
from pathlib import Path

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = Path(__file__).parent
CSRC = ROOT / "detectmateservice_amd" / "ops" / "csrc"

sources = [
    str(CSRC / "bindings.cpp"),
    str(CSRC / "codec.cpp"),
    str(CSRC / "frame_reader.cpp"),
    str(CSRC / "shm_ring.cpp"),
    str(CSRC / "gemm_bf16.hip"),
    str(CSRC / "layernorm.hip"),
    str(CSRC / "attention.hip"),
    str(CSRC / "attention_mfma.hip"),
    str(CSRC / "bert_fused.hip"),
    str(CSRC / "template_match.hip"),
    str(CSRC / "hashset.hip"),
    str(CSRC / "edit_distance.hip"),
]

setup(
    name="detectmateservice_amd_kernels",
    ext_modules=[
        CUDAExtension(
            name="detectmateservice_amd.ops._dmx_C",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
