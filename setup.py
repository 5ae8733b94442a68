"""Build the waternet_amd native HIP extension IN-TREE for gfx950.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

hipcc cross-compiles without a GPU; the resulting waternet_amd/_C*.so
travels to the GPU box with the repo snapshot.
"""

import os
from pathlib import Path

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = Path(__file__).resolve().parent
CSRC = ROOT / "waternet_amd" / "csrc"

sources = [
    str(CSRC / "bindings.cpp"),
    str(CSRC / "conv_mfma.hip"),
    str(CSRC / "elementwise.hip"),
    str(CSRC / "pool.hip"),
    str(CSRC / "ssim.hip"),
    str(CSRC / "preprocess.hip"),
]

setup(
    name="waternet_amd_C",
    ext_modules=[
        CUDAExtension(
            name="waternet_amd._C",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": [
                    "-O3",
                    "-std=c++17",
                    "--offload-arch=gfx950",
                ],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
