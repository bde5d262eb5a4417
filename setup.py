"""In-tree build of the tensorlink_amd HIP extension for gfx950.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built .so lands next to the sources inside the package so it travels
with repo snapshots to GPU boxes (it is git-ignored but not gpurun-ignored).
"""

import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup  # noqa: E402
from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

SRC_DIR = os.path.join("tensorlink_amd", "ops", "csrc")
SOURCES = [
    os.path.join(SRC_DIR, f)
    for f in sorted(os.listdir(SRC_DIR))
    if f.endswith((".hip", ".cpp"))
]

setup(
    name="tensorlink_amd_ext",
    ext_modules=[
        CUDAExtension(
            name="tensorlink_amd._C",
            sources=SOURCES,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
