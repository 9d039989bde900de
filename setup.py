"""Build the tfmesos_amd HIP extension in-tree for MI355X (gfx950).

    python setup.py build_ext --inplace

Produces tfmesos_amd/_C*.so next to the package sources so the built
artifact ships with the repo snapshot to GPU boxes.
"""

import os

from setuptools import setup, find_packages

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

CSRC = os.path.join("tfmesos_amd", "ops", "csrc")
sources = [
    os.path.join(CSRC, "ext.hip"),
    os.path.join(CSRC, "apply.hip"),
    os.path.join(CSRC, "gemm.hip"),
    os.path.join(CSRC, "conv.hip"),
    os.path.join(CSRC, "bn.hip"),
    os.path.join(CSRC, "pool.hip"),
    os.path.join(CSRC, "softmax_xent.hip"),
    os.path.join(CSRC, "embedding.hip"),
    os.path.join(CSRC, "elementwise.hip"),
]

setup(
    name="tfmesos_amd",
    version="0.1.0",
    packages=find_packages(exclude=["tests"]),
    entry_points={
        "console_scripts": ["tfa_run = tfmesos_amd.cli:main"],
    },
    ext_modules=[
        CUDAExtension(
            name="tfmesos_amd._C",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
