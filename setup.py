"""Build the in-tree CDNA4 HIP extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built `_sonata_hip*.so` lands in sonata_amd/ops/ (in-tree: it travels
with repo snapshots; no JIT cache involved).
"""

import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

ROOT = os.path.dirname(os.path.abspath(__file__))

ext = CUDAExtension(
    name="sonata_amd.ops._sonata_hip",
    sources=[
        "csrc/ext.cpp",
        "csrc/elementwise.hip",
        "csrc/conv1d.hip",
        "csrc/conv1d_cl.hip",
        "csrc/resblock_cl.hip",
        "csrc/attention_cl.hip",
        "csrc/engine/vits_engine.cpp",
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": [
            "-O3",
            "-std=c++17",
            "--offload-arch=gfx950",
            "-mcumode",
        ],
    },
)

setup(
    name="sonata_amd_hip",
    version="0.1.0",
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
