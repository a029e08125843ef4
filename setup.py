"""Build the parallax_amd HIP extension in-tree for gfx950 (MI355X).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built .so lands at parallax_amd/ops/_C.*.so (git-ignored; it travels to the
GPU box with the gpurun snapshot).
"""

import os

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "parallax_amd", "ops", "csrc")

sources = [
    os.path.join(CSRC, "bindings.cpp"),
    os.path.join(CSRC, "lt_gemm.cpp"),
    os.path.join(CSRC, "elementwise.hip"),
    os.path.join(CSRC, "paged_attention.hip"),
    os.path.join(CSRC, "paged_attention_mfma.hip"),
    os.path.join(CSRC, "mla_attention.hip"),
    os.path.join(CSRC, "prefill_attention.hip"),
    os.path.join(CSRC, "moe.hip"),
    os.path.join(CSRC, "skinny_gemm.hip"),
    os.path.join(CSRC, "indexer.hip"),
    os.path.join(CSRC, "sampler.hip"),
]
sources = [s for s in sources if os.path.exists(s)]

setup(
    name="parallax_amd_ext",
    entry_points={
        "console_scripts": ["parallax_amd = parallax_amd.cli:main"],
    },
    ext_modules=[
        CUDAExtension(
            name="parallax_amd.ops._C",
            sources=sources,
            libraries=["hipblaslt"],
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
