"""Build the deeprest_amd HIP extension in-tree for gfx950.

  PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The resulting deeprest_amd/_C*.so stays inside the package directory so it
travels with repo snapshots to GPU machines (no JIT cache dependence).
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "deeprest_amd", "csrc")

sources = [
    os.path.join(CSRC, f)
    for f in ["bindings.cpp", "featurize.cpp", "layernorm.hip", "pinball.hip",
              "adam.hip", "gru.hip", "attention.hip"]
]

setup(
    name="deeprest_amd_ext",
    ext_modules=[
        CUDAExtension(
            name="deeprest_amd._C",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
