"""In-tree build of the gfx950 HIP extension.

    python setup.py build_ext --inplace

Produces isolation_forest_amd/ops/_iforest_hip*.so (which travels to the
GPU box with the repo snapshot). Target is MI355X (gfx950) ONLY — no
multi-arch fat binaries, no CUDA path.
"""

import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup  # noqa: E402
from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

HERE = os.path.dirname(os.path.abspath(__file__))

ext = CUDAExtension(
    name="isolation_forest_amd.ops._iforest_hip",
    sources=[
        "isolation_forest_amd/ops/hip/bindings.cpp",
        "isolation_forest_amd/ops/hip/forest_kernels.hip",
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": [
            "-O3",
            "-std=c++17",
            "--offload-arch=gfx950",
            # determinism contract: no fma contraction — device doubles must
            # be bit-identical to the numpy oracle (utils/det_math.py); the
            # hot float32 paths use explicit __f{add,mul,div}_rn already.
            "-ffp-contract=off",
        ],
    },
)

setup(
    name="isolation-forest-amd",
    version="0.1.0",
    packages=[
        "isolation_forest_amd",
        "isolation_forest_amd.core",
        "isolation_forest_amd.models",
        "isolation_forest_amd.ops",
        "isolation_forest_amd.parallel",
        "isolation_forest_amd.persist",
        "isolation_forest_amd.utils",
        "isolation_forest_amd.onnx",
        "isolation_forest_amd.serving",
    ],
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
