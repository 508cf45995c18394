"""Build the in-tree gfx950 HIP extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built .so lands in caffeonspark_amd/ops/ and travels with the repo
snapshot to GPU boxes (it is git-ignored but NOT gpurun-ignored).
"""

import glob
import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup  # noqa: E402
from torch.utils import cpp_extension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "caffeonspark_amd", "ops", "csrc")

sources = sorted(
    glob.glob(os.path.join(CSRC, "*.cpp")) +
    [f for f in glob.glob(os.path.join(CSRC, "*.hip"))
     if not f.endswith("_hip.hip")])   # skip hipify-generated copies

setup(
    name="cosamd-hip",
    ext_modules=[
        cpp_extension.CUDAExtension(
            name="caffeonspark_amd.ops._cosamd_hip",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": cpp_extension.BuildExtension},
)
