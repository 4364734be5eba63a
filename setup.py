"""In-tree build of the dlaf_amd HIP extension for gfx950 (MI355X).

Build with:  python setup.py build_ext --inplace
The resulting dlaf_amd/_hip.*.so travels with the repo snapshot to GPU boxes.
"""

import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup  # noqa: E402
from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

setup(
    name="dlaf_amd_hip",
    ext_modules=[
        CUDAExtension(
            name="dlaf_amd._hip",
            sources=[
                "csrc/ext.cpp",
                "csrc/band_chase.cpp",
                "csrc/gemm_tiles.hip",
                "csrc/gemm_tiles_v2.hip",
                "csrc/bt_apply.hip",
                "csrc/factor.hip",
                "csrc/panel_qr.hip",
                "csrc/secular.hip",
                "csrc/rocblas_batch.cpp",
                "csrc/chase_gpu.hip",
            ],
            libraries=["rocblas"],
            extra_compile_args={
                "cxx": ["-O3"],
                "nvcc": ["-O3"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
