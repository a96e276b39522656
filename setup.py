"""In-tree build of the gfx950 HIP extension.

`python setup.py build_ext --inplace` drops `_d4pg_hip*.so` into
d4pg_amd/ops/ (the loader imports it from there; no JIT cache involved, so
the .so travels with the repo snapshot to GPU boxes).
"""

import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup  # noqa: E402
from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

HERE = os.path.dirname(os.path.abspath(__file__))


setup(
    name="d4pg_amd_hip",
    ext_modules=[
        CUDAExtension(
            # package-qualified name => `build_ext --inplace` drops the .so
            # into d4pg_amd/ops/ (in-tree, travels with the repo snapshot)
            name="d4pg_amd.ops._d4pg_hip",
            sources=["d4pg_amd/ops/hip/ext.hip"],
            extra_compile_args={
                "cxx": ["-O3"],
                "nvcc": ["-O3", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
