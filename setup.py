# In-tree build of the gfx950 HIP extension:
#   PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
# (driven by __graft_entry__.build(); the built .so lands inside
# rocnrdma_amd/ops/ and travels to the GPU box with the repo snapshot)
import os

from setuptools import find_packages, setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

setup(
    name="rocnrdma_amd",
    version="0.1.0",
    packages=find_packages(include=["rocnrdma_amd", "rocnrdma_amd.*"]),
    ext_modules=[
        CUDAExtension(
            name="rocnrdma_amd.ops._p2p_ext",
            sources=[
                "rocnrdma_amd/ops/csrc/p2p_ext.cpp",
                "rocnrdma_amd/ops/csrc/p2p_kernels.hip",
            ],
            extra_compile_args={
                "cxx": ["-O3"],
                "nvcc": ["-O3", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
