"""In-tree build of the gfx950 HIP extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built gtopkssgd_amd/_C*.so stays in the source tree (git-ignored)
so it travels with repo snapshots to GPU boxes.
"""
import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

setup(
    name="gtopkssgd_amd",
    version="0.1.0",
    description="MI355X-native gTop-k sparsified S-SGD framework",
    packages=["gtopkssgd_amd", "gtopkssgd_amd.models", "gtopkssgd_amd.ops",
              "gtopkssgd_amd.parallel", "gtopkssgd_amd.utils",
              "gtopkssgd_amd.data"],
    ext_modules=[
        CUDAExtension(
            name="gtopkssgd_amd._C",
            sources=["csrc/gtopk_ext.hip", "csrc/fusedbn.hip"],
            extra_compile_args={
                "cxx": ["-O3"],
                "nvcc": ["-O3", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
