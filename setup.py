"""In-tree build of the gfx950 HIP extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built gtopkssgd_amd/_C*.so stays in the source tree (git-ignored)
so it travels with repo snapshots to GPU boxes.
"""
import glob
import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

# ninja's depfiles do not see through torch's hipify step: editing a
# csrc/*.h leaves the stale *_hip.o and silently relinks old kernels.
# Touch every .hip whose headers are newer so hipify + compile rerun.
_hdr_mtime = max((os.path.getmtime(h) for h in glob.glob("csrc/*.h")),
                 default=0.0)
for _src in glob.glob("csrc/*.hip"):
    if os.path.getmtime(_src) < _hdr_mtime:
        os.utime(_src, None)

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
