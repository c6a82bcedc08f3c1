"""Build the in-tree HIP extension flowhip._C for gfx950.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The .so lands in flowhip/ (git-ignored; it travels to the GPU box with the
gpurun snapshot). hipcc cross-compiles without a GPU present.
"""

import glob
import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

# exclude the *_hip.* copies torch's hipify writes next to the sources on
# a previous build — globbing them too would compile every TU twice
sources = [s for s in sorted(glob.glob("csrc/*.hip")) + sorted(glob.glob("csrc/*.cpp"))
           if not s.endswith(("_hip.hip", "_hip.cpp"))]

setup(
    name="flowhip",
    version="0.1.0",
    packages=["flowhip"],
    ext_modules=[
        CUDAExtension(
            name="flowhip._C",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3"],
                "nvcc": ["-O3", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
