"""Build the CDNA4 HIP extension in-tree:

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The .so lands at substratus_amd/_C*.so (git-ignored; it travels to the GPU box
with the gpurun snapshot).
"""

import os

from setuptools import find_packages, setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
HIP_DIR = os.path.join(ROOT, "substratus_amd", "ops", "hip")

# Hand-written kernel sources only. torch's CUDAExtension hipify pass emits
# *_hip.hip/*_hip.cpp siblings into the build dir; exclude any that leak
# in-tree so the generated copies can never shadow the originals.
sources = [os.path.join(HIP_DIR, f) for f in sorted(os.listdir(HIP_DIR))
           if f.endswith(".hip") and not f.endswith("_hip.hip")]

setup(
    name="substratus_amd",
    version="0.1.0",
    packages=find_packages(include=["substratus_amd", "substratus_amd.*"]),
    # wheels carry the built _C.so + the GEMM tables; kernel SOURCES ride
    # the sdist via MANIFEST.in (which can exclude the *_hip.hip siblings
    # torch's hipify regenerates in-tree on every build)
    package_data={"substratus_amd": ["data/tunables/*.csv"]},

    entry_points={"console_scripts": [
        "sub = substratus_amd.cli.main:main",
        "nbwatch = substratus_amd.containertools.nbwatch:watch",
    ]},
    ext_modules=[
        CUDAExtension(
            name="substratus_amd._C",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
