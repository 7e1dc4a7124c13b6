"""Build glt_amd: MI355X-native (gfx950) GNN sampling/training engine.

In-tree build:  python setup.py build_ext --inplace
The HIP sources are compiled by hipcc with --offload-arch=gfx950 (works
without a GPU present); CPU C++ sources by the host compiler.
"""
import os
import glob

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
os.environ.setdefault("MAX_JOBS", str(min(16, os.cpu_count() or 8)))

from setuptools import setup, find_packages  # noqa: E402
from torch.utils.cpp_extension import CUDAExtension, BuildExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))

sources = sorted(
    glob.glob("glt_amd/csrc/*.cpp")
    + glob.glob("glt_amd/csrc/cpu/*.cpp")
    + [f for f in glob.glob("glt_amd/csrc/hip/*.hip")
       # exclude the *_hip.hip copies torch's hipify step regenerates
       if not f.endswith("_hip.hip")]
)

setup(
    name="glt_amd",
    version="0.1.0",
    description="MI355X-native graph learning engine (GLT-compatible)",
    packages=find_packages(include=["glt_amd", "glt_amd.*"]),
    ext_modules=[
        CUDAExtension(
            name="glt_amd._C",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
