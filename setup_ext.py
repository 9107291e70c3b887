"""Build the stmgcn_amd HIP extension IN-TREE for gfx950.

Run: PYTORCH_ROCM_ARCH=gfx950 python setup_ext.py
(the built stmgcn_amd/_C*.so travels with the repo snapshot to GPU boxes).
"""
import os
import sys

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
os.environ.setdefault("MAX_JOBS", str(os.cpu_count() or 8))

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

REPO = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(REPO, "stmgcn_amd", "csrc")

# torch's build step writes hipify copies (*_hip.cpp / *_hip.hip) next to the
# sources; exclude them or a rebuild would compile duplicates.
sources = [os.path.join(CSRC, f) for f in sorted(os.listdir(CSRC))
           if f.endswith((".cpp", ".hip")) and "_hip." not in f]

setup(
    name="stmgcn_amd_ext",
    ext_modules=[
        CUDAExtension(
            name="stmgcn_amd._C",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
    script_args=["build_ext", "--inplace"] if len(sys.argv) == 1 else None,
)
