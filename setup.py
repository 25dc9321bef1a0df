"""In-tree build of the MI355X (gfx950) HIP extension.

Build with:  PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
The resulting .so lives inside the package (torch_on_k8s_amd/ops/_C...so)
so it travels with repo snapshots; it is git-ignored.
"""
import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "torch_on_k8s_amd", "ops", "csrc")

# "_hip" files are byproducts of torch's extension preprocessing of a
# previous build -- never feed them back in as sources.
sources = [
    os.path.join(CSRC, f)
    for f in sorted(os.listdir(CSRC))
    if f.endswith((".cpp", ".hip")) and "_hip." not in f
]

setup(
    name="torch-on-k8s-amd-kernels",
    version="0.1.0",
    ext_modules=[
        CUDAExtension(
            name="torch_on_k8s_amd.ops._C",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
