"""Build the daft_amd HIP/CDNA4 kernel extension in-tree:

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Sources are hand-written HIP (csrc/*.hip) targeting gfx950 only."""
import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

setup(
    name="daft_amd_native",
    ext_modules=[
        CUDAExtension(
            name="daft_amd._native",
            sources=[
                "csrc/binding.cpp",
                "csrc/kernels.hip",
                "csrc/sort.hip",
                "csrc/strings.hip",
                "csrc/multimodal.hip",
                "csrc/fusedexpr.hip",
                "csrc/fusedjit.hip",
                "csrc/bpe.hip",
                "csrc/editdist.hip",
            ],
            libraries=["hiprtc"],
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=True)},
)
