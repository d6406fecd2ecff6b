"""Build the kubeai_amd gfx950 HIP extension in-tree.

Usage:  PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The .so lands inside kubeai_amd/ (kubeai_amd._C) so it travels with the
repo snapshot to GPU boxes. CPU-only boxes cross-compile fine (hipcc needs
no GPU present).
"""
import os
import glob

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils import cpp_extension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "kubeai_amd", "csrc")

sources = sorted(
    glob.glob(os.path.join(CSRC, "*.cpp")) + glob.glob(os.path.join(CSRC, "*.hip"))
)

setup(
    name="kubeai_amd_C",
    ext_modules=[
        cpp_extension.CUDAExtension(
            name="kubeai_amd._C",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": [
                    "-O3",
                    "-std=c++17",
                    "--offload-arch=gfx950",
                ],
            },
        )
    ],
    cmdclass={"build_ext": cpp_extension.BuildExtension},
)
