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

# torch's hipify/ninja pipeline tracks no header dependencies for .hip
# objects: a common.h edit would silently ship stale kernels. Touch every
# .hip newer than its headers so the pipeline rebuilds them.
_hdr_mtime = max(
    (os.path.getmtime(h) for h in glob.glob(os.path.join(CSRC, "*.h"))),
    default=0.0,
)
for _src in list(sources):
    if (
        _src.endswith(".hip")
        and not _src.endswith("_hip.hip")
        and os.path.exists(_src)
        and os.path.getmtime(_src) < _hdr_mtime
    ):
        os.utime(_src, None)
        _gen = _src[:-4] + "_hip.hip"
        if os.path.exists(_gen):
            os.remove(_gen)
            sources = [s for s in sources if s != _gen]
# drop hipify artifacts whose base source was deleted or touched
sources = [s for s in sources if os.path.exists(s)]

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
