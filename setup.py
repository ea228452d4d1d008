"""Build the ravnest_amd CDNA4 HIP extension IN-TREE:

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Produces ravnest_amd/_C*.so next to the package sources (the built .so
travels to the GPU box with the repo snapshot; a JIT cache would not).
"""
import os
from pathlib import Path

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

CSRC = Path(__file__).parent / "ravnest_amd" / "csrc"
# hipify writes *_hip.hip copies next to the sources; exclude them
sources = [str(CSRC / "bindings.cpp")] + sorted(
    str(p) for p in CSRC.glob("*.hip") if not p.name.endswith("_hip.hip"))

setup(
    name="ravnest_amd_ext",
    version="0.1.0",
    ext_modules=[
        CUDAExtension(
            name="ravnest_amd._C",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
