"""In-tree build of the gfx950 native extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built `.so` lands inside the package (`llm_instance_gateway_amd/_C...so`)
so the repo snapshot carries it to GPU boxes (the JIT cache would not travel).
"""

import os
from pathlib import Path

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

import pybind11
from setuptools import Extension
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

ROOT = Path(__file__).parent
CSRC = ROOT / "llm_instance_gateway_amd" / "ops" / "csrc"
RCSRC = ROOT / "llm_instance_gateway_amd" / "router" / "csrc"

sources = sorted(
    str(p)
    for p in CSRC.glob("*.hip")
    # hipify writes shadow copies named *_hip.hip next to the originals
    if not p.name.endswith("_hip.hip")
)

# LIG_DEBUG=1: host-side index/range validation in the bindings (make
# build-debug) — slot ids vs stack sizes, unit tables vs batch bounds —
# catching bad inputs before a kernel dereferences them.
debug_flags = ["-DLIG_DEBUG=1", "-g"] if os.environ.get("LIG_DEBUG") == "1" else []

ext = CUDAExtension(
    name="llm_instance_gateway_amd._C",
    sources=sources,
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"] + debug_flags,
        "nvcc": [
            "-O3",
            "-std=c++17",
            "--offload-arch=gfx950",
        ] + debug_flags,
    },
)

# pure-C++ native router (no GPU/torch dependency): plain pybind11 module
router_ext = Extension(
    name="llm_instance_gateway_amd._router",
    sources=[str(RCSRC / "router_native.cpp")],
    include_dirs=[pybind11.get_include()],
    extra_compile_args=["-O3", "-std=c++17"],
    language="c++",
)

setup(
    name="llm_instance_gateway_amd",
    version="0.1.0",
    packages=["llm_instance_gateway_amd"],
    ext_modules=[ext, router_ext],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
