# Build for torchdistx_amd: an in-tree native core (_C: fake tensors +
# deferred-init tape against the PyTorch-ROCm dispatcher) and the CDNA4 HIP
# kernel extension (_K: init kernels + fused optimizer step) compiled for
# gfx950 only.
#
# Usage: python setup.py build_ext --inplace
# (The .so files land inside torchdistx_amd/ and travel with the repo.)

import os

from setuptools import setup

import torch
from torch.utils import cpp_extension

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "torchdistx_amd", "csrc")

BASE_VERSION = "0.2.0"


def version_with_variant() -> str:
    """Version with a local variant tag naming the compute stack the wheel
    was built against (rocmX.Y from the torch in the build env), so wheels
    from different stacks are distinguishable — the same scheme the wheel
    ecosystem uses for +cpu / +cuXXX / +rocmX.Y variants. Override with
    TDX_VERSION_VARIANT (empty string = no tag, e.g. for sdists)."""
    variant = os.environ.get("TDX_VERSION_VARIANT")
    if variant is None:
        hip = getattr(torch.version, "hip", None)
        if hip:
            variant = "rocm" + ".".join(hip.split(".")[:2])
        else:
            variant = "cpu"
    return BASE_VERSION + ("+" + variant if variant else "")

ext_modules = [
    cpp_extension.CppExtension(
        name="torchdistx_amd._C",
        sources=[
            os.path.join(CSRC, "core", "stack_utils.cc"),
            os.path.join(CSRC, "core", "fake.cc"),
            os.path.join(CSRC, "core", "deferred_init.cc"),
            os.path.join(CSRC, "core", "native_redirect.cc"),
            os.path.join(CSRC, "core", "tdx_ops.cc"),
            os.path.join(CSRC, "core", "module.cc"),
        ],
        extra_compile_args=["-O2", "-std=c++17", "-fvisibility=hidden"],
    ),
]

hip_sources = [
    os.path.join(CSRC, "hip", "init_kernels.hip"),
    os.path.join(CSRC, "hip", "anyprecision_adamw.hip"),
    os.path.join(CSRC, "hip", "kernels_module.cc"),
]
if all(os.path.exists(s) for s in hip_sources) and torch.version.hip:
    ext_modules.append(
        cpp_extension.CUDAExtension(  # drives hipcc on ROCm builds
            name="torchdistx_amd._K",
            sources=hip_sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    )

setup(
    name="torchdistx_amd",
    version=version_with_variant(),
    description=(
        "MI355X-native fake-tensor / deferred-init framework with "
        "torchdistx's capabilities"
    ),
    packages=[
        "torchdistx",
        "torchdistx.slowmo",
        "torchdistx.optimizers",
        "torchdistx_amd",
        "torchdistx_amd.slowmo",
        "torchdistx_amd.optimizers",
        "torchdistx_amd.parallel",
        "torchdistx_amd.models",
        "torchdistx_amd.ops",
        "torchdistx_amd.utils",
    ],
    package_data={"torchdistx_amd": ["_C.pyi", "_K.pyi", "py.typed"]},
    ext_modules=ext_modules,
    cmdclass={"build_ext": cpp_extension.BuildExtension},
)
