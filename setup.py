"""In-tree build of the gfx950 HIP extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The .so lands in rainbowiqn_amd/ops/ so it travels with the repo snapshot to
GPU boxes (no JIT cache dependence)."""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "rainbowiqn_amd", "ops", "csrc")

ext = CUDAExtension(
    name="rainbowiqn_amd.ops._rainbowiqn_hip",
    sources=[
        os.path.join(CSRC, "bindings.cpp"),
        os.path.join(CSRC, "replay.hip"),
        os.path.join(CSRC, "loss.hip"),
        os.path.join(CSRC, "adam.hip"),
        os.path.join(CSRC, "rng.hip"),
        os.path.join(CSRC, "env.hip"),
        os.path.join(CSRC, "model_fwd.hip"),
        os.path.join(CSRC, "model_bwd.hip"),
        os.path.join(CSRC, "model_fp8.hip"),
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
    },
)

setup(
    name="rainbowiqn-amd",
    version="0.1.0",
    packages=["rainbowiqn_amd", "rainbowiqn_amd.models", "rainbowiqn_amd.ops",
              "rainbowiqn_amd.parallel", "rainbowiqn_amd.replay",
              "rainbowiqn_amd.runtime", "rainbowiqn_amd.utils"],
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=True)},
)
