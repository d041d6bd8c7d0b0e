"""In-tree build of the pdrl_amd HIP extension for MI355X (gfx950).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built .so lands in pdrl_amd/ops/ so it travels with the repo snapshot
(not a site-packages or JIT-cache install).
"""
import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import (  # noqa: E402
    BuildExtension,
    CppExtension,
    CUDAExtension,
)

SRC = [
    "pdrl_amd/ops/csrc/bindings.cpp",
    "pdrl_amd/ops/csrc/seq_lstm.hip",
    "pdrl_amd/ops/csrc/wgrad.hip",
    "pdrl_amd/ops/csrc/losses.hip",
    "pdrl_amd/ops/csrc/vmpo_loss.hip",
    "pdrl_amd/ops/csrc/ppoc_loss.hip",
    "pdrl_amd/ops/csrc/sac_loss.hip",
    "pdrl_amd/ops/csrc/sacc_loss.hip",
    "pdrl_amd/ops/csrc/scans.hip",
    "pdrl_amd/ops/csrc/multi_tensor.hip",
    "pdrl_amd/ops/csrc/megastep.hip",
    "pdrl_amd/ops/csrc/fwd_loss.hip",
]

setup(
    name="pdrl_amd_hip_ops",
    ext_modules=[
        CUDAExtension(
            name="pdrl_amd.ops._hip_ops",
            sources=SRC,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        ),
        # CPU batched actor for the worker processes (x86-64-v3 = AVX2:
        # portable across the EPYC hosts; -march=native would pin to the
        # build box)
        CppExtension(
            name="pdrl_amd.ops._cpu_actor",
            sources=["pdrl_amd/ops/csrc/cpu_actor.cpp"],
            extra_compile_args=["-O3", "-std=c++17", "-march=x86-64-v3",
                                "-ffast-math"],
        ),
    ],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
