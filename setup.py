"""Build the rl_amd native extension (CPU segment trees + CDNA4 HIP kernels).

In-tree build: ``python setup.py build_ext --inplace`` produces
``rl_amd/_C*.so`` next to the package so it ships with the source tree.
Target arch is gfx950 only (MI355X) — set via PYTORCH_ROCM_ARCH.
"""
import os
import sys

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup  # noqa: E402

import torch  # noqa: E402
from torch.utils import cpp_extension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "rl_amd", "csrc")

with_hip = torch.version.hip is not None

sources = [os.path.join(CSRC, "bindings.cpp")]
extra_cflags = ["-O3", "-std=c++17"]
define_macros = []

if with_hip:
    sources += [
        os.path.join(CSRC, "value_scan.hip"),
        os.path.join(CSRC, "segment_tree_hip.hip"),
        os.path.join(CSRC, "rnn_scan.hip"),
        os.path.join(CSRC, "fused_actor.hip"),
        os.path.join(CSRC, "wgrad.hip"),
        os.path.join(CSRC, "env_step.hip"),
        os.path.join(CSRC, "loss_ops.hip"),
        os.path.join(CSRC, "fused_mlp.hip"),
        os.path.join(CSRC, "rollout_fused.hip"),
    ]
    define_macros.append(("RL_AMD_WITH_HIP", None))
    ext_cls = cpp_extension.CUDAExtension
    extra = {
        "extra_compile_args": {
            "cxx": extra_cflags,
            "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
        }
    }
else:
    ext_cls = cpp_extension.CppExtension
    extra = {"extra_compile_args": {"cxx": extra_cflags}}

setup(
    name="rl_amd_C",
    ext_modules=[
        ext_cls(
            name="rl_amd._C",
            sources=sources,
            define_macros=define_macros,
            **extra,
        )
    ],
    cmdclass={"build_ext": cpp_extension.BuildExtension.with_options(no_python_abi_suffix=False)},
)
