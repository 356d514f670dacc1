"""In-tree build of the gfx950 HIP extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The .so lands next to the Python wrappers
(colearn_federated_learning_amd/ops/_colearn_hip*.so) so repo snapshots
carry it to GPU boxes.  Sources are native HIP written for CDNA4; torch's
build machinery mechanically copies them to ``*_hip.hip`` (its hipify
pass).  There is no CUDA in them to translate — the pass only adds the
``hip_runtime.h`` include and rewrites the ``<<<>>>`` launches to the
equivalent ``hipLaunchKernelGGL`` spelling — and the generated copies
are gitignored.
"""

import os

from setuptools import find_packages, setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

CSRC = os.path.join("colearn_federated_learning_amd", "ops", "csrc")

ext = CUDAExtension(
    name="colearn_federated_learning_amd.ops._colearn_hip",
    sources=[
        os.path.join(CSRC, "bindings.cpp"),
        os.path.join(CSRC, "fused_optim.hip"),
        os.path.join(CSRC, "dp_topk.hip"),
        os.path.join(CSRC, "norms.hip"),
        os.path.join(CSRC, "gemm_bf16.hip"),
        os.path.join(CSRC, "conv_bf16.hip"),
        os.path.join(CSRC, "head.hip"),
        os.path.join(CSRC, "shadow.hip"),
        os.path.join(CSRC, "attn.hip"),
        os.path.join(CSRC, "elemwise.hip"),
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17"],
    },
)

setup(
    name="colearn_federated_learning_amd",
    version="0.1.0",
    packages=find_packages(include=["colearn_federated_learning_amd*"]),
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
