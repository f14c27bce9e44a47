"""Build the in-tree HIP extension for gfx950.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built ``_gossip_kernels*.so`` lands inside
``stochastic_gradient_push_amd/ops/`` so it ships with the source tree
(reference packaging parity: reference setup.py:22-47; this package also
installs the trainer CLI as a script).
"""

import os

from setuptools import find_packages, setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "stochastic_gradient_push_amd", "ops", "csrc")

ext = CUDAExtension(
    name="stochastic_gradient_push_amd.ops._gossip_kernels",
    sources=[
        os.path.join(CSRC, "bindings.cpp"),
        os.path.join(CSRC, "comm_core.cpp"),
        os.path.join(CSRC, "gossip_kernels.hip"),
        os.path.join(CSRC, "gemm1x1_kernels.hip"),
        os.path.join(CSRC, "conv3x3_kernels.hip"),
        os.path.join(CSRC, "bn_kernels.hip"),
    ],
    libraries=["rccl"],
    extra_compile_args={
        "cxx": ["-O3"],
        "nvcc": ["-O3", "--offload-arch=gfx950"],
    },
)

setup(
    name="stochastic_gradient_push_amd",
    version="0.1.0",
    description=(
        "MI355X-native gossip-based distributed SGD (SGP/OSGP/D-PSGD/"
        "AD-PSGD) with HIP/CDNA4 kernels and RCCL p2p over xGMI"
    ),
    packages=find_packages(include=["stochastic_gradient_push_amd*"]),
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension},
    python_requires=">=3.8",
    scripts=["gossip_sgd.py", "gossip_sgd_adpsgd.py"],
)
