"""In-tree build of the rl_replicas_amd HIP extension for gfx950.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Produces rl_replicas_amd/_hip_ops*.so next to the package sources so
the built artifact travels with the repo snapshot to GPU boxes.
torch.utils.cpp_extension's GPU-extension path drives hipcc for the
.hip translation units on ROCm.
"""
import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

HIP_DIR = os.path.join("rl_replicas_amd", "ops", "hip")

ext = CUDAExtension(
    name="rl_replicas_amd._hip_ops",
    sources=[
        os.path.join(HIP_DIR, "bindings.hip"),
        os.path.join(HIP_DIR, "mlp_kernels.hip"),
        os.path.join(HIP_DIR, "loss_kernels.hip"),
        os.path.join(HIP_DIR, "sample_kernels.hip"),
        os.path.join(HIP_DIR, "env_kernels.hip"),
        os.path.join(HIP_DIR, "rollout_kernels.hip"),
        os.path.join(HIP_DIR, "update_kernels.hip"),
        os.path.join(HIP_DIR, "offpolicy_kernels.hip"),
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
    },
)

setup(
    name="rl_replicas_amd",
    version="0.1.0",
    packages=[
        "rl_replicas",
        "rl_replicas_amd",
        "rl_replicas_amd.algorithms",
        "rl_replicas_amd.envs",
        "rl_replicas_amd.networks",
        "rl_replicas_amd.ops",
        "rl_replicas_amd.optimizers",
        "rl_replicas_amd.parallel",
        "rl_replicas_amd.policies",
        "rl_replicas_amd.samplers",
    ],
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension},
)
