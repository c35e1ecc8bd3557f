"""Build the mine_amd HIP extension in-tree for gfx950 (MI355X).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Produces mine_amd/ops/_mine_hip*.so next to the Python sources (the .so
must live in-tree so it ships with repo snapshots).
"""
import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "mine_amd", "ops", "csrc")

ext = CUDAExtension(
    name="mine_amd.ops._mine_hip",
    sources=[
        os.path.join(CSRC, "bindings.hip"),
        os.path.join(CSRC, "render_kernels.hip"),
        os.path.join(CSRC, "ssim_kernels.hip"),
        os.path.join(CSRC, "pad_kernels.hip"),
        os.path.join(CSRC, "bn_kernels.hip"),
        os.path.join(CSRC, "head_kernels.hip"),
        os.path.join(CSRC, "conv_kernels.hip"),
        os.path.join(CSRC, "wrw_kernels.hip"),
        os.path.join(CSRC, "resample_kernels.hip"),
        os.path.join(CSRC, "igemm_kernels.hip"),
        os.path.join(CSRC, "loss_kernels.hip"),
    ],
    extra_compile_args={
        "cxx": ["-O3"],
        "nvcc": ["-O3", "--offload-arch=gfx950"],
    },
)

setup(
    name="mine_amd",
    version="0.1.0",
    packages=["mine_amd"],
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
