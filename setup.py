"""Build the in-tree gfx950 HIP extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The resulting _hip_ops*.so lands in tosem2021_amd/ (in-tree, so it travels
with the repo snapshot to GPU boxes; it is git-ignored).
"""
import glob
import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

# torch's hipify writes generated csrc/*_hip.hip next to the sources and can
# SKIP regeneration even when the source is newer — delete them up front so
# every build compiles the current kernels.
for _f in glob.glob(os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                 "csrc", "*_hip.hip")):
    os.remove(_f)

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

ext = CUDAExtension(
    name="tosem2021_amd._hip_ops",
    sources=[
        "csrc/ops.cpp",
        "csrc/lt_gemm.cpp",
        "csrc/layernorm.hip",
        "csrc/bias_gelu.hip",
        "csrc/softmax.hip",
        "csrc/adamw.hip",
        "csrc/repack.hip",
        "csrc/flash_attn.hip",
        "csrc/wgrad_gemm.hip",
        "csrc/pool.hip",
    ],
    libraries=["hipblaslt"],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
    },
)

setup(
    name="tosem2021-amd-ops",
    version="0.1.0",
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=True)},
)
