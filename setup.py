"""Build the in-tree CDNA4 (gfx950) HIP extension for fl4health_amd.

Usage:  PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
"""
import os

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "fl4health_amd", "ops", "csrc")

setup(
    name="fl4health_amd",
    version="0.2.0",
    packages=["fl4health_amd"],
    ext_modules=[
        CUDAExtension(
            name="fl4health_amd._C",
            sources=[
                os.path.join(CSRC, "bindings.cpp"),
                os.path.join(CSRC, "flat_ops.hip"),
                os.path.join(CSRC, "bn_ops.hip"),
                os.path.join(CSRC, "mmd_ops.hip"),
                os.path.join(CSRC, "conv_ops.hip"),
                os.path.join(CSRC, "contrastive_ops.hip"),
                os.path.join(CSRC, "in_ops.hip"),
            ],
            extra_compile_args={
                # FL4_ASAN=1: host-side AddressSanitizer build of the binding
                # layer (run pytest with LD_PRELOAD=$(gcc -print-file-name=libasan.so)
                # ASAN_OPTIONS=detect_leaks=0). Device-side ASAN on gfx950
                # needs an XNACK-enabled driver stack and is not wired here.
                "cxx": ["-O3"] + (["-fsanitize=address", "-fno-omit-frame-pointer"]
                                  if os.environ.get("FL4_ASAN") == "1" else []),
                "nvcc": ["-O3", "--offload-arch=gfx950"],
            },
            extra_link_args=(["-fsanitize=address"] if os.environ.get("FL4_ASAN") == "1" else []),
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
