"""Build the in-tree HIP extension for gfx950 (MI355X).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

hipcc cross-compiles without a GPU; the resulting
ai_rtc_agent_amd/ops/_C*.so travels with the repo snapshot.
"""
import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "ai_rtc_agent_amd", "ops", "csrc")

sources = [
    os.path.join(CSRC, f)
    for f in ["ext.cpp", "vcn.cpp", "h264sw.cpp", "dtls.cpp", "elementwise.hip", "norms.hip", "conv2d.hip", "attention.hip", "fp8.hip", "conv2d_fp8.hip"]
]

setup(
    name="ai_rtc_agent_amd_C",
    ext_modules=[
        CUDAExtension(
            name="ai_rtc_agent_amd.ops._C",
            sources=sources,
            libraries=["ssl", "crypto"],  # dtls.cpp (system OpenSSL 3)
            extra_compile_args={
                "cxx": ["-O3"],
                "nvcc": ["-O3", "--offload-arch=gfx950", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
