"""Build the CDNA4 HIP engine extension in-tree.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Produces tfservingcache_amd/engine/_tfsc_engine*.so (gitignored; ships to
the GPU box with the gpurun snapshot).
"""
import os
import sys

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils import cpp_extension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "tfservingcache_amd", "engine", "csrc")

sources = [
    os.path.join(CSRC, "executor.cpp"),
    os.path.join(CSRC, "ops", "ops_memory.hip"),
    os.path.join(CSRC, "ops", "gemm.hip"),
    os.path.join(CSRC, "ops", "gemm_fp8.hip"),
    os.path.join(CSRC, "ops", "conv.hip"),
    os.path.join(CSRC, "ops", "attention.hip"),
    os.path.join(CSRC, "fastpath.cpp"),
    os.path.join(CSRC, "frontend.cpp"),
    os.path.join(CSRC, "rest_frontend.cpp"),
]

# nghttp2 (HTTP/2 framing/HPACK for the native gRPC front-end) ships in
# this image's conda tree; probe the usual prefixes so other ROCm bases
# (system libnghttp2-dev) work too
def _find_nghttp2():
    for inc, lib in [("/opt/conda/include", "/opt/conda/lib"),
                     ("/usr/include", "/usr/lib/x86_64-linux-gnu"),
                     ("/usr/local/include", "/usr/local/lib")]:
        if os.path.exists(os.path.join(inc, "nghttp2", "nghttp2.h")):
            return inc, lib
    raise RuntimeError(
        "nghttp2 headers not found (needed by the native gRPC "
        "front-end, engine/csrc/frontend.cpp) — install libnghttp2-dev "
        "or point NGHTTP2_INC/NGHTTP2_LIB at a prefix")


NGHTTP2_INC, NGHTTP2_LIB = (
    (os.environ.get("NGHTTP2_INC"), os.environ.get("NGHTTP2_LIB"))
    if os.environ.get("NGHTTP2_INC") else _find_nghttp2())

setup(
    name="tfsc_engine",
    ext_modules=[
        cpp_extension.CUDAExtension(
            name="tfservingcache_amd.engine._tfsc_engine",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17", f"-I{NGHTTP2_INC}"],
                "nvcc": ["-O3", "-std=c++17"],
            },
            library_dirs=[NGHTTP2_LIB],
            libraries=["nghttp2"],
            extra_link_args=[f"-Wl,-rpath,{NGHTTP2_LIB}"],
        )
    ],
    cmdclass={"build_ext": cpp_extension.BuildExtension.with_options(
        no_python_abi_suffix=False)},
)
