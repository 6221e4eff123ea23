"""In-tree build of the HIP extension (`_hipops`) for gfx950.

Build:  PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
The built .so lands inside tensorflow_web_deploy_amd/ so it travels to
GPU boxes with the repo snapshot.
"""
import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup  # noqa: E402
from torch.utils import cpp_extension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "tensorflow_web_deploy_amd", "csrc")

ext = cpp_extension.CUDAExtension(
    name="tensorflow_web_deploy_amd._hipops",
    sources=[
        os.path.join(CSRC, "bindings.cpp"),
        os.path.join(CSRC, "gemm_conv.hip"),
        os.path.join(CSRC, "conv_direct.hip"),
        os.path.join(CSRC, "pool_reduce.hip"),
        os.path.join(CSRC, "elementwise.hip"),
        os.path.join(CSRC, "layernorm.hip"),
        os.path.join(CSRC, "preprocess.hip"),
        os.path.join(CSRC, "attn_fused.hip"),
        os.path.join(CSRC, "depthwise.hip"),
        os.path.join(CSRC, "jpeg_gpu.hip"),
        os.path.join(CSRC, "gemm_8p.hip"),
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
    },
)

# Torch-free native modules (plain Extension, NOT CppExtension: they
# must import without libtorch — the server CLI loads the frontend
# before torch, and the shm-ring decode workers load the JPEG decoder
# without torch at all).
import pybind11  # noqa: E402
from setuptools import Extension  # noqa: E402

httpd = Extension(
    name="tensorflow_web_deploy_amd._httpd",
    sources=[os.path.join(CSRC, "httpd.cpp")],
    include_dirs=[pybind11.get_include()],
    extra_compile_args=["-O2", "-std=c++17"],
    language="c++",
)

jpegc = Extension(
    name="tensorflow_web_deploy_amd._jpegc",
    sources=[os.path.join(CSRC, "jpegc.cpp")],
    include_dirs=[pybind11.get_include()],
    extra_compile_args=["-O3", "-std=c++17"],
    language="c++",
)

setup(
    name="tensorflow_web_deploy_amd",
    version="0.1.0",
    packages=["tensorflow_web_deploy_amd"],
    ext_modules=[ext, httpd, jpegc],
    cmdclass={"build_ext": cpp_extension.BuildExtension},
)
