"""In-tree build of the native extension (gfx950).

    python setup.py build_ext --inplace

Produces ``min_tfs_client_amd/_native*.so`` next to the package so the
built artefact travels with the repo snapshot to GPU boxes (build system
analogue of reference setup.py:28-98, which compiles protos instead — our
wire layer needs no codegen, the native build is the codec + HIP kernels).
"""
import os

from setuptools import find_namespace_packages, setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import (  # noqa: E402
    BuildExtension,
    CppExtension,
    CUDAExtension,
)

_CSRC = os.path.join("min_tfs_client_amd", "ops", "csrc")

ext = CUDAExtension(
    name="min_tfs_client_amd._native",
    sources=[
        os.path.join(_CSRC, "native.cpp"),
        os.path.join(_CSRC, "pack_kernels.hip"),
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950",
                 # numerics are pinned to torch's unfused mul/add; hipcc
                 # contracts even __fmul_rn/__fadd_rn into fma otherwise
                 "-ffp-contract=off"],
    },
)

# native gRPC (HTTP/2) transport: C++ (sockets + HPACK) + host-side HIP
# runtime API only (staging.h pinned pipeline for the overlapped send) —
# no kernels, so plain g++ with -lamdhip64 suffices
_ROCM = os.environ.get("ROCM_PATH", "/opt/rocm")
transport_ext = CppExtension(
    name="min_tfs_client_amd._transport",
    sources=[os.path.join(_CSRC, "grpc_transport.cpp")],
    include_dirs=[os.path.join(_ROCM, "include")],
    library_dirs=[os.path.join(_ROCM, "lib")],
    libraries=["amdhip64", "c10_hip", "torch_hip"],
    extra_compile_args={"cxx": ["-O3", "-std=c++17"]},
)

setup(
    name="min-tfs-client-amd",
    version="0.1.0",
    description="MI355X-native TensorFlow-Serving client + serving "
                "framework (byte-compatible wire format, HIP pack path)",
    # the tensorflow/tensorflow_serving pb2 shims and the min_tfs_client
    # alias are namespace packages, like the reference wheel's generated
    # modules (reference setup.py:100-101 find_namespace_packages)
    packages=find_namespace_packages(
        include=["min_tfs_client_amd", "min_tfs_client_amd.*",
                 "min_tfs_client", "tensorflow", "tensorflow.*",
                 "tensorflow_serving", "tensorflow_serving.*"]),
    package_data={"min_tfs_client_amd": ["py.typed"]},
    ext_modules=[ext, transport_ext],
    install_requires=["numpy", "grpcio>=1.21", "protobuf>=3.8", "torch"],
    python_requires=">=3.10",
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
