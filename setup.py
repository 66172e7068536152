"""Build the in-tree gfx950 HIP extension.

Usage: PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
(The built .so lands in torchrec_amd/ops/ and ships to the GPU box with the
repo snapshot; it is git-ignored so history stays source-only.)
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CppExtension, CUDAExtension  # noqa: E402

CSRC = os.path.join(os.path.dirname(os.path.abspath(__file__)), "torchrec_amd", "ops", "csrc")

ext = CUDAExtension(
    name="torchrec_amd.ops._hip_ops",
    sources=[
        os.path.join(CSRC, "jagged_ops.hip"),
        os.path.join(CSRC, "tbe.hip"),
        os.path.join(CSRC, "quant_tbe.hip"),
        os.path.join(CSRC, "interaction.hip"),
        os.path.join(CSRC, "interaction_mfma.hip"),
        os.path.join(CSRC, "mlp_ops.hip"),
        os.path.join(CSRC, "lt_gemm.hip"),
        os.path.join(CSRC, "cache.hip"),
        os.path.join(CSRC, "bindings.hip"),
    ],
    libraries=["hipblaslt"],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
    },
)

infer_ext = CppExtension(
    name="torchrec_amd.inference._batching",
    sources=[
        os.path.join(
            os.path.dirname(os.path.abspath(__file__)),
            "torchrec_amd",
            "inference",
            "csrc",
            "batching_queue.cpp",
        )
    ],
    extra_compile_args={"cxx": ["-O3", "-std=c++17"]},
)

de_ext = CppExtension(
    name="torchrec_amd.dynamic_embedding._id_transformer",
    sources=[
        os.path.join(
            os.path.dirname(os.path.abspath(__file__)),
            "torchrec_amd",
            "dynamic_embedding",
            "csrc",
            "id_transformer.cpp",
        )
    ],
    extra_compile_args={"cxx": ["-O3", "-std=c++17"]},
)

from setuptools import find_packages

setup(
    name="torchrec_amd",
    version="0.1.0",
    description=(
        "MI355X-native sparse/recommender-systems training framework "
        "(TorchRec capabilities; CDNA4 HIP kernels; RCCL over xGMI)"
    ),
    packages=find_packages(include=["torchrec_amd", "torchrec_amd.*"]),
    python_requires=">=3.10",
    ext_modules=[ext, infer_ext, de_ext],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=True)},
)
