"""Build the in-tree gfx950 HIP extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

hipcc cross-compiles without a GPU; the built .so lives inside the package
(neuronx_distributed_inference_amd/ops/) so the repo snapshot that travels to
a GPU box carries it.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

CSRC = os.path.join("neuronx_distributed_inference_amd", "ops", "csrc")

sources = [
    os.path.join(CSRC, "bindings.cpp"),
    os.path.join(CSRC, "elementwise.hip"),
    os.path.join(CSRC, "kv_update.hip"),
    os.path.join(CSRC, "attention_decode.hip"),
    os.path.join(CSRC, "sampling.hip"),
    os.path.join(CSRC, "attention_prefill.hip"),
    os.path.join(CSRC, "qkv_fused.hip"),
    os.path.join(CSRC, "fp8_quant.hip"),
    os.path.join(CSRC, "skinny_gemm.hip"),
    os.path.join(CSRC, "skinny_gemm2.hip"),
    os.path.join(CSRC, "skinny_gemm3.hip"),
    os.path.join(CSRC, "skinny_gemm4.hip"),
    os.path.join(CSRC, "skinny_gemm5.hip"),
    os.path.join(CSRC, "moe_grouped.hip"),
    os.path.join(CSRC, "router_topk.hip"),
    os.path.join(CSRC, "lt_gemm.cpp"),
]

extra = {
    "cxx": ["-O3", "-std=c++17"],
    "nvcc": [
        "-O3",
        "-std=c++17",
        "--offload-arch=gfx950",
        "-fno-gpu-rdc",
    ],
}

setup(
    name="neuronx_distributed_inference_amd",
    version="0.1.0",
    packages=["neuronx_distributed_inference_amd"],
    ext_modules=[
        CUDAExtension(
            name="neuronx_distributed_inference_amd.ops._nxdi_C",
            sources=sources,
            extra_compile_args=extra,
            libraries=["hipblaslt"],
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
