"""Build the gfx950 HIP extension in-tree:

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Produces llmapigateway_amd/ops/_C*.so (git-ignored; ships to the GPU box
with the gpurun snapshot).
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "llmapigateway_amd", "ops", "csrc")

sources = [
    os.path.join(CSRC, "ext.cpp"),
    os.path.join(CSRC, "elementwise.hip"),
    os.path.join(CSRC, "attention_decode.hip"),
    os.path.join(CSRC, "attention_prefill.hip"),
    os.path.join(CSRC, "sampling.hip"),
    os.path.join(CSRC, "gemm_skinny.hip"),
    os.path.join(CSRC, "gemm_m256.hip"),
]

setup(
    name="llmapigateway_amd_ext",
    ext_modules=[
        CUDAExtension(
            name="llmapigateway_amd.ops._C",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                # mfma-vgpr-form: keep MFMA accumulators in arch VGPRs —
                # the AGPR form copies every accumulator through
                # v_accvgpr_read/write pairs each loop iteration
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950",
                         "-mllvm", "-amdgpu-mfma-vgpr-form"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
