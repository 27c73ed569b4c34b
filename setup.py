"""Build the in-tree gfx950 HIP extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The .so lands at gradient_accumulation_tf_estimator_amd/ops/_ga_hip*.so so it
travels with repo snapshots to GPU boxes (no JIT cache involved).
"""

import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup  # noqa: E402
from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

HERE = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(HERE, "gradient_accumulation_tf_estimator_amd", "ops", "csrc")

setup(
    name="gradient_accumulation_tf_estimator_amd",
    version="0.1.0",
    packages=["gradient_accumulation_tf_estimator_amd"],
    ext_modules=[
        CUDAExtension(
            name="gradient_accumulation_tf_estimator_amd.ops._ga_hip",
            sources=[
                os.path.join(CSRC, "ga_kernels.hip"),
                os.path.join(CSRC, "fused_ln_gelu.hip"),
                os.path.join(CSRC, "blas_acc.hip"),
                os.path.join(CSRC, "lt_gemm.hip"),
                os.path.join(CSRC, "grouped_wgrad.hip"),
                os.path.join(CSRC, "attn.hip"),
                os.path.join(CSRC, "wgrad_mfma.hip"),
                os.path.join(CSRC, "ffn_mfma.hip"),
                os.path.join(CSRC, "cls_head.hip"),
                os.path.join(CSRC, "linear_small.hip"),
                os.path.join(CSRC, "ga_bindings.hip"),
            ],
            libraries=["hipblaslt"],
            # GA_ASAN=1: instrument the HOST side (bindings, launch plumbing,
            # hipBLASLt workspace management) with AddressSanitizer; device
            # code is unchanged. Run via tools/asan_smoke.sh (LD_PRELOADs
            # libasan). Device-side ASAN needs xnack+ and is not exercised
            # on this pool.
            # Compile-only instrumentation: the __asan_* references stay
            # undefined in the .so (legal for shared objects) and resolve
            # from the LD_PRELOADed clang runtime at run time -- the link
            # step is g++, whose libasan must not be mixed in.
            extra_compile_args={
                "cxx": ["-O3"] + (["-fsanitize=address", "-g"]
                                  if os.environ.get("GA_ASAN") else []),
                "nvcc": ["-O3", "--offload-arch=gfx950"]
                + (["-fsanitize=address", "-g",
                    "-Xarch_device", "-fno-sanitize=all"]
                   if os.environ.get("GA_ASAN") else []),
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
