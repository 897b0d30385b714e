import os

from setuptools import find_packages, setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ext_modules = [
    CUDAExtension(
        name="hydragnn_amd.ops._hip_ops",
        sources=["hydragnn_amd/ops/csrc/hip_ops.hip",
                 "hydragnn_amd/ops/csrc/etp.hip",
                 "hydragnn_amd/ops/csrc/mfma_linear.hip",
                 "hydragnn_amd/ops/csrc/varlen_attn.hip",
                 "hydragnn_amd/ops/csrc/gemv.hip",
                 "hydragnn_amd/ops/csrc/radius.hip",
                 "hydragnn_amd/ops/csrc/fused_adamw.hip",
                 "hydragnn_amd/ops/csrc/irreps_linear.hip"],
        extra_compile_args={
            "cxx": ["-O3"],
            "nvcc": ["-O3", "--offload-arch=gfx950"],
        },
    )
]

setup(
    name="hydragnn_amd",
    version="0.1.0",
    description=(
        "MI355X-native multi-headed GNN training framework "
        "(HydraGNN-capability, CDNA4-first)"
    ),
    packages=find_packages(include=["hydragnn_amd", "hydragnn_amd.*"]),
    ext_modules=ext_modules,
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=True)},
)
