"""In-tree build of the polyrl_amd HIP extension for gfx950 (MI355X).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built .so lands next to the package (polyrl_amd/_hip_C*.so) so it travels
with a repo snapshot to GPU boxes.
"""
import glob
import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

# hipify writes *_hip.hip copies of each source; a header-only edit does not
# refresh them, which leaves stale objects in incremental builds. Remove the
# copies so every build re-derives them from the true sources.
for _f in glob.glob(os.path.join("polyrl_amd", "ops", "csrc", "*_hip.hip")):
    os.remove(_f)

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

CSRC = os.path.join("polyrl_amd", "ops", "csrc")

sources = [
    os.path.join(CSRC, f)
    for f in (
        "bindings.cpp",
        "gemm_tuned.cpp",
        "elementwise.hip",
        "rmsnorm_train.hip",
        "rope_train.hip",
        "logprobs.hip",
        "sampling.hip",
        "kv_cache.hip",
        "attention_decode.hip",
        "attention_prefill.hip",
        "attention_backward.hip",
        "attention_backward_v3.hip",
    )
]

setup(
    name="polyrl_amd_hip",
    ext_modules=[
        CUDAExtension(
            name="polyrl_amd._hip",
            sources=sources,
            libraries=["hipblaslt"],
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
