"""Build the in-tree gfx950 HIP extension:

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Produces nanorlhf_amd/_C.*.so (git-ignored; ships to GPU boxes with the
source snapshot)."""
import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "nanorlhf_amd", "csrc")

sources = [
    os.path.join(CSRC, f)
    for f in ["bindings.cpp", "elementwise.hip", "logprob.hip", "adamw.hip",
              "sampling.hip", "kvcache.hip", "attention.hip", "masked.hip",
              "lora.hip"]
]

setup(
    name="nanorlhf_amd_C",
    ext_modules=[
        CUDAExtension(
            name="nanorlhf_amd._C",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
