"""Build the kllms_amd gfx950 HIP extension in-tree.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Produces kllms_amd/_C*.so next to the package sources so the snapshot that
ships to a GPU box carries the built extension.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

HIP_DIR = os.path.join("kllms_amd", "ops", "hip")
SOURCES = [
    os.path.join(HIP_DIR, f)
    for f in (
        "bindings.hip",
        "elementwise.hip",
        "attn_decode.hip",
        "attn_prefill.hip",
        "sampling.hip",
        "mfma_selftest.hip",
        "allreduce.hip",
        "moe.hip",
        "levenshtein.hip",
    )
]

setup(
    name="kllms_amd_C",
    ext_modules=[
        CUDAExtension(
            name="kllms_amd._C",
            sources=SOURCES,
            extra_compile_args={
                "cxx": ["-O3"],
                "nvcc": ["-O3", "--offload-arch=gfx950", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
