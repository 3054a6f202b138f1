"""In-tree build of the MI355X HIP extension.

`python setup.py build_ext --inplace` produces
`vainplex_openclaw_amd/_hip_ops.<abi>.so` next to the package so the
snapshot shipped to a GPU box carries the built extension (no JIT cache
dependency). Cross-compiles for gfx950 without a GPU present.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension

SOURCES = [
    "csrc/bindings.hip",
    "csrc/sha256_merkle.hip",
    "csrc/pattern_scan.hip",
    "csrc/encoder.hip",
    "csrc/gemm_nt.hip",
    "csrc/topk_recall.hip",
    "csrc/firewall.hip",
    "csrc/edit_distance.hip",
    "csrc/fact_probe.hip",
    "csrc/host_envelope.cpp",
]

setup(
    name="vainplex_openclaw_amd",
    version="0.1.0",
    packages=["vainplex_openclaw_amd"],
    ext_modules=[
        CUDAExtension(
            name="vainplex_openclaw_amd._hip_ops",
            sources=SOURCES,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
