"""Build colossalai_amd._C — the gfx950 HIP kernel extension.

Built IN-TREE (``python setup.py build_ext --inplace``) so the resulting .so
travels with the repo snapshot to GPU boxes. gfx950 only — no multi-arch
fatbins, no CUDA paths.
"""

import os

from setuptools import find_packages, setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CppExtension, CUDAExtension  # noqa: E402

THIS_DIR = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(THIS_DIR, "colossalai_amd", "csrc")
CSRC_CPU = os.path.join(THIS_DIR, "colossalai_amd", "csrc_cpu")

sources = [os.path.join(CSRC, f) for f in sorted(os.listdir(CSRC)) if f.endswith((".hip", ".cpp")) and not f.endswith("_hip.hip")]
cpu_sources = [os.path.join(CSRC_CPU, f) for f in sorted(os.listdir(CSRC_CPU)) if f.endswith(".cpp")]

ext = CUDAExtension(
    name="colossalai_amd._C",
    sources=sources,
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": [  # hipcc on ROCm
            "-O3",
            "-std=c++17",
            "--offload-arch=gfx950",
            "-fgpu-flush-denormals-to-zero",
        ],
    },
)

cpu_ext = CppExtension(
    name="colossalai_amd._C_cpu",
    sources=cpu_sources,
    extra_compile_args=["-O3", "-std=c++17", "-fopenmp", "-mavx2", "-mfma", "-ffast-math"],
    extra_link_args=["-fopenmp"],
)

setup(
    name="colossalai_amd",
    version="0.1.0",
    packages=find_packages(include=["colossalai_amd", "colossalai_amd.*"]),
    ext_modules=[ext, cpu_ext],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
