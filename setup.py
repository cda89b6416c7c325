"""Build the machin_amd gfx950 HIP extension in-tree.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Produces machin_amd/ops/_machin_hip*.so next to the package sources so
the built artifact travels with any snapshot of the repo.
"""
import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

HIP_DIR = os.path.join("machin_amd", "ops", "hip")

ext = CUDAExtension(
    name="machin_amd.ops._machin_hip",
    sources=[
        os.path.join(HIP_DIR, "bindings.cpp"),
        os.path.join(HIP_DIR, "sumtree.hip"),
        os.path.join(HIP_DIR, "scans.hip"),
        os.path.join(HIP_DIR, "projection.hip"),
        os.path.join(HIP_DIR, "multi_tensor.hip"),
        os.path.join(HIP_DIR, "distributions.hip"),
        os.path.join(HIP_DIR, "elementwise.hip"),
        os.path.join(HIP_DIR, "conv1_wrw.hip"),
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17"],
    },
)

setup(
    name="machin_amd",
    version="0.1.0",
    packages=[
        "machin_amd",
        "machin_amd.frame",
        "machin_amd.frame.buffers",
        "machin_amd.frame.algorithms",
        "machin_amd.frame.noise",
        "machin_amd.model",
        "machin_amd.model.nets",
        "machin_amd.model.algorithms",
        "machin_amd.parallel",
        "machin_amd.parallel.distributed",
        "machin_amd.parallel.server",
        "machin_amd.env",
        "machin_amd.env.wrappers",
        "machin_amd.env.envs",
        "machin_amd.auto",
        "machin_amd.utils",
        "machin_amd.ops",
    ],
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension},
)
