"""In-tree build of the g2vec_amd HIP extension for gfx950 (MI355X).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built g2vec_amd/_C*.so stays in-tree (git-ignored) so it travels with
repo snapshots to GPU hosts.
"""
import os

from setuptools import find_packages, setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

setup(
    name="g2vec_amd",
    version="0.1.0",
    packages=find_packages(include=["g2vec_amd", "g2vec_amd.*"]),
    package_data={"g2vec_amd": ["data/*.npz"]},
    ext_modules=[
        CUDAExtension(
            name="g2vec_amd._C",
            sources=["g2vec_amd/ops/csrc/bindings.hip"],
            extra_compile_args={
                "cxx": ["-O3"],
                # G2VEC_DEBUG=1: device bounds asserts + host debug info
                "nvcc": ["-O3", "-std=c++17"] + (
                    ["-DG2VEC_DEBUG=1", "-g"]
                    if os.environ.get("G2VEC_DEBUG") == "1" else []),
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
