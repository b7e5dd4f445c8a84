"""In-tree extension build for distributed_cluster_gpus_amd.

Builds:
  * distributed_cluster_gpus_amd.ops._des_core     — native scalar DES core
    (C++17, pybind11; no GPU needed)
  * distributed_cluster_gpus_amd.ops._sim_hip      — batched MI355X replica
    engine kernels (HIP, gfx950) — built separately via ops/build_hip.py with
    hipcc (torch.utils.cpp_extension), see __graft_entry__.build().

Usage: python setup.py build_ext --inplace
"""
import pybind11
from setuptools import Extension, setup

ext = Extension(
    "distributed_cluster_gpus_amd.ops._des_core",
    sources=["distributed_cluster_gpus_amd/ops/csrc/des_core.cpp"],
    include_dirs=[pybind11.get_include()],
    language="c++",
    extra_compile_args=["-O3", "-std=c++17", "-fvisibility=hidden"],
)

setup(
    name="distributed_cluster_gpus_amd",
    version="0.1.0",
    packages=["distributed_cluster_gpus_amd"],
    ext_modules=[ext],
)
