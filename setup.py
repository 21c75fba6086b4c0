"""Build registrar_amd._core (pybind11 extension) in-tree.

Usage: python setup.py build_ext --inplace
The standalone daemon/ensemble binaries are built by the Makefile from the
same csrc/ sources.
"""
import glob
import os

from pybind11.setup_helpers import Pybind11Extension, build_ext
from setuptools import setup

CSRC = os.path.join("registrar_amd", "csrc")

ext = Pybind11Extension(
    "registrar_amd._core",
    sources=[
        os.path.join(CSRC, "bindings.cpp"),
        os.path.join(CSRC, "ensemble.cpp"),
        os.path.join(CSRC, "zkclient.cpp"),
        os.path.join(CSRC, "registrar.cpp"),
        os.path.join(CSRC, "health.cpp"),
        os.path.join(CSRC, "orchestrator.cpp"),
        os.path.join(CSRC, "gpu.cpp"),
    ],
    # a header edit must rebuild every TU (setuptools skips unchanged .cpp
    # otherwise — a jute.hpp fix once shipped stale objects)
    depends=sorted(glob.glob(os.path.join(CSRC, "*.hpp"))),
    cxx_std=17,
    extra_compile_args=["-O2", "-g", "-Wall", "-pthread"],
    extra_link_args=["-pthread"],
)

setup(
    name="registrar_amd",
    version="0.1.0",
    description="MI355X-host-native service-registration framework (registrar-compatible)",
    packages=["registrar_amd"],
    ext_modules=[ext],
    cmdclass={"build_ext": build_ext},
    python_requires=">=3.8",
)
