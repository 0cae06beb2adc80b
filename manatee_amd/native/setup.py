"""Build the native codec extension in-tree:

    python setup.py build_ext --inplace

(__graft_entry__.build() runs this automatically.)"""

from pybind11.setup_helpers import Pybind11Extension, build_ext
from setuptools import setup

setup(
    name="manatee-amd-native",
    ext_modules=[
        Pybind11Extension(
            "_codec", ["codec.cpp"],
            cxx_std=17,
            extra_compile_args=["-O3"],
        ),
        Pybind11Extension(
            "_jutec", ["jutec.cpp"],
            cxx_std=17,
            extra_compile_args=["-O3"],
        ),
    ],
    cmdclass={"build_ext": build_ext},
)
