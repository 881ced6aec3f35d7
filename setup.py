"""setup.py: builds the native pieces in-tree.

``python setup.py build_ext --inplace`` compiles the _devnative C++
extension (g++/pybind11) and the gfx950 attestation library (hipcc) so
both .so files sit inside the package — they ship with a repo snapshot
and need no JIT cache.
"""

from setuptools import setup
from setuptools.command.build_ext import build_ext


class BuildNative(build_ext):
    def run(self):
        from k8s_cc_manager_amd.device.native.build import build as build_dev
        from k8s_cc_manager_amd.ops.build import build as build_attest

        build_dev()
        build_attest()


setup(cmdclass={"build_ext": BuildNative}, ext_modules=[])
