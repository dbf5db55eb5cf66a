"""Build integration: `python setup.py build_ext --inplace` compiles the
gfx950 HIP extension in-tree via hipcc (see keystone_amd/hip/__init__.py;
the .so is intentionally kept inside the package so it travels with the
source checkout rather than a site-packages install)."""

from setuptools import setup
from setuptools.command.build_ext import build_ext


class HipBuild(build_ext):
    def run(self):
        import sys
        sys.path.insert(0, ".")
        from keystone_amd import hip
        hip.build(force=False)


setup(cmdclass={"build_ext": HipBuild}, ext_modules=[])
