"""setup shim: builds the gfx950 native extension in-tree before packaging.

The extension is compiled by build_native.py (hipcc --offload-arch=gfx950)
into spark_tfrecord_amd/_native.so and shipped as package data — there is no
portable-wheel story for a single-arch HIP binary, and in-tree builds keep
the .so next to the sources for the GPU test harness.
"""

from setuptools import setup
from setuptools.command.build_py import build_py


class BuildWithNative(build_py):
    def run(self):
        import build_native

        build_native.build()
        super().run()


setup(
    cmdclass={"build_py": BuildWithNative},
    package_data={"spark_tfrecord_amd": ["_native.so"]},
)
