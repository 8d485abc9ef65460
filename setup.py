"""Build entry: `python setup.py build_ext --inplace` compiles the gfx950
HIP extension in-tree via deepspeed_amd/ops/build.py (direct hipcc)."""
import sys

from setuptools import Command, setup, find_packages


class BuildExt(Command):
    user_options = [("inplace", "i", "build in-tree (always true)")]

    def initialize_options(self):
        self.inplace = True

    def finalize_options(self):
        pass

    def run(self):
        from deepspeed_amd.ops.build import build
        build(verbose=True)


setup(
    name="deepspeed_amd",
    version="0.1.0",
    description="MI355X-native large-scale training framework "
                "(DeepSpeed-compatible API)",
    packages=find_packages(include=["deepspeed_amd*"]),
    cmdclass={"build_ext": BuildExt},
    python_requires=">=3.10",
    scripts=["bin/deepspeed", "bin/ds_report"],
)
