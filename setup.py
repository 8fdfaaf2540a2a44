"""In-tree build: `python setup.py build_ext --inplace` compiles the gfx950
HIP extension (hipcc, no hipify) into alpa_amd/ops/_hip_ops.so."""
from setuptools import Command, find_packages, setup


class BuildHipExt(Command):
    user_options = [("inplace", "i", "build in-tree (always true)")]

    def initialize_options(self):
        self.inplace = True

    def finalize_options(self):
        pass

    def run(self):
        from alpa_amd.ops.build_ext import build_all
        build_all()


setup(
    name="alpa_amd",
    version="0.1.0",
    description="MI355X-native auto-parallelization framework "
                "(Alpa capabilities; PyTorch-ROCm + gfx950 HIP + RCCL)",
    packages=find_packages(include=["alpa_amd", "alpa_amd.*"]),
    cmdclass={"build_ext": BuildHipExt},
    python_requires=">=3.9",
)
