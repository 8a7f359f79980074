"""setup.py — `python setup.py build_ext --inplace` builds the gfx950 HIP
extension in-tree (delegates to video_features_amd/ops/build.py, which drives
hipcc directly; PYTORCH_ROCM_ARCH overrides the offload arch)."""
import sys

from setuptools import Command, find_packages, setup


class BuildHip(Command):
    user_options = [('inplace', 'i', 'build in-tree (always true here)')]

    def initialize_options(self):
        self.inplace = True

    def finalize_options(self):
        pass

    def run(self):
        from video_features_amd.ops.build import build
        build()


setup(
    name='video_features_amd',
    version='0.1.0',
    packages=find_packages(include=['video_features_amd*']),
    package_data={'video_features_amd.ops': ['_vfa_hip.so'],
                  'video_features_amd.utils': ['*.json']},
    cmdclass={'build_ext': BuildHip},
    python_requires='>=3.10',
)
