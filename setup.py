"""Build the in-tree gfx950 HIP extension.

``python setup.py build_ext --inplace`` (or any build command) produces
spark_ensemble_amd/_hip_ops.so via tools/build_ext.py (direct hipcc, native
HIP — no hipify).
"""

import sys

from setuptools import setup
from setuptools.command.build_ext import build_ext as _build_ext


class BuildHip(_build_ext):
    def run(self):
        sys.path.insert(0, "tools")
        from build_ext import build

        build()


setup(
    name="spark_ensemble_amd",
    version="0.1.0",
    description="MI355X-native ensemble learning framework",
    packages=[
        "spark_ensemble_amd",
        "spark_ensemble_amd.boosting",
        "spark_ensemble_amd.classification",
        "spark_ensemble_amd.ensemble",
        "spark_ensemble_amd.models",
        "spark_ensemble_amd.ops",
        "spark_ensemble_amd.parallel",
        "spark_ensemble_amd.regression",
        "spark_ensemble_amd.utils",
    ],
    package_data={"spark_ensemble_amd": ["*.so"]},
    cmdclass={"build_ext": BuildHip},
    ext_modules=[],
)
