"""Install byteps_amd (pure-Python package + prebuilt in-tree native core).

The native core is built by ``python -m byteps_amd.ops.build`` (one hipcc
invocation, gfx950).  ``pip install -e .`` registers the ``bpslaunch``
entry point.
"""

import os
import subprocess
import sys

from setuptools import setup, find_packages
from setuptools.command.build_py import build_py


class BuildWithNative(build_py):
    def run(self):
        try:
            subprocess.run(
                [sys.executable, "-m", "byteps_amd.ops.build"],
                check=True,
                cwd=os.path.dirname(os.path.abspath(__file__)))
        except Exception as e:  # allow pure-python install on non-ROCm hosts
            print("warning: native core build skipped: %s" % e,
                  file=sys.stderr)
        super().run()


setup(
    name="byteps_amd",
    version="0.1.0",
    description="MI355X-native gradient-synchronization framework "
                "(BytePS-capability, HIP/CDNA4 + RCCL + native PS)",
    packages=find_packages(include=["byteps_amd", "byteps_amd.*"]),
    package_data={"byteps_amd.ops": ["_core.so", "csrc/*"],
                  "byteps_amd": ["tuning/*.csv"]},
    python_requires=">=3.8",
    cmdclass={"build_py": BuildWithNative},
    entry_points={
        "console_scripts": [
            "bpslaunch = byteps_amd.launcher.launch:main",
        ],
    },
)
