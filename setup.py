"""pip-installable packaging for uccl_amd (parity with the reference's
wheel: a single package shipping the native engine + plugin .so)."""
import subprocess
import sys
from pathlib import Path

from setuptools import setup
from setuptools.command.build_py import build_py


class BuildNative(build_py):
    def run(self):
        subprocess.check_call([sys.executable, "-m", "uccl_amd._build"],
                              cwd=Path(__file__).parent)
        super().run()


setup(
    name="uccl-amd",
    version="0.1.0",
    description="MI355X-native GPU communication framework "
                "(collectives, P2P, EP, multipath transport)",
    packages=["uccl_amd", "uccl_amd.collective", "uccl_amd.p2p",
              "uccl_amd.ep", "uccl_amd.transport", "uccl_amd.ukernel",
              "uccl_amd.utils"],
    package_data={"uccl_amd": ["*.so", "lib/*.so", "csrc/**/*"]},
    cmdclass={"build_py": BuildNative},
    python_requires=">=3.10",
)
