"""Build/install for the MI355X gpushare device plugin.

`python setup.py build_ext --inplace` (or `make build`) compiles the three
native extensions in-tree via gpushare_amd/native/build.py (g++ for the
amdsmi shim + devlist codec, hipcc --offload-arch=gfx950 for the canary).
"""

import subprocess
import sys

from setuptools import Command, find_packages, setup


class BuildNative(Command):
    description = "build native extensions in-tree (g++ + hipcc/gfx950)"
    user_options = []

    def initialize_options(self):
        pass

    def finalize_options(self):
        pass

    def run(self):
        subprocess.run(
            [sys.executable, "-m", "gpushare_amd.native.build"], check=True
        )


setup(
    name="gpushare-amd-device-plugin",
    version="0.1.0",
    description="MI355X-native Kubernetes GPU-sharing device plugin "
    "(aliyun.com/gpu-mem, amdsmi + KFD + /dev/dri injection)",
    packages=find_packages(include=["gpushare_amd*"]),
    package_data={"gpushare_amd": ["*.so"]},
    python_requires=">=3.9",
    install_requires=["grpcio", "protobuf", "pyyaml", "pybind11"],
    cmdclass={"build_native": BuildNative},
    entry_points={
        "console_scripts": [
            "amdgpushare-device-plugin=gpushare_amd.cli.daemon:main",
            "kubectl-inspect-gpushare=gpushare_amd.cli.inspect:main",
            "kubectl-inspect-gpushare-v2=gpushare_amd.cli.inspect:main",
            "gpushare-podgetter=gpushare_amd.cli.podgetter:main",
            "gpushare-scheduler-extender=gpushare_amd.extender.__main__:main",
            "gpushare-top=gpushare_amd.cli.top:main",
        ]
    },
)
