"""Package metadata + in-tree native build.

``python setup.py build_ext --inplace`` delegates to build_native.py so
the .so files land inside the package (they must travel with the source
tree, not a JIT cache).
"""

import os
import sys

from setuptools import Command, find_packages, setup

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


class BuildNative(Command):
    user_options = [("inplace", "i", "build in-tree (always true)")]

    def initialize_options(self):
        self.inplace = True

    def finalize_options(self):
        pass

    def run(self):
        import build_native
        build_native.build()


setup(
    name="kubevirt-gpu-device-plugin-amd",
    version="0.1.0",
    description="MI355X-native KubeVirt GPU device plugin "
                "(vfio passthrough + MxGPU SR-IOV VFs)",
    packages=find_packages(include=["kubevirt_gpu_device_plugin_amd*"]),
    package_data={
        "kubevirt_gpu_device_plugin_amd": ["utils/amd_pci.ids", "*.so"],
    },
    python_requires=">=3.8",
    install_requires=["grpcio", "protobuf"],
    entry_points={
        "console_scripts": [
            "kubevirt-gpu-device-plugin-amd="
            "kubevirt_gpu_device_plugin_amd.cmd.main:main",
        ],
    },
    cmdclass={"build_ext": BuildNative},
)
