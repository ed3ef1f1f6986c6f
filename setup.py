"""setup.py — in-tree native builds: `python setup.py build_ext --inplace`."""

import sys

from setuptools import Command, find_packages, setup


class BuildNative(Command):
    description = "build in-tree native extensions (_amdhal, _hiphealth)"
    user_options = [("inplace", "i", "build into the package dir (always on)")]

    def initialize_options(self):
        self.inplace = True

    def finalize_options(self):
        pass

    def run(self):
        sys.path.insert(0, ".")
        from k8s_dra_driver_amd import build_native

        build_native.build()


setup(
    name="k8s-dra-driver-amd",
    version="0.1.0",
    description="MI355X-native Kubernetes Dynamic Resource Allocation driver",
    packages=find_packages(include=["k8s_dra_driver_amd*"]),
    package_data={"k8s_dra_driver_amd": ["*.so"]},
    python_requires=">=3.9",
    cmdclass={"build_ext": BuildNative},
    entry_points={
        "console_scripts": [
            "amd-dra-kubeletplugin=k8s_dra_driver_amd.plugin.main:main",
            "amd-dra-controller=k8s_dra_driver_amd.controller.main:main",
            "amd-dra-ctl=k8s_dra_driver_amd.ctl:main",
            "amd-dra-workload=k8s_dra_driver_amd.workload:main",
            "amd-dra-demo=k8s_dra_driver_amd.demo_runner:main",
        ]
    },
)
