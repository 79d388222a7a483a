from setuptools import find_packages, setup

setup(
    name="k8s-device-plugin-amd",
    version="0.2.0",
    description=(
        "MI355X-native Kubernetes device plugin and node labeller for "
        "AMD Instinct GPUs"
    ),
    packages=find_packages(exclude=["tests"]),
    package_data={
        "k8s_device_plugin_amd.native": ["amdgpu.ids", "*.so"],
    },
    python_requires=">=3.9",
    install_requires=["grpcio", "protobuf"],
    extras_require={"labeller": ["requests"]},
    entry_points={
        "console_scripts": [
            "amd-device-plugin=k8s_device_plugin_amd.cli:device_plugin_main",
            "amd-node-labeller=k8s_device_plugin_amd.cli:labeller_main",
            "amd-partitionctl=k8s_device_plugin_amd.cli:partition_main",
        ]
    },
)
