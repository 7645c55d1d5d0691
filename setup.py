import os

from setuptools import find_packages, setup

setup(
    name="clearml-serving-amd",
    version="0.1.0",
    description="MI355X-native multi-model serving framework "
                "(clearml-serving capabilities, CDNA4/HIP compute tier)",
    packages=find_packages(exclude=["tests", "examples"]),
    python_requires=">=3.8",
    install_requires=[
        "numpy",
        "fastapi",
        "uvicorn",
        "prometheus-client",
        "requests",
    ],
    entry_points={
        "console_scripts": [
            "clearml-serving-amd = clearml_serving_amd.cli:main",
        ],
    },
)
