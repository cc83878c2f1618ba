from setuptools import find_packages, setup

setup(
    name="fengshen_amd",
    version="0.1.0",
    description=("MI355X-native Chinese foundation-model training framework "
                 "(Fengshenbang-LM capabilities, ground-up AMD CDNA4 rebuild)"),
    packages=find_packages(include=["fengshen_amd", "fengshen_amd.*"]),
    python_requires=">=3.10",
    install_requires=[
        "torch>=2.0",
        "numpy",
        "transformers",
        "pybind11",
    ],
    extras_require={
        "serving": ["fastapi", "uvicorn", "pydantic"],
        "data": ["datasets"],
    },
    entry_points={
        "console_scripts": [
            "fengshen-pipeline=fengshen_amd.cli.fengshen_pipeline:main",
        ],
    },
    package_data={"fengshen_amd": ["ops/_C.so", "data/_helpers.so",
                                   "ops/csrc/*", "data/csrc/*",
                                   "workspace/*/*"]},
)
