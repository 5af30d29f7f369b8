from setuptools import find_packages, setup

setup(
    name="xgboost_amd",
    version="0.1.0",
    description="MI355X-native gradient-boosted trees (dmlc/xgboost "
                "capabilities, CDNA4 HIP kernels, RCCL collectives)",
    packages=find_packages(include=["xgboost_amd", "xgboost_amd.*"]),
    package_data={"xgboost_amd.ops": ["*.so", "cpp/*"]},
    python_requires=">=3.9",
    install_requires=["numpy", "torch", "scipy"],
)
