from setuptools import setup, find_packages

setup(
    name="dfno_amd",
    version="0.1.0",
    description="MI355X-native distributed Fourier Neural Operator framework",
    packages=find_packages(include=["dfno_amd", "dfno_amd.*"]),
    python_requires=">=3.9",
    install_requires=["numpy", "torch"],
)
