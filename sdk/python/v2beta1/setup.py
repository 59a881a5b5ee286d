from setuptools import find_packages, setup

setup(
    name="kubeflow-mpi",
    version="2.0.0a0+amd",
    description="MPIJob v2beta1 client models (MI355X-native mpi-operator stack)",
    packages=find_packages(include=["mpijob", "mpijob.*"]),
    python_requires=">=3.8",
)
