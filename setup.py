from setuptools import find_packages, setup

setup(
    name="skycomputing_amd",
    version="0.2.0",
    description=(
        "MI355X-native load-balanced pipeline-parallel training framework "
        "(capabilities of hpcaitech/SkyComputing, rebuilt for gfx950/CDNA4)"
    ),
    packages=find_packages(include=["skycomputing_amd", "skycomputing_amd.*"]),
    package_data={"skycomputing_amd.ops": ["hip/*.hip", "hip/*.h", "hip/*.so"]},
    python_requires=">=3.10",
    install_requires=["torch>=2.4", "numpy"],
)
