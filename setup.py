"""Install sofa_amd (pure-Python package + in-tree native helpers).

`pip install -e .` gives the `sofa` console command; native components build
in-tree via `python -m sofa_amd.native.build` (also invoked on first
`sofa record`).  Replaces the reference's install.sh PREFIX-copy scheme.
"""

from setuptools import find_packages, setup

setup(
    name="sofa-amd",
    version="0.1.0",
    description="MI355X-native whole-system performance profiler",
    packages=find_packages(include=["sofa_amd", "sofa_amd.*"]),
    package_data={
        "sofa_amd": [
            "sofaboard/*",
            "native/*/*.cc",
            "native/*/*.hip",
            "native/*/*.h",
            "native/collector/*.h",
            "pystacks_inject/*.py",
        ]
    },
    python_requires=">=3.8",
    install_requires=["numpy", "pandas"],
    entry_points={"console_scripts": ["sofa=sofa_amd.cli:main"]},
)
