"""setup.py — explicit metadata for the image's setuptools 59 (no PEP 621).

Also builds the gfx950 HIP engine in-tree when hipcc is available
(``python setup.py build_ext`` equivalent via the custom build step is not
needed — ops/build.py owns the hipcc invocation; __graft_entry__.build()
and `python -m agentbom_amd.ops.build` are the supported entry points).
"""

from setuptools import find_packages, setup

setup(
    name="agentbom-amd",
    version="0.1.0",
    description=(
        "MI355X-native AI-BOM security scanner and blast-radius graph engine "
        "(agent-bom capabilities, CDNA4 HIP kernels, RCCL over xGMI)"
    ),
    python_requires=">=3.10",
    packages=find_packages(include=["agentbom_amd*"]),
    include_package_data=True,
    package_data={"agentbom_amd.ops": ["csrc/*", "_abom_gpu.so"],
                  "agentbom_amd": ["data/*.json"]},
    entry_points={"console_scripts": ["agent-bom = agentbom_amd.cli:cli_main"]},
)
