"""setup.py fallback for pre-PEP-621 setuptools (<61): the ROCm image
ships setuptools 59.6, which cannot read pyproject's [project] table.
Mirrors pyproject.toml — keep the two in sync."""

from setuptools import find_packages, setup

setup(
    name="agentainer-amd",
    version="0.2.0",
    description=("MI355X-native multi-tenant LLM-agent runtime "
                 "(Agentainer-compatible CLI/REST, CDNA4 HIP kernels, "
                 "RCCL over xGMI)"),
    python_requires=">=3.10",
    packages=find_packages(include=["agentainer_amd*"]),
    package_data={"agentainer_amd.ops": ["csrc/*", "*.so"]},
    install_requires=["click", "fastapi", "uvicorn", "httpx", "pyyaml",
                      "numpy"],
    entry_points={"console_scripts": ["agentainer=agentainer_amd.cli:main"]},
)
