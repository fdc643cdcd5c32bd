"""Package metadata for megatron_amd (reference setup.py:1-10 packages
megatron.core; here the whole MI355X-native framework is installable).

The HIP extension is built in-tree (`python -m megatron_amd.ops.build` or
`__graft_entry__.build()`), not at pip-install time, so installation works
on machines without hipcc; GPU hosts build the extension once and the .so
lives next to the sources.
"""

from setuptools import find_packages, setup

setup(
    name="megatron_amd",
    version="0.1",
    description=(
        "MI355X-native (CDNA4/gfx950) large-scale LLM training framework: "
        "PyTorch-ROCm + hand-written HIP kernels + RCCL over xGMI"
    ),
    packages=find_packages(include=("megatron_amd", "megatron_amd.*")),
    package_data={"megatron_amd.ops": ["csrc/*.hip", "csrc/*.h", "*.so"]},
    python_requires=">=3.10",
)
