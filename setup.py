"""Build script: python package + in-tree HIP extension for gfx950.

Build the extension in-tree (the .so travels with repo snapshots):

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
"""

from __future__ import annotations

import os
from pathlib import Path

from setuptools import find_packages, setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

ext_modules = []
cmdclass = {}
try:
    from torch.utils.cpp_extension import BuildExtension, CUDAExtension

    csrc = Path(__file__).parent / "sheeprl_amd" / "ops" / "csrc"
    # exclude torch-hipify outputs: X_hip.hip generated from an existing X.hip
    def _is_hipify_output(p: Path) -> bool:
        return p.name.endswith("_hip.hip") and (p.parent / (p.name[: -len("_hip.hip")] + ".hip")).exists()

    sources = sorted(
        str(p) for p in csrc.glob("*.hip") if not _is_hipify_output(p)
    ) + sorted(str(p) for p in csrc.glob("*.cpp"))
    if sources:
        ext_modules.append(
            CUDAExtension(
                name="sheeprl_amd.ops._sheep_hip",
                sources=sources,
                extra_compile_args={
                    "cxx": ["-O3", "-std=c++17"],
                    "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
                },
            )
        )
        cmdclass["build_ext"] = BuildExtension.with_options(no_python_abi_suffix=False)
except Exception as e:  # pragma: no cover
    print(f"[setup.py] torch cpp_extension unavailable ({e}); building pure-python package")

setup(
    name="sheeprl-amd",
    version="0.1.0",
    description="MI355X-native distributed deep-RL framework (sheeprl capability set)",
    packages=find_packages(include=["sheeprl_amd", "sheeprl_amd.*"]),
    package_data={"sheeprl_amd": ["configs/**/*.yaml", "ops/csrc/*"]},
    python_requires=">=3.10",
    ext_modules=ext_modules,
    cmdclass=cmdclass,
    entry_points={
        "console_scripts": [
            "sheeprl-amd=sheeprl_amd.cli:main",
            "sheeprl-amd-eval=sheeprl_amd.cli:evaluation",
            "sheeprl-amd-registration=sheeprl_amd.cli:registration",
            "sheeprl-amd-agents=sheeprl_amd.cli:available_agents",
        ]
    },
)
