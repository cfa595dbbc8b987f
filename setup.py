"""setuptools<61 fallback: mirrors pyproject.toml's [project] metadata so an
offline editable/sdist install works with the system setuptools (59.x here).
Newer pips with build isolation read pyproject.toml directly."""
from setuptools import find_packages, setup

setup(
    name="flaxdiff-amd",
    version="0.2.0",
    description=("MI355X-native diffusion training + sampling library "
                 "(FlaxDiff capabilities on PyTorch-ROCm with hand-written "
                 "CDNA4 HIP kernels)"),
    python_requires=">=3.10",
    license="MIT",
    packages=find_packages(include=["flaxdiff_amd*"]),
    py_modules=["training", "bench", "bench_sample", "demo_hilbert_curve"],
    package_data={"flaxdiff_amd.ops.hip": ["*.hip", "*.cpp", "*.h", "*.so"]},
    install_requires=["torch>=2.4", "numpy", "einops", "tqdm", "pyyaml"],
    extras_require={
        "data": ["datasets", "safetensors"],
        "text": ["transformers", "sentencepiece"],
        "metrics": ["scipy"],
        "serve": ["fastapi", "uvicorn"],
    },
    entry_points={"console_scripts": ["flaxdiff-train=training:main"]},
)
