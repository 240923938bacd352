"""xotorch_amd packaging: `pip install -e .` gives the `xot` CLI; build_ext
compiles the gfx950 HIP extension in-tree (reference parity: setup.py:162's
console script; the accelerator-specific wheel logic is unnecessary here —
this framework targets ROCm/MI355X only)."""
import subprocess
import sys
from pathlib import Path

from setuptools import Command, find_packages, setup
from setuptools.command.build_ext import build_ext as _build_ext


class HipBuildExt(_build_ext):
  def run(self):
    subprocess.run([sys.executable, str(Path(__file__).parent / "xotorch_amd" / "ops" / "build.py")], check=True)


setup(
  name="xotorch_amd",
  version="0.1.0",
  description="MI355X-native peer-partitioned LLM inference & training runtime",
  packages=find_packages(include=["xotorch_amd", "xotorch_amd.*"]),
  package_data={"xotorch_amd": ["api/tinychat/*", "ops/hip/*.hip", "train/data/lora/*.jsonl"]},
  python_requires=">=3.10",
  install_requires=[
    "numpy",
    "msgpack",
    "aiohttp",
    "rich",
    "safetensors",
    "psutil",
  ],
  extras_require={
    "tokenizers": ["transformers", "tokenizers", "sentencepiece"],
  },
  entry_points={"console_scripts": ["xot = xotorch_amd.cli:run"]},
  cmdclass={"build_ext": HipBuildExt},
)
