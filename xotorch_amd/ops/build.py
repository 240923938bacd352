"""In-tree build of the CDNA4 HIP extension (no network, no JIT cache).

Invoked by __graft_entry__.build(). Compiles xotorch_amd/ops/hip/hip_ops.hip
with hipcc --offload-arch=gfx950 into xotorch_amd/ops/_hip_ops.so so the
built artifact travels with the repo snapshot to the GPU box.
"""
from __future__ import annotations

import subprocess
import sys
import sysconfig
from pathlib import Path

PKG_DIR = Path(__file__).resolve().parent
SRC = PKG_DIR / "hip" / "hip_ops.hip"
OUT = PKG_DIR / "_hip_ops.so"


def _torch_paths():
  import torch
  from torch.utils import cpp_extension
  return cpp_extension.include_paths(), cpp_extension.library_paths(), torch._C._GLIBCXX_USE_CXX11_ABI


def needs_build() -> bool:
  if not OUT.exists():
    return True
  return SRC.stat().st_mtime > OUT.stat().st_mtime


def build(force: bool = False, verbose: bool = True) -> Path:
  if not force and not needs_build():
    return OUT
  includes, libpaths, abi = _torch_paths()
  py_inc = sysconfig.get_paths()["include"]
  cmd = [
    "hipcc", "--offload-arch=gfx950", "-O3", "-std=c++17", "-fPIC", "-shared",
    f"-DTORCH_EXTENSION_NAME=_hip_ops", "-DTORCH_API_INCLUDE_EXTENSION_H",
    f"-D_GLIBCXX_USE_CXX11_ABI={1 if abi else 0}", "-DUSE_ROCM", "-D__HIP_PLATFORM_AMD__",
  ]
  for inc in includes:
    cmd += ["-I", inc]
  cmd += ["-I", py_inc, str(SRC), "-o", str(OUT)]
  for lp in libpaths:
    cmd += ["-L", lp]
  cmd += ["-ltorch", "-ltorch_hip", "-lc10", "-lc10_hip", "-ltorch_python"]
  if verbose:
    print("[xotorch_amd] building HIP extension:", " ".join(cmd), file=sys.stderr)
  subprocess.run(cmd, check=True)
  return OUT


if __name__ == "__main__":
  build(force="--force" in sys.argv)
