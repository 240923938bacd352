"""Shard-aware HF weight downloader with ranged resume and hash verification.

Capability parity with the reference's downloader stack
(/root/reference/xotorch/download/new_shard_download.py:72-308 and
hf/hf_helpers.py:82-98): file list from the HF API, shard-aware allow
patterns driven by model.safetensors.index.json's weight_map (only the files
containing this shard's layers are fetched), `.partial` files resumed with
HTTP Range, sha256 verification, bounded parallelism, progress events, and
the Singleton/Cached decorator semantics (dedup concurrent ensure_shard
calls, memoize (engine, shard) → path).
"""
from __future__ import annotations

import asyncio
import hashlib
import json
import os
import time
from abc import ABC, abstractmethod
from dataclasses import dataclass, field
from pathlib import Path
from typing import Callable, Dict, List, Optional, Tuple

from xotorch_amd.helpers import DEBUG, XOT_HOME, AsyncCallbackSystem
from xotorch_amd.models.llama import hf_key_map
from xotorch_amd.models.config import config_from_hf
from xotorch_amd.models.registry import get_repo
from xotorch_amd.shard import Shard

HF_ENDPOINT = os.getenv("HF_ENDPOINT", "https://huggingface.co")


@dataclass
class RepoFileProgressEvent:
  repo_id: str
  file_path: str
  downloaded: int
  total: int
  speed: float
  status: str  # in_progress | complete

  def to_dict(self):
    return self.__dict__.copy()


@dataclass
class RepoProgressEvent:
  repo_id: str
  completed_files: int
  total_files: int
  downloaded_bytes: int
  total_bytes: int
  status: str
  file_progress: Dict[str, RepoFileProgressEvent] = field(default_factory=dict)

  def to_dict(self):
    d = self.__dict__.copy()
    d["file_progress"] = {k: v.to_dict() for k, v in self.file_progress.items()}
    return d


class ShardDownloader(ABC):
  @abstractmethod
  async def ensure_shard(self, shard: Shard, engine_classname: str) -> Path:
    ...

  @property
  def on_progress(self) -> AsyncCallbackSystem:
    raise NotImplementedError


class NoopShardDownloader(ShardDownloader):
  """Test downloader: returns a fixed (possibly nonexistent) path."""

  def __init__(self, path: Optional[Path] = None):
    self.path = path or Path("/tmp/noop-shard")
    self._on_progress = AsyncCallbackSystem()

  async def ensure_shard(self, shard: Shard, engine_classname: str) -> Path:
    return self.path

  @property
  def on_progress(self) -> AsyncCallbackSystem:
    return self._on_progress


def models_dir() -> Path:
  d = XOT_HOME / "models"
  d.mkdir(parents=True, exist_ok=True)
  return d


async def fetch_file_list(session, repo_id: str, revision: str = "main", path: str = "") -> List[dict]:
  url = f"{HF_ENDPOINT}/api/models/{repo_id}/tree/{revision}{('/' + path) if path else ''}"
  async with session.get(url) as resp:
    if resp.status != 200:
      raise RuntimeError(f"failed to list {repo_id}: HTTP {resp.status}")
    entries = await resp.json()
  files: List[dict] = []
  for e in entries:
    if e.get("type") == "file":
      files.append(e)
    elif e.get("type") == "directory":
      files.extend(await fetch_file_list(session, repo_id, revision, e["path"]))
  return files


def shard_allow_patterns(shard: Shard, weight_map: Optional[Dict[str, str]], cfg=None) -> List[str]:
  """Which repo files this shard needs (reference hf_helpers.py:82-98)."""
  default = ["*.json", "*.py", "tokenizer.model", "*.tiktoken", "*.txt", "*.model"]
  if weight_map is None:
    return ["*"]
  if cfg is None:
    # conservative: derive needed keys by layer-prefix match
    needed = set()
    for key, fn in weight_map.items():
      if key.startswith("model.layers."):
        lid = int(key.split(".")[2])
        if shard.start_layer <= lid <= shard.end_layer:
          needed.add(fn)
      else:
        # embeddings / norm / head live on first & last shards
        if shard.is_first_layer or shard.is_last_layer:
          needed.add(fn)
  else:
    mapping = hf_key_map(shard, cfg)
    needed = {weight_map[k] for k in mapping if k in weight_map}
  return default + sorted(needed)


def _matches(path: str, patterns: List[str]) -> bool:
  from fnmatch import fnmatch
  return any(fnmatch(path, p) for p in patterns)


async def download_file(session, repo_id: str, revision: str, file: dict, target_dir: Path,
                        on_progress: Callable[[int, int], None] = lambda a, b: None) -> Path:
  """Ranged-resume download with hash verification (reference :141-168)."""
  out = target_dir / file["path"]
  out.parent.mkdir(parents=True, exist_ok=True)
  total = file.get("size", 0)
  if out.exists() and (total == 0 or out.stat().st_size == total):
    on_progress(total, total)
    return out
  partial = out.with_suffix(out.suffix + ".partial")
  resume = partial.stat().st_size if partial.exists() else 0
  url = f"{HF_ENDPOINT}/{repo_id}/resolve/{revision}/{file['path']}"
  headers = {"Range": f"bytes={resume}-"} if resume else {}
  n_read = resume
  async with session.get(url, headers=headers) as resp:
    if resp.status == 416:  # already fully downloaded
      pass
    elif resp.status not in (200, 206):
      raise RuntimeError(f"download {url}: HTTP {resp.status}")
    else:
      if resp.status == 200:
        n_read = 0
        mode = "wb"
      else:
        mode = "ab"
      # plain buffered writes: a 1 MB write is ~microseconds and keeps the
      # loop responsive without an aiofiles dependency
      with open(partial, mode) as f:
        async for chunk in resp.content.iter_chunked(1 << 20):
          f.write(chunk)
          n_read += len(chunk)
          on_progress(n_read, total)
  # verify
  lfs = file.get("lfs") or {}
  expected = lfs.get("oid")
  if expected:
    h = hashlib.sha256()
    with open(partial, "rb") as f:
      for block in iter(lambda: f.read(1 << 22), b""):
        h.update(block)
    if h.hexdigest() != expected:
      partial.unlink(missing_ok=True)
      raise RuntimeError(f"sha256 mismatch for {file['path']}")
  partial.rename(out)
  on_progress(total, total)
  return out


class HFShardDownloader(ShardDownloader):
  def __init__(self, max_parallel: int = 4, revision: str = "main"):
    self.max_parallel = max_parallel
    self.revision = revision
    self._on_progress = AsyncCallbackSystem()
    self._active: Dict[Tuple[str, Shard], asyncio.Task] = {}
    self._cache: Dict[Tuple[str, Shard], Path] = {}
    self._progress_state: Dict[str, dict] = {}

  @property
  def on_progress(self) -> AsyncCallbackSystem:
    return self._on_progress

  def progress(self) -> dict:
    return self._progress_state

  async def ensure_shard(self, shard: Shard, engine_classname: str) -> Path:
    key = (engine_classname, shard)
    if key in self._cache:
      return self._cache[key]
    if key in self._active:  # singleton: dedupe concurrent downloads
      return await self._active[key]
    task = asyncio.create_task(self._download_shard(shard, engine_classname))
    self._active[key] = task
    try:
      path = await task
      self._cache[key] = path
      return path
    finally:
      self._active.pop(key, None)

  async def delete_model(self, model_id: str, engine_classname: str) -> bool:
    import shutil
    repo = get_repo(model_id, engine_classname)
    if repo is None:
      return False
    d = models_dir() / repo.replace("/", "--")
    if d.exists():
      shutil.rmtree(d)
      self._cache = {k: v for k, v in self._cache.items() if v != d}
      return True
    return False

  async def _download_shard(self, shard: Shard, engine_classname: str) -> Path:
    import aiohttp
    repo = get_repo(shard.model_id, engine_classname)
    if repo in (None, "dummy"):
      raise ValueError(f"no repo for model {shard.model_id} / engine {engine_classname}")
    target = models_dir() / repo.replace("/", "--")
    target.mkdir(parents=True, exist_ok=True)
    timeout = aiohttp.ClientTimeout(total=None, connect=30, sock_read=120)
    async with aiohttp.ClientSession(timeout=timeout) as session:
      files = await fetch_file_list(session, repo, self.revision)
      # first pass: config + index so allow-patterns can be shard-aware
      meta_files = [f for f in files if f["path"] in ("config.json", "model.safetensors.index.json")]
      for f in meta_files:
        await download_file(session, repo, self.revision, f, target)
      weight_map = None
      idx = target / "model.safetensors.index.json"
      if idx.exists():
        weight_map = json.loads(idx.read_text()).get("weight_map")
      cfg = None
      if (target / "config.json").exists():
        try:
          cfg = config_from_hf(target / "config.json", shard.model_id)
        except Exception:
          cfg = None
      patterns = shard_allow_patterns(shard, weight_map, cfg)
      wanted = [f for f in files if _matches(f["path"], patterns)]
      total_bytes = sum(f.get("size", 0) for f in wanted)
      state = {"repo": repo, "total_files": len(wanted), "completed_files": 0,
               "total_bytes": total_bytes, "downloaded_bytes": 0, "status": "in_progress"}
      self._progress_state[repo] = state
      sem = asyncio.Semaphore(self.max_parallel)
      start = time.time()
      done_bytes = {"n": 0}

      async def one(f):
        async with sem:
          base = done_bytes["n"]

          def prog(n, total):
            state["downloaded_bytes"] = base + n
            ev = RepoProgressEvent(repo, state["completed_files"], state["total_files"],
                                   state["downloaded_bytes"], total_bytes,
                                   "in_progress")
            self._on_progress.trigger_all(shard, ev)
          await download_file(session, repo, self.revision, f, target, prog)
          done_bytes["n"] += f.get("size", 0)
          state["completed_files"] += 1
      await asyncio.gather(*(one(f) for f in wanted))
      state["status"] = "complete"
      self._on_progress.trigger_all(
        shard, RepoProgressEvent(repo, state["completed_files"], state["total_files"],
                                 state["downloaded_bytes"], total_bytes, "complete"))
    return target


def new_shard_downloader(kind: str = "hf") -> ShardDownloader:
  if kind == "noop":
    return NoopShardDownloader()
  return HFShardDownloader()
