"""Downloader tests (offline): shard-aware allow patterns, ranged resume
against a local aiohttp server, sha256 verification, singleton dedup."""
import asyncio
import hashlib
import json

import pytest
from aiohttp import web

from xotorch_amd.download.downloader import (
  HFShardDownloader,
  NoopShardDownloader,
  download_file,
  shard_allow_patterns,
  _matches,
)
from xotorch_amd.models.config import config_from_hf
from xotorch_amd.models.registry import BUILTIN_CONFIGS
from xotorch_amd.shard import Shard


def run(coro):
  return asyncio.new_event_loop().run_until_complete(coro)


def test_allow_patterns_shard_aware():
  cfg = config_from_hf(BUILTIN_CONFIGS["llama-3-70b"], "llama-3-70b")
  weight_map = {}
  for lid in range(80):
    for part in ("self_attn.q_proj", "self_attn.k_proj", "self_attn.v_proj", "self_attn.o_proj",
                 "mlp.gate_proj", "mlp.up_proj", "mlp.down_proj",
                 "input_layernorm", "post_attention_layernorm"):
      weight_map[f"model.layers.{lid}.{part}.weight"] = f"model-{lid // 10:05d}.safetensors"
  weight_map["model.embed_tokens.weight"] = "model-00000.safetensors"
  weight_map["model.norm.weight"] = "model-00007.safetensors"
  weight_map["lm_head.weight"] = "model-00007.safetensors"
  mid = Shard("llama-3-70b", 30, 39, 80)
  pats = shard_allow_patterns(mid, weight_map, cfg)
  assert "model-00003.safetensors" in pats
  assert "model-00000.safetensors" not in pats
  assert not _matches("model-00005.safetensors", pats)
  assert _matches("config.json", pats)
  first = Shard("llama-3-70b", 0, 9, 80)
  assert "model-00000.safetensors" in shard_allow_patterns(first, weight_map, cfg)
  last = Shard("llama-3-70b", 70, 79, 80)
  assert "model-00007.safetensors" in shard_allow_patterns(last, weight_map, cfg)


def test_download_resume_and_hash(tmp_path):
  payload = bytes(range(256)) * 512  # 128 KB
  sha = hashlib.sha256(payload).hexdigest()

  async def go():
    ranged_calls = []

    async def handler(request):
      rng = request.headers.get("Range")
      if rng:
        start = int(rng.split("=")[1].rstrip("-"))
        ranged_calls.append(start)
        return web.Response(status=206, body=payload[start:])
      return web.Response(body=payload)

    app = web.Application()
    app.router.add_get("/repo/resolve/main/weights.bin", handler)
    app.router.add_get("/repo/resolve/main/weights2.bin", handler)
    runner = web.AppRunner(app)
    await runner.setup()
    site = web.TCPSite(runner, "127.0.0.1", 0)
    await site.start()
    port = site._server.sockets[0].getsockname()[1]

    import aiohttp
    import xotorch_amd.download.downloader as dl
    old = dl.HF_ENDPOINT
    dl.HF_ENDPOINT = f"http://127.0.0.1:{port}"
    try:
      file = {"path": "weights.bin", "size": len(payload), "lfs": {"oid": sha}}
      # simulate an interrupted download: half the payload in .partial
      (tmp_path / "weights.bin.partial").write_bytes(payload[: len(payload) // 2])
      async with aiohttp.ClientSession() as session:
        out = await dl.download_file(session, "repo", "main", file, tmp_path)
      assert out.read_bytes() == payload
      assert ranged_calls == [len(payload) // 2], "must resume from the partial offset"
      # second call: already complete, no network
      async with aiohttp.ClientSession() as session:
        out2 = await dl.download_file(session, "repo", "main", file, tmp_path)
      assert out2.read_bytes() == payload
      # corrupt hash must raise
      bad = {"path": "weights2.bin", "size": len(payload), "lfs": {"oid": "0" * 64}}
      with pytest.raises(RuntimeError, match="sha256"):
        async with aiohttp.ClientSession() as session:
          await dl.download_file(session, "repo", "main", bad, tmp_path)
    finally:
      dl.HF_ENDPOINT = old
      await runner.cleanup()
    return True

  assert run(go())


def test_noop_downloader():
  async def go():
    d = NoopShardDownloader()
    p = await d.ensure_shard(Shard("dummy", 0, 3, 4), "DummyEngine")
    return p is not None
  assert run(go())
