"""ChatGPT-compatible API tests against a dummy-engine Node (no GPU)."""
import asyncio
import json

import pytest
from aiohttp.test_utils import TestClient, TestServer

from xotorch_amd.api.chatgpt import ChatGPTAPI
from xotorch_amd.engine.dummy import DummyEngine
from xotorch_amd.orchestration.node import Node


def run(coro):
  return asyncio.new_event_loop().run_until_complete(coro)


async def make_client():
  node = Node("api-node", None, DummyEngine(), None, max_generate_tokens=4)
  await node.start()
  api = ChatGPTAPI(node, "DummyEngine", default_model="dummy")
  client = TestClient(TestServer(api.app))
  await client.start_server()
  return node, client


def test_models_and_health():
  async def go():
    node, client = await make_client()
    r = await client.get("/healthcheck")
    assert (await r.json())["status"] == "ok"
    r = await client.get("/v1/models")
    data = await r.json()
    ids = [m["id"] for m in data["data"]]
    assert "llama-3-70b" in ids and "dummy" in ids
    r = await client.get("/modelpool")
    assert "model pool" in await r.json()
    r = await client.get("/initial_models")
    assert "llama-3.2-1b" in await r.json()
    r = await client.get("/v1/topology")
    assert "api-node" in (await r.json())["nodes"]
    await client.close()
    await node.stop()
    return True
  assert run(go())


def test_chat_completion_non_streaming():
  async def go():
    node, client = await make_client()
    r = await client.post("/v1/chat/completions", json={
      "model": "dummy", "messages": [{"role": "user", "content": "hi there"}],
    })
    assert r.status == 200, await r.text()
    data = await r.json()
    assert data["object"] == "chat.completion"
    assert data["choices"][0]["message"]["content"]
    assert data["usage"]["completion_tokens"] >= 1
    await client.close()
    await node.stop()
    return True
  assert run(go())


def test_chat_completion_streaming():
  async def go():
    node, client = await make_client()
    r = await client.post("/v1/chat/completions", json={
      "model": "dummy", "stream": True,
      "messages": [{"role": "user", "content": "stream me"}],
    })
    assert r.status == 200
    body = await r.content.read()
    text = body.decode()
    chunks = [l[6:] for l in text.splitlines() if l.startswith("data: ")]
    assert chunks[-1] == "[DONE]"
    parsed = [json.loads(c) for c in chunks[:-1]]
    assert any(p["choices"][0].get("delta", {}).get("content") for p in parsed)
    assert parsed[-1]["choices"][0]["finish_reason"] == "stop"
    await client.close()
    await node.stop()
    return True
  assert run(go())


def test_invalid_model_rejected():
  async def go():
    node, client = await make_client()
    r = await client.post("/v1/chat/completions", json={
      "model": "not-a-model", "messages": [{"role": "user", "content": "x"}],
    })
    assert r.status == 400
    await client.close()
    await node.stop()
    return True
  assert run(go())


def test_tinychat_served():
  async def go():
    node, client = await make_client()
    r = await client.get("/")
    assert r.status == 200
    assert "xotorch_amd" in await r.text()
    await client.close()
    await node.stop()
    return True
  assert run(go())


def test_max_tokens_enforced():
  """`max_tokens` must cap generation per request (it travels in
  inference_state and survives engine hops — engines merge, not replace,
  the state dict)."""
  async def go():
    node, client = await make_client()
    try:
      resp = await client.post("/v1/chat/completions", json={
        "model": "dummy", "messages": [{"role": "user", "content": "hello"}],
        "max_tokens": 3,
      })
      j = await resp.json()
      text = j["choices"][0]["message"]["content"]
      # DummyTokenizer decodes each token as "dummy "
      assert len(text.split()) <= 3, text
    finally:
      await client.close()
      await node.stop()
  run(go())


def test_token_encode_route():
  async def go():
    node, client = await make_client()
    r = await client.post("/v1/chat/token/encode", json={
      "model": "dummy", "messages": [{"role": "user", "content": "hello world"}]})
    data = await r.json()
    assert r.status == 200 and data["num_tokens"] > 0 and isinstance(data["tokens"], list)
    await client.close()
    await node.stop()
    return True
  assert run(go())


def test_cors_headers_on_every_route():
  async def go():
    node, client = await make_client()
    for path in ("/healthcheck", "/v1/models", "/v1/topology"):
      r = await client.get(path)
      assert r.headers.get("Access-Control-Allow-Origin") == "*", path
    r = await client.options("/v1/chat/completions")
    assert r.status == 204
    assert "POST" in r.headers.get("Access-Control-Allow-Methods", "")
    await client.close()
    await node.stop()
    return True
  assert run(go())


def test_post_download_route():
  async def go():
    node, client = await make_client()
    # no downloader wired -> 503 with a clear message; unknown model -> 400
    r = await client.post("/download", json={"model": "nope"})
    assert r.status == 400
    r = await client.post("/download", json={"model": "llama-3-8b"})
    assert r.status == 503
    await client.close()
    await node.stop()
    return True
  assert run(go())


def test_stop_sequences():
  """OpenAI `stop` param: generation is truncated before the stop string in
  both non-streaming and streaming modes, and the request is cancelled."""
  async def go():
    node, client = await make_client()
    # discover the untruncated output first
    r = await client.post("/v1/chat/completions", json={
      "model": "dummy", "messages": [{"role": "user", "content": "hello"}]})
    full = (await r.json())["choices"][0]["message"]["content"]
    assert len(full) >= 3, full
    stop = full[1:3]  # substring from the middle
    want = full[: full.find(stop)]
    r = await client.post("/v1/chat/completions", json={
      "model": "dummy", "stop": stop,
      "messages": [{"role": "user", "content": "hello"}]})
    data = await r.json()
    assert data["choices"][0]["message"]["content"] == want
    assert data["choices"][0]["finish_reason"] == "stop"
    # streaming: concatenated deltas match the same truncation
    r = await client.post("/v1/chat/completions", json={
      "model": "dummy", "stream": True, "stop": [stop],
      "messages": [{"role": "user", "content": "hello"}]})
    text = ""
    async for line in r.content:
      line = line.decode().strip()
      if not line.startswith("data: ") or line == "data: [DONE]":
        continue
      for ch in json.loads(line[6:])["choices"]:
        text += ch["delta"].get("content") or ""
    assert text == want, (text, want)
    await client.close()
    await node.stop()
    return True
  assert run(go())


def test_streaming_carries_final_batch_text():
  """Streaming deltas concatenated must equal the non-streaming content —
  the final chunk carries both text and finish_reason (a chunk with
  finish_reason used to drop its text, losing the last token batch)."""
  async def go():
    node, client = await make_client()
    r = await client.post("/v1/chat/completions", json={
      "model": "dummy", "messages": [{"role": "user", "content": "same seed"}]})
    full = (await r.json())["choices"][0]["message"]["content"]
    r = await client.post("/v1/chat/completions", json={
      "model": "dummy", "stream": True,
      "messages": [{"role": "user", "content": "same seed"}]})
    text = ""
    async for line in r.content:
      line = line.decode().strip()
      if not line.startswith("data: ") or line == "data: [DONE]":
        continue
      for ch in json.loads(line[6:])["choices"]:
        text += ch.get("delta", {}).get("content") or ""
    assert text == full, (text, full)
    await client.close()
    await node.stop()
    return True
  assert run(go())


def test_stream_options_include_usage():
  """OpenAI stream_options.include_usage: a final usage chunk with empty
  choices precedes [DONE]."""
  async def go():
    node, client = await make_client()
    r = await client.post("/v1/chat/completions", json={
      "model": "dummy", "stream": True, "stream_options": {"include_usage": True},
      "messages": [{"role": "user", "content": "hi"}]})
    chunks = []
    async for line in r.content:
      line = line.decode().strip()
      if line.startswith("data: ") and line != "data: [DONE]":
        chunks.append(json.loads(line[6:]))
    assert chunks[-1]["choices"] == []
    u = chunks[-1]["usage"]
    assert u["completion_tokens"] >= 1
    assert u.get("prompt_tokens", 0) >= 1
    assert u.get("total_tokens") == u["prompt_tokens"] + u["completion_tokens"]
    await client.close()
    await node.stop()
    return True
  assert run(go())
