"""ChatGPT-compatible HTTP API + web chat UI host.

Route parity with the reference's API
(/root/reference/xotorch/api/chatgpt_api.py:208-229): /v1/chat/completions
(SSE streaming + non-streaming), /v1/models, /modelpool, /initial_models,
/v1/topology, /healthcheck, /quit, model delete/download progress, static
tinychat hosting. Tokens flow from Node.on_token callbacks into per-request
asyncio queues.
"""
from __future__ import annotations

import asyncio
import json
import time
import uuid
from pathlib import Path
from typing import Callable, Dict, List, Optional

from aiohttp import web

from xotorch_amd.helpers import DEBUG, shutdown
from xotorch_amd.models.registry import build_base_shard, get_repo, get_supported_models, model_cards, pretty_name
from xotorch_amd.engine.tokenizers import resolve_tokenizer


class Message:
  def __init__(self, role: str, content):
    self.role = role
    self.content = content

  def to_dict(self):
    return {"role": self.role, "content": self.content}


class ChatCompletionRequest:
  def __init__(self, model: str, messages: List[Message], temperature: float, tools=None, max_tokens=None, stream=False):
    self.model = model
    self.messages = messages
    self.temperature = temperature
    self.tools = tools
    self.max_tokens = max_tokens
    self.stream = stream


def build_prompt(tokenizer, messages: List[Message], tools=None) -> str:
  conv = [m.to_dict() for m in messages]
  try:
    return tokenizer.apply_chat_template(
      conversation=conv, tokenize=False, add_generation_prompt=True, tools=tools
    )
  except Exception:
    return "\n".join(f"{m.role}: {m.content}" for m in messages) + "\nassistant:"


def generate_completion(request_id: str, tokens: List[int], decoded: str, model: str,
                        finish_reason: Optional[str], stream: bool, object_type: str,
                        prompt_tokens: Optional[int] = None) -> dict:
  completion = {
    "id": f"chatcmpl-{request_id}",
    "object": object_type,
    "created": int(time.time()),
    "model": model,
    "system_fingerprint": "xotorch_amd-0.1.0",
    "choices": [{
      "index": 0,
      "logprobs": None,
      "finish_reason": finish_reason,
    }],
  }
  if stream:
    # the final chunk may carry both text and finish_reason: dropping the
    # text would lose the last token batch on streaming clients
    completion["choices"][0]["delta"] = {"role": "assistant", "content": decoded} if decoded else {}
  else:
    completion["choices"][0]["message"] = {"role": "assistant", "content": decoded}
    completion["usage"] = {"completion_tokens": len(tokens)}
    if prompt_tokens is not None:
      completion["usage"]["prompt_tokens"] = prompt_tokens
      completion["usage"]["total_tokens"] = prompt_tokens + len(tokens)
  return completion


class ChatGPTAPI:
  def __init__(self, node, inference_engine_classname: str = "TorchEngine",
               response_timeout: float = 120.0, on_chat_completion_request: Optional[Callable] = None,
               default_model: Optional[str] = None, shard_downloader=None):
    self.node = node
    self.inference_engine_classname = inference_engine_classname
    self.response_timeout = response_timeout
    self.on_chat_completion_request = on_chat_completion_request
    self.default_model = default_model or "llama-3.2-1b"
    self.shard_downloader = shard_downloader
    self.token_queues: Dict[str, asyncio.Queue] = {}
    self.prev_token_lens: Dict[str, int] = {}

    self.app = web.Application(client_max_size=100 * 1024 * 1024)
    self.app.router.add_post("/v1/chat/completions", self.handle_post_chat_completions)
    self.app.router.add_post("/chat/completions", self.handle_post_chat_completions)
    self.app.router.add_get("/v1/models", self.handle_get_models)
    self.app.router.add_get("/models", self.handle_get_models)
    self.app.router.add_get("/v1/topology", self.handle_get_topology)
    self.app.router.add_get("/topology", self.handle_get_topology)
    self.app.router.add_get("/modelpool", self.handle_model_support)
    self.app.router.add_get("/initial_models", self.handle_get_initial_models)
    self.app.router.add_get("/healthcheck", self.handle_healthcheck)
    self.app.router.add_post("/quit", self.handle_quit)
    self.app.router.add_delete("/models/{model_name}", self.handle_delete_model)
    self.app.router.add_get("/v1/download/progress", self.handle_get_download_progress)
    self.app.router.add_post("/v1/chat/token/encode", self.handle_post_chat_token_encode)
    self.app.router.add_post("/download", self.handle_post_download)
    static_dir = Path(__file__).parent / "tinychat"
    if static_dir.exists():
      self.app.router.add_get("/", self.handle_root)
      self.app.router.add_static("/static/", static_dir, name="static")
    self.app.middlewares.append(self._timeout_middleware)
    self.app.middlewares.append(self._log_middleware)
    self.app.middlewares.append(self._cors_middleware)
    # node token plumbing
    if node is not None:
      node.on_token.register("chatgpt-api-token-handler").on_next(self._on_token)
    self._runner = None

  @web.middleware
  async def _log_middleware(self, request, handler):
    # request log (reference chatgpt_api.py:255-260), DEBUG-gated
    from xotorch_amd.helpers import DEBUG
    if DEBUG >= 2:
      print(f"[api] {request.method} {request.path}")
    return await handler(request)

  @web.middleware
  async def _cors_middleware(self, request, handler):
    # permissive CORS on every route (reference wraps each route in
    # aiohttp_cors, chatgpt_api.py:208-223; tinychat runs cross-origin)
    if request.method == "OPTIONS":
      resp = web.Response(status=204)
    else:
      try:
        resp = await handler(request)
      except web.HTTPException as e:
        resp = e
    resp.headers["Access-Control-Allow-Origin"] = "*"
    resp.headers["Access-Control-Allow-Methods"] = "GET, POST, DELETE, OPTIONS"
    resp.headers["Access-Control-Allow-Headers"] = "Content-Type, Authorization"
    if isinstance(resp, web.HTTPException):
      raise resp
    return resp

  @web.middleware
  async def _timeout_middleware(self, request, handler):
    try:
      return await asyncio.wait_for(handler(request), timeout=self.response_timeout * 10)
    except asyncio.TimeoutError:
      return web.json_response({"detail": "response timed out"}, status=408)

  def _on_token(self, request_id: str, tokens, is_finished: bool):
    q = self.token_queues.get(request_id)
    if q is not None:
      q.put_nowait((tokens, is_finished))

  async def handle_root(self, request):
    return web.FileResponse(Path(__file__).parent / "tinychat" / "index.html")

  async def handle_healthcheck(self, request):
    return web.json_response({"status": "ok"})

  async def handle_quit(self, request):
    response = web.json_response({"detail": "quitting"})
    await response.prepare(request)
    await response.write_eof()
    asyncio.get_running_loop().call_later(0.2, lambda: asyncio.ensure_future(
      shutdown("api-quit", asyncio.get_running_loop(), getattr(self.node, "server", None))))
    return response

  async def handle_get_models(self, request):
    models = [
      {"id": mid, "object": "model", "owned_by": "xotorch_amd", "ready": True, "name": pretty_name(mid)}
      for mid in model_cards
    ]
    return web.json_response({"object": "list", "data": models})

  async def handle_get_initial_models(self, request):
    out = {}
    local_repos = set()
    try:
      from xotorch_amd.download.downloader import models_dir
      md = models_dir()
      if md.exists():
        local_repos = {p.name for p in md.iterdir() if p.is_dir()}
    except Exception:
      pass
    for mid in get_supported_models():
      repo = get_repo(mid, self.inference_engine_classname) or ""
      folder = repo.replace("/", "--")
      out[mid] = {
        "name": pretty_name(mid),
        "downloaded": (folder in local_repos) if local_repos else None,
        "download_percentage": None,
        "total_size": None, "total_downloaded": None, "loading": False,
      }
    return web.json_response(out)

  async def handle_model_support(self, request):
    return web.json_response({"model pool": {mid: pretty_name(mid) for mid in get_supported_models()}})

  async def handle_get_topology(self, request):
    topo = getattr(self.node, "current_topology", None)
    return web.json_response(topo.to_json() if topo else {})

  async def handle_delete_model(self, request):
    model_name = request.match_info["model_name"]
    if model_name not in model_cards:
      return web.json_response({"detail": f"unknown model {model_name}"}, status=404)
    if self.shard_downloader is not None and hasattr(self.shard_downloader, "delete_model"):
      deleted = await self.shard_downloader.delete_model(model_name, self.inference_engine_classname)
      return web.json_response({"status": "success" if deleted else "not_found"})
    return web.json_response({"status": "no_downloader"})

  async def handle_get_download_progress(self, request):
    if self.shard_downloader is not None and hasattr(self.shard_downloader, "progress"):
      return web.json_response(self.shard_downloader.progress())
    return web.json_response({})

  async def handle_post_chat_token_encode(self, request):
    """Tokenize a chat-templated conversation without running it
    (reference chatgpt_api.py:210-211)."""
    data = await request.json()
    model_id = data.get("model") or self.default_model
    messages = [Message(m.get("role", "user"), m.get("content", "")) for m in data.get("messages", [])]
    tokenizer = getattr(self.node.inference_engine, "tokenizer", None)
    if tokenizer is None:
      repo = get_repo(model_id, self.inference_engine_classname)
      try:
        tokenizer = await resolve_tokenizer(repo)
      except Exception:
        from xotorch_amd.engine.tokenizers import DummyTokenizer
        tokenizer = DummyTokenizer()
    prompt = build_prompt(tokenizer, messages, data.get("tools"))
    tokens = tokenizer.encode(prompt)
    return web.json_response({"length": len(prompt), "num_tokens": len(tokens), "tokens": list(map(int, tokens))})

  async def handle_post_download(self, request):
    """Kick off a model download (reference chatgpt_api.py:221)."""
    data = await request.json()
    model_id = data.get("model")
    if not model_id or model_id not in model_cards:
      return web.json_response({"detail": f"unknown model {model_id}"}, status=400)
    shard = build_base_shard(model_id, self.inference_engine_classname)
    if self.shard_downloader is None or shard is None:
      return web.json_response({"detail": "no downloader available"}, status=503)
    asyncio.create_task(self.shard_downloader.ensure_shard(shard, self.inference_engine_classname))
    return web.json_response({"status": "started", "model": model_id})

  async def handle_post_chat_completions(self, request):
    data = await request.json()
    stream = bool(data.get("stream", False))
    model_id = data.get("model") or self.default_model
    if model_id not in model_cards:
      return web.json_response(
        {"detail": f"invalid model: {model_id}. supported: {list(model_cards)}"}, status=400)
    messages = [Message(m.get("role", "user"), m.get("content", "")) for m in data.get("messages", [])]
    chat_request = ChatCompletionRequest(model_id, messages, data.get("temperature", 0.0),
                                         data.get("tools"), data.get("max_tokens"), stream)
    shard = build_base_shard(model_id, self.inference_engine_classname)
    if shard is None:
      return web.json_response({"detail": f"no shard for model {model_id}"}, status=400)
    tokenizer = getattr(self.node.inference_engine, "tokenizer", None)
    if tokenizer is None or getattr(self.node.inference_engine, "shard", None) is None or \
       self.node.inference_engine.shard.model_id != model_id:
      repo = get_repo(model_id, self.inference_engine_classname)
      try:
        tokenizer = await resolve_tokenizer(repo)
      except Exception:
        from xotorch_amd.engine.tokenizers import DummyTokenizer
        tokenizer = DummyTokenizer()
    prompt = build_prompt(tokenizer, messages, chat_request.tools)
    if self.on_chat_completion_request is not None:
      try:
        self.on_chat_completion_request(str(uuid.uuid4()), chat_request, prompt)
      except Exception:
        pass
    request_id = str(uuid.uuid4())
    self.token_queues[request_id] = asyncio.Queue()
    try:
      state = {}
      if chat_request.max_tokens:
        state["max_tokens"] = int(chat_request.max_tokens)
      if chat_request.temperature:
        state["temperature"] = float(chat_request.temperature)
      if data.get("top_p") is not None:
        state["top_p"] = float(data["top_p"])
      try:
        n_prompt = len(tokenizer.encode(prompt))
      except Exception:
        n_prompt = None
      stops = data.get("stop")
      if isinstance(stops, str):
        stops = [stops]
      elif isinstance(stops, list):
        stops = [x for x in stops if isinstance(x, str) and x][:4]  # OpenAI caps at 4
      else:
        stops = None
      stops = stops or None
      try:
        await self.node.process_prompt(shard, prompt, request_id, state or None)
      except ValueError as e:
        # e.g. prompt longer than the serving context — a client error
        return web.json_response({"detail": str(e)}, status=400)
      if stream:
        include_usage = bool((data.get("stream_options") or {}).get("include_usage"))
        return await self._stream_response(request, request_id, model_id, tokenizer, stops,
                                           include_usage, n_prompt)
      return await self._full_response(request_id, model_id, tokenizer, stops, n_prompt)
    finally:
      self.token_queues.pop(request_id, None)
      self.prev_token_lens.pop(request_id, None)

  @staticmethod
  def _find_stop(text: str, stops) -> int:
    """Earliest index of any stop string in text, or -1."""
    cut = -1
    for st in stops or ():
      i = text.find(st)
      if i != -1 and (cut == -1 or i < cut):
        cut = i
    return cut

  async def _stream_response(self, request, request_id, model_id, tokenizer, stops=None,
                             include_usage=False, n_prompt=None):
    response = web.StreamResponse(status=200, headers={
      "Content-Type": "text/event-stream", "Cache-Control": "no-cache",
    })
    await response.prepare(request)
    all_tokens: List[int] = []
    finished = False
    # stop-sequence handling: hold back max(len)-1 chars so a stop string
    # split across token boundaries is still caught before being emitted
    text_all, sent = "", 0
    hold = max((len(st) for st in stops), default=1) - 1 if stops else 0
    try:
      while not finished:
        tokens, finished = await asyncio.wait_for(self.token_queues[request_id].get(), self.response_timeout)
        all_tokens.extend(tokens)
        text_all += tokenizer.decode(tokens) if tokens else ""
        cut = self._find_stop(text_all, stops)
        if cut != -1:
          delta, fin = text_all[sent:cut], "stop"
          finished = True
          cancel = getattr(self.node, "cancel_request", None)
          if cancel is not None:
            cancel(request_id)
        else:
          safe = len(text_all) if finished else max(sent, len(text_all) - hold)
          delta, fin = text_all[sent:safe], ("stop" if finished else None)
        sent += len(delta)
        chunk = generate_completion(request_id, tokens, delta, model_id,
                                    fin, True, "chat.completion.chunk")
        await response.write(f"data: {json.dumps(chunk)}\n\n".encode())
    except (asyncio.CancelledError, ConnectionResetError):
      # client went away mid-stream: free the slot/session instead of
      # generating to max_tokens for nobody
      cancel = getattr(self.node, "cancel_request", None)
      if cancel is not None:
        cancel(request_id)
      raise
    if include_usage:
      usage_chunk = generate_completion(request_id, [], "", model_id, None, True,
                                        "chat.completion.chunk")
      usage_chunk["choices"] = []
      usage_chunk["usage"] = {"completion_tokens": len(all_tokens)}
      if n_prompt is not None:
        usage_chunk["usage"]["prompt_tokens"] = n_prompt
        usage_chunk["usage"]["total_tokens"] = n_prompt + len(all_tokens)
      await response.write(f"data: {json.dumps(usage_chunk)}\n\n".encode())
    await response.write(b"data: [DONE]\n\n")
    await response.write_eof()
    return response

  async def _full_response(self, request_id, model_id, tokenizer, stops=None, n_prompt=None):
    all_tokens: List[int] = []
    finished = False
    text_all = ""
    while not finished:
      tokens, finished = await asyncio.wait_for(self.token_queues[request_id].get(), self.response_timeout)
      all_tokens.extend(tokens)
      if stops:
        text_all += tokenizer.decode(tokens) if tokens else ""
        cut = self._find_stop(text_all, stops)
        if cut != -1:
          cancel = getattr(self.node, "cancel_request", None)
          if cancel is not None:
            cancel(request_id)
          return web.json_response(
            generate_completion(request_id, all_tokens, text_all[:cut], model_id,
                                "stop", False, "chat.completion", n_prompt))
    decoded = tokenizer.decode(all_tokens) if all_tokens else ""
    return web.json_response(
      generate_completion(request_id, all_tokens, decoded, model_id, "stop", False,
                          "chat.completion", n_prompt))

  async def run(self, host: str = "0.0.0.0", port: int = 52415):
    self._runner = web.AppRunner(self.app)
    await self._runner.setup()
    site = web.TCPSite(self._runner, host, port)
    await site.start()

  async def stop(self):
    if self._runner is not None:
      await self._runner.cleanup()
