"""Terminal chat REPL with per-request token-speed stats.

Parity with /root/reference/xotorch/viz/chat_tui.py:11-165: prompt loop →
node.process_prompt, stream tokens from the on_token callback, print
`Final stats: {n} tokens | {tok/s} | {TFLOPS}` per request.
"""
from __future__ import annotations

import asyncio
import time
import uuid

from xotorch_amd.models.registry import build_base_shard
from xotorch_amd.engine.tokenizers import resolve_tokenizer


async def run_chat_tui(node, model_id: str, engine_classname: str = "TorchEngine",
                       max_tokens: int = 512, input_fn=input, print_fn=print):
  from xotorch_amd.models.registry import get_repo
  shard = build_base_shard(model_id, engine_classname)
  if shard is None:
    raise ValueError(f"unknown model {model_id}")
  tokenizer = await resolve_tokenizer(get_repo(model_id, engine_classname))
  total_flops = sum(c.flops.fp16 for _, c in node.topology.all_nodes()) or 1.0
  print_fn(f"chat with {model_id} — /quit to exit")
  while True:
    try:
      prompt_text = await asyncio.get_running_loop().run_in_executor(None, input_fn, "you> ")
    except (EOFError, KeyboardInterrupt):
      break
    if prompt_text.strip() in ("/quit", "/exit", "q"):
      break
    if not prompt_text.strip():
      continue
    try:
      prompt = tokenizer.apply_chat_template(
        conversation=[{"role": "user", "content": prompt_text}], tokenize=False, add_generation_prompt=True)
    except Exception:
      prompt = prompt_text
    request_id = str(uuid.uuid4())
    done = asyncio.Event()
    state = {"tokens": [], "t0": None}

    def on_token(rid, tokens, is_finished):
      if rid != request_id:
        return
      if state["t0"] is None:
        state["t0"] = time.perf_counter()
      state["tokens"].extend(tokens)
      try:
        print_fn(tokenizer.decode(tokens), end="", flush=True)
      except TypeError:
        print_fn(tokenizer.decode(tokens))
      if is_finished:
        done.set()

    cb = node.on_token.register(f"tui-{request_id}")
    cb.on_next(on_token)
    t_start = time.perf_counter()
    await node.process_prompt(shard, prompt, request_id)
    try:
      await asyncio.wait_for(done.wait(), timeout=600)
    except asyncio.TimeoutError:
      print_fn("\n[timed out]")
    node.on_token.deregister(f"tui-{request_id}")
    n = len(state["tokens"])
    elapsed = time.perf_counter() - (state["t0"] or t_start)
    tps = n / elapsed if elapsed > 0 else 0.0
    print_fn(f"\nFinal stats: {n} tokens | {tps:.1f} tokens/sec | {total_flops:.0f} TFLOPS cluster")
