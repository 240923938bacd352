"""`xot` CLI: daemon / run / eval / train verbs.

Command parity with the reference's entrypoint
(/root/reference/xotorch/main.py:74-108,226-384): no-command starts the
daemon (discovery + node + ChatGPT API + optional TUI); `run` does one-shot
generation; `train`/`eval` drive the ring training path; `--resume-checkpoint`
actually loads (the reference parses but ignores it).
"""
from __future__ import annotations

import argparse
import asyncio
import json
import os
import signal
import sys
import time
import uuid
from pathlib import Path

from xotorch_amd.helpers import DEBUG, find_available_port, get_or_create_node_id
from xotorch_amd.models.registry import build_base_shard, get_supported_models


def build_parser() -> argparse.ArgumentParser:
  p = argparse.ArgumentParser("xot", description="MI355X-native distributed LLM runtime")
  p.add_argument("command", nargs="?", choices=["run", "eval", "train", "serve"], help="one-shot verb (default: daemon)")
  p.add_argument("model_name", nargs="?", help="model id (see --list-models)")
  p.add_argument("--list-models", action="store_true")
  p.add_argument("--prompt", type=str, default=None,
                 help="run: one-shot prompt; omit for the interactive chat TUI")
  p.add_argument("--max-generate-tokens", type=int, default=1024)
  p.add_argument("--default-temp", type=float, default=0.0)
  p.add_argument("--inference-engine", type=str, default=None, choices=[None, "torch", "hip", "dummy"])
  p.add_argument("--node-id", type=str, default=None)
  p.add_argument("--node-host", type=str, default="0.0.0.0")
  p.add_argument("--node-port", type=int, default=None)
  p.add_argument("--listen-port", type=int, default=5678)
  p.add_argument("--broadcast-port", type=int, default=5678)
  p.add_argument("--discovery-module", type=str, choices=["udp", "manual", "none"], default="udp")
  p.add_argument("--discovery-config", type=str, default=None)
  p.add_argument("--discovery-timeout", type=int, default=30)
  p.add_argument("--wait-for-peers", type=int, default=0)
  p.add_argument("--chatgpt-api-port", type=int, default=52415)
  p.add_argument("--chatgpt-api-response-timeout", type=int, default=120)
  p.add_argument("--disable-tui", action="store_true")
  p.add_argument("--gpus", type=int, default=1, help="serve: GPUs for the single-node RCCL ring")
  p.add_argument("--slots", type=int, default=8, help="serve: concurrent KV slots (continuous batching)")
  p.add_argument("--no-graphs", action="store_true", help="serve: disable hipGraph decode capture")
  # train/eval
  p.add_argument("--data", type=str, default=None)
  p.add_argument("--epochs", type=int, default=1)
  p.add_argument("--batch-size", type=int, default=1)
  p.add_argument("--save-every", type=int, default=5)
  p.add_argument("--save-checkpoint-dir", type=str, default="checkpoints")
  p.add_argument("--resume-checkpoint", type=str, default=None)
  return p


def pick_engine(args) -> str:
  if args.inference_engine:
    return args.inference_engine
  import torch
  return "hip" if torch.cuda.is_available() else "torch"


async def make_node(args):
  from xotorch_amd.engine.interface import get_inference_engine
  from xotorch_amd.download.downloader import new_shard_downloader
  from xotorch_amd.orchestration.discovery import ManualDiscovery, UDPDiscovery
  from xotorch_amd.orchestration.node import Node
  from xotorch_amd.orchestration.server import Server

  node_id = args.node_id or get_or_create_node_id()
  node_port = args.node_port or find_available_port()
  downloader = new_shard_downloader("noop" if os.getenv("XOT_OFFLINE") == "1" else "hf")
  engine = get_inference_engine(pick_engine(args), downloader)
  if args.discovery_module == "udp":
    discovery = UDPDiscovery(node_id, node_port, args.listen_port, args.broadcast_port,
                             discovery_timeout=args.discovery_timeout)
  elif args.discovery_module == "manual":
    if not args.discovery_config:
      raise SystemExit("--discovery-module manual requires --discovery-config")
    discovery = ManualDiscovery(args.discovery_config, node_id)
  else:
    discovery = None
  node = Node(node_id, None, engine, discovery,
              max_generate_tokens=args.max_generate_tokens,
              default_sample_temperature=args.default_temp)
  node.server = Server(node, args.node_host, node_port)
  return node


async def run_model_cli(args):
  node = await make_node(args)
  await node.start(args.wait_for_peers)
  engine_classname = {"torch": "TorchEngine", "hip": "HIPEngine", "dummy": "DummyEngine"}[pick_engine(args)]
  shard = build_base_shard(args.model_name, engine_classname)
  if shard is None:
    raise SystemExit(f"unknown model {args.model_name}")
  if args.prompt is None:
    # interactive chat REPL (reference main.py:381 -> viz/chat_tui.py:11)
    from xotorch_amd.viz.chat_tui import run_chat_tui
    try:
      await run_chat_tui(node, args.model_name, engine_classname)
    finally:
      await node.stop()
    return
  from xotorch_amd.engine.tokenizers import resolve_tokenizer
  from xotorch_amd.models.registry import get_repo
  tokenizer = await resolve_tokenizer(get_repo(args.model_name, engine_classname))
  try:
    prompt = tokenizer.apply_chat_template(
      conversation=[{"role": "user", "content": args.prompt}], tokenize=False, add_generation_prompt=True)
  except Exception:
    prompt = args.prompt
  request_id = str(uuid.uuid4())
  done = asyncio.Event()
  tokens = []
  t_first = [None]

  def on_token(rid, toks, is_finished):
    if rid != request_id:
      return
    if t_first[0] is None:
      t_first[0] = time.perf_counter()
    tokens.extend(toks)
    if is_finished:
      done.set()

  node.on_token.register("cli").on_next(on_token)
  t0 = time.perf_counter()
  await node.process_prompt(shard, prompt, request_id)
  await asyncio.wait_for(done.wait(), timeout=300)
  dt = time.perf_counter() - (t_first[0] or t0)
  print(tokenizer.decode(tokens))
  print(f"\nFinal stats: {len(tokens)} tokens | {len(tokens)/max(dt,1e-9):.1f} tokens/sec | "
        f"TTFT {1000*((t_first[0] or t0)-t0):.0f} ms", file=sys.stderr)
  await node.stop()


async def train_model_cli(args, train: bool = True):
  import numpy as np
  node = await make_node(args)
  await node.start(args.wait_for_peers)
  engine_classname = {"torch": "TorchEngine", "hip": "HIPEngine", "dummy": "DummyEngine"}[pick_engine(args)]
  shard = build_base_shard(args.model_name, engine_classname)
  if shard is None:
    raise SystemExit(f"unknown model {args.model_name}")
  from xotorch_amd.engine.tokenizers import resolve_tokenizer
  from xotorch_amd.models.registry import get_repo
  from xotorch_amd.train.dataset import iterate_batches, load_dataset
  tokenizer = await resolve_tokenizer(get_repo(args.model_name, engine_classname))
  data_dir = args.data or str(Path(__file__).parent / "train" / "data" / "lora")
  train_set, valid_set, test_set = load_dataset(data_dir, lambda t: tokenizer.encode(t))
  if args.resume_checkpoint:
    await node.inference_engine.load_checkpoint(node.get_current_shard(shard), args.resume_checkpoint)
  if train:
    for epoch in range(args.epochs):
      losses = []
      for inputs, targets, lengths in iterate_batches(train_set, args.batch_size, seed=epoch):
        loss, _ = await node.enqueue_example(shard, inputs, targets, lengths, train=True)
        losses.append(loss)
        print(f"epoch {epoch} loss {loss:.4f}")
      if (epoch + 1) % args.save_every == 0:
        await node.coordinate_save(shard, epoch + 1, args.save_checkpoint_dir)
      print(f"epoch {epoch} mean loss {np.mean(losses):.4f}")
  else:
    losses = []
    for inputs, targets, lengths in iterate_batches(test_set or valid_set or train_set, args.batch_size, shuffle=False):
      loss, _ = await node.enqueue_example(shard, inputs, targets, lengths, train=False)
      losses.append(loss)
    print(f"eval loss: {np.mean(losses):.4f}" if losses else "no eval data")
  await node.stop()


async def daemon(args):
  from xotorch_amd.api.chatgpt import ChatGPTAPI
  from xotorch_amd.viz.topology_viz import TopologyViz
  node = await make_node(args)
  viz = None
  if not args.disable_tui and sys.stdout.isatty():
    viz = TopologyViz()
    node.topology_viz = viz
    viz.start()
  await node.start(args.wait_for_peers)
  engine_classname = type(node.inference_engine).__name__
  api = ChatGPTAPI(node, engine_classname, response_timeout=args.chatgpt_api_response_timeout,
                   shard_downloader=getattr(node.inference_engine, "shard_downloader", None))
  await api.run(port=args.chatgpt_api_port)
  print(f"ChatGPT API on http://localhost:{args.chatgpt_api_port}/v1/chat/completions", file=sys.stderr)
  stop = asyncio.Event()
  loop = asyncio.get_running_loop()
  for sig in (signal.SIGINT, signal.SIGTERM):
    try:
      loop.add_signal_handler(sig, stop.set)
    except NotImplementedError:
      pass
  try:
    await stop.wait()
  except asyncio.CancelledError:
    pass  # /quit cancels the main task; shut down cleanly below
  finally:
    if viz:
      viz.stop()
    await api.stop()
    await node.stop()


def serve_ring(args):
  """Single-node multi-GPU serving: torchrun-spawn the RCCL ring workers."""
  import subprocess
  script = Path(__file__).parent / "serve_ring.py"
  cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         f"--nproc-per-node={args.gpus}", "--master-addr", "127.0.0.1",
         "--master-port", str(find_available_port("127.0.0.1")), str(script),
         "--model", args.model_name or "llama-3-8b",
         "--port", str(args.chatgpt_api_port),
         "--slots", str(args.slots)]
  if args.no_graphs:
    cmd.append("--no-graphs")
  raise SystemExit(subprocess.run(cmd).returncode)


def run():
  args = build_parser().parse_args()
  if args.list_models:
    for mid in get_supported_models():
      print(mid)
    return
  try:
    import uvloop
    uvloop.install()
  except ImportError:
    pass
  if args.command == "run":
    asyncio.run(run_model_cli(args))
  elif args.command == "train":
    asyncio.run(train_model_cli(args, train=True))
  elif args.command == "eval":
    asyncio.run(train_model_cli(args, train=False))
  elif args.command == "serve":
    serve_ring(args)
  else:
    try:
      asyncio.run(daemon(args))
    except (KeyboardInterrupt, asyncio.CancelledError):
      pass


if __name__ == "__main__":
  run()
