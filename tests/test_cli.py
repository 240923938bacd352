"""CLI surface: parser semantics + the one-shot `xot run` path end to end
with the dummy engine (no network, no GPU)."""
import asyncio

import pytest

from xotorch_amd.cli import build_parser, pick_engine, run_model_cli


def test_parser_verbs_and_defaults():
  p = build_parser()
  a = p.parse_args([])
  assert a.command is None and a.default_temp == 0.0 and a.chatgpt_api_port == 52415
  a = p.parse_args(["run", "llama-3.2-1b", "--prompt", "hi", "--max-generate-tokens", "7"])
  assert (a.command, a.model_name, a.prompt, a.max_generate_tokens) == ("run", "llama-3.2-1b", "hi", 7)
  a = p.parse_args(["serve", "llama-3-70b", "--gpus", "8", "--slots", "16", "--no-graphs"])
  assert (a.command, a.gpus, a.slots, a.no_graphs) == ("serve", 8, 16, True)
  a = p.parse_args(["train", "dummy", "--data", "d.jsonl", "--epochs", "2"])
  assert (a.command, a.data, a.epochs) == ("train", "d.jsonl", 2)


def test_pick_engine_explicit_and_default():
  p = build_parser()
  assert pick_engine(p.parse_args(["run", "dummy", "--inference-engine", "dummy"])) == "dummy"
  # default resolution never raises and returns a known engine name
  assert pick_engine(p.parse_args(["run", "dummy"])) in ("torch", "hip", "dummy")


def test_run_one_shot_dummy(capsys):
  """`xot run dummy --prompt ...` produces tokens and exits cleanly."""
  p = build_parser()
  args = p.parse_args(["run", "dummy", "--prompt", "hello world",
                       "--inference-engine", "dummy", "--discovery-module", "none",
                       "--max-generate-tokens", "5", "--disable-tui"])
  asyncio.new_event_loop().run_until_complete(run_model_cli(args))
  out = capsys.readouterr().out
  assert out.strip(), "no generated text printed"


def test_tools_and_entry_scripts_parse():
  """Every tools/ script, bench.py and __graft_entry__.py must at least
  parse — they run remotely on GPU boxes where a syntax error is costly."""
  import ast
  from pathlib import Path
  root = Path(__file__).resolve().parent.parent
  files = sorted((root / "tools").glob("*.py")) + [root / "bench.py", root / "__graft_entry__.py"]
  assert len(files) > 8
  for f in files:
    ast.parse(f.read_text(), filename=str(f))


def test_train_and_eval_cli_end_to_end(tmp_path, capsys):
  """`xot train`/`xot eval` through the real torch engine (tiny builtin
  "dummy" llama on CPU) over a tiny jsonl set: losses print, a checkpoint
  lands in --save-checkpoint-dir."""
  import json
  for split in ("train", "valid", "test"):
    with open(tmp_path / f"{split}.jsonl", "w") as f:
      for i in range(2):
        f.write(json.dumps({"text": f"{split} example {i} lorem ipsum"}) + "\n")
  ckpt = tmp_path / "ckpts"
  p = build_parser()
  args = p.parse_args(["train", "dummy", "--inference-engine", "torch",
                       "--discovery-module", "none", "--data", str(tmp_path),
                       "--epochs", "1", "--save-every", "1",
                       "--save-checkpoint-dir", str(ckpt), "--disable-tui"])
  from xotorch_amd.cli import train_model_cli
  asyncio.new_event_loop().run_until_complete(train_model_cli(args, train=True))
  out = capsys.readouterr().out
  assert "mean loss" in out
  assert any(ckpt.rglob("*")), "no checkpoint written"
  args = p.parse_args(["eval", "dummy", "--inference-engine", "torch",
                       "--discovery-module", "none", "--data", str(tmp_path),
                       "--disable-tui"])
  asyncio.new_event_loop().run_until_complete(train_model_cli(args, train=False))
  assert "eval loss" in capsys.readouterr().out
  # resume from the saved checkpoint and keep training
  saved = next(ckpt.rglob("*.safetensors"))
  args = p.parse_args(["train", "dummy", "--inference-engine", "torch",
                       "--discovery-module", "none", "--data", str(tmp_path),
                       "--epochs", "1", "--save-every", "5",
                       "--resume-checkpoint", str(saved), "--disable-tui"])
  asyncio.new_event_loop().run_until_complete(train_model_cli(args, train=True))
  assert "mean loss" in capsys.readouterr().out


def test_daemon_entry_http_smoke():
  """The bare `xot` daemon entry point in a subprocess: API comes up,
  healthcheck + models respond, clean shutdown."""
  import json as _json
  import os
  import subprocess
  import sys
  import time as _t
  import urllib.request
  from xotorch_amd.helpers import find_available_port

  port = find_available_port("127.0.0.1")
  code = ("import sys; sys.argv = ['xot', '--disable-tui', '--discovery-module', 'none', "
          f"'--inference-engine', 'dummy', '--chatgpt-api-port', '{port}']; "
          "from xotorch_amd.cli import run; run()")
  env = dict(os.environ, XOT_OFFLINE="1")
  proc = subprocess.Popen([sys.executable, "-c", code],
                          stdout=subprocess.PIPE, stderr=subprocess.STDOUT, env=env)
  try:
    deadline = _t.time() + 60
    up = False
    while _t.time() < deadline:
      try:
        urllib.request.urlopen(f"http://127.0.0.1:{port}/healthcheck", timeout=2)
        up = True
        break
      except Exception:
        if proc.poll() is not None:
          out = proc.stdout.read().decode(errors="replace")
          raise AssertionError(f"daemon died early:\n{out[-2000:]}")
        _t.sleep(0.3)
    assert up, "daemon API never came up"
    with urllib.request.urlopen(f"http://127.0.0.1:{port}/v1/models", timeout=10) as r:
      data = _json.loads(r.read())
    assert any(m["id"] == "llama-3-70b" for m in data["data"])
  finally:
    proc.terminate()
    try:
      proc.wait(timeout=15)
    except subprocess.TimeoutExpired:
      proc.kill()
