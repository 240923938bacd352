"""Tokenizer resolution with graceful fallbacks.

Parity with /root/reference/xotorch/inference/tokenizers.py:11-63: resolve
from a local download dir, else the HF repo id; AutoProcessor→AutoTokenizer
fallback chain; a deterministic DummyTokenizer for plumbing tests.
"""
from __future__ import annotations

from pathlib import Path
from typing import List, Optional, Union


class DummyTokenizer:
  """Deterministic fake tokenizer (eos=69) for engine-plumbing tests."""

  def __init__(self):
    self.eos_token_id = 69
    self.vocab_size = 256

  def apply_chat_template(self, conversation=None, tokenize=True, add_generation_prompt=True, tools=None, **kwargs):
    messages = conversation or []
    text = " ".join(m.get("content", "") if isinstance(m, dict) else str(m) for m in messages)
    if tokenize:
      return self.encode(text)
    return text

  def encode(self, text: str, **kwargs) -> List[int]:
    return [(ord(c) % 200) + 1 for c in (text or "dummy")][:64] or [1]

  def decode(self, tokens, **kwargs) -> str:
    return "dummy " * max(1, len(tokens))


async def resolve_tokenizer(model_path_or_repo: Union[str, Path, None], timeout: float = 20.0):
  """Resolve a tokenizer without ever wedging the caller: local paths load
  directly; hub repos are resolved in a worker thread with a timeout (an
  air-gapped box can block for minutes inside DNS/TCP otherwise) and fall
  back to the deterministic DummyTokenizer."""
  import asyncio
  import os
  if model_path_or_repo in (None, "dummy"):
    return DummyTokenizer()
  p = Path(str(model_path_or_repo))
  local = p.exists()
  if not local:
    os.environ.setdefault("HF_HUB_ETAG_TIMEOUT", "5")
    os.environ.setdefault("HF_HUB_DOWNLOAD_TIMEOUT", "10")
  try:
    return await asyncio.wait_for(
      asyncio.get_running_loop().run_in_executor(None, _resolve_tokenizer, model_path_or_repo),
      timeout=None if local else timeout,
    )
  except Exception:  # timeout, no network, missing repo, bad config — all
    return DummyTokenizer()  # degrade to the deterministic fallback


def _resolve_tokenizer(model_path_or_repo: Union[str, Path]):
  from transformers import AutoTokenizer
  try:
    try:
      from transformers import AutoProcessor
      proc = AutoProcessor.from_pretrained(str(model_path_or_repo), trust_remote_code=True)
      tok = getattr(proc, "tokenizer", None) or proc
      _patch(tok)
      return tok
    except Exception:
      pass
    tok = AutoTokenizer.from_pretrained(str(model_path_or_repo), trust_remote_code=True)
    _patch(tok)
    return tok
  except Exception as e:
    raise RuntimeError(f"could not resolve tokenizer for {model_path_or_repo}: {e}")


def _patch(tok):
  """Fill in eos/encode/decode when a processor lacks them."""
  if not hasattr(tok, "eos_token_id") or tok.eos_token_id is None:
    tok.eos_token_id = getattr(tok, "vocab_size", 2) - 1
  return tok
