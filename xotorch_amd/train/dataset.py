"""jsonl train/valid/test loading + padded batching.

Parity with /root/reference/xotorch/train/dataset.py:9-79: each jsonl line is
{"text": ...} (or a prompt/completion pair rendered by the caller); batches
are padded to the longest sequence and returned as (inputs, targets, lengths)
with targets = inputs shifted left by one.
"""
from __future__ import annotations

import json
import random
from pathlib import Path
from typing import Callable, Iterator, List, Tuple

import numpy as np

MAX_WARN_LEN = 2048


def batch_with_lengths(batch: List[List[int]], pad_id: int = 0):
  B = len(batch)
  lengths = [len(x) for x in batch]
  maxlen = max(lengths)
  if maxlen > MAX_WARN_LEN:
    print(f"[train] warning: sequence of {maxlen} tokens exceeds {MAX_WARN_LEN}; consider pre-splitting")
  inputs = np.full((B, maxlen), pad_id, dtype=np.int64)
  for i, x in enumerate(batch):
    inputs[i, :len(x)] = x
  targets = np.roll(inputs, -1, axis=1)
  targets[:, -1] = pad_id
  return inputs, targets, np.asarray(lengths, dtype=np.int64)


def iterate_batches(dataset: List[List[int]], batch_size: int, shuffle: bool = True,
                    seed: int = 0) -> Iterator[Tuple[np.ndarray, np.ndarray, np.ndarray]]:
  order = list(range(len(dataset)))
  if shuffle:
    random.Random(seed).shuffle(order)
  for i in range(0, len(order) - batch_size + 1, batch_size):
    batch = [dataset[j] for j in order[i:i + batch_size]]
    yield batch_with_lengths(batch)


def load_dataset(data_dir: str, encode: Callable[[str], List[int]]):
  """Load train/valid/test jsonl files from a directory; returns 3 token lists."""
  d = Path(data_dir)

  def load_split(name: str) -> List[List[int]]:
    path = d / f"{name}.jsonl"
    if not path.exists():
      return []
    out = []
    with open(path) as f:
      for line in f:
        line = line.strip()
        if not line:
          continue
        rec = json.loads(line)
        text = rec.get("text")
        if text is None and "prompt" in rec:
          text = rec["prompt"] + rec.get("completion", "")
        if text:
          out.append(list(encode(text)))
    return out

  return load_split("train"), load_split("valid"), load_split("test")
