"""Shared utilities: debug flags, async pub-sub callbacks, ports, node ids.

Capability parity with the reference's helpers
(/root/reference/xotorch/helpers.py:19-20,104-149,47-76,182-205) with a
simpler surface: the AsyncCallbackSystem is the backbone every layer above
uses for token streams, status gossip and download progress.
"""
from __future__ import annotations

import asyncio
import os
import random
import socket
import uuid
from pathlib import Path
from typing import Any, Callable, Dict, Generic, List, Optional, Tuple, TypeVar

DEBUG = int(os.getenv("DEBUG", "0"))
DEBUG_DISCOVERY = int(os.getenv("DEBUG_DISCOVERY", "0"))

XOT_HOME = Path(os.getenv("XOT_HOME", Path.home() / ".xotorch_amd"))

T = TypeVar("T", bound=Tuple)


class AsyncCallback(Generic[T]):
  """One awaitable callback slot: wait(condition) / on_next(fn) / set(args)."""

  def __init__(self):
    self.condition: asyncio.Condition = asyncio.Condition()
    self.result: Optional[T] = None
    self.observers: List[Callable[..., None]] = []

  async def wait(self, check_condition: Callable[..., bool], timeout: Optional[float] = None) -> T:
    async with self.condition:
      await asyncio.wait_for(
        self.condition.wait_for(lambda: self.result is not None and check_condition(*self.result)),
        timeout,
      )
      assert self.result is not None
      return self.result

  def on_next(self, callback: Callable[..., None]) -> None:
    self.observers.append(callback)

  def set(self, *args: Any) -> None:
    self.result = args  # type: ignore[assignment]
    for observer in self.observers:
      observer(*args)
    loop = None
    try:
      loop = asyncio.get_running_loop()
    except RuntimeError:
      pass
    if loop is not None:
      loop.create_task(self._notify())
    else:
      asyncio.run(self._notify())

  async def _notify(self) -> None:
    async with self.condition:
      self.condition.notify_all()


K = TypeVar("K")


class AsyncCallbackSystem(Generic[K, T]):
  """Keyed registry of AsyncCallbacks with trigger_all fan-out."""

  def __init__(self):
    self.callbacks: Dict[K, AsyncCallback[T]] = {}

  def register(self, name: K) -> AsyncCallback[T]:
    if name not in self.callbacks:
      self.callbacks[name] = AsyncCallback[T]()
    return self.callbacks[name]

  def deregister(self, name: K) -> None:
    self.callbacks.pop(name, None)

  def trigger(self, name: K, *args: Any) -> None:
    if name in self.callbacks:
      self.callbacks[name].set(*args)

  def trigger_all(self, *args: Any) -> None:
    for cb in list(self.callbacks.values()):
      cb.set(*args)


class PrefixDict(Generic[K, T]):
  """Dict whose items can be looked up by key prefix (used by the API router)."""

  def __init__(self):
    self._d: Dict[str, T] = {}

  def __setitem__(self, key: str, value: T):
    self._d[key] = value

  def items(self):
    return self._d.items()

  def find_prefix(self, argument: str) -> List[Tuple[str, T]]:
    return [(k, v) for k, v in self._d.items() if argument.startswith(k)]

  def find_longest_prefix(self, argument: str) -> Optional[Tuple[str, T]]:
    matches = self.find_prefix(argument)
    if not matches:
      return None
    return max(matches, key=lambda x: len(x[0]))


_handed_out_ports: set = set()


def find_available_port(host: str = "", min_port: int = 49152, max_port: int = 65535) -> int:
  for _ in range(200):
    port = random.randint(min_port, max_port)
    if port in _handed_out_ports:
      continue
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
      try:
        s.bind((host, port))
        _handed_out_ports.add(port)
        return port
      except OSError:
        continue
  raise RuntimeError("no available ports")


def get_or_create_node_id() -> str:
  """Persist a stable node id under XOT_HOME (override with XOT_UUID)."""
  if os.getenv("XOT_UUID"):
    return os.environ["XOT_UUID"]
  try:
    XOT_HOME.mkdir(parents=True, exist_ok=True)
    id_file = XOT_HOME / "node_id"
    if id_file.exists():
      node_id = id_file.read_text().strip()
      if node_id:
        return node_id
    node_id = str(uuid.uuid4())
    id_file.write_text(node_id)
    return node_id
  except Exception:
    return str(uuid.uuid4())


def pretty_print_bytes(size_in_bytes: int) -> str:
  for unit, div in (("TB", 1024**4), ("GB", 1024**3), ("MB", 1024**2), ("KB", 1024)):
    if size_in_bytes >= div:
      return f"{size_in_bytes / div:.2f} {unit}"
  return f"{size_in_bytes} B"


def pretty_print_bytes_per_second(bps: float) -> str:
  return pretty_print_bytes(int(bps)) + "/s"


async def shutdown(signal_name, loop, server=None):
  """Graceful shutdown: cancel outstanding tasks, stop the loop."""
  if DEBUG >= 1:
    print(f"received exit signal {signal_name}...")
  if server is not None:
    try:
      await server.stop()
    except Exception:
      pass
  tasks = [t for t in asyncio.all_tasks(loop) if t is not asyncio.current_task()]
  for task in tasks:
    task.cancel()
  await asyncio.gather(*tasks, return_exceptions=True)
  loop.stop()


def interface_priority(ifname: str) -> int:
  """Rank a NIC for peer connections (reference helpers.py:284-315):
  Thunderbolt/USB4 > Ethernet > WiFi > cellular > virtual/loopback."""
  n = (ifname or "").lower()
  if n.startswith(("tb", "thunderbolt", "usb4")):
    return 5
  if n.startswith(("en", "eth", "enp", "eno", "ens")):
    return 4
  if n.startswith(("wl", "wifi", "wlan")):
    return 3
  if n.startswith(("ww", "cell")):
    return 2
  if n.startswith(("lo", "docker", "veth", "br-", "virbr", "tun", "tap", "utun")):
    return 1
  return 2
