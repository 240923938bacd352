"""PeerHandle: per-peer RPC facade (reference parity:
/root/reference/xotorch/networking/peer_handle.py:9-56 and
grpc/grpc_peer_handle.py:19-224, re-based on the msgpack/TCP wire)."""
from __future__ import annotations

import asyncio
from abc import ABC, abstractmethod
from typing import Optional, Tuple

import numpy as np

from xotorch_amd.orchestration import wire
from xotorch_amd.parallel.topology import DeviceCapabilities, Topology
from xotorch_amd.shard import Shard


class PeerHandle(ABC):
  @abstractmethod
  def id(self) -> str: ...

  @abstractmethod
  def addr(self) -> str: ...

  @abstractmethod
  def description(self) -> str: ...

  @abstractmethod
  def device_capabilities(self) -> DeviceCapabilities: ...

  @abstractmethod
  async def connect(self) -> None: ...

  @abstractmethod
  async def is_connected(self) -> bool: ...

  @abstractmethod
  async def disconnect(self) -> None: ...

  @abstractmethod
  async def health_check(self) -> bool: ...

  @abstractmethod
  async def send_prompt(self, shard: Shard, prompt: str, request_id: str,
                        inference_state: Optional[dict] = None) -> None: ...

  @abstractmethod
  async def send_tensor(self, shard: Shard, tensor: np.ndarray, request_id: str,
                        inference_state: Optional[dict] = None) -> None: ...

  @abstractmethod
  async def send_example(self, shard: Shard, example: np.ndarray, target: np.ndarray,
                         length: np.ndarray, request_id: str, train: bool = False
                         ) -> Tuple[float, Optional[np.ndarray]]: ...

  @abstractmethod
  async def send_result(self, request_id: str, result, is_finished: bool) -> None: ...

  @abstractmethod
  async def send_opaque_status(self, request_id: str, status: str) -> None: ...

  @abstractmethod
  async def collect_topology(self, visited: set, max_depth: int) -> Topology: ...


class TCPPeerHandle(PeerHandle):
  def __init__(self, peer_id: str, address: str, description: str = "",
               capabilities: Optional[DeviceCapabilities] = None):
    self._id = peer_id
    self._addr = address
    host, port = address.rsplit(":", 1)
    self.host, self.port = host, int(port)
    self._description = description
    self._caps = capabilities or DeviceCapabilities(model="unknown", chip="unknown", memory=0)
    self._connected = False

  def id(self) -> str:
    return self._id

  def addr(self) -> str:
    return self._addr

  def description(self) -> str:
    return self._description

  def device_capabilities(self) -> DeviceCapabilities:
    return self._caps

  async def connect(self) -> None:
    self._connected = await self.health_check()
    if not self._connected:
      raise ConnectionError(f"peer {self._id} at {self._addr} not healthy")

  async def is_connected(self) -> bool:
    return self._connected

  async def disconnect(self) -> None:
    self._connected = False

  async def _rpc(self, msg: dict, timeout: float = 60.0) -> dict:
    return await wire.request(self.host, self.port, msg, timeout)

  async def health_check(self) -> bool:
    try:
      reply = await self._rpc({"type": "health"}, timeout=5.0)
      return reply.get("ok", False)
    except Exception:
      return False

  async def send_prompt(self, shard, prompt, request_id, inference_state=None) -> None:
    await self._rpc({
      "type": "prompt", "shard": shard.to_dict(), "prompt": prompt,
      "request_id": request_id, "inference_state": inference_state,
    })

  async def send_tensor(self, shard, tensor, request_id, inference_state=None) -> None:
    await self._rpc({
      "type": "tensor", "shard": shard.to_dict(), "tensor": wire.pack_tensor(tensor),
      "request_id": request_id, "inference_state": inference_state,
    })

  async def send_example(self, shard, example, target, length, request_id, train=False):
    reply = await self._rpc({
      "type": "example", "shard": shard.to_dict(), "example": wire.pack_tensor(example),
      "target": wire.pack_tensor(target), "length": wire.pack_tensor(length),
      "request_id": request_id, "train": train,
    }, timeout=600.0)
    return reply.get("loss", 0.0), wire.unpack_tensor(reply.get("grads"))

  async def send_result(self, request_id, result, is_finished) -> None:
    msg = {"type": "result", "request_id": request_id, "is_finished": is_finished}
    if isinstance(result, np.ndarray):
      msg["tensor_result"] = wire.pack_tensor(result)
    else:
      msg["result"] = list(result)
    await self._rpc(msg)

  async def send_opaque_status(self, request_id, status) -> None:
    await self._rpc({"type": "status", "request_id": request_id, "status": status})

  async def collect_topology(self, visited: set, max_depth: int) -> Topology:
    reply = await self._rpc({"type": "topology", "visited": list(visited), "max_depth": max_depth})
    return Topology.from_json(reply.get("topology", {}))
