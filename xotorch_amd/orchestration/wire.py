"""Wire protocol for the host control/data plane between Nodes.

Replaces the reference's gRPC/protobuf layer
(/root/reference/xotorch/networking/grpc/node_service.proto:5-117) with a
dependency-free asyncio TCP protocol: 4-byte length-prefixed msgpack frames,
tensors as raw bytes + shape + dtype. Same RPC surface: SendPrompt,
SendTensor, SendExample, CollectTopology, SendResult, SendOpaqueStatus,
HealthCheck. On a single 8-GPU node, bulk activations move over RCCL/xGMI
(parallel/ring.py) — this plane carries control, gossip and multi-host hops.
"""
from __future__ import annotations

import asyncio
import struct
from typing import Any, Dict, Optional, Tuple

import msgpack
import numpy as np

MAX_FRAME = 256 * 1024 * 1024  # parity with reference's 256 MB message cap


def pack_tensor(arr: Optional[np.ndarray]) -> Optional[dict]:
  if arr is None:
    return None
  arr = np.asarray(arr)
  shape = list(arr.shape)  # before ascontiguousarray: it promotes 0-d to 1-d
  arr = np.ascontiguousarray(arr)
  return {"shape": shape, "dtype": str(arr.dtype), "data": arr.tobytes()}


def unpack_tensor(d: Optional[dict]) -> Optional[np.ndarray]:
  if d is None:
    return None
  return np.frombuffer(d["data"], dtype=np.dtype(d["dtype"])).reshape(d["shape"]).copy()


async def read_frame(reader: asyncio.StreamReader) -> Dict[str, Any]:
  hdr = await reader.readexactly(4)
  (n,) = struct.unpack("!I", hdr)
  if n > MAX_FRAME:
    raise ValueError(f"frame too large: {n}")
  payload = await reader.readexactly(n)
  return msgpack.unpackb(payload, raw=False)


def write_frame(writer: asyncio.StreamWriter, msg: Dict[str, Any]) -> None:
  payload = msgpack.packb(msg, use_bin_type=True)
  if len(payload) > MAX_FRAME:
    # fail on the SEND side with a clear error; otherwise the receiver
    # rejects the oversized frame and kills the connection with no clue
    # at the sender (e.g. a >256 MB activation hop at large batch)
    raise ValueError(f"frame too large to send: {len(payload)} > {MAX_FRAME} bytes")
  writer.write(struct.pack("!I", len(payload)) + payload)


async def request(host: str, port: int, msg: Dict[str, Any], timeout: float = 30.0) -> Dict[str, Any]:
  """One-shot RPC: connect, send, await reply, close."""
  reader, writer = await asyncio.wait_for(asyncio.open_connection(host, port), timeout)
  try:
    write_frame(writer, msg)
    await writer.drain()
    return await asyncio.wait_for(read_frame(reader), timeout)
  finally:
    writer.close()
    try:
      await writer.wait_closed()
    except Exception:
      pass
