"""TCP server exposing a Node on the wire protocol (reference parity:
/root/reference/xotorch/networking/grpc/grpc_server.py:17-169)."""
from __future__ import annotations

import asyncio
import json
import traceback
from typing import Optional

import numpy as np

from xotorch_amd.helpers import DEBUG
from xotorch_amd.orchestration import wire
from xotorch_amd.shard import Shard


class Server:
  def __init__(self, node, host: str = "0.0.0.0", port: int = 50051):
    self.node = node
    self.host, self.port = host, port
    self._server: Optional[asyncio.AbstractServer] = None

  async def start(self):
    if self._server is not None:
      return  # idempotent: Node.start() also starts the server
    self._server = await asyncio.start_server(self._handle, self.host, self.port)

  async def stop(self):
    if self._server is not None:
      self._server.close()
      await self._server.wait_closed()
      self._server = None

  async def _handle(self, reader: asyncio.StreamReader, writer: asyncio.StreamWriter):
    try:
      while True:
        try:
          msg = await wire.read_frame(reader)
        except (asyncio.IncompleteReadError, ConnectionResetError):
          break
        reply = await self._dispatch(msg)
        wire.write_frame(writer, reply)
        await writer.drain()
    except Exception:
      if DEBUG >= 1:
        traceback.print_exc()
    finally:
      writer.close()
      try:
        await writer.wait_closed()
      except Exception:
        pass

  async def _dispatch(self, msg: dict) -> dict:
    t = msg.get("type")
    try:
      if t == "health":
        return {"ok": True}
      if t == "prompt":
        shard = Shard.from_dict(msg["shard"])
        asyncio.create_task(
          self.node.process_prompt(shard, msg["prompt"], msg.get("request_id"), msg.get("inference_state"))
        )
        return {"ok": True}
      if t == "tensor":
        shard = Shard.from_dict(msg["shard"])
        tensor = wire.unpack_tensor(msg["tensor"])
        asyncio.create_task(
          self.node.process_tensor(shard, tensor, msg.get("request_id"), msg.get("inference_state"))
        )
        return {"ok": True}
      if t == "example":
        shard = Shard.from_dict(msg["shard"])
        loss, grads = await self.node.process_example(
          shard, wire.unpack_tensor(msg["example"]), wire.unpack_tensor(msg["target"]),
          wire.unpack_tensor(msg["length"]), msg.get("train", False), msg.get("request_id"),
        )
        out = {"ok": True, "loss": float(loss)}
        if grads is not None:
          out["grads"] = wire.pack_tensor(np.asarray(grads))
        return out
      if t == "result":
        result = msg.get("result")
        if result is None and msg.get("tensor_result") is not None:
          result = wire.unpack_tensor(msg["tensor_result"])
        # finished results release this stage's KV session too (handle_result)
        await self.node.handle_result(msg["request_id"], result, msg["is_finished"])
        return {"ok": True}
      if t == "status":
        self.node.on_opaque_status.trigger_all(msg.get("request_id", ""), msg["status"])
        return {"ok": True}
      if t == "topology":
        topo = await self.node.collect_topology(set(msg.get("visited", [])), msg.get("max_depth", 4))
        return {"ok": True, "topology": topo.to_json()}
      return {"ok": False, "error": f"unknown message type {t}"}
    except Exception as e:
      if DEBUG >= 1:
        traceback.print_exc()
      return {"ok": False, "error": str(e)}
