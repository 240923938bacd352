"""Wire-protocol unit tests: framing, tensor round-trip, server dispatch
edges (reference surface: networking/grpc/node_service.proto + grpc_server)."""
import asyncio

import numpy as np
import pytest

from xotorch_amd.orchestration import wire


def test_tensor_roundtrip():
  for arr in (np.random.randn(3, 5).astype(np.float32),
              np.random.randn(2, 1, 4).astype(np.float16),
              np.arange(7, dtype=np.int64)):
    d = wire.pack_tensor(arr)
    back = wire.unpack_tensor(d)
    assert back.dtype == arr.dtype and back.shape == arr.shape
    assert np.array_equal(back, arr)
  assert wire.pack_tensor(None) is None
  assert wire.unpack_tensor(None) is None


def test_frame_roundtrip_and_limits():
  async def go():
    server_got = []

    async def handle(reader, writer):
      msg = await wire.read_frame(reader)
      server_got.append(msg)
      wire.write_frame(writer, {"ok": True, "echo": msg["x"]})
      await writer.drain()
      writer.close()

    srv = await asyncio.start_server(handle, "127.0.0.1", 0)
    port = srv.sockets[0].getsockname()[1]
    reply = await wire.request("127.0.0.1", port, {"x": 41, "blob": b"\x00" * 1024})
    assert reply == {"ok": True, "echo": 41}
    assert server_got[0]["blob"] == b"\x00" * 1024
    srv.close()
    await srv.wait_closed()
  asyncio.run(go())


def test_oversize_frame_rejected():
  async def go():
    async def handle(reader, writer):
      # advertise an over-cap frame length; client must refuse to read it
      import struct
      writer.write(struct.pack("!I", wire.MAX_FRAME + 1))
      await writer.drain()
      await asyncio.sleep(0.2)
      writer.close()

    srv = await asyncio.start_server(handle, "127.0.0.1", 0)
    port = srv.sockets[0].getsockname()[1]
    reader, writer = await asyncio.open_connection("127.0.0.1", port)
    with pytest.raises(ValueError):
      await wire.read_frame(reader)
    writer.close()
    srv.close()
    await srv.wait_closed()
  asyncio.run(go())


def test_server_unknown_type():
  async def go():
    from xotorch_amd.orchestration.server import Server
    srv = Server(node=None, host="127.0.0.1", port=0)
    reply = await srv._dispatch({"type": "nonsense"})
    assert reply["ok"] is False and "unknown" in reply["error"]
  asyncio.run(go())


def test_malformed_frames_fail_cleanly():
  """Garbage length prefixes, truncated payloads and non-msgpack bytes must
  raise (never hang or return garbage) — a crashing peer cannot wedge the
  control plane."""
  async def go():
    async def serve_bytes(payload):
      got = {}

      async def handle(reader, writer):
        try:
          got["msg"] = await asyncio.wait_for(wire.read_frame(reader), 2)
        except Exception as e:
          got["err"] = type(e).__name__
        writer.close()

      srv = await asyncio.start_server(handle, "127.0.0.1", 0)
      port = srv.sockets[0].getsockname()[1]
      r, w = await asyncio.open_connection("127.0.0.1", port)
      w.write(payload)
      await w.drain()
      w.close()
      await asyncio.sleep(0.1)
      srv.close()
      await srv.wait_closed()
      return got

    # oversize length prefix (wire framing is 4-byte !I)
    got = await serve_bytes((0xFFFFFFFF).to_bytes(4, "big"))
    assert got.get("err") == "ValueError", got
    # truncated payload (claims 100 bytes, sends 3)
    got = await serve_bytes((100).to_bytes(4, "big") + b"abc")
    assert "err" in got, got
    # valid length, non-msgpack garbage
    body = b"\xc1\xff\x00garbage"
    got = await serve_bytes(len(body).to_bytes(4, "big") + body)
    assert "err" in got, got
  asyncio.run(go())


def test_tensor_roundtrip_property_fuzz():
  """Hypothesis fuzz: pack/unpack preserves dtype, shape and bytes for all
  wire-legal dtypes incl. 0-d and empty tensors."""
  from hypothesis import given, settings, strategies as st

  dtypes = [np.float32, np.float16, np.int64, np.int32, np.uint8, np.bool_]

  @settings(max_examples=200, deadline=None)
  @given(
    dt=st.sampled_from(dtypes),
    shape=st.lists(st.integers(min_value=0, max_value=5), min_size=0, max_size=4),
    seed=st.integers(min_value=0, max_value=2**31 - 1),
  )
  def check(dt, shape, seed):
    rng = np.random.default_rng(seed)
    arr = (rng.random(shape) * 100).astype(dt)
    back = wire.unpack_tensor(wire.pack_tensor(arr))
    assert back.dtype == arr.dtype and back.shape == arr.shape
    assert np.array_equal(back, arr)

  check()
