"""Elastic reconnect test (the reference covers this only with a manual shell
script, test/reconnect.sh): peer loss shrinks the ring, rejoin restores it,
and requests keep working after each transition."""
import asyncio
import json

import pytest

from xotorch_amd.engine.dummy import DummyEngine
from xotorch_amd.helpers import find_available_port
from xotorch_amd.models.registry import build_base_shard
from xotorch_amd.orchestration.discovery import ManualDiscovery
from xotorch_amd.orchestration.node import Node
from xotorch_amd.orchestration.server import Server


def run(coro):
  return asyncio.new_event_loop().run_until_complete(coro)


async def decode_once(node, shard, rid):
  done = asyncio.Event()
  got = []

  def on_token(r, toks, fin):
    if r != rid:
      return
    got.extend(toks)
    if fin:
      done.set()

  node.on_token.register(f"cb-{rid}").on_next(on_token)
  await node.process_prompt(shard, "ping", rid)
  await asyncio.wait_for(done.wait(), 30)
  node.on_token.deregister(f"cb-{rid}")
  return got


@pytest.mark.timeout(180)
def test_peer_loss_and_rejoin(tmp_path):
  async def go():
    ports = [find_available_port("127.0.0.1") for _ in range(2)]
    cfg_path = tmp_path / "topo.json"
    cfg_path.write_text(json.dumps({"peers": {
      "ra": {"address": f"127.0.0.1:{ports[0]}"},
      "rb": {"address": f"127.0.0.1:{ports[1]}"},
    }}))
    nodes = []
    for name, port in zip(("ra", "rb"), ports):
      disc = ManualDiscovery(str(cfg_path), name, interval=0.3)
      n = Node(name, None, DummyEngine(), disc, max_generate_tokens=3)
      n.server = Server(n, "127.0.0.1", port)
      await n.server.start()
      nodes.append(n)
    a, b = nodes
    for n in nodes:
      await n.start(wait_for_peers=1)
    shard = build_base_shard("dummy", "DummyEngine")
    assert len(a.peers) == 1
    got = await decode_once(a, shard, "r1")
    assert got

    # kill b: a must shrink to a single-node ring and keep serving
    await b.stop()
    for _ in range(60):
      await a.update_peers()
      if not a.peers:
        break
      await asyncio.sleep(0.3)
    assert not a.peers
    await a.collect_topology(set())
    mine = a.get_current_shard(shard)
    assert mine.get_layer_count() == shard.n_layers
    got = await decode_once(a, shard, "r2")
    assert got

    # restart b on the same port: a must re-add it
    disc_b = ManualDiscovery(str(cfg_path), "rb", interval=0.3)
    b2 = Node("rb", None, DummyEngine(), disc_b, max_generate_tokens=3)
    b2.server = Server(b2, "127.0.0.1", ports[1])
    await b2.server.start()
    await b2.start(wait_for_peers=1)
    for _ in range(60):
      await a.update_peers()
      if a.peers:
        break
      await asyncio.sleep(0.3)
    assert [p.id() for p in a.peers] == ["rb"]
    await a.collect_topology(set())
    got = await decode_once(a, shard, "r3")
    assert got
    await a.stop()
    await b2.stop()
    return True
  assert run(go())
