"""UDP discovery test: two instances with mirrored ports discover each other
over loopback (reference pattern: networking/udp/test_udp_discovery.py)."""
import asyncio

import pytest

from xotorch_amd.engine.dummy import DummyEngine
from xotorch_amd.helpers import find_available_port
from xotorch_amd.orchestration.discovery import UDPDiscovery
from xotorch_amd.orchestration.node import Node
from xotorch_amd.orchestration.server import Server


def run(coro):
  return asyncio.new_event_loop().run_until_complete(coro)


@pytest.mark.timeout(90)
def test_udp_discovery_pair():
  async def go():
    # two TCP node servers so health checks pass
    tcp_a, tcp_b = find_available_port("127.0.0.1"), find_available_port("127.0.0.1")
    node_a = Node("udp-a", None, DummyEngine(), None)
    node_a.server = Server(node_a, "127.0.0.1", tcp_a)
    node_b = Node("udp-b", None, DummyEngine(), None)
    node_b.server = Server(node_b, "127.0.0.1", tcp_b)
    await node_a.server.start()
    await node_b.server.start()
    udp1, udp2 = find_available_port("127.0.0.1"), find_available_port("127.0.0.1")
    da = UDPDiscovery("udp-a", tcp_a, listen_port=udp1, broadcast_port=udp2, broadcast_interval=0.2)
    db = UDPDiscovery("udp-b", tcp_b, listen_port=udp2, broadcast_port=udp1, broadcast_interval=0.2)
    await da.start()
    await db.start()
    try:
      peers_a = await asyncio.wait_for(da.discover_peers(wait_for_peers=1), 30)
      peers_b = await asyncio.wait_for(db.discover_peers(wait_for_peers=1), 30)
      assert [p.id() for p in peers_a] == ["udp-b"]
      assert [p.id() for p in peers_b] == ["udp-a"]
      assert peers_a[0].device_capabilities().memory > 0
      # health-checked handle points at b's TCP server
      assert await peers_a[0].health_check()
    finally:
      await da.stop()
      await db.stop()
      await node_a.server.stop()
      await node_b.server.stop()
    return True
  assert run(go())
