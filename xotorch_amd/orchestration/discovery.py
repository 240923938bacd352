"""Peer discovery: manual JSON config (hot-reloaded) and UDP broadcast.

Capability parity with the reference's discovery layer
(/root/reference/xotorch/networking/discovery.py:6-18,
manual/manual_discovery.py:13-101, udp/udp_discovery.py:51-246):
- ManualDiscovery: static JSON topology file, reloaded on mtime change,
  peers health-checked every `interval`; unhealthy peers are dropped and
  re-added when they come back.
- UDPDiscovery: periodic JSON presence broadcast + listener; stale/unhealthy
  peers cleaned up after `discovery_timeout`.
"""
from __future__ import annotations

import asyncio
import json
import socket
import time
from abc import ABC, abstractmethod
from pathlib import Path
from typing import Callable, Dict, List, Optional

from xotorch_amd.helpers import DEBUG_DISCOVERY
from xotorch_amd.orchestration.peer import PeerHandle, TCPPeerHandle
from xotorch_amd.parallel.topology import DeviceCapabilities, device_capabilities


class Discovery(ABC):
  @abstractmethod
  async def start(self) -> None: ...

  @abstractmethod
  async def stop(self) -> None: ...

  @abstractmethod
  async def discover_peers(self, wait_for_peers: int = 0) -> List[PeerHandle]: ...


class ManualDiscovery(Discovery):
  def __init__(self, config_path: str, node_id: str,
               create_peer_handle: Callable[[str, str, str, DeviceCapabilities], PeerHandle] = None,
               interval: float = 5.0):
    self.config_path = Path(config_path)
    self.node_id = node_id
    self.create_peer_handle = create_peer_handle or (
      lambda pid, addr, desc, caps: TCPPeerHandle(pid, addr, desc, caps)
    )
    self.interval = interval
    self.known_peers: Dict[str, PeerHandle] = {}
    self._mtime: float = 0.0
    self._peers_cfg: Dict[str, dict] = {}
    self._task: Optional[asyncio.Task] = None

  def _load(self):
    mtime = self.config_path.stat().st_mtime
    if mtime == self._mtime:
      return
    self._mtime = mtime
    raw = json.loads(self.config_path.read_text())
    peers = raw.get("peers", raw)
    if not isinstance(peers, dict):
      raise ValueError("manual discovery config must map peer-id -> {address, ...}")
    self._peers_cfg = {pid: cfg for pid, cfg in peers.items() if pid != self.node_id}

  async def start(self):
    self._load()
    self._task = asyncio.create_task(self._loop())

  async def stop(self):
    if self._task:
      self._task.cancel()
      try:
        await self._task
      except asyncio.CancelledError:
        pass

  async def _loop(self):
    while True:
      try:
        self._load()
        await self._refresh()
      except FileNotFoundError:
        pass
      except Exception as e:
        if DEBUG_DISCOVERY >= 1:
          print(f"manual discovery error: {e}")
      await asyncio.sleep(self.interval)

  async def _refresh(self):
    for pid, cfg in self._peers_cfg.items():
      addr = f"{cfg.get('address', cfg.get('host', '127.0.0.1'))}:{cfg.get('port', 50051)}" \
        if "port" in cfg or "host" in cfg else cfg["address"]
      if pid not in self.known_peers:
        caps = DeviceCapabilities.from_dict(cfg.get("device_capabilities", {})) \
          if cfg.get("device_capabilities") else None
        handle = self.create_peer_handle(pid, addr, cfg.get("description", ""), caps)
        if await handle.health_check():
          self.known_peers[pid] = handle
      else:
        if not await self.known_peers[pid].health_check():
          del self.known_peers[pid]
    # drop peers removed from the config
    for pid in list(self.known_peers):
      if pid not in self._peers_cfg:
        del self.known_peers[pid]

  async def discover_peers(self, wait_for_peers: int = 0) -> List[PeerHandle]:
    self._load()
    await self._refresh()
    while len(self.known_peers) < wait_for_peers:
      await asyncio.sleep(1.0)
      self._load()
      await self._refresh()
    return list(self.known_peers.values())


class UDPDiscovery(Discovery):
  def __init__(self, node_id: str, node_port: int, listen_port: int, broadcast_port: Optional[int] = None,
               create_peer_handle: Callable = None, broadcast_interval: float = 1.0,
               discovery_timeout: float = 30.0, device_caps: Optional[DeviceCapabilities] = None,
               allowed_node_ids: Optional[List[str]] = None):
    self.node_id = node_id
    self.node_port = node_port
    self.listen_port = listen_port
    self.broadcast_port = broadcast_port or listen_port
    self.create_peer_handle = create_peer_handle or (
      lambda pid, addr, desc, caps: TCPPeerHandle(pid, addr, desc, caps)
    )
    self.broadcast_interval = broadcast_interval
    self.discovery_timeout = discovery_timeout
    self.device_caps = device_caps
    self.allowed_node_ids = allowed_node_ids
    self.known_peers: Dict[str, tuple] = {}  # id -> (handle, first_seen, last_seen)
    self._tasks: List[asyncio.Task] = []
    self._transport = None

  async def start(self):
    if self.device_caps is None:
      self.device_caps = await asyncio.get_running_loop().run_in_executor(None, device_capabilities)
    self._tasks = [
      asyncio.create_task(self._broadcast_loop()),
      asyncio.create_task(self._listen()),
      asyncio.create_task(self._cleanup_loop()),
    ]

  async def stop(self):
    for t in self._tasks:
      t.cancel()
    for t in self._tasks:
      try:
        await t
      except asyncio.CancelledError:
        pass
    if self._transport is not None:
      self._transport.close()

  async def discover_peers(self, wait_for_peers: int = 0) -> List[PeerHandle]:
    while len(self.known_peers) < wait_for_peers:
      await asyncio.sleep(0.2)
    return [h for h, _, _ in self.known_peers.values()]

  async def _broadcast_loop(self):
    from xotorch_amd.helpers import interface_priority
    try:
      ifname = next((n for n in __import__("os").listdir("/sys/class/net") if not n.startswith("lo")), "eth0")
    except Exception:
      ifname = "eth0"
    msg = json.dumps({
      "type": "discovery", "node_id": self.node_id, "node_port": self.node_port,
      "device_capabilities": self.device_caps.to_dict(),
      "interface_name": ifname, "interface_priority": interface_priority(ifname),
    }).encode()
    sock = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
    sock.setsockopt(socket.SOL_SOCKET, socket.SO_BROADCAST, 1)
    sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
    sock.setblocking(False)
    loop = asyncio.get_running_loop()
    while True:
      try:
        await loop.run_in_executor(None, sock.sendto, msg, ("255.255.255.255", self.broadcast_port))
        await loop.run_in_executor(None, sock.sendto, msg, ("127.0.0.1", self.broadcast_port))
      except Exception as e:
        if DEBUG_DISCOVERY >= 2:
          print(f"broadcast error: {e}")
      await asyncio.sleep(self.broadcast_interval)

  async def _listen(self):
    loop = asyncio.get_running_loop()

    class Proto(asyncio.DatagramProtocol):
      def __init__(p):  # noqa: N805
        p.queue = asyncio.Queue()

      def datagram_received(p, data, addr):  # noqa: N805
        p.queue.put_nowait((data, addr))

    transport, proto = await loop.create_datagram_endpoint(
      Proto, local_addr=("0.0.0.0", self.listen_port), reuse_port=True if hasattr(socket, "SO_REUSEPORT") else None
    )
    self._transport = transport
    while True:
      data, addr = await proto.queue.get()
      await self._on_msg(data, addr)

  async def _on_msg(self, data: bytes, addr):
    try:
      msg = json.loads(data.decode())
    except Exception:
      return
    if msg.get("type") != "discovery":
      return
    pid = msg.get("node_id")
    if not pid or pid == self.node_id:
      return
    if self.allowed_node_ids and pid not in self.allowed_node_ids:
      return
    now = time.time()
    prio = int(msg.get("interface_priority", 2))
    if pid in self.known_peers:
      handle, first, _ = self.known_peers[pid]
      # a higher-priority interface replaces the stored connection
      # (reference udp_discovery.py:180-186)
      new_addr = f"{addr[0]}:{msg.get('node_port')}"
      if prio > getattr(handle, "_iface_priority", 2) and new_addr != handle.addr():
        caps = DeviceCapabilities.from_dict(msg.get("device_capabilities", {}))
        replacement = self.create_peer_handle(pid, new_addr, "udp", caps)
        replacement._iface_priority = prio
        if await replacement.health_check():
          self.known_peers[pid] = (replacement, first, now)
          return
      self.known_peers[pid] = (handle, first, now)
      return
    caps = DeviceCapabilities.from_dict(msg.get("device_capabilities", {}))
    peer_addr = f"{addr[0]}:{msg.get('node_port')}"
    handle = self.create_peer_handle(pid, peer_addr, "udp", caps)
    handle._iface_priority = prio
    if await handle.health_check():
      if DEBUG_DISCOVERY >= 1:
        print(f"discovered peer {pid} at {peer_addr}")
      self.known_peers[pid] = (handle, now, now)

  async def _cleanup_loop(self):
    while True:
      now = time.time()
      for pid in list(self.known_peers):
        handle, first, last = self.known_peers[pid]
        if now - last > self.discovery_timeout or not await handle.health_check():
          if DEBUG_DISCOVERY >= 1:
            print(f"removing stale peer {pid}")
          self.known_peers.pop(pid, None)
      await asyncio.sleep(self.broadcast_interval * 5)
