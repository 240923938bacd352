"""Topology graph + device capability probe.

Parity with the reference's topology layer
(/root/reference/xotorch/topology/topology.py:21-75 and
device_capabilities.py:22-52,167-348) redesigned for an MI355X-first fleet:
the probe goes through torch's ROCm runtime (device name, HBM size) and
amd-smi/rocm-smi when present, with a static peak-TFLOPS table for the
Instinct parts we schedule for. CPU-only hosts report system RAM so the
gloo-based test rings still partition sensibly.
"""
from __future__ import annotations

import os
from dataclasses import dataclass, field
from typing import Dict, Iterable, Optional, Set, Tuple


@dataclass(frozen=True)
class DeviceFlops:
  # TFLOPS
  fp32: float = 0.0
  fp16: float = 0.0
  int8: float = 0.0

  def to_dict(self):
    return {"fp32": self.fp32, "fp16": self.fp16, "int8": self.int8}


@dataclass(frozen=True)
class DeviceCapabilities:
  model: str
  chip: str
  memory: int  # MB
  flops: DeviceFlops = field(default_factory=DeviceFlops)

  def to_dict(self):
    return {"model": self.model, "chip": self.chip, "memory": self.memory, "flops": self.flops.to_dict()}

  @classmethod
  def from_dict(cls, d):
    fl = d.get("flops", {}) or {}
    return cls(model=d.get("model", "unknown"), chip=d.get("chip", "unknown"), memory=int(d.get("memory", 0)),
               flops=DeviceFlops(fp32=fl.get("fp32", 0.0), fp16=fl.get("fp16", 0.0), int8=fl.get("int8", 0.0)))


UNKNOWN_DEVICE_CAPABILITIES = DeviceCapabilities(model="unknown", chip="unknown", memory=0)

# Dense (not 2:1-sparse) peak TFLOPS for the chips this framework targets.
CHIP_FLOPS: Dict[str, DeviceFlops] = {
  "AMD INSTINCT MI355X": DeviceFlops(fp32=157.3, fp16=2500.0, int8=5000.0),
  "AMD INSTINCT MI350X": DeviceFlops(fp32=144.0, fp16=2300.0, int8=4600.0),
  "AMD INSTINCT MI325X": DeviceFlops(fp32=163.4, fp16=1307.4, int8=2614.9),
  "AMD INSTINCT MI300X": DeviceFlops(fp32=163.4, fp16=1307.4, int8=2614.9),
  "AMD INSTINCT MI250X": DeviceFlops(fp32=47.9, fp16=383.0, int8=383.0),
}


def _chip_key(name: str) -> str:
  up = name.upper()
  for key in CHIP_FLOPS:
    if key in up or key.replace("AMD ", "") in up:
      return key
  return up


def device_capabilities() -> DeviceCapabilities:
  """Probe this process's device. GPU (ROCm) if visible, else host RAM."""
  try:
    import torch
    if torch.cuda.is_available():
      idx = torch.cuda.current_device()
      props = torch.cuda.get_device_properties(idx)
      name = props.name
      key = _chip_key(name)
      flops = CHIP_FLOPS.get(key, DeviceFlops())
      mem_mb = props.total_memory // (1024 * 1024)
      return DeviceCapabilities(model=name, chip=key, memory=int(mem_mb), flops=flops)
  except Exception:
    pass
  # CPU fallback: system memory so ring partitioning still weights sensibly.
  try:
    import psutil
    mem_mb = psutil.virtual_memory().total // (1024 * 1024)
  except Exception:
    mem_mb = 8192
  cpu = os.uname().machine if hasattr(os, "uname") else "cpu"
  return DeviceCapabilities(model=f"cpu-{cpu}", chip="cpu", memory=int(mem_mb))


@dataclass(frozen=True)
class PeerConnection:
  from_id: str
  to_id: str
  description: Optional[str] = None


class Topology:
  """Node→capabilities map plus a directed peer edge set, mergeable from gossip."""

  def __init__(self):
    self.nodes: Dict[str, DeviceCapabilities] = {}
    self.peer_graph: Dict[str, Set[PeerConnection]] = {}
    self.active_node_id: Optional[str] = None

  def update_node(self, node_id: str, capabilities: DeviceCapabilities):
    self.nodes[node_id] = capabilities

  def get_node(self, node_id: str) -> Optional[DeviceCapabilities]:
    return self.nodes.get(node_id)

  def all_nodes(self) -> Iterable[Tuple[str, DeviceCapabilities]]:
    return self.nodes.items()

  def add_edge(self, from_id: str, to_id: str, description: Optional[str] = None):
    conn = PeerConnection(from_id=from_id, to_id=to_id, description=description)
    self.peer_graph.setdefault(from_id, set()).add(conn)

  def get_neighbors(self, node_id: str) -> Set[str]:
    return {c.to_id for c in self.peer_graph.get(node_id, set())}

  def merge(self, other: "Topology", merging_peer_id: Optional[str] = None):
    for node_id, cap in other.nodes.items():
      # never downgrade a known node to a zero-memory placeholder (gossip
      # replies list already-visited peers with handle-level stub caps)
      if cap.memory == 0 and node_id in self.nodes and self.nodes[node_id].memory > 0:
        continue
      self.update_node(node_id, cap)
    for node_id, conns in other.peer_graph.items():
      for conn in conns:
        self.add_edge(conn.from_id, conn.to_id, conn.description)
    if merging_peer_id is not None and other.active_node_id is not None:
      self.add_edge(merging_peer_id, other.active_node_id)

  def to_json(self) -> dict:
    return {
      "nodes": {nid: cap.to_dict() for nid, cap in self.nodes.items()},
      "peer_graph": {
        nid: [{"from": c.from_id, "to": c.to_id, "description": c.description} for c in conns]
        for nid, conns in self.peer_graph.items()
      },
      "active_node_id": self.active_node_id,
    }

  @classmethod
  def from_json(cls, d: dict) -> "Topology":
    t = cls()
    for nid, cap in (d.get("nodes") or {}).items():
      t.update_node(nid, DeviceCapabilities.from_dict(cap))
    for nid, conns in (d.get("peer_graph") or {}).items():
      for c in conns:
        t.add_edge(c["from"], c["to"], c.get("description"))
    t.active_node_id = d.get("active_node_id")
    return t

  def __str__(self):
    return f"Topology(nodes={list(self.nodes)}, edges={sum(len(v) for v in self.peer_graph.values())})"
