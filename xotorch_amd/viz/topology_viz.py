"""Terminal topology visualization (rich Live), reference parity:
/root/reference/xotorch/viz/topology_viz.py:20-378 — ring of nodes with
capabilities + partition ranges, total-FLOPS bar, last prompt/response panel,
download progress rows."""
from __future__ import annotations

from typing import Dict, List, Optional

from rich.console import Console, Group
from rich.live import Live
from rich.panel import Panel
from rich.table import Table
from rich.text import Text

from xotorch_amd.helpers import pretty_print_bytes
from xotorch_amd.parallel.partitioning import Partition
from xotorch_amd.parallel.topology import Topology


class TopologyViz:
  def __init__(self, chatgpt_api_endpoints: Optional[List[str]] = None):
    self.topology = Topology()
    self.partitions: List[Partition] = []
    self.node_id: Optional[str] = None
    self.chatgpt_api_endpoints = chatgpt_api_endpoints or []
    self.prompts: List[str] = []
    self.responses: Dict[str, str] = {}
    self.download_rows: Dict[str, dict] = {}
    self.console = Console()
    self._live: Optional[Live] = None

  def start(self):
    if self._live is None:
      self._live = Live(self._render(), console=self.console, refresh_per_second=4)
      self._live.start()

  def stop(self):
    if self._live is not None:
      self._live.stop()
      self._live = None

  def update_visualization(self, topology: Topology, partitions: List[Partition], node_id: str = None):
    self.topology = topology
    self.partitions = partitions
    self.node_id = node_id or self.node_id
    self._refresh()

  def update_prompt(self, request_id: str, prompt: str):
    self.prompts = ([prompt] + self.prompts)[:3]
    self._refresh()

  def update_response(self, request_id: str, response: str):
    self.responses[request_id] = response
    self._refresh()

  def update_download(self, node_id: str, progress: dict):
    self.download_rows[node_id] = progress
    self._refresh()

  def _refresh(self):
    if self._live is not None:
      self._live.update(self._render())

  def _render(self):
    table = Table(title="ring topology", expand=True)
    table.add_column("node")
    table.add_column("device")
    table.add_column("memory")
    table.add_column("fp16 TFLOPS", justify="right")
    table.add_column("layers", justify="left")
    part_by_node = {p.node_id: p for p in self.partitions}
    total_flops = 0.0
    for nid, caps in self.topology.all_nodes():
      p = part_by_node.get(nid)
      rng = f"[{p.start:.3f}, {p.end:.3f})" if p else "-"
      marker = "→ " if nid == self.node_id else "  "
      total_flops += caps.flops.fp16
      table.add_row(marker + nid[:12], caps.model, pretty_print_bytes(caps.memory * 1024 * 1024),
                    f"{caps.flops.fp16:.1f}", rng)
    bar_len = 40
    # "GPU poor/rich" bar: 2.5 PF (one MI355X) pegs the middle
    frac = min(1.0, total_flops / 5000.0)
    bar = Text("GPU poor " + "█" * int(frac * bar_len) + "░" * (bar_len - int(frac * bar_len)) + " GPU rich")
    blocks = [table, bar]
    if self.prompts:
      chat = Group(*[Text(f"> {p[:120]}") for p in self.prompts],
                   *[Text(f"< {r[:120]}") for r in list(self.responses.values())[-3:]])
      blocks.append(Panel(chat, title="last requests"))
    if self.download_rows:
      dl = Table(title="downloads", expand=True)
      dl.add_column("node")
      dl.add_column("repo")
      dl.add_column("progress")
      for nid, pr in self.download_rows.items():
        pct = 100.0 * pr.get("downloaded_bytes", 0) / max(1, pr.get("total_bytes", 1))
        dl.add_row(nid[:12], str(pr.get("repo_id", "")), f"{pct:.1f}%")
      blocks.append(dl)
    return Group(*blocks)
