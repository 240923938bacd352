"""Viz rendering with synthetic topology/progress (reference viz/test_topology_viz.py)."""
from rich.console import Console

from xotorch_amd.parallel.partitioning import Partition
from xotorch_amd.parallel.topology import DeviceCapabilities, DeviceFlops, Topology
from xotorch_amd.viz.topology_viz import TopologyViz


def test_topology_viz_renders():
  viz = TopologyViz()
  topo = Topology()
  for i in range(8):
    topo.update_node(f"gpu{i}", DeviceCapabilities(
      model="AMD Instinct MI355X", chip="AMD INSTINCT MI355X", memory=294912,
      flops=DeviceFlops(fp16=2500.0)))
  parts = [Partition(f"gpu{i}", i / 8, (i + 1) / 8) for i in range(8)]
  viz.update_visualization(topo, parts, "gpu0")
  viz.update_prompt("r1", "what is a wavefront?")
  viz.update_response("r1", "a 64-lane SIMT group")
  viz.update_download("gpu0", {"repo_id": "meta/llama", "downloaded_bytes": 512, "total_bytes": 1024})
  console = Console(record=True, width=120)
  console.print(viz._render())
  text = console.export_text()
  assert "gpu0" in text and "MI355X" in text
  assert "GPU poor" in text
  assert "50.0%" in text  # download progress


def test_chat_tui_importable():
  from xotorch_amd.viz.chat_tui import run_chat_tui
  assert callable(run_chat_tui)
