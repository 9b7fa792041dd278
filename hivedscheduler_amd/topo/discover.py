"""Consume rocm-topo-discover output: measured node topology -> scheduler
config fragments.

The native tool (native/rocm_topo_discover.cpp) emits one YAML document per
node with `physicalCells` (pair/quad grouping derived from the measured
link-weight matrix, per-leaf measured hbmBytes, per-link xgmiLinks table).
This module parses that document into PhysicalCellSpec objects and merges
node fragments into a full cluster Config — the measured replacement for the
reference's hand-transcribed cellTypes YAML (doc/user-manual.md:44-72).
"""
from __future__ import annotations

import subprocess
from typing import Dict, List, Optional, Tuple

import yaml

from ..api.types import (
    Config,
    PhysicalCellSpec,
    PhysicalClusterSpec,
    VirtualCellSpec,
    VirtualClusterSpec,
)
from ..api import config as apicfg
from .mi355x import mi355x_cell_types


def parse_discovery_output(text: str) -> dict:
    """Parse one node's rocm-topo-discover YAML document."""
    doc = yaml.safe_load(text)
    if not isinstance(doc, dict) or "physicalCells" not in doc:
        raise ValueError("not a rocm-topo-discover document (no physicalCells)")
    return doc


def discovery_to_cell_specs(doc: dict) -> List[PhysicalCellSpec]:
    """The node's physicalCells fragment as typed specs (hbmBytes and
    xgmiLinks carried through to the core)."""
    return [PhysicalCellSpec.from_dict(c) for c in doc["physicalCells"]]


def run_discovery(binary: str = "rocm-topo-discover",
                  node_name: Optional[str] = None, timeout: float = 120.0) -> dict:
    """Run the native tool on this node and parse its output."""
    cmd = [binary]
    if node_name:
        cmd += ["--node-name", node_name]
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=timeout)
    if out.returncode != 0:
        raise RuntimeError(f"{binary} failed rc={out.returncode}: {out.stderr[:500]}")
    return parse_discovery_output(out.stdout)


def cluster_config_from_discovery(
    docs: List[dict],
    vcs: Optional[Dict[str, List[Tuple[str, int]]]] = None,
) -> Config:
    """Merge per-node discovery documents into a scheduler Config.

    vcs: VC name -> [(cellTypePath, number)]; default one VC owning every
    discovered node-level cell of the first node's type.
    """
    cells: List[PhysicalCellSpec] = []
    for doc in docs:
        cells.extend(discovery_to_cell_specs(doc))
    if not cells:
        raise ValueError("no physical cells discovered")
    if vcs is None:
        top_type = cells[0].cellType
        count = sum(1 for c in cells if c.cellType == top_type)
        vcs = {"VC1": [(top_type, count)]}
    cfg = Config(
        physicalCluster=PhysicalClusterSpec(
            cellTypes=mi355x_cell_types(),
            physicalCells=cells,
        ),
        virtualClusters={
            vc: VirtualClusterSpec(
                virtualCells=[VirtualCellSpec(cellType=t, cellNumber=n) for t, n in quota]
            )
            for vc, quota in vcs.items()
        },
    )
    apicfg.infer_physical_cluster(cfg.physicalCluster)
    return cfg
