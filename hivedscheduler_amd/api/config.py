"""Config load + defaulting + recursive cell-spec inference + hot-reload watch.

Parity with reference pkg/api/config.go:
- NewConfig defaults (l.87-118)
- recursive inference of omitted cell addresses/children (l.120-167)
- viper-style file watch -> exit(0) so the orchestrator restarts with the new
  config (work-preserving reconfiguration, l.202-217)
"""
from __future__ import annotations

import os
import threading
import time
from typing import Dict, List, Optional

import yaml

from . import constants
from .types import (
    CellTypeSpec,
    Config,
    PhysicalCellSpec,
    PhysicalClusterSpec,
    VirtualClusterSpec,
    WebServerError,
)


def init_raw_config(path: Optional[str] = None) -> Config:
    """Read the YAML config file located by `path` or $CONFIG."""
    if path is None:
        path = os.environ.get(constants.EnvNameConfigFilePath, constants.DefaultConfigFilePath)
    with open(path, "r") as f:
        raw = yaml.safe_load(f) or {}
    return new_config(raw)


def new_config(raw: dict) -> Config:
    cfg = Config(
        kubeApiServerAddress=raw.get("kubeApiServerAddress"),
        kubeConfigFilePath=raw.get("kubeConfigFilePath"),
        webServerAddress=raw.get("webServerAddress") or constants.DefaultWebServerAddress,
        forcePodBindThreshold=int(raw.get("forcePodBindThreshold", 3)),
        waitingPodSchedulingBlockMilliSec=int(raw.get("waitingPodSchedulingBlockMilliSec", 0)),
        physicalCluster=PhysicalClusterSpec.from_dict(raw.get("physicalCluster") or {}),
        virtualClusters={
            k: VirtualClusterSpec.from_dict(v or {}) for k, v in (raw.get("virtualClusters") or {}).items()
        },
    )
    infer_physical_cluster(cfg.physicalCluster)
    return cfg


# ---------------------------------------------------------------------------
# Cell-spec inference
# ---------------------------------------------------------------------------


def is_leaf_type(cell_types: Dict[str, CellTypeSpec], type_name: str) -> bool:
    ct = cell_types.get(type_name)
    return ct is None or not ct.childCellType


def type_chain(cell_types: Dict[str, CellTypeSpec], top_type: str) -> List[str]:
    """Top-down list of type names from `top_type` to the leaf type."""
    chain = [top_type]
    seen = {top_type}
    t = top_type
    while not is_leaf_type(cell_types, t):
        t = cell_types[t].childCellType  # type: ignore[union-attr]
        if t in seen:
            raise WebServerError.bad_request(f"cellTypes contains a cycle at {t!r}")
        seen.add(t)
        chain.append(t)
    return chain


def infer_physical_cluster(pc: PhysicalClusterSpec) -> None:
    """Fill in omitted cellChildren and cellAddress fields in-place.

    Address defaults (reference pkg/api/config.go:120-167 + design-config doc):
    - below node level: relative index of the cell at its level within its node
    - at/above node level: node address must be user-given (node name); above
      node, relative index within the same top-level cell
    - top-level: index in the physicalCells array
    """
    for i, cell in enumerate(pc.physicalCells):
        if not cell.cellType:
            raise WebServerError.bad_request(f"physicalCells[{i}]: cellType is required")
        chain = type_chain(pc.cellTypes, cell.cellType)
        if not cell.cellAddress:
            cell.cellAddress = str(i)
        _infer_cell(pc.cellTypes, cell, chain, 0, _NodeCounters())


class _NodeCounters:
    """Per-level running index used to assign default addresses within a scope."""

    def __init__(self) -> None:
        self.counters: Dict[int, int] = {}

    def next(self, depth: int) -> int:
        v = self.counters.get(depth, 0)
        self.counters[depth] = v + 1
        return v


def _is_node_level(cell_types: Dict[str, CellTypeSpec], type_name: str) -> bool:
    ct = cell_types.get(type_name)
    return ct is not None and ct.isNodeLevel


def _infer_cell(
    cell_types: Dict[str, CellTypeSpec],
    cell: PhysicalCellSpec,
    chain: List[str],
    depth: int,
    scope: _NodeCounters,
) -> None:
    type_name = chain[depth]
    cell.cellType = type_name
    is_node = _is_node_level(cell_types, type_name)
    if is_node and not cell.cellAddress:
        raise WebServerError.bad_request(
            f"node-level cell of type {type_name!r} must specify cellAddress (the K8s node name)"
        )
    if depth == len(chain) - 1:
        if cell.cellChildren:
            raise WebServerError.bad_request(f"leaf cell {cell.cellAddress!r} must not have children")
        return
    child_type = chain[depth + 1]
    child_num = cell_types[type_name].childCellNumber
    if not cell.cellChildren:
        cell.cellChildren = [PhysicalCellSpec(cellType=child_type) for _ in range(child_num)]
    elif len(cell.cellChildren) != child_num:
        raise WebServerError.bad_request(
            f"cell {cell.cellAddress!r} of type {type_name!r} has {len(cell.cellChildren)} children, "
            f"cellTypes says {child_num}"
        )
    # Entering a node resets the per-level default-address scope: below-node
    # addresses are relative to the node; above-node, to the top-level cell.
    child_scope = _NodeCounters() if is_node else scope
    for child in cell.cellChildren:
        child.cellType = child_type
        if not child.cellAddress:
            child.cellAddress = str(child_scope.next(depth + 1))
        _infer_cell(cell_types, child, chain, depth + 1, child_scope)


# ---------------------------------------------------------------------------
# Config watch: exit(0) on change so the orchestrator restarts us with the new
# config (work-preserving reconfiguration; state is rebuilt from pod-bind-info
# annotations on restart).
# ---------------------------------------------------------------------------


def watch_config(path: str, poll_interval_s: float = 2.0) -> threading.Thread:
    def _watch() -> None:
        try:
            last = os.stat(path).st_mtime
        except OSError:
            last = 0.0
        while True:
            time.sleep(poll_interval_s)
            try:
                cur = os.stat(path).st_mtime
            except OSError:
                continue
            if cur != last:
                # Reference pkg/api/config.go:202-217: exit cleanly; the
                # StatefulSet restarts us and AddAllocatedPod replay reconciles.
                os._exit(0)

    t = threading.Thread(target=_watch, name="config-watch", daemon=True)
    t.start()
    return t
