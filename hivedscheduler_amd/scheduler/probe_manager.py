"""Post-bind placement validation workflow.

When an affinity group becomes fully allocated, the scheduler enqueues a
probe task for each (node, leaf cells) placement. Node agents poll
GET /v1/health/probes/{node}, run the native RCCL all-reduce probe over
exactly those GPUs (HIP_VISIBLE_DEVICES), and POST the result back. Results
are attached to the group (inspect API) and a busbw far below the xGMI
expectation marks the placement's cells bad.

The reference binds and hopes (scheduler.go:594-627); this closes the loop
with measured fabric bandwidth.
"""
from __future__ import annotations

import threading
import time
from typing import Dict, List, Optional


class ProbeManager:
    def __init__(self, min_busbw_gbps_per_cell: Optional[Dict[int, float]] = None):
        self.lock = threading.Lock()
        # node -> list of pending tasks
        self.pending: Dict[str, List[dict]] = {}
        # group -> results
        self.results: Dict[str, List[dict]] = {}
        # minimum healthy busbw by cell size (xGMI all-reduce is per-link
        # bound: a healthy pair sustains >100 GB/s busbw; use a conservative
        # floor so only truly degraded links trip it)
        self.min_busbw = min_busbw_gbps_per_cell or {2: 50.0, 4: 50.0, 8: 50.0}
        # daemon hygiene: cap retained groups and expire unpolled tasks so a
        # long-lived scheduler doesn't grow without bound
        self.max_groups = 1000
        self.task_ttl_s = 3600.0

    def enqueue_group(self, group_name: str, placements: Dict[str, List[int]]) -> None:
        """placements: node -> leaf cell indices used by the group there."""
        with self.lock:
            for node, cells in placements.items():
                if not cells:
                    continue
                self.pending.setdefault(node, []).append({
                    "group": group_name,
                    "node": node,
                    "leafCellIndices": sorted(cells),
                    "enqueued": time.time(),
                })

    def poll(self, node: str) -> List[dict]:
        """Agent polling: returns and drains this node's pending tasks
        (expired tasks — e.g. for nodes whose agent never came — are
        dropped)."""
        with self.lock:
            now = time.time()
            for n in list(self.pending):
                fresh = [t for t in self.pending[n] if now - t["enqueued"] < self.task_ttl_s]
                if fresh:
                    self.pending[n] = fresh
                else:
                    del self.pending[n]
            return self.pending.pop(node, [])

    def report(self, result: dict) -> dict:
        """Agent result: {'group', 'node', 'leafCellIndices', 'busbw_gbps',
        'algbw_gbps', 'ok', ...}. Returns the health verdict."""
        with self.lock:
            group = result.get("group", "")
            self.results.setdefault(group, []).append(result)
            while len(self.results) > self.max_groups:
                self.results.pop(next(iter(self.results)))
            n = len(result.get("leafCellIndices", []))
            floor = self.min_busbw.get(n, 0.0)
            healthy = bool(result.get("ok", False)) and (
                n <= 1 or float(result.get("busbw_gbps", 0.0)) >= floor)
            result["healthy"] = healthy
            result["min_busbw_gbps"] = floor
            return {"group": group, "healthy": healthy}

    def group_results(self, group_name: str) -> List[dict]:
        with self.lock:
            return list(self.results.get(group_name, []))

    def drop_group(self, group_name: str) -> None:
        with self.lock:
            self.results.pop(group_name, None)
            for node in list(self.pending):
                self.pending[node] = [t for t in self.pending[node]
                                      if t["group"] != group_name]
                if not self.pending[node]:
                    del self.pending[node]
