"""Node health agent (DaemonSet-style): grounds cell healthiness in measured
CDNA4 facts and reports GPU-level health to the scheduler.

Per SURVEY.md §2.2, this replaces the reference's K8s-NodeReady-only health
signal with: rocm-smi device state, HIP health-probe kernels (HBM bandwidth,
MFMA verification), and the RCCL xGMI pair probe. Degraded xGMI links mark
the PAIR cell bad (via its endpoint leaf), not the whole node.

Runs on each GPU node: `python -m hivedscheduler_amd.agent --scheduler URL`.
Reports POST to the scheduler's /v1/health/nodes/{node} endpoint:
  {"gpus": {"0": {"healthy": true, ...}, ...}}
"""
from __future__ import annotations

import json
import logging
import os
import socket
import subprocess
import time
from typing import Dict, List, Optional

log = logging.getLogger("hivedscheduler.agent")


def _rocm_smi_gpu_state() -> Dict[int, bool]:
    """GPU liveness via rocm-smi JSON (a GPU missing or erroring is bad)."""
    try:
        out = subprocess.run(["rocm-smi", "--showuse", "--json"], capture_output=True,
                             text=True, timeout=60)
        if out.returncode != 0:
            return {}
        data = json.loads(out.stdout)
        state = {}
        for key in data:
            if key.lower().startswith("card"):
                idx = int(key[4:])
                state[idx] = True
        return state
    except Exception as e:
        log.warning("rocm-smi query failed: %s", e)
        return {}


def collect_node_health(
    deep: bool = False,
    min_hbm_gbps: float = 2000.0,
    min_pair_busbw_gbps: float = 50.0,
    probe_pairs: bool = False,
    sweep: bool = False,
) -> dict:
    """One health sweep over all visible GPUs. deep=True also runs the HIP
    kernels (HBM + MFMA); sweep=True adds the 16 GiB HBM stuck-bit pattern
    sweep (any error => leaf bad); probe_pairs=True RCCL-probes each xGMI
    pair.

    The stuck-bit scan transiently holds up to 16 GiB of HBM; on a GPU
    running tenant jobs this can make the tenant's own allocations fail.
    Keep sweep on the low-frequency cadence and prefer idle GPUs (the
    scheduler knows which leaves are unallocated).
    """
    report: dict = {"node": socket.gethostname(), "time": time.time(), "gpus": {}}
    alive = _rocm_smi_gpu_state()
    try:
        import torch

        n = torch.cuda.device_count() if torch.cuda.is_available() else 0
    except Exception:
        n = 0
    n = max(n, len(alive))
    for i in range(n):
        gpu = {"healthy": alive.get(i, True)}
        report["gpus"][str(i)] = gpu
    if deep and n > 0:
        from ..ops import gpu_health_report

        for i in range(n):
            try:
                # best-of-3 bandwidth: a tenant job on the GPU can halve a
                # single triad sample (see profiles/interference_r01.md)
                rep = gpu_health_report(i, quick=True, deep=sweep, bw_samples=3)
                report["gpus"][str(i)].update(
                    hbm_gbps=round(rep["hbm_gbps"], 1),
                    mfma_ok=rep["mfma_ok"],
                    cu_coverage=rep["cu_coverage"],
                    lds_errors=rep["lds_errors"],
                )
                if "hbm_sweep" in rep:
                    report["gpus"][str(i)]["hbm_sweep_errors"] = rep["hbm_sweep"]["errors"]
                if not rep["healthy"] or rep["hbm_gbps"] < min_hbm_gbps:
                    report["gpus"][str(i)]["healthy"] = False
            except Exception as e:
                report["gpus"][str(i)].update(healthy=False, error=str(e)[:200])
    if probe_pairs and n >= 2:
        from ..probe import CellProbeRunner

        runner = CellProbeRunner(min_busbw_gbps=min_pair_busbw_gbps)
        if runner.available():
            report["pairs"] = {}
            links = []
            for p in range(n // 2):
                pair = [2 * p, 2 * p + 1]
                try:
                    res = runner.probe_cell(pair, size_mb=64, iters=10)
                    report["pairs"][f"{pair[0]}-{pair[1]}"] = res
                    if res.get("ok"):
                        # first-class LINK report: a degraded xGMI link marks
                        # the link (scheduler avoids co-placing the endpoints)
                        # while both GPUs stay schedulable for 1-GPU work
                        links.append({"a": pair[0], "b": pair[1],
                                      "healthy": bool(res.get("healthy", True)),
                                      "gbps": float(res.get("busbw_gbps") or 0.0)})
                except Exception as e:
                    report["pairs"][f"{pair[0]}-{pair[1]}"] = {"ok": False, "error": str(e)[:200]}
            if links:
                report["links"] = links
    return report


def p2p_link_matrix(n: Optional[int] = None, size_mb: int = 64, iters: int = 5,
                    min_link_gbps: float = 40.0) -> dict:
    """Measure the full xGMI p2p bandwidth matrix with the HIP copy kernel
    (ops.p2p_gbps) and derive per-link health. This is the cheapest way to
    localize a degraded link to ONE pair when a >=4-GPU collective probe
    reads low: every GPU pair on an 8x MI355X node is directly connected
    (7 links x ~153 GB/s per direction), so matrix[i][j] well below the
    floor indicts exactly link i<->j.

    Returns {"matrix": {"i-j": gbps}, "links": [{a, b, gbps, healthy}]}
    shaped for the scheduler's /v1/health/nodes intake.
    """
    import torch

    from ..ops import get_ops

    ops = get_ops()
    count = n if n is not None else (torch.cuda.device_count() if torch.cuda.is_available() else 0)
    out: dict = {"matrix": {}, "links": []}
    for i in range(count):
        for j in range(i + 1, count):
            try:
                # per-direction copies; a link is as sick as its worse direction
                fwd = ops.p2p_gbps(i, j, size_mb, iters)
                rev = ops.p2p_gbps(j, i, size_mb, iters)
                gbps = min(fwd, rev)
            except Exception as e:
                log.warning("p2p probe %d<->%d failed: %s", i, j, e)
                out["links"].append({"a": i, "b": j, "healthy": False, "gbps": 0.0,
                                     "error": str(e)[:200]})
                continue
            out["matrix"][f"{i}-{j}"] = round(gbps, 1)
            out["links"].append({"a": i, "b": j, "healthy": bool(gbps >= min_link_gbps),
                                 "gbps": round(gbps, 1)})
    return out


class NodeHealthAgent:
    """Periodic health loop posting reports to the scheduler."""

    def __init__(self, scheduler_url: str, node_name: Optional[str] = None,
                 interval_s: float = 60.0, deep_every: int = 10,
                 sweep_every: int = 60, probe_pairs: bool = True,
                 p2p_matrix_every: int = 30):
        self.scheduler_url = scheduler_url.rstrip("/")
        self.node_name = node_name or socket.gethostname()
        self.interval_s = interval_s
        self.deep_every = deep_every
        # stuck-bit HBM sweep cadence (GPU must be idle-ish; 16 GiB ~ 5 s/GPU)
        self.sweep_every = sweep_every
        # pair probes default ON: the xGMI checks are the point of the agent
        self.probe_pairs = probe_pairs
        # full p2p matrix cadence (28 pairs x 2 directions, ~1 min on 8 GPUs):
        # localizes a degraded link to one pair when collective probes read low
        self.p2p_matrix_every = p2p_matrix_every
        self._rounds = 0

    def post_report(self, report: dict) -> bool:
        import requests

        url = f"{self.scheduler_url}/v1/health/nodes/{self.node_name}"
        try:
            r = requests.post(url, json=report, timeout=30)
            return r.status_code < 300
        except Exception as e:
            log.warning("health report post failed: %s", e)
            return False

    def run_placement_probes(self) -> List[dict]:
        """Poll post-bind probe tasks for this node, run the native RCCL
        probe over each placement's GPUs, and post the results back."""
        import requests

        from ..probe import CellProbeRunner

        url = f"{self.scheduler_url}/v1/health/probes/{self.node_name}"
        try:
            tasks = requests.get(url, timeout=30).json()
        except Exception as e:
            log.warning("probe poll failed: %s", e)
            return []
        runner = CellProbeRunner()
        results = []
        for task in tasks:
            cells = [int(i) for i in task.get("leafCellIndices", [])]
            if runner.available():
                res = runner.probe_cell(cells, size_mb=64, iters=10)
            else:
                res = {"ok": False, "error": "rccl-cell-probe binary missing"}
            res.update(group=task.get("group", ""), node=self.node_name,
                       leafCellIndices=cells)
            try:
                requests.post(f"{self.scheduler_url}/v1/health/probes", json=res, timeout=30)
            except Exception as e:
                log.warning("probe result post failed: %s", e)
            results.append(res)
        return results

    def run_once(self) -> dict:
        deep = (self._rounds % self.deep_every) == 0
        sweep = (self._rounds % self.sweep_every) == 0
        matrix = (self._rounds % self.p2p_matrix_every) == 0
        self._rounds += 1
        report = collect_node_health(deep=deep, probe_pairs=self.probe_pairs and deep,
                                     sweep=sweep and deep)
        report["node"] = self.node_name
        if matrix and deep:
            try:
                m = p2p_link_matrix()
                if m["links"]:
                    report["p2p_matrix"] = m["matrix"]
                    # matrix verdicts override pair-probe verdicts (finer)
                    report["links"] = m["links"]
            except Exception as e:
                log.warning("p2p matrix sweep failed: %s", e)
        self.post_report(report)
        report["placement_probes"] = self.run_placement_probes()
        return report

    def run_forever(self) -> None:  # pragma: no cover
        while True:
            try:
                self.run_once()
            except Exception as e:
                log.error("health sweep failed: %s", e)
            time.sleep(self.interval_s)
