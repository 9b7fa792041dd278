"""RCCL-over-xGMI all-reduce placement probe.

After the scheduler binds an affinity group, the probe validates the placement
on the live node: an all-reduce over the group's GPUs measures bus bandwidth;
a result far below the xGMI expectation (7 links x ~153 GB/s per GPU,
all-reduce is per-link bound) indicates a degraded link, and the pair/quad
cell gets marked bad.

Two entry points:
- allreduce_probe(): inside an initialized torch.distributed process group
  (backend "nccl" IS RCCL on ROCm); also works on gloo for CPU-only tests.
- CellProbeRunner: spawns the native single-process probe binary
  (native/rccl-cell-probe) over an isolated GPU set (HIP_VISIBLE_DEVICES).
"""
from __future__ import annotations

import json
import os
import subprocess
import time
from typing import Dict, List, Optional


def busbw_from_algbw(algbw_gbps: float, world: int) -> float:
    """Ring all-reduce bus bandwidth: busbw = algbw * 2*(n-1)/n."""
    if world <= 1:
        return algbw_gbps
    return algbw_gbps * 2.0 * (world - 1) / world


def allreduce_probe(
    sizes_mb: List[int] = (16, 64, 256),
    iters: int = 20,
    warmup: int = 5,
    device: Optional[str] = None,
    group=None,
) -> Dict[str, dict]:
    """Measure all-reduce algbw/busbw for each size on the current process
    group. Returns {size_mb: {algbw_gbps, busbw_gbps, ms}}."""
    import torch
    import torch.distributed as dist

    assert dist.is_initialized(), "torch.distributed must be initialized"
    world = dist.get_world_size(group)
    use_cuda = torch.cuda.is_available() and (device is None or device.startswith("cuda"))
    dev = torch.device(device or ("cuda" if use_cuda else "cpu"))
    results: Dict[str, dict] = {}
    for size_mb in sizes_mb:
        numel = size_mb * 1024 * 1024 // 4
        t = torch.ones(numel, dtype=torch.float32, device=dev)
        for _ in range(warmup):
            dist.all_reduce(t, group=group)
        if use_cuda:
            torch.cuda.synchronize()
        dist.barrier(group)
        t0 = time.perf_counter()
        for _ in range(iters):
            dist.all_reduce(t, group=group)
        if use_cuda:
            torch.cuda.synchronize()
        elapsed = time.perf_counter() - t0
        ms = elapsed / iters * 1e3
        bytes_ = numel * 4
        algbw = bytes_ / (elapsed / iters) / 1e9
        results[str(size_mb)] = {
            "size_mb": size_mb,
            "ms": ms,
            "algbw_gbps": algbw,
            "busbw_gbps": busbw_from_algbw(algbw, world),
            "world": world,
        }
    return results


class CellProbeRunner:
    """Runs the native rccl-cell-probe binary over a scheduler-placed cell
    (set of leaf/GPU indices on one node) and classifies link health."""

    def __init__(self, binary: Optional[str] = None, min_busbw_gbps: float = 50.0):
        if binary is None:
            binary = os.path.join(
                os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__)))),
                "native", "rccl-cell-probe")
        self.binary = binary
        self.min_busbw_gbps = min_busbw_gbps

    def available(self) -> bool:
        return os.path.exists(self.binary) and os.access(self.binary, os.X_OK)

    def probe_cell(self, leaf_indices: List[int], size_mb: int = 64, iters: int = 20,
                   timeout_s: float = 300.0, p2p_matrix: bool = False) -> dict:
        """All-reduce over the given GPU indices (one node). Returns the
        parsed probe JSON plus a health verdict. A hung probe (RCCL init can
        take minutes on a cold box; a truly wedged link hangs forever) is
        killed at timeout_s and reported as not-ok rather than raising — the
        agent loop must survive it.

        p2p_matrix=True also measures every pair's p2p copy bandwidth
        (worse direction), localizing a low collective busbw to specific
        links: the result gains "p2p_matrix" ({"i-j": gbps} in VISIBLE-device
        ordinals) and "suspect_links" ([[leafA, leafB, gbps], ...] in the
        caller's leaf indices) for pairs below min_busbw_gbps.
        """
        env = dict(os.environ)
        env["HIP_VISIBLE_DEVICES"] = ",".join(str(i) for i in leaf_indices)
        cmd = [self.binary, "--size-mb", str(size_mb), "--iters", str(iters)]
        if p2p_matrix and len(leaf_indices) > 1:
            cmd.append("--p2p-matrix")
        try:
            out = subprocess.run(cmd, env=env, capture_output=True, text=True,
                                 timeout=timeout_s)
        except subprocess.TimeoutExpired:
            return {"ok": False, "error": f"probe timed out after {timeout_s:.0f}s",
                    "timeout": True, "leaf_indices": leaf_indices}
        if out.returncode != 0:
            return {"ok": False, "error": out.stderr.strip()[-2000:], "leaf_indices": leaf_indices}
        result = json.loads(out.stdout.strip().splitlines()[-1])
        result["leaf_indices"] = leaf_indices
        result["ok"] = True
        n = len(leaf_indices)
        result["healthy"] = bool(n <= 1 or result.get("busbw_gbps", 0.0) >= self.min_busbw_gbps)
        if result.get("p2p_matrix"):
            suspects = []
            for key, gbps in result["p2p_matrix"].items():
                if gbps < self.min_busbw_gbps:
                    vi, vj = (int(x) for x in key.split("-"))
                    suspects.append([leaf_indices[vi], leaf_indices[vj], gbps])
            result["suspect_links"] = suspects
            if suspects:
                result["healthy"] = False
        return result
