#!/usr/bin/env python3
"""Driver for rocprofv3 kernel profiling of the placement-validation path.

Runs, in one process, everything the scheduler executes on the GPU box after a
bind: the HIP health-probe kernels (MFMA check + HBM triad, ops/hived_ops.hip)
and the RCCL all-reduce cell probe over cuda:0 (probe/allreduce.py). rocprofv3
wraps this script to produce the per-kernel stats committed under profiles/.

Reference analog: HiveD has no GPU-side code at all; this is
the MI355X-native placement-validation layer mandated by BASELINE.json.
"""
import os
import sys

sys.path.insert(0, os.environ.get("GRAFT_REPO_ROOT", os.path.dirname(os.path.abspath(__file__))))


def main() -> None:
    import torch

    assert torch.cuda.is_available(), "needs a GPU"

    from hivedscheduler_amd.ops import gpu_health_report
    from hivedscheduler_amd.probe import allreduce_probe, CellProbeRunner

    rep = gpu_health_report(0, quick=False)
    print("health:", {k: rep[k] for k in ("hbm_gbps", "mfma_ok", "healthy")})

    # single-rank RCCL probe (ring over self) — exercises the exact rccl
    # kernels the post-bind validation runs per placed cell
    runner = CellProbeRunner()
    if runner.available():
        out = runner.probe_cell([0], size_mb=64, iters=10)
        print("rccl probe:", out)
    else:
        # fall back to in-process torch.distributed single-rank allreduce
        import torch.distributed as dist

        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29571")
        os.environ.setdefault("RANK", "0")
        os.environ.setdefault("WORLD_SIZE", "1")
        torch.cuda.set_device(0)
        dist.init_process_group("nccl")
        print("probe:", allreduce_probe(sizes_mb=(64,), iters=10, warmup=3))
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
