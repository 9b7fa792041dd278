#!/usr/bin/env python3
"""Measure health-probe readings under tenant load.

The node agent runs its health kernels on GPUs that may be busy with tenant
jobs. This experiment measures what the probes report while a large bf16 GEMM
loop saturates the GPU from another thread/stream, to calibrate the agent's
health thresholds (a busy-GPU reading must not be classified as sick
hardware).

Writes one JSON line: idle vs loaded triad bandwidth, MFMA check integrity,
and the GEMM slowdown caused by probing.
"""
import json
import os
import sys
import threading
import time

sys.path.insert(0, os.environ.get("GRAFT_REPO_ROOT", os.path.dirname(os.path.abspath(__file__))))


def main() -> None:
    import torch

    from hivedscheduler_amd.ops import get_ops

    ops = get_ops()
    torch.cuda.set_device(0)

    # tenant workload: sustained bf16 GEMM on its own stream
    n = 8192
    a = torch.randn(n, n, dtype=torch.bfloat16, device="cuda")
    b = torch.randn(n, n, dtype=torch.bfloat16, device="cuda")
    stream = torch.cuda.Stream()
    stop = threading.Event()
    gemm_iters = [0]

    def tenant():
        with torch.cuda.stream(stream):
            while not stop.is_set():
                torch.mm(a, b)
                gemm_iters[0] += 1
                if gemm_iters[0] % 8 == 0:
                    stream.synchronize()
        stream.synchronize()

    # baselines (idle)
    idle_triad = ops.hbm_triad_gbps(512, 5)
    A = (torch.randn(16, 32) / 8).bfloat16().cuda()
    B = (torch.randn(32, 16) / 8).bfloat16().cuda()
    ref = A.float() @ B.float()

    def mfma_err():
        tiles = ops.mfma_check(A, B, 2048, 1)
        spread = (tiles - tiles[0].unsqueeze(0)).abs().max().item()
        err = (tiles[0] - ref.cuda()).abs().max().item()
        return err, spread

    idle_mfma_err, idle_spread = mfma_err()

    # tenant GEMM throughput alone
    t = threading.Thread(target=tenant)
    t.start()
    time.sleep(3)
    torch.cuda.synchronize()
    it0, t0 = gemm_iters[0], time.perf_counter()
    time.sleep(5)
    gemm_alone_ips = (gemm_iters[0] - it0) / (time.perf_counter() - t0)

    # probes under load
    loaded_triads = [ops.hbm_triad_gbps(512, 5) for _ in range(3)]
    loaded_mfma_err, loaded_spread = mfma_err()
    it1, t1 = gemm_iters[0], time.perf_counter()
    for _ in range(3):
        ops.hbm_triad_gbps(512, 5)
    gemm_probed_ips = (gemm_iters[0] - it1) / (time.perf_counter() - t1)

    stop.set()
    t.join()

    tflops = 2 * n**3 / 1e12
    out = {
        "idle_triad_gbps": round(idle_triad, 1),
        "loaded_triad_gbps": [round(x, 1) for x in loaded_triads],
        "idle_mfma": {"err": idle_mfma_err, "cross_cu_spread": idle_spread},
        "loaded_mfma": {"err": loaded_mfma_err, "cross_cu_spread": loaded_spread},
        "gemm_alone_tflops": round(gemm_alone_ips * tflops, 1),
        "gemm_during_probe_tflops": round(gemm_probed_ips * tflops, 1),
        "probe_slowdown_on_tenant_pct": round(100 * (1 - gemm_probed_ips / max(gemm_alone_ips, 1e-9)), 1),
    }
    print(json.dumps(out))


if __name__ == "__main__":
    main()
