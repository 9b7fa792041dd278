"""HIP (gfx950) GPU health-probe kernels.

Import `get_ops()` to obtain the compiled extension. On a GPU box the native
extension is REQUIRED: failure to load raises (no silent eager fallback).
"""
from __future__ import annotations

import os
import sys

_ops = None


def get_ops():
    """Load the hived_ops HIP extension, building it if necessary."""
    global _ops
    if _ops is None:
        from .build import build

        _ops = build(verbose=False)
    return _ops


def gpu_health_report(device: int = 0, quick: bool = True, deep: bool = False,
                      bw_samples: int = 1) -> dict:
    """Run the health-probe kernels on one GPU and return measured facts.

    Feeds leaf-cell healthiness: HBM bandwidth deficit, MFMA mismatch, or
    (deep=True) any stuck-bit error in the HBM pattern sweep marks the GPU's
    leaf cell bad. The deep sweep scans a bounded span (16 GiB quick / 64 GiB
    otherwise) of the 288 GB HBM3E in 4 GiB chunks.

    bw_samples: take the BEST of N triad samples. On a GPU busy with a tenant
    job a single sample can read under half the real bandwidth (measured:
    1.53 TB/s vs 4.6 TB/s idle on the same healthy GPU — profiles/
    interference_r01.md); genuinely sick HBM is low in EVERY sample. Agents
    probing live nodes should use bw_samples>=3.
    """
    import torch

    ops = get_ops()
    torch.cuda.set_device(device)
    size_mb = 256 if quick else 2048
    iters = 5 if quick else 20
    report = {"device": device, "info": ops.device_info(device)}
    samples = [ops.hbm_triad_gbps(size_mb, iters) for _ in range(max(1, bw_samples))]
    report["hbm_gbps"] = max(samples)
    if len(samples) > 1:
        report["hbm_gbps_samples"] = [round(s, 1) for s in samples]

    torch.manual_seed(0)
    A = (torch.randn(16, 32, dtype=torch.float32) / 8).bfloat16().cuda(device)
    B = (torch.randn(32, 16, dtype=torch.float32) / 8).bfloat16().cuda(device)
    tiles = ops.mfma_check(A, B, 2048, 1)  # 8192 waves >> 256 CUs
    ref = (A.float() @ B.float())
    max_err = (tiles - ref.unsqueeze(0)).abs().max().item()
    tile_spread = (tiles - tiles[0].unsqueeze(0)).abs().max().item()
    report["mfma_max_err_vs_fp32"] = max_err
    report["mfma_cross_cu_spread"] = tile_spread  # must be exactly 0
    report["mfma_ok"] = bool(tile_spread == 0.0 and max_err < 0.1)
    # Low-precision pipes (MI355X's headline datapaths): fp8 e4m3 MFMA, the
    # MX block-scaled fp8 path (K=128), and MX fp4. A GPU whose bf16 pipes
    # work but whose fp8/fp4 units are broken passes the bf16 check alone.
    for key, fn in (("mfma_fp8", _mfma_fp8_probe), ("mfma_mx8", _mfma_mx_probe_fp8),
                    ("mfma_fp4", _mfma_mx_probe_fp4)):
        err, spread = fn(ops, device)
        report[f"{key}_max_err_vs_fp32"] = err
        report[f"{key}_cross_cu_spread"] = spread
        report[f"{key}_ok"] = bool(spread == 0.0 and err < 0.1)
    report["mfma_lowprec_ok"] = bool(report["mfma_fp8_ok"] and report["mfma_mx8_ok"]
                                     and report["mfma_fp4_ok"])
    # CU coverage: a big grid must place waves on every CU of every XCD; a
    # fused-off / hung CU shows up as missing (xcc, se, sh, cu) tuples
    words = ops.cu_coverage(4096).cpu().numpy()
    cus = {(int(w) >> 16, (int(w) >> 13) & 0x7, (int(w) >> 12) & 0x1, (int(w) >> 8) & 0xF)
           for w in words}
    report["cu_coverage"] = len(cus)
    expected_cus = report["info"]["multiProcessorCount"]
    report["cu_coverage_ok"] = bool(len(cus) >= expected_cus)
    report["lds_errors"] = int(ops.lds_check(2048, 7))
    report["healthy"] = bool(report["mfma_ok"] and report["mfma_lowprec_ok"]
                             and report["cu_coverage_ok"]
                             and report["lds_errors"] == 0
                             and report["hbm_gbps"] > 1000.0)
    if deep:
        sweep = ops.hbm_sweep(16 if quick else 64, 4, 1)
        report["hbm_sweep"] = sweep
        # A skipped (zero-byte) sweep is not evidence of health, but neither
        # is it evidence of sickness (a busy tenant GPU can leave too little
        # free HBM): record it, don't flip `healthy` on it.
        if not sweep.get("skipped"):
            report["healthy"] = bool(report["healthy"] and sweep["errors"] == 0)
    return report


_FP4_E2M1_VALUES = [0.0, 0.5, 1.0, 1.5, 2.0, 3.0, 4.0, 6.0,
                    -0.0, -0.5, -1.0, -1.5, -2.0, -3.0, -4.0, -6.0]


def _mfma_fp8_probe(ops, device):
    """fp8 e4m3 16x16x32 MFMA vs a torch fp32 reference decoded from the
    same raw e4m3 bytes. Returns (max_err, cross_cu_spread)."""
    import torch

    torch.manual_seed(1)
    A8 = (torch.randn(16, 32) / 8).to(torch.float8_e4m3fn).cuda(device)
    B8 = (torch.randn(32, 16) / 8).to(torch.float8_e4m3fn).cuda(device)
    tiles = ops.mfma_check_fp8(A8.view(torch.uint8), B8.view(torch.uint8), 2048)
    ref = A8.float() @ B8.float()
    return ((tiles - ref.unsqueeze(0)).abs().max().item(),
            (tiles - tiles[0].unsqueeze(0)).abs().max().item())


def _mfma_mx_probe_fp8(ops, device):
    """MX block-scaled fp8 (16x16x128 f8f6f4, fmt=0) with unit scales."""
    import torch

    torch.manual_seed(2)
    A8 = (torch.randn(16, 128) / 16).to(torch.float8_e4m3fn).cuda(device)
    B8 = (torch.randn(128, 16) / 16).to(torch.float8_e4m3fn).cuda(device)
    tiles = ops.mfma_check_mx(A8.view(torch.uint8), B8.view(torch.uint8), 2048, 0)
    ref = A8.float() @ B8.float()
    return ((tiles - ref.unsqueeze(0)).abs().max().item(),
            (tiles - tiles[0].unsqueeze(0)).abs().max().item())


def _mfma_mx_probe_fp4(ops, device):
    """MX block-scaled fp4 (e2m1 codes, fmt=4) with unit scales. Values are
    exactly representable and the K=128 dot products stay small, so the fp32
    reference matches bit-for-bit on healthy hardware."""
    import torch

    g = torch.Generator().manual_seed(3)
    A4 = torch.randint(0, 16, (16, 128), generator=g, dtype=torch.uint8).cuda(device)
    B4 = torch.randint(0, 16, (128, 16), generator=g, dtype=torch.uint8).cuda(device)
    lut = torch.tensor(_FP4_E2M1_VALUES, dtype=torch.float32).cuda(device)
    tiles = ops.mfma_check_mx(A4, B4, 2048, 4)
    ref = lut[A4.long()] @ lut[B4.long()]
    return ((tiles - ref.unsqueeze(0)).abs().max().item(),
            (tiles - tiles[0].unsqueeze(0)).abs().max().item())
