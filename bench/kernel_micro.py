"""Standalone kernel microbenchmarks (GPU) — per-kernel wall time via HIP
events, for tuning without running the whole model.

Usage: python bench/kernel_micro.py [--kernel lstm_wgrad|atb_wgrad|lstm|all]
Prints one JSON line per measurement.
"""
from __future__ import annotations

import argparse
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def timeit(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters * 1e3  # us


def bench_lstm_wgrad(C):
    # bench-1024 shape: S = 32*1024, T = 8, L = 3, cin = 1
    S, T, L, H = 32 * 1024, 8, 3, 64
    S_pad = (S + 63) // 64 * 64
    R = T * S_pad
    dA = torch.randn(L, T, S_pad, 4 * H, device="cuda", dtype=torch.bfloat16)
    hseq = torch.randn(L, T, S_pad, H, device="cuda", dtype=torch.bfloat16)
    x = torch.randn(S, T, 1, device="cuda", dtype=torch.bfloat16)
    us = timeit(lambda: C.lstm_wgrad(dA, hseq, x))
    gb = (dA.numel() + hseq.numel() + x.numel()) * 2 / 1e9
    print(json.dumps({"kernel": "lstm_wgrad", "us": round(us, 1),
                      "eff_tb_s": round(gb / (us * 1e-6) / 1e3, 2),
                      "chunks": os.environ.get("STMGCN_WGRAD_CHUNKS", "auto")}))


def bench_atb(C):
    for rows, M, N in [(32768, 192, 64), (32768, 24, 8)]:
        A = torch.randn(rows, M, device="cuda", dtype=torch.bfloat16)
        B = torch.randn(rows, N, device="cuda", dtype=torch.bfloat16)
        us = timeit(lambda: C.atb_wgrad(A, B, True))
        gb = (A.numel() + B.numel()) * 2 / 1e9
        print(json.dumps({"kernel": f"atb_{M}x{N}", "us": round(us, 1),
                          "eff_tb_s": round(gb / (us * 1e-6) / 1e3, 2)}))
        # library comparison
        us2 = timeit(lambda: (A.float().T @ B.float(), B.sum(0)))
        print(json.dumps({"kernel": f"atb_{M}x{N}_torch", "us": round(us2, 1)}))


def bench_lstm(C):
    from stmgcn_amd.ops.hip_ops import FusedLSTMFn
    S, T, L = 32 * 1024, 8, 3
    x = torch.randn(S, T, 1, device="cuda", dtype=torch.bfloat16)
    ws = []
    for l in range(L):
        in_l = 1 if l == 0 else 64
        ws += [torch.randn(256, in_l, device="cuda", dtype=torch.bfloat16) * 0.1,
               torch.randn(256, 64, device="cuda", dtype=torch.bfloat16) * 0.1,
               torch.randn(256, device="cuda", dtype=torch.bfloat16) * 0.1,
               torch.randn(256, device="cuda", dtype=torch.bfloat16) * 0.1]
    us = timeit(lambda: FusedLSTMFn.apply(x, "lstm", False, False, *ws), iters=30)
    print(json.dumps({"kernel": "lstm_fwd_infer", "us": round(us, 1)}))

    def train_step():
        xg = x.clone().requires_grad_(True)
        wsg = [w.clone().requires_grad_(True) for w in ws]
        out = FusedLSTMFn.apply(xg, "lstm", False, True, *wsg)
        out.float().square().sum().backward()
    us = timeit(train_step, iters=20, warmup=5)
    print(json.dumps({"kernel": "lstm_fwd_bwd_wgrad", "us": round(us, 1)}))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--kernel", default="all")
    args = p.parse_args()
    from stmgcn_amd.ops.functional import require_hip
    C = require_hip()
    if args.kernel in ("lstm_wgrad", "all"):
        bench_lstm_wgrad(C)
    if args.kernel in ("atb_wgrad", "all"):
        bench_atb(C)
    if args.kernel in ("lstm", "all"):
        bench_lstm(C)


if __name__ == "__main__":
    main()
