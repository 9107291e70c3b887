"""SpMM roofline measurement.

The ChebConv recurrence step is a gather-bound SpMM: for each (row, c) it
reads the row's CSR segment and gathers x[col, :] rows. Its bandwidth
ceiling is NOT the dense STREAM rate — random row gathers fetch whole
cache lines per touched row — so we compare against a measured GATHER
ceiling (torch.index_select over the same column distribution) as well as
a dense copy ceiling, at the deep/large BASELINE shapes.

Usage (GPU): python bench/spmm_roofline.py
Prints one JSON line per shape; paste into profiles/r02_spmm_roofline.md.
"""
from __future__ import annotations

import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from stmgcn_amd.data.synthetic import make_synthetic_dataset
from stmgcn_amd.graph import SupportGenerator
from stmgcn_amd.ops.functional import require_hip


def timeit(fn, iters=30, warmup=5):
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


def main():
    C = require_hip()
    dev = torch.device("cuda")
    # (name, N, B, channels, cheby_K) — deep-4096 (fp16) / large-16384 (bf16)
    cases = [("deep-4096", 4096, 16, 64, 2, torch.float16),
             ("large-16384", 16384, 8, 64, 3, torch.bfloat16)]
    for name, N, B, ch, K, dtype in cases:
        raw = make_synthetic_dataset(n_nodes=N, n_steps=8, m_graphs=3, seed=1,
                                     day_timesteps=1)
        gen = SupportGenerator("chebyshev", K)
        # the random "semantic" graph is the worst locality case
        csr = gen.process_csr(torch.from_numpy(raw["semantic_adj"]).float()).to(dev)
        x = torch.randn(B, N, ch, device=dev, dtype=dtype)
        us = timeit(lambda: C.spmm_axpby(x, None, csr.row_ptr, csr.col_idx,
                                         csr.vals, 1.0, 0.0))
        nnz = int(csr.col_idx.numel())
        # bytes actually requested: out write + per-nnz gather of a ch-row
        # segment + csr metadata (col idx + val per nnz)
        req_gb = (B * N * ch * 2 + nnz * B * ch * 2 + nnz * 8) / 1e9
        # gather ceiling: index_select of the same rows from the same x
        idx = csr.col_idx.long()
        xg = x.reshape(B * N, ch)
        gus = timeit(lambda: torch.index_select(xg, 0, idx))
        g_gb = (nnz * ch * 2 * 2) / 1e9   # read + write per gathered row
        # dense copy ceiling
        y = torch.empty_like(x)
        cus = timeit(lambda: y.copy_(x))
        c_gb = B * N * ch * 2 * 2 / 1e9
        print(json.dumps({
            "case": name, "N": N, "B": B, "ch": ch, "K_supports": csr.K_supports,
            "nnz": nnz, "avg_degree": round(nnz / N, 1),
            "spmm_us": round(us, 1),
            "spmm_eff_tb_s": round(req_gb / (us * 1e-6) / 1e3, 2),
            "gather_ceiling_tb_s": round(g_gb / (gus * 1e-6) / 1e3, 2),
            "dense_copy_tb_s": round(c_gb / (cus * 1e-6) / 1e3, 2),
            "pct_of_gather_ceiling": round(
                100 * (req_gb / us) / (g_gb / gus) * (1), 1),
        }))


if __name__ == "__main__":
    main()
