"""Synthetic region-demand data.

The reference expects ./data/data_dict.npz with keys taxi / neighbor_adj /
trans_adj / semantic_adj (Data_Container.py:16-28) but ships no data and no
generator. BASELINE.json mandates synthetic tensors + random-init weights, so
this module is the canonical source: demand with daily+weekly periodic
structure, a grid neighbor graph, and two random sparse symmetric graphs.
All graphs are guaranteed free of isolated nodes (symmetric normalization of
an isolated node is NaN in the reference, GCN.py:108-111).
"""
from __future__ import annotations

import math
import os
from typing import Dict

import numpy as np


def _grid_neighbor_adj(n_nodes: int) -> np.ndarray:
    """4-neighbor grid adjacency on the most-square factorization of N."""
    rows = int(math.sqrt(n_nodes))
    while n_nodes % rows != 0:
        rows -= 1
    cols = n_nodes // rows
    A = np.zeros((n_nodes, n_nodes), dtype=np.float32)
    for r in range(rows):
        for c in range(cols):
            i = r * cols + c
            if r + 1 < rows:
                j = (r + 1) * cols + c
                A[i, j] = A[j, i] = 1.0
            if c + 1 < cols:
                j = r * cols + (c + 1)
                A[i, j] = A[j, i] = 1.0
    return A


def _random_sparse_sym_adj(n_nodes: int, avg_degree: int, rng: np.random.Generator,
                           weighted: bool = False) -> np.ndarray:
    """Symmetric random graph with ~avg_degree edges per node, no isolated nodes."""
    A = np.zeros((n_nodes, n_nodes), dtype=np.float32)
    n_edges = n_nodes * avg_degree // 2
    src = rng.integers(0, n_nodes, size=n_edges)
    dst = rng.integers(0, n_nodes, size=n_edges)
    w = rng.uniform(0.1, 1.0, size=n_edges).astype(np.float32) if weighted else np.ones(n_edges, np.float32)
    keep = src != dst
    A[src[keep], dst[keep]] = w[keep]
    A = np.maximum(A, A.T)
    # ring fallback guarantees no isolated node
    ring = np.arange(n_nodes)
    A[ring, (ring + 1) % n_nodes] = np.maximum(A[ring, (ring + 1) % n_nodes], 1.0)
    A[(ring + 1) % n_nodes, ring] = A[ring, (ring + 1) % n_nodes]
    return A


def make_synthetic_dataset(n_nodes: int = 58, n_steps: int = 24 * 365,
                           m_graphs: int = 3, seed: int = 0,
                           day_timesteps: int = 24) -> Dict[str, np.ndarray]:
    """Demand tensor (T, N, 1) with daily/weekly periodicity plus M adjacency
    matrices, shaped like the reference's expected data_dict.npz."""
    rng = np.random.default_rng(seed)
    t = np.arange(n_steps, dtype=np.float32)[:, None]          # (T,1)
    phase = rng.uniform(0, 2 * np.pi, size=(1, n_nodes)).astype(np.float32)
    base = rng.uniform(20.0, 200.0, size=(1, n_nodes)).astype(np.float32)
    daily = 0.5 * np.sin(2 * np.pi * t / day_timesteps + phase)
    weekly = 0.25 * np.sin(2 * np.pi * t / (7 * day_timesteps) + 2 * phase)
    noise = 0.1 * rng.standard_normal((n_steps, n_nodes)).astype(np.float32)
    taxi = base * (1.0 + daily + weekly + noise)
    taxi = np.clip(taxi, 0.0, None).astype(np.float32)[..., None]  # (T,N,1)

    out: Dict[str, np.ndarray] = {"taxi": taxi}
    keys = ["neighbor_adj", "trans_adj", "semantic_adj"]
    makers = [
        lambda: _grid_neighbor_adj(n_nodes),
        lambda: _random_sparse_sym_adj(n_nodes, avg_degree=8, rng=rng, weighted=True),
        lambda: _random_sparse_sym_adj(n_nodes, avg_degree=16, rng=rng, weighted=True),
    ]
    for k in range(m_graphs):
        out[keys[k]] = makers[k]()
    return out


def write_synthetic_npz(path: str, **kwargs) -> str:
    os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
    np.savez_compressed(path, **make_synthetic_dataset(**kwargs))
    return path
