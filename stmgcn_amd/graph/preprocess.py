"""Adjacency -> graph-convolution supports, CSR-first.

Behavioral parity target: the reference's Adj_Preprocessor (GCN.py:50-135),
which materializes a dense (K_supports, N, N) stack offline:

  chebyshev:  A_hat = D^-1/2 A D^-1/2 ; L = I - A_hat ; L~ = (2/lmax) L - I ;
              supports = Chebyshev polynomials T_0..T_K of L~      (GCN.py:66-75)
  localpool:  single support I + A_hat                             (GCN.py:68-70)
  random_walk_diffusion: supports = T_0..T_K of P^T, P = D^-1 A    (GCN.py:77-81)
  recurrence: T_0 = I, T_1 = G, T_k = 2 G T_{k-1} - T_{k-2}        (GCN.py:125-135)

The reference's torch.eig call raises on torch>=1.10, so lambda_max == 2 is
ALWAYS used (GCN.py:116-121). We default to that observable
behavior ("fixed2") and offer a real power-iteration mode.

MI355X-native difference: at N=16384 a dense (K+1, N, N) fp32 stack is 3.2 GB
per graph and O(N^3) to build. We therefore keep only the sparse
GENERATOR matrix G in CSR and run the K-hop recurrence inside the HIP ChebConv
kernel; dense stacks remain available for parity tests and the CPU oracle.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional

import torch


def symmetric_normalize(adj: torch.Tensor) -> torch.Tensor:
    """D^-1/2 A D^-1/2 with isolated-node guard (deg 0 -> row/col of zeros).

    The reference (GCN.py:108-111) has no inf guard here — an isolated node
    produces inf/nan. We guard (the rw normalizer in the reference does,
    GCN.py:100-105); for graphs without isolated nodes the output is identical.
    """
    deg = adj.sum(dim=1)
    d_inv_sqrt = deg.float().pow(-0.5)
    d_inv_sqrt[torch.isinf(d_inv_sqrt)] = 0.0
    return adj * d_inv_sqrt.unsqueeze(1) * d_inv_sqrt.unsqueeze(0)


def random_walk_normalize(adj: torch.Tensor) -> torch.Tensor:
    """P = D^-1 A (reference GCN.py:100-105, inf guard included there)."""
    d_inv = adj.sum(dim=1).float().pow(-1.0)
    d_inv[torch.isinf(d_inv)] = 0.0
    return adj * d_inv.unsqueeze(1)


def power_iteration_lmax(L: torch.Tensor, iters: int = 500, tol: float = 1e-9) -> float:
    """Largest eigenvalue of symmetric PSD L by power iteration (the honest
    replacement for the reference's dead torch.eig path). Converges from
    below; exit requires the Rayleigh quotient to stall for 3 consecutive
    iterations (a single small delta is NOT convergence when the spectral
    gap is tiny — the estimate can still be several percent short)."""
    v = torch.randn(L.shape[0], generator=torch.Generator().manual_seed(0), dtype=L.dtype)
    v = v / v.norm()
    lam = 0.0
    stall = 0
    for _ in range(iters):
        w = L @ v
        nw = w.norm()
        if nw == 0:
            return 0.0
        v_new = w / nw
        lam_new = float(v_new @ (L @ v_new))
        stall = stall + 1 if abs(lam_new - lam) < tol else 0
        lam, v = lam_new, v_new
        if stall >= 3:
            break
    return lam


@dataclass
class CSRSupport:
    """Sparse generator matrix G for the in-kernel support recurrence.

    kind == "cheby":  supports are T_0=I, T_1=G, T_k = 2 G T_{k-1} - T_{k-2},
                      K_s = K+1 of them (chebyshev & rw-diffusion families).
    kind == "single": one support, the matrix G itself (localpool).

    Carries BOTH G and G^T in CSR: backward applies T_k(G)^T = T_k(G^T)
    (rw-diffusion's P^T is asymmetric; for the symmetric chebyshev/localpool
    generators the two are identical but stored uniformly).
    """
    row_ptr: torch.Tensor   # int32 (N+1,)
    col_idx: torch.Tensor   # int32 (nnz,)
    vals: torch.Tensor      # fp32  (nnz,)
    n_nodes: int
    K_supports: int
    kind: str               # "cheby" | "single"
    row_ptr_t: Optional[torch.Tensor] = None   # CSR of G^T
    col_idx_t: Optional[torch.Tensor] = None
    vals_t: Optional[torch.Tensor] = None

    def to(self, device) -> "CSRSupport":
        return CSRSupport(self.row_ptr.to(device), self.col_idx.to(device),
                          self.vals.to(device), self.n_nodes, self.K_supports,
                          self.kind,
                          None if self.row_ptr_t is None else self.row_ptr_t.to(device),
                          None if self.col_idx_t is None else self.col_idx_t.to(device),
                          None if self.vals_t is None else self.vals_t.to(device))

    @property
    def nnz(self) -> int:
        return self.col_idx.numel()

    def dense_generator(self) -> torch.Tensor:
        G = torch.zeros(self.n_nodes, self.n_nodes, dtype=self.vals.dtype)
        for i in range(self.n_nodes):
            s, e = int(self.row_ptr[i]), int(self.row_ptr[i + 1])
            G[i, self.col_idx[s:e].long()] = self.vals[s:e]
        return G

    def dense_supports(self) -> torch.Tensor:
        """Materialize the (K_s, N, N) stack — parity/oracle use only."""
        G = self.dense_generator()
        if self.kind == "single":
            return G.unsqueeze(0)
        return torch.stack(chebyshev_polynomials(G, self.K_supports - 1), dim=0)


def chebyshev_polynomials(G: torch.Tensor, K: int) -> List[torch.Tensor]:
    """T_0..T_K of G via the first-kind recurrence (reference GCN.py:125-135)."""
    N = G.shape[0]
    out: List[torch.Tensor] = [torch.eye(N, dtype=G.dtype, device=G.device)]
    if K >= 1:
        out.append(G)
    for k in range(2, K + 1):
        out.append(2.0 * (G @ out[k - 1]) - out[k - 2])
    return out


def _csr_arrays(G: torch.Tensor, prune_eps: float):
    mask = G.abs() > prune_eps
    N = G.shape[0]
    counts = mask.sum(dim=1, dtype=torch.int32)
    row_ptr = torch.zeros(N + 1, dtype=torch.int32)
    row_ptr[1:] = torch.cumsum(counts, dim=0).to(torch.int32)
    idx = mask.nonzero(as_tuple=False)  # sorted row-major -> cols sorted per row
    col_idx = idx[:, 1].to(torch.int32).contiguous()
    vals = G[mask].to(torch.float32).contiguous()
    return row_ptr, col_idx, vals


def dense_to_csr(G: torch.Tensor, prune_eps: float = 0.0) -> CSRSupport:
    """Dense (N,N) -> CSR of G and of G^T (int32 indices, fp32 values).
    Rows kept sorted by column for coalesced in-kernel gathers."""
    rp, ci, v = _csr_arrays(G, prune_eps)
    rpt, cit, vt = _csr_arrays(G.T.contiguous(), prune_eps)
    return CSRSupport(rp, ci, v, G.shape[0], 0, "", rpt, cit, vt)


class SupportGenerator:
    """MI355X-native replacement for the reference Adj_Preprocessor.

    process(adj)      -> dense (K_supports, N, N) stack (reference parity)
    process_csr(adj)  -> CSRSupport of the generator matrix for the HIP kernel
    """

    def __init__(self, kernel_type: str, K: int, lambda_max_mode: str = "fixed2"):
        if kernel_type not in ("chebyshev", "localpool", "random_walk_diffusion"):
            raise ValueError(
                "kernel_type must be one of [chebyshev, localpool, random_walk_diffusion]")
        self.kernel_type = kernel_type
        # localpool ignores K (reference GCN.py:54)
        self.K = K if kernel_type != "localpool" else 1
        self.lambda_max_mode = lambda_max_mode

    # -- generator matrix ---------------------------------------------------
    def generator(self, adj: torch.Tensor) -> torch.Tensor:
        adj = adj.float()
        if self.kernel_type == "localpool":
            return torch.eye(adj.shape[0]) + symmetric_normalize(adj)
        if self.kernel_type == "chebyshev":
            L = torch.eye(adj.shape[0]) - symmetric_normalize(adj)
            if self.lambda_max_mode == "fixed2":
                lmax = 2.0  # the reference's only reachable path (quirk 1)
            elif self.lambda_max_mode == "power_iteration":
                # 1% spectral safety margin, clamped to the sym-normalized
                # Laplacian's hard bound of 2: power iteration approaches
                # lmax from BELOW, and an underestimate pushes the scaled
                # spectrum outside [-1,1] (Chebyshev bound violated) while a
                # slight overestimate merely compresses it.
                lmax = min(power_iteration_lmax(L) * 1.01, 2.0)
            else:
                raise ValueError(f"bad lambda_max_mode {self.lambda_max_mode!r}")
            return (2.0 / lmax) * L - torch.eye(L.shape[0])
        # random_walk_diffusion: Chebyshev series of P^T (reference GCN.py:77-81)
        return random_walk_normalize(adj).T.contiguous()

    # -- dense supports (parity surface) ------------------------------------
    def process(self, adj: torch.Tensor) -> torch.Tensor:
        G = self.generator(adj)
        if self.kernel_type == "localpool":
            return G.unsqueeze(0)
        return torch.stack(chebyshev_polynomials(G, self.K), dim=0)

    # -- CSR generator (HIP kernel surface) ----------------------------------
    def process_csr(self, adj: torch.Tensor, prune_eps: float = 0.0) -> CSRSupport:
        G = self.generator(adj)
        csr = dense_to_csr(G, prune_eps)
        if self.kernel_type == "localpool":
            csr.K_supports, csr.kind = 1, "single"
        else:
            csr.K_supports, csr.kind = self.K + 1, "cheby"
        return csr


class Adj_Preprocessor(SupportGenerator):
    """Drop-in API-compatibility alias (reference class name, GCN.py:50)."""
    pass
