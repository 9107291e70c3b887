"""Pure-PyTorch oracle implementations of every fused op.

These define the exact math the HIP kernels must reproduce (the numeric-parity
test oracles) and serve as the CPU execution path. Each
docstring cites the reference call site whose math it replicates.
"""
from __future__ import annotations

from typing import List, Optional

import torch
import torch.nn.functional as F


def gconv_mix_dense(A: torch.Tensor, x: torch.Tensor, W: torch.Tensor,
                    b: Optional[torch.Tensor], activation: Optional[str]) -> torch.Tensor:
    """K-support dense graph convolution (reference GCN.py:24-43).

    A: (K, N, N) support stack; x: (B, N, Cin); W: (K*Cin, Cout); b: (Cout,).
    y = act(concat_k(A_k @ x) @ W + b)
    """
    K = A.shape[0]
    feats = [torch.einsum("ij,bjp->bip", A[k], x) for k in range(K)]
    feat = torch.cat(feats, dim=-1)                 # (B, N, K*Cin)
    y = torch.einsum("bip,pq->biq", feat, W)
    if b is not None:
        y = y + b
    if activation == "relu":
        y = torch.relu(y)
    elif activation not in (None, "none", "linear"):
        raise ValueError(f"unsupported activation {activation!r}")
    return y


def cheb_supports_apply(csr, x: torch.Tensor) -> torch.Tensor:
    """Support stack application via the sparse generator: returns
    (B, K_s, N, Cin) where slice k is T_k(G) @ x (oracle for the HIP
    in-kernel recurrence)."""
    G = torch.sparse_csr_tensor(
        csr.row_ptr.to(torch.int64), csr.col_idx.to(torch.int64),
        csr.vals.to(x.dtype), size=(csr.n_nodes, csr.n_nodes), device=x.device)
    B, N, C = x.shape
    xt = x.permute(1, 0, 2).reshape(N, B * C)       # (N, B*C)
    outs: List[torch.Tensor] = []
    if csr.kind == "single":
        outs.append(torch.sparse.mm(G, xt))
    else:
        s_prev2 = xt                                 # T_0 x = x
        outs.append(s_prev2)
        if csr.K_supports > 1:
            s_prev = torch.sparse.mm(G, xt)          # T_1 x = G x
            outs.append(s_prev)
        for _ in range(2, csr.K_supports):
            s_cur = 2.0 * torch.sparse.mm(G, s_prev) - s_prev2
            outs.append(s_cur)
            s_prev2, s_prev = s_prev, s_cur
    S = torch.stack(outs, dim=0)                     # (K, N, B*C)
    return S.reshape(len(outs), N, B, C).permute(2, 0, 1, 3).contiguous()


def gconv_mix_csr(csr, x: torch.Tensor, W: torch.Tensor,
                  b: Optional[torch.Tensor], activation: Optional[str]) -> torch.Tensor:
    """CSR path of gconv_mix: same math as gconv_mix_dense with the support
    stack generated on the fly by the Chebyshev recurrence."""
    B, N, C = x.shape
    S = cheb_supports_apply(csr, x)                 # (B, K, N, C)
    feat = S.permute(0, 2, 1, 3).reshape(B, N, csr.K_supports * C)
    y = feat @ W
    if b is not None:
        y = y + b
    if activation == "relu":
        y = torch.relu(y)
    return y


def contextual_gate(obs_seq: torch.Tensor, gconv_out: torch.Tensor,
                    fc_weight: torch.Tensor, fc_bias: torch.Tensor) -> torch.Tensor:
    """Contextual gating (reference STMGCN.py:36-44, eqs. 6-9).

    obs_seq: (B, T, N, C); gconv_out: (B, N, T) = GCN(A, sum_c obs (B,N,T)).
    x_hat = gconv_out + x_seq (residual, eq.6); z = mean_n x_hat (eq.7);
    s = sigmoid(FC(relu(FC(z)))) with ONE weight-tied FC applied twice
    (quirk 2, STMGCN.py:43); out = obs_seq * s (eq.9).
    """
    x_seq = obs_seq.sum(dim=-1).permute(0, 2, 1)    # (B, N, T)
    x_hat = gconv_out + x_seq                        # eq.6 residual
    z = x_hat.mean(dim=1)                            # (B, T) node pool, eq.7
    s = torch.sigmoid(F.linear(torch.relu(F.linear(z, fc_weight, fc_bias)),
                               fc_weight, fc_bias))  # eq.8, tied weights
    return torch.einsum("btnc,bt->btnc", obs_seq, s)  # eq.9


def lstm_forward(x: torch.Tensor, weights: List[torch.Tensor],
                 h0: torch.Tensor, c0: torch.Tensor,
                 return_sequences: bool = False):
    """Multi-layer LSTM, batch_first, nn.LSTM-compatible math
    (reference STMGCN.py:21-22,47-50; cuDNN semantics).

    x: (B', T, Cin); weights: flat list [w_ih, w_hh, b_ih, b_hh] per layer,
    w_ih: (4H, Cin_l), gate order i|f|g|o. h0/c0: (L, B', H).
    Returns h_last (B', H) or the full top-layer sequence (B', T, H).
    """
    L = len(weights) // 4
    B, T, _ = x.shape
    layer_in = x
    top_seq = None
    for l in range(L):
        w_ih, w_hh, b_ih, b_hh = weights[4 * l: 4 * l + 4]
        H = w_hh.shape[1]
        h, c = h0[l], c0[l]
        outs = []
        # hoist the input projection out of the time loop (one GEMM per layer)
        x_proj = layer_in @ w_ih.T + b_ih            # (B', T, 4H)
        for t in range(T):
            gates = x_proj[:, t] + h @ w_hh.T + b_hh
            i, f, g, o = gates.split(H, dim=-1)
            i, f, o = torch.sigmoid(i), torch.sigmoid(f), torch.sigmoid(o)
            g = torch.tanh(g)
            c = f * c + i * g
            h = o * torch.tanh(c)
            outs.append(h)
        layer_in = torch.stack(outs, dim=1)          # (B', T, H)
        top_seq = layer_in
    return top_seq if return_sequences else top_seq[:, -1]


def gru_forward(x: torch.Tensor, weights: List[torch.Tensor],
                h0: torch.Tensor, return_sequences: bool = False):
    """Multi-layer GRU, batch_first, nn.GRU-compatible math (gate order r|z|n;
    the deep-variant CGRNN cell, BASELINE.json configs[3])."""
    L = len(weights) // 4
    layer_in = x
    top_seq = None
    for l in range(L):
        w_ih, w_hh, b_ih, b_hh = weights[4 * l: 4 * l + 4]
        H = w_hh.shape[1]
        h = h0[l]
        outs = []
        x_proj = layer_in @ w_ih.T + b_ih            # (B', T, 3H)
        for t in range(layer_in.shape[1]):
            h_proj = h @ w_hh.T + b_hh
            xr, xz, xn = x_proj[:, t].split(H, dim=-1)
            hr, hz, hn = h_proj.split(H, dim=-1)
            r = torch.sigmoid(xr + hr)
            z = torch.sigmoid(xz + hz)
            n = torch.tanh(xn + r * hn)
            h = (1.0 - z) * n + z * h
            outs.append(h)
        layer_in = torch.stack(outs, dim=1)
        top_seq = layer_in
    return top_seq if return_sequences else top_seq[:, -1]


def branch_fuse_head(branch_feats: List[torch.Tensor], fc_weight: torch.Tensor,
                     fc_bias: torch.Tensor) -> torch.Tensor:
    """Sum-fuse M branch outputs + FC regression head
    (reference STMGCN.py:116-118)."""
    fused = torch.stack(branch_feats, dim=-1).sum(dim=-1)   # (B, N, G)
    return F.linear(fused, fc_weight, fc_bias)              # (B, N, C)
