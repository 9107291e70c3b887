"""Op dispatch: CDNA4 HIP kernels on GPU, pure-PyTorch oracle on CPU.

GPU tensors REQUIRE the in-tree HIP extension (stmgcn_amd/_C*.so, built by
__graft_entry__.build()); there is no silent eager fallback on a GPU device —
a missing extension raises. CPU tensors run the oracle math from
reference_impl (also the numeric-parity test oracle).
"""
from __future__ import annotations

from typing import List, Optional

import torch

from . import reference_impl as ref
from ..graph.preprocess import CSRSupport

_HIP = None
_HIP_ERR: Optional[str] = None


def impl_mode() -> str:
    """"hip" (default): GPU tensors require the HIP extension, fail-loud.
    "torch": explicit opt-in eager-PyTorch execution on GPU — used ONLY to
    measure the stock-PyTorch comparison floor and for A/B
    numerics debugging. Set STMGCN_IMPL=torch to opt in."""
    import os
    return os.environ.get("STMGCN_IMPL", "hip")


def _load_hip():
    global _HIP, _HIP_ERR
    if _HIP is not None or _HIP_ERR is not None:
        return _HIP
    try:
        from .. import _C  # in-tree extension: stmgcn_amd/_C*.so
        _HIP = _C
    except ImportError as e:  # record why, fail loudly at first GPU use
        _HIP_ERR = str(e)
    return _HIP


def hip_available() -> bool:
    return _load_hip() is not None


def require_hip():
    if _load_hip() is None:
        raise RuntimeError(
            "stmgcn_amd HIP extension (_C) is not built/importable on a GPU "
            f"device path — refusing to fall back to eager. Import error: {_HIP_ERR}. "
            "Build it with: python -c 'import __graft_entry__ as g; g.build()'")
    return _HIP


def gconv_mix(A, x: torch.Tensor, W: torch.Tensor, b: Optional[torch.Tensor],
              activation: Optional[str]) -> torch.Tensor:
    """K-support graph convolution y = act(concat_k(T_k x) @ W + b).

    A is either a dense (K, N, N) support stack (reference parity path,
    GCN.py:24-43) or a CSRSupport (in-kernel Chebyshev recurrence path)."""
    if isinstance(A, CSRSupport):
        if x.is_cuda and impl_mode() == "hip":
            from .hip_ops import ChebGconvFn
            # grad mode is disabled inside Function.forward, so the
            # training decision (save recurrence states for wgrad?) must be
            # made here
            training = torch.is_grad_enabled() and (
                x.requires_grad or W.requires_grad
                or (b is not None and b.requires_grad))
            return ChebGconvFn.apply(x, W, b, A, activation, training)
        return ref.gconv_mix_csr(A, x, W, b, activation)
    return ref.gconv_mix_dense(A, x, W, b, activation)


def contextual_gate(obs_seq: torch.Tensor, gconv_out: torch.Tensor,
                    fc_weight: torch.Tensor, fc_bias: torch.Tensor) -> torch.Tensor:
    """Contextual gating (eqs.6-9, weight-tied FC — reference STMGCN.py:36-44)."""
    if obs_seq.is_cuda and impl_mode() == "hip":
        from .hip_ops import contextual_gate_hip
        return contextual_gate_hip(obs_seq, gconv_out, fc_weight, fc_bias)
    return ref.contextual_gate(obs_seq, gconv_out, fc_weight, fc_bias)


def rnn_forward(cell: str, x: torch.Tensor, weights: List[torch.Tensor],
                h0: torch.Tensor, c0: Optional[torch.Tensor],
                return_sequences: bool = False) -> torch.Tensor:
    """Multi-layer LSTM/GRU over (B*N, T, C) — the dominant-FLOP op
    (reference STMGCN.py:47-50). GPU: persistent fused HIP kernel."""
    if x.is_cuda and impl_mode() == "hip":
        from .hip_ops import FusedRNNFn
        return FusedRNNFn.apply(cell, x, h0, c0, return_sequences, *weights)
    if x.is_cuda:  # stock-torch floor mode: the reference's nn.LSTM backend
        from .hip_ops import _vf_rnn
        return _vf_rnn(cell, x, h0, c0, return_sequences, list(weights))
    if cell == "lstm":
        return ref.lstm_forward(x, weights, h0, c0, return_sequences)
    if cell == "gru":
        return ref.gru_forward(x, weights, h0, return_sequences)
    raise ValueError(f"unknown rnn cell {cell!r}")


def seqsum_permute(obs_seq: torch.Tensor) -> torch.Tensor:
    """K3: (B,T,N,C) -> (B,N,T) feature-sum + transpose (STMGCN.py:36,39)."""
    if obs_seq.is_cuda and impl_mode() == "hip":
        from .hip_ops import SeqsumPermuteFn
        return SeqsumPermuteFn.apply(obs_seq)
    return obs_seq.sum(dim=-1).permute(0, 2, 1)


def mse_loss(pred: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
    """K8: fused MSE loss + grad on GPU; torch elsewhere."""
    if pred.is_cuda and impl_mode() == "hip":
        from .hip_ops import FusedMSELossFn
        return FusedMSELossFn.apply(pred, target)
    return torch.nn.functional.mse_loss(pred, target)


def branch_fuse_head(branch_feats: List[torch.Tensor], fc_weight: torch.Tensor,
                     fc_bias: torch.Tensor) -> torch.Tensor:
    """Sum over M branches + FC head (reference STMGCN.py:116-118)."""
    if branch_feats[0].is_cuda and impl_mode() == "hip":
        from .hip_ops import branch_fuse_head_hip
        return branch_fuse_head_hip(branch_feats, fc_weight, fc_bias)
    return ref.branch_fuse_head(branch_feats, fc_weight, fc_bias)
