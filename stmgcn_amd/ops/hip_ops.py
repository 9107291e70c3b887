"""autograd.Function wrappers over the CDNA4 HIP kernels (stmgcn_amd._C).

Status (round 1, in progress):
  - ChebGconvFn: HIP cheb_apply / cheb_combine (in-kernel support recurrence,
    SURVEY K1) + rocBLAS mix GEMMs. DONE.
  - FusedRNNFn / contextual_gate / branch_fuse_head: fused kernels in
    fused_rnn.hip / cg_gate.hip — being brought up; the interim GPU path
    composes torch ops so the end-to-end GPU slice runs (cheb kernels are
    already the native load-bearing path).
"""
from __future__ import annotations

from typing import List, Optional

import torch

from . import reference_impl as ref
from .functional import require_hip
from ..graph.preprocess import CSRSupport


class ChebGconvFn(torch.autograd.Function):
    """y = act(concat_k(T_k(G) x) @ W + b) with the K-hop Chebyshev
    recurrence computed by the HIP kernel on CSR G (never materializing T_k;
    reference materializes dense stacks, GCN.py:95/34-42). Mix GEMMs go to
    rocBLAS (plain library GEMMs); dW/db likewise.

    Backward: dX = sum_k T_k(G)^T (dZ W_k^T) = cheb_combine over G^T CSR
    (Clenshaw); dW = feat^T dZ; db = sum dZ.
    """

    @staticmethod
    def forward(ctx, x: torch.Tensor, W: torch.Tensor, b: Optional[torch.Tensor],
                csr: CSRSupport, activation: Optional[str]):
        C = require_hip()
        x = x.contiguous()
        S = C.cheb_apply(x, csr.row_ptr, csr.col_idx, csr.vals,
                         csr.K_supports, csr.kind == "single")
        B_, N, K, Cin = S.shape
        feat = S.view(B_, N, K * Cin)
        y = feat @ W.to(feat.dtype)
        if b is not None:
            y = y + b
        if activation == "relu":
            y = torch.relu(y)
        ctx.save_for_backward(feat, W, y if activation == "relu" else None)
        ctx.csr = csr
        ctx.act = activation
        ctx.has_b = b is not None
        ctx.cin = Cin
        return y

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        C = require_hip()
        feat, W, y = ctx.saved_tensors
        csr: CSRSupport = ctx.csr
        if ctx.act == "relu":
            dz = dy * (y > 0).to(dy.dtype)
        else:
            dz = dy
        dz = dz.contiguous()
        B_, N, KC = feat.shape
        Cout = dz.shape[-1]
        dW = (feat.reshape(-1, KC).T.to(torch.float32)
              @ dz.reshape(-1, Cout).to(torch.float32)).to(W.dtype)
        db = dz.sum(dim=(0, 1)).to(W.dtype) if ctx.has_b else None
        U = (dz @ W.to(dz.dtype).T).view(B_, N, csr.K_supports, ctx.cin).contiguous()
        dX = C.cheb_combine(U, csr.row_ptr_t, csr.col_idx_t, csr.vals_t,
                            csr.kind == "single")
        return dX, dW, db, None, None


def contextual_gate_hip(obs_seq, gconv_out, fc_weight, fc_bias):
    # TODO(round1): fused cg_gate.hip kernel; interim torch path.
    return ref.contextual_gate(obs_seq, gconv_out, fc_weight, fc_bias)


def _vf_rnn(cell, x, h0, c0, return_sequences, weights):
    """torch native fused RNN (what nn.LSTM calls). Used for the stock floor
    (STMGCN_IMPL=torch) and as the interim path while fused_rnn.hip lands."""
    L = len(weights) // 4
    if cell == "lstm":
        out, _, _ = torch._VF.lstm(x, (h0, c0), list(weights), True, L, 0.0,
                                   torch.is_grad_enabled(), False, True)
    else:
        out, _ = torch._VF.gru(x, h0, list(weights), True, L, 0.0,
                               torch.is_grad_enabled(), False, True)
    return out if return_sequences else out[:, -1]


class FusedLSTMFn(torch.autograd.Function):
    """Persistent fused multi-layer LSTM: one kernel for
    all layers x timesteps (forward), one dgrad kernel for the BPTT; weight
    gradients are two plain library GEMMs per layer over the streamed
    gate-preactivation grads dA (dW = dA^T @ [h_prev | x]).

    Constraints (checked): bf16/f16, H == 64, C_in in {1, 64}, T <= 16,
    L <= 8, zero initial states (the reference zero-inits every forward —
    STMGCN.py:93-98,109, quirk 8)."""

    @staticmethod
    def forward(ctx, x, ret_seq, training, *weights):
        C = require_hip()
        L = len(weights) // 4
        w_ih = [weights[4 * l + 0].contiguous() for l in range(L)]
        w_hh = [weights[4 * l + 1].contiguous() for l in range(L)]
        b_ih = [weights[4 * l + 2].float().contiguous() for l in range(L)]
        b_hh = [weights[4 * l + 3].float().contiguous() for l in range(L)]
        outs = C.lstm_fwd(x.contiguous(), w_ih, w_hh, b_ih, b_hh, ret_seq, training)
        ctx.ret_seq = ret_seq
        ctx.L = L
        if training:
            out, hseq, cseq, gates = outs
            ctx.save_for_backward(x, cseq, gates, hseq, *w_ih, *w_hh)
        else:
            out = outs[0]
        return out

    @staticmethod
    def backward(ctx, dout):
        C = require_hip()
        L = ctx.L
        x, cseq, gates, hseq = ctx.saved_tensors[:4]
        w_ih = ctx.saved_tensors[4:4 + L]
        w_hh = ctx.saved_tensors[4 + L:4 + 2 * L]
        w_ihT = [w.t().contiguous() for w in w_ih]
        w_hhT = [w.t().contiguous() for w in w_hh]
        x = x.contiguous()
        dx, dA = C.lstm_bwd(dout, x, cseq, gates, w_ihT, w_hhT, ctx.ret_seq)
        S, Tst, cin = x.shape
        # ---- weight grads: plain GEMMs over the dA stream (rocBLAS) --------
        # dA: (L, Tst, S_pad, 4H); hseq: (L, Tst, S_pad, H)
        S_pad = dA.shape[2]
        H = hseq.shape[-1]
        grads = []
        h_prev = torch.zeros_like(hseq[:, :1])
        for l in range(L):
            dA_l = dA[l].reshape(-1, 4 * H)                      # (Tst*S_pad, 4H)
            # h_{t-1}: shift hseq[l] right by one step
            hp = torch.cat([h_prev[0], hseq[l][:-1]], dim=0).reshape(-1, H)
            dw_hh = (dA_l.t().float() @ hp.float()).to(w_hh[l].dtype)
            if l == 0:
                xs = x.permute(1, 0, 2).reshape(Tst * S, cin)    # (Tst*S, C)
                dA_x = dA[l][:, :S].reshape(-1, 4 * H)
                dw_ih = (dA_x.t().float() @ xs.float()).to(w_ih[l].dtype)
            else:
                xl = hseq[l - 1].reshape(-1, H)
                dw_ih = (dA_l.t().float() @ xl.float()).to(w_ih[l].dtype)
            db = dA_l.sum(dim=0).to(w_ih[l].dtype)
            grads += [dw_ih, dw_hh, db, db.clone()]
        return (dx, None, None, *grads)


class FusedRNNFn:
    """Dispatch: LSTM -> fused HIP kernels (bf16/f16); GRU and fp32 -> torch
    native fused RNN (interim; GRU HIP kernels land with the deep variant)."""

    @staticmethod
    def apply(cell, x, h0, c0, return_sequences, *weights):
        if (cell == "lstm" and x.dtype in (torch.bfloat16, torch.float16)
                and x.shape[-1] in (1, 64) and weights[1].shape[1] == 64
                and x.shape[1] <= 16):
            training = torch.is_grad_enabled() and (
                x.requires_grad or any(w.requires_grad for w in weights))
            return FusedLSTMFn.apply(x, return_sequences, training, *weights)
        return _vf_rnn(cell, x, h0, c0, return_sequences, weights)


def branch_fuse_head_hip(branch_feats, fc_weight, fc_bias):
    # TODO(round1): fused multi-graph-sum + FC head kernel.
    return ref.branch_fuse_head(branch_feats, fc_weight, fc_bias)
