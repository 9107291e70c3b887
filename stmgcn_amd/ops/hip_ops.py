"""autograd.Function wrappers over the CDNA4 HIP kernels (stmgcn_amd._C).

Kernel coverage (all landed; SURVEY §2.4 op numbers in parentheses):
  - ChebGconvFn (K1/K2/K10): fully fused ChebConv — in-kernel CSR support
    recurrence with the mix GEMM + bias + act as an MFMA epilogue of each
    step; backward = Clenshaw over G^T with the U = dz W^T GEMM fused
    in-kernel + one multi-source atb_wgrad launch for every dW_k.
  - FusedLSTMFn / FusedRNNFn (K5/K6/K10): persistent multi-layer LSTM/GRU
    (fused_rnn.hip) + one batched wgrad launch for ALL layer weight grads
    (wgrad.hip). GRU rides the 4-slot packing documented on FusedLSTMFn.
  - SeqsumPermuteFn (K3), GateFn (K4), HeadFn (K7), FusedMSELossFn (K8);
    the K9 Adam kernel is driven by train/fused_adam.py.
Numerics tests compare every Function against the fp32 oracle in
reference_impl.py (tests/test_gpu_kernels.py).
"""
from __future__ import annotations

from typing import Optional

import torch

import os

from . import reference_impl as ref
from .functional import require_hip
from ..graph.preprocess import CSRSupport

# ---------------------------------------------------------------------------
# Fallback policy: the HIP kernels serve fixed shapes (H=64, T<=16, ...).
# Off-shape on the GPU hot path is a HARD ERROR unless the caller explicitly
# sets STMGCN_ALLOW_FALLBACK=1 — a silent torch fallback would bench stock
# PyTorch while reporting impl="hip" (fail-loud principle, functional.py).
# Every allowed fallback is counted so bench.py can surface impl="hip+fallback".
_fallback_count = 0


def fallback_count() -> int:
    return _fallback_count


def reset_fallback_count() -> None:
    global _fallback_count
    _fallback_count = 0


def _fallback(what: str) -> None:
    global _fallback_count
    if os.environ.get("STMGCN_ALLOW_FALLBACK", "0") != "1":
        raise RuntimeError(
            f"HIP kernel path cannot serve {what}; refusing a silent torch "
            "fallback on the GPU hot path. Set STMGCN_ALLOW_FALLBACK=1 to run "
            "the torch op anyway (reported as impl=hip+fallback), or "
            "STMGCN_IMPL=torch for the full stock-PyTorch path.")
    _fallback_count += 1


class ChebGconvFn(torch.autograd.Function):
    """y = act(concat_k(T_k(G) x) @ W + b) — fully fused ChebConv
    (reference GCN.py:34-42 materializes dense (K,N,N)
    stacks offline and a (B,N,K_s,C) feature concat per forward).

    Fused path (bf16/f16, C/Cout <= 64 and mult-of-8, K_s <= 4 — every
    BASELINE config): the K_s recurrence steps run in-kernel with the mix
    GEMM + bias + ReLU folded into each step's MFMA epilogue. Training keeps
    the recurrence states p_k (their HBM writes are mandatory anyway — the
    step dependency crosses workgroups) and the backward computes dX via
    Clenshaw over G^T with U_j = dz W_j^T fused in-kernel, plus ALL
    supports' dW_k in one atb_wgrad_multi launch over [x, p_1, ..]. Zero
    library GEMMs, zero (B,N,K_s,C) concat stacks in either direction.

    Stack path (fp32 parity / larger widths): cheb_apply support stack +
    rocBLAS mix, Clenshaw combine backward.
    """

    @staticmethod
    def forward(ctx, x: torch.Tensor, W: torch.Tensor, b: Optional[torch.Tensor],
                csr: CSRSupport, activation: Optional[str],
                training: bool = True):
        C = require_hip()
        x = x.contiguous()
        Cin = x.shape[-1]
        Cout = W.shape[1]
        Wd = W.to(x.dtype).contiguous()
        fused = (x.dtype in (torch.bfloat16, torch.float16)
                 and Cin <= 64 and Cout <= 64
                 and Cin % 8 == 0 and Cout % 8 == 0
                 and csr.K_supports <= 4)   # atb_wgrad_multi: <= 4 sources
        ctx.fused = fused
        ctx.csr = csr
        ctx.act = activation
        ctx.has_b = b is not None
        ctx.cin = Cin
        if fused:
            outs = C.cheb_gconv_fused_fwd(
                x, csr.row_ptr, csr.col_idx, csr.vals, Wd,
                b.to(x.dtype).contiguous() if b is not None else None,
                csr.K_supports, csr.kind == "single",
                1 if activation == "relu" else 0, training)
            y = outs[0]
            # outs[1:] = recurrence states p_1..p_{K_s-1} (p_0 == x), kept
            # for the wgrad — their HBM writes were mandatory regardless
            ctx.save_for_backward(x, Wd, W,
                                  y if activation == "relu" else None,
                                  *outs[1:])
            return y
        S = C.cheb_apply(x, csr.row_ptr, csr.col_idx, csr.vals,
                         csr.K_supports, csr.kind == "single")
        B_, N, K, _ = S.shape
        feat = S.view(B_, N, K * Cin)
        if b is not None:
            y = torch.addmm(b.to(feat.dtype), feat.view(-1, K * Cin), Wd)
            y = y.view(B_, N, -1)
        else:
            y = feat @ Wd
        if activation == "relu":
            y = torch.relu_(y)
        ctx.save_for_backward(feat, W, y if activation == "relu" else None)
        return y

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        C = require_hip()
        csr: CSRSupport = ctx.csr
        if ctx.fused:
            x, Wd, W, y = ctx.saved_tensors[:4]
            ps = ctx.saved_tensors[4:]       # p_1..p_{K_s-1} from forward
            if ctx.act == "relu":
                dz = (dy * (y > 0).to(dy.dtype)).contiguous()
            else:
                dz = dy.contiguous()
            single = csr.kind == "single"
            K_s = csr.K_supports
            Cin, Cout = ctx.cin, dz.shape[-1]
            dX = C.cheb_gconv_fused_bwd_dx(dz, Wd, csr.row_ptr_t,
                                           csr.col_idx_t, csr.vals_t, K_s,
                                           single)
            # ---- dW/db: ONE multi-source MFMA reduction over the saved
            # recurrence states (dz streamed once, no concat stack)
            dWf = torch.zeros(K_s * Cin, Cout, dtype=torch.float32,
                              device=dz.device)
            db_f = (torch.zeros(Cout, dtype=torch.float32, device=dz.device)
                    if ctx.has_b else None)
            dz2 = dz.reshape(-1, Cout)
            srcs = [p.reshape(-1, Cin) for p in ps] if single \
                else [x.reshape(-1, Cin)] + [p.reshape(-1, Cin) for p in ps]
            C.atb_wgrad_multi(srcs, dz2, dWf, db_f)
            dW = dWf.to(W.dtype)
            db = db_f.to(W.dtype) if ctx.has_b else None
            return dX, dW, db, None, None, None

        feat, W, y = ctx.saved_tensors
        if ctx.act == "relu":
            dz = dy * (y > 0).to(dy.dtype)
        else:
            dz = dy
        dz = dz.contiguous()
        B_, N, KC = feat.shape
        Cout = dz.shape[-1]
        # tall-skinny reduction GEMM kernel (wgrad.hip): dW = feat^T dZ and
        # db = colsum(dZ) in one launch, fp32 accumulate. (bf16/f16 only —
        # the MFMA tile is 16x16x32_bf16/f16; fp32 parity runs rocBLAS.
        # M/N must be multiples of 8: the kernel issues 16-byte fragment
        # loads at column offsets cb*8 — a ragged tail would read OOB.)
        if (dz.dtype in (torch.bfloat16, torch.float16) and KC <= 256
                and Cout <= 64 and KC % 8 == 0 and Cout % 8 == 0):
            outs = C.atb_wgrad(feat.reshape(-1, KC), dz.reshape(-1, Cout), ctx.has_b)
            dW = outs[0].to(W.dtype)
            db = outs[1].to(W.dtype) if ctx.has_b else None
        else:
            dW = (feat.reshape(-1, KC).T @ dz.reshape(-1, Cout)).to(W.dtype)
            db = dz.sum(dim=(0, 1)).to(W.dtype) if ctx.has_b else None
        U = (dz @ W.to(dz.dtype).T).view(B_, N, csr.K_supports, ctx.cin).contiguous()
        dX = C.cheb_combine(U, csr.row_ptr_t, csr.col_idx_t, csr.vals_t,
                            csr.kind == "single")
        return dX, dW, db, None, None, None


def contextual_gate_hip(obs_seq, gconv_out, fc_weight, fc_bias):
    if obs_seq.shape[1] <= 16:
        return GateFn.apply(obs_seq, gconv_out, fc_weight, fc_bias)
    _fallback(f"contextual gate with T={obs_seq.shape[1]} > 16")
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
    """Persistent fused multi-layer LSTM/GRU: one kernel
    for all layers x timesteps (forward), one dgrad kernel for the BPTT, one
    batched reduction-GEMM kernel (wgrad.hip) for ALL weight gradients.

    GRU (the deep variant, BASELINE configs[3]) rides the same 4-slot MFMA
    kernels through weight packing — q-slots [r | z | n_input | n_hidden]:
      W_ih_packed = [Wr; Wz; Wn; 0],  W_hh_packed = [Ur; Uz; 0; Un]
      b_ih_packed = [bir; biz; bin; 0], b_hh_packed = [bhr; bhz; 0; bhn]
    and the packed dA stream [dr|dz|dn|dBn] flows through the shared dgrad
    GEMMs and wgrad kernel; grads are un-packed by slicing below.

    Constraints (checked): bf16/f16, H == 64, C_in in {1, 64}, T <= 16,
    L <= 8, zero initial states (the reference zero-inits every forward —
    STMGCN.py:93-98,109, quirk 8)."""

    @staticmethod
    def forward(ctx, x, cell, ret_seq, training, *weights):
        C = require_hip()
        L = len(weights) // 4
        gru = cell == "gru"
        if gru:
            H = weights[1].shape[1]
            w_ih, w_hh, b_ih, b_hh = [], [], [], []
            for l in range(L):
                wi, wh, bi, bh = weights[4 * l:4 * l + 4]
                wp = wi.new_zeros(4 * H, wi.shape[1])
                wp[:3 * H] = wi
                hp = wh.new_zeros(4 * H, H)
                hp[:2 * H] = wh[:2 * H]
                hp[3 * H:] = wh[2 * H:]
                bip = wi.new_zeros(4 * H)
                bhp = torch.zeros_like(bip)
                bip[:3 * H] = bi
                bhp[:2 * H] = bh[:2 * H]
                bhp[3 * H:] = bh[2 * H:]
                w_ih.append(wp.contiguous())
                w_hh.append(hp.contiguous())
                b_ih.append(bip)
                b_hh.append(bhp)
        else:
            w_ih = [weights[4 * l + 0].contiguous() for l in range(L)]
            w_hh = [weights[4 * l + 1].contiguous() for l in range(L)]
            b_ih = [weights[4 * l + 2].contiguous() for l in range(L)]
            b_hh = [weights[4 * l + 3].contiguous() for l in range(L)]
        outs = C.lstm_fwd(x.contiguous(), w_ih, w_hh, b_ih, b_hh, ret_seq,
                          training, gru)
        ctx.ret_seq = ret_seq
        ctx.L = L
        ctx.gru = gru
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
        w_ih = ctx.saved_tensors[4:4 + L]        # packed for GRU
        w_hh = ctx.saved_tensors[4 + L:4 + 2 * L]
        w_ihT = [w.t().contiguous() for w in w_ih]
        w_hhT = [w.t().contiguous() for w in w_hh]
        x = x.contiguous()
        dx, dA = C.lstm_bwd(dout, x, cseq, gates, w_ihT, w_hhT, ctx.ret_seq,
                            ctx.gru)
        S, Tst, cin = x.shape
        # ---- weight grads: ONE batched reduction-GEMM launch for all layers
        # (wgrad.hip; replaces 5 hipBLASLt GEMMs + 3 cat + 3 colsum per call
        # that measured 47.8% of step time — profiles/r01_*)
        dwih, dwhh, dbf = C.lstm_wgrad(dA, hseq, x)
        grads = []
        H = 64
        for l in range(L):
            dt = w_ih[l].dtype
            cin_l = cin if l == 0 else H
            if ctx.gru:
                dw_ih = dwih[l, :3 * H, :cin_l].to(dt)
                dw_hh = torch.cat([dwhh[l, :2 * H], dwhh[l, 3 * H:]]).to(dt)
                db_ih = dbf[l, :3 * H].to(dt)
                db_hh = torch.cat([dbf[l, :2 * H], dbf[l, 3 * H:]]).to(dt)
            else:
                dw_ih = dwih[l, :, :cin_l].to(dt)
                dw_hh = dwhh[l].to(dt)
                db_ih = dbf[l].to(dt)
                db_hh = db_ih.clone()
            grads += [dw_ih, dw_hh, db_ih, db_hh]
        return (dx, None, None, None, *grads)


class FusedRNNFn:
    """Dispatch: LSTM/GRU -> fused HIP kernels (bf16/f16, H=64);
    fp32 or off-shape -> torch native fused RNN (the stock floor)."""

    @staticmethod
    def apply(cell, x, h0, c0, return_sequences, *weights):
        gmul = 4 if cell == "lstm" else 3
        if (cell in ("lstm", "gru")
                and x.dtype in (torch.bfloat16, torch.float16)
                and x.shape[-1] in (1, 64)
                and weights[1].shape == (gmul * 64, 64)
                and x.shape[1] <= 16):
            training = torch.is_grad_enabled() and (
                x.requires_grad or any(w.requires_grad for w in weights))
            return FusedLSTMFn.apply(x, cell, return_sequences, training, *weights)
        _fallback(
            f"fused {cell} with dtype={x.dtype}, C_in={x.shape[-1]}, "
            f"T={x.shape[1]}, hidden={tuple(weights[1].shape)} "
            "(served: bf16/f16, C_in in {1,64}, T<=16, H=64)")
        return _vf_rnn(cell, x, h0, c0, return_sequences, weights)


def branch_fuse_head_hip(branch_feats, fc_weight, fc_bias):
    if (fc_weight.shape[0] == 1 and fc_weight.shape[1] <= 64
            and len(branch_feats) <= 3):
        return HeadFn.apply(fc_weight, fc_bias, *branch_feats)
    _fallback(
        f"fused head with C_out={fc_weight.shape[0]}, G={fc_weight.shape[1]}, "
        f"M={len(branch_feats)} (served: C_out=1, G<=64, M<=3)")
    return ref.branch_fuse_head(branch_feats, fc_weight, fc_bias)


class SeqsumPermuteFn(torch.autograd.Function):
    """K3: x_seq = obs.sum(-1).permute(0,2,1) as one kernel (reference
    STMGCN.py:36,39); backward broadcasts over T,C."""

    @staticmethod
    def forward(ctx, obs):
        C = require_hip()
        ctx.C = obs.shape[-1]
        return C.seqsum_permute(obs.contiguous())

    @staticmethod
    def backward(ctx, dxs):
        C = require_hip()
        return C.seqsum_permute_bwd(dxs, ctx.C)


class GateFn(torch.autograd.Function):
    """K4: fused contextual gate. The kernel folds the gradient of the
    residual x_seq path (x_seq = sum_c obs) directly into dobs, so the xs
    input gets no separate grad here; the gconv path (xs -> gconv -> g) flows
    through dg and the autograd graph of xs as usual."""

    @staticmethod
    def forward(ctx, obs, g, fcw, fcb):
        C = require_hip()
        obs = obs.contiguous()
        with torch.no_grad():
            xs = C.seqsum_permute(obs)
        out, z, u, s = C.gate_fwd(obs, g.contiguous(), xs, fcw.to(obs.dtype),
                                  fcb.to(obs.dtype))
        ctx.save_for_backward(obs, fcw, z, u, s)
        return out

    @staticmethod
    def backward(ctx, dout):
        C = require_hip()
        obs, fcw, z, u, s = ctx.saved_tensors
        dobs, dg, dw_part, db_part = C.gate_bwd(dout, obs, fcw.to(obs.dtype), z, u, s)
        dw = dw_part.sum(dim=0).to(fcw.dtype)
        db = db_part.sum(dim=0).to(fcw.dtype)
        return dobs, dg, dw, db


class HeadFn(torch.autograd.Function):
    """K7: y = (sum_m feats_m) @ w^T + b fused (reference STMGCN.py:116-118)."""

    @staticmethod
    def forward(ctx, fc_weight, fc_bias, *feats):
        C = require_hip()
        y, fsum = C.head_fwd(list(feats), fc_weight.view(-1).to(feats[0].dtype),
                             fc_bias.to(feats[0].dtype))
        ctx.save_for_backward(fc_weight, fsum)
        ctx.M = len(feats)
        return y

    @staticmethod
    def backward(ctx, dy):
        C = require_hip()
        fc_weight, fsum = ctx.saved_tensors
        G = fsum.shape[-1]
        dfeat = C.head_bwd(dy, fc_weight.view(-1).to(dy.dtype), G)
        # dw = dy^T fsum, db = sum(dy) as one reduction kernel (the last
        # library GEMM on the step was exactly this 1xG tall-skinny)
        dwf, dbf = C.head_wgrad(dy, fsum)
        dw = dwf.view(1, G).to(fc_weight.dtype)
        db = dbf.to(fc_weight.dtype)
        return (dw, db) + (dfeat,) * ctx.M


class FusedMSELossFn(torch.autograd.Function):
    """K8: mean((pred-target)^2) with the grad fused (diff saved once)."""

    @staticmethod
    def forward(ctx, pred, target):
        C = require_hip()
        loss, diff = C.mse_fwd(pred, target)
        ctx.save_for_backward(diff)
        return loss

    @staticmethod
    def backward(ctx, gout):
        C = require_hip()
        (diff,) = ctx.saved_tensors
        dpred = C.mse_bwd(diff, gout.detach().reshape(1).float().contiguous())
        return dpred, None
