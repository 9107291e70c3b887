"""ST-MGCN model family, MI355X-first.

Checkpoint contract: the base model's state_dict must carry the
reference's exact 56-key schema —
  rnn_list.{m}.gconv_temporal_feats.{W,b}
  rnn_list.{m}.fc.{weight,bias}
  rnn_list.{m}.lstm.{weight_ih,weight_hh,bias_ih,bias_hh}_l{0..L-1}
  gcn_list.{m}.{W,b}
  fc.{weight,bias}
and the class MUST be named ST_MGCN (the trainer keys the checkpoint filename
and dispatch on the class name — reference Model_Trainer.py:11,34, quirk 9).

All compute routes through stmgcn_amd.ops (HIP kernels on GPU, oracle on CPU).
The forward accepts either dense (K, N, N) support stacks (reference parity,
STMGCN.py:100-119) or CSRSupport generators (the MI355X path — the support
recurrence runs inside the ChebConv kernel, never materializing T_k).
"""
from __future__ import annotations

import math
from typing import List, Optional, Sequence, Union

import torch
import torch.nn as nn

from .. import ops
from ..graph.preprocess import CSRSupport

AdjLike = Union[torch.Tensor, CSRSupport]

# Branch-concurrency stream pool: the M graph branches are independent
# (reference runs them sequentially, STMGCN.py:112-115); on GPU each branch
# runs on its own HIP stream so the small gconv/gate kernels of one branch
# overlap the big persistent-RNN kernels of another. torch autograd replays
# each op's backward on its forward stream, so the overlap carries to the
# backward pass; fork/join events keep this capturable in a hipGraph.
# STMGCN_DETERMINISTIC=1 disables the multi-stream path (SURVEY §5
# race-detection: a serialized debug mode for bisecting stream-ordering bugs).
import os as _os

_STREAMS: List[torch.cuda.Stream] = []


def deterministic_mode() -> bool:
    return _os.environ.get("STMGCN_DETERMINISTIC", "0") == "1"


def _branch_streams(n: int) -> List[torch.cuda.Stream]:
    while len(_STREAMS) < n:
        _STREAMS.append(torch.cuda.Stream())
    return _STREAMS[:n]


def _norm_activation(act):
    """Accept the reference's nn-module wiring (Main.py:64 passes
    gconv_activation=nn.ReLU) alongside the string form used here."""
    if act is None or isinstance(act, str):
        return act
    if act is nn.ReLU or isinstance(act, nn.ReLU):
        return "relu"
    raise ValueError(
        f"unsupported gconv activation {act!r}: pass 'relu', None, or nn.ReLU")


def _record_stream(t: torch.Tensor, stream: torch.cuda.Stream) -> None:
    # During hipGraph capture the graph's private memory pool keeps
    # allocations stable, and some torch/ROCm builds reject record_stream
    # inside capture — skip it there so --graph keeps the multi-stream path.
    if not torch.cuda.is_current_stream_capturing():
        t.record_stream(stream)


class GCN(nn.Module):
    """K-support graph convolution op (reference GCN.py:7-46).

    Parameters named W/b to match the checkpoint schema (GCN.py:17-22);
    xavier-normal W, zero b (GCN.py:20-22)."""

    def __init__(self, K: int, input_dim: int, hidden_dim: int, bias: bool = True,
                 activation: Optional[str] = "relu"):
        super().__init__()
        self.K = K
        self.input_dim = input_dim
        self.hidden_dim = hidden_dim
        self.activation = _norm_activation(activation)
        self.W = nn.Parameter(torch.empty(K * input_dim, hidden_dim))
        nn.init.xavier_normal_(self.W)
        self.b = nn.Parameter(torch.zeros(hidden_dim)) if bias else None

    def forward(self, A: AdjLike, x: torch.Tensor) -> torch.Tensor:
        K_s = A.K_supports if isinstance(A, CSRSupport) else A.shape[0]
        if K_s != self.K:
            raise ValueError(f"support count mismatch: adj has {K_s}, module expects {self.K}")
        return ops.gconv_mix(A, x, self.W, self.b, self.activation)

    def extra_repr(self) -> str:
        return f"K={self.K}, in={self.input_dim}, out={self.hidden_dim}, act={self.activation}"


class CGRNNCellParams(nn.Module):
    """Parameter container for the fused multi-layer LSTM/GRU.

    Exposes nn.LSTM-compatible parameter names (weight_ih_l{k} etc.,
    gate order i|f|g|o for LSTM, r|z|n for GRU) so the checkpoint schema
    matches the reference's `rnn_list.{m}.lstm.*` keys byte-for-byte.
    Initialization matches nn.LSTM: U(-1/sqrt(H), 1/sqrt(H)) on every tensor.
    """

    def __init__(self, cell: str, input_dim: int, hidden_dim: int, num_layers: int):
        super().__init__()
        assert cell in ("lstm", "gru")
        self.cell = cell
        self.input_dim = input_dim
        self.hidden_dim = hidden_dim
        self.num_layers = num_layers
        gmul = 4 if cell == "lstm" else 3
        stdv = 1.0 / math.sqrt(hidden_dim)
        for l in range(num_layers):
            in_l = input_dim if l == 0 else hidden_dim
            for name, shape in (
                    (f"weight_ih_l{l}", (gmul * hidden_dim, in_l)),
                    (f"weight_hh_l{l}", (gmul * hidden_dim, hidden_dim)),
                    (f"bias_ih_l{l}", (gmul * hidden_dim,)),
                    (f"bias_hh_l{l}", (gmul * hidden_dim,))):
                p = nn.Parameter(torch.empty(*shape))
                nn.init.uniform_(p, -stdv, stdv)
                setattr(self, name, p)

    def flat_weights(self) -> List[torch.Tensor]:
        out: List[torch.Tensor] = []
        for l in range(self.num_layers):
            out += [getattr(self, f"weight_ih_l{l}"), getattr(self, f"weight_hh_l{l}"),
                    getattr(self, f"bias_ih_l{l}"), getattr(self, f"bias_hh_l{l}")]
        return out

    def forward(self, x: torch.Tensor, h0: torch.Tensor, c0: Optional[torch.Tensor],
                return_sequences: bool = False) -> torch.Tensor:
        return ops.rnn_forward(self.cell, x, self.flat_weights(), h0, c0,
                               return_sequences)


class CG_LSTM(nn.Module):
    """One per-graph branch: contextual gating + shared RNN
    (reference STMGCN.py:7-57). Module attribute names define the checkpoint
    schema — gconv_temporal_feats, fc (weight-tied, applied twice: quirk 2),
    lstm."""

    def __init__(self, K: int, seq_len: int, input_dim: int, lstm_hidden_dim: int,
                 lstm_num_layers: int, cell: str = "lstm",
                 gconv_use_bias: bool = True, gconv_activation: str = "relu"):
        super().__init__()
        self.seq_len = seq_len
        self.input_dim = input_dim
        self.hidden_dim = lstm_hidden_dim
        self.num_layers = lstm_num_layers
        self.cell = cell
        # temporal GCN over the T-step series viewed as node features (B,N,T)
        self.gconv_temporal_feats = GCN(K, seq_len, seq_len, bias=gconv_use_bias,
                                        activation=gconv_activation)
        # ONE shared FC applied twice in the gate (reference STMGCN.py:20,43)
        self.fc = nn.Linear(seq_len, seq_len)
        self.lstm = CGRNNCellParams(cell, input_dim, lstm_hidden_dim, lstm_num_layers)

    def forward(self, adj: AdjLike, obs_seq: torch.Tensor,
                return_sequences: bool = False) -> torch.Tensor:
        """obs_seq: (B, T, N, C) -> (B, N, H) (or (B, T, N, H) if sequences)."""
        B, T, N, C = obs_seq.shape
        x_seq = ops.seqsum_permute(obs_seq)                   # (B,N,T), K3
        g = self.gconv_temporal_feats(adj, x_seq)             # (B,N,T), K1+K2
        gated = ops.contextual_gate(obs_seq, g, self.fc.weight, self.fc.bias)  # K4
        flat = gated.permute(0, 2, 1, 3).reshape(B * N, T, C)  # node-major seqs
        h0 = obs_seq.new_zeros(self.num_layers, B * N, self.hidden_dim)
        c0 = h0.clone() if self.cell == "lstm" else None
        out = self.lstm(flat, h0, c0, return_sequences)       # K5/K6
        if return_sequences:
            return out.reshape(B, N, T, self.hidden_dim).permute(0, 2, 1, 3)
        return out.reshape(B, N, self.hidden_dim)


class ST_MGCN(nn.Module):
    """Spatiotemporal multi-graph convolution network
    (reference STMGCN.py:61-119): M parallel (CG_LSTM -> GCN) branches,
    sum-fused, FC regression head."""

    def __init__(self, M: int, seq_len: int, n_nodes: int, input_dim: int,
                 lstm_hidden_dim: int, lstm_num_layers: int, gcn_hidden_dim: int,
                 sta_kernel_config: dict, gconv_use_bias: bool = True,
                 gconv_activation: str = "relu", rnn_cell: str = "lstm"):
        super().__init__()
        self.M = M
        self.n_nodes = n_nodes
        self.seq_len = seq_len
        self.input_dim = input_dim
        sta_K = self.get_support_K(sta_kernel_config)
        self.sta_K = sta_K
        self.rnn_list = nn.ModuleList([
            CG_LSTM(sta_K, seq_len, input_dim, lstm_hidden_dim, lstm_num_layers,
                    cell=rnn_cell, gconv_use_bias=gconv_use_bias,
                    gconv_activation=gconv_activation)
            for _ in range(M)])
        self.gcn_list = nn.ModuleList([
            GCN(sta_K, lstm_hidden_dim, gcn_hidden_dim, bias=gconv_use_bias,
                activation=gconv_activation)
            for _ in range(M)])
        self.fc = nn.Linear(gcn_hidden_dim, input_dim)

    @staticmethod
    def get_support_K(config: dict) -> int:
        """Kernel config -> support count (reference STMGCN.py:80-91):
        localpool -> 1, chebyshev -> K+1, random_walk_diffusion -> 2K+1.
        (rw-diffusion is unusable end-to-end in the reference — quirk 3.)"""
        kt, K = config["kernel_type"], config.get("K", 2)
        if kt == "localpool":
            return 1
        if kt == "chebyshev":
            return K + 1
        if kt == "random_walk_diffusion":
            return 2 * K + 1
        raise ValueError(f"unknown kernel_type {kt!r}")

    def forward(self, obs_seq: torch.Tensor, sta_adj_list: Sequence[AdjLike]) -> torch.Tensor:
        """obs_seq: (B, T, N, C); sta_adj_list: M supports -> (B, N, C).

        Branches are independent (reference runs them sequentially,
        STMGCN.py:112-115); on GPU the fused kernels batch the M dimension
        into the grid / run on concurrent HIP streams."""
        if len(sta_adj_list) != self.M:
            raise ValueError(f"expected {self.M} adjacencies, got {len(sta_adj_list)}")
        feat_list = []
        if obs_seq.is_cuda and self.M > 1 and not deterministic_mode():
            main = torch.cuda.current_stream()
            streams = _branch_streams(self.M)
            fork = torch.cuda.Event()
            fork.record(main)
            for m in range(self.M):
                streams[m].wait_event(fork)
                with torch.cuda.stream(streams[m]):
                    _record_stream(obs_seq, streams[m])
                    h = self.rnn_list[m](sta_adj_list[m], obs_seq)
                    f = self.gcn_list[m](sta_adj_list[m], h)
                feat_list.append(f)
            for m in range(self.M):
                main.wait_stream(streams[m])
                _record_stream(feat_list[m], main)
        else:
            for m in range(self.M):
                h = self.rnn_list[m](sta_adj_list[m], obs_seq)     # (B,N,H)
                feat_list.append(self.gcn_list[m](sta_adj_list[m], h))  # (B,N,G)
        return ops.branch_fuse_head(feat_list, self.fc.weight, self.fc.bias)  # K7


class STMGCNBlock(nn.Module):
    """One stacked block for the deep variant: sequence-in, sequence-out.
    The CGRNN returns the full top-layer sequence and the post-RNN GCN is
    applied per timestep, so blocks compose; the reference has no stacking
    (this realizes BASELINE.json configs[3])."""

    def __init__(self, M: int, seq_len: int, input_dim: int, lstm_hidden_dim: int,
                 lstm_num_layers: int, gcn_hidden_dim: int, sta_K: int,
                 rnn_cell: str = "gru", gconv_use_bias: bool = True,
                 gconv_activation: str = "relu"):
        super().__init__()
        self.M = M
        self.rnn_list = nn.ModuleList([
            CG_LSTM(sta_K, seq_len, input_dim, lstm_hidden_dim, lstm_num_layers,
                    cell=rnn_cell, gconv_use_bias=gconv_use_bias,
                    gconv_activation=gconv_activation)
            for _ in range(M)])
        self.gcn_list = nn.ModuleList([
            GCN(sta_K, lstm_hidden_dim, gcn_hidden_dim, bias=gconv_use_bias,
                activation=gconv_activation)
            for _ in range(M)])

    def forward(self, x: torch.Tensor, sta_adj_list: Sequence[AdjLike]) -> torch.Tensor:
        """x: (B, T, N, C) -> (B, T, N, G) (sum-fused over branches;
        branch-concurrent HIP streams as in ST_MGCN.forward)."""
        B, T, N, _ = x.shape

        def branch(m):
            seq = self.rnn_list[m](sta_adj_list[m], x, return_sequences=True)  # (B,T,N,H)
            flat = seq.reshape(B * T, N, seq.shape[-1])
            return self.gcn_list[m](sta_adj_list[m], flat).reshape(B, T, N, -1)

        outs = []
        if x.is_cuda and self.M > 1 and not deterministic_mode():
            main = torch.cuda.current_stream()
            streams = _branch_streams(self.M)
            fork = torch.cuda.Event()
            fork.record(main)
            for m in range(self.M):
                streams[m].wait_event(fork)
                with torch.cuda.stream(streams[m]):
                    _record_stream(x, streams[m])
                    outs.append(branch(m))
            for m in range(self.M):
                main.wait_stream(streams[m])
                _record_stream(outs[m], main)
        else:
            outs = [branch(m) for m in range(self.M)]
        out = outs[0]
        for g in outs[1:]:
            out = out + g
        return out


class StackedSTMGCN(nn.Module):
    """Deep variant: n_blocks stacked ST-MGCN blocks (GRU CGRNN), final
    last-step FC head (BASELINE.json configs[3])."""

    def __init__(self, M: int, seq_len: int, n_nodes: int, input_dim: int,
                 lstm_hidden_dim: int, lstm_num_layers: int, gcn_hidden_dim: int,
                 sta_kernel_config: dict, n_blocks: int = 4, rnn_cell: str = "gru",
                 gconv_use_bias: bool = True, gconv_activation: str = "relu"):
        super().__init__()
        sta_K = ST_MGCN.get_support_K(sta_kernel_config)
        self.M, self.n_blocks = M, n_blocks
        dims = [input_dim] + [gcn_hidden_dim] * n_blocks
        self.blocks = nn.ModuleList([
            STMGCNBlock(M, seq_len, dims[i], lstm_hidden_dim, lstm_num_layers,
                        gcn_hidden_dim, sta_K, rnn_cell=rnn_cell,
                        gconv_use_bias=gconv_use_bias,
                        gconv_activation=gconv_activation)
            for i in range(n_blocks)])
        self.fc = nn.Linear(gcn_hidden_dim, input_dim)

    def forward(self, obs_seq: torch.Tensor, sta_adj_list: Sequence[AdjLike]) -> torch.Tensor:
        x = obs_seq
        for blk in self.blocks:
            x = blk(x, sta_adj_list)
        return self.fc(x[:, -1])                     # (B, N, C)


def build_model(cfg) -> nn.Module:
    """Config -> model (STMGCNConfig from stmgcn_amd.config)."""
    kconf = {"kernel_type": cfg.kernel_type, "K": cfg.cheby_K}
    if cfg.n_blocks > 1:
        return StackedSTMGCN(cfg.m_graphs, cfg.seq_len, cfg.n_nodes, cfg.input_dim,
                             cfg.lstm_hidden_dim, cfg.lstm_num_layers,
                             cfg.gcn_hidden_dim, kconf, n_blocks=cfg.n_blocks,
                             rnn_cell=cfg.rnn_cell,
                             gconv_use_bias=cfg.gconv_use_bias,
                             gconv_activation=cfg.gconv_activation)
    return ST_MGCN(cfg.m_graphs, cfg.seq_len, cfg.n_nodes, cfg.input_dim,
                   cfg.lstm_hidden_dim, cfg.lstm_num_layers, cfg.gcn_hidden_dim,
                   kconf, gconv_use_bias=cfg.gconv_use_bias,
                   gconv_activation=cfg.gconv_activation, rnn_cell=cfg.rnn_cell)
