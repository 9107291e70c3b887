"""GPU numerics tests: every HIP kernel vs the fp32 pure-PyTorch oracle
(SURVEY §4 item 1). All marked @pytest.mark.gpu."""
import os

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

from stmgcn_amd.data.synthetic import _random_sparse_sym_adj
from stmgcn_amd.graph import SupportGenerator
from stmgcn_amd.ops import reference_impl as ref
from stmgcn_amd.ops.functional import require_hip


def _csr(n=64, seed=0, kernel="chebyshev", K=2):
    rng = np.random.default_rng(seed)
    A = torch.from_numpy(_random_sparse_sym_adj(n, 8, rng, weighted=True))
    return SupportGenerator(kernel, K).process_csr(A)


@pytest.mark.parametrize("dtype,tol", [(torch.float32, 1e-5), (torch.bfloat16, 3e-2)])
@pytest.mark.parametrize("C", [8, 64])
def test_cheb_apply_matches_oracle(dtype, tol, C):
    _C = require_hip()
    csr = _csr()
    dev = torch.device("cuda")
    csr_d = csr.to(dev)
    x = torch.randn(4, csr.n_nodes, C, device=dev, dtype=dtype)
    S = _C.cheb_apply(x, csr_d.row_ptr, csr_d.col_idx, csr_d.vals,
                      csr_d.K_supports, csr_d.kind == "single")
    assert S.shape == (4, csr.n_nodes, csr.K_supports, C)
    S_ref = ref.cheb_supports_apply(csr, x.float().cpu())  # (B,K,N,C)
    got = S.permute(0, 2, 1, 3).float().cpu()
    torch.testing.assert_close(got, S_ref, rtol=tol, atol=tol)


def test_cheb_apply_localpool():
    _C = require_hip()
    csr = _csr(kernel="localpool")
    dev = torch.device("cuda")
    csr_d = csr.to(dev)
    x = torch.randn(2, csr.n_nodes, 16, device=dev)
    S = _C.cheb_apply(x, csr_d.row_ptr, csr_d.col_idx, csr_d.vals, 1, True)
    S_ref = ref.cheb_supports_apply(csr, x.float().cpu())
    torch.testing.assert_close(S.permute(0, 2, 1, 3).cpu(), S_ref, rtol=1e-5, atol=1e-5)


@pytest.mark.parametrize("K", [2, 3])
def test_cheb_combine_matches_sum(K):
    """combine(U) == sum_k T_k(G) @ U_k against dense support stacks."""
    _C = require_hip()
    csr = _csr(K=K)
    dense = csr.dense_supports()                      # (K_s, N, N)
    dev = torch.device("cuda")
    csr_d = csr.to(dev)
    B, C = 3, 8
    U = torch.randn(B, csr.n_nodes, csr.K_supports, C, device=dev)
    Z = _C.cheb_combine(U, csr_d.row_ptr, csr_d.col_idx, csr_d.vals, False)
    Uc = U.cpu()
    Z_ref = sum(torch.einsum("ij,bjc->bic", dense[k], Uc[:, :, k])
                for k in range(csr.K_supports))
    torch.testing.assert_close(Z.cpu(), Z_ref, rtol=1e-4, atol=1e-4)


@pytest.mark.parametrize("dtype,rtol,atol", [
    (torch.float32, 1e-4, 1e-5), (torch.bfloat16, 5e-2, 5e-2)])
def test_cheb_gconv_forward_backward(dtype, rtol, atol):
    from stmgcn_amd.ops.hip_ops import ChebGconvFn
    csr = _csr()
    dev = torch.device("cuda")
    csr_d = csr.to(dev)
    B, N, Cin, Cout = 4, csr.n_nodes, 8, 16
    torch.manual_seed(0)
    x = torch.randn(B, N, Cin, device=dev, dtype=dtype, requires_grad=True)
    W = torch.randn(csr.K_supports * Cin, Cout, device=dev, dtype=dtype,
                    requires_grad=True) * 0.2
    W.retain_grad()
    b = torch.zeros(Cout, device=dev, dtype=dtype, requires_grad=True)
    y = ChebGconvFn.apply(x, W, b, csr_d, "relu")
    loss = (y.float() ** 2).sum()
    loss.backward()

    x_ref = x.detach().float().cpu().requires_grad_(True)
    W_ref = W.detach().float().cpu().requires_grad_(True)
    b_ref = b.detach().float().cpu().requires_grad_(True)
    y_ref = ref.gconv_mix_csr(csr, x_ref, W_ref, b_ref, "relu")
    (y_ref ** 2).sum().backward()

    torch.testing.assert_close(y.float().cpu(), y_ref.detach(), rtol=rtol, atol=atol * 10)
    torch.testing.assert_close(x.grad.float().cpu(), x_ref.grad, rtol=rtol * 10, atol=atol * 50)
    torch.testing.assert_close(W.grad.float().cpu(), W_ref.grad, rtol=rtol * 10, atol=atol * 50)
    torch.testing.assert_close(b.grad.float().cpu(), b_ref.grad, rtol=rtol * 10, atol=atol * 50)


def test_model_gpu_matches_cpu_fp32(monkeypatch):
    """Full ST_MGCN forward on GPU (HIP path, fp32) vs CPU oracle.

    fp32 + H=16 is off-shape for the fused RNN — this parity test explicitly
    opts into the (counted) torch fallback for that op."""
    monkeypatch.setenv("STMGCN_ALLOW_FALLBACK", "1")
    from stmgcn_amd.models import ST_MGCN
    n, M = 32, 2
    rng = np.random.default_rng(1)
    adjs_raw = [torch.from_numpy(_random_sparse_sym_adj(n, 6, rng, weighted=True))
                for _ in range(M)]
    gen = SupportGenerator("chebyshev", 2)
    csr_cpu = [gen.process_csr(a) for a in adjs_raw]
    torch.manual_seed(0)
    model = ST_MGCN(M=M, seq_len=5, n_nodes=n, input_dim=1, lstm_hidden_dim=16,
                    lstm_num_layers=2, gcn_hidden_dim=16,
                    sta_kernel_config={"kernel_type": "chebyshev", "K": 2})
    x = torch.randn(3, 5, n, 1)
    y_cpu = model(x, csr_cpu)

    dev = torch.device("cuda")
    model_g = model.to(dev)
    csr_gpu = [c.to(dev) for c in csr_cpu]
    y_gpu = model_g(x.to(dev), csr_gpu)
    torch.testing.assert_close(y_gpu.cpu(), y_cpu, rtol=1e-4, atol=1e-4)


def test_off_shape_raises_without_optin(monkeypatch):
    """Off-shape on the GPU hip path must be a hard error, not a silent
    torch fallback: H=128 RNN, T=32 gate, C_out=2 head."""
    monkeypatch.delenv("STMGCN_ALLOW_FALLBACK", raising=False)
    from stmgcn_amd.ops import hip_ops
    from stmgcn_amd.ops.hip_ops import FusedRNNFn, branch_fuse_head_hip, contextual_gate_hip
    dev = torch.device("cuda")
    H = 128
    x = torch.randn(8, 4, 1, device=dev, dtype=torch.bfloat16)
    ws = [torch.randn(4 * H, 1, device=dev, dtype=torch.bfloat16),
          torch.randn(4 * H, H, device=dev, dtype=torch.bfloat16),
          torch.randn(4 * H, device=dev, dtype=torch.bfloat16),
          torch.randn(4 * H, device=dev, dtype=torch.bfloat16)]
    h0 = torch.zeros(1, 8, H, device=dev, dtype=torch.bfloat16)
    with pytest.raises(RuntimeError, match="STMGCN_ALLOW_FALLBACK"):
        FusedRNNFn.apply("lstm", x, h0, h0.clone(), False, *ws)
    obs = torch.randn(2, 32, 16, 1, device=dev, dtype=torch.bfloat16)
    g = torch.randn(2, 16, 32, device=dev, dtype=torch.bfloat16)
    fcw = torch.randn(32, 32, device=dev, dtype=torch.bfloat16)
    fcb = torch.randn(32, device=dev, dtype=torch.bfloat16)
    with pytest.raises(RuntimeError, match="STMGCN_ALLOW_FALLBACK"):
        contextual_gate_hip(obs, g, fcw, fcb)
    feats = [torch.randn(2, 16, 8, device=dev, dtype=torch.bfloat16)]
    w2 = torch.randn(2, 8, device=dev, dtype=torch.bfloat16)
    b2 = torch.randn(2, device=dev, dtype=torch.bfloat16)
    with pytest.raises(RuntimeError, match="STMGCN_ALLOW_FALLBACK"):
        branch_fuse_head_hip(feats, w2, b2)
    # with the explicit opt-in the fallback runs and is counted
    monkeypatch.setenv("STMGCN_ALLOW_FALLBACK", "1")
    hip_ops.reset_fallback_count()
    out = FusedRNNFn.apply("lstm", x, h0, h0.clone(), False, *ws)
    assert out.shape == (8, H)
    assert hip_ops.fallback_count() == 1


def test_train_step_gpu_bf16_finite():
    """One full train step at the bench config shape (reduced batch) in bf16:
    finite loss and finite grads through the HIP path."""
    from stmgcn_amd import PRESETS
    from stmgcn_amd.data.synthetic import make_synthetic_dataset
    from stmgcn_amd.models import build_model
    from torch import nn, optim
    cfg = PRESETS["bench-1024"].replace(batch_size=4)
    dev = torch.device("cuda")
    raw = make_synthetic_dataset(n_nodes=cfg.n_nodes, n_steps=32,
                                 m_graphs=cfg.m_graphs, seed=3, day_timesteps=1)
    gen = SupportGenerator(cfg.kernel_type, cfg.cheby_K)
    adjs = [gen.process_csr(torch.from_numpy(raw[k]).float()).to(dev)
            for k in raw if k.endswith("_adj")]
    model = build_model(cfg).to(device=dev, dtype=torch.bfloat16)
    x = torch.randn(4, cfg.seq_len, cfg.n_nodes, 1, device=dev, dtype=torch.bfloat16)
    y = torch.randn(4, cfg.n_nodes, 1, device=dev, dtype=torch.bfloat16)
    opt = optim.Adam(model.parameters(), lr=1e-3)
    loss = nn.MSELoss()(model(x, adjs), y)
    loss.backward()
    for p in model.parameters():
        assert p.grad is None or torch.isfinite(p.grad.float()).all()
    opt.step()
    assert torch.isfinite(loss.float())


def test_mfma_probe_layout():
    """Validate the assumed 16x16x32 bf16 MFMA fragment maps on hardware.
    ASYMMETRIC operands (transpose-detecting, guide §5.4 rule 16)."""
    _C = require_hip()
    torch.manual_seed(0)
    A = (torch.randn(16, 32) * 0.5).bfloat16().cuda()
    B = (torch.arange(32 * 16).float().reshape(32, 16) % 7 - 3).bfloat16().cuda() * 0.3
    D = _C.mfma_probe(A, B)
    ref_D = A.float().cpu() @ B.float().cpu()
    torch.testing.assert_close(D.cpu(), ref_D, rtol=1e-2, atol=1e-2)


def _lstm_oracle_weights(L, cin, H=64, seed=0):
    torch.manual_seed(seed)
    ws = []
    for l in range(L):
        in_l = cin if l == 0 else H
        ws += [torch.randn(4 * H, in_l) * 0.2, torch.randn(4 * H, H) * 0.2,
               torch.randn(4 * H) * 0.1, torch.randn(4 * H) * 0.1]
    return ws


@pytest.mark.parametrize("S,T,L,cin,ret_seq", [
    (64, 8, 3, 1, False), (100, 8, 3, 1, False), (64, 5, 2, 1, True),
    (64, 8, 2, 64, False), (128, 4, 1, 64, True)])
def test_fused_lstm_forward_matches_oracle(S, T, L, cin, ret_seq):
    from stmgcn_amd.ops.hip_ops import FusedLSTMFn
    ws = _lstm_oracle_weights(L, cin)
    x = torch.randn(S, T, cin)
    out_ref = ref.lstm_forward(x, ws, torch.zeros(L, S, 64), torch.zeros(L, S, 64),
                               ret_seq)
    dev = torch.device("cuda")
    ws_g = [w.bfloat16().to(dev) for w in ws]
    out = FusedLSTMFn.apply(x.bfloat16().to(dev), "lstm", ret_seq, False, *ws_g)
    assert out.shape == out_ref.shape
    err = (out.float().cpu() - out_ref).abs().max().item()
    scale = out_ref.abs().max().item() + 1e-6
    assert err / scale < 0.05, f"rel err {err/scale}"


@pytest.mark.parametrize("S,T,L,cin,ret_seq", [
    (64, 8, 3, 1, False), (100, 6, 2, 1, True), (64, 8, 2, 64, False),
    # S=96: ceil(S/64)*64 != ceil(S/32)*32 — regression for the binding
    # padding mismatch vs the kernels' 32-row tiles
    (96, 8, 2, 64, False)])
def test_fused_lstm_backward_matches_oracle(S, T, L, cin, ret_seq):
    from stmgcn_amd.ops.hip_ops import FusedLSTMFn
    ws = _lstm_oracle_weights(L, cin)
    x = torch.randn(S, T, cin)

    x_ref = x.clone().requires_grad_(True)
    ws_ref = [w.clone().requires_grad_(True) for w in ws]
    out_ref = ref.lstm_forward(x_ref, ws_ref, torch.zeros(L, S, 64),
                               torch.zeros(L, S, 64), ret_seq)
    loss_ref = (out_ref.float() ** 2).sum()
    loss_ref.backward()

    dev = torch.device("cuda")
    x_g = x.bfloat16().to(dev).requires_grad_(True)
    ws_g = [w.bfloat16().to(dev).requires_grad_(True) for w in ws]
    out = FusedLSTMFn.apply(x_g, "lstm", ret_seq, True, *ws_g)
    (out.float() ** 2).sum().backward()

    def relerr(a, b):
        return ((a.float().cpu() - b).abs().max() / (b.abs().max() + 1e-6)).item()

    assert relerr(out.detach(), out_ref.detach()) < 0.05
    assert relerr(x_g.grad, x_ref.grad) < 0.08, f"dx {relerr(x_g.grad, x_ref.grad)}"
    for i, (wg, wr) in enumerate(zip(ws_g, ws_ref)):
        e = relerr(wg.grad, wr.grad)
        assert e < 0.08, f"weight {i} grad rel err {e}"


def test_seqsum_permute_matches_torch():
    from stmgcn_amd.ops.hip_ops import SeqsumPermuteFn
    obs = torch.randn(3, 6, 40, 2, device="cuda", requires_grad=True)
    out = SeqsumPermuteFn.apply(obs)
    ref_out = obs.sum(-1).permute(0, 2, 1)
    torch.testing.assert_close(out, ref_out)
    g = torch.randn_like(out)
    d1 = torch.autograd.grad(out, obs, g, retain_graph=True)[0]
    d2 = torch.autograd.grad(ref_out, obs, g)[0]
    torch.testing.assert_close(d1, d2)


@pytest.mark.parametrize("dtype,tol", [(torch.float32, 1e-4), (torch.bfloat16, 3e-2)])
def test_gate_kernel_matches_oracle(dtype, tol):
    from stmgcn_amd.ops.hip_ops import GateFn
    torch.manual_seed(0)
    B, T, N, C = 4, 8, 96, 1
    obs = torch.randn(B, T, N, C)
    g = torch.randn(B, N, T)
    w = torch.randn(T, T) * 0.4
    b = torch.randn(T) * 0.1

    obs_r = obs.clone().requires_grad_(True)
    g_r = g.clone().requires_grad_(True)
    w_r = w.clone().requires_grad_(True)
    b_r = b.clone().requires_grad_(True)
    out_r = ref.contextual_gate(obs_r, g_r, w_r, b_r)
    (out_r.float() ** 2).sum().backward()

    dev = torch.device("cuda")
    obs_g = obs.to(dev, dtype).requires_grad_(True)
    g_g = g.to(dev, dtype).requires_grad_(True)
    w_g = w.to(dev, dtype).requires_grad_(True)
    b_g = b.to(dev, dtype).requires_grad_(True)
    out = GateFn.apply(obs_g, g_g, w_g, b_g)
    (out.float() ** 2).sum().backward()

    def ck(a, b_, what):
        e = ((a.float().cpu() - b_).abs().max() / (b_.abs().max() + 1e-6)).item()
        assert e < tol * 3, f"{what}: rel err {e}"

    ck(out.detach(), out_r.detach(), "out")
    ck(obs_g.grad, obs_r.grad, "dobs")
    ck(g_g.grad, g_r.grad, "dg")
    ck(w_g.grad, w_r.grad, "dw")
    ck(b_g.grad, b_r.grad, "db")


def test_head_kernel_matches_oracle():
    from stmgcn_amd.ops.hip_ops import HeadFn
    torch.manual_seed(0)
    B, N, G, M = 3, 50, 64, 3
    feats = [torch.randn(B, N, G) for _ in range(M)]
    w = torch.randn(1, G) * 0.3
    b = torch.randn(1) * 0.1

    feats_r = [f.clone().requires_grad_(True) for f in feats]
    w_r = w.clone().requires_grad_(True)
    b_r = b.clone().requires_grad_(True)
    y_r = ref.branch_fuse_head(feats_r, w_r, b_r)
    (y_r ** 2).sum().backward()

    dev = torch.device("cuda")
    feats_g = [f.to(dev).requires_grad_(True) for f in feats]
    w_g = w.to(dev).requires_grad_(True)
    b_g = b.to(dev).requires_grad_(True)
    y = HeadFn.apply(w_g, b_g, *feats_g)
    (y ** 2).sum().backward()

    torch.testing.assert_close(y.cpu(), y_r.detach(), rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(w_g.grad.cpu(), w_r.grad, rtol=1e-3, atol=1e-3)
    torch.testing.assert_close(b_g.grad.cpu(), b_r.grad, rtol=1e-3, atol=1e-3)
    for fg, fr in zip(feats_g, feats_r):
        torch.testing.assert_close(fg.grad.cpu(), fr.grad, rtol=1e-3, atol=1e-3)


def test_mse_kernel_matches_torch():
    from stmgcn_amd.ops.hip_ops import FusedMSELossFn
    pred = torch.randn(32, 100, device="cuda", requires_grad=True)
    tgt = torch.randn(32, 100, device="cuda")
    loss = FusedMSELossFn.apply(pred, tgt)
    loss_ref = torch.nn.functional.mse_loss(pred, tgt)
    torch.testing.assert_close(loss, loss_ref, rtol=1e-4, atol=1e-5)
    (d1,) = torch.autograd.grad(loss * 3.0, pred, retain_graph=True)
    (d2,) = torch.autograd.grad(loss_ref * 3.0, pred)
    torch.testing.assert_close(d1, d2, rtol=1e-4, atol=1e-5)


def test_fused_adam_matches_torch_adam():
    """FusedAdam (flat arena, fp32 master) vs torch.optim.Adam on identical
    fp32 params/grads over several steps."""
    from stmgcn_amd.train import FusedAdam
    torch.manual_seed(0)
    dev = torch.device("cuda")
    shapes = [(16, 8), (32,), (7, 5, 3)]
    base = [torch.randn(*s) for s in shapes]
    grads = [[torch.randn(*s) for s in shapes] for _ in range(5)]

    ps_t = [torch.nn.Parameter(b.clone().to(dev)) for b in base]
    opt_t = torch.optim.Adam(ps_t, lr=1e-2, weight_decay=1e-2)
    ps_f = [torch.nn.Parameter(b.clone().to(dev)) for b in base]
    opt_f = FusedAdam(ps_f, lr=1e-2, weight_decay=1e-2)
    for gs in grads:
        for p, g in zip(ps_t, gs):
            p.grad = g.to(dev)
        opt_t.step()
        opt_f.zero_grad()
        for p, g in zip(ps_f, gs):
            p.grad = g.to(dev).to(p.dtype)
        opt_f.step()
    for pt, pf in zip(ps_t, ps_f):
        torch.testing.assert_close(pf.data, pt.data, rtol=1e-5, atol=1e-6)


def test_full_model_bf16_hip_vs_torch_impl():
    """End-to-end forward+backward: HIP kernel path vs STMGCN_IMPL=torch
    eager path on identical bf16 weights/inputs."""
    import os
    from stmgcn_amd.models import ST_MGCN
    n, M, T = 64, 3, 8
    rng = np.random.default_rng(5)
    adjs_raw = [torch.from_numpy(_random_sparse_sym_adj(n, 8, rng, weighted=True))
                for _ in range(M)]
    gen = SupportGenerator("chebyshev", 2)
    torch.manual_seed(0)
    model = ST_MGCN(M=M, seq_len=T, n_nodes=n, input_dim=1, lstm_hidden_dim=64,
                    lstm_num_layers=3, gcn_hidden_dim=64,
                    sta_kernel_config={"kernel_type": "chebyshev", "K": 2})
    dev = torch.device("cuda")
    model = model.to(dev, torch.bfloat16)
    x = torch.randn(4, T, n, 1, device=dev, dtype=torch.bfloat16)

    csr = [gen.process_csr(a).to(dev) for a in adjs_raw]
    y_hip = model(x, csr)
    loss_h = (y_hip.float() ** 2).sum()
    gh = torch.autograd.grad(loss_h, model.fc.weight, retain_graph=False)[0]

    os.environ["STMGCN_IMPL"] = "torch"
    try:
        dense = [gen.process(a).to(dev, torch.bfloat16) for a in adjs_raw]
        y_t = model(x, dense)
        loss_t = (y_t.float() ** 2).sum()
        gt = torch.autograd.grad(loss_t, model.fc.weight)[0]
    finally:
        os.environ["STMGCN_IMPL"] = "hip"
    rel = ((y_hip.float() - y_t.float()).abs().max() /
           (y_t.float().abs().max() + 1e-6)).item()
    assert rel < 0.06, f"forward rel err {rel}"
    relg = ((gh.float() - gt.float()).abs().max() / (gt.float().abs().max() + 1e-6)).item()
    assert relg < 0.12, f"fc grad rel err {relg}"


@pytest.mark.gpu
@pytest.mark.parametrize("rows,M,N", [(5000, 192, 64), (4097, 24, 8),
                                      (1024, 256, 64), (33, 16, 16)])
def test_atb_wgrad_matches_gemm(rows, M, N):
    """wgrad.hip atb kernel: C = A^T @ B + colsum(B) vs fp32 GEMM oracle
    (covers the gconv dW/db shapes incl. zero-padded odd M/N and row tails)."""
    from stmgcn_amd.ops.functional import require_hip
    C = require_hip()
    torch.manual_seed(0)
    A = torch.randn(rows, M, device="cuda", dtype=torch.bfloat16) * 0.5
    B = torch.randn(rows, N, device="cuda", dtype=torch.bfloat16) * 0.5
    out, db = C.atb_wgrad(A, B, True)
    ref = A.float().T @ B.float()
    db_ref = B.float().sum(dim=0)
    torch.testing.assert_close(out, ref, rtol=2e-2, atol=2e-2 * rows ** 0.5)
    torch.testing.assert_close(db, db_ref, rtol=2e-2, atol=2e-2 * rows ** 0.5)


def _gru_oracle(x, ws, L, ret_seq):
    """CPU fp32 GRU oracle (torch native fused GRU, zero init states)."""
    h0 = torch.zeros(L, x.shape[0], 64)
    out, _ = torch._VF.gru(x, h0, [w.float() for w in ws], True, L, 0.0,
                           False, False, True)
    return out if ret_seq else out[:, -1]


def _gru_weights(L, cin, H=64, seed=3):
    torch.manual_seed(seed)
    ws = []
    for l in range(L):
        in_l = cin if l == 0 else H
        ws += [torch.randn(3 * H, in_l) * 0.2, torch.randn(3 * H, H) * 0.2,
               torch.randn(3 * H) * 0.1, torch.randn(3 * H) * 0.1]
    return ws


@pytest.mark.gpu
@pytest.mark.parametrize("S,T,L,cin,ret_seq", [
    (64, 8, 3, 1, False), (100, 8, 2, 1, True), (64, 6, 2, 64, True)])
def test_fused_gru_forward_matches_oracle(S, T, L, cin, ret_seq):
    """GRU on the packed 4-slot LSTM kernels (fused_rnn.hip) vs torch GRU."""
    from stmgcn_amd.ops.hip_ops import FusedLSTMFn
    ws = _gru_weights(L, cin)
    x = torch.randn(S, T, cin)
    out_ref = _gru_oracle(x, ws, L, ret_seq)
    dev = torch.device("cuda")
    ws_g = [w.bfloat16().to(dev) for w in ws]
    out = FusedLSTMFn.apply(x.bfloat16().to(dev), "gru", ret_seq, False, *ws_g)
    assert out.shape == out_ref.shape
    err = (out.float().cpu() - out_ref).abs().max().item()
    scale = out_ref.abs().max().item() + 1e-6
    assert err / scale < 0.05, f"rel err {err/scale}"


@pytest.mark.gpu
@pytest.mark.parametrize("S,T,L,cin,ret_seq", [
    (64, 8, 3, 1, False), (64, 6, 2, 64, True)])
def test_fused_gru_backward_matches_oracle(S, T, L, cin, ret_seq):
    from stmgcn_amd.ops.hip_ops import FusedLSTMFn
    ws = _gru_weights(L, cin)
    x = torch.randn(S, T, cin)

    x_ref = x.clone().requires_grad_(True)
    ws_ref = [w.clone().requires_grad_(True) for w in ws]
    out_ref = _gru_oracle(x_ref, ws_ref, L, ret_seq)
    (out_ref.float() ** 2).sum().backward()

    dev = torch.device("cuda")
    x_g = x.bfloat16().to(dev).requires_grad_(True)
    ws_g = [w.bfloat16().to(dev).requires_grad_(True) for w in ws]
    out = FusedLSTMFn.apply(x_g, "gru", ret_seq, True, *ws_g)
    (out.float() ** 2).sum().backward()

    def relerr(a, b):
        return ((a.float().cpu() - b).abs().max() / (b.abs().max() + 1e-6)).item()

    assert relerr(out.detach(), out_ref.detach()) < 0.05
    assert relerr(x_g.grad, x_ref.grad) < 0.08, f"dx {relerr(x_g.grad, x_ref.grad)}"
    for i, (wg, wr) in enumerate(zip(ws_g, ws_ref)):
        e = relerr(wg.grad, wr.grad)
        assert e < 0.08, f"weight {i} grad rel err {e}"


@pytest.mark.gpu
def test_fused_trainer_path_loss_decreases():
    """End-to-end MI355X trainer path: fused MSE loss + flat-arena FusedAdam
    on the bench config (small batch); loss must drop over 30 steps and all
    parameters must stay finite."""
    from stmgcn_amd import PRESETS
    from stmgcn_amd.graph import SupportGenerator
    from stmgcn_amd.models import build_model
    from stmgcn_amd.ops import mse_loss
    from stmgcn_amd.train import FusedAdam

    dev = torch.device("cuda")
    cfg = PRESETS["bench-1024"].replace(n_nodes=128, batch_size=8)
    torch.manual_seed(0)
    model = build_model(cfg).to(device=dev, dtype=torch.bfloat16)
    gen = SupportGenerator(cfg.kernel_type, cfg.cheby_K, cfg.lambda_max_mode)
    adjs = []
    for _ in range(cfg.m_graphs):
        a = torch.rand(cfg.n_nodes, cfg.n_nodes)
        a = ((a + a.T) > 1.6).float()
        a.fill_diagonal_(0)
        adjs.append(gen.process_csr(a).to(dev))
    x = torch.randn(cfg.batch_size, cfg.seq_len, cfg.n_nodes, 1,
                    device=dev, dtype=torch.bfloat16)
    # learnable target: y = mean over time (so the model can actually fit)
    y = x.mean(dim=1)
    opt = FusedAdam(model.parameters(), lr=5e-3, weight_decay=0.0)
    losses = []
    for _ in range(30):
        opt.zero_grad()
        loss = mse_loss(model(x, adjs), y)
        loss.backward()
        opt.step()
        losses.append(float(loss.detach()))
    assert all(torch.isfinite(p.float()).all() for p in model.parameters())
    assert losses[-1] < 0.5 * losses[0], f"no learning: {losses[0]} -> {losses[-1]}"


@pytest.mark.gpu
def test_rccl_allreduce_inside_hipgraph():
    """Capture an RCCL all-reduce inside a hipGraph and replay it (world=1
    process group on the 1-GPU CI box — the capture path is what multi-rank
    DP would exercise; bench.py keys its multi-rank graph default off this
    capability)."""
    import os
    import torch.distributed as dist
    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29541")
        dist.init_process_group("nccl", rank=0, world_size=1)
    torch.cuda.set_device(0)
    x = torch.ones(1024, device="cuda")
    # warmup on a side stream (NCCL communicator init must precede capture)
    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        dist.all_reduce(x)
    torch.cuda.current_stream().wait_stream(s)
    torch.cuda.synchronize()
    g = torch.cuda.CUDAGraph()
    try:
        with torch.cuda.graph(g):
            dist.all_reduce(x)
            x.mul_(0.5)
    except Exception as e:
        pytest.skip(f"RCCL graph capture unsupported here: {e}")
    x.fill_(2.0)
    g.replay()
    torch.cuda.synchronize()
    torch.testing.assert_close(x, torch.ones_like(x))
    dist.destroy_process_group()


@pytest.mark.gpu
@pytest.mark.parametrize("cell", ["lstm", "gru"])
def test_fused_rnn_fp16_forward_backward(cell):
    """fp16 variants of the fused RNN kernels (the deep-4096 preset dtype)
    vs the fp32 oracle."""
    from stmgcn_amd.ops.hip_ops import FusedLSTMFn
    S, T, L, cin = 96, 8, 2, 64
    if cell == "lstm":
        ws = _lstm_oracle_weights(L, cin, seed=9)
        x = torch.randn(S, T, cin)
        xr = x.clone().requires_grad_(True)
        wr = [w.clone().requires_grad_(True) for w in ws]
        out_ref = ref.lstm_forward(xr, wr, torch.zeros(L, S, 64),
                                   torch.zeros(L, S, 64), False)
    else:
        ws = _gru_weights(L, cin, seed=9)
        x = torch.randn(S, T, cin)
        xr = x.clone().requires_grad_(True)
        wr = [w.clone().requires_grad_(True) for w in ws]
        out_ref = _gru_oracle(xr, wr, L, False)
    (out_ref.float() ** 2).sum().backward()

    xg = x.half().cuda().requires_grad_(True)
    wg = [w.half().cuda().requires_grad_(True) for w in ws]
    out = FusedLSTMFn.apply(xg, cell, False, True, *wg)
    (out.float() ** 2).sum().backward()

    def relerr(a, b):
        return ((a.float().cpu() - b).abs().max() / (b.abs().max() + 1e-6)).item()

    assert relerr(out.detach(), out_ref.detach()) < 0.05
    assert relerr(xg.grad, xr.grad) < 0.1
    for i, (a, b) in enumerate(zip(wg, wr)):
        assert relerr(a.grad, b.grad) < 0.1, f"weight {i}"


@pytest.mark.gpu
def test_trainer_hipgraph_capture_path():
    """ModelTrainer(use_graph=True): capture on second full-size batch,
    replay thereafter; losses stay finite and training still converges to
    the same ballpark as eager."""
    from stmgcn_amd import PRESETS
    from stmgcn_amd.graph import SupportGenerator
    from stmgcn_amd.models import build_model
    from stmgcn_amd.ops import mse_loss
    from stmgcn_amd.train import FusedAdam, ModelTrainer

    dev = torch.device("cuda")
    cfg = PRESETS["bench-1024"].replace(n_nodes=128, batch_size=8)
    gen = SupportGenerator(cfg.kernel_type, cfg.cheby_K, cfg.lambda_max_mode)
    torch.manual_seed(3)
    adjs = []
    for _ in range(cfg.m_graphs):
        a = torch.rand(cfg.n_nodes, cfg.n_nodes)
        a = ((a + a.T) > 1.6).float()
        a.fill_diagonal_(0)
        adjs.append(gen.process_csr(a).to(dev))
    x = torch.randn(16, cfg.seq_len, cfg.n_nodes, 1, device=dev, dtype=torch.bfloat16)
    y = x.mean(dim=1)

    losses = {}
    for graph in (False, True):
        torch.manual_seed(5)
        model = build_model(cfg).to(device=dev, dtype=torch.bfloat16)
        tr = ModelTrainer(model=model, loss=mse_loss, optimizer=FusedAdam,
                          lr=5e-3, wd=0.0, n_epochs=1, use_graph=graph)
        ls = []
        for step in range(12):
            ls.append(tr._train_step(x[:8], y[:8], adjs))  # returns float
        losses[graph] = ls
        assert all(np.isfinite(v) for v in ls)
    assert len(tr._graphs) == 1, "graph was never captured"
    # both runs converge; allow the extra capture-warmup steps' perturbation
    assert losses[True][-1] < 0.7 * losses[True][0]
    assert losses[False][-1] < 0.7 * losses[False][0]
    # a ragged batch must NOT clobber the full-batch graph (per-shape cache)
    full_key = next(iter(tr._graphs))
    v = tr._train_step(x[:5], y[:5], adjs)   # eager warmup for the new shape
    assert np.isfinite(v) and full_key in tr._graphs
    v = tr._train_step(x[:8], y[:8], adjs)   # full shape still replays
    assert np.isfinite(v)


@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float16])
@pytest.mark.parametrize("cin,cout,kernel,K", [
    (64, 64, "chebyshev", 2),   # MFMA fwd epilogue + MFMA bwd U-GEMM
    (8, 8, "chebyshev", 2),     # temporal-gconv shape (scalar epilogues)
    (64, 64, "chebyshev", 3),   # K_s=4: ping/pong recurrence + deep Clenshaw
    (16, 64, "localpool", 2),   # single-support path
    (64, 32, "chebyshev", 2),   # MFMA fwd, scalar bwd (Cout % 32 != 0)
])
def test_cheb_gconv_fused_parity(dtype, cin, cout, kernel, K):
    """Fully fused ChebConv (no support stack, no library GEMMs) vs the fp32
    CPU oracle: forward, dX, dW, db."""
    from stmgcn_amd.ops.hip_ops import ChebGconvFn
    csr = _csr(kernel=kernel, K=K)
    dev = torch.device("cuda")
    csr_d = csr.to(dev)
    B, N = 3, csr.n_nodes
    torch.manual_seed(1)
    x = torch.randn(B, N, cin, device=dev, dtype=dtype, requires_grad=True)
    W = (torch.randn(csr.K_supports * cin, cout, device=dev, dtype=dtype,
                     requires_grad=True) * 0.15).detach().requires_grad_(True)
    b = torch.randn(cout, device=dev, dtype=dtype) * 0.1
    b.requires_grad_(True)
    y = ChebGconvFn.apply(x, W, b, csr_d, "relu")
    (y.float() ** 2).sum().backward()

    x_ref = x.detach().float().cpu().requires_grad_(True)
    W_ref = W.detach().float().cpu().requires_grad_(True)
    b_ref = b.detach().float().cpu().requires_grad_(True)
    y_ref = ref.gconv_mix_csr(csr, x_ref, W_ref, b_ref, "relu")
    (y_ref ** 2).sum().backward()

    def relerr(a, r):
        return ((a.float().cpu() - r).abs().max() / (r.abs().max() + 1e-6)).item()

    assert relerr(y.detach(), y_ref.detach()) < 0.05, "fwd"
    assert relerr(x.grad, x_ref.grad) < 0.08, "dX"
    assert relerr(W.grad, W_ref.grad) < 0.08, "dW"
    assert relerr(b.grad, b_ref.grad) < 0.08, "db"


def test_cheb_gconv_fused_no_bias_no_act():
    """Fused path with bias=None / activation=None (GCN(bias=False))."""
    from stmgcn_amd.ops.hip_ops import ChebGconvFn
    csr = _csr()
    dev = torch.device("cuda")
    csr_d = csr.to(dev)
    torch.manual_seed(2)
    x = torch.randn(2, csr.n_nodes, 64, device=dev, dtype=torch.bfloat16,
                    requires_grad=True)
    W = (torch.randn(csr.K_supports * 64, 64, device=dev,
                     dtype=torch.bfloat16) * 0.15).requires_grad_(True)
    y = ChebGconvFn.apply(x, W, None, csr_d, None)
    (y.float() ** 2).sum().backward()
    x_ref = x.detach().float().cpu().requires_grad_(True)
    W_ref = W.detach().float().cpu().requires_grad_(True)
    y_ref = ref.gconv_mix_csr(csr, x_ref, W_ref, None, None)
    (y_ref ** 2).sum().backward()
    for got, want in [(y.detach(), y_ref.detach()), (x.grad, x_ref.grad),
                      (W.grad, W_ref.grad)]:
        rel = ((got.float().cpu() - want).abs().max() /
               (want.abs().max() + 1e-6)).item()
        assert rel < 0.08


def test_perf_regression_gate():
    """a short bench run must stay above a samples/s floor
    on the pure-hip path — catches silent fallbacks (impl would report
    hip+fallback) and performance cliffs at driver time. The floor (9000)
    is ~20% under the measured ~11k steady state on a fresh MI355X."""
    import json
    import subprocess
    import sys
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, os.path.join(repo, "bench.py"),
         "--steps", "10", "--warmup", "5"],
        capture_output=True, text=True, timeout=600, cwd=repo)
    assert r.returncode == 0, r.stderr[-2000:]
    rec = json.loads([l for l in r.stdout.splitlines() if l.startswith("{")][0])
    assert rec["config"]["impl"] == "hip", rec["config"]
    assert rec["config"]["hipgraph"] is True, "whole-step capture regressed"
    assert rec["value"] > 9000, f"perf cliff: {rec['value']:.0f} samples/s"
