"""Model-level tests: 56-key checkpoint schema, 288,587-parameter budget,
forward shapes, CSR==dense path equivalence, gradient flow, deep variant."""
import numpy as np
import pytest
import torch

from stmgcn_amd import PRESETS
from stmgcn_amd.data.synthetic import _random_sparse_sym_adj
from stmgcn_amd.graph import SupportGenerator
from stmgcn_amd.models import ST_MGCN, StackedSTMGCN, build_model


def _adjs(n, m=3, seed=0):
    rng = np.random.default_rng(seed)
    return [torch.from_numpy(_random_sparse_sym_adj(n, 6, rng, weighted=True))
            for _ in range(m)]


def _model(n_nodes=58, seq_len=5, M=3):
    torch.manual_seed(0)
    return ST_MGCN(M=M, seq_len=seq_len, n_nodes=n_nodes, input_dim=1,
                   lstm_hidden_dim=64, lstm_num_layers=3, gcn_hidden_dim=64,
                   sta_kernel_config={"kernel_type": "chebyshev", "K": 2})


def test_parameter_budget_and_state_dict_schema():
    """SURVEY-verified ground truth: 288,587 params, 56 state_dict keys with
    the reference's exact naming."""
    model = _model()
    n_params = sum(p.numel() for p in model.parameters())
    assert n_params == 288587
    sd = model.state_dict()
    assert len(sd) == 56
    for m in range(3):
        for key in [f"rnn_list.{m}.gconv_temporal_feats.W",
                    f"rnn_list.{m}.gconv_temporal_feats.b",
                    f"rnn_list.{m}.fc.weight", f"rnn_list.{m}.fc.bias",
                    f"gcn_list.{m}.W", f"gcn_list.{m}.b"]:
            assert key in sd, key
        for l in range(3):
            for p in ["weight_ih", "weight_hh", "bias_ih", "bias_hh"]:
                assert f"rnn_list.{m}.lstm.{p}_l{l}" in sd
    assert "fc.weight" in sd and "fc.bias" in sd
    # shape spot checks vs the reference schema
    assert sd["rnn_list.0.gconv_temporal_feats.W"].shape == (15, 5)
    assert sd["rnn_list.0.lstm.weight_ih_l0"].shape == (256, 1)
    assert sd["rnn_list.0.lstm.weight_hh_l1"].shape == (256, 64)
    assert sd["gcn_list.0.W"].shape == (192, 64)
    assert sd["fc.weight"].shape == (1, 64)


def test_forward_shape():
    model = _model()
    gen = SupportGenerator("chebyshev", 2)
    adjs = [gen.process(a) for a in _adjs(58)]
    x = torch.randn(4, 5, 58, 1)
    y = model(x, adjs)
    assert y.shape == (4, 58, 1)


def test_support_count_contract():
    assert ST_MGCN.get_support_K({"kernel_type": "chebyshev", "K": 2}) == 3
    assert ST_MGCN.get_support_K({"kernel_type": "localpool", "K": 2}) == 1
    assert ST_MGCN.get_support_K({"kernel_type": "random_walk_diffusion", "K": 2}) == 5


def test_csr_and_dense_paths_match():
    """The CSR in-kernel-recurrence path must equal the dense support-stack
    path bit-for-fp32-tolerance (forward AND backward)."""
    model = _model(n_nodes=32)
    gen = SupportGenerator("chebyshev", 2)
    raw = _adjs(32)
    dense = [gen.process(a) for a in raw]
    csr = [gen.process_csr(a) for a in raw]
    x = torch.randn(3, 5, 32, 1, requires_grad=True)
    y_dense = model(x, dense)
    g_dense = torch.autograd.grad(y_dense.sum(), x, retain_graph=False)[0]
    x2 = x.detach().clone().requires_grad_(True)
    y_csr = model(x2, csr)
    g_csr = torch.autograd.grad(y_csr.sum(), x2)[0]
    torch.testing.assert_close(y_csr, y_dense, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(g_csr, g_dense, rtol=1e-4, atol=1e-5)


def test_grad_flows_to_all_params():
    model = _model(n_nodes=16)
    gen = SupportGenerator("chebyshev", 2)
    adjs = [gen.process(a) for a in _adjs(16)]
    y = model(torch.randn(2, 5, 16, 1), adjs)
    y.sum().backward()
    for name, p in model.named_parameters():
        assert p.grad is not None, name
        assert p.grad.abs().sum() > 0 or "bias" in name or name.endswith(".b"), name


def test_gate_weight_tying():
    """One shared FC applied twice in the gate (quirk 2): its grad must
    receive contributions from both applications (non-None and used twice ->
    grad differs from a single-application clone)."""
    from stmgcn_amd.ops.reference_impl import contextual_gate
    torch.manual_seed(0)
    obs = torch.randn(2, 5, 7, 1)
    g = torch.randn(2, 7, 5)
    w = torch.randn(5, 5, requires_grad=True)
    b = torch.randn(5, requires_grad=True)
    out = contextual_gate(obs, g, w, b)
    assert out.shape == obs.shape
    out.sum().backward()
    assert w.grad is not None and b.grad is not None
    # oracle formula check
    x_seq = obs.sum(-1).permute(0, 2, 1)
    z = (g + x_seq).mean(1)
    s = torch.sigmoid(torch.relu(z @ w.T + b) @ w.T + b)
    expected = obs * s[:, :, None, None]
    torch.testing.assert_close(out, expected, rtol=1e-5, atol=1e-6)


def test_lstm_matches_nn_lstm():
    """CGRNNCellParams CPU math == torch.nn.LSTM on identical weights."""
    from stmgcn_amd.models import CGRNNCellParams
    torch.manual_seed(0)
    cell = CGRNNCellParams("lstm", 3, 16, 2)
    ref = torch.nn.LSTM(3, 16, num_layers=2, batch_first=True)
    ref.load_state_dict(cell.state_dict())
    x = torch.randn(5, 7, 3)
    h0 = torch.zeros(2, 5, 16)
    c0 = torch.zeros(2, 5, 16)
    out_ref, _ = ref(x, (h0, c0))
    got = cell(x, h0, c0, return_sequences=True)
    torch.testing.assert_close(got, out_ref, rtol=1e-5, atol=1e-6)


def test_gru_matches_nn_gru():
    from stmgcn_amd.models import CGRNNCellParams
    torch.manual_seed(0)
    cell = CGRNNCellParams("gru", 3, 16, 2)
    ref = torch.nn.GRU(3, 16, num_layers=2, batch_first=True)
    ref.load_state_dict(cell.state_dict())
    x = torch.randn(5, 7, 3)
    h0 = torch.zeros(2, 5, 16)
    out_ref, _ = ref(x, h0)
    got = cell(x, h0, None, return_sequences=True)
    torch.testing.assert_close(got, out_ref, rtol=1e-5, atol=1e-6)


def test_deep_variant_forward():
    cfg = PRESETS["deep-4096"].replace(n_nodes=16, batch_size=2)
    model = build_model(cfg)
    assert isinstance(model, StackedSTMGCN)
    gen = SupportGenerator(cfg.kernel_type, cfg.cheby_K)
    adjs = [gen.process(a) for a in _adjs(16, m=cfg.m_graphs)]
    y = model(torch.randn(2, cfg.seq_len, 16, 1), adjs)
    assert y.shape == (2, 16, 1)
    y.sum().backward()


def test_checkpoint_roundtrip(tmp_path):
    model = _model(n_nodes=16)
    path = tmp_path / "ck.pkl"
    torch.save({"epoch": 3, "state_dict": model.state_dict()}, path)
    model2 = _model(n_nodes=16)
    ck = torch.load(path, weights_only=False)
    model2.load_state_dict(ck["state_dict"])
    for (n1, p1), (n2, p2) in zip(model.named_parameters(), model2.named_parameters()):
        assert n1 == n2
        torch.testing.assert_close(p1, p2)


def test_nn_module_activation_accepted():
    """The reference wires gconv_activation=nn.ReLU (Main.py:64); the ctor
    accepts the module form and it matches the string form exactly."""
    from torch import nn
    torch.manual_seed(0)
    m1 = ST_MGCN(M=1, seq_len=4, n_nodes=12, input_dim=1, lstm_hidden_dim=8,
                 lstm_num_layers=1, gcn_hidden_dim=8,
                 sta_kernel_config={"kernel_type": "chebyshev", "K": 2},
                 gconv_activation=nn.ReLU)
    torch.manual_seed(0)
    m2 = ST_MGCN(M=1, seq_len=4, n_nodes=12, input_dim=1, lstm_hidden_dim=8,
                 lstm_num_layers=1, gcn_hidden_dim=8,
                 sta_kernel_config={"kernel_type": "chebyshev", "K": 2},
                 gconv_activation="relu")
    gen = SupportGenerator("chebyshev", 2)
    adjs = [gen.process(a) for a in _adjs(12, m=1)]
    x = torch.randn(2, 4, 12, 1)
    torch.testing.assert_close(m1(x, adjs), m2(x, adjs))
    with pytest.raises(ValueError, match="activation"):
        ST_MGCN(M=1, seq_len=4, n_nodes=12, input_dim=1, lstm_hidden_dim=8,
                lstm_num_layers=1, gcn_hidden_dim=8,
                sta_kernel_config={"kernel_type": "chebyshev", "K": 2},
                gconv_activation=torch.nn.Tanh)
