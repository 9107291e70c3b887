"""Property-based tests (hypothesis) for graph preprocessing and windowing —
invariants that hold for ANY valid input, complementing the fixed-seed parity
tests."""
import numpy as np
import torch
from hypothesis import given, settings, strategies as st

from stmgcn_amd.data.container import sliding_windows
from stmgcn_amd.graph import SupportGenerator
from stmgcn_amd.graph.preprocess import dense_to_csr, chebyshev_polynomials


def _rand_adj(n, seed, density=0.3):
    rng = np.random.default_rng(seed)
    a = (rng.random((n, n)) < density).astype(np.float32) * rng.random((n, n)).astype(np.float32)
    a = np.triu(a, 1)
    a = a + a.T
    return torch.from_numpy(a)


@settings(max_examples=25, deadline=None)
@given(n=st.integers(4, 40), seed=st.integers(0, 10_000), K=st.integers(1, 4))
def test_chebyshev_supports_invariants(n, seed, K):
    """T_0 == I; T_1 == G; supports of a symmetric generator are symmetric;
    CSR round-trip reproduces the dense generator exactly."""
    A = _rand_adj(n, seed)
    gen = SupportGenerator("chebyshev", K)
    stack = gen.process(A)
    assert stack.shape == (K + 1, n, n)
    torch.testing.assert_close(stack[0], torch.eye(n))
    G = gen.generator(A)
    torch.testing.assert_close(stack[1], G)
    for k in range(K + 1):
        torch.testing.assert_close(stack[k], stack[k].T, rtol=1e-4, atol=1e-5)
    csr = gen.process_csr(A)
    torch.testing.assert_close(csr.dense_generator(), G)
    torch.testing.assert_close(csr.dense_supports(), stack, rtol=1e-5, atol=1e-5)


@settings(max_examples=25, deadline=None)
@given(n=st.integers(3, 30), seed=st.integers(0, 10_000))
def test_csr_transpose_consistency(n, seed):
    """The stored G^T CSR is exactly the transpose of the G CSR."""
    A = _rand_adj(n, seed)
    G = SupportGenerator("chebyshev", 2).generator(A)
    csr = dense_to_csr(G)
    Gt = torch.zeros(n, n)
    for i in range(n):
        s, e = int(csr.row_ptr_t[i]), int(csr.row_ptr_t[i + 1])
        Gt[i, csr.col_idx_t[s:e].long()] = csr.vals_t[s:e]
    torch.testing.assert_close(Gt, G.T)


@settings(max_examples=20, deadline=None)
@given(steps=st.integers(180, 400), n=st.integers(2, 10),
       serial=st.integers(1, 6), daily=st.integers(0, 2), weekly=st.integers(0, 1))
def test_sliding_windows_contents(steps, n, serial, daily, weekly):
    """Window slices reproduce the reference construction: concat order
    weekly|daily|serial, periodic sequences oldest-first, y = data[i]
    (reference Data_Container.py:82-86,125-146)."""
    if serial + daily + weekly == 0:
        return
    data = np.arange(steps * n, dtype=np.float32).reshape(steps, n, 1)
    x, y = sliding_windows(data, serial, daily, weekly, day_timesteps=24)
    start = max(serial, daily * 24, weekly * 24 * 7)
    assert x.shape[0] == steps - start == y.shape[0]
    T = serial + daily + weekly
    assert x.shape[1] == T
    i = start  # first sample
    np.testing.assert_array_equal(y[0], data[i])
    expect = []
    for w in range(weekly, 0, -1):
        expect.append(data[i - (weekly * 24 * 7) * w])
    for d in range(daily, 0, -1):
        expect.append(data[i - (daily * 24) * d])
    for s_ in range(serial, 0, -1):
        expect.append(data[i - s_])
    np.testing.assert_array_equal(x[0], np.stack(expect))


@settings(max_examples=15, deadline=None)
@given(n=st.integers(4, 32), seed=st.integers(0, 1000))
def test_lambda_max_modes(n, seed):
    """power_iteration lambda is a valid spectral bound: scaled Laplacian
    eigenvalues lie in [-1, 1] (up to tolerance)."""
    A = _rand_adj(n, seed, density=0.5)
    gen = SupportGenerator("chebyshev", 2, lambda_max_mode="power_iteration")
    G = gen.generator(A)  # (2/lmax) L - I
    ev = torch.linalg.eigvalsh(G.double())
    # power iteration approximates lmax from below on near-degenerate
    # spectra -> allow a few percent overshoot (Chebyshev basis tolerates it)
    assert ev.max().item() <= 1.05
    assert ev.min().item() >= -1.05
