"""Multi-process DP correctness on CPU (gloo, world_size 2) — SURVEY §4 item 5.

The reference has zero distributed code; these tests pin the
new framework's DP semantics without a GPU:
  - GradReducer DP=2 gradients == single-process full-batch gradients
  - k optimizer steps under DP=2 == k steps at DP=1 (same seed, same data)
  - bucket partitioning covers every parameter exactly once

The same GradReducer runs over RCCL on the GPU box (backend selection is the
only difference — parallel/ddp.py:42-54), so gloo equivalence here is the CI
proxy for the xGMI path.
"""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from stmgcn_amd import PRESETS
from stmgcn_amd.graph import SupportGenerator
from stmgcn_amd.models import build_model
from stmgcn_amd.parallel import GradReducer


def _free_port():
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _make_problem(seed=0):
    """Tiny ST_MGCN + fixed synthetic batch, deterministic."""
    cfg = PRESETS["cpu-small"].replace(n_nodes=16, seq_len=4, batch_size=8,
                                       lstm_hidden_dim=8, gcn_hidden_dim=8,
                                       lstm_num_layers=2, m_graphs=2,
                                       obs_len=[4, 0, 0])
    torch.manual_seed(seed)
    model = build_model(cfg)
    gen = SupportGenerator(cfg.kernel_type, cfg.cheby_K, cfg.lambda_max_mode)
    g = torch.Generator().manual_seed(seed + 1)
    adjs = []
    for _ in range(cfg.m_graphs):
        a = torch.rand(cfg.n_nodes, cfg.n_nodes, generator=g)
        a = ((a + a.T) > 1.4).float()
        a.fill_diagonal_(0)
        adjs.append(gen.process(a))
    x = torch.randn(cfg.batch_size, cfg.seq_len, cfg.n_nodes, cfg.input_dim,
                    generator=g)
    y = torch.randn(cfg.batch_size, cfg.n_nodes, cfg.input_dim, generator=g)
    return cfg, model, adjs, x, y


def _reference_grads_and_steps(n_steps=3):
    """DP=1 oracle: full-batch grads after step 1, params after n_steps."""
    cfg, model, adjs, x, y = _make_problem()
    opt = torch.optim.Adam(model.parameters(), lr=cfg.lr,
                           weight_decay=cfg.weight_decay)
    first_grads = None
    for s in range(n_steps):
        opt.zero_grad()
        loss = torch.nn.functional.mse_loss(model(x, adjs), y)
        loss.backward()
        if s == 0:
            first_grads = [p.grad.detach().clone() for p in model.parameters()]
        opt.step()
    final = [p.detach().clone() for p in model.parameters()]
    return first_grads, final


def _worker(rank, world, port, bucket_cap_mb, n_steps, out_path):
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        cfg, model, adjs, x, y = _make_problem()
        # per-rank contiguous shard of the batch (mirrors DeviceLoader)
        B = x.shape[0]
        shard = slice(rank * B // world, (rank + 1) * B // world)
        xs, ys = x[shard], y[shard]
        reducer = GradReducer(model, bucket_cap_mb=bucket_cap_mb)
        opt = torch.optim.Adam(model.parameters(), lr=cfg.lr,
                               weight_decay=cfg.weight_decay)
        first_grads = None
        for s in range(n_steps):
            reducer.zero_grad()
            # mean loss over the GLOBAL batch = mean over shard, then the
            # reducer's 1/world rescale finishes the global mean
            loss = torch.nn.functional.mse_loss(model(xs, adjs), ys)
            loss.backward()
            reducer.reduce()
            if s == 0:
                first_grads = [p.grad.detach().clone() for p in model.parameters()]
            opt.step()
        if rank == 0:
            torch.save((first_grads,
                        [p.detach().clone() for p in model.parameters()],
                        [b.numel() for b in reducer.flat]), out_path)
    finally:
        dist.destroy_process_group()


def _run_dp(tmp_path, world=2, bucket_cap_mb=25.0, n_steps=3):
    ctx = mp.get_context("spawn")
    out_path = str(tmp_path / "dp_result.pt")
    port = _free_port()
    procs = [ctx.Process(target=_worker, args=(r, world, port, bucket_cap_mb,
                                               n_steps, out_path))
             for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0, f"worker exited with {p.exitcode}"
    return torch.load(out_path, weights_only=False)


@pytest.mark.parametrize("bucket_cap_mb", [25.0, 0.001])
def test_dp2_matches_dp1(bucket_cap_mb, tmp_path):
    """DP=2 (sharded batch + all-reduce) == DP=1 full batch: gradients after
    the first step and parameters after 3 Adam steps. bucket_cap 0.01 MB
    forces many buckets (exercises the overlap hook path end-to-end)."""
    ref_grads, ref_final = _reference_grads_and_steps()
    dp_grads, dp_final, bucket_sizes = _run_dp(tmp_path, bucket_cap_mb=bucket_cap_mb)
    n_params = sum(g.numel() for g in ref_grads)
    assert sum(bucket_sizes) == n_params  # every param in exactly one bucket
    if bucket_cap_mb < 0.01:
        assert len(bucket_sizes) > 1
    for rg, dg in zip(ref_grads, dp_grads):
        torch.testing.assert_close(dg, rg, rtol=1e-5, atol=1e-6)
    for rp, dp_ in zip(ref_final, dp_final):
        torch.testing.assert_close(dp_, rp, rtol=1e-5, atol=1e-6)


def test_bucket_partition_reverse_order():
    """Buckets fill in reverse parameter order and partition all params."""
    _, model, _, _, _ = _make_problem()
    # no process group: GradReducer world==1, no hooks/broadcast, layout only
    red = GradReducer(model, bucket_cap_mb=0.001)
    params = [p for p in model.parameters() if p.requires_grad]
    flat_order = [p for b in red.buckets for p in b]
    assert flat_order == list(reversed(params))
    assert sum(p.numel() for p in flat_order) == sum(p.numel() for p in params)
    # every grad view shares storage with its bucket's flat buffer
    for bucket, buf in zip(red.buckets, red.flat):
        for p in bucket:
            assert p.grad.data_ptr() >= buf.data_ptr()
            assert p.grad.data_ptr() < buf.data_ptr() + buf.numel() * buf.element_size()
