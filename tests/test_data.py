"""Data-pipeline tests: windowing parity with a literal re-statement of the
reference semantics (Data_Container.py:125-146), split lengths, normalization
round trip, loader sharding equivalence."""
import numpy as np
import torch

from stmgcn_amd.data import DataInput, DataGenerator, DeviceLoader, make_synthetic_dataset
from stmgcn_amd.data.container import sliding_windows


def _naive_windows(data, serial_len, daily_len, weekly_len, day_ts):
    """Literal loop transcription of the reference's windowing semantics
    (oracle only — get_feats + get_periodic_skip_seq + weekly|daily|serial
    concat order, oldest-first per component)."""
    start = max(serial_len, daily_len * day_ts, weekly_len * day_ts * 7)
    xs, ys = [], []
    for i in range(start, data.shape[0]):
        comps = []
        if weekly_len > 0:
            p = weekly_len * day_ts * 7
            comps.append(np.stack([data[i - p * w] for w in range(1, weekly_len + 1)])[::-1])
        if daily_len > 0:
            p = daily_len * day_ts
            comps.append(np.stack([data[i - p * d] for d in range(1, daily_len + 1)])[::-1])
        if serial_len > 0:
            comps.append(data[i - serial_len:i])
        xs.append(np.concatenate(comps, axis=0))
        ys.append(data[i])
    return np.stack(xs), np.stack(ys)


def test_sliding_windows_match_reference_semantics():
    rng = np.random.default_rng(0)
    data = rng.standard_normal((400, 6, 1)).astype(np.float32)
    for obs in [(3, 1, 1), (4, 2, 2), (5, 0, 0), (2, 3, 0)]:
        got_x, got_y = sliding_windows(data, *obs, day_timesteps=24)
        exp_x, exp_y = _naive_windows(data, *obs, day_ts=24)
        np.testing.assert_allclose(got_x, exp_x)
        np.testing.assert_allclose(got_y, exp_y)
        assert got_x.shape[1] == sum(obs)


def test_reference_split_lengths():
    """Defaults (0101/0630/0701/0731, dt=1, val 0.2) -> SURVEY-verified
    {train: 3476, validate: 868, test: 744}."""
    gen = DataGenerator(dt=1, obs_len=(3, 1, 1),
                        train_test_dates=["0101", "0630", "0701", "0731"],
                        val_ratio=0.2)
    assert gen.mode_len == {"train": 3476, "validate": 868, "test": 744}
    assert gen.day_timesteps == 24


def test_first_sample_indices():
    """First window anchor = t=168 at defaults: serial t-3..t-1, daily [t-24],
    weekly [t-168]."""
    data = np.arange(400, dtype=np.float32).reshape(-1, 1, 1)
    x, y = sliding_windows(data, 3, 1, 1, 24)
    assert y[0, 0, 0] == 168.0
    np.testing.assert_allclose(x[0, :, 0, 0], [0.0, 144.0, 165.0, 166.0, 167.0])


def test_minmax_normalize_roundtrip():
    di = DataInput(M_adj=1, data_dir="", norm_opt=True)
    x = np.random.default_rng(0).uniform(5, 50, size=(100, 4, 1))
    nx = di.minmax_normalize(x)
    assert nx.min() == -1.0 and nx.max() == 1.0
    np.testing.assert_allclose(di.minmax_denormalize(nx), x, rtol=1e-12)


def test_load_dict_gates_adjacency_count():
    raw = make_synthetic_dataset(n_nodes=16, n_steps=200, m_graphs=3)
    di = DataInput(M_adj=2, data_dir="", norm_opt=False)
    out = di.load_dict(raw)
    assert "taxi" in out and "neighbor_adj" in out


def test_device_loader_shard_equivalence():
    """DP=2 block sharding: concat of rank slices == the DP=1 global batch."""
    x = torch.arange(40.0).reshape(20, 2)
    y = torch.arange(20.0).reshape(20, 1)
    full = DeviceLoader(x, y, batch_size=8, rank=0, world_size=1)
    r0 = DeviceLoader(x, y, batch_size=4, rank=0, world_size=2)
    r1 = DeviceLoader(x, y, batch_size=4, rank=1, world_size=2)
    fb = list(full)
    b0, b1 = list(r0), list(r1)
    assert len(b0) == len(b1) == 2  # 20 // 8
    for i in range(2):
        torch.testing.assert_close(torch.cat([b0[i][0], b1[i][0]]), fb[i][0])
        torch.testing.assert_close(torch.cat([b0[i][1], b1[i][1]]), fb[i][1])


def test_device_loader_shuffle_deterministic():
    x = torch.arange(30.0).reshape(30, 1)
    y = x.clone()
    ld = DeviceLoader(x, y, batch_size=10, shuffle=True, seed=7)
    ld.set_epoch(3)
    a = [bx.clone() for bx, _ in ld]
    b = [bx.clone() for bx, _ in ld]
    for t1, t2 in zip(a, b):
        torch.testing.assert_close(t1, t2)


def test_get_data_loader_short_synthetic_rescales():
    raw = make_synthetic_dataset(n_nodes=9, n_steps=24 * 30, m_graphs=1)
    di = DataInput(M_adj=1, data_dir="", norm_opt=True)
    data = di.load_dict(raw)
    gen = DataGenerator(dt=1, obs_len=(3, 1, 1),
                        train_test_dates=["0101", "0630", "0701", "0731"])
    loaders = gen.get_data_loader(data, batch_size=16, device="cpu")
    total = sum(gen.mode_len.values())
    assert total == 24 * 30 - 168
    for mode in ["train", "validate", "test"]:
        assert len(loaders[mode]) > 0


def test_device_loader_drop_last_when_sharded():
    """Sharded (world>1) loaders DROP the ragged final global batch — a
    partial batch cannot be split evenly across ranks and would desync the
    per-rank step counts (the all-reduce would deadlock). Unsharded loaders
    keep it (reference DataLoader semantics)."""
    import torch
    from stmgcn_amd.data.container import DeviceLoader
    n, b, world = 21, 4, 2                 # global batch 8 -> 2 full, 5 dropped
    x = torch.arange(n, dtype=torch.float32).reshape(n, 1)
    y = x.clone()
    counts = []
    for rank in range(world):
        dl = DeviceLoader(x, y, batch_size=b, rank=rank, world_size=world)
        batches = list(dl)
        counts.append(len(batches))
        assert all(xb.shape[0] == b for xb, _ in batches)   # only full batches
    assert counts[0] == counts[1] == len(dl) == n // (b * world)
    # unsharded: ragged tail kept
    dl1 = DeviceLoader(x, y, batch_size=b)
    sizes = [xb.shape[0] for xb, _ in dl1]
    assert sum(sizes) == n and sizes[-1] == n % b


def test_std_normalize_pair_and_load_prints(capsys, tmp_path):
    """Reference API-surface parity: the (dead in the reference) std
    normalize/denormalize pair round-trips, and load_data reproduces the
    reference's prints (Data_Container.py:15-17,33,43-51)."""
    import numpy as np
    from stmgcn_amd.data import DataInput
    rng = np.random.default_rng(0)
    x = rng.normal(5.0, 2.0, size=(50, 4, 1))
    di = DataInput(M_adj=1, data_dir="", norm_opt=True)
    z = di.std_normalize(x)
    assert abs(z.mean()) < 1e-9 and abs(z.std() - 1.0) < 1e-9
    np.testing.assert_allclose(di.std_denormalize(z), x, rtol=1e-12)
    out = capsys.readouterr().out
    assert out.startswith("mean:") and "std:" in out

    npz = tmp_path / "d.npz"
    np.savez(npz, taxi=x, neighbor_adj=np.eye(4))
    di2 = DataInput(M_adj=1, data_dir=str(npz), norm_opt=True)
    di2.load_data()
    out = capsys.readouterr().out
    assert "Loading data..." in out
    assert "Available keys:" in out
    assert "min:" in out and "max:" in out
