"""Data loading, normalization, windowing, device-resident datasets.

Parity surface: reference Data_Container.py. Behavioral contract replicated:
  - min-max normalization to [-1, 1] with GLOBAL min/max, denorm state held on
    the DataInput instance (Data_Container.py:31-41, quirk 13)
  - sliding windows start at max(serial, daily*day_ts, weekly*day_ts*7)
    (Data_Container.py:127)
  - periodic skip sequences stride p_steps = len*day_ts (*7 weekly), emitted
    oldest-first via reversal (Data_Container.py:135-146)
  - window concat order weekly | daily | serial, zero-length components
    dropped (Data_Container.py:82-86, quirk 12)
  - date->split-length mapping via a calendar year's date list
    (Data_Container.py:102-112); the reference's start_idx day/timestep unit
    bug (quirk 4) is NOT replicated — splits here start at window index 0
    exactly as the reference's defaults reduce to
  - splits are moved to the target device EAGERLY at dataset construction
    (Data_Container.py:88-89): steady-state training does no H2D traffic;
    on MI355X the whole dataset sits in 288 GB HBM3E

MI355X-native additions:
  - vectorized windowing (the reference uses an O(T) python append loop,
    Data_Container.py:128-133) built on views — no window materialization
    until the device copy
  - DeviceLoader: zero-copy batch slicing of device-resident splits with
    deterministic block sharding for data parallelism. Rank r of w takes the
    contiguous sub-slice [r*b, (r+1)*b) of every global batch, so DP=w with
    per-rank batch b is sample-identical to DP=1 with batch w*b (the SURVEY §4
    test-strategy equivalence harness relies on this).
"""
from __future__ import annotations

import datetime
from typing import Dict, List, Optional, Sequence

import numpy as np
import torch


class DataInput:
    """npz loading + normalization (reference Data_Container.py:8-51)."""

    ADJ_KEYS = ["neighbor_adj", "trans_adj", "semantic_adj"]

    def __init__(self, M_adj: int, data_dir: str, norm_opt: bool = True):
        self.M_sta = M_adj
        self.data_dir = data_dir
        self.norm_opt = norm_opt
        self._min: Optional[float] = None
        self._max: Optional[float] = None

    def load_data(self) -> Dict[str, np.ndarray]:
        print("Loading data...")                 # reference Data_Container.py:15
        npz = np.load(self.data_dir)
        print("Available keys:", list(npz.keys()))
        dataset: Dict[str, np.ndarray] = {}
        dataset["taxi"] = self.minmax_normalize(npz["taxi"]) if self.norm_opt else npz["taxi"]
        for k in range(self.M_sta):
            key = self.ADJ_KEYS[k]
            dataset[key] = npz[key]
        return dataset

    def load_dict(self, data: Dict[str, np.ndarray]) -> Dict[str, np.ndarray]:
        """Same as load_data but from an in-memory dict (synthetic pipelines)."""
        out = dict(data)
        if self.norm_opt:
            out["taxi"] = self.minmax_normalize(out["taxi"])
        return out

    def minmax_normalize(self, x: np.ndarray) -> np.ndarray:
        self._max, self._min = float(x.max()), float(x.min())
        print("min:", self._min, "max:", self._max)  # Data_Container.py:33
        x = (x - self._min) / (self._max - self._min)
        return 2.0 * x - 1.0

    def minmax_denormalize(self, x: np.ndarray) -> np.ndarray:
        if self._max is None:
            raise RuntimeError("minmax_denormalize called before normalize")
        return (self._max - self._min) * ((x + 1.0) / 2.0) + self._min

    # std pair: dead code in the reference (Data_Container.py:43-51 — nothing
    # calls it) but part of the DataInput API surface; reproduced for parity.
    def std_normalize(self, x: np.ndarray) -> np.ndarray:
        self._mean, self._std = float(x.mean()), float(x.std())
        print("mean:", round(self._mean, 4), "std:", round(self._std, 4))
        return (x - self._mean) / self._std

    def std_denormalize(self, x: np.ndarray) -> np.ndarray:
        return x * self._std + self._mean


def sliding_windows(data: np.ndarray, serial_len: int, daily_len: int,
                    weekly_len: int, day_timesteps: int):
    """Vectorized equivalent of the reference's get_feats + get_periodic_skip_seq
    (Data_Container.py:125-146). Returns (x_seq, y) where x_seq is the
    weekly|daily|serial concat (n_samples, T, N, C), oldest-first per component.
    """
    T_total = data.shape[0]
    start = max(serial_len, daily_len * day_timesteps, weekly_len * day_timesteps * 7)
    anchors = np.arange(start, T_total)                       # one sample per anchor t
    offsets: List[np.ndarray] = []
    if weekly_len > 0:
        p = weekly_len * day_timesteps * 7
        # d = weekly_len .. 1 (reference builds d=1..len then reverses)
        offsets.append(-p * np.arange(weekly_len, 0, -1))
    if daily_len > 0:
        p = daily_len * day_timesteps
        offsets.append(-p * np.arange(daily_len, 0, -1))
    if serial_len > 0:
        offsets.append(np.arange(-serial_len, 0))
    off = np.concatenate(offsets)                             # (T_win,)
    idx = anchors[:, None] + off[None, :]                     # (S, T_win)
    x_seq = data[idx]                                         # (S, T_win, N, C)
    y = data[anchors]                                         # (S, N, C)
    return x_seq, y


class TaxiDataset(torch.utils.data.Dataset):
    """Device-resident windowed split (reference Data_Container.py:54-90)."""

    def __init__(self, device, x_seq: np.ndarray, y: np.ndarray, mode: str,
                 mode_len: Dict[str, int], start_idx: int = 0,
                 dtype: torch.dtype = torch.float32):
        self.mode = mode
        self.mode_len = mode_len
        ofs = start_idx
        if mode == "validate":
            ofs += mode_len["train"]
        elif mode == "test":
            ofs += mode_len["train"] + mode_len["validate"]
        n = mode_len[mode]
        self.x = torch.from_numpy(np.ascontiguousarray(x_seq[ofs:ofs + n])).to(dtype).to(device)
        self.y = torch.from_numpy(np.ascontiguousarray(y[ofs:ofs + n])).to(dtype).to(device)

    def __len__(self):
        return self.mode_len[self.mode]

    def __getitem__(self, i):
        return self.x[i], self.y[i]


class DeviceLoader:
    """Batch iterator over device-resident tensors — zero per-step H2D copies,
    zero collation, deterministic DP block sharding.

    Global batches are consecutive slices of the (time-ordered) split; rank r
    of world w reads the contiguous [r*b, (r+1)*b) sub-slice of each global
    batch (b = per-rank batch). drop_last applies only when sharded (a ragged
    final batch cannot be split evenly across ranks).
    """

    def __init__(self, x: torch.Tensor, y: torch.Tensor, batch_size: int,
                 rank: int = 0, world_size: int = 1, shuffle: bool = False,
                 seed: int = 0):
        assert x.shape[0] == y.shape[0]
        self.x, self.y = x, y
        self.batch_size = batch_size            # per-rank
        self.rank, self.world = rank, world_size
        self.global_batch = batch_size * world_size
        self.shuffle = shuffle
        self.seed = seed
        self.epoch = 0
        n = x.shape[0]
        if world_size > 1:
            self.n_batches = n // self.global_batch
        else:
            self.n_batches = (n + batch_size - 1) // batch_size

    def set_epoch(self, epoch: int):
        self.epoch = epoch

    def __len__(self):
        return self.n_batches

    def __iter__(self):
        n = self.x.shape[0]
        if self.shuffle:
            g = torch.Generator()
            g.manual_seed(self.seed + self.epoch)
            perm = torch.randperm(n, generator=g).to(self.x.device)
        for i in range(self.n_batches):
            lo = i * self.global_batch + self.rank * self.batch_size
            hi = min(lo + self.batch_size, n)
            if self.shuffle:
                sel = perm[lo:hi]
                yield self.x[sel], self.y[sel]
            else:
                yield self.x[lo:hi], self.y[lo:hi]


class DataGenerator:
    """Date-keyed splits + loaders (reference Data_Container.py:94-146).

    train_test_dates = [train_start, train_end, test_start, test_end] as
    'MMDD' strings within one (non-leap by default) year.
    """

    def __init__(self, dt: int, obs_len: Sequence[int], train_test_dates: List[str],
                 val_ratio: float = 0.2, year: int = 2017):
        self.day_timesteps = 24 // dt
        self.serial_len, self.daily_len, self.weekly_len = obs_len
        self.train_test_dates = train_test_dates
        self.val_ratio = val_ratio
        self.start_idx, self.mode_len = self.date2len(year)

    def date2len(self, year: int):
        d0 = datetime.date(year, 1, 1)

        def day_index(mmdd: str) -> int:
            d = datetime.date(year, int(mmdd[:2]), int(mmdd[2:]))
            return (d - d0).days

        tr_s, tr_e, te_s, te_e = (day_index(x) for x in self.train_test_dates)
        train_len = (tr_e + 1 - tr_s) * self.day_timesteps
        validate_len = int(train_len * self.val_ratio)
        train_len -= validate_len
        test_len = (te_e + 1 - te_s) * self.day_timesteps
        # NOTE: the reference returns the train start DAY index and later uses
        # it as a SAMPLE index (quirk 4); with the default 0101 start both are
        # 0. We always start the splits at window index 0 — identical
        # observable behavior at the defaults, well-defined for other dates.
        return 0, {"train": train_len, "validate": validate_len, "test": test_len}

    def window(self, taxi: np.ndarray):
        return sliding_windows(taxi, self.serial_len, self.daily_len,
                               self.weekly_len, self.day_timesteps)

    def get_data_loader(self, data: Dict[str, np.ndarray], batch_size: int,
                        device, rank: int = 0, world_size: int = 1,
                        shuffle_train: bool = False,
                        dtype: torch.dtype = torch.float32) -> Dict[str, DeviceLoader]:
        x_seq, y = self.window(data["taxi"])
        avail = x_seq.shape[0]
        need = sum(self.mode_len.values())
        if need > avail:
            # synthetic short runs: rescale split lengths proportionally
            scale = avail / need
            tr = int(self.mode_len["train"] * scale)
            va = int(self.mode_len["validate"] * scale)
            self.mode_len = {"train": tr, "validate": va, "test": avail - tr - va}
        loaders: Dict[str, DeviceLoader] = {}
        for mode in ["train", "validate", "test"]:
            ds = TaxiDataset(device, x_seq, y, mode, self.mode_len, self.start_idx,
                             dtype=dtype)
            loaders[mode] = DeviceLoader(
                ds.x, ds.y, batch_size,
                rank=rank if mode == "train" else 0,
                world_size=world_size if mode == "train" else 1,
                shuffle=shuffle_train and mode == "train")
        return loaders
