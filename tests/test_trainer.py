"""End-to-end micro-training on synthetic data:
checkpoint layout, early stopping, test metrics, Main.py wiring."""
import os
import subprocess
import sys

import numpy as np
import pytest
import torch
from torch import nn, optim

from stmgcn_amd.data import DataInput, DataGenerator, make_synthetic_dataset
from stmgcn_amd.graph import SupportGenerator
from stmgcn_amd.models import ST_MGCN
from stmgcn_amd.train import ModelTrainer

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _setup(n_nodes=16, m=2, n_steps=24 * 30):
    raw = make_synthetic_dataset(n_nodes=n_nodes, n_steps=n_steps, m_graphs=m)
    di = DataInput(M_adj=m, data_dir="", norm_opt=True)
    data = di.load_dict(raw)
    gen = SupportGenerator("chebyshev", 2)
    adjs = [gen.process(torch.from_numpy(data[k]).float())
            for k in data if k.endswith("_adj")]
    dgen = DataGenerator(dt=1, obs_len=(3, 1, 1),
                         train_test_dates=["0101", "0630", "0701", "0731"])
    loaders = dgen.get_data_loader(data, batch_size=32, device="cpu")
    torch.manual_seed(0)
    model = ST_MGCN(M=m, seq_len=5, n_nodes=n_nodes, input_dim=1,
                    lstm_hidden_dim=16, lstm_num_layers=2, gcn_hidden_dim=16,
                    sta_kernel_config={"kernel_type": "chebyshev", "K": 2})
    return di, adjs, loaders, model


def test_micro_train_checkpoint_and_test(tmp_path):
    di, adjs, loaders, model = _setup()
    trainer = ModelTrainer(model=model, loss=nn.MSELoss(), optimizer=optim.Adam,
                           lr=2e-3, wd=1e-4, n_epochs=2,
                           metrics_path=str(tmp_path / "metrics.jsonl"))
    trainer.train(loaders, adjs, modes=["train", "validate"],
                  model_dir=str(tmp_path))
    ckpt_path = tmp_path / "ST_MGCN_best_model.pkl"
    assert ckpt_path.exists()
    ck = torch.load(ckpt_path, weights_only=False)
    assert set(ck.keys()) == {"epoch", "state_dict"}
    assert ck["epoch"] >= 1
    assert len(ck["state_dict"]) == len(model.state_dict())

    results = trainer.test(loaders, adjs, modes=["train", "test"],
                           model_dir=str(tmp_path), data_class=di)
    for mode in ["train", "test"]:
        assert np.isfinite(results[mode]["RMSE"])
        assert results[mode]["RMSE"] >= 0
    assert (tmp_path / "metrics.jsonl").exists()


def test_training_reduces_loss(tmp_path):
    """Val loss after a few epochs must drop below the epoch-1 value
    (sanity: the whole stack actually learns on periodic synthetic data)."""
    di, adjs, loaders, model = _setup()
    trainer = ModelTrainer(model=model, loss=nn.MSELoss(), optimizer=optim.Adam,
                           lr=5e-3, wd=0.0, n_epochs=4)
    losses = []

    import stmgcn_amd.train.trainer as trmod
    orig = trainer._allreduce_scalar
    trainer._allreduce_scalar = lambda v: (losses.append(v), orig(v))[1]
    trainer.train(loaders, adjs, modes=["train", "validate"], model_dir=str(tmp_path))
    assert len(losses) == 4
    assert min(losses[1:]) < losses[0]


def test_early_stopping(tmp_path):
    di, adjs, loaders, model = _setup(n_nodes=9, m=1, n_steps=24 * 10)
    small = model.__class__(
        M=1, seq_len=5, n_nodes=9, input_dim=1, lstm_hidden_dim=8,
        lstm_num_layers=1, gcn_hidden_dim=8,
        sta_kernel_config={"kernel_type": "chebyshev", "K": 2})
    trainer = ModelTrainer(model=small, loss=nn.MSELoss(), optimizer=optim.Adam,
                           lr=0.0, wd=0.0, n_epochs=50)
    # force a strictly worsening val loss -> patience exhausts after epoch 1
    seq = iter(range(1, 100))
    trainer._allreduce_scalar = lambda v: float(next(seq))
    trainer.train(loaders, adjs[:1], modes=["train", "validate"],
                  model_dir=str(tmp_path), early_stopper=3)
    ck = torch.load(tmp_path / "ST_MGCN_best_model.pkl", weights_only=False)
    assert ck["epoch"] == 1  # only epoch 1 improved; stop fired before 50


def test_trainer_rejects_unknown_model():
    with pytest.raises(ValueError):
        ModelTrainer(model=nn.Linear(3, 3), loss=nn.MSELoss(),
                     optimizer=optim.Adam, lr=1e-3, wd=0, n_epochs=1)


def test_main_cli_sparse_path(tmp_path):
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "Main.py"),
         "-device", "cpu", "--synthetic", "--preset", "cpu-small",
         "--nodes", "9", "--epochs", "1", "--batch-size", "16", "--sparse",
         "--model-dir", str(tmp_path)],
        capture_output=True, text=True, timeout=600, cwd=REPO)
    assert out.returncode == 0, out.stderr[-2000:]
    assert (tmp_path / "ST_MGCN_best_model.pkl").exists()


def test_step_timer_cpu():
    """StepTimer works without a GPU (perf_counter fallback) and summarizes."""
    from stmgcn_amd.utils.profiling import StepTimer
    t = StepTimer(capacity=4)
    for _ in range(6):
        with t:
            sum(range(1000))
    ms = t.ms()
    assert len(ms) == 4 and all(m >= 0 for m in ms)
    s = t.summary()
    assert s["n"] == 4 and s["p50_ms"] >= 0


def test_deterministic_mode_env(monkeypatch):
    from stmgcn_amd.models.stmgcn import deterministic_mode
    monkeypatch.delenv("STMGCN_DETERMINISTIC", raising=False)
    assert not deterministic_mode()
    monkeypatch.setenv("STMGCN_DETERMINISTIC", "1")
    assert deterministic_mode()


def test_main_cli_end_to_end(tmp_path):
    """Full CLI path on CPU: synthetic data, 2 epochs, checkpoint written
    with the reference layout, test metrics printed (reference Main.py flag
    surface + wiring order)."""
    import subprocess
    import sys
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out_dir = tmp_path / "output"
    r = subprocess.run(
        [sys.executable, os.path.join(repo, "Main.py"),
         "--synthetic", "--preset", "cpu-small", "--epochs", "2",
         "--nodes", "24", "-cpt", "3", "1", "1",
         "-date", "0101", "0109", "0110", "0112",
         "--model-dir", str(out_dir),
         "--metrics", str(tmp_path / "metrics.jsonl")],
        capture_output=True, text=True, timeout=420, cwd=repo)
    assert r.returncode == 0, r.stderr[-2000:]
    assert "Training starts at" in r.stdout
    assert "true RMSE" in r.stdout
    ckpt = out_dir / "ST_MGCN_best_model.pkl"
    assert ckpt.exists()
    saved = torch.load(ckpt, weights_only=False)
    assert set(saved) == {"epoch", "state_dict"}
    assert (tmp_path / "metrics.jsonl").read_text().strip()


def test_resume_from_best_checkpoint(tmp_path):
    """Failure recovery: kill after 2 epochs, resume — weights
    and optimizer state reload, training continues from the saved epoch,
    and the reference checkpoint file format is unchanged (no extra keys)."""
    import torch.optim as optim
    from stmgcn_amd import PRESETS
    from stmgcn_amd.graph import SupportGenerator
    from stmgcn_amd.models import build_model
    from stmgcn_amd.train import ModelTrainer

    cfg = PRESETS["cpu-small"].replace(n_nodes=16, batch_size=8, seq_len=4,
                                       lstm_hidden_dim=8, gcn_hidden_dim=8,
                                       lstm_num_layers=1, m_graphs=1,
                                       obs_len=[4, 0, 0])
    torch.manual_seed(0)
    gen = SupportGenerator(cfg.kernel_type, cfg.cheby_K, cfg.lambda_max_mode)
    a = torch.rand(16, 16)
    a = ((a + a.T) > 1.4).float()
    a.fill_diagonal_(0)
    adjs = [gen.process(a)]
    x = torch.randn(24, cfg.seq_len, 16, 1)
    y = torch.randn(24, 16, 1)
    from stmgcn_amd.data.container import DeviceLoader
    loaders = {m: DeviceLoader(x, y, 8) for m in ["train", "validate"]}

    def make_trainer():
        torch.manual_seed(1)
        model = build_model(cfg)
        return ModelTrainer(model=model, loss=torch.nn.MSELoss(),
                            optimizer=optim.Adam, lr=1e-2, wd=0.0, n_epochs=2)

    t1 = make_trainer()
    t1.train(loaders, adjs, ["train", "validate"], str(tmp_path))
    saved = torch.load(tmp_path / "ST_MGCN_best_model.pkl", weights_only=False)
    assert set(saved) == {"epoch", "state_dict"}      # reference layout intact
    assert (tmp_path / "ST_MGCN_best_model.optim.pkl").exists()

    t2 = make_trainer()                                # fresh process stand-in
    start = t2.resume(str(tmp_path))
    assert start == saved["epoch"] > 0
    for k, v in t2.model.state_dict().items():
        torch.testing.assert_close(v, saved["state_dict"][k])
    assert t2.optimizer.state_dict()["state"]          # optimizer moments loaded
    # the sidecar seeds the best-val floor: the first post-resume epoch must
    # not overwrite a better pre-crash checkpoint with a worse one
    assert np.isfinite(t2._resume_best_val)
    t2.n_epochs = start + 1
    t2.train(loaders, adjs, ["train", "validate"], str(tmp_path),
             start_epoch=start)                        # continues, no crash


def test_resume_does_not_clobber_better_checkpoint(tmp_path):
    """Crash/resume cycle: if every post-resume epoch is WORSE than the
    pre-crash best, the best checkpoint file must survive untouched
    (train() used to restart at val_loss=inf and overwrite)."""
    make, loaders, adjs = _tiny_setup(seed=5)
    t1 = make(n_epochs=2, lr=1e-2)
    vals1 = iter([1.0, 0.25])             # best = epoch 2 @ 0.25
    t1._allreduce_scalar = lambda v: float(next(vals1))
    t1.train(loaders, adjs, ["train", "validate"], str(tmp_path))
    best = torch.load(tmp_path / "ST_MGCN_best_model.pkl", weights_only=False)
    assert best["epoch"] == 2

    t2 = make(n_epochs=4, lr=1e-2)        # "restarted process"
    start = t2.resume(str(tmp_path))
    assert t2._resume_best_val == pytest.approx(0.25)
    vals2 = iter([0.9, 1.5])              # both worse than 0.25
    t2._allreduce_scalar = lambda v: float(next(vals2))
    t2.train(loaders, adjs, ["train", "validate"], str(tmp_path),
             start_epoch=start)
    after = torch.load(tmp_path / "ST_MGCN_best_model.pkl", weights_only=False)
    assert after["epoch"] == 2            # pre-crash best not overwritten
    for k, v in after["state_dict"].items():
        torch.testing.assert_close(v, best["state_dict"][k])


def _tiny_setup(seed=0):
    import torch.optim as optim
    from stmgcn_amd import PRESETS
    from stmgcn_amd.data.container import DeviceLoader
    from stmgcn_amd.graph import SupportGenerator
    from stmgcn_amd.models import build_model
    from stmgcn_amd.train import ModelTrainer

    cfg = PRESETS["cpu-small"].replace(n_nodes=12, batch_size=6, seq_len=4,
                                       lstm_hidden_dim=4, gcn_hidden_dim=4,
                                       lstm_num_layers=1, m_graphs=1,
                                       obs_len=[4, 0, 0])
    torch.manual_seed(seed)
    gen = SupportGenerator(cfg.kernel_type, cfg.cheby_K, cfg.lambda_max_mode)
    a = torch.rand(12, 12)
    a = ((a + a.T) > 1.4).float()
    a.fill_diagonal_(0)
    adjs = [gen.process(a)]
    x = torch.randn(12, cfg.seq_len, 12, 1)
    y = torch.randn(12, 12, 1)
    loaders = {m: DeviceLoader(x, y, 6) for m in ["train", "validate"]}

    def make(n_epochs, lr=1e-3):
        torch.manual_seed(seed + 1)
        model = build_model(cfg)
        return ModelTrainer(model=model, loss=torch.nn.MSELoss(),
                            optimizer=optim.Adam, lr=lr, wd=0.0,
                            n_epochs=n_epochs)
    return make, loaders, adjs


def test_early_stopping_fires(tmp_path, capsys):
    """Patience exhausts -> 'Early stopping at epoch N..' and train returns
    before n_epochs (reference Model_Trainer.py:55-60)."""
    make, loaders, adjs = _tiny_setup()
    tr = make(n_epochs=50, lr=1.0)   # divergent lr -> val loss worsens
    tr.train(loaders, adjs, ["train", "validate"], str(tmp_path),
             early_stopper=2)
    out = capsys.readouterr().out
    assert "Early stopping at epoch" in out
    # stopped well before 50 epochs
    stopped = int(out.split("Early stopping at epoch")[1].split("..")[0])
    assert stopped < 50


def test_final_save_is_best_not_last(tmp_path):
    """The end-of-training save preserves the BEST-epoch weights even when a
    later, worse epoch ran afterwards. (The reference aliases live tensors in
    its checkpoint dict, so its final save silently writes last-epoch weights
    under the best epoch number — fixed by cloning at best-epoch time.)
    Forced val losses make best=epoch 2, last=epoch 3 deterministically."""
    make, loaders, adjs = _tiny_setup(seed=3)
    tr = make(n_epochs=3, lr=1e-2)
    vals = iter([1.0, 0.5, 2.0])          # improves at 2, worsens at 3
    tr._allreduce_scalar = lambda v: float(next(vals))
    tr.train(loaders, adjs, ["train", "validate"], str(tmp_path),
             early_stopper=50)
    saved = torch.load(tmp_path / "ST_MGCN_best_model.pkl", weights_only=False)
    assert saved["epoch"] == 2
    live = tr.model.state_dict()
    # epoch 3 trained with lr>0 -> live weights moved past the saved snapshot
    assert any(not torch.equal(v, live[k])
               for k, v in saved["state_dict"].items())
