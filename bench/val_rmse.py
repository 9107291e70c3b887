"""val-RMSE half of the north-star metric (BASELINE.json: "train samples/sec
(whole node) + val RMSE, 1024-region 3-graph").

Trains ST-MGCN on the synthetic region-demand dataset (the reference ships
no data — SURVEY §6) through the full Main.py stack (DataInput ->
Adj_Preprocessor -> ST_MGCN -> ModelTrainer) and prints one JSON line with
the best val loss and denormalized test RMSE/MAE/MAPE, for both the HIP
kernel path and the stock-torch floor when requested.

Usage: python bench/val_rmse.py [--preset bench-1024] [--epochs 5]
       [--impl hip|torch] [--nodes N]
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch
from torch import nn, optim


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--preset", default="bench-1024")
    p.add_argument("--epochs", type=int, default=5)
    p.add_argument("--impl", default="hip", choices=["hip", "torch"])
    p.add_argument("--nodes", type=int, default=None)
    p.add_argument("--days", type=int, default=21, help="synthetic dataset length")
    p.add_argument("--seed", type=int, default=7)
    p.add_argument("--graph", action="store_true",
                   help="hipGraph-captured train steps (HIP path)")
    args = p.parse_args()
    os.environ["STMGCN_IMPL"] = args.impl

    from stmgcn_amd import PRESETS, ops
    from stmgcn_amd.data import DataInput, DataGenerator, make_synthetic_dataset
    from stmgcn_amd.graph import Adj_Preprocessor
    from stmgcn_amd.models import build_model
    from stmgcn_amd.train import ModelTrainer

    cfg = PRESETS[args.preset].replace(n_epochs=args.epochs)
    if args.nodes:
        cfg = cfg.replace(n_nodes=args.nodes)
    use_gpu = torch.cuda.is_available()
    device = torch.device("cuda:0" if use_gpu else "cpu")
    dtype = {"fp32": torch.float32, "bf16": torch.bfloat16,
             "fp16": torch.float16}[cfg.dtype] if use_gpu else torch.float32

    torch.manual_seed(args.seed)
    np.random.seed(args.seed)
    raw = make_synthetic_dataset(n_nodes=cfg.n_nodes, m_graphs=cfg.m_graphs,
                                 n_steps=args.days * 24, seed=args.seed)
    data_in = DataInput(M_adj=cfg.m_graphs, data_dir="<synthetic>", norm_opt=True)
    data = data_in.load_dict(raw)

    pre = Adj_Preprocessor(kernel_type=cfg.kernel_type, K=cfg.cheby_K,
                           lambda_max_mode=cfg.lambda_max_mode)
    hip_path = (use_gpu and args.impl == "hip" and ops.hip_available()
                and dtype in (torch.bfloat16, torch.float16))
    adjs = []
    for key in data:
        if key.endswith("_adj"):
            a = torch.from_numpy(data[key]).float()
            adjs.append(pre.process_csr(a).to(device) if hip_path
                        else pre.process(a).to(device=device, dtype=dtype))

    from datetime import date, timedelta
    d0 = date(2017, 1, 1)
    fmt = lambda d: f"{d.month:02d}{d.day:02d}"
    train_end = d0 + timedelta(days=args.days - 8)
    dates = ["0101", fmt(train_end), fmt(train_end + timedelta(days=1)),
             fmt(d0 + timedelta(days=args.days - 2))]
    gen = DataGenerator(dt=cfg.dt, obs_len=tuple(cfg.obs_len),
                        train_test_dates=dates, val_ratio=0.2)
    loaders = gen.get_data_loader(data, cfg.batch_size, device, dtype=dtype)

    model = build_model(cfg).to(device=device, dtype=dtype)
    if hip_path and cfg.loss == "MSE":
        from stmgcn_amd.ops import mse_loss as loss
        from stmgcn_amd.train import FusedAdam as opt_cls
    else:
        loss = {"MSE": nn.MSELoss(), "MAE": nn.L1Loss(),
                "Huber": nn.SmoothL1Loss()}[cfg.loss]
        opt_cls = optim.Adam

    with tempfile.TemporaryDirectory() as model_dir:
        trainer = ModelTrainer(model=model, loss=loss, optimizer=opt_cls,
                               lr=cfg.lr, wd=cfg.weight_decay,
                               n_epochs=cfg.n_epochs,
                               use_graph=args.graph and hip_path)
        trainer.train(data_loader=loaders, sta_adj_list=adjs,
                      modes=["train", "validate"], model_dir=model_dir,
                      early_stopper=cfg.early_stop_patience)
        results = trainer.test(data_loader=loaders, sta_adj_list=adjs,
                               modes=["test"], model_dir=model_dir,
                               data_class=data_in)
    rec = {"metric": "val_rmse", "impl": args.impl, "preset": args.preset,
           "n_nodes": cfg.n_nodes, "epochs": cfg.n_epochs,
           "dtype": cfg.dtype if use_gpu else "fp32", "data": "synthetic",
           "test_RMSE": results["test"]["RMSE"],
           "test_MAE": results["test"]["MAE"],
           "test_MAPE": results["test"]["MAPE"]}
    print(json.dumps(rec))


if __name__ == "__main__":
    main()
