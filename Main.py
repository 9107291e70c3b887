"""ST-MGCN on MI355X — CLI entry point.

Keeps the reference's flag surface (-device / -model / -date / -cpt,
reference Main.py:21-34) and wiring order (data -> adjacency supports ->
model -> trainer -> test), plus MI355X extensions:

  --preset       named config from stmgcn_amd.config.PRESETS
  --nodes/--epochs/--batch-size  overrides
  --sparse       use CSR supports + the in-kernel Chebyshev recurrence
                 (the MI355X-native path; dense stacks = reference parity)
  --synthetic    generate the synthetic dataset in-memory (the reference
                 expects ./data/data_dict.npz but ships no data)
  --dtype        fp32 | bf16 | fp16 compute dtype

Distributed: launch under torchrun (one process per GPU, RCCL); rank/world
are read from the environment and the train split is block-sharded.
"""
from __future__ import annotations

import argparse
import os

import numpy as np
import torch
from torch import nn, optim

from stmgcn_amd import PRESETS
from stmgcn_amd.data import DataInput, DataGenerator, make_synthetic_dataset
from stmgcn_amd.graph import Adj_Preprocessor
from stmgcn_amd.models import build_model
from stmgcn_amd.parallel import GradReducer, init_distributed, cleanup_distributed
from stmgcn_amd.train import ModelTrainer

DTYPES = {"fp32": torch.float32, "bf16": torch.bfloat16, "fp16": torch.float16}


def main():
    p = argparse.ArgumentParser(description="Run ST-MGCN (MI355X-native)")
    p.add_argument("-device", "--device", type=str, default="cuda:0" if torch.cuda.is_available() else "cpu",
                   help="cpu or cuda:N")
    p.add_argument("-model", "--model_name", type=str, choices=["STMGCN"], default="STMGCN")
    p.add_argument("-date", "--dates", type=str, nargs="+",
                   default=["0101", "0630", "0701", "0731"],
                   help="train_start train_end test_start test_end (MMDD)")
    p.add_argument("-cpt", "--obs_len", type=int, nargs="+", default=[3, 1, 1],
                   help="serial/daily/weekly observation lengths")
    p.add_argument("--preset", type=str, default="reference", choices=sorted(PRESETS))
    p.add_argument("--data", type=str, default="./data/data_dict.npz")
    p.add_argument("--synthetic", action="store_true",
                   help="generate synthetic data in-memory (no npz needed)")
    p.add_argument("--nodes", type=int, default=None)
    p.add_argument("--epochs", type=int, default=None)
    p.add_argument("--batch-size", type=int, default=None)
    p.add_argument("--dtype", type=str, default=None, choices=sorted(DTYPES))
    p.add_argument("--sparse", action="store_true", help="CSR supports (HIP recurrence path)")
    p.add_argument("--model-dir", type=str, default="./output")
    p.add_argument("--metrics", type=str, default=None, help="JSONL metrics path")
    p.add_argument("--graph", action="store_true",
                   help="capture full-size train steps in a hipGraph (HIP path)")
    p.add_argument("--resume", action="store_true",
                   help="resume from the best-val checkpoint in --model-dir")
    args = p.parse_args()

    cfg = PRESETS[args.preset]
    if args.nodes:
        cfg = cfg.replace(n_nodes=args.nodes)
    if args.epochs:
        cfg = cfg.replace(n_epochs=args.epochs)
    if args.batch_size:
        cfg = cfg.replace(batch_size=args.batch_size)
    if args.dtype:
        cfg = cfg.replace(dtype=args.dtype)
    if args.obs_len != [3, 1, 1] or args.preset == "reference":
        cfg = cfg.replace(obs_len=list(args.obs_len), seq_len=sum(args.obs_len))

    env = init_distributed()
    rank, world = env["rank"], env["world_size"]
    if not str(args.device).startswith("cuda"):
        device = torch.device("cpu")
    elif world > 1 or "LOCAL_RANK" in os.environ:
        device = torch.device(f"cuda:{env['local_rank']}")  # one rank per GPU
    else:
        # single-process: honor -device cuda:N (reference Main.py:22-23)
        device = torch.device(args.device)
    dtype = DTYPES[cfg.dtype]
    if device.type == "cpu":
        dtype = torch.float32  # CPU oracle path runs fp32

    # ---- data (L1) ----
    data_in = DataInput(M_adj=cfg.m_graphs, data_dir=args.data, norm_opt=True)
    if args.synthetic or not os.path.exists(args.data):
        raw = make_synthetic_dataset(n_nodes=cfg.n_nodes, m_graphs=cfg.m_graphs)
        data = data_in.load_dict(raw)
    else:
        data = data_in.load_data()

    # ---- graph supports (L2) ----
    pre = Adj_Preprocessor(kernel_type=cfg.kernel_type, K=cfg.cheby_K,
                           lambda_max_mode=cfg.lambda_max_mode)
    sta_adj_list = []
    for key in data:
        if key.endswith("_adj"):
            adj = torch.from_numpy(data[key]).float()
            if args.sparse:
                sta_adj_list.append(pre.process_csr(adj).to(device))
            else:
                sta_adj_list.append(pre.process(adj).to(device=device, dtype=dtype))
    assert len(sta_adj_list) == cfg.m_graphs

    # ---- loaders ----
    gen = DataGenerator(dt=cfg.dt, obs_len=tuple(cfg.obs_len),
                        train_test_dates=args.dates, val_ratio=0.2)
    loaders = gen.get_data_loader(data, cfg.batch_size, device, rank=rank,
                                  world_size=world, shuffle_train=cfg.shuffle,
                                  dtype=dtype)

    # ---- model (L4) ----
    model = build_model(cfg).to(device=device, dtype=dtype)

    # MI355X path: fused MSE loss + flat-arena FusedAdam (one Adam launch,
    # one flat RCCL all-reduce for DP); torch Adam + GradReducer otherwise.
    from stmgcn_amd import ops as _ops
    hip_path = (device.type == "cuda" and _ops.hip_available()
                and dtype in (torch.bfloat16, torch.float16)
                and os.environ.get("STMGCN_IMPL", "hip") == "hip")
    if hip_path and cfg.loss == "MSE":
        from stmgcn_amd.ops import mse_loss as loss
    else:
        loss = {"MSE": nn.MSELoss(), "MAE": nn.L1Loss(), "Huber": nn.SmoothL1Loss()}[cfg.loss]
    if hip_path:
        from stmgcn_amd.train import FusedAdam as opt_cls
        reducer = None
    else:
        opt_cls = optim.Adam
        reducer = GradReducer(model) if world > 1 else None

    trainer = ModelTrainer(model=model, loss=loss, optimizer=opt_cls,
                           lr=cfg.lr, wd=cfg.weight_decay, n_epochs=cfg.n_epochs,
                           grad_reducer=reducer, rank=rank, world_size=world,
                           metrics_path=args.metrics,
                           use_graph=args.graph and hip_path)

    os.makedirs(args.model_dir, exist_ok=True)
    start_epoch = trainer.resume(args.model_dir) if args.resume else 0
    trainer.train(data_loader=loaders, sta_adj_list=sta_adj_list,
                  modes=["train", "validate"], model_dir=args.model_dir,
                  early_stopper=cfg.early_stop_patience, start_epoch=start_epoch)
    if rank == 0:
        print("Test: on Month", args.dates[2][:2], "Model", args.model_name)
        trainer.test(data_loader=loaders, sta_adj_list=sta_adj_list,
                     modes=["train", "test"], model_dir=args.model_dir,
                     data_class=data_in)
    cleanup_distributed()


if __name__ == "__main__":
    main()
