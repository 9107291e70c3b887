"""Flagship training benchmark — the driver contract.

Measures the BASELINE.json north-star metric: whole-node train samples/sec of
the 3-graph ST-MGCN at 1024 regions, seq_len 8, bf16, on synthetic
region-demand tensors with random-init weights (the reference publishes no
numbers; BASELINE.md's floor is the stock-PyTorch impl measured with
--impl torch on the same config).

Usage (driver): python -m torch.distributed.run --nnodes=1 --nproc-per-node N
  --master-addr 127.0.0.1 --master-port P bench.py --gpus N --steps K --warmup W
Single process: python bench.py [--gpus 1] [--steps K] [--warmup W]

One training step = forward + loss + backward (+ DP all-reduce) + Adam step.
Timed region: barrier + synchronize, K steps, barrier + synchronize; elapsed
is MAX over ranks; rank 0 prints ONE JSON line.
"""
from __future__ import annotations

import argparse
import json
import os
import time

import numpy as np
import torch
from torch import nn, optim

from stmgcn_amd import PRESETS
from stmgcn_amd.data.synthetic import make_synthetic_dataset
from stmgcn_amd.graph import SupportGenerator
from stmgcn_amd.models import build_model
from stmgcn_amd.parallel import GradReducer, init_distributed, cleanup_distributed

DTYPES = {"fp32": torch.float32, "bf16": torch.bfloat16, "fp16": torch.float16}

# Self-established comparison floors: the reference publishes
# no numbers, so the floor is the stock-PyTorch implementation (--impl torch)
# measured on the same config / 1x MI355X. samples/s at N=1 GPU.
BASELINE_FLOOR = {"bench-1024": 1198.0}


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--preset", type=str, default="bench-1024")
    p.add_argument("--batch-size", type=int, default=None, help="per-rank batch")
    p.add_argument("--impl", type=str, default=None, choices=["hip", "torch"],
                   help="override STMGCN_IMPL (torch = stock-PyTorch floor)")
    p.add_argument("--dtype", type=str, default=None, choices=sorted(DTYPES))
    p.add_argument("--profile-trace", type=str, default=None,
                   help="write a torch profiler trace to this path")
    p.add_argument("--graph", type=str, default="auto", choices=["auto", "on", "off"],
                   help="capture the whole train step in a hipGraph and replay it")
    p.add_argument("--mode", type=str, default="train", choices=["train", "infer"],
                   help="train: fwd+loss+bwd+opt (the driver contract); "
                        "infer: eval-mode forward only (the serving path)")
    args = p.parse_args()

    if args.impl:
        os.environ["STMGCN_IMPL"] = args.impl
    # Stock-floor measurability: MIOpen's RNN find can run for >10 min at the
    # deep/large shapes (MIOPEN_FIND_MODE only governs convolutions), so the
    # floor can optionally be measured on torch's decomposed native RNN path.
    if os.environ.get("STMGCN_TORCH_RNN_NATIVE", "0") == "1":
        torch.backends.cudnn.enabled = False
    cfg = PRESETS[args.preset]
    if args.batch_size:
        cfg = cfg.replace(batch_size=args.batch_size)
    if args.dtype:
        cfg = cfg.replace(dtype=args.dtype)

    env = init_distributed()
    rank, world = env["rank"], env["world_size"]
    use_gpu = torch.cuda.is_available()
    # RCCL all-reduce inside hipGraph capture is verified by
    # tests/test_gpu_kernels.py::test_rccl_allreduce_inside_hipgraph, so the
    # whole-step capture (fwd+loss+bwd+all-reduce+Adam) stays the default for
    # multi-rank runs too; capture failure falls back to eager below.
    device = torch.device(f"cuda:{env['local_rank']}") if use_gpu else torch.device("cpu")
    if use_gpu:
        torch.cuda.set_device(device)
    dtype = DTYPES[cfg.dtype] if use_gpu else torch.float32

    torch.manual_seed(1234 + rank)
    # ---- synthetic inputs of the target shape (weak scaling: per-GPU fixed) --
    B, T, N, C = cfg.batch_size, cfg.seq_len, cfg.n_nodes, cfg.input_dim
    raw = make_synthetic_dataset(n_nodes=N, n_steps=max(T * 8, 64), m_graphs=cfg.m_graphs,
                                 seed=7, day_timesteps=1)
    gen = SupportGenerator(cfg.kernel_type, cfg.cheby_K, cfg.lambda_max_mode)
    adj_keys = [k for k in raw if k.endswith("_adj")]
    if use_gpu and os.environ.get("STMGCN_IMPL", "hip") == "hip":
        adjs = [gen.process_csr(torch.from_numpy(raw[k]).float()).to(device) for k in adj_keys]
    else:
        adjs = [gen.process(torch.from_numpy(raw[k]).float()).to(device=device, dtype=dtype)
                for k in adj_keys]
    x = torch.randn(B, T, N, C, device=device, dtype=dtype)
    y = torch.randn(B, N, C, device=device, dtype=dtype)

    # ---- model + optimizer --------------------------------------------------
    model = build_model(cfg).to(device=device, dtype=dtype)
    want_graph = args.graph != "off" and use_gpu
    hip_mode = use_gpu and os.environ.get("STMGCN_IMPL", "hip") == "hip"
    if hip_mode and cfg.loss == "MSE":
        from stmgcn_amd.ops import mse_loss as criterion
    else:
        criterion = {"MSE": nn.MSELoss(), "MAE": nn.L1Loss(),
                     "Huber": nn.SmoothL1Loss()}[cfg.loss]
    if hip_mode:
        from stmgcn_amd.train import FusedAdam
        # one flat arena; DP = one RCCL all-reduce over it (reduce())
        opt = FusedAdam(model.parameters(), lr=cfg.lr,
                        weight_decay=cfg.weight_decay)
        reducer = None
    else:
        reducer = GradReducer(model) if world > 1 else None
        opt = optim.Adam(model.parameters(), lr=cfg.lr,
                         weight_decay=cfg.weight_decay,
                         capturable=want_graph, foreach=True)

    infer = args.mode == "infer"
    if infer:
        model.eval()

    def step():
        if infer:
            with torch.no_grad():
                return model(x, adjs)
        if reducer is not None:
            reducer.zero_grad()
        else:
            opt.zero_grad(set_to_none=False)
        loss = criterion(model(x, adjs), y)
        loss.backward()
        if reducer is not None:
            reducer.reduce()
        elif hip_mode and world > 1:
            opt.reduce()
        opt.step()
        return loss

    def barrier_sync():
        if world > 1:
            torch.distributed.barrier()
        if use_gpu:
            torch.cuda.synchronize()

    for _ in range(max(args.warmup, 2 if want_graph else 0)):
        step()
    barrier_sync()

    # ---- hipGraph capture of the whole training step ------------------------
    # (forward + loss + backward (+ DP all-reduce) + Adam as ONE replayable
    # graph — the model is launch-bound at this scale; see MI355X_MICROARCH
    # "graph-replay-floor")
    graph_ok = False
    if want_graph:
        try:
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                step()
            torch.cuda.current_stream().wait_stream(side)
            torch.cuda.synchronize()
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                step()
            torch.cuda.synchronize()
            graph_ok = True
        except Exception as e:
            if args.graph == "on":
                raise
            print(f"# graph capture unavailable, eager fallback: {e}", flush=True)
            graph_ok = False

    run_step = (lambda: g.replay()) if graph_ok else step
    for _ in range(args.warmup):
        run_step()
    barrier_sync()

    prof_ctx = None
    if args.profile_trace:
        prof_ctx = torch.profiler.profile(
            activities=[torch.profiler.ProfilerActivity.CPU,
                        torch.profiler.ProfilerActivity.CUDA])
        prof_ctx.__enter__()

    t0 = time.perf_counter()
    for _ in range(args.steps):
        run_step()
    barrier_sync()
    elapsed = time.perf_counter() - t0

    if prof_ctx is not None:
        prof_ctx.__exit__(None, None, None)
        if rank == 0:
            prof_ctx.export_chrome_trace(args.profile_trace)

    # impl truthfulness: if any op fell back to torch under
    # STMGCN_ALLOW_FALLBACK=1, the JSON must not claim a pure-hip run
    impl = os.environ.get("STMGCN_IMPL", "hip") if use_gpu else "torch-cpu"
    if impl == "hip":
        from stmgcn_amd.ops import hip_ops
        if hip_ops.fallback_count() > 0:
            impl = "hip+fallback"

    # max over ranks
    if world > 1:
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if torch.distributed.get_backend() == "nccl" else "cpu")
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(t.item())

    n_gpus = world if use_gpu else world  # ranks == GPUs in the driver launch
    samples_per_sec = n_gpus * B * args.steps / elapsed
    # weak scaling: the floor scales with n_gpus for a whole-job ratio
    # (train-mode floors only — no published/measured infer floor)
    floor = BASELINE_FLOOR.get(args.preset) if (use_gpu and not infer) else None
    vs_baseline = (samples_per_sec / (floor * n_gpus)) if floor else None
    if rank == 0:
        rec = {
            "metric": "infer_samples_per_sec" if infer else "train_samples_per_sec",
            "value": samples_per_sec,
            "unit": "samples/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1e3,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": vs_baseline,
            "dtype": cfg.dtype if use_gpu else "fp32",
            "data": "synthetic",
            "config": {
                "model": model.__class__.__name__,
                "preset": args.preset,
                "n_nodes": N, "seq_len": T, "m_graphs": cfg.m_graphs,
                "lstm_hidden": cfg.lstm_hidden_dim, "gcn_hidden": cfg.gcn_hidden_dim,
                "global_batch": B * n_gpus,
                "parallelism": f"dp{n_gpus}",
                "hipgraph": graph_ok,
                "impl": impl,
                "max_mem_gb": round(torch.cuda.max_memory_allocated() / 2**30, 1)
                              if use_gpu else 0.0,
            },
        }
        print(json.dumps(rec))
    cleanup_distributed()


if __name__ == "__main__":
    main()
