"""Pin the bench.py driver contract (the harness runs
`python bench.py --gpus N --steps K --warmup W` and parses ONE JSON line
from rank 0): required keys, sane values, CPU single-process operation."""
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED_KEYS = {
    "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
    "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
}


def _run_bench(*extra):
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"),
         "--steps", "2", "--warmup", "1", *extra],
        capture_output=True, text=True, timeout=600, cwd=REPO)
    assert r.returncode == 0, r.stderr[-2000:]
    json_lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(json_lines) == 1, f"expected exactly one JSON line: {r.stdout}"
    return json.loads(json_lines[0])


def test_bench_json_contract():
    rec = _run_bench()
    assert REQUIRED_KEYS.issubset(rec.keys()), REQUIRED_KEYS - set(rec.keys())
    assert rec["metric"] == "train_samples_per_sec"
    assert rec["unit"] == "samples/s"
    assert rec["n_gpus"] == 1 and rec["steps"] == 2 and rec["warmup"] == 1
    assert rec["higher_is_better"] is True
    assert rec["scaling"] == "weak"
    assert rec["data"] == "synthetic"
    assert rec["value"] > 0 and rec["ms_per_step"] > 0
    cfg = rec["config"]
    assert cfg["n_nodes"] == 1024 and cfg["seq_len"] == 8 and cfg["m_graphs"] == 3
    assert cfg["global_batch"] == 32 and cfg["parallelism"] == "dp1"


def test_bench_torch_floor_mode():
    rec = _run_bench("--impl", "torch")
    assert rec["config"]["impl"].startswith("torch")


def test_bench_torchrun_world2_cpu():
    """The driver's exact multi-rank launch (torch.distributed.run, nnodes=1,
    one JSON line from rank 0) works end-to-end: rendezvous over 127.0.0.1,
    gloo on CPU, max-over-ranks elapsed reduce, global-batch reporting.
    Pre-hardens the 8-GPU SCALE run."""
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(port), os.path.join(REPO, "bench.py"),
         "--gpus", "2", "--steps", "2", "--warmup", "1",
         "--preset", "cpu-small"],
        capture_output=True, text=True, timeout=600, cwd=REPO)
    assert r.returncode == 0, r.stderr[-2000:]
    json_lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(json_lines) == 1, f"expected exactly one JSON line: {r.stdout}"
    rec = json.loads(json_lines[0])
    assert rec["n_gpus"] == 2
    assert rec["config"]["parallelism"] == "dp2"
    assert rec["config"]["global_batch"] == 16      # 8 per rank x 2 ranks
    assert rec["value"] > 0
