"""ST-MGCN on MI355X — a from-scratch, MI355X-native spatiotemporal multi-graph
forecasting framework with the capabilities of underdoc-wang/ST-MGCN (AAAI'19).

Layer map (mirrors SURVEY.md §1, rebuilt MI355X-first):

  L6 CLI        Main.py (repo root) — reference-compatible argparse surface
  L5 Training   stmgcn_amd.train.ModelTrainer — epoch loop, early stop,
                checkpointing, metrics (reference Model_Trainer.py:8-114 parity)
  L4 Model      stmgcn_amd.models.ST_MGCN / CG_LSTM (reference STMGCN.py:7-119)
  L3 Ops        stmgcn_amd.ops — autograd.Function wrappers over hand-written
                CDNA4 HIP kernels (CSR ChebConv with in-kernel K-hop recurrence,
                fused contextual gate, persistent fused LSTM/GRU, fused head,
                fused loss, multi-tensor Adam); pure-PyTorch oracle on CPU
  L2 Graph      stmgcn_amd.graph — adjacency -> support generator; CSR-first
                (reference GCN.py:50-135 emits dense (K+1,N,N) stacks; we keep
                the sparse generator matrix and run the recurrence in-kernel)
  L1 Data       stmgcn_amd.data — synthetic npz, normalization, windowing,
                device-resident datasets, per-rank DP shards
                (reference Data_Container.py)

Distributed: one process per GPU, torch.distributed over RCCL (backend "nccl"
on ROCm) with a flat bucketed gradient all-reduce sized for the 7-link xGMI
point-to-point clique. See stmgcn_amd.parallel.
"""

__version__ = "0.1.0"

from . import graph  # noqa: F401
from . import data  # noqa: F401
from . import models  # noqa: F401
from . import train  # noqa: F401
from .config import STMGCNConfig, PRESETS  # noqa: F401
