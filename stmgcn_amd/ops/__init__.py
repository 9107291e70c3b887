from .functional import (  # noqa: F401
    gconv_mix, contextual_gate, rnn_forward, branch_fuse_head,
    seqsum_permute, mse_loss, hip_available, require_hip,
)
