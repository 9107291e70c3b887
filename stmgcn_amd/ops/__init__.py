from .functional import (  # noqa: F401
    gconv_mix, contextual_gate, rnn_forward, branch_fuse_head,
    hip_available, require_hip,
)
