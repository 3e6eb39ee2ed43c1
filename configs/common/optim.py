"""Default AdamW optimizer config (reference: configs/common/optim.py)."""

from libai_amd.config import LazyCall
from libai_amd.optim import FusedAdamW, get_default_optimizer_params

optim = LazyCall(FusedAdamW)(
    params=LazyCall(get_default_optimizer_params)(
        # model inserted by build_optimizer
        clip_grad_max_norm=1.0,
        clip_grad_norm_type=2.0,
        weight_decay_norm=0.0,
        weight_decay_bias=0.0,
    ),
    lr=1e-4,
    weight_decay=0.01,
    betas=(0.9, 0.999),
    eps=1e-8,
    clip_grad=1.0,
)
