from sheeprl_amd.ops._ext import get_ext, has_ext, require_ext, use_hip
from sheeprl_amd.ops.functional import (
    bernoulli_log_prob,
    gae,
    lambda_values,
    mse_log_prob,
    symexp,
    symlog,
    symlog_mse_log_prob,
    two_hot_decoder,
    two_hot_encoder,
    twohot_from_support,
)
from sheeprl_amd.ops.categorical import categorical_st
from sheeprl_amd.ops.fused import (
    ema_update_,
    gru_gates,
    kl_balanced,
    layer_norm_act,
    masked_lerp,
    normalize_obs,
    twohot_log_prob,
)

__all__ = [
    "get_ext",
    "has_ext",
    "require_ext",
    "use_hip",
    "symlog",
    "symexp",
    "two_hot_encoder",
    "two_hot_decoder",
    "twohot_from_support",
    "gae",
    "lambda_values",
    "mse_log_prob",
    "symlog_mse_log_prob",
    "bernoulli_log_prob",
    "layer_norm_act",
    "kl_balanced",
    "twohot_log_prob",
    "categorical_st",
    "gru_gates",
    "ema_update_",
    "normalize_obs",
    "masked_lerp",
]
