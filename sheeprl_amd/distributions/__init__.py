from sheeprl_amd.distributions.dists import (
    LogProbCategorical,
    BernoulliSafeMode,
    MSEDistribution,
    OneHotCategoricalST,
    SymlogDistribution,
    TanhNormal,
    TruncatedNormal,
    TwoHotEncodingDistribution,
    unimix_logits,
)

__all__ = [
    "SymlogDistribution",
    "MSEDistribution",
    "TwoHotEncodingDistribution",
    "OneHotCategoricalST",
    "BernoulliSafeMode",
    "TruncatedNormal",
    "TanhNormal",
    "unimix_logits",
    "LogProbCategorical",
]
