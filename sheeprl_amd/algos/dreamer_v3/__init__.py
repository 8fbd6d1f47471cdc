from sheeprl_amd.algos.dreamer_v3 import dreamer_v3, evaluate  # noqa: F401
