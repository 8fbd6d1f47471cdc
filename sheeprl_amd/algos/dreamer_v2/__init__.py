from sheeprl_amd.algos.dreamer_v2 import dreamer_v2  # noqa: F401
