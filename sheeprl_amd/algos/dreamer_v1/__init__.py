from sheeprl_amd.algos.dreamer_v1 import dreamer_v1  # noqa: F401
