from sheeprl_amd.algos.sac import evaluate, sac  # noqa: F401
