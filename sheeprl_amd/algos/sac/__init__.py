from sheeprl_amd.algos.sac import evaluate, sac, sac_decoupled  # noqa: F401
