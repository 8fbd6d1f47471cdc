from sheeprl_amd.algos.sac_ae import sac_ae  # noqa: F401
