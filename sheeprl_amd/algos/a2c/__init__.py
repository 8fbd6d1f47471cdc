from sheeprl_amd.algos.a2c import a2c  # noqa: F401
