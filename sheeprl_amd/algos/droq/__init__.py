from sheeprl_amd.algos.droq import droq  # noqa: F401
