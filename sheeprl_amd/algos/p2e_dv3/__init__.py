from sheeprl_amd.algos.p2e_dv3 import p2e_dv3_exploration, p2e_dv3_finetuning  # noqa: F401
