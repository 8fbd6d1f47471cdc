from sheeprl_amd.algos.p2e_dv2 import p2e_dv2_exploration, p2e_dv2_finetuning  # noqa: F401
