from sheeprl_amd.algos.p2e_dv1 import p2e_dv1_exploration, p2e_dv1_finetuning  # noqa: F401
