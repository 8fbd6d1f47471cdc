from sheeprl_amd.algos.ppo_recurrent import ppo_recurrent  # noqa: F401
