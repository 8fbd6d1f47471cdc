from sheeprl_amd.algos.ppo import evaluate, ppo, ppo_decoupled  # noqa: F401
