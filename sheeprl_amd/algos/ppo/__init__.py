from sheeprl_amd.algos.ppo import evaluate, ppo  # noqa: F401  (registers ppo + eval)
