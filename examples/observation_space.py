"""Print the observation/action space an algorithm will see for a config
(parity: examples/observation_space.py): compose the config, build one env
through the factory and show the spaces plus the cnn/mlp key selection."""

import sys

from sheeprl_amd.config.compose import compose
from sheeprl_amd.envs import make_env

if __name__ == "__main__":
    overrides = sys.argv[1:] or ["exp=dreamer_v3", "env=dummy"]
    cfg = compose(overrides)
    env = make_env(cfg, cfg.seed, 0, None, "")()
    print("env id            :", cfg.env.id)
    print("observation space :", env.observation_space)
    print("action space      :", env.action_space)
    print("cnn encoder keys  :", cfg.algo.cnn_keys.encoder)
    print("mlp encoder keys  :", cfg.algo.mlp_keys.encoder)
    env.close()
