from sheeprl_amd.config.compose import compose, load_yaml, merge, resolve, save_config, instantiate

__all__ = ["compose", "load_yaml", "merge", "resolve", "save_config", "instantiate"]
