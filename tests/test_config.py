import pytest

from sheeprl_amd.config import compose


def test_compose_basic():
    cfg = compose(["exp=ppo"])
    assert cfg.algo.name == "ppo"
    assert cfg.env.id == "cartpole"
    assert cfg.runtime.devices == 1


def test_compose_dotted_override():
    cfg = compose(["exp=ppo", "algo.rollout_steps=7", "env.num_envs=3"])
    assert cfg.algo.rollout_steps == 7
    assert cfg.env.num_envs == 3


def test_compose_group_override():
    cfg = compose(["exp=ppo", "env=dummy"])
    assert cfg.env.id == "dummy_discrete"


def test_interpolation():
    cfg = compose(["exp=ppo"])
    assert cfg.exp_name == "ppo_cartpole"
    assert "ppo" in cfg.root_dir


def test_missing_exp_raises():
    with pytest.raises(Exception):
        compose([])


def test_exp_inherits_group_chain():
    cfg = compose(["exp=ppo_benchmarks"])
    assert cfg.algo.total_steps == 65536
    assert cfg.env.num_envs == 1
    assert cfg.runtime.accelerator == "cpu"


def test_exp_child_group_override_beats_base():
    """A child exp's `override /algo:` must beat its base exp's selection
    (regression: the bench silently ran the XL model instead of S)."""
    from sheeprl_amd.config import compose

    cfg = compose(["exp=dreamer_v3_100k_ms_pacman"])
    assert cfg.algo.world_model.recurrent_model.recurrent_state_size == 512
    assert cfg.algo.dense_units == 512


def test_registry_completeness():
    """Every registered algorithm has an evaluator and a composable config."""
    import sheeprl_amd.algos  # noqa: F401
    from sheeprl_amd.config import compose
    from sheeprl_amd.utils.registry import algorithm_registry, evaluation_registry

    assert len(algorithm_registry) == 17
    missing = set(algorithm_registry) - set(evaluation_registry)
    assert not missing, f"algorithms without evaluators: {missing}"
    for name in algorithm_registry:
        extra = []
        if name.endswith("_finetuning"):
            extra = ["checkpoint.exploration_ckpt_path=/tmp/x.ckpt"]
        cfg = compose([f"exp={name}", "env=dummy", *extra])
        assert cfg.algo.name == name
