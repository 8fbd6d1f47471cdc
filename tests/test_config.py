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


def test_external_algorithm_via_search_path(tmp_path, monkeypatch):
    """End-to-end external-algorithm flow: configs from SHEEPRL_AMD_SEARCH_PATH,
    the module imported via the import_modules config key, entrypoint resolved
    from the registry (docs/register_external_algorithm.md)."""
    import subprocess
    import sys
    import textwrap

    pkg = tmp_path / "myext"
    pkg.mkdir()
    (pkg / "__init__.py").write_text("")
    (pkg / "algo.py").write_text(textwrap.dedent("""
        from sheeprl_amd.utils.registry import register_algorithm

        @register_algorithm(name="my_algo")
        def main(runtime, cfg):
            print("MY_ALGO_RAN", cfg.algo.total_steps)
    """))
    cfgdir = tmp_path / "configs"
    (cfgdir / "algo").mkdir(parents=True)
    (cfgdir / "exp").mkdir()
    (cfgdir / "algo" / "my_algo.yaml").write_text(
        "name: my_algo\ntotal_steps: 7\nper_rank_batch_size: 1\n"
    )
    (cfgdir / "exp" / "my_algo.yaml").write_text(
        "defaults:\n  - override /algo: my_algo\nimport_modules: [myext.algo]\n"
    )
    env = dict(**__import__("os").environ)
    env["SHEEPRL_AMD_SEARCH_PATH"] = str(cfgdir)
    repo_root = __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__)))
    env["PYTHONPATH"] = f"{tmp_path}:{repo_root}:{env.get('PYTHONPATH', '')}"
    out = subprocess.run(
        [sys.executable, "-m", "sheeprl_amd", "exp=my_algo", "env=dummy",
         "buffer.size=8", "runtime.devices=1", "runtime.accelerator=cpu"],
        capture_output=True, text=True, env=env, cwd=str(tmp_path), timeout=240,
    )
    assert "MY_ALGO_RAN 7" in out.stdout, out.stdout + out.stderr


def test_cli_group_selection_beats_exp_override():
    # dreamer_v3_100k_ms_pacman's defaults select env=atari; an explicit CLI
    # group selection must win over the exp's override (compose.py:173-179).
    cfg = compose(["exp=dreamer_v3_100k_ms_pacman", "env=dummy", "env.id=dummy"])
    assert cfg.env.id == "dummy"
    assert cfg.algo.name == "dreamer_v3"
