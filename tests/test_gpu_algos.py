"""GPU integration smokes: each algorithm family runs a tiny training loop on
cuda:0 through the real CLI (catches GPU-only dtype/layout issues outside the
DreamerV3 bench path)."""

import os

import pytest
import torch

requires_gpu = pytest.mark.gpu


@pytest.fixture(scope="module", autouse=True)
def _check_env():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")


def _run(tmp_path, extra, precision="fp32"):
    from sheeprl_amd.cli import run

    cwd = os.getcwd()
    os.chdir(tmp_path)
    try:
        run(
            [
                "env=dummy",
                "runtime.accelerator=cuda",
                f"runtime.precision={precision}",
                "runtime.devices=1",
                "checkpoint.every=0",
                "checkpoint.save_last=False",
                "metric.log_every=1",
                "env.num_envs=2",
                "seed=0",
                "dry_run=False",
                *extra,
            ]
        )
    finally:
        os.chdir(cwd)


TINY_DV = [
    "algo.dense_units=8",
    "algo.mlp_layers=1",
    "algo.world_model.encoder.cnn_channels_multiplier=2",
    "algo.world_model.recurrent_model.recurrent_state_size=8",
    "algo.world_model.transition_model.hidden_size=8",
    "algo.world_model.representation_model.hidden_size=8",
    "algo.world_model.discrete_size=4",
    "algo.world_model.stochastic_size=4",
    "algo.per_rank_batch_size=2",
    "algo.per_rank_sequence_length=4",
    "algo.horizon=3",
    "algo.total_steps=16",
    "algo.learning_starts=4",
    "algo.replay_ratio=0.5",
    "buffer.size=64",
    "algo.run_test=False",
]


@requires_gpu
@pytest.mark.timeout(300)
def test_gpu_ppo(tmp_path):
    _run(tmp_path, ["exp=ppo", "algo.total_steps=64", "algo.rollout_steps=8",
                    "algo.update_epochs=1", "algo.per_rank_batch_size=8",
                    "algo.cnn_keys.encoder=[rgb]", "algo.mlp_keys.encoder=[state]",
                    "algo.run_test=True"])


@requires_gpu
@pytest.mark.timeout(300)
def test_gpu_a2c(tmp_path):
    _run(tmp_path, ["exp=a2c", "algo.total_steps=40", "algo.rollout_steps=5", "algo.run_test=False"])


@requires_gpu
@pytest.mark.timeout(300)
def test_gpu_sac(tmp_path):
    _run(tmp_path, ["exp=sac", "env=dummy", "env.id=dummy_continuous", "algo.total_steps=32",
                    "algo.learning_starts=8", "algo.per_rank_batch_size=16",
                    "algo.replay_ratio=0.5", "buffer.size=256", "algo.run_test=True"])


@requires_gpu
@pytest.mark.timeout(300)
def test_gpu_droq(tmp_path):
    _run(tmp_path, ["exp=droq", "env=dummy", "env.id=dummy_continuous", "algo.total_steps=32",
                    "algo.learning_starts=8", "algo.per_rank_batch_size=16",
                    "algo.replay_ratio=0.5", "buffer.size=256", "algo.run_test=False"])


@requires_gpu
@pytest.mark.timeout(300)
def test_gpu_sac_ae(tmp_path):
    _run(tmp_path, ["exp=sac_ae", "env.id=dummy_continuous", "algo.total_steps=24",
                    "algo.learning_starts=8", "algo.per_rank_batch_size=8",
                    "algo.replay_ratio=0.5", "algo.cnn_keys.encoder=[rgb]",
                    "algo.mlp_keys.encoder=[state]", "buffer.size=128", "algo.run_test=False"])


@requires_gpu
@pytest.mark.timeout(300)
def test_gpu_ppo_recurrent(tmp_path):
    _run(tmp_path, ["exp=ppo_recurrent", "algo.total_steps=64", "algo.rollout_steps=8",
                    "algo.update_epochs=1", "algo.per_rank_batch_size=2", "algo.run_test=False"])


@requires_gpu
@pytest.mark.timeout(300)
def test_gpu_dreamer_v2(tmp_path):
    _run(tmp_path, ["exp=dreamer_v2", "algo.per_rank_pretrain_steps=1",
                    "algo.mlp_keys.encoder=[state]", *TINY_DV])


@requires_gpu
@pytest.mark.timeout(300)
def test_gpu_dreamer_v1(tmp_path):
    _run(tmp_path, ["exp=dreamer_v1",
                    "algo.world_model.encoder.dense_units=8",
                    "algo.world_model.recurrent_model.dense_units=8",
                    "algo.world_model.observation_model.dense_units=8",
                    "algo.world_model.reward_model.dense_units=8",
                    "algo.actor.dense_units=8", "algo.actor.mlp_layers=1",
                    "algo.critic.dense_units=8", "algo.critic.mlp_layers=1",
                    "algo.per_rank_pretrain_steps=1",
                    "algo.mlp_keys.encoder=[state]",
                    *[a for a in TINY_DV if "discrete_size" not in a and "transition_model" not in a
                      and "representation_model" not in a and "cnn_channels" not in a
                      and "recurrent_state_size" not in a and "dense_units=8" != a.split(".")[-1]],
                    "algo.world_model.stochastic_size=4",
                    "algo.world_model.transition_model.hidden_size=8",
                    "algo.world_model.representation_model.hidden_size=8",
                    "algo.world_model.encoder.cnn_channels_multiplier=2",
                    "algo.world_model.recurrent_model.recurrent_state_size=8"])


@requires_gpu
@pytest.mark.timeout(300)
def test_gpu_dreamer_v3_bf16(tmp_path):
    """bf16-true DV3 on GPU through the real CLI (incl. graph capture path)."""
    _run(tmp_path, ["exp=dreamer_v3", "algo.mlp_keys.encoder=[state]", *TINY_DV],
         precision="bf16")


@requires_gpu
@pytest.mark.timeout(300)
def test_gpu_p2e_dv3_bf16(tmp_path):
    """bf16-true Plan2Explore-DV3 on GPU (exercises the ensemble dtype path)."""
    _run(tmp_path, ["exp=p2e_dv3_exploration", "algo.mlp_keys.encoder=[state]",
                    "algo.ensembles.n=2", "algo.ensembles.dense_units=8",
                    "algo.ensembles.mlp_layers=1", *TINY_DV],
         precision="bf16")


@requires_gpu
@pytest.mark.timeout(300)
def test_gpu_p2e_dv1_graphs(tmp_path):
    """P2E-DV1 exploration with the (round-2 default) hipGraph capture."""
    _run(tmp_path, ["exp=p2e_dv1_exploration", "algo.mlp_keys.encoder=[state]",
                    "algo.ensembles.n=2", "algo.ensembles.dense_units=8",
                    "algo.ensembles.mlp_layers=1",
                    "algo.world_model.encoder.dense_units=8",
                    "algo.world_model.recurrent_model.dense_units=8",
                    "algo.world_model.observation_model.dense_units=8",
                    "algo.world_model.reward_model.dense_units=8",
                    "algo.actor.dense_units=8", "algo.actor.mlp_layers=1",
                    "algo.critic.dense_units=8", "algo.critic.mlp_layers=1",
                    "algo.world_model.stochastic_size=4",
                    "algo.world_model.transition_model.hidden_size=8",
                    "algo.world_model.representation_model.hidden_size=8",
                    "algo.world_model.encoder.cnn_channels_multiplier=2",
                    "algo.world_model.recurrent_model.recurrent_state_size=8",
                    "algo.per_rank_batch_size=2", "algo.per_rank_sequence_length=4",
                    "algo.horizon=3", "algo.total_steps=16", "algo.learning_starts=4",
                    "algo.replay_ratio=0.5", "buffer.size=64", "algo.run_test=False"],
         precision="bf16")


@requires_gpu
@pytest.mark.timeout(300)
def test_gpu_p2e_dv2_graphs(tmp_path):
    """P2E-DV2 exploration with the (round-2 default) hipGraph capture."""
    _run(tmp_path, ["exp=p2e_dv2_exploration", "algo.mlp_keys.encoder=[state]",
                    "algo.ensembles.n=2", "algo.ensembles.dense_units=8",
                    "algo.ensembles.mlp_layers=1", "algo.per_rank_pretrain_steps=1",
                    *TINY_DV],
         precision="bf16")


@requires_gpu
@pytest.mark.timeout(420)
def test_gpu_ppo_bench_graphs(tmp_path):
    """bench --algo ppo captures both the player forward and the minibatch
    step in hipGraphs and emits a valid JSON line (bf16)."""
    import json as _json
    import subprocess
    import sys as _sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ)
    env["PYTHONPATH"] = f"{repo}:{env.get('PYTHONPATH', '')}"
    out = subprocess.run(
        [_sys.executable, os.path.join(repo, "bench.py"), "--algo", "ppo", "--steps", "3",
         "--warmup", "1", "--override", "algo.rollout_steps=32", "--override", "env.num_envs=4",
         "--override", "algo.per_rank_batch_size=64", "--override", "algo.update_epochs=2"],
        capture_output=True, text=True, env=env, cwd=str(tmp_path), timeout=390,
    )
    assert out.returncode == 0, out.stdout[-1500:] + out.stderr[-1500:]
    assert "minibatch step captured" in out.stderr, out.stderr[-800:]
    assert "player forward captured" in out.stderr, out.stderr[-800:]
    rec = _json.loads([l for l in out.stdout.splitlines() if l.startswith("{")][0])
    assert rec["config"]["model"] == "ppo_pixel" and rec["value"] > 0
