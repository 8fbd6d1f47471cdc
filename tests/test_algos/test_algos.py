"""Algorithm smoke tests through the real CLI with dummy envs and tiny
configs (the reference's test strategy, SURVEY.md §4: dry_run + num_envs=2 +
cpu + devices in {1,2})."""

import os
import sys
from unittest import mock

import pytest

from sheeprl_amd.cli import run


@pytest.fixture(params=["1", "2"])
def devices(request):
    return request.param


def standard_args(tmp_path, extra):
    return [
        "env=dummy",
        "runtime.accelerator=cpu",
        "dry_run=True",
        "checkpoint.every=0",
        "checkpoint.save_last=True",
        "metric.log_every=1",
        "metric.log_level=1",
        "env.num_envs=2",
        "seed=0",
        *extra,
    ]


def _run(tmp_path, args, devices="1"):
    cwd = os.getcwd()
    os.chdir(tmp_path)
    try:
        run(args + [f"runtime.devices={devices}"])
    finally:
        os.chdir(cwd)


@pytest.mark.timeout(180)
def test_ppo(tmp_path, devices):
    _run(
        tmp_path,
        standard_args(
            tmp_path,
            [
                "exp=ppo",
                "algo.total_steps=64",
                "algo.rollout_steps=8",
                "algo.update_epochs=1",
                "algo.per_rank_batch_size=8",
                "algo.run_test=True",
            ],
        ),
        devices,
    )


@pytest.mark.timeout(180)
def test_ppo_pixel(tmp_path):
    _run(
        tmp_path,
        standard_args(
            tmp_path,
            [
                "exp=ppo",
                "algo.total_steps=32",
                "algo.rollout_steps=8",
                "algo.update_epochs=1",
                "algo.per_rank_batch_size=8",
                "algo.cnn_keys.encoder=[rgb]",
                "algo.mlp_keys.encoder=[state]",
                "algo.run_test=False",
            ],
        ),
    )


@pytest.mark.timeout(180)
def test_ppo_continuous(tmp_path):
    _run(
        tmp_path,
        standard_args(
            tmp_path,
            [
                "exp=ppo",
                "env.id=dummy_continuous",
                "algo.total_steps=32",
                "algo.rollout_steps=8",
                "algo.update_epochs=1",
                "algo.per_rank_batch_size=8",
                "algo.run_test=False",
            ],
        ),
    )


@pytest.mark.timeout(180)
def test_ppo_multidiscrete(tmp_path):
    _run(
        tmp_path,
        standard_args(
            tmp_path,
            [
                "exp=ppo",
                "env.id=dummy_multidiscrete",
                "algo.total_steps=32",
                "algo.rollout_steps=8",
                "algo.update_epochs=1",
                "algo.per_rank_batch_size=8",
                "algo.run_test=False",
            ],
        ),
    )


@pytest.mark.timeout(300)
def test_dreamer_v3(tmp_path, devices):
    _run(
        tmp_path,
        standard_args(
            tmp_path,
            [
                "exp=dreamer_v3",
                "algo=dreamer_v3_S",
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
                "algo.mlp_keys.encoder=[state]",
                "algo.total_steps=16",
                "algo.learning_starts=4",
                "algo.replay_ratio=0.5",
                "buffer.size=64",
                "algo.run_test=True",
                "dry_run=False",
            ],
        ),
        devices,
    )


@pytest.mark.timeout(300)
def test_dreamer_v3_episode_boundary_resets(tmp_path):
    # partial player resets (inference-tensor in-place update) only trigger
    # when an episode ends mid-run: force short episodes with several envs
    _run(
        tmp_path,
        standard_args(
            tmp_path,
            [
                "exp=dreamer_v3",
                "algo=dreamer_v3_S",
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
                "algo.mlp_keys.encoder=[state]",
                "algo.total_steps=48",
                "algo.learning_starts=8",
                "algo.replay_ratio=0.25",
                "env.num_envs=2",
                "env.max_episode_steps=5",
                "buffer.size=128",
                "algo.run_test=False",
                "dry_run=False",
            ],
        ),
    )


@pytest.mark.timeout(300)
def test_dreamer_v3_decoupled_rssm(tmp_path):
    # the sequence-parallel representation variant (reference agent.py:501)
    _run(
        tmp_path,
        standard_args(
            tmp_path,
            [
                "exp=dreamer_v3",
                "algo.world_model.decoupled_rssm=True",
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
                "algo.mlp_keys.encoder=[state]",
                "algo.total_steps=24",
                "algo.learning_starts=4",
                "algo.replay_ratio=0.5",
                "buffer.size=64",
                "algo.run_test=False",
                "dry_run=False",
            ],
        ),
    )


@pytest.mark.timeout(300)
def test_dreamer_v3_checkpoint_resume(tmp_path):
    args = standard_args(
        tmp_path,
        [
            "exp=dreamer_v3",
            "algo=dreamer_v3_S",
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
            "algo.mlp_keys.encoder=[state]",
            "algo.total_steps=16",
            "algo.learning_starts=4",
            "algo.replay_ratio=0.5",
            "buffer.size=64",
            "buffer.checkpoint=True",
            "algo.run_test=False",
            "dry_run=False",
            "checkpoint.save_last=True",
        ],
    )
    _run(tmp_path, args)
    import glob

    ckpts = glob.glob(str(tmp_path / "logs" / "runs" / "**" / "ckpt_*.ckpt"), recursive=True)
    assert ckpts, "no checkpoint written"
    _run(tmp_path, args + [f"checkpoint.resume_from={ckpts[-1]}"])


@pytest.mark.timeout(180)
def test_sac(tmp_path, devices):
    _run(
        tmp_path,
        standard_args(
            tmp_path,
            [
                "exp=sac",
                "env=dummy",
                "env.id=dummy_continuous",
                "algo.total_steps=32",
                "algo.learning_starts=8",
                "algo.per_rank_batch_size=16",
                "algo.replay_ratio=0.5",
                "buffer.size=256",
                "algo.run_test=True",
                "dry_run=False",
            ],
        ),
        devices,
    )


@pytest.mark.timeout(180)
def test_a2c(tmp_path, devices):
    _run(
        tmp_path,
        standard_args(
            tmp_path,
            [
                "exp=a2c",
                "env=dummy",
                "algo.total_steps=40",
                "algo.rollout_steps=5",
                "algo.run_test=True",
                "dry_run=False",
            ],
        ),
        devices,
    )


@pytest.mark.timeout(180)
def test_droq(tmp_path):
    _run(
        tmp_path,
        standard_args(
            tmp_path,
            [
                "exp=droq",
                "env=dummy",
                "env.id=dummy_continuous",
                "algo.total_steps=32",
                "algo.learning_starts=8",
                "algo.per_rank_batch_size=16",
                "algo.replay_ratio=1.0",
                "buffer.size=256",
                "algo.run_test=True",
                "dry_run=False",
            ],
        ),
    )


@pytest.mark.timeout(300)
def test_ppo_decoupled(tmp_path):
    _run(
        tmp_path,
        standard_args(
            tmp_path,
            [
                "exp=ppo_decoupled",
                "env=dummy",
                "algo.total_steps=64",
                "algo.rollout_steps=8",
                "algo.update_epochs=1",
                "algo.per_rank_batch_size=8",
                "algo.run_test=False",
                "dry_run=False",
            ],
        ),
        devices="2",
    )


@pytest.mark.timeout(420)
def test_ppo_decoupled_checkpoint_resume(tmp_path):
    args = standard_args(
        tmp_path,
        [
            "exp=ppo_decoupled",
            "env=dummy",
            "algo.total_steps=64",
            "algo.rollout_steps=8",
            "algo.update_epochs=1",
            "algo.per_rank_batch_size=8",
            "algo.run_test=False",
            "dry_run=False",
            "checkpoint.save_last=True",
        ],
    )
    _run(tmp_path, args, devices="2")
    import glob

    ckpts = glob.glob(str(tmp_path / "logs" / "runs" / "**" / "ckpt_*.ckpt"), recursive=True)
    assert ckpts, "no decoupled checkpoint written"
    import torch as _t

    state = _t.load(ckpts[-1], map_location="cpu", weights_only=False)
    assert "optimizer" in state and state["optimizer"] is not None, "trainer optimizer state missing"
    _run(tmp_path, args + [f"checkpoint.resume_from={ckpts[-1]}"], devices="2")


@pytest.mark.timeout(420)
def test_sac_decoupled_checkpoint_resume(tmp_path):
    args = standard_args(
        tmp_path,
        [
            "exp=sac_decoupled",
            "env=dummy",
            "algo.total_steps=32",
            "algo.learning_starts=4",
            "algo.per_rank_batch_size=8",
            "algo.replay_ratio=0.5",
            "buffer.size=128",
            "algo.run_test=False",
            "dry_run=False",
            "checkpoint.save_last=True",
        ],
    )
    _run(tmp_path, args, devices="2")
    import glob

    ckpts = glob.glob(str(tmp_path / "logs" / "runs" / "**" / "ckpt_*.ckpt"), recursive=True)
    assert ckpts, "no decoupled checkpoint written"
    import torch as _t

    state = _t.load(ckpts[-1], map_location="cpu", weights_only=False)
    assert "qf_optimizer" in state, "trainer optimizer states missing"
    _run(tmp_path, args + [f"checkpoint.resume_from={ckpts[-1]}"], devices="2")


def test_decoupled_requires_multi_device(tmp_path):
    with pytest.raises(Exception):
        _run(
            tmp_path,
            standard_args(tmp_path, ["exp=ppo_decoupled", "env=dummy", "algo.run_test=False"]),
            devices="1",
        )


@pytest.mark.timeout(300)
def test_sac_decoupled(tmp_path):
    _run(
        tmp_path,
        standard_args(
            tmp_path,
            [
                "exp=sac_decoupled",
                "env=dummy",
                "env.id=dummy_continuous",
                "algo.total_steps=32",
                "algo.learning_starts=8",
                "algo.per_rank_batch_size=8",
                "algo.replay_ratio=0.5",
                "buffer.size=256",
                "algo.run_test=False",
                "dry_run=False",
            ],
        ),
        devices="2",
    )


@pytest.mark.timeout(300)
def test_ppo_recurrent(tmp_path, devices):
    _run(
        tmp_path,
        standard_args(
            tmp_path,
            [
                "exp=ppo_recurrent",
                "env=dummy",
                "algo.total_steps=64",
                "algo.rollout_steps=8",
                "algo.update_epochs=1",
                "algo.per_rank_batch_size=2",
                "algo.run_test=True",
                "dry_run=False",
            ],
        ),
        devices,
    )


@pytest.mark.timeout(300)
def test_sac_ae(tmp_path):
    _run(
        tmp_path,
        standard_args(
            tmp_path,
            [
                "exp=sac_ae",
                "env.id=dummy_continuous",
                "algo.total_steps=24",
                "algo.learning_starts=8",
                "algo.per_rank_batch_size=8",
                "algo.replay_ratio=0.5",
                "algo.cnn_keys.encoder=[rgb]",
                "algo.mlp_keys.encoder=[state]",
                "buffer.size=128",
                "algo.run_test=True",
                "dry_run=False",
            ],
        ),
    )


@pytest.mark.timeout(300)
def test_dreamer_v2(tmp_path, devices):
    _run(
        tmp_path,
        standard_args(
            tmp_path,
            [
                "exp=dreamer_v2",
                "env=dummy",
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
                "algo.per_rank_pretrain_steps=1",
                "algo.horizon=3",
                "algo.mlp_keys.encoder=[state]",
                "algo.total_steps=16",
                "algo.learning_starts=4",
                "algo.replay_ratio=0.5",
                "buffer.size=64",
                "algo.run_test=True",
                "dry_run=False",
            ],
        ),
        devices,
    )


@pytest.mark.timeout(300)
def test_dreamer_v1(tmp_path, devices):
    _run(
        tmp_path,
        standard_args(
            tmp_path,
            [
                "exp=dreamer_v1",
                "env=dummy",
                "algo.world_model.stochastic_size=4",
                "algo.world_model.encoder.cnn_channels_multiplier=2",
                "algo.world_model.encoder.dense_units=8",
                "algo.world_model.recurrent_model.recurrent_state_size=8",
                "algo.world_model.recurrent_model.dense_units=8",
                "algo.world_model.transition_model.hidden_size=8",
                "algo.world_model.representation_model.hidden_size=8",
                "algo.world_model.observation_model.dense_units=8",
                "algo.world_model.reward_model.dense_units=8",
                "algo.actor.dense_units=8",
                "algo.actor.mlp_layers=1",
                "algo.critic.dense_units=8",
                "algo.critic.mlp_layers=1",
                "algo.per_rank_batch_size=2",
                "algo.per_rank_sequence_length=4",
                "algo.per_rank_pretrain_steps=1",
                "algo.horizon=3",
                "algo.mlp_keys.encoder=[state]",
                "algo.total_steps=16",
                "algo.learning_starts=4",
                "algo.replay_ratio=0.5",
                "buffer.size=64",
                "algo.run_test=True",
                "dry_run=False",
            ],
        ),
        devices,
    )


P2E_TINY = [
    "algo.dense_units=8",
    "algo.mlp_layers=1",
    "algo.world_model.encoder.cnn_channels_multiplier=2",
    "algo.world_model.recurrent_model.recurrent_state_size=8",
    "algo.world_model.transition_model.hidden_size=8",
    "algo.world_model.representation_model.hidden_size=8",
    "algo.world_model.discrete_size=4",
    "algo.world_model.stochastic_size=4",
    "algo.ensembles.n=2",
    "algo.per_rank_batch_size=2",
    "algo.per_rank_sequence_length=4",
    "algo.horizon=3",
    "algo.mlp_keys.encoder=[state]",
    "algo.total_steps=16",
    "algo.learning_starts=4",
    "algo.replay_ratio=0.5",
    "buffer.size=64",
    "dry_run=False",
]


@pytest.mark.timeout(300)
def test_p2e_dv3_exploration_then_finetuning(tmp_path):
    _run(
        tmp_path,
        standard_args(
            tmp_path,
            ["exp=p2e_dv3_exploration", "env=dummy", "algo.run_test=True",
             "checkpoint.save_last=True", *P2E_TINY],
        ),
    )
    import glob

    ckpts = sorted(glob.glob(str(tmp_path / "logs" / "runs" / "**" / "ckpt_*.ckpt"), recursive=True))
    assert ckpts, "exploration produced no checkpoint"
    _run(
        tmp_path,
        standard_args(
            tmp_path,
            ["exp=p2e_dv3_finetuning", "env=dummy", "algo.run_test=False",
             f"checkpoint.exploration_ckpt_path={ckpts[-1]}", *P2E_TINY],
        ),
    )


@pytest.mark.timeout(300)
def test_p2e_dv2_exploration_then_finetuning(tmp_path):
    args = [
        "env=dummy",
        "algo.dense_units=8",
        "algo.mlp_layers=1",
        "algo.world_model.encoder.cnn_channels_multiplier=2",
        "algo.world_model.recurrent_model.recurrent_state_size=8",
        "algo.world_model.transition_model.hidden_size=8",
        "algo.world_model.representation_model.hidden_size=8",
        "algo.world_model.discrete_size=4",
        "algo.world_model.stochastic_size=4",
        "algo.ensembles.n=2",
        "algo.per_rank_batch_size=2",
        "algo.per_rank_sequence_length=4",
        "algo.per_rank_pretrain_steps=1",
        "algo.horizon=3",
        "algo.mlp_keys.encoder=[state]",
        "algo.total_steps=16",
        "algo.learning_starts=4",
        "algo.replay_ratio=0.5",
        "buffer.size=64",
        "dry_run=False",
    ]
    _run(tmp_path, standard_args(tmp_path, ["exp=p2e_dv2_exploration", "algo.run_test=False",
                                            "checkpoint.save_last=True", *args]))
    import glob

    ckpts = sorted(glob.glob(str(tmp_path / "logs" / "runs" / "**" / "ckpt_*.ckpt"), recursive=True))
    assert ckpts
    _run(tmp_path, standard_args(tmp_path, ["exp=p2e_dv2_finetuning", "algo.run_test=False",
                                            f"checkpoint.exploration_ckpt_path={ckpts[-1]}", *args]))


@pytest.mark.timeout(300)
def test_p2e_dv1_exploration_then_finetuning(tmp_path):
    args = [
        "env=dummy",
        "env.id=dummy_continuous",
        "algo.world_model.stochastic_size=4",
        "algo.world_model.encoder.cnn_channels_multiplier=2",
        "algo.world_model.encoder.dense_units=8",
        "algo.world_model.recurrent_model.recurrent_state_size=8",
        "algo.world_model.recurrent_model.dense_units=8",
        "algo.world_model.transition_model.hidden_size=8",
        "algo.world_model.representation_model.hidden_size=8",
        "algo.world_model.observation_model.dense_units=8",
        "algo.world_model.reward_model.dense_units=8",
        "algo.actor.dense_units=8",
        "algo.actor.mlp_layers=1",
        "algo.critic.dense_units=8",
        "algo.critic.mlp_layers=1",
        "algo.ensembles.n=2",
        "algo.ensembles.dense_units=8",
        "algo.ensembles.mlp_layers=1",
        "algo.per_rank_batch_size=2",
        "algo.per_rank_sequence_length=4",
        "algo.per_rank_pretrain_steps=1",
        "algo.horizon=3",
        "algo.mlp_keys.encoder=[state]",
        "algo.total_steps=16",
        "algo.learning_starts=4",
        "algo.replay_ratio=0.5",
        "buffer.size=64",
        "dry_run=False",
    ]
    _run(tmp_path, standard_args(tmp_path, ["exp=p2e_dv1_exploration", "algo.run_test=False",
                                            "checkpoint.save_last=True", *args]))
    import glob

    ckpts = sorted(glob.glob(str(tmp_path / "logs" / "runs" / "**" / "ckpt_*.ckpt"), recursive=True))
    assert ckpts
    _run(tmp_path, standard_args(tmp_path, ["exp=p2e_dv1_finetuning", "algo.run_test=False",
                                            f"checkpoint.exploration_ckpt_path={ckpts[-1]}", *args]))


@pytest.mark.timeout(300)
def test_eval_and_registration_cli(tmp_path):
    """Train tiny PPO, then exercise `eval` and `register` CLI paths
    (parity: tests/test_algos/test_cli.py eval smoke + registration)."""
    _run(
        tmp_path,
        standard_args(
            tmp_path,
            [
                "exp=ppo",
                "algo.total_steps=32",
                "algo.rollout_steps=8",
                "algo.update_epochs=1",
                "algo.per_rank_batch_size=8",
                "algo.run_test=False",
                "checkpoint.save_last=True",
                "dry_run=False",
            ],
        ),
    )
    import glob
    import os

    from sheeprl_amd.cli import evaluation, registration

    ckpts = sorted(glob.glob(str(tmp_path / "logs" / "runs" / "**" / "ckpt_*.ckpt"), recursive=True))
    assert ckpts
    cwd = os.getcwd()
    os.chdir(tmp_path)
    try:
        evaluation([f"checkpoint_path={ckpts[-1]}", "runtime.accelerator=cpu"])
        registration([f"checkpoint_path={ckpts[-1]}", f"registry_dir={tmp_path}/registry"])
        assert (tmp_path / "registry").exists()
    finally:
        os.chdir(cwd)


@pytest.mark.timeout(300)
def test_eval_cli_dreamer_v3(tmp_path):
    """Train tiny DV3 with save_last, then run the `eval` CLI on its
    checkpoint (parity: sheeprl-eval over the flagship family)."""
    _run(
        tmp_path,
        standard_args(
            tmp_path,
            [
                "exp=dreamer_v3",
                "algo=dreamer_v3_S",
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
                "algo.mlp_keys.encoder=[state]",
                "algo.total_steps=16",
                "algo.learning_starts=4",
                "algo.replay_ratio=0.5",
                "buffer.size=64",
                "algo.run_test=False",
                "checkpoint.save_last=True",
                "dry_run=False",
            ],
        ),
    )
    import glob
    import os

    from sheeprl_amd.cli import evaluation

    ckpts = sorted(glob.glob(str(tmp_path / "logs" / "runs" / "**" / "ckpt_*.ckpt"), recursive=True))
    assert ckpts
    cwd = os.getcwd()
    os.chdir(tmp_path)
    try:
        evaluation([f"checkpoint_path={ckpts[-1]}", "runtime.accelerator=cpu"])
    finally:
        os.chdir(cwd)


def test_minedojo_actor_masking():
    import torch
    from sheeprl_amd.algos.dreamer_v3.agent import MinedojoActor

    torch.manual_seed(0)
    a = MinedojoActor(24, [19, 5, 7], False, dense_units=16, mlp_layers=2, unimix=0.01)
    state = torch.randn(3, 4, 24)
    mask = {
        "mask_action_type": torch.ones(3, 4, 19, dtype=torch.bool),
        "mask_craft_smelt": torch.zeros(3, 4, 5, dtype=torch.bool),
        "mask_equip_place": torch.ones(3, 4, 7, dtype=torch.bool),
        "mask_destroy": torch.zeros(3, 4, 7, dtype=torch.bool),
    }
    mask["mask_action_type"][..., 3] = False
    mask["mask_craft_smelt"][..., 1] = True
    for greedy in (False, True):
        acts, dists = a(state, greedy=greedy, mask=mask)
        assert (acts[0].argmax(-1) != 3).all()
        crafted = acts[0].argmax(-1) == 15
        if crafted.any():
            assert (acts[1].argmax(-1)[crafted] == 1).all()


def test_graph_capture_plumbing_cpu(tmp_path, monkeypatch):
    """Exercise every main's hipGraph-capture wiring on CPU with an eager
    stand-in for CUDAGraphStep — catches argument-plumbing regressions in the
    GPU-only code path."""
    from sheeprl_amd.parallel import graphs as graphs_mod

    class EagerStep:
        def __init__(self, fn, example_inputs, warmup=3, pool=None):
            self.fn = fn
            for _ in range(1):
                fn(example_inputs)

        def __call__(self, inputs):
            self.fn(inputs)

    monkeypatch.setattr(graphs_mod, "CUDAGraphStep", EagerStep)
    monkeypatch.setenv("SHEEPRL_AMD_FORCE_GRAPHS", "1")
    tiny = [
        "algo.dense_units=8", "algo.mlp_layers=1",
        "algo.world_model.encoder.cnn_channels_multiplier=2",
        "algo.world_model.recurrent_model.recurrent_state_size=8",
        "algo.world_model.transition_model.hidden_size=8",
        "algo.world_model.representation_model.hidden_size=8",
        "algo.world_model.discrete_size=4", "algo.world_model.stochastic_size=4",
        "algo.per_rank_batch_size=2", "algo.per_rank_sequence_length=4",
        "algo.horizon=3", "algo.total_steps=20", "algo.learning_starts=4",
        "algo.replay_ratio=0.5", "buffer.size=64", "algo.run_test=False",
        "algo.mlp_keys.encoder=[state]",
    ]
    for exp, extra in [
        ("dreamer_v3", []),
        ("dreamer_v2", ["algo.per_rank_pretrain_steps=1"]),
        ("p2e_dv3_exploration", ["algo.hip_graphs=true", "algo.ensembles.n=2",
                                 "algo.ensembles.dense_units=8", "algo.ensembles.mlp_layers=1"]),
    ]:
        _run(tmp_path, [f"exp={exp}", "env=dummy", "runtime.accelerator=cpu", "dry_run=False",
                        "checkpoint.every=0", "metric.log_every=100", "env.num_envs=2", "seed=0",
                        "algo.cnn_keys.encoder=[]", *tiny, *extra])


@pytest.mark.timeout(420)
def test_ppo_decoupled_two_trainers(tmp_path):
    """world=3: player + TWO trainers — exercises the trainers' DDP group
    all-reduce, equal-size chunk padding across trainers and the rank-1
    metric/weight broadcasts (reference Join-context case)."""
    _run(
        tmp_path,
        standard_args(
            tmp_path,
            [
                "exp=ppo_decoupled",
                "env=dummy",
                "algo.total_steps=64",
                "algo.rollout_steps=9",  # odd size -> uneven chunks get padded
                "algo.update_epochs=1",
                "algo.per_rank_batch_size=4",
                "algo.run_test=False",
                "dry_run=False",
            ],
        ),
        devices="3",
    )


@pytest.mark.timeout(420)
def test_sac_decoupled_two_trainers(tmp_path):
    """world=3 SAC decoupled: sample-chunk scatter across two trainers."""
    _run(
        tmp_path,
        standard_args(
            tmp_path,
            [
                "exp=sac_decoupled",
                "env=dummy",
                "env.id=dummy_continuous",
                "algo.total_steps=32",
                "algo.learning_starts=8",
                "algo.per_rank_batch_size=6",
                "algo.replay_ratio=0.5",
                "buffer.size=256",
                "algo.run_test=False",
                "dry_run=False",
            ],
        ),
        devices="3",
    )
