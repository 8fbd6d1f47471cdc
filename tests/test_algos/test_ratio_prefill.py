"""Round-2 parity regressions: replay-ratio prefill accounting and the
fast unit-Normal log_prob."""

import math

import pytest
import torch
import torch.distributions as td


def test_fast_unit_normal_matches_td_independent():
    """_FastUnitNormal.log_prob == Independent(Normal(loc, 1), n).log_prob
    (same math, minus torch's scale-broadcast temporaries)."""
    from sheeprl_amd.algos.dreamer_v2.dreamer_v2 import _FastUnitNormal

    torch.manual_seed(0)
    for shape, n in [((6, 4, 3, 8, 8), 3), ((5, 7, 1), 1), ((9,), 0)]:
        loc = torch.randn(*shape)
        x = torch.randn(*shape)
        fast = _FastUnitNormal(loc, n)
        ref = td.Independent(td.Normal(loc, torch.ones(())), n) if n else td.Normal(loc, torch.ones(()))
        assert torch.allclose(fast.log_prob(x), ref.log_prob(x), atol=1e-5)
        assert torch.equal(fast.mean, loc) and torch.equal(fast.mode, loc)


def test_ratio_counts_start_after_prefill(tmp_path, monkeypatch):
    """The first train iteration must NOT pay a learning_starts-sized
    gradient-step backlog (reference subtracts the prefill policy steps
    before asking Ratio; sheeprl dreamer_v3.py:661)."""
    import sheeprl_amd.algos.dreamer_v3.dreamer_v3 as dv3
    from sheeprl_amd.cli import run

    calls = []
    orig = dv3.train

    def counting_train(*a, **k):
        calls.append(1)
        return orig(*a, **k)

    monkeypatch.setattr(dv3, "train", counting_train)
    # metric.log_level=0 flips the class-level timer/aggregator disable
    # flags; restore them so later tests see the default state
    from sheeprl_amd.utils.metric import MetricAggregator
    from sheeprl_amd.utils.timer import timer

    monkeypatch.setattr(timer, "disabled", timer.disabled, raising=False)
    monkeypatch.setattr(MetricAggregator, "disabled", MetricAggregator.disabled, raising=False)
    run([
        "exp=dreamer_v3", "algo=dreamer_v3_S", "env=dummy", "runtime.accelerator=cpu",
        "env.num_envs=1", "seed=0",
        "algo.dense_units=8", "algo.mlp_layers=1",
        "algo.world_model.encoder.cnn_channels_multiplier=2",
        "algo.world_model.recurrent_model.recurrent_state_size=8",
        "algo.world_model.transition_model.hidden_size=8",
        "algo.world_model.representation_model.hidden_size=8",
        "algo.world_model.discrete_size=4", "algo.world_model.stochastic_size=4",
        "algo.per_rank_batch_size=2", "algo.per_rank_sequence_length=4", "algo.horizon=3",
        "algo.mlp_keys.encoder=[state]", "algo.total_steps=48", "algo.learning_starts=16",
        "algo.replay_ratio=1", "buffer.size=64", "algo.run_test=False", "dry_run=False",
        f"root_dir={tmp_path}", "metric.log_level=0", "checkpoint.every=0",
    ])
    # post-prefill policy steps = 48 - 16 = 32 at ratio 1 -> ~32 grad steps;
    # the pre-fix behavior paid a 16-step backlog on the first train call
    # (total ~48).  Allow slack for the prefill-minus-one-iteration rule.
    assert 28 <= len(calls) <= 36, len(calls)
