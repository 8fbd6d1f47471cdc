"""Utility-surface tests: Ratio checkpointing, schedules, timers, metrics,
callbacks (reference tests cover the same utilities)."""

import time

import numpy as np
import pytest
import torch

from sheeprl_amd.utils.metric import MetricAggregator, MeanMetric, SumMetric, MaxMetric
from sheeprl_amd.utils.timer import timer
from sheeprl_amd.utils.utils import Ratio, polynomial_decay, normalize_tensor, safeatanh, safetanh


def test_ratio_checkpoint_roundtrip():
    r = Ratio(0.5)
    issued = [r(i * 10) for i in range(1, 6)]
    state = r.state_dict()
    r2 = Ratio(123.0).load_state_dict(state)
    # continuing from the restored state matches continuing the original
    assert r2(60) == r(60)
    assert r2.state_dict() == r.state_dict()


def test_ratio_pretrain():
    r = Ratio(1.0, pretrain_steps=7)
    first = r(4)
    assert first >= 7
    assert r(5) == 1


def test_polynomial_decay_bounds():
    assert polynomial_decay(0, initial=1.0, final=0.0, max_decay_steps=10) == 1.0
    assert polynomial_decay(10, initial=1.0, final=0.0, max_decay_steps=10) == 0.0
    assert polynomial_decay(99, initial=1.0, final=0.25, max_decay_steps=10) == 0.25
    mid = polynomial_decay(5, initial=1.0, final=0.0, max_decay_steps=10, power=2.0)
    assert 0 < mid < 1


def test_normalize_tensor_masked():
    t = torch.tensor([1.0, 2.0, 3.0, 100.0])
    mask = torch.tensor([True, True, True, False])
    out = normalize_tensor(t, mask=mask)
    sel = t[mask]
    assert torch.allclose(out[:3], (sel - sel.mean()) / (sel.std() + 1e-8))


def test_safetanh_atanh_roundtrip():
    x = torch.linspace(-5, 5, 21)
    y = safetanh(x)
    assert y.abs().max() < 1.0
    assert torch.allclose(safeatanh(y), x.clamp(-5, 5), atol=1e-2)


def test_metric_aggregator_nan_and_types():
    agg = MetricAggregator({"a": "mean", "b": "sum", "c": "max"})
    agg.update("a", 1.0)
    agg.update("a", float("nan"))  # dropped
    agg.update("a", 3.0)
    agg.update("b", 2.0)
    agg.update("b", 5.0)
    agg.update("c", -1.0)
    agg.update("c", 4.0)
    out = agg.compute()
    assert out["a"] == 2.0 and out["b"] == 7.0 and out["c"] == 4.0
    agg.reset()
    assert "a" not in agg.compute()  # empty -> NaN -> dropped


def test_timer_context_accumulates():
    timer.reset()
    with timer("Time/test_section"):
        time.sleep(0.01)
    with timer("Time/test_section"):
        time.sleep(0.01)
    vals = timer.compute()
    assert vals.get("Time/test_section", 0) >= 0.02
    timer.reset()


def test_checkpoint_keep_last(tmp_path):
    from sheeprl_amd.utils.callback import CheckpointCallback

    cb = CheckpointCallback(keep_last=2)
    ckdir = tmp_path / "checkpoint"
    ckdir.mkdir()
    for i in range(5):
        (ckdir / f"ckpt_{i}_0.ckpt").write_bytes(b"x")
        cb._prune(str(ckdir))
    left = sorted(p.name for p in ckdir.glob("*.ckpt"))
    assert len(left) == 2 and "ckpt_4_0.ckpt" in left


def test_ckpt_rb_truncation_mark_and_restore():
    """The buffer-consistency trick (callback.py:40-71): the slot before the
    write head is marked truncated for the snapshot, then restored."""
    from sheeprl_amd.data.buffers import EnvIndependentReplayBuffer, EpisodeBuffer, ReplayBuffer
    from sheeprl_amd.utils.callback import CheckpointCallback

    rb = ReplayBuffer(8, 2)
    step = {
        "obs": np.zeros((1, 2, 3), np.float32),
        "truncated": np.zeros((1, 2, 1), np.float32),
    }
    for _ in range(3):
        rb.add(step)
    restore = CheckpointCallback._ckpt_rb(rb)
    marked = np.asarray(rb["truncated"][(rb._pos - 1) % rb.buffer_size])
    assert (marked == 1).all()
    CheckpointCallback._restore_rb(rb, restore)
    assert (np.asarray(rb["truncated"][(rb._pos - 1) % rb.buffer_size]) == 0).all()

    # EnvIndependent recursion: every sub-buffer marked then restored
    eib = EnvIndependentReplayBuffer(8, n_envs=2)
    sub_step = {
        "obs": np.zeros((1, 1, 3), np.float32),
        "truncated": np.zeros((1, 1, 1), np.float32),
    }
    for _ in range(2):
        eib.add({k: np.repeat(v, 2, axis=1) for k, v in sub_step.items()})
    restore = CheckpointCallback._ckpt_rb(eib)
    for b in eib.buffer:
        assert (np.asarray(b["truncated"][(b._pos - 1) % b.buffer_size]) == 1).all()
    CheckpointCallback._restore_rb(eib, restore)
    for b in eib.buffer:
        assert (np.asarray(b["truncated"][(b._pos - 1) % b.buffer_size]) == 0).all()

    # EpisodeBuffer: open (unfinished) episodes dropped from the snapshot,
    # put back afterwards
    eb = EpisodeBuffer(40, 2, n_envs=1, obs_keys=("obs",))
    eb.add({
        "obs": np.zeros((3, 1, 2), np.float32),
        "terminated": np.zeros((3, 1, 1), np.float32),
        "truncated": np.zeros((3, 1, 1), np.float32),
    })
    assert any(o is not None for o in eb._open)
    restore = CheckpointCallback._ckpt_rb(eb)
    assert all(o is None for o in eb._open)
    CheckpointCallback._restore_rb(eb, restore)
    assert any(o is not None for o in eb._open)
