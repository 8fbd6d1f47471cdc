"""CPU-path numerics tests for the ops layer (the GPU kernels are compared
against these same references in tests/test_gpu_kernels.py)."""

import numpy as np
import pytest
import torch

from sheeprl_amd import ops


def test_symlog_symexp_roundtrip():
    x = torch.randn(64) * 10
    assert torch.allclose(ops.symexp(ops.symlog(x)), x, atol=1e-5)


def test_symlog_grad():
    x = torch.randn(32, dtype=torch.float64, requires_grad=True)
    assert torch.autograd.gradcheck(lambda t: ops.symlog(t).sum(), (x,), eps=1e-6)


def test_symexp_grad():
    x = (torch.rand(32, dtype=torch.float64) * 2 - 1).requires_grad_()
    assert torch.autograd.gradcheck(lambda t: ops.symexp(t).sum(), (x,), eps=1e-6)


def test_twohot_roundtrip():
    x = torch.tensor([[0.0], [1.5], [-3.2], [100.0]])
    enc = ops.two_hot_encoder(x, support_range=20, num_buckets=41)
    assert enc.shape == (4, 41)
    assert torch.allclose(enc.sum(-1), torch.ones(4), atol=1e-5)
    dec = ops.two_hot_decoder(enc, support_range=20)
    assert torch.allclose(dec, x, atol=1e-3, rtol=1e-3)


def test_twohot_weights_two_nonzero():
    x = torch.tensor([[0.7]])
    enc = ops.twohot_from_support(x, torch.linspace(-5, 5, 11))
    nz = (enc != 0).sum()
    assert nz <= 2
    assert torch.allclose((enc * torch.linspace(-5, 5, 11)).sum(), torch.tensor(0.7), atol=1e-5)


def test_gae_matches_naive():
    T, N = 16, 4
    torch.manual_seed(0)
    rewards = torch.randn(T, N, 1)
    values = torch.randn(T, N, 1)
    dones = (torch.rand(T, N, 1) < 0.1)
    next_value = torch.randn(N, 1)
    gamma, lam = 0.99, 0.95
    ret, adv = ops.gae(rewards, values, dones, next_value, T, gamma, lam)
    # naive recomputation
    not_dones = (~dones).float()
    lastgaelam = torch.zeros(N, 1)
    expected = torch.zeros(T, N, 1)
    nnt, nv = not_dones[-1], next_value
    for t in reversed(range(T)):
        if t < T - 1:
            nnt, nv = not_dones[t], values[t + 1]
        delta = rewards[t] + gamma * nv * nnt - values[t]
        lastgaelam = delta + gamma * lam * nnt * lastgaelam
        expected[t] = lastgaelam
    assert torch.allclose(adv, expected, atol=1e-5)
    assert torch.allclose(ret, expected + values, atol=1e-5)


def test_lambda_values_forward_and_grad():
    T, B = 8, 6
    torch.manual_seed(1)
    rewards = torch.randn(T, B, dtype=torch.float64, requires_grad=True)
    next_values = torch.randn(T, B, dtype=torch.float64, requires_grad=True)
    continues = torch.rand(T, B, dtype=torch.float64)
    lmbda = 0.95

    def reference(r, nv):
        out = []
        nxt = nv[-1]
        for t in reversed(range(T)):
            nxt = r[t] + continues[t] * ((1 - lmbda) * nv[t] + lmbda * nxt)
            out.append(nxt)
        return torch.stack(out[::-1])

    got = ops.lambda_values(rewards, next_values, continues, lmbda)
    want = reference(rewards, next_values)
    assert torch.allclose(got, want, atol=1e-9)

    g = torch.randn_like(got)
    got.backward(g)
    gr1, gn1 = rewards.grad.clone(), next_values.grad.clone()
    rewards.grad = None
    next_values.grad = None
    reference(rewards, next_values).backward(g)
    assert torch.allclose(gr1, rewards.grad, atol=1e-9)
    assert torch.allclose(gn1, next_values.grad, atol=1e-9)


def test_layer_norm_act_matches_torch():
    torch.manual_seed(2)
    x = torch.randn(5, 7, 16, requires_grad=True)
    w = torch.randn(16, requires_grad=True)
    b = torch.randn(16, requires_grad=True)
    y = ops.layer_norm_act(x, w, b, eps=1e-3, act="silu")
    ref = torch.nn.functional.silu(torch.nn.functional.layer_norm(x, (16,), w, b, eps=1e-3))
    assert torch.allclose(y, ref, atol=1e-5)
    g = torch.randn_like(y)
    y.backward(g)
    gx1, gw1, gb1 = x.grad.clone(), w.grad.clone(), b.grad.clone()
    x.grad = w.grad = b.grad = None
    ref2 = torch.nn.functional.silu(torch.nn.functional.layer_norm(x, (16,), w, b, eps=1e-3))
    ref2.backward(g)
    assert torch.allclose(gx1, x.grad, atol=1e-4)
    assert torch.allclose(gw1, w.grad, atol=1e-4)
    assert torch.allclose(gb1, b.grad, atol=1e-4)


def test_gru_gates_matches_composed():
    torch.manual_seed(3)
    B, H = 4, 8
    y = torch.randn(B, 3 * H, requires_grad=True)
    h = torch.randn(B, H, requires_grad=True)
    w = torch.randn(3 * H, requires_grad=True)
    b = torch.randn(3 * H, requires_grad=True)

    def reference(y, h, w, b):
        z = torch.nn.functional.layer_norm(y, (3 * H,), w, b, eps=1e-3)
        r, c, u = z.chunk(3, -1)
        r = torch.sigmoid(r)
        c = torch.tanh(r * c)
        u = torch.sigmoid(u - 1)
        return u * c + (1 - u) * h

    got = ops.gru_gates(y, h, w, b, eps=1e-3)
    want = reference(y, h, w, b)
    assert torch.allclose(got, want, atol=1e-5)
    g = torch.randn_like(got)
    got.backward(g)
    grads1 = [t.grad.clone() for t in (y, h, w, b)]
    for t in (y, h, w, b):
        t.grad = None
    reference(y, h, w, b).backward(g)
    for g1, t in zip(grads1, (y, h, w, b)):
        assert torch.allclose(g1, t.grad, atol=1e-4), t.shape


def test_ema_update():
    tgt = [torch.ones(4), torch.zeros(3)]
    src = [torch.zeros(4), torch.ones(3)]
    ops.ema_update_(tgt, src, tau=0.25)
    assert torch.allclose(tgt[0], torch.full((4,), 0.75))
    assert torch.allclose(tgt[1], torch.full((3,), 0.25))


def test_normalize_obs():
    x = torch.randint(0, 256, (2, 3, 8, 8), dtype=torch.uint8)
    y = ops.normalize_obs(x)
    assert y.dtype == torch.float32
    assert y.min() >= -0.5 and y.max() <= 0.5


def test_fused_adam_matches_torch_adam():
    torch.manual_seed(4)
    from sheeprl_amd.optim import FusedAdam

    p1 = torch.nn.Parameter(torch.randn(10))
    p2 = torch.nn.Parameter(p1.detach().clone())
    o1 = FusedAdam([p1], lr=1e-2)
    o2 = torch.optim.Adam([p2], lr=1e-2)
    for _ in range(5):
        g = torch.randn(10)
        p1.grad = g.clone()
        p2.grad = g.clone()
        o1.step()
        o2.step()
    assert torch.allclose(p1, p2, atol=1e-6)


def test_ratio_scheduler():
    from sheeprl_amd.utils.utils import Ratio

    r = Ratio(ratio=0.5, pretrain_steps=3)
    assert r(0) == 3
    assert r(8) == 4
    assert r(9) == 0
    assert r(10) == 1
    st = r.state_dict()
    r2 = Ratio(0.5).load_state_dict(st)
    assert r2(12) == 1


def test_ema_update_mixed_layout_falls_back_correctly():
    """ema_update_ must stay correct when target and source layouts differ
    (the HIP fast path requires identical strides; mismatches use foreach)."""
    import torch

    from sheeprl_amd.ops.fused import ema_update_

    t = torch.nn.Conv2d(4, 8, 3).weight.detach().clone().contiguous(memory_format=torch.channels_last)
    s = torch.randn_like(t).contiguous()  # standard layout
    expect = 0.1 * s + 0.9 * t
    ema_update_([t], [s], tau=0.1)
    assert torch.allclose(t, expect, atol=1e-6)


def test_fused_adam_mixed_layout_grads_match_torch_adam():
    """FusedAdam must update channels_last params with standard-layout grads
    correctly (the unused-parameter gradsync path can produce this mix)."""
    import torch

    from sheeprl_amd.optim import FusedAdam

    torch.manual_seed(0)
    base = torch.randn(8, 4, 3, 3)
    p1 = torch.nn.Parameter(base.clone().contiguous(memory_format=torch.channels_last))
    p2 = torch.nn.Parameter(base.clone().contiguous(memory_format=torch.channels_last))
    g = torch.randn(8, 4, 3, 3)  # standard layout
    p1.grad = g.clone()
    p2.grad = g.clone()
    opt1 = FusedAdam([p1], lr=1e-2)
    opt2 = torch.optim.Adam([p2], lr=1e-2)
    for _ in range(3):
        opt1.step()
        opt2.step()
    assert torch.allclose(p1, p2, atol=1e-6)


def test_rmsprop_tf_semantics():
    """TF-style RMSprop: ones-init square_avg and eps INSIDE the sqrt
    (reference optim/rmsprop_tf.py:63-156); verified against a hand rollout."""
    import torch

    from sheeprl_amd.optim import RMSpropTF

    torch.manual_seed(0)
    p = torch.nn.Parameter(torch.tensor([1.0, -2.0]))
    g = torch.tensor([0.5, 0.25])
    lr, alpha, eps = 0.1, 0.9, 1e-10
    opt = RMSpropTF([p], lr=lr, alpha=alpha, eps=eps)
    v = torch.ones(2)
    expect = p.detach().clone()
    for _ in range(3):
        p.grad = g.clone()
        opt.step()
        v = alpha * v + (1 - alpha) * g * g
        expect = expect - lr * g / (v + eps).sqrt()
    assert torch.allclose(p, expect, atol=1e-6)

    # momentum variant accumulates buf = mom*buf + g/sqrt(v+eps)
    p2 = torch.nn.Parameter(torch.tensor([1.0, -2.0]))
    opt2 = RMSpropTF([p2], lr=lr, alpha=alpha, eps=eps, momentum=0.5)
    v2, buf = torch.ones(2), torch.zeros(2)
    expect2 = p2.detach().clone()
    for _ in range(3):
        p2.grad = g.clone()
        opt2.step()
        v2 = alpha * v2 + (1 - alpha) * g * g
        buf = 0.5 * buf + g / (v2 + eps).sqrt()
        expect2 = expect2 - lr * buf
    assert torch.allclose(p2, expect2, atol=1e-6)


def test_fused_adam_zero_grad_unlatches_on_fallback():
    """The in-kernel grad-zeroing flag must be re-derived every step: if a step
    falls off the multi-tensor path (here: the CPU eager path), zero_grad must
    go back to actually zeroing grads."""
    from sheeprl_amd.optim import FusedAdam

    p = torch.nn.Parameter(torch.randn(8))
    opt = FusedAdam([p], lr=1e-2)
    # simulate a previous step having taken the in-kernel-zeroing path
    opt._zeroes_grads_in_kernel = True
    p.grad = torch.ones(8)
    opt.step()  # eager CPU path -> must clear the latch
    assert opt._zeroes_grads_in_kernel is False
    opt.zero_grad(set_to_none=False)
    assert p.grad is not None and p.grad.abs().sum().item() == 0.0


def test_ppo_value_loss_clipped_half_factor():
    """Clipped value loss carries the reference's 0.5 factor
    (sheeprl/algos/ppo/loss.py:61)."""
    from sheeprl_amd.algos.ppo.loss import value_loss

    torch.manual_seed(0)
    nv, ov, ret = torch.randn(32), torch.randn(32), torch.randn(32)
    v = value_loss(nv, ov, ret, clip_coef=0.2, clip_vloss=True)
    unc = (nv - ret) ** 2
    cl = (ov + (nv - ov).clamp(-0.2, 0.2) - ret) ** 2
    assert torch.allclose(v, 0.5 * torch.max(unc, cl).mean())
    # unclipped branch stays plain MSE
    assert torch.allclose(value_loss(nv, ov, ret, 0.2, False), torch.nn.functional.mse_loss(nv, ret))


def test_make_optimizer_honours_config_name():
    """The algo configs' ``optimizer.name`` field selects the optimizer class
    (reference parity: sheeprl configs/optim Hydra group)."""
    from sheeprl_amd.optim import FusedAdam, RMSpropTF, make_optimizer
    from sheeprl_amd.utils.dotdict import DotDict

    p = [torch.nn.Parameter(torch.randn(4))]
    o = make_optimizer(p, DotDict({"name": "adam", "lr": 3e-4, "eps": 1e-5}))
    assert isinstance(o, FusedAdam)
    assert o.param_groups[0]["lr"] == 3e-4 and o.param_groups[0]["eps"] == 1e-5

    o = make_optimizer(p, DotDict({"name": "rmsprop_tf", "lr": 1e-2, "alpha": 0.95, "momentum": 0.9}))
    assert isinstance(o, RMSpropTF)
    assert o.param_groups[0]["alpha"] == 0.95 and o.param_groups[0]["momentum"] == 0.9

    o = make_optimizer(p, DotDict({"name": "sgd", "lr": 1e-1, "weight_decay": None}))
    assert isinstance(o, torch.optim.SGD) and o.param_groups[0]["weight_decay"] == 0.0

    with pytest.raises(ValueError):
        make_optimizer(p, DotDict({"name": "lamb", "lr": 1e-3}))
