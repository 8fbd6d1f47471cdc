"""Model-primitive unit tests (parity surface: the reference's
tests/test_models.py): shapes, flatten behavior, GRU-cell math, dict
encoder/decoder round trips."""

import numpy as np
import pytest
import torch

from sheeprl_amd.models import (
    CNN,
    DeCNN,
    LayerNormGRUCell,
    MLP,
    MultiDecoder,
    MultiEncoder,
    NatureCNN,
    cnn_forward,
)


def test_mlp_shapes_and_flatten():
    m = MLP(12, 5, [16, 16], activation="relu")
    assert m(torch.randn(7, 12)).shape == (7, 5)
    m2 = MLP(2 * 3 * 4, 6, [8], flatten_dim=1)
    assert m2(torch.randn(5, 2, 3, 4)).shape == (5, 6)
    # no output layer -> last hidden width
    m3 = MLP(4, None, [9])
    assert m3(torch.randn(3, 4)).shape == (3, 9)


def test_mlp_layer_norm_matches_eager():
    torch.manual_seed(0)
    m = MLP(6, None, [8], activation="silu", layer_norm=True, layer_norm_eps=1e-3)
    x = torch.randn(11, 6)
    blk = m.model[0]
    ref = torch.nn.functional.silu(
        torch.nn.functional.layer_norm(blk.linear(x), (8,), blk.ln_weight, blk.ln_bias, 1e-3)
    )
    assert torch.allclose(m(x), ref, atol=1e-5)


def test_cnn_decnn_round_trip_shapes():
    enc = CNN(3, [8, 16], [4, 4], [2, 2], [1, 1], activation="relu")
    y = enc(torch.randn(2, 3, 32, 32))
    assert y.dim() == 4 and y.shape[:2] == (2, 16)
    dec = DeCNN(16, [8, 3], [4, 4], [2, 2], [1, 1], activation="relu")
    z = dec(y)
    assert z.shape[1] == 3


def test_nature_cnn_feature_dim():
    m = NatureCNN(4, features_dim=128, screen_size=64)
    assert m(torch.randn(5, 4, 64, 64)).shape == (5, 128)


def test_layer_norm_gru_cell_math():
    torch.manual_seed(1)
    H, D = 6, 4
    cell = LayerNormGRUCell(D, H, bias=False, layer_norm=True)
    x = torch.randn(3, D)
    h = torch.randn(3, H)
    out = cell(x, h)
    # manual Hafner-gate reference
    z = torch.nn.functional.layer_norm(
        cell.linear(torch.cat((h, x), -1)), (3 * H,), cell.ln_weight, cell.ln_bias, cell.ln_eps
    )
    r, c, u = z.chunk(3, -1)
    r = torch.sigmoid(r)
    c = torch.tanh(r * c)
    u = torch.sigmoid(u - 1)
    ref = u * c + (1 - u) * h
    assert torch.allclose(out, ref, atol=1e-5), (out - ref).abs().max()


def test_multi_encoder_decoder_dict_flow():
    from sheeprl_amd.envs import spaces
    from sheeprl_amd.algos.dreamer_v3.agent import build_agent  # noqa: F401 (import check)

    class CnnEnc(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.keys = ["rgb"]
            self.model = CNN(3, [4], [3], [2], [0], activation="relu")
            self.output_dim = 4 * 15 * 15

        def forward(self, obs):
            x = torch.cat([obs[k] for k in self.keys], dim=-3)
            return cnn_forward(self.model, x, x.shape[-3:], flatten=True)

    class MlpEnc(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.keys = ["state"]
            self.model = MLP(5, None, [7])
            self.output_dim = 7

        def forward(self, obs):
            return self.model(torch.cat([obs[k] for k in self.keys], -1))

    enc = MultiEncoder(CnnEnc(), MlpEnc())
    obs = {"rgb": torch.randn(2, 3, 3, 31, 31), "state": torch.randn(2, 3, 5)}
    out = enc(obs)
    assert out.shape == (2, 3, 4 * 15 * 15 + 7)


def test_cnn_forward_folds_leading_dims():
    m = CNN(1, [2], [3], [1], [0], activation="relu")
    x = torch.randn(4, 5, 1, 8, 8)
    y = cnn_forward(m, x, (1, 8, 8), flatten=True)
    assert y.shape[:2] == (4, 5)


def test_player_states_carry_no_autograd():
    """PlayerDV3.init_states must not leak autograd references from the RSSM
    parameters into the player state (a displaced no_grad decorator did, and
    the stale AccumulateGrad nodes crashed hipGraph capture)."""
    from sheeprl_amd.algos.dreamer_v3.agent import RSSM, RecurrentModel, Actor, PlayerDV3

    S, K, H, DU, P, A, E = 2, 4, 8, 8, 8, 3, 6
    SK = S * K
    rssm = RSSM(
        RecurrentModel(SK + A, H, DU),
        MLP(E + H, SK, [P], activation="silu", layer_norm=True),
        MLP(H, SK, [P], activation="silu", layer_norm=True),
        discrete=K,
    )

    class Enc(torch.nn.Module):
        cnn_keys = []
        mlp_keys = ["state"]

        def forward(self, obs):
            return torch.zeros(1, obs["state"].shape[1], E)

    actor = Actor(SK + H, [A], False, dense_units=DU, mlp_layers=1)
    player = PlayerDV3(Enc(), rssm, actor, [A], num_envs=2, stochastic_size=S,
                       recurrent_state_size=H, device=torch.device("cpu"), discrete_size=K)
    player.init_states()
    for name in ("recurrent_state", "stochastic_state", "actions"):
        t = getattr(player, name)
        assert t.grad_fn is None and not t.requires_grad, f"{name} carries autograd state"
