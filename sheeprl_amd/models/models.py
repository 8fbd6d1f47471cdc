"""Model primitives.

Parity surface with sheeprl/models/models.py: MLP (:16),
CNN (:122), DeCNN (:205), NatureCNN (:288), LayerNormGRUCell (:331),
MultiEncoder (:413), MultiDecoder (:478), LayerNormChannelLast (:507),
LayerNorm (:521).

MI355X design: every Linear/Conv is followed (when configured) by a FUSED
LayerNorm(+SiLU) op from ``sheeprl_amd.ops`` — one kernel instead of three
eager ops — and the GRU cell's post-GEMM math is one fused kernel
(``ops.gru_gates``).  GEMMs/convs go through hipBLASLt/MIOpen via torch.
"""

from __future__ import annotations

from typing import Any, Callable, Dict, Optional, Sequence, Tuple, Union

import torch
from torch import Tensor, nn

from sheeprl_amd import ops

ModuleType = Optional[Callable[..., nn.Module]]
_ACTS: Dict[str, Callable[[], nn.Module]] = {
    "silu": nn.SiLU,
    "relu": nn.ReLU,
    "tanh": nn.Tanh,
    "elu": nn.ELU,
    "gelu": nn.GELU,
    "identity": nn.Identity,
    "none": nn.Identity,
}


class _LinearFastBias(torch.autograd.Function):
    """F.linear with the bias gradient computed as a GEMM against a ones row.

    torch's autograd computes grad_bias with a bf16 column-reduce that was
    measured at ~300 us for head-sized tensors ([16k, 255]) on MI355X; the
    hipBLASLt GEMV path is ~5 us.  grad_input/grad_weight are the same GEMMs
    autograd would emit.
    """

    @staticmethod
    def forward(ctx, x: Tensor, w: Tensor, b: Tensor) -> Tensor:
        ctx.save_for_backward(x, w)
        x2 = x.reshape(-1, x.shape[-1])
        y = torch.addmm(b, x2, w.t())
        return y.view(*x.shape[:-1], w.shape[0])

    @staticmethod
    def backward(ctx, g: Tensor):
        x, w = ctx.saved_tensors
        g2 = g.reshape(-1, g.shape[-1])
        gx = (g2 @ w).view(x.shape)
        x2 = x.reshape(-1, x.shape[-1])
        gw = g2.t() @ x2
        ones = torch.ones(1, g2.shape[0], device=g.device, dtype=g2.dtype)
        gb = torch.mm(ones, g2).view(-1)
        return gx, gw, gb


class _ConvBiasAdd(torch.autograd.Function):
    """Broadcast bias add over conv output with a fast channels-last
    per-channel sum in backward (torch's bf16 channels-last bias-grad reduce
    was measured at 1.3 ms on the DV3 decoder's final deconv; the HIP kernel
    is ~10 us).  grad_input passes through untouched."""

    @staticmethod
    def forward(ctx, x: Tensor, b: Tensor) -> Tensor:
        ctx.bias_dtype = b.dtype
        return x + b.view(1, -1, *([1] * (x.dim() - 2)))

    @staticmethod
    def backward(ctx, g: Tensor):
        from sheeprl_amd.ops._ext import require_ext

        C = g.shape[1]
        if g.is_contiguous(memory_format=torch.channels_last):
            gb = require_ext().chlast_bias_sum(g, C).to(ctx.bias_dtype)
        else:
            dims = [0] + list(range(2, g.dim()))
            gb = g.sum(dims)
        return g, gb


class FastBiasConv2d(nn.Conv2d):
    """nn.Conv2d drop-in (same state dict) adding bias outside the conv so
    its gradient uses the fast channels-last reduction."""

    def forward(self, x: Tensor) -> Tensor:
        if self.bias is not None and x.is_cuda and torch.is_grad_enabled():
            y = self._conv_forward(x, self.weight, None)
            return _ConvBiasAdd.apply(y, self.bias)
        return super().forward(x)


class FastBiasConvTranspose2d(nn.ConvTranspose2d):
    """nn.ConvTranspose2d drop-in with the fast bias-gradient path."""

    def forward(self, x: Tensor, output_size=None) -> Tensor:
        if self.bias is not None and x.is_cuda and torch.is_grad_enabled():
            num_spatial_dims = 2
            output_padding = self._output_padding(
                x, output_size, self.stride, self.padding, self.kernel_size,
                num_spatial_dims, self.dilation,
            )
            y = torch.nn.functional.conv_transpose2d(
                x, self.weight, None, self.stride, self.padding, output_padding, self.groups, self.dilation
            )
            return _ConvBiasAdd.apply(y, self.bias)
        return super().forward(x, output_size)


class FastLinear(nn.Linear):
    """nn.Linear drop-in (same state dict) using the fast-bias-grad path on
    CUDA when gradients are being recorded."""

    def forward(self, x: Tensor) -> Tensor:
        if self.bias is not None and x.is_cuda and torch.is_grad_enabled():
            return _LinearFastBias.apply(x, self.weight, self.bias)
        return super().forward(x)


def get_activation(act: Union[str, ModuleType, None]) -> Callable[[], nn.Module]:
    if act is None:
        return nn.Identity
    if isinstance(act, str):
        key = act.lower().rsplit(".", 1)[-1]
        if key in _ACTS:
            return _ACTS[key]
        raise ValueError(f"unknown activation '{act}'")
    return act


class LayerNorm(nn.Module):
    """Fused LayerNorm (parity: models.py:521 — a LayerNorm whose input is
    cast to fp32 internally; our fused kernel accumulates in fp32)."""

    def __init__(self, normalized_shape: int, eps: float = 1e-5, elementwise_affine: bool = True) -> None:
        super().__init__()
        self.normalized_shape = int(normalized_shape)
        self.eps = eps
        if elementwise_affine:
            self.weight = nn.Parameter(torch.ones(self.normalized_shape))
            self.bias = nn.Parameter(torch.zeros(self.normalized_shape))
        else:
            self.register_buffer("weight", torch.ones(self.normalized_shape))
            self.register_buffer("bias", torch.zeros(self.normalized_shape))

    def forward(self, x: Tensor) -> Tensor:
        return ops.layer_norm_act(x, self.weight, self.bias, self.eps, "none")


class LayerNormChannelLast(nn.Module):
    """LayerNorm over channels of an NCHW tensor (parity: models.py:507-518:
    permute to NHWC, normalize over C, permute back, dtype preserved)."""

    def __init__(self, normalized_shape: int, eps: float = 1e-5, elementwise_affine: bool = True) -> None:
        super().__init__()
        self.normalized_shape = int(normalized_shape)
        self.eps = eps
        if elementwise_affine:
            self.weight = nn.Parameter(torch.ones(self.normalized_shape))
            self.bias = nn.Parameter(torch.zeros(self.normalized_shape))
        else:
            self.register_buffer("weight", torch.ones(self.normalized_shape))
            self.register_buffer("bias", torch.zeros(self.normalized_shape))

    def forward(self, x: Tensor) -> Tensor:
        if x.dim() != 4:
            raise ValueError(f"expected NCHW input, got {x.shape}")
        y = x.permute(0, 2, 3, 1)
        y = ops.layer_norm_act(y, self.weight, self.bias, self.eps, "none")
        return y.permute(0, 3, 1, 2)


class DenseBlock(nn.Module):
    """Linear -> (fused LayerNorm+activation).  When norm is off, the
    activation runs standalone.  SiLU+LN is one kernel on GPU."""

    def __init__(
        self,
        in_features: int,
        out_features: int,
        *,
        bias: bool = True,
        layer_norm: bool = False,
        layer_norm_eps: float = 1e-3,
        activation: Union[str, ModuleType, None] = None,
    ) -> None:
        super().__init__()
        self.linear = FastLinear(in_features, out_features, bias=bias and not layer_norm)
        self.layer_norm = layer_norm
        act_cls = get_activation(activation)
        self._act_name = "silu" if act_cls is nn.SiLU else "none"
        self.act = act_cls() if not (layer_norm and self._act_name == "silu") else None
        if layer_norm:
            self.ln_weight = nn.Parameter(torch.ones(out_features))
            self.ln_bias = nn.Parameter(torch.zeros(out_features))
            self.ln_eps = layer_norm_eps

    def forward(self, x: Tensor) -> Tensor:
        y = self.linear(x)
        if self.layer_norm:
            y = ops.layer_norm_act(y, self.ln_weight, self.ln_bias, self.ln_eps, self._act_name)
            if self.act is not None and not isinstance(self.act, nn.Identity):
                y = self.act(y)
            return y
        if self.act is not None:
            y = self.act(y)
        return y


class MLP(nn.Module):
    """Configurable Linear stack (parity: models.py:16-119)."""

    def __init__(
        self,
        input_dims: int,
        output_dim: Optional[int] = None,
        hidden_sizes: Sequence[int] = (),
        activation: Union[str, ModuleType, None] = nn.ReLU,
        layer_norm: bool = False,
        layer_norm_eps: float = 1e-3,
        flatten_dim: Optional[int] = None,
        norm_layer: Any = None,
        act_fun_args: Any = None,
        output_activation: Union[str, ModuleType, None] = None,
        bias: bool = True,
    ) -> None:
        super().__init__()
        self.input_dims = input_dims
        self.flatten_dim = flatten_dim
        dims = [input_dims, *hidden_sizes]
        blocks = []
        for i in range(len(dims) - 1):
            blocks.append(
                DenseBlock(
                    dims[i],
                    dims[i + 1],
                    bias=bias,
                    layer_norm=layer_norm,
                    layer_norm_eps=layer_norm_eps,
                    activation=activation,
                )
            )
        if output_dim is not None:
            blocks.append(
                DenseBlock(dims[-1], output_dim, bias=bias, layer_norm=False, activation=output_activation)
            )
        self.model = nn.Sequential(*blocks)
        self.output_dim = output_dim if output_dim is not None else dims[-1]

    def forward(self, x: Tensor) -> Tensor:
        if self.flatten_dim is not None:
            x = x.flatten(self.flatten_dim)
        return self.model(x)


class ConvBlock(nn.Module):
    def __init__(
        self,
        in_ch: int,
        out_ch: int,
        kernel_size: int,
        stride: int = 1,
        padding: int = 0,
        *,
        transpose: bool = False,
        layer_norm: bool = False,
        layer_norm_eps: float = 1e-3,
        activation: Union[str, ModuleType, None] = None,
        bias: bool = True,
    ) -> None:
        super().__init__()
        conv_cls = FastBiasConvTranspose2d if transpose else FastBiasConv2d
        self.conv = conv_cls(in_ch, out_ch, kernel_size, stride, padding, bias=bias and not layer_norm)
        self.layer_norm = layer_norm
        act_cls = get_activation(activation)
        self._act_name = "silu" if act_cls is nn.SiLU else "none"
        self.act = act_cls() if not (layer_norm and self._act_name == "silu") else None
        if layer_norm:
            self.ln_weight = nn.Parameter(torch.ones(out_ch))
            self.ln_bias = nn.Parameter(torch.zeros(out_ch))
            self.ln_eps = layer_norm_eps

    def forward(self, x: Tensor) -> Tensor:
        if x.is_cuda and x.dim() == 4:
            # NHWC path: MIOpen picks CK/igemm kernels and the channel
            # LayerNorm becomes a zero-copy row norm (permute of a
            # channels_last tensor is contiguous NHWC)
            x = x.contiguous(memory_format=torch.channels_last)
        y = self.conv(x)
        if self.layer_norm:
            z = y.permute(0, 2, 3, 1)
            z = ops.layer_norm_act(z, self.ln_weight, self.ln_bias, self.ln_eps, self._act_name)
            y = z.permute(0, 3, 1, 2)
            if not y.is_cuda:
                y = y.contiguous()
            if self.act is not None and not isinstance(self.act, nn.Identity):
                y = self.act(y)
            return y
        if self.act is not None:
            y = self.act(y)
        return y


class CNN(nn.Module):
    """Stride-2 Conv2d stack (parity: models.py:122-203)."""

    def __init__(
        self,
        in_channels: int,
        hidden_channels: Sequence[int],
        kernel_sizes: Optional[Sequence[int]] = None,
        strides: Optional[Sequence[int]] = None,
        paddings: Optional[Sequence[int]] = None,
        activation: Union[str, ModuleType, None] = nn.ReLU,
        layer_norm: bool = False,
        layer_norm_eps: float = 1e-3,
    ) -> None:
        super().__init__()
        n = len(hidden_channels)
        kernel_sizes = list(kernel_sizes or [4] * n)
        strides = list(strides or [2] * n)
        paddings = list(paddings or [1] * n)
        chans = [in_channels, *hidden_channels]
        self.model = nn.Sequential(
            *[
                ConvBlock(
                    chans[i],
                    chans[i + 1],
                    kernel_sizes[i],
                    strides[i],
                    paddings[i],
                    layer_norm=layer_norm,
                    layer_norm_eps=layer_norm_eps,
                    activation=activation,
                )
                for i in range(n)
            ]
        )
        self.output_channels = chans[-1]

    def forward(self, x: Tensor) -> Tensor:
        return self.model(x)


class DeCNN(nn.Module):
    """ConvTranspose2d stack (parity: models.py:205-285)."""

    def __init__(
        self,
        in_channels: int,
        hidden_channels: Sequence[int],
        kernel_sizes: Optional[Sequence[int]] = None,
        strides: Optional[Sequence[int]] = None,
        paddings: Optional[Sequence[int]] = None,
        activation: Union[str, ModuleType, None] = nn.ReLU,
        layer_norm: bool = False,
        layer_norm_eps: float = 1e-3,
        last_layer_plain: bool = True,
    ) -> None:
        super().__init__()
        n = len(hidden_channels)
        kernel_sizes = list(kernel_sizes or [4] * n)
        strides = list(strides or [2] * n)
        paddings = list(paddings or [1] * n)
        chans = [in_channels, *hidden_channels]
        blocks = []
        for i in range(n):
            last = i == n - 1
            blocks.append(
                ConvBlock(
                    chans[i],
                    chans[i + 1],
                    kernel_sizes[i],
                    strides[i],
                    paddings[i],
                    transpose=True,
                    layer_norm=layer_norm and not (last and last_layer_plain),
                    layer_norm_eps=layer_norm_eps,
                    activation=None if (last and last_layer_plain) else activation,
                )
            )
        self.model = nn.Sequential(*blocks)

    def forward(self, x: Tensor) -> Tensor:
        return self.model(x)


class NatureCNN(nn.Module):
    """DQN Nature encoder: 3 convs (8/4, 4/2, 3/1) + Linear
    (parity: models.py:288-328, used by the PPO pixel encoder)."""

    def __init__(self, in_channels: int, features_dim: int = 512, screen_size: int = 64) -> None:
        super().__init__()
        self.conv = nn.Sequential(
            nn.Conv2d(in_channels, 32, 8, 4),
            nn.ReLU(),
            nn.Conv2d(32, 64, 4, 2),
            nn.ReLU(),
            nn.Conv2d(64, 64, 3, 1),
            nn.ReLU(),
            nn.Flatten(),
        )
        with torch.no_grad():
            n_flat = self.conv(torch.zeros(1, in_channels, screen_size, screen_size)).shape[1]
        self.fc = nn.Sequential(nn.Linear(n_flat, features_dim), nn.ReLU())
        self.output_dim = features_dim

    def forward(self, x: Tensor) -> Tensor:
        return self.fc(self.conv(x))


class LayerNormGRUCell(nn.Module):
    """GRU cell with LayerNorm after the input projection
    (parity: models.py:331-410; gate math at :396-403 — the Hafner variant:
    update = sigmoid(u - 1)).  The post-GEMM math is one fused HIP kernel."""

    def __init__(
        self,
        input_size: int,
        hidden_size: int,
        bias: bool = True,
        layer_norm: bool = True,
        layer_norm_eps: float = 1e-3,
    ) -> None:
        super().__init__()
        self.input_size = input_size
        self.hidden_size = hidden_size
        self.linear = nn.Linear(input_size + hidden_size, 3 * hidden_size, bias=bias and not layer_norm)
        self.use_layer_norm = layer_norm
        self.ln_eps = layer_norm_eps
        self.ln_weight = nn.Parameter(torch.ones(3 * hidden_size))
        self.ln_bias = nn.Parameter(torch.zeros(3 * hidden_size))
        if not layer_norm:
            # without LN the fused gate kernel still runs with identity affine
            self.ln_weight.requires_grad_(True)
            self.ln_bias.requires_grad_(True)

    def forward(self, input: Tensor, hx: Tensor) -> Tensor:
        squeeze = False
        if input.dim() == 3:
            input = input.squeeze(0)
            squeeze = True
        if hx.dim() == 3:
            hx = hx.squeeze(0)
        y = self.linear(torch.cat((hx, input), -1))
        if self.use_layer_norm:
            out = ops.gru_gates(y, hx, self.ln_weight, self.ln_bias, self.ln_eps)
        else:
            reset, cand, update = torch.chunk(y, 3, -1)
            reset = torch.sigmoid(reset)
            cand = torch.tanh(reset * cand)
            update = torch.sigmoid(update - 1)
            out = update * cand + (1 - update) * hx
        return out.unsqueeze(0) if squeeze else out


class MultiEncoder(nn.Module):
    """Dict-obs fusion: concat of CNN features and MLP features
    (parity: models.py:413-475)."""

    def __init__(self, cnn_encoder: Optional[nn.Module], mlp_encoder: Optional[nn.Module]) -> None:
        super().__init__()
        if cnn_encoder is None and mlp_encoder is None:
            raise ValueError("at least one of cnn_encoder / mlp_encoder is required")
        self.cnn_encoder = cnn_encoder
        self.mlp_encoder = mlp_encoder
        self.cnn_output_dim = getattr(cnn_encoder, "output_dim", 0) if cnn_encoder else 0
        self.mlp_output_dim = getattr(mlp_encoder, "output_dim", 0) if mlp_encoder else 0
        self.output_dim = self.cnn_output_dim + self.mlp_output_dim

    def forward(self, obs: Dict[str, Tensor]) -> Tensor:
        feats = []
        if self.cnn_encoder is not None:
            feats.append(self.cnn_encoder(obs))
        if self.mlp_encoder is not None:
            feats.append(self.mlp_encoder(obs))
        return torch.cat(feats, dim=-1)


class MultiDecoder(nn.Module):
    """Dict reconstruction from latent (parity: models.py:478-504)."""

    def __init__(self, cnn_decoder: Optional[nn.Module], mlp_decoder: Optional[nn.Module]) -> None:
        super().__init__()
        self.cnn_decoder = cnn_decoder
        self.mlp_decoder = mlp_decoder

    def forward(self, latent: Tensor) -> Dict[str, Tensor]:
        out: Dict[str, Tensor] = {}
        if self.cnn_decoder is not None:
            out.update(self.cnn_decoder(latent))
        if self.mlp_decoder is not None:
            out.update(self.mlp_decoder(latent))
        return out


def cnn_forward(module: nn.Module, x: Tensor, input_dim: Tuple[int, ...], flatten: bool = True) -> Tensor:
    """Fold leading [T, B] dims around a conv stack
    (parity: sheeprl/utils/model.py:165-223)."""
    batch_shape = x.shape[: -len(input_dim)]
    flat = x.reshape(-1, *input_dim)
    y = module(flat)
    if flatten:
        y = y.flatten(1)
    return y.reshape(*batch_shape, *y.shape[1:])
