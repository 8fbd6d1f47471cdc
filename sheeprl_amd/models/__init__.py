from sheeprl_amd.models.models import (
    CNN,
    DeCNN,
    DenseBlock,
    LayerNorm,
    LayerNormChannelLast,
    LayerNormGRUCell,
    MLP,
    MultiDecoder,
    MultiEncoder,
    NatureCNN,
    cnn_forward,
    get_activation,
)

__all__ = [
    "MLP",
    "CNN",
    "DeCNN",
    "DenseBlock",
    "NatureCNN",
    "LayerNorm",
    "LayerNormChannelLast",
    "LayerNormGRUCell",
    "MultiEncoder",
    "MultiDecoder",
    "cnn_forward",
    "get_activation",
]
