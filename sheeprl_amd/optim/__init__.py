from sheeprl_amd.optim.fused import FusedAdam, RMSpropTF

__all__ = ["FusedAdam", "RMSpropTF"]
