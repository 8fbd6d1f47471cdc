from sheeprl_amd.optim.fused import FusedAdam, RMSpropTF, make_optimizer

__all__ = ["FusedAdam", "RMSpropTF", "make_optimizer"]
