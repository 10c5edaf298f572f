from .ddp import DistributedGrads, init_distributed, broadcast_parameters

__all__ = ["DistributedGrads", "init_distributed", "broadcast_parameters"]
