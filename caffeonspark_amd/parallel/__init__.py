from .ddp import DistributedSync, init_distributed  # noqa: F401
