from msbn.utils.logging import DDPLogger, comm_log, is_master, master_print  # noqa: F401
from msbn.utils.checkpoint import load_checkpoint, save_checkpoint  # noqa: F401
from msbn.utils import env  # noqa: F401
from msbn.utils.graphs import GraphedStep  # noqa: F401

__all__ = [
    "DDPLogger",
    "comm_log",
    "is_master",
    "master_print",
    "save_checkpoint",
    "load_checkpoint",
    "env",
    "GraphedStep",
]
