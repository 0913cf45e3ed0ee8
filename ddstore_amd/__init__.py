"""ddstore_amd -- MI355X-native distributed in-HBM sample store.

Brand-new framework with the capabilities and API surface of ORNL/DDStore
(see SURVEY.md), designed for CDNA4/gfx950: shards in HBM3E, one-sided
peer reads over xGMI via hipIpc, batched hand-written HIP gather kernels,
RCCL (torch.distributed "nccl") metadata plane, POSIX-shm CPU compatibility
path. See ddstore_amd/csrc for the native core.
"""
from .comm import Comm, as_comm
from .store import DDStore
from .distdataset import DistDataset, nsplit
from .prefetch import PrefetchLoader
from .reshuffle import reshuffle_epoch
from . import debug, io

__version__ = "0.1.0"

__all__ = [
    "Comm",
    "as_comm",
    "DDStore",
    "DistDataset",
    "nsplit",
    "PrefetchLoader",
    "reshuffle_epoch",
    "debug",
    "io",
    "__version__",
]
