from . import comm, sync_bn  # noqa: F401
from .ddp import DistributedDataParallel  # noqa: F401
from .flat import FlatState  # noqa: F401
