"""Multi-GPU parallelism: sharded GFKB search over torch.distributed.

On MI355X nodes the backend is RCCL over xGMI (``backend="nccl"`` on
ROCm); CPU tests use gloo with the same code path.
"""

from kakveda_amd.parallel.dist import (  # noqa: F401
    get_world,
    init_from_env,
    is_distributed,
)
from kakveda_amd.parallel.sharded import ShardedStore  # noqa: F401
