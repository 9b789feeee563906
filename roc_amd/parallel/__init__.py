from .partition import GraphShard, build_shard, edge_balanced_bounds, rebalance_bounds  # noqa: F401
from .halo import halo_exchange  # noqa: F401
