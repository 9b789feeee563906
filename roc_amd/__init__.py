"""roc_amd — MI355X-native distributed full-graph GNN training framework.

A from-scratch rebuild of the capabilities of jiazhihao/ROC (MLSys'20)
designed for AMD Instinct MI355X (gfx950/CDNA4):

- hand-written HIP kernels (MFMA GEMM, CSR SpMM aggregation, fused
  degree-norm, Philox dropout, fused softmax-CE + metrics, fused Adam)
- one process per GPU, RCCL collectives over xGMI (halo exchange by
  all_to_all, flat-bucket gradient all-reduce)
- HBM-resident activations (288 GB/GPU) with optional host-DRAM offload
- edge-balanced contiguous vertex partitioning with cost-model rebalance
"""
__version__ = "0.1.0"

from . import graph  # noqa: F401
from .graph import (CSRGraph, load_lux, synthetic_graph,  # noqa: F401
                    synthetic_dataset, reorder_graph, apply_ordering,
                    ORDERINGS)
from .parallel.partition import (GraphShard, build_shard,  # noqa: F401
                                 edge_balanced_bounds, edge_tensor)
from .optim import AdamOptimizer  # noqa: F401
from .engine import Trainer  # noqa: F401
from .models import build_model, GCN, GraphSAGE, GIN, GAT  # noqa: F401
