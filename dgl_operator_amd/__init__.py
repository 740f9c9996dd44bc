"""dgl_operator_amd — an MI355X-native distributed-GNN training framework + job operator.

A from-scratch rebuild of the capabilities of Qihoo360/dgl-operator for AMD
Instinct MI355X (gfx950): PyTorch-ROCm trainer with hand-written HIP/CDNA4
kernels for the GNN hot path (g-SpMM, g-SDDMM, edge-softmax, neighbor
sampling, KGE scoring, sparse Adagrad) and RCCL-over-xGMI collectives for the
distributed plane. The reference delegates its math to DGL/DGL-KE
(see /root/reference SURVEY.md §2.4); here the compute plane is native.

Subpackages
-----------
graph         CSR/CSC graph containers, message-flow blocks, R-MAT generator,
              graph partitioning (reference: dgl.distributed.partition_graph).
ops           Autograd ops backed by HIP kernels on GPU and pure-PyTorch fp32
              reference implementations on CPU.
nn            SAGEConv / GraphConv / GATConv layers
              (reference: dgl.nn.* used by examples/*/code/*.py).
models        GraphSAGE (DistSAGE), GCN, GAT link predictor, KGE model zoo
              (reference: examples/GraphSAGE_dist/code/train_dist.py,
              examples/DGL-KE/hotfix/kvserver.py score functions).
distributed   Process-group setup (RCCL/gloo), partition book, DistGraph
              with alltoallv feature pulls, sharded KVStore w/ sparse Adagrad
              (reference: examples/DGL-KE/hotfix/dis_kvstore.py).
operator_plane DGLJob API types, reconciler state machine, fake cluster,
              watcher loop (reference: controllers/dgljob_controller.go,
              watcher-loop/). Implemented in Python because this image has
              no Go toolchain; semantics follow the reference faithfully.
tools         dglrun workflow CLI, launch/dispatch/revise_hostfile
              (reference: python/dglrun/).
"""

__version__ = "0.2.0"

try:  # compute plane needs torch; the control-plane-only manager image
    from . import graph  # noqa: F401
    from . import fn  # noqa: F401
    from .graph import (  # noqa: F401
        Graph, Block, NID, EID, batch_graphs, batch_num_edges,
        to_bidirected, unbatch,
    )
except ImportError:  # pragma: no cover
    graph = None
