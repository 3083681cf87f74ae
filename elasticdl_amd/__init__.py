"""elasticdl_amd: an MI355X-native elastic deep-learning training framework.

A from-scratch rebuild of the capabilities of sql-machine-learning/elasticdl
(reference surveyed in SURVEY.md) designed for AMD Instinct MI355X (gfx950):

- PyTorch-ROCm workers, one process per GPU, RCCL (torch.distributed "nccl")
  collectives over xGMI with elastic communicator re-formation.
- A GPU-resident parameter server: sparse embedding tables live in HBM3E,
  served by hand-written HIP/CDNA4 kernels (fused row-wise optimizers,
  gather, sparse-gradient dedup) exposed through a C++ torch extension.
- A Kubernetes-native master (dynamic task sharding, pod watch/relaunch,
  elastic rendezvous) with gRPC control plane.
"""

__version__ = "0.1.0"


def __getattr__(name):
    """Lazy top-level conveniences (torch import stays off the module
    import path until needed)."""
    _exports = {
        "EdlEmbedding": ("elasticdl_amd.layers.embedding", "EdlEmbedding"),
        "FusedDense": ("elasticdl_amd.ops.functional", "FusedDense"),
        "DistributedOptimizer": (
            "elasticdl_amd.collective.distributed_optimizer",
            "DistributedOptimizer",
        ),
        "ElasticAllReduceController": (
            "elasticdl_amd.collective.controller",
            "ElasticAllReduceController",
        ),
        "PSEngine": ("elasticdl_amd.ps.engine", "PSEngine"),
        "FusedBatchNorm2d": (
            "elasticdl_amd.layers.batch_norm", "FusedBatchNorm2d",
        ),
        "BNReLU": ("elasticdl_amd.layers.batch_norm", "BNReLU"),
        "embedding_column": (
            "elasticdl_amd.preprocessing.feature_column", "embedding_column",
        ),
        "DenseFeatures": (
            "elasticdl_amd.preprocessing.feature_column", "DenseFeatures",
        ),
    }
    if name in _exports:
        import importlib

        mod, attr = _exports[name]
        return getattr(importlib.import_module(mod), attr)
    raise AttributeError(name)
