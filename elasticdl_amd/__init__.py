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
