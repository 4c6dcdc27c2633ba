"""bnsgcn_amd — MI355X-native partition-parallel full-graph GNN training engine.

A from-scratch AMD-native framework with the capability set of BNS-GCN
(MLSys'22, GATECH-EIC/BNS-GCN): METIS/random graph partitions pinned
one-per-GPU, per-layer boundary-node-sampled halo exchange over RCCL/xGMI,
and hand-written HIP/CDNA4 (gfx950) kernels for the graph + dense hot ops.

Layering (bottom-up):
  graph/     CSR graph core, synthetic named-shape datasets, partitioner + store
  ops/       compute ops: HIP gfx950 kernels with pure-torch CPU references
  parallel/  process groups, boundary discovery, BNS sampling, halo exchange,
             bucketed gradient reduction
  models/    GCN / GraphSAGE / GAT, LayerNorm/SyncBN wiring
  runtime/   training loop, precompute, evaluation, checkpointing, CLI config
  utils/     event-based timers, logging

This is NOT a port of the reference: the reference (pure Python over
DGL/CUDA + gloo) tells us WHAT to build; the architecture
here is MI355X-first (device-resident alltoallv over the fully connected
xGMI clique, inner/halo-split aggregation for comm/compute overlap,
counter-based Philox sampling shared by sender and receiver so no per-epoch
ID exchange is needed).
"""

__version__ = "0.1.0"
