"""nerrf-amd: MI355X-native temporal-graph anomaly detection + rollback engine.

A from-scratch AMD CDNA4 implementation of the NERRF capability set
(reference: github.com/Itz-Agasta/nerrf): eBPF trace ingest over the
nerrf.trace gRPC contract, 30-60 s temporal dependency graphs in HBM,
GraphSAGE-T + BiLSTM anomaly detection with hand-written HIP kernels,
batched MCTS rollback planning, and RCCL data-parallel training over xGMI.
"""
__version__ = "0.2.0"
