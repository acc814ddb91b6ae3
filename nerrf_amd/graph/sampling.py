"""CSR adjacency + fixed-fanout neighbor sampling.

GraphSAGE-style fixed fanout turns irregular neighborhood aggregation into a
dense gather over an [N, K] index matrix — the shape the CDNA4 gather-GEMM
kernel (nerrf_amd/ops/hip/gather_sage.hip) is built around.  Sampling is
deterministic given a seed so CPU reference and GPU runs see identical
neighborhoods.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional, Tuple

import numpy as np


@dataclass
class CSRGraph:
    indptr: np.ndarray  # [N+1] int64
    indices: np.ndarray  # [E] int64 neighbor node ids
    weights: np.ndarray  # [E] float32

    @property
    def num_nodes(self) -> int:
        return int(self.indptr.shape[0] - 1)


def to_csr(
    edge_index: np.ndarray,
    num_nodes: int,
    edge_weight: Optional[np.ndarray] = None,
    symmetric: bool = True,
) -> CSRGraph:
    """Destination-indexed CSR (row = dst, cols = incoming srcs).

    GraphSAGE aggregates *incoming* messages; `symmetric=True` adds the
    reverse direction so information flows both ways along each edge.
    """
    src, dst = edge_index[0], edge_index[1]
    w = edge_weight if edge_weight is not None else np.ones(src.shape[0], dtype=np.float32)
    if symmetric:
        src = np.concatenate([src, edge_index[1]])
        dst = np.concatenate([dst, edge_index[0]])
        w = np.concatenate([w, w])
    order = np.argsort(dst, kind="stable")
    dst_s, src_s, w_s = dst[order], src[order], w[order]
    counts = np.bincount(dst_s, minlength=num_nodes)
    indptr = np.zeros(num_nodes + 1, dtype=np.int64)
    np.cumsum(counts, out=indptr[1:])
    return CSRGraph(indptr=indptr, indices=src_s.astype(np.int64), weights=w_s.astype(np.float32))


def sample_fanout(
    csr: CSRGraph,
    fanout: int,
    seed: int = 0,
    self_fill: bool = True,
) -> Tuple[np.ndarray, np.ndarray]:
    """Sample `fanout` incoming neighbors per node.

    Returns (idx [N, K] int64, w [N, K] float32) — neighbor ids and their
    causality weights.  Nodes with fewer than K neighbors repeat what they
    have (sampling with replacement); isolated nodes point at themselves with
    weight 1 when `self_fill` (aggregation degenerates to the node's own
    features, the GraphSAGE convention for isolated vertices).
    """
    n = csr.num_nodes
    k = fanout
    rng = np.random.default_rng(seed)
    deg = np.diff(csr.indptr)
    if len(csr.indices) == 0:
        idx = np.tile(np.arange(n, dtype=np.int64)[:, None], (1, k))
        return idx, np.ones((n, k), dtype=np.float32)
    # vectorised: random offsets modulo degree
    rand = rng.integers(0, 1 << 62, size=(n, k))
    safe_deg = np.maximum(deg, 1)
    offs = (rand % safe_deg[:, None]).astype(np.int64)
    flat = np.clip(csr.indptr[:-1][:, None] + offs, 0, len(csr.indices) - 1)
    idx = csr.indices[flat]
    w = csr.weights[flat].astype(np.float32)
    if self_fill:
        isolated = deg == 0
        if isolated.any():
            idx[isolated] = np.arange(n, dtype=np.int64)[isolated, None]
            w[isolated] = 1.0
    return idx, w


def reverse_index(idx: np.ndarray, w: np.ndarray) -> Tuple[np.ndarray, np.ndarray, np.ndarray]:
    """Reverse index of a sampled-fanout matrix for the backward kernel.

    Returns (rev_dst, rev_src, rev_w), each [N*K], sorted by destination:
    entry e says "source row rev_src[e] gathered from node rev_dst[e] with
    normalised weight rev_w[e] (= w/denom)".  Built once per window batch;
    reused by every layer and training step on that batch.  The sort order
    is what lets the GPU backward run as a segmented reduce.
    """
    n, k = idx.shape
    denom = np.maximum(w.sum(axis=1, keepdims=True), 1e-6)
    wn = (w / denom).astype(np.float32).reshape(-1)
    flat_dst = idx.reshape(-1)
    src_n = np.repeat(np.arange(n, dtype=np.int64), k)
    order = np.argsort(flat_dst, kind="stable")
    return flat_dst[order], src_n[order], wn[order]


def edge_reverse_index(endpoints: np.ndarray) -> Tuple[np.ndarray, np.ndarray, np.ndarray]:
    """Reverse index for an edge-endpoint gather (K=1, unit weights):
    returns (rev_dst, rev_src, rev_w) sorted by destination node."""
    order = np.argsort(endpoints, kind="stable")
    return (
        endpoints[order].astype(np.int64),
        order.astype(np.int64),
        np.ones(len(order), dtype=np.float32),
    )


def sample_fanout_torch(edge_index, edge_weight, num_nodes: int, fanout: int,
                        seed: int = 0, symmetric: bool = True):
    """Device-side CSR build + fixed-fanout sampling (torch tensors in/out).

    The serving tick was spending ~13.6 ms per 600k-event window on the
    HOST argsort inside to_csr/sample_fanout while every consumer of the
    [N, K] neighbor matrices lives on the GPU; this computes them where
    they are used.  Same algorithm as the numpy pair (dst-indexed CSR,
    random offsets modulo degree, self-fill for isolated nodes); the RNG
    stream differs (torch generator), which only changes WHICH valid
    neighbors are drawn.
    """
    import torch

    dev = edge_index.device
    src, dst = edge_index[0], edge_index[1]
    w = edge_weight.float()
    if symmetric:
        src = torch.cat([src, edge_index[1]])
        dst = torch.cat([dst, edge_index[0]])
        w = torch.cat([w, w])
    order = torch.argsort(dst, stable=True)
    src_s, w_s = src[order], w[order]
    counts = torch.bincount(dst[order], minlength=num_nodes)
    indptr = torch.zeros(num_nodes + 1, dtype=torch.int64, device=dev)
    torch.cumsum(counts, 0, out=indptr[1:])
    deg = counts
    k = fanout
    if src_s.numel() == 0:
        idx = torch.arange(num_nodes, device=dev).unsqueeze(1).expand(num_nodes, k).contiguous()
        return idx, torch.ones(num_nodes, k, device=dev)
    g = torch.Generator(device=dev)
    g.manual_seed(int(seed) & 0x7FFFFFFF)
    rand = torch.randint(0, 1 << 62, (num_nodes, k), generator=g, device=dev)
    safe_deg = deg.clamp(min=1)
    offs = rand % safe_deg.unsqueeze(1)
    flat = (indptr[:-1].unsqueeze(1) + offs).clamp(0, src_s.numel() - 1)
    idx = src_s[flat]
    wk = w_s[flat]
    isolated = deg == 0
    if bool(isolated.any()):
        own = torch.arange(num_nodes, device=dev)
        idx[isolated] = own[isolated].unsqueeze(1)
        wk[isolated] = 1.0
    return idx, wk
