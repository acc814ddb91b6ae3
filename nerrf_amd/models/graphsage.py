"""GraphSAGE-T: temporal GraphSAGE for edge/node anomaly classification.

Spec (reference docs architecture.mdx:49-53): 28 layers, ~2 M parameter
ceiling, edge anomaly classification, ROC-AUC >= 0.90 target.  This is a
from-scratch MI355X-first design: fixed-fanout sampled aggregation so the hot
op is a dense weighted gather-mean + GEMM (MFMA-shaped), residual + LayerNorm
so 28 layers train stably, bf16-friendly throughout.

The neighbor aggregation runs through `nerrf_amd.ops.gather_mean`, which
dispatches to the hand-written CDNA4 HIP kernel on ROCm devices and to a
pure-PyTorch reference implementation on CPU.
"""
from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import gather_mean, gather_rows


@dataclass
class SageConfig:
    in_dim: int = 32
    hidden: int = 128
    layers: int = 28
    fanout: int = 16
    dropout: float = 0.1
    edge_head_hidden: int = 64


class SageLayer(nn.Module):
    """h' = h + LN(act(W_s h + W_n agg(h)))  — pre-activation residual."""

    def __init__(self, dim: int, dropout: float = 0.0) -> None:
        super().__init__()
        self.w_self = nn.Linear(dim, dim, bias=False)
        self.w_nbr = nn.Linear(dim, dim, bias=True)
        self.norm = nn.LayerNorm(dim)
        self.dropout = dropout

    def forward(self, h: torch.Tensor, nbr_idx: torch.Tensor, nbr_w: torch.Tensor, rev=None) -> torch.Tensor:
        agg = gather_mean(h, nbr_idx, nbr_w, rev)
        if h.is_cuda and h.dtype == torch.bfloat16 and h.shape[-1] == 128:
            # fused layer tail (ops/hip/sage_ln_act.hip): one kernel fwd,
            # one bwd, replacing ~5/7 eager launches per layer x28 layers
            from ..ops.sage_tail import sage_layer_tail

            return sage_layer_tail(
                h, self.w_self(h), self.w_nbr(agg), self.norm.weight,
                self.norm.bias, self.dropout, self.training,
            )
        z = F.gelu(self.w_self(h) + self.w_nbr(agg))
        if self.dropout > 0 and self.training:
            z = F.dropout(z, self.dropout)
        return h + self.norm(z)


class GraphSAGET(nn.Module):
    def __init__(self, cfg: SageConfig | None = None) -> None:
        super().__init__()
        self.cfg = cfg or SageConfig()
        c = self.cfg
        self.input_proj = nn.Linear(c.in_dim, c.hidden)
        self.layers = nn.ModuleList(SageLayer(c.hidden, c.dropout) for _ in range(c.layers))
        self.node_head = nn.Linear(c.hidden, 1)
        # edge head: [h_src * h_dst, |h_src - h_dst|, weight, ts] -> score
        self.edge_head = nn.Sequential(
            nn.Linear(2 * c.hidden + 2, c.edge_head_hidden),
            nn.GELU(),
            nn.Linear(c.edge_head_hidden, 1),
        )

    def encode(self, x: torch.Tensor, nbr_idx: torch.Tensor, nbr_w: torch.Tensor, rev=None) -> torch.Tensor:
        if (
            not self.training
            and getattr(self, "use_fused_inference", False)
            and not torch.is_grad_enabled()
        ):
            # serving path: each layer = ONE MFMA kernel (ops/hip/sage_fused.hip)
            from ..ops import sage_encode_fused

            return sage_encode_fused(self, x, nbr_idx, nbr_w)
        h = self.input_proj(x)
        for layer in self.layers:
            h = layer(h, nbr_idx, nbr_w, rev)
        return h

    def forward(
        self,
        x: torch.Tensor,  # [N, F]
        nbr_idx: torch.Tensor,  # [N, K] int64
        nbr_w: torch.Tensor,  # [N, K] float — causality weights
        edge_index: torch.Tensor | None = None,  # [2, E]
        edge_weight: torch.Tensor | None = None,  # [E]
        edge_ts: torch.Tensor | None = None,  # [E]
        nbr_rev=None,  # optional reverse CSR (sampling.reverse_index tensors)
        edge_rev=None,  # optional (rev for edge_index[0], rev for edge_index[1])
    ):
        h = self.encode(x, nbr_idx, nbr_w, nbr_rev)
        node_logit = self.node_head(h).squeeze(-1)
        edge_logit = None
        if edge_index is not None and edge_index.numel():
            er0 = edge_rev[0] if edge_rev is not None else None
            er1 = edge_rev[1] if edge_rev is not None else None
            hs, hd = gather_rows(h, edge_index[0], er0), gather_rows(h, edge_index[1], er1)
            ew = edge_weight if edge_weight is not None else torch.ones(
                edge_index.shape[1], device=h.device, dtype=h.dtype
            )
            ets = edge_ts if edge_ts is not None else torch.zeros_like(ew)
            feat = torch.cat(
                [hs * hd, (hs - hd).abs(), ew.unsqueeze(-1).to(h.dtype), ets.unsqueeze(-1).to(h.dtype)],
                dim=-1,
            )
            edge_logit = self.edge_head(feat).squeeze(-1)
        return node_logit, edge_logit

    def num_parameters(self) -> int:
        return sum(p.numel() for p in self.parameters())
