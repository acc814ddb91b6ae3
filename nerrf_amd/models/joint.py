"""Joint GNN + LSTM anomaly model and loss.

Spec: joint loss over GraphSAGE-T edge/node classification and LSTM sequence
prediction (reference ROADMAP.md:62-69 — "LSTM on edge sequences + joint
loss", ROC-AUC >= 0.90 CI gate).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from .graphsage import GraphSAGET, SageConfig
from .lstm import BiLSTMDetector, LSTMConfig


@dataclass
class JointConfig:
    sage: SageConfig = field(default_factory=SageConfig)
    lstm: LSTMConfig = field(default_factory=LSTMConfig)
    w_node: float = 1.0
    w_edge: float = 1.0
    w_seq: float = 1.0
    pos_weight: float = 4.0  # attacks are the minority class


class NerrfJointModel(nn.Module):
    def __init__(self, cfg: JointConfig | None = None) -> None:
        super().__init__()
        self.cfg = cfg or JointConfig()
        self.gnn = GraphSAGET(self.cfg.sage)
        self.lstm = BiLSTMDetector(self.cfg.lstm)
        # buffer (not a per-call torch.tensor): loss() must stay
        # hipGraph-capturable — scalar H2D copies are forbidden mid-capture
        self.register_buffer(
            "pos_weight_buf", torch.tensor(float(self.cfg.pos_weight)), persistent=False
        )

    def forward(self, batch: Dict[str, torch.Tensor]):
        """batch keys: x, nbr_idx, nbr_w, edge_index, edge_weight, edge_ts,
        seq_feats, seq_lengths (+ optional y_node, y_edge, y_seq)."""
        node_logit, edge_logit = self.gnn(
            batch["x"],
            batch["nbr_idx"],
            batch["nbr_w"],
            batch.get("edge_index"),
            batch.get("edge_weight"),
            batch.get("edge_ts"),
            batch.get("nbr_rev"),
            batch.get("edge_rev"),
        )
        seq_logit = None
        if batch.get("seq_feats") is not None and batch["seq_feats"].shape[0] > 0:
            seq_logit = self.lstm(batch["seq_feats"], batch["seq_lengths"])
        return node_logit, edge_logit, seq_logit

    def loss(
        self,
        node_logit: torch.Tensor,
        edge_logit: Optional[torch.Tensor],
        seq_logit: Optional[torch.Tensor],
        batch: Dict[str, torch.Tensor],
    ) -> Dict[str, torch.Tensor]:
        cfg = self.cfg
        pw = self.pos_weight_buf
        zero = node_logit.new_zeros(())
        losses = {"node": zero, "edge": zero, "seq": zero}
        if batch.get("y_node") is not None:
            losses["node"] = F.binary_cross_entropy_with_logits(
                node_logit.float(), batch["y_node"].float(), pos_weight=pw.float()
            )
        if edge_logit is not None and batch.get("y_edge") is not None and edge_logit.numel():
            losses["edge"] = F.binary_cross_entropy_with_logits(
                edge_logit.float(), batch["y_edge"].float(), pos_weight=pw.float()
            )
        if seq_logit is not None and batch.get("y_seq") is not None and seq_logit.numel():
            losses["seq"] = F.binary_cross_entropy_with_logits(
                seq_logit.float(), batch["y_seq"].float(), pos_weight=pw.float()
            )
        total = cfg.w_node * losses["node"] + cfg.w_edge * losses["edge"] + cfg.w_seq * losses["seq"]
        losses["total"] = total
        return losses

    def num_parameters(self) -> int:
        return sum(p.numel() for p in self.parameters())
