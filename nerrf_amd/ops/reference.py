"""Pure-PyTorch reference implementations of the fused ops.

These are the numerics ground truth: every HIP kernel is validated against
these in fp32 (tests/test_ops_gpu.py) and they are the CPU execution path.
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch


def gather_mean_ref(
    h: torch.Tensor,  # [N, D]
    idx: torch.Tensor,  # [N, K] int64
    w: torch.Tensor,  # [N, K] float
) -> torch.Tensor:
    """Weighted mean of gathered neighbor rows."""
    n, k = idx.shape
    gathered = h[idx.reshape(-1)].reshape(n, k, h.shape[1])  # [N, K, D]
    wf = w.to(h.dtype).unsqueeze(-1)
    denom = w.sum(dim=1, keepdim=True).to(h.dtype).clamp_min(1e-6)
    return (gathered * wf).sum(dim=1) / denom


def lstm_pointwise_fwd_ref(
    gates_pre: torch.Tensor,  # [B, 4H] pre-activation (x_t W_ih + h W_hh + b)
    c: torch.Tensor,  # [B, H]
    h_prev: torch.Tensor,  # [B, H] (returned unchanged where mask == 0)
    mask: Optional[torch.Tensor],  # [B] or None
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Returns (h_new, c_new, gates_act[B,4H] saved for backward)."""
    hdim = c.shape[1]
    i = torch.sigmoid(gates_pre[:, :hdim])
    f = torch.sigmoid(gates_pre[:, hdim : 2 * hdim])
    g = torch.tanh(gates_pre[:, 2 * hdim : 3 * hdim])
    o = torch.sigmoid(gates_pre[:, 3 * hdim :])
    c_new = f * c + i * g
    h_new = o * torch.tanh(c_new)
    if mask is not None:
        m = mask.to(c.dtype).unsqueeze(-1)
        c_new = m * c_new + (1 - m) * c
        h_new = m * h_new + (1 - m) * h_prev
    gates_act = torch.cat([i, f, g, o], dim=1)
    return h_new, c_new, gates_act


def lstm_pointwise_bwd_ref(
    grad_h: torch.Tensor,  # [B, H]
    grad_c: torch.Tensor,  # [B, H]
    gates_act: torch.Tensor,  # [B, 4H]
    c: torch.Tensor,  # [B, H] previous cell
    mask: Optional[torch.Tensor],
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Returns (grad_gates_pre [B,4H], grad_c_prev [B,H], grad_h_prev_passthrough [B,H]).

    The unmasked c_new is recomputed from (f, i, g, c); where mask == 0 the
    upstream grads are routed straight through to (h_prev, c_prev).
    """
    hdim = c.shape[1]
    i = gates_act[:, :hdim]
    f = gates_act[:, hdim : 2 * hdim]
    g = gates_act[:, 2 * hdim : 3 * hdim]
    o = gates_act[:, 3 * hdim :]
    if mask is not None:
        m = mask.to(c.dtype).unsqueeze(-1)
    else:
        m = torch.ones_like(c[:, :1])
    tanh_c_new = torch.tanh(f * c + i * g)
    # grads on the *active* branch
    gh = grad_h * m
    gc = grad_c * m
    do = gh * tanh_c_new
    dc = gc + gh * o * (1 - tanh_c_new * tanh_c_new)
    di = dc * g
    df = dc * c
    dg = dc * i
    dc_prev = dc * f + grad_c * (1 - m)
    d_ip = di * i * (1 - i)
    d_fp = df * f * (1 - f)
    d_gp = dg * (1 - g * g)
    d_op = do * o * (1 - o)
    grad_gates = torch.cat([d_ip, d_fp, d_gp, d_op], dim=1)
    grad_h_pass = grad_h * (1 - m)  # flows straight to h_prev where masked
    return grad_gates, dc_prev, grad_h_pass


def gather_mean_bwd_ref(
    grad_out: torch.Tensor,  # [N, D]
    idx: torch.Tensor,  # [N, K]
    w: torch.Tensor,  # [N, K]
    num_nodes: int,
) -> torch.Tensor:
    """grad wrt h: scatter-add of normalised weights x grad_out rows."""
    n, k = idx.shape
    denom = w.sum(dim=1, keepdim=True).clamp_min(1e-6)
    wn = (w / denom).to(grad_out.dtype)  # [N, K]
    grad_h = torch.zeros(num_nodes, grad_out.shape[1], device=grad_out.device, dtype=grad_out.dtype)
    contrib = wn.unsqueeze(-1) * grad_out.unsqueeze(1)  # [N, K, D]
    grad_h.index_add_(0, idx.reshape(-1), contrib.reshape(n * k, -1))
    return grad_h
