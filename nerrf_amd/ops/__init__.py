"""Fused-op public API with CPU-reference / CDNA4-native dispatch.

Ops:
  * gather_mean(h, idx, w[, rev]) — weighted neighbor aggregation (GraphSAGE-T)
  * gather_rows(h, idx[, rev])    — row gather with fast scatter backward
  * lstm_cell(xg, h, c, w_hh, b, mask) — fused LSTM recurrent step
  * lstm_sequence(...)            — whole-sequence fused LSTM (time-major)

On CPU the pure-PyTorch reference runs; on ROCm devices the in-tree HIP
extension is mandatory (missing extension => RuntimeError, never a silent
eager fallback).
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch

from . import reference as _ref
from .native import get_native, native_available  # noqa: F401 (re-export)

__all__ = ["gather_mean", "lstm_cell", "lstm_sequence", "native_available"]


class _GatherMeanFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, h, idx, w, rev_dst=None, rev_src=None, rev_w=None):
        if rev_dst is not None:
            ctx.save_for_backward(idx, w, rev_dst, rev_src, rev_w)
        else:
            ctx.save_for_backward(idx, w)
        ctx.num_nodes = h.shape[0]
        ext = get_native(h)
        if ext is not None:
            return ext.gather_mean_fwd(h, idx, w)
        return _ref.gather_mean_ref(h, idx, w)

    @staticmethod
    def backward(ctx, grad_out: torch.Tensor):
        saved = ctx.saved_tensors
        idx, w = saved[0], saved[1]
        rev = saved[2:] if len(saved) > 2 else None
        ext = get_native(grad_out)
        if ext is not None:
            if rev is not None:
                grad_h = ext.gather_mean_bwd_csr(
                    grad_out.contiguous(), rev[0], rev[1], rev[2], ctx.num_nodes
                )
            else:
                grad_h = ext.gather_mean_bwd(grad_out.contiguous(), idx, w, ctx.num_nodes)
        else:
            grad_h = _ref.gather_mean_bwd_ref(grad_out, idx, w, ctx.num_nodes)
        return grad_h, None, None, None, None, None


def gather_mean(h: torch.Tensor, idx: torch.Tensor, w: torch.Tensor, rev=None) -> torch.Tensor:
    """Weighted mean over sampled neighbors.

    h: [N, D] node features; idx: [N, K] int64; w: [N, K] float (treated as
    constants — causality weights are data, not parameters).
    rev: optional (rev_dst, rev_src, rev_w) reverse index (graph.sampling
    .reverse_index) enabling the load-balanced segmented-reduce backward.
    """
    if rev is not None:
        return _GatherMeanFn.apply(
            h.contiguous(), idx.contiguous(), w.detach().contiguous(),
            rev[0], rev[1], rev[2],
        )
    return _GatherMeanFn.apply(h.contiguous(), idx.contiguous(), w.detach().contiguous())


def gather_rows(h: torch.Tensor, idx: torch.Tensor, rev=None) -> torch.Tensor:
    """h[idx] with a fast scatter-add backward.

    Replaces advanced indexing on the GPU: eager `h[idx]` backward lowers to
    torch's sort-based `indexing_backward_kernel_many_indices` (~3 ms for
    288k edge endpoints).  With `rev` (sorted (dst, src, w) from
    graph.sampling.edge_reverse_index) the backward is the load-balanced
    segmented reduce; otherwise an fp32 atomic scatter.
    """
    if not h.is_cuda:
        return h[idx]
    ones = torch.ones(idx.shape[0], 1, device=h.device, dtype=torch.float32)
    if rev is not None:
        return _GatherMeanFn.apply(
            h.contiguous(), idx.reshape(-1, 1).contiguous(), ones,
            rev[0], rev[1], rev[2],
        )
    return _GatherMeanFn.apply(h.contiguous(), idx.reshape(-1, 1).contiguous(), ones)


class _LSTMCellFn(torch.autograd.Function):
    """Fused recurrent step: gates = xg + h @ W_hh^T + b; pointwise; mask.

    The recurrent GEMM stays in hipBLASLt (plain library GEMM); the fused HIP
    kernel covers the 4-gate pointwise + state update (the launch-bound part),
    both forward and backward.
    """

    @staticmethod
    def forward(ctx, xg, h, c, w_hh, b, mask):
        ext = get_native(h)
        if ext is not None:
            hg = torch.mm(h, w_hh.t())
            h_new = torch.empty_like(c)
            c_new = torch.empty_like(c)
            gates_act = torch.empty_like(xg)
            ext.lstm_pointwise_fwd(
                hg, xg, b, c, h,
                mask if mask is not None else torch.empty(0, device=h.device),
                h_new, c_new, gates_act,
            )
        else:
            gates_pre = torch.addmm(b, h, w_hh.t()) + xg  # [B, 4H]
            h_new, c_new, gates_act = _ref.lstm_pointwise_fwd_ref(gates_pre, c, h, mask)
        ctx.save_for_backward(gates_act, c, h, w_hh, mask if mask is not None else torch.empty(0))
        return h_new, c_new

    @staticmethod
    def backward(ctx, grad_h: torch.Tensor, grad_c: torch.Tensor):
        gates_act, c, h_prev, w_hh, mask_t = ctx.saved_tensors
        mask = mask_t if mask_t.numel() else None
        ext = get_native(grad_h)
        grad_h = grad_h.contiguous()
        grad_c = grad_c.contiguous()
        if ext is not None:
            grad_gates = torch.empty_like(gates_act)
            grad_c_prev = torch.empty_like(c)
            grad_h_pass = torch.empty_like(c)
            ext.lstm_pointwise_bwd(
                grad_h, torch.empty(0, device=grad_h.device), grad_c, gates_act, c,
                mask if mask is not None else torch.empty(0, device=grad_h.device),
                grad_gates, grad_c_prev, grad_h_pass,
                torch.empty(0, device=grad_h.device),
            )
        else:
            grad_gates, grad_c_prev, grad_h_pass = _ref.lstm_pointwise_bwd_ref(
                grad_h, grad_c, gates_act, c, mask
            )
        grad_xg = grad_gates
        grad_hprev = torch.mm(grad_gates, w_hh) + grad_h_pass
        grad_whh = torch.mm(grad_gates.t(), h_prev)
        grad_b = grad_gates.sum(dim=0)
        return grad_xg, grad_hprev, grad_c_prev, grad_whh, grad_b, None


def lstm_cell(
    xg: torch.Tensor,
    h: torch.Tensor,
    c: torch.Tensor,
    w_hh: torch.Tensor,
    b: torch.Tensor,
    mask: Optional[torch.Tensor] = None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """One fused LSTM step. xg: [B,4H] precomputed input projection."""
    m = None
    if mask is not None:
        m = mask.detach().contiguous().to(xg.dtype)
    return _LSTMCellFn.apply(
        xg.contiguous(), h.contiguous(), c.contiguous(), w_hh, b, m
    )


from .lstm_seq import lstm_sequence  # noqa: E402  (re-export)


def sage_encode_fused(gnn, x, nbr_idx, nbr_w):
    """Inference-only fused GraphSAGE-T encode (bf16, hidden=128 on GPU).

    Runs the input projection in hipBLASLt, then each of the 28 layers as one
    fused MFMA kernel (gather + dual GEMM + GELU + LayerNorm + residual —
    ops/hip/sage_fused.hip).  Falls back to the module path when the shape /
    dtype / device contract is not met.
    """
    h = gnn.input_proj(x)
    if not (
        h.is_cuda
        and h.dtype == torch.bfloat16
        and h.shape[1] == 128
        and nbr_idx.shape[1] <= 64
    ):
        for layer in gnn.layers:
            h = layer(h, nbr_idx, nbr_w)
        return h
    ext = get_native(h)
    idx = nbr_idx.contiguous()
    w = nbr_w.detach().to(torch.float32).contiguous()
    h = h.contiguous()
    for layer in gnn.layers:
        h = ext.sage_layer_fwd(
            h, idx, w,
            layer.w_self.weight, layer.w_nbr.weight, layer.w_nbr.bias,
            layer.norm.weight, layer.norm.bias,
        )
    return h
