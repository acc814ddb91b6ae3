"""Dual-direction LSTM input projection with custom streaming GEMMs.

Both directions of a BiLSTM layer project the SAME input: xg_f = x @ W_f^T,
xg_b = x @ W_b^T ([M ~ 6.4M, 512] x [1024, 512] at production shapes).  As
two hipBLASLt calls the A panel streams from HBM twice and each call's
column tiles re-read it again (2.3-2.9 TB/s effective after tuning —
profiles/train_kstats_vec_r01.txt).  `proj_fwd_dual` (stream_gemm.hip)
stages each 128-row A strip in LDS once and produces both outputs; the
backward's input gradient sums both directions in one pass
(`proj_dgrad_dual`).  Weight grads stay on hipBLASLt for now (huge-K
reduction — own kernel is round-2 follow-up work).

Measured round 2: the custom kernels run at 0.88x (fwd) / 0.77x (dgrad)
of tuned hipBLASLt at production shapes (profiles/PROFILES.md ladder), so
they are OPT-IN (NERRF_STREAM_PROJ=1) and the model uses the concatenated
single-GEMM layout instead (models/lstm.py).  Kept as the validated
baseline for the round-3 pipelined rewrite.
"""
from __future__ import annotations

import os

import torch

from .native import get_native


def _use_stream(x: torch.Tensor, w_f: torch.Tensor, w_b: torch.Tensor) -> bool:
    return (
        x.is_cuda
        and x.dtype == torch.bfloat16
        and os.environ.get("NERRF_STREAM_PROJ", "0") == "1"
        and x.dim() == 2
        and x.shape[1] == 512
        and tuple(w_f.shape) == (1024, 512)
        and tuple(w_b.shape) == (1024, 512)
        and x.stride(1) == 1
        and w_f.is_contiguous()
        and w_b.is_contiguous()
    )


class _DualProjFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w_f, w_b):
        ext = get_native(x)
        ctx.save_for_backward(x, w_f, w_b)
        if ext is not None and hasattr(ext, "proj_fwd_dual") and _use_stream(x, w_f, w_b):
            m = x.shape[0]
            c1 = torch.empty(m, 1024, device=x.device, dtype=x.dtype)
            c2 = torch.empty(m, 1024, device=x.device, dtype=x.dtype)
            ext.proj_fwd_dual(x, w_f, w_b, c1, c2)
            return c1, c2
        return torch.matmul(x, w_f.t()), torch.matmul(x, w_b.t())

    @staticmethod
    def backward(ctx, g1, g2):
        x, w_f, w_b = ctx.saved_tensors
        ext = get_native(x)
        g1 = g1.contiguous()
        g2 = g2.contiguous()
        gx = None
        if ctx.needs_input_grad[0]:
            if (
                ext is not None
                and hasattr(ext, "proj_dgrad_dual")
                and _use_stream(x, w_f, w_b)
            ):
                gx = torch.empty_like(x)
                ext.proj_dgrad_dual(
                    g1, g2, w_f.t().contiguous(), w_b.t().contiguous(), gx
                )
            else:
                gx = torch.matmul(g1, w_f) + torch.matmul(g2, w_b)
        gwf = gwb = None
        if ctx.needs_input_grad[1] or ctx.needs_input_grad[2]:
            if (
                ext is not None
                and hasattr(ext, "proj_wgrad")
                and _use_stream(x, w_f, w_b)
                and os.environ.get("NERRF_STREAM_WGRAD", "1") == "1"
            ):
                # chunk count sized so the grid is a few block-waves deep
                n_chunks = max(16, min(256, x.shape[0] // 8192))
                dw1, dw2 = ext.proj_wgrad(g1, g2, x, n_chunks)
                gwf = dw1.to(x.dtype)
                gwb = dw2.to(x.dtype)
            else:
                gwf = torch.matmul(g1.t(), x) if ctx.needs_input_grad[1] else None
                gwb = torch.matmul(g2.t(), x) if ctx.needs_input_grad[2] else None
        return gx, gwf, gwb


def dual_projection(x: torch.Tensor, w_f: torch.Tensor, w_b: torch.Tensor):
    """(x @ w_f^T, x @ w_b^T) — one A pass on GPU for the BiLSTM layer."""
    return _DualProjFn.apply(x, w_f, w_b)
