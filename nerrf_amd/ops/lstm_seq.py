"""Time-major fused LSTM sequence op.

One autograd Function per (layer, direction) covering all T timesteps:
  forward : per step exactly 1 hipBLASLt GEMM (h @ W_hh^T, beta=0) and 1 fused
            HIP kernel (add xg + bias, 4-gate pointwise, masked state update)
            writing straight into time-major output buffers;
  backward: per step 1 fused gate-gradient kernel + 1 GEMM, then the weight
            gradients as TWO large batched GEMMs over all timesteps
            (grad_gates [T*B,4H]^T @ h_in [T*B,H]) and one reduction for bias.

This removes the per-step elementwise adds / stacks / reduces that dominated
the naive schedule (profiles/: 84% of busy time before the restructure).
Everything is time-major [T, B, ...]; the model transposes once per sequence.
"""
from __future__ import annotations

from typing import Optional

import torch

from . import reference as _ref
from .native import get_native


class _LSTMSeqFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, xg, h0, c0, w_hh, bias, mask, reverse: bool, infer: bool = False):
        """xg: [T, B, 4H] (time-major input projection); mask: [T, B] or None.

        Returns h_all: [T, B, H] (time-major hidden states).
        """
        t_len, batch, gdim = xg.shape
        hdim = gdim // 4
        dev, dt = xg.device, xg.dtype
        ext = get_native(xg)
        # inference (no_grad at the lstm_sequence call site) skips the
        # backward-only activated-gates store: 32 MB of HBM writes per step
        # at serving batch sizes.  (grad mode is always off inside
        # Function.forward, so the flag is computed by the wrapper.)
        h_all = torch.empty(t_len, batch, hdim, device=dev, dtype=dt)
        c_all = torch.empty(t_len, batch, hdim, device=dev, dtype=dt)
        gates_all = torch.empty(
            0 if infer else t_len, batch, gdim, device=dev, dtype=dt
        )
        g_none = torch.empty(0, device=dev, dtype=dt)
        h = h0.contiguous()
        c = c0.contiguous()
        steps = range(t_len - 1, -1, -1) if reverse else range(t_len)
        if ext is not None:
            empty_mask = torch.empty(0, device=dev)
            # The fully-fused MFMA step (lstm_step_fused.hip) is numerically
            # validated but measured slower than hipBLASLt-GEMM + fused
            # pointwise at this shape (215us vs 48us per step: 1 block/CU and
            # a single-buffered W slice expose full L2 latency); keep it
            # opt-in until the pipelined variant lands.
            import os

            fused = (
                os.environ.get("NERRF_FUSED_LSTM", "0") == "1"
                and dt == torch.bfloat16 and hdim == 256 and w_hh.is_contiguous()
            )
            if fused:
                # fully-fused MFMA step: no separate GEMM, no gates HBM round trip
                bias_c = bias.contiguous()
                # k-slice-contiguous weight tiling (see lstm_step_fused.hip)
                w_tiled = w_hh.reshape(gdim, hdim // 32, 32).permute(1, 0, 2).contiguous()
                # the fused kernel always writes gates; reuse one scratch
                # slab in inference instead of a [T, B, 4H] history
                g_scratch = (
                    torch.empty(batch, gdim, device=dev, dtype=dt) if infer else None
                )
                for ti in steps:
                    ext.lstm_step_fused(
                        h, w_tiled, xg[ti], bias_c, c,
                        mask[ti] if mask is not None else empty_mask,
                        h_all[ti], c_all[ti],
                        g_scratch if infer else gates_all[ti], False,
                    )
                    h = h_all[ti]
                    c = c_all[ti]
            else:
                w_hh_t = w_hh.t().contiguous()
                hg = torch.empty(batch, gdim, device=dev, dtype=dt)
                for ti in steps:
                    torch.mm(h, w_hh_t, out=hg)
                    ext.lstm_pointwise_fwd(
                        hg, xg[ti], bias, c, h,
                        mask[ti] if mask is not None else empty_mask,
                        h_all[ti], c_all[ti], g_none if infer else gates_all[ti],
                    )
                    h = h_all[ti]
                    c = c_all[ti]
        else:
            for ti in steps:
                gates_pre = torch.addmm(bias, h, w_hh.t()) + xg[ti]
                h_new, c_new, g_act = _ref.lstm_pointwise_fwd_ref(
                    gates_pre, c, h, mask[ti] if mask is not None else None
                )
                h_all[ti] = h_new
                c_all[ti] = c_new
                if not infer:
                    gates_all[ti] = g_act
                h, c = h_new, c_new
        ctx.save_for_backward(
            gates_all, h_all, c_all, h0, c0, w_hh,
            mask if mask is not None else torch.empty(0, device=dev),
        )
        ctx.reverse = reverse
        return h_all

    @staticmethod
    def backward(ctx, grad_out):
        gates_all, h_all, c_all, h0, c0, w_hh, mask_t = ctx.saved_tensors
        reverse = ctx.reverse
        mask = mask_t if mask_t.numel() else None
        t_len, batch, hdim = h_all.shape
        gdim = 4 * hdim
        dev, dt = h_all.device, h_all.dtype
        ext = get_native(h_all)
        grad_out = grad_out.contiguous()

        grad_gates_all = torch.empty(t_len, batch, gdim, device=dev, dtype=dt)
        grad_h = torch.zeros(batch, hdim, device=dev, dtype=dt)
        grad_c = torch.zeros(batch, hdim, device=dev, dtype=dt)
        grad_h_pass = torch.empty(batch, hdim, device=dev, dtype=dt)
        grad_c_prev = torch.empty(batch, hdim, device=dev, dtype=dt)

        # iterate in the opposite order of forward
        steps = range(t_len) if reverse else range(t_len - 1, -1, -1)
        empty_mask = torch.empty(0, device=dev)
        for ti in steps:
            # c/h input of step ti = previous step's output (or h0/c0 at start)
            first = (ti == t_len - 1) if reverse else (ti == 0)
            if first:
                c_in = c0
            else:
                c_in = c_all[ti + 1] if reverse else c_all[ti - 1]
            if ext is not None:
                # grad_out[ti] is folded inside the kernel (no separate add)
                ext.lstm_pointwise_bwd(
                    grad_h.contiguous(), grad_out[ti], grad_c.contiguous(),
                    gates_all[ti], c_in.contiguous(),
                    mask[ti] if mask is not None else empty_mask,
                    grad_gates_all[ti], grad_c_prev, grad_h_pass,
                )
                grad_h = torch.addmm(grad_h_pass, grad_gates_all[ti], w_hh)
            else:
                gg, gcp, ghp = _ref.lstm_pointwise_bwd_ref(
                    grad_h + grad_out[ti], grad_c, gates_all[ti], c_in,
                    mask[ti] if mask is not None else None,
                )
                grad_gates_all[ti] = gg
                grad_c_prev = gcp
                grad_h_pass = ghp
                grad_h = torch.mm(grad_gates_all[ti], w_hh) + grad_h_pass
            grad_c, grad_c_prev = grad_c_prev, grad_c  # ping-pong buffers

        # weight grads over all timesteps: the h input of step ti is h_all
        # shifted by one step, whose bulk is a CONTIGUOUS prefix/suffix of
        # h_all — two GEMMs (bulk + the h0 boundary row) instead of
        # materialising a shifted copy of the whole history
        gg2 = grad_gates_all.reshape(t_len * batch, gdim)
        if t_len > 1:
            if reverse:
                bulk_gg = grad_gates_all[:-1].reshape((t_len - 1) * batch, gdim)
                bulk_h = h_all[1:].reshape((t_len - 1) * batch, hdim)
                edge_gg = grad_gates_all[t_len - 1]
            else:
                bulk_gg = grad_gates_all[1:].reshape((t_len - 1) * batch, gdim)
                bulk_h = h_all[:-1].reshape((t_len - 1) * batch, hdim)
                edge_gg = grad_gates_all[0]
            grad_whh = torch.mm(bulk_gg.t(), bulk_h)
            grad_whh = torch.addmm(grad_whh, edge_gg.t(), h0.to(dt))
        else:
            grad_whh = torch.mm(gg2.t(), h0.to(dt))
        grad_bias = gg2.sum(dim=0)
        grad_xg = grad_gates_all
        grad_h0 = grad_h
        grad_c0 = grad_c
        return grad_xg, grad_h0, grad_c0, grad_whh, grad_bias, None, None, None


def lstm_sequence(
    xg: torch.Tensor,  # [T, B, 4H]
    h0: torch.Tensor,
    c0: torch.Tensor,
    w_hh: torch.Tensor,
    bias: torch.Tensor,
    mask: Optional[torch.Tensor] = None,  # [T, B]
    reverse: bool = False,
) -> torch.Tensor:
    m = mask.detach().contiguous().to(torch.float32) if mask is not None else None
    infer = not torch.is_grad_enabled() or not (
        xg.requires_grad or h0.requires_grad or c0.requires_grad
        or w_hh.requires_grad or bias.requires_grad
    )
    return _LSTMSeqFn.apply(xg.contiguous(), h0, c0, w_hh, bias, m, reverse, infer)
