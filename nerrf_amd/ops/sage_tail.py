"""Fused GraphSAGE training-layer tail: h + LN(dropout(GELU(zs+zn)))*g+b.

One HIP kernel forward + one backward (ops/hip/sage_ln_act.hip) replaces
the ~5 forward / ~7 backward eager launches per layer; the dropout mask is
a counter-based RNG replayed in backward, so it never exists in memory.
Falls back to eager off-GPU / off-shape.
"""
from __future__ import annotations

import torch

from .native import get_native


class _SageTailFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, h, zs, zn, gamma, beta, drop_p: float, seed: int):
        ext = get_native(h)
        y, s_save, stats, _ = ext.sage_ln_act_fwd(
            h.contiguous(), zs.contiguous(), zn.contiguous(), gamma, beta,
            float(drop_p), int(seed), False,
        )
        ctx.save_for_backward(s_save, stats, gamma)
        ctx.drop_p = float(drop_p)
        ctx.seed = int(seed)
        return y

    @staticmethod
    def backward(ctx, dy):
        s_save, stats, gamma = ctx.saved_tensors
        ext = get_native(dy)
        dz, dgamma, dbeta = ext.sage_ln_act_bwd(
            dy.contiguous(), s_save, stats, gamma, ctx.drop_p, ctx.seed
        )
        # residual: dh = dy; both GEMM branches share dz
        return dy, dz, dz, dgamma.to(gamma.dtype), dbeta.to(gamma.dtype), None, None


_seed_counter = [12345]


def sage_layer_tail(h, zs, zn, gamma, beta, drop_p: float, training: bool):
    """Fused tail when on GPU bf16 D=128; eager otherwise."""
    import torch.nn.functional as F

    use = (
        h.is_cuda
        and h.dtype == torch.bfloat16
        and h.shape[-1] == 128
        and zs.shape == h.shape
        and zn.shape == h.shape
    )
    p = drop_p if training else 0.0
    if use:
        _seed_counter[0] = (_seed_counter[0] * 6364136223846793005 + 1) & 0xFFFFFFFF
        return _SageTailFn.apply(h, zs, zn, gamma, beta, p, _seed_counter[0])
    z = F.gelu(zs + zn)
    if p > 0:
        z = F.dropout(z, p)
    return h + F.layer_norm(z, (h.shape[-1],), gamma, beta, 1e-5)
