"""A/B: rec_gemm_fwd (rec_gemm.hip) vs tuned hipBLASLt on the recurrent
step shape hg[B,1024] = h[B,256] @ W_hh[1024,256]^T.

Run on a GPU box:  python tools/rec_gemm_ab.py
"""
import os
import sys
import time

import torch

_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, _ROOT)

from nerrf_amd.perf import enable_tuned_gemms  # noqa: E402
from nerrf_amd.ops.native import load_extension  # noqa: E402

enable_tuned_gemms()
ext = load_extension(required=True)
dev = "cuda:0"
torch.manual_seed(0)

for m in (16384, 65536):
    a = (torch.randn(m, 256, device=dev) * 0.1).to(torch.bfloat16)
    w = (torch.randn(1024, 256, device=dev) * 0.1).to(torch.bfloat16)
    wt = w.t().contiguous()
    c_ref = torch.empty(m, 1024, device=dev, dtype=torch.bfloat16)
    c_k = torch.empty_like(c_ref)

    # correctness vs fp32 reference
    ext.rec_gemm_fwd(a, w, c_k)
    ref32 = torch.matmul(a.float(), w.float().t())
    err = (c_k.float() - ref32).abs().max().item()
    rel = err / ref32.abs().max().item()
    print(f"M={m}: max abs err {err:.4e} (rel {rel:.2e})")
    assert rel < 2e-2, "numerics FAIL"

    # strided-A variant (column slab, row stride 512)
    slab = (torch.randn(m, 512, device=dev) * 0.1).to(torch.bfloat16)
    a_s = slab[:, 256:]
    ext.rec_gemm_fwd(a_s, w, c_k)
    ref32s = torch.matmul(a_s.float(), w.float().t())
    errs = (c_k.float() - ref32s).abs().max().item()
    print(f"M={m} strided: max abs err {errs:.4e}")
    assert errs / ref32s.abs().max().item() < 2e-2, "strided numerics FAIL"

    def bench(fn, n=200):
        for _ in range(20):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(n):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / n * 1e6

    t_blas = bench(lambda: torch.mm(a, wt, out=c_ref))
    t_k = bench(lambda: ext.rec_gemm_fwd(a, w, c_k))
    gb = (m * 256 * 2 + m * 1024 * 2 + 1024 * 256 * 2) / 1e9
    print(f"M={m}: blas {t_blas:.1f} us ({gb / (t_blas * 1e-6):.2f} TB/s eff)"
          f" | rec_gemm {t_k:.1f} us ({gb / (t_k * 1e-6):.2f} TB/s eff)"
          f" | speedup {t_blas / t_k:.2f}x")

# ---- dgrad: grad_h = gg @ W_hh (+ grad_h_pass) -----------------------------
for m in (16384, 65536):
    gg = (torch.randn(m, 1024, device=dev) * 0.1).to(torch.bfloat16)
    w = (torch.randn(1024, 256, device=dev) * 0.1).to(torch.bfloat16)
    wt = w.t().contiguous()
    dpass = (torch.randn(m, 256, device=dev) * 0.1).to(torch.bfloat16)
    out = torch.empty(m, 256, device=dev, dtype=torch.bfloat16)
    empty = torch.empty(0, device=dev, dtype=torch.bfloat16)

    ext.rec_gemm_dgrad(gg, wt, dpass, out)
    ref32 = torch.addmm(dpass.float(), gg.float(), w.float())
    err = (out.float() - ref32).abs().max().item()
    rel = err / ref32.abs().max().item()
    print(f"dgrad M={m}: max abs err {err:.4e} (rel {rel:.2e})")
    assert rel < 2e-2, "dgrad numerics FAIL"
    # no-addend + strided-A (gg slab, row stride 2048)
    slab = (torch.randn(m, 2048, device=dev) * 0.1).to(torch.bfloat16)
    gg_s = slab[:, :1024]
    ext.rec_gemm_dgrad(gg_s, wt, empty, out)
    refs = torch.matmul(gg_s.float(), w.float())
    errs = (out.float() - refs).abs().max().item()
    print(f"dgrad M={m} strided/no-d: max abs err {errs:.4e}")
    assert errs / refs.abs().max().item() < 2e-2, "dgrad strided FAIL"

    def bench2(fn, n=200):
        for _ in range(20):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(n):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / n * 1e6

    t_blas = bench2(lambda: torch.addmm(dpass, gg, w, out=out))
    t_k = bench2(lambda: ext.rec_gemm_dgrad(gg, wt, dpass, out))
    gb = (m * 1024 * 2 + 2 * m * 256 * 2 + 1024 * 256 * 2) / 1e9
    print(f"dgrad M={m}: blas(addmm) {t_blas:.1f} us ({gb / (t_blas * 1e-6) / 1e3:.2f} TB/s)"
          f" | rec_dgrad {t_k:.1f} us ({gb / (t_k * 1e-6) / 1e3:.2f} TB/s)"
          f" | speedup {t_blas / t_k:.2f}x")
