"""Interleaved A/B: streaming projection GEMMs (stream_gemm.hip) vs
hipBLASLt at production shapes (M = T*B ~ 6.4M, K=512, N=1024, dual).

Usage: python tools/proj_ab.py [M]
"""
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from nerrf_amd.ops.native import load_extension

ext = load_extension(required=True)
dev = "cuda:0"
M = int(sys.argv[1]) if len(sys.argv) > 1 else 6_414_000
torch.manual_seed(0)
x = (torch.randn(M, 512, device=dev) * 0.3).to(torch.bfloat16)
w1 = (torch.randn(1024, 512, device=dev) * 0.05).to(torch.bfloat16)
w2 = (torch.randn(1024, 512, device=dev) * 0.05).to(torch.bfloat16)
w1t = w1.t().contiguous(); w2t = w2.t().contiguous()
c1 = torch.empty(M, 1024, device=dev, dtype=torch.bfloat16)
c2 = torch.empty_like(c1)
g1 = (torch.randn(M, 1024, device=dev) * 0.2).to(torch.bfloat16)
g2 = (torch.randn(M, 1024, device=dev) * 0.2).to(torch.bfloat16)
dx = torch.empty(M, 512, device=dev, dtype=torch.bfloat16)


def fwd_blas():
    torch.mm(x, w1.t(), out=c1)
    torch.mm(x, w2.t(), out=c2)


def fwd_stream():
    ext.proj_fwd_dual(x, w1, w2, c1, c2)


def dgrad_blas():
    torch.mm(g1, w1, out=dx)
    dx.addmm_(g2, w2)


def dgrad_stream():
    ext.proj_dgrad_dual(g1, g2, w1t, w2t, dx)


dwf32 = torch.empty(1024, 512, device=dev, dtype=torch.bfloat16)


def wgrad_blas():
    global dwf32
    a = torch.mm(g1.t(), x)
    b = torch.mm(g2.t(), x)
    dwf32 = a


def wgrad_stream():
    ext.proj_wgrad(g1, g2, x, max(16, min(256, M // 8192)))


def tm(fn, n=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / n * 1e3


for rnd in range(3):
    fb = tm(fwd_blas); fs = tm(fwd_stream)
    db = tm(dgrad_blas); ds = tm(dgrad_stream)
    wb = tm(wgrad_blas, n=10); ws = tm(wgrad_stream, n=10)
    print(f"  wgrad blas {wb:.2f} ms  stream {ws:.2f} ms ({wb/ws:.2f}x)")
    # effective bytes: fwd reads A once + writes 2C (+W); dgrad reads 2A + writes C
    fwd_gb = (M * 512 * 2 + 2 * M * 1024 * 2) / 1e9
    dg_gb = (2 * M * 1024 * 2 + M * 512 * 2) / 1e9
    print(f"round {rnd}: fwd blas {fb:.2f} ms  stream {fs:.2f} ms ({fb/fs:.2f}x, "
          f"{fwd_gb/fs*1e3:.2f} TB/s alg)   dgrad blas {db:.2f} ms  stream {ds:.2f} ms "
          f"({db/ds:.2f}x, {dg_gb/ds*1e3:.2f} TB/s alg)")

# correctness spot check on a slice
fwd_stream()
r1 = x[:4096].float() @ w1.float().t()
print("fwd max err:", (c1[:4096].float() - r1).abs().max().item())
dgrad_stream()
rref = g1[:4096].float() @ w1.float() + g2[:4096].float() @ w2.float()
print("dgrad max err:", (dx[:4096].float() - rref).abs().max().item())
