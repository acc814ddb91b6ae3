"""Interleaved A/B: fused recurrent step (lstm_rec_fused.hip) vs the split
GEMM+pointwise path, fwd and bwd, at production shapes (B=64k, H=256).

Usage: python tools/rec_ab.py [B]
"""
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from nerrf_amd.ops.native import load_extension

ext = load_extension(required=True)
dev = "cuda:0"
B, H = (int(sys.argv[1]) if len(sys.argv) > 1 else 64140), 256
G = 4 * H
torch.manual_seed(0)
h = (torch.randn(B, H, device=dev) * 0.3).to(torch.bfloat16)
c = (torch.randn(B, H, device=dev) * 0.3).to(torch.bfloat16)
w = (torch.randn(G, H, device=dev) * 0.1).to(torch.bfloat16)
xg = (torch.randn(B, G, device=dev) * 0.2).to(torch.bfloat16)
bias = torch.randn(G, device=dev).to(torch.bfloat16)
mask = (torch.rand(B, device=dev) > 0.1).float()
empty = torch.empty(0, device=dev)
w_t = w.t().contiguous()
h_out = torch.empty_like(c); c_out = torch.empty_like(c); gates = torch.empty_like(xg)
hg = torch.empty(B, G, device=dev, dtype=torch.bfloat16)

# bwd buffers
gh = (torch.randn(B, H, device=dev) * 0.2).to(torch.bfloat16)
gout = (torch.randn(B, H, device=dev) * 0.2).to(torch.bfloat16)
gc = (torch.randn(B, H, device=dev) * 0.2).to(torch.bfloat16)
gacts = torch.sigmoid(torch.randn(B, G, device=dev)).to(torch.bfloat16)
gg = torch.empty(B, G, device=dev, dtype=torch.bfloat16)
gcp = torch.empty_like(c); ghp = torch.empty_like(c); gh_out = torch.empty_like(c)


def fwd_split():
    torch.mm(h, w_t, out=hg)
    ext.lstm_pointwise_fwd(hg, xg, bias, c, h, mask, h_out, c_out, gates)


def fwd_fused():
    ext.lstm_rec_fwd(h, w, xg, bias, c, mask, h_out, c_out, gates)


def bwd_split():
    ext.lstm_pointwise_bwd(gh, gout, gc, gacts, c, mask, gg, gcp, ghp, torch.empty(0, device=dev))
    torch.addmm(ghp, gg, w, out=gh_out)


def bwd_fused():
    ext.lstm_rec_bwd(gh, gout, gc, gacts, c, w_t, mask, gg, gcp, gh_out)


def tm(fn, n=200):
    for _ in range(20):
        fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / n * 1e6


for rnd in range(3):
    fs = tm(fwd_split); ff = tm(fwd_fused)
    bs = tm(bwd_split); bf = tm(bwd_fused)
    print(f"round {rnd}: fwd split {fs:.1f} us  fused {ff:.1f} us  ({fs/ff:.2f}x)   "
          f"bwd split {bs:.1f} us  fused {bf:.1f} us  ({bs/bf:.2f}x)")

# correctness spot checks vs fp32
fwd_fused()
hf = h_out.float().clone(); cf = c_out.float().clone()
gp = h.float() @ w.float().t() + xg.float() + bias.float()
i = torch.sigmoid(gp[:, :H]); f_ = torch.sigmoid(gp[:, H:2 * H])
g_ = torch.tanh(gp[:, 2 * H:3 * H]); o = torch.sigmoid(gp[:, 3 * H:])
cn = f_ * c.float() + i * g_
hn = o * torch.tanh(cn)
m = mask.unsqueeze(1)
cn = m * cn + (1 - m) * c.float(); hn = m * hn + (1 - m) * h.float()
print("fwd max err h:", (hf - hn).abs().max().item(),
      " c:", (cf - cn).abs().max().item())

bwd_fused()
ggf = gg.float().clone(); gcpf = gcp.float().clone(); ghof = gh_out.float().clone()
bwd_split()
print("bwd max diff gg:", (ggf - gg.float()).abs().max().item(),
      " gcp:", (gcpf - gcp.float()).abs().max().item(),
      " gh_out:", (ghof - gh_out.float()).abs().max().item())
