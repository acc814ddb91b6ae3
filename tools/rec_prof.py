"""Run ONE rec kernel repeatedly for clean rocprofv3 capture.
Usage: python tools/rec_prof.py {fwd|bwd|fwd_split|bwd_split|proj|projblas} [B] [iters]
"""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from nerrf_amd.ops.native import load_extension

ext = load_extension(required=True)
dev = "cuda:0"
which = sys.argv[1]
B = int(sys.argv[2]) if len(sys.argv) > 2 else 64140
iters = int(sys.argv[3]) if len(sys.argv) > 3 else 50
H = 256; G = 1024
torch.manual_seed(0)
h = (torch.randn(B, H, device=dev) * 0.3).to(torch.bfloat16)
c = (torch.randn(B, H, device=dev) * 0.3).to(torch.bfloat16)
w = (torch.randn(G, H, device=dev) * 0.1).to(torch.bfloat16)
xgt = (torch.randn(B, G, device=dev) * 0.2).to(torch.bfloat16)
bias = torch.randn(G, device=dev).to(torch.bfloat16)
mask = (torch.rand(B, device=dev) > 0.1).float()
w_t = w.t().contiguous()
h_out = torch.empty_like(c); c_out = torch.empty_like(c); gates = torch.empty_like(xgt)
hg = torch.empty(B, G, device=dev, dtype=torch.bfloat16)
gh = (torch.randn(B, H, device=dev) * 0.2).to(torch.bfloat16)
gout = (torch.randn(B, H, device=dev) * 0.2).to(torch.bfloat16)
gc = (torch.randn(B, H, device=dev) * 0.2).to(torch.bfloat16)
gacts = torch.sigmoid(torch.randn(B, G, device=dev)).to(torch.bfloat16)
gg = torch.empty(B, G, device=dev, dtype=torch.bfloat16)
gcp = torch.empty_like(c); gh_out = torch.empty_like(c); ghp = torch.empty_like(c)

if which == "proj":
    M = 1_600_000
    x = (torch.randn(M, 512, device=dev) * 0.3).to(torch.bfloat16)
    w1 = (torch.randn(1024, 512, device=dev) * 0.05).to(torch.bfloat16)
    w2 = (torch.randn(1024, 512, device=dev) * 0.05).to(torch.bfloat16)
    c1 = torch.empty(M, 1024, device=dev, dtype=torch.bfloat16); c2 = torch.empty_like(c1)

def run():
    if which == "fwd":
        ext.lstm_rec_fwd(h, w, xgt, bias, c, mask, h_out, c_out, gates)
    elif which == "bwd":
        ext.lstm_rec_bwd(gh, gout, gc, gacts, c, w_t, mask, gg, gcp, gh_out)
    elif which == "fwd_split":
        torch.mm(h, w_t, out=hg)
        ext.lstm_pointwise_fwd(hg, xgt, bias, c, h, mask, h_out, c_out, gates)
    elif which == "bwd_split":
        ext.lstm_pointwise_bwd(gh, gout, gc, gacts, c, mask, gg, gcp, ghp, torch.empty(0, device=dev))
        torch.addmm(ghp, gg, w, out=gh_out)
    elif which == "proj":
        ext.proj_fwd_dual(x, w1, w2, c1, c2)

for _ in range(5):
    run()
torch.cuda.synchronize()
for _ in range(iters):
    run()
torch.cuda.synchronize()
print("done", which)
