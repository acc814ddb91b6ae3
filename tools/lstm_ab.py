import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from nerrf_amd.ops.native import load_extension

ext = load_extension(required=True)
dev = "cuda:0"
B, H = (int(sys.argv[1]) if len(sys.argv) > 1 else 16000), 256
torch.manual_seed(0)
h = (torch.randn(B, H, device=dev) * 0.3).to(torch.bfloat16)
c = (torch.randn(B, H, device=dev) * 0.3).to(torch.bfloat16)
w = (torch.randn(4 * H, H, device=dev) * 0.1).to(torch.bfloat16)
xg = (torch.randn(B, 4 * H, device=dev) * 0.2).to(torch.bfloat16)
bias = torch.randn(4 * H, device=dev).to(torch.bfloat16)
mask = torch.empty(0, device=dev)
w_t = w.t().contiguous()
w_tiled = w.reshape(4 * H, H // 32, 32).permute(1, 0, 2).contiguous()
h_out = torch.empty_like(c); c_out = torch.empty_like(c); gates = torch.empty_like(xg)
hg = torch.empty(B, 4 * H, device=dev, dtype=torch.bfloat16)

def unfused():
    torch.mm(h, w_t, out=hg)
    ext.lstm_pointwise_fwd(hg, xg, bias, c, h, mask, h_out, c_out, gates)

def fused():
    ext.lstm_step_fused(h, w_tiled, xg, bias, c, mask, h_out, c_out, gates, False)

def tm(fn, n=300):
    for _ in range(20): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / n * 1e6

# interleaved A/B
for rnd in range(3):
    ua = tm(unfused); fa = tm(fused)
    print(f"round {rnd}: unfused {ua:.1f} us   fused {fa:.1f} us   ratio {ua/fa:.2f}x")
# correctness spot check
fused()
hf = h_out.clone()
unfused()
import math
gp = (h.float() @ w.float().t() + xg.float() + bias.float())
i = torch.sigmoid(gp[:, :H]); f_ = torch.sigmoid(gp[:, H:2*H])
g = torch.tanh(gp[:, 2*H:3*H]); o = torch.sigmoid(gp[:, 3*H:])
cn = f_ * c.float() + i * g
href = o * torch.tanh(cn)
err = (hf.float() - href).abs().max().item()
print("fused max err vs fp32 ref:", err)
