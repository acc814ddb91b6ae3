"""Per-shape timing of every projection GEMM in the training step vs its
traffic roofline (M = T*B at production scale). Run on the GPU box."""
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from nerrf_amd.perf import enable_tuned_gemms

enable_tuned_gemms()
dev = "cuda:0"
M = int(sys.argv[1]) if len(sys.argv) > 1 else 6_414_000
dt = torch.bfloat16


def tm(fn, n=8):
    for _ in range(2):
        fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / n * 1e3


def gemm_case(name, m, k, n, ta=False):
    a = (torch.randn(k, m, device=dev) * 0.1).to(dt).t() if ta else (torch.randn(m, k, device=dev) * 0.1).to(dt)
    b = (torch.randn(k, n, device=dev) * 0.1).to(dt)
    out = torch.empty(m, n, device=dev, dtype=dt)

    def run():
        torch.mm(a, b, out=out)

    ms = tm(run)
    gb = (m * k + k * n + m * n) * 2 / 1e9
    print(f"{name:28s} [{m:>9},{k:>5}]x[{k:>5},{n:>5}] {ms:8.2f} ms  {gb/ms*1e3:7.2f} TB/s eff  (traffic {gb:.1f} GB)")
    del a, b, out
    torch.cuda.empty_cache()


# forward projections (cat weights)
gemm_case("L2 proj fwd", M, 512, 2048)
gemm_case("L1 proj fwd", M, 16, 2048)
# dgrad
gemm_case("L2 proj dgrad", M, 2048, 512)
# wgrads: dW = g^T @ x  -> mm(g.t(), x): [2048, M] x [M, K]
gemm_case("L2 proj wgrad", 2048, M, 512, ta=True)
gemm_case("L1 proj wgrad", 2048, M, 16, ta=True)
# recurrent shapes
B = 64140
gemm_case("rec fwd h@Whh^T", B, 256, 1024)
gemm_case("rec bwd gg@Whh", B, 1024, 256)
# in-layer W_hh wgrad (bulk)
gemm_case("Whh wgrad", 1024, (B * 99), 256, ta=True)
