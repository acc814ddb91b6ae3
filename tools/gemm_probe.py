import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

dev = "cuda:0"
def tm(fn, n=10):
    for _ in range(3): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / n * 1e3

T, B = 100, 16000
M = T * B
shapes = [
    ("xproj-l2 [1.6M,512]@[512,1024]", M, 512, 1024),
    ("wgrad-l2 [1024,1.6M]@[1.6M,512]T", None, None, None),
    ("xproj-l1 [1.6M,16]@[16,1024]", M, 16, 1024),
]
a = torch.randn(M, 512, device=dev, dtype=torch.bfloat16)
w = torch.randn(1024, 512, device=dev, dtype=torch.bfloat16)
gg = torch.randn(M, 1024, device=dev, dtype=torch.bfloat16)

def io_tbps(bytes_, ms): return bytes_ / (ms * 1e-3) / 1e12

# baseline mm
t1 = tm(lambda: torch.mm(a, w.t()))
b1 = (M*512 + 512*1024 + M*1024) * 2
print(f"xproj mm: {t1:.2f} ms  {io_tbps(b1, t1):.2f} TB/s")
# wgrad: gg^T @ a  -> [1024, 512]
t2 = tm(lambda: torch.mm(gg.t(), a))
b2 = (M*1024 + M*512 + 1024*512) * 2
print(f"wgrad mm: {t2:.2f} ms  {io_tbps(b2, t2):.2f} TB/s")
# with tunableop vendored table
from nerrf_amd.perf import enable_tuned_gemms
enable_tuned_gemms()
t1b = tm(lambda: torch.mm(a, w.t()))
t2b = tm(lambda: torch.mm(gg.t(), a))
print(f"xproj tuned: {t1b:.2f} ms  {io_tbps(b1, t1b):.2f} TB/s")
print(f"wgrad tuned: {t2b:.2f} ms  {io_tbps(b2, t2b):.2f} TB/s")
# live tuning on these shapes
torch.cuda.tunable.tuning_enable(True)
torch.mm(a, w.t()); torch.mm(gg.t(), a)
torch.cuda.synchronize()
torch.cuda.tunable.tuning_enable(False)
t1c = tm(lambda: torch.mm(a, w.t()))
t2c = tm(lambda: torch.mm(gg.t(), a))
print(f"xproj live-tuned: {t1c:.2f} ms  {io_tbps(b1, t1c):.2f} TB/s")
print(f"wgrad live-tuned: {t2c:.2f} ms  {io_tbps(b2, t2c):.2f} TB/s")
# reference: pure HBM copy speed on same volume
src = torch.randn(M*1024//2, device=dev, dtype=torch.float32)
dst = torch.empty_like(src)
t3 = tm(lambda: dst.copy_(src))
print(f"copy 3.3GB: {t3:.2f} ms  {io_tbps(src.numel()*8, t3):.2f} TB/s")
# fp32 output variant (wgrad in fp32 accumulate out)
wf = torch.randn(512, 1024, device=dev, dtype=torch.bfloat16)
t4 = tm(lambda: torch.mm(a, wf))
print(f"xproj NN-layout: {t4:.2f} ms")
