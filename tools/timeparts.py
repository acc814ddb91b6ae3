import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time, torch
from bench import build_bench_batches
from nerrf_amd.models.joint import NerrfJointModel, JointConfig

dev = torch.device("cuda:0")
b_np = build_bench_batches(0, 1, "full")[0]
b = b_np.to_torch(device=dev, dtype=torch.bfloat16)
model = NerrfJointModel(JointConfig()).to(device=dev, dtype=torch.bfloat16)
opt = torch.optim.AdamW(model.parameters(), lr=1e-3, foreach=True)

def tm(fn, n=5):
    for _ in range(2): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / n * 1000

state = {}
def fwd():
    state["out"] = model(b)
def loss_fn():
    nl, el, sl = state["out"]
    state["losses"] = model.loss(nl, el, sl, b)
def bwd():
    fwd(); loss_fn()
    model.zero_grad(set_to_none=True)
    state["losses"]["total"].backward()
def zg():
    opt.zero_grad(set_to_none=False)
def opt_step():
    opt.step()
def full_step():
    nl, el, sl = model(b)
    losses = model.loss(nl, el, sl, b)
    opt.zero_grad(set_to_none=False)
    losses["total"].backward()
    opt.step()

print("fwd ms", tm(fwd))
print("loss ms", tm(loss_fn))
print("fwd+loss+bwd ms", tm(bwd))
print("zero_grad ms", tm(zg))
print("opt_step ms", tm(opt_step))
print("full_step ms", tm(full_step))
