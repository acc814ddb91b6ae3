import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time, torch
from bench import build_bench_batches
from nerrf_amd.models.joint import NerrfJointModel, JointConfig
from nerrf_amd.models.graphsage import SageConfig
from nerrf_amd.models.lstm import LSTMConfig

dev = torch.device("cuda:0")
b_np = build_bench_batches(0, 1, "full")[0]
b = b_np.to_torch(device=dev, dtype=torch.bfloat16)
model = NerrfJointModel(JointConfig()).to(device=dev, dtype=torch.bfloat16)
opt = torch.optim.AdamW(model.parameters(), lr=1e-3, foreach=True)

def tm(fn, n=5):
    for _ in range(2): fn()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/n*1000

def gnn_fwd():
    model.gnn(b["x"], b["nbr_idx"], b["nbr_w"], b["edge_index"], b["edge_weight"], b["edge_ts"])
def lstm_fwd():
    model.lstm(b["seq_feats"], b["seq_lengths"])
def gnn_fwdbwd():
    nl, el = model.gnn(b["x"], b["nbr_idx"], b["nbr_w"], b["edge_index"], b["edge_weight"], b["edge_ts"])
    (nl.float().sum()+el.float().sum()).backward()
    model.zero_grad(set_to_none=True)
def lstm_fwdbwd():
    sl = model.lstm(b["seq_feats"], b["seq_lengths"])
    sl.float().sum().backward()
    model.zero_grad(set_to_none=True)
def full_step():
    nl, el, sl = model(b)
    losses = model.loss(nl, el, sl, b)
    opt.zero_grad(set_to_none=False)
    losses["total"].backward()
    opt.step()

print("gnn_fwd ms", tm(gnn_fwd))
print("lstm_fwd ms", tm(lstm_fwd))
print("gnn_fwdbwd ms", tm(gnn_fwdbwd))
print("lstm_fwdbwd ms", tm(lstm_fwdbwd))
print("full_step ms", tm(full_step))
