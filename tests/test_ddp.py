"""Multi-process data-parallel tests (gloo backend, world_size=2, CPU)."""
import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from nerrf_amd.eval import roc_auc


def _find_free_port():
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _worker_allreduce(rank, world, port, results):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from nerrf_amd.parallel.ddp import GradAllReducer

        torch.manual_seed(100 + rank)  # different data per rank
        model = torch.nn.Sequential(
            torch.nn.Linear(8, 16), torch.nn.GELU(), torch.nn.Linear(16, 1)
        )
        # identical init across ranks
        torch.manual_seed(7)
        for p in model.parameters():
            torch.nn.init.normal_(p)
        reducer = GradAllReducer(model, bucket_bytes=128)  # force several buckets
        x = torch.randn(4, 8) * (rank + 1)
        y = model(x).sum()
        y.backward()
        reducer.finalize()
        grads = torch.cat([p.grad.flatten() for p in model.parameters()])
        results[rank] = grads.numpy()
    finally:
        dist.destroy_process_group()


def test_grad_allreduce_averages_across_ranks():
    port = _find_free_port()
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_worker_allreduce, args=(2, port, results), nprocs=2, join=True)
    g0, g1 = results[0], results[1]
    # after all-reduce both ranks hold identical averaged grads
    assert np.allclose(g0, g1, atol=1e-6)


def _worker_expected_avg(rank, world, port, results):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from nerrf_amd.parallel.ddp import GradAllReducer

        model = torch.nn.Linear(4, 1, bias=False)
        with torch.no_grad():
            model.weight.fill_(1.0)
        reducer = GradAllReducer(model)
        x = torch.full((1, 4), float(rank + 1))  # rank0 grads=1, rank1 grads=2
        model(x).sum().backward()
        reducer.finalize()
        results[rank] = model.weight.grad.numpy().copy()
    finally:
        dist.destroy_process_group()


def test_grad_allreduce_value():
    port = _find_free_port()
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_worker_expected_avg, args=(2, port, results), nprocs=2, join=True)
    # average of grad 1 and grad 2 = 1.5
    assert np.allclose(results[0], 1.5, atol=1e-6)
    assert np.allclose(results[1], 1.5, atol=1e-6)


def test_roc_auc_against_sklearn():
    try:
        from sklearn.metrics import roc_auc_score
    except ImportError:
        pytest.skip("sklearn unavailable")
    rng = np.random.default_rng(0)
    y = rng.integers(0, 2, 500)
    s = rng.random(500) * 0.5 + y * rng.random(500) * 0.5
    assert abs(roc_auc(y, s) - roc_auc_score(y, s)) < 1e-9
    # with heavy ties
    s_t = np.round(s, 1)
    assert abs(roc_auc(y, s_t) - roc_auc_score(y, s_t)) < 1e-9


def _worker_unused_head(rank, world, port, results):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from nerrf_amd.parallel.ddp import GradAllReducer

        class TwoHeads(torch.nn.Module):
            def __init__(self):
                super().__init__()
                self.trunk = torch.nn.Linear(4, 8)
                self.used = torch.nn.Linear(8, 1)
                self.unused = torch.nn.Linear(8, 1)

            def forward(self, x):
                return self.used(self.trunk(x))

        torch.manual_seed(3)
        model = TwoHeads()
        reducer = GradAllReducer(model, bucket_bytes=64)  # many buckets
        x = torch.full((2, 4), float(rank + 1))
        model(x).sum().backward()
        # the unused head's bucket hooks never fire; finalize must still
        # all-reduce it (as zeros) without deadlocking
        reducer.finalize()
        results[rank] = {
            "used": model.used.weight.grad.numpy().copy(),
            "unused": model.unused.weight.grad.numpy().copy(),
        }
    finally:
        dist.destroy_process_group()


def test_grad_allreduce_unused_head_no_deadlock():
    port = _find_free_port()
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_worker_unused_head, args=(2, port, results), nprocs=2, join=True)
    assert np.allclose(results[0]["used"], results[1]["used"], atol=1e-6)
    assert np.allclose(results[0]["unused"], 0.0)
    assert np.allclose(results[1]["unused"], 0.0)


def test_roc_auc_degenerate_classes():
    import math

    assert math.isnan(roc_auc(np.zeros(5), np.random.rand(5)))
    assert math.isnan(roc_auc(np.ones(5), np.random.rand(5)))


def test_precision_recall_edge_cases():
    from nerrf_amd.eval import best_f1, precision_recall_f1

    y = np.array([0, 0, 1, 1])
    s = np.array([0.1, 0.2, 0.8, 0.9])
    m = precision_recall_f1(y, s, 0.5)
    assert m["precision"] == 1.0 and m["recall"] == 1.0 and m["f1"] == 1.0
    # no predictions above threshold
    m = precision_recall_f1(y, s, 0.95)
    assert m["precision"] == 0.0 and m["recall"] == 0.0 and m["f1"] == 0.0
    b = best_f1(y, s)
    assert b["f1"] == 1.0


def _worker_divergent_head(rank, world, port, results):
    """Ranks use DIFFERENT heads => grad-arrival order differs across ranks.

    Launch order must still be identical (fixed bucket-index order) or the
    collectives mismatch/hang (ADVICE r1 finding on GradAllReducer).
    """
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from nerrf_amd.parallel.ddp import GradAllReducer

        class TwoHeads(torch.nn.Module):
            def __init__(self):
                super().__init__()
                self.trunk = torch.nn.Linear(4, 8)
                self.head_a = torch.nn.Linear(8, 1)
                self.head_b = torch.nn.Linear(8, 1)

        torch.manual_seed(11)
        model = TwoHeads()
        reducer = GradAllReducer(model, bucket_bytes=64)  # ~1 param per bucket
        launch_order = []
        orig_launch = reducer._launch

        def recording_launch(bi):
            launch_order.append(bi)
            orig_launch(bi)

        reducer._launch = recording_launch
        x = torch.full((2, 4), float(rank + 1))
        h = model.trunk(x)
        if rank == 0:
            # both heads -> every bucket fills during backward
            loss = model.head_a(h).sum() + model.head_b(h).sum()
        else:
            # head_b unused -> its buckets only fill at finalize()
            loss = model.head_a(h).sum()
        loss.backward()
        reducer.finalize()
        results[rank] = {
            "order": list(launch_order),
            "b_w": model.head_b.weight.grad.numpy().copy(),
            "a_w": model.head_a.weight.grad.numpy().copy(),
            "trunk_w": model.trunk.weight.grad.numpy().copy(),
        }
    finally:
        dist.destroy_process_group()


def test_grad_allreduce_divergent_head_fixed_order():
    port = _find_free_port()
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_worker_divergent_head, args=(2, port, results), nprocs=2, join=True)
    r0, r1 = results[0], results[1]
    # identical launch sequence on both ranks, strictly index-ordered
    assert r0["order"] == r1["order"] == sorted(r0["order"])
    # both ranks converge to the same averaged grads
    assert np.allclose(r0["b_w"], r1["b_w"], atol=1e-6)
    assert np.allclose(r0["a_w"], r1["a_w"], atol=1e-6)
    assert np.allclose(r0["trunk_w"], r1["trunk_w"], atol=1e-6)
    # head_b grads: rank0's contribution halved (rank1 contributed zeros)
    assert np.abs(r0["b_w"]).max() > 0.0


def _worker_overlap(rank, world, port, results):
    """Bucket reduces must be ISSUED during backward (overlap), not batched
    at finalize: at least one _launch has to happen before finalize() runs
    when every bucket fills."""
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from nerrf_amd.parallel.ddp import GradAllReducer

        torch.manual_seed(5)
        model = torch.nn.Sequential(
            torch.nn.Linear(8, 32), torch.nn.GELU(), torch.nn.Linear(32, 1)
        )
        reducer = GradAllReducer(model, bucket_bytes=256)
        n_buckets = len(reducer.buckets)
        events = []
        orig_launch = reducer._launch

        def recording_launch(bi):
            events.append(("launch", bi))
            orig_launch(bi)

        reducer._launch = recording_launch
        orig_finalize = reducer.finalize

        def recording_finalize():
            events.append(("finalize", -1))
            orig_finalize()

        model(torch.randn(4, 8)).sum().backward()
        recording_finalize()
        launches_before_finalize = [
            e for e in events[: events.index(("finalize", -1))] if e[0] == "launch"
        ]
        results[rank] = {
            "n_buckets": n_buckets,
            "overlapped": len(launches_before_finalize),
        }
    finally:
        dist.destroy_process_group()


def test_grad_allreduce_overlaps_backward():
    port = _find_free_port()
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_worker_overlap, args=(2, port, results), nprocs=2, join=True)
    for r in (results[0], results[1]):
        assert r["n_buckets"] >= 2
        # every bucket fills during backward, so every reduce is issued
        # before finalize — that IS the overlap contract
        assert r["overlapped"] == r["n_buckets"]
