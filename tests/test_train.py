"""Training loop / config / checkpoint tests (CPU, small)."""
import json

import torch

from nerrf_amd.checkpoint import load_checkpoint, save_checkpoint
from nerrf_amd.config import load_config
from nerrf_amd.models.joint import JointConfig, NerrfJointModel
from nerrf_amd.models.graphsage import SageConfig
from nerrf_amd.models.lstm import LSTMConfig
from nerrf_amd.train import run_toy, run_training


def small_cfg(tmp_path, epochs=1):
    cfg = load_config(
        None,
        [
            "data.n_scenarios=2",
            "data.benign_rate_hz=120",
            "data.duration_s=60",
            "optim.epochs=%d" % epochs,
            "run.eval_holdout=2",
            "run.log_every=1000",
            f"run.checkpoint_dir={tmp_path}/ckpt",
            "model.sage.layers=4",
            "model.sage.hidden=32",
            "model.lstm.hidden=32",
        ],
    )
    return cfg


def test_config_overrides(tmp_path):
    cfg = small_cfg(tmp_path)
    assert cfg.data.n_scenarios == 2
    assert cfg.model.sage.layers == 4
    assert cfg.optim.epochs == 1


def test_config_yaml(tmp_path):
    p = tmp_path / "c.yaml"
    p.write_text("optim:\n  lr: 0.01\n  epochs: 7\nmodel:\n  sage:\n    hidden: 48\n")
    cfg = load_config(p, ["optim.epochs=2"])
    assert cfg.optim.lr == 0.01
    assert cfg.optim.epochs == 2  # override wins
    assert cfg.model.sage.hidden == 48


def test_run_toy(capsys):
    out = run_toy("datasets/traces/toy_trace.csv")
    assert out["nodes"] > 0 and out["edges"] > 0
    printed = capsys.readouterr().out
    assert json.loads(printed.strip().splitlines()[-1])["mode"] == "toy"


def test_training_improves_and_reports(tmp_path):
    cfg = small_cfg(tmp_path, epochs=2)
    report = run_training(cfg)
    assert "node_auc" in report
    assert report["node_auc"] > 0.7  # detects the synthetic attack well above chance
    # checkpoint written
    assert (tmp_path / "ckpt" / "checkpoint.json").exists()


def test_checkpoint_roundtrip(tmp_path):
    cfg = JointConfig(sage=SageConfig(layers=2, hidden=32), lstm=LSTMConfig(hidden=16))
    m = NerrfJointModel(cfg)
    opt = torch.optim.AdamW(m.parameters(), lr=1e-3)
    # take one step so optimizer has state
    loss = sum((p * p).sum() for p in m.parameters())
    loss.backward()
    opt.step()
    save_checkpoint(tmp_path / "ck", m, opt, step=5, epoch=1, metrics={"auc": 0.99}, config=cfg)

    m2 = NerrfJointModel(cfg)
    opt2 = torch.optim.AdamW(m2.parameters(), lr=1e-3)
    manifest, _ = load_checkpoint(tmp_path / "ck", m2, opt2)
    assert manifest["step"] == 5
    assert manifest["metrics"]["auc"] == 0.99
    for (k1, p1), (k2, p2) in zip(m.state_dict().items(), m2.state_dict().items()):
        assert k1 == k2
        assert torch.equal(p1, p2)
    # optimizer state restored
    assert len(opt2.state_dict()["state"]) == len(opt.state_dict()["state"])


def test_checkpoint_resume_training(tmp_path):
    cfg = small_cfg(tmp_path, epochs=1)
    run_training(cfg)
    # resume for one more epoch from the saved checkpoint
    cfg2 = small_cfg(tmp_path, epochs=2)
    report = run_training(cfg2, resume=str(tmp_path / "ckpt"))
    assert "node_auc" in report


def test_pretrained_checkpoint_loads_and_detects():
    """The vendored GPU-trained checkpoint loads and detects an attack."""
    from nerrf_amd.data.synth import SynthConfig, generate
    from nerrf_amd.serve.engine import StreamingEngine, load_model_from_checkpoint

    model = load_model_from_checkpoint("checkpoints/pretrained")
    assert model.num_parameters() > 2_000_000  # joint (GNN 0.95M + LSTM 2.1M)
    engine = StreamingEngine(model=model, device="cpu", alarm_threshold=0.7)
    engine.store.window_s = 1e9
    arr, _ = generate(SynthConfig(seed=77, duration_s=40, benign_rate_hz=60, n_victim_files=10))
    engine.ingest_events(arr)
    det = engine.score_window()
    assert det.alarm
    # the trained model itself scores victim files high (not just indicators)
    hot = [p for p, s in det.file_scores.items() if s > 0.5 and "/app/uploads/" in p]
    assert len(hot) >= 10


def test_hparam_random_search(tmp_path):
    """tune.random_search: deterministic sampling, per-trial checkpoints,
    best-by-composite selection, resumable search.json artifact."""
    import json

    from nerrf_amd.tune import DEFAULT_SPACE, random_search, sample_trial

    # deterministic sampling: same (seed, trial) -> same overrides
    a = sample_trial(DEFAULT_SPACE, seed=7, trial=3)
    b = sample_trial(DEFAULT_SPACE, seed=7, trial=3)
    assert a == b
    assert set(a) == set(DEFAULT_SPACE)
    assert a != sample_trial(DEFAULT_SPACE, seed=7, trial=4)
    # ranged keys sample inside their bounds
    assert 2e-4 <= float(a["optim.lr"]) <= 5e-3

    res = random_search(
        n_trials=2,
        space={"optim.lr": (1e-3, 3e-3), "model.pos_weight": [2.0, 4.0]},
        base_overrides=[
            "optim.epochs=1", "data.n_scenarios=2", "data.duration_s=40",
            "data.benign_rate_hz=120", "model.sage.layers=2",
            "model.sage.hidden=32", "model.lstm.hidden=32",
            "run.eval_holdout=1", "run.log_every=1000",
        ],
        out_dir=str(tmp_path / "hs"),
        seed=1,
    )
    assert len(res.trials) == 2
    best = res.best
    assert best.score == max(t.score for t in res.trials)
    assert (tmp_path / "hs" / "trial_00").is_dir()
    summary = json.loads((tmp_path / "hs" / "search.json").read_text())
    assert summary["best"] == best.trial
    assert len(summary["trials"]) == 2
    assert "seq_f1" in res.trials[0].metrics or "node_auc" in res.trials[0].metrics
