"""Edge-case and error-path coverage: config, checkpoint, rollback, eval."""
import json

import numpy as np
import pytest
import torch

from nerrf_amd.checkpoint import load_checkpoint, save_checkpoint
from nerrf_amd.config import load_config
from nerrf_amd.eval import RecoveryMetrics, best_f1, precision_recall_f1, roc_auc
from nerrf_amd.models.joint import JointConfig, NerrfJointModel
from nerrf_amd.models.graphsage import SageConfig
from nerrf_amd.models.lstm import LSTMConfig
from nerrf_amd.serve.rollback import execute_rollback


def test_config_unknown_key_rejected():
    with pytest.raises(KeyError):
        load_config(None, ["optim.does_not_exist=1"])
    with pytest.raises(KeyError):
        load_config(None, ["nope.lr=1"])


def test_config_yaml_unknown_key_rejected(tmp_path):
    p = tmp_path / "c.yaml"
    p.write_text("optim:\n  learning_rate: 0.1\n")  # wrong key name
    with pytest.raises(KeyError):
        load_config(p)


def test_config_bad_override_format():
    with pytest.raises(ValueError):
        load_config(None, ["optim.lr"])


def test_checkpoint_missing_dir_raises(tmp_path):
    m = NerrfJointModel(JointConfig(sage=SageConfig(layers=2, hidden=16), lstm=LSTMConfig(hidden=8)))
    with pytest.raises(FileNotFoundError):
        load_checkpoint(tmp_path / "nope", m)


def test_checkpoint_corrupt_manifest_raises(tmp_path):
    d = tmp_path / "ck"
    d.mkdir()
    (d / "checkpoint.json").write_text("{not json")
    m = NerrfJointModel(JointConfig(sage=SageConfig(layers=2, hidden=16), lstm=LSTMConfig(hidden=8)))
    with pytest.raises(json.JSONDecodeError):
        load_checkpoint(d, m)


def test_checkpoint_wrong_architecture_fails_loudly(tmp_path):
    small = NerrfJointModel(JointConfig(sage=SageConfig(layers=2, hidden=16), lstm=LSTMConfig(hidden=8)))
    save_checkpoint(tmp_path / "ck", small, step=1)
    big = NerrfJointModel(JointConfig(sage=SageConfig(layers=3, hidden=32), lstm=LSTMConfig(hidden=8)))
    with pytest.raises(RuntimeError):
        load_checkpoint(tmp_path / "ck", big)


def test_rollback_empty_directory(tmp_path):
    res = execute_rollback(tmp_path, validate_in_sandbox=False)
    assert res.files_restored == 0
    assert res.files_failed == 0


def test_rollback_sandbox_empty_dir_rejects(tmp_path):
    # nothing to restore -> the gate refuses (restored == 0)
    res = execute_rollback(tmp_path, validate_in_sandbox=True)
    assert res.files_restored == 0


def test_roc_auc_degenerate():
    assert np.isnan(roc_auc(np.zeros(10), np.random.rand(10)))
    assert np.isnan(roc_auc(np.ones(10), np.random.rand(10)))
    assert roc_auc([0, 1], [0.1, 0.9]) == 1.0
    assert roc_auc([1, 0], [0.1, 0.9]) == 0.0
    assert roc_auc([0, 1, 0, 1], [0.5, 0.5, 0.5, 0.5]) == 0.5  # all tied


def test_precision_recall_empty_predictions():
    m = precision_recall_f1(np.array([1, 1, 0]), np.array([0.1, 0.2, 0.3]), threshold=0.9)
    assert m["precision"] == 0.0 and m["recall"] == 0.0 and m["f1"] == 0.0


def test_best_f1_perfect_separation():
    y = np.array([0, 0, 1, 1])
    s = np.array([0.1, 0.2, 0.8, 0.9])
    best = best_f1(y, s)
    assert best["f1"] == 1.0
    assert 0.2 < best["threshold"] <= 0.8


def test_recovery_metrics():
    r = RecoveryMetrics(detect_ts=10.0, recover_ts=14.5, bytes_lost=0,
                        files_restored=9, files_total=10)
    d = r.as_dict()
    assert d["mttr_s"] == 4.5
    assert d["restore_rate"] == 0.9


def test_gather_mean_empty_fanout_cpu():
    from nerrf_amd.ops import gather_mean

    h = torch.randn(3, 4)
    idx = torch.zeros(3, 1, dtype=torch.int64)
    w = torch.zeros(3, 1)  # all-zero weights -> denom clamps, output ~0
    out = gather_mean(h, idx, w)
    assert torch.isfinite(out).all()


def test_build_edges_and_flags_matches_build_graph():
    from nerrf_amd.data.synth import SynthConfig, generate
    from nerrf_amd.graph.constructor import (
        build_edges_and_flags,
        build_graph,
        build_graph_parts,
    )

    arr, _ = generate(SynthConfig(seed=33, duration_s=40, benign_rate_hz=80))
    parts = build_graph_parts(arr)
    ed = build_edges_and_flags(parts)
    g = build_graph(arr, parts=parts)
    assert np.array_equal(ed["edge_index"], g.edge_index)
    assert np.allclose(ed["edge_weight"], g.edge_weight)
    assert np.allclose(np.log1p(ed["in_deg"]), g.x[:, 2])
    assert np.allclose(ed["suspicious"], g.x[:, 13])


def test_config_sequence_override_forms():
    """Sequence-typed overrides parse as item tuples, never char-splits."""
    for ov in (
        "data.scenario_kinds=(lockbit,supply_chain)",
        "data.scenario_kinds=lockbit,supply_chain",
        "data.scenario_kinds=[lockbit, supply_chain]",
    ):
        cfg = load_config(None, [ov])
        assert cfg.data.scenario_kinds == ("lockbit", "supply_chain"), ov


def test_operating_point_fixed_fp_budget():
    import numpy as np

    from nerrf_amd.eval import operating_point

    y = np.array([1, 1, 1, 1, 0, 0, 0, 0, 0, 0])
    s = np.array([0.95, 0.9, 0.85, 0.4, 0.5, 0.3, 0.2, 0.1, 0.05, 0.01])
    op = operating_point(y, s, max_fp_frac=0.05)
    # flagging the top 3 keeps precision 1.0; the 4th pick (0.5) is a FP
    assert op["precision"] >= 0.95
    assert op["recall"] == 0.75
    assert op["threshold"] > 0.5
    # degenerate: no positives
    op0 = operating_point(np.zeros(5), np.random.default_rng(0).random(5))
    assert op0["recall"] == 0.0
    # generous budget flags more
    op2 = operating_point(y, s, max_fp_frac=0.5)
    assert op2["recall"] >= op["recall"]


def test_flags_doc_covers_all_env_flags():
    """docs/flags.md must document every NERRF_* runtime flag in the code
    (guards the catalogue against rot as flags are added)."""
    import os
    import re

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    flags = set()
    scan_dirs = ["nerrf_amd", "tools"]
    for d in scan_dirs:
        for base, _dirs, files in os.walk(os.path.join(root, d)):
            if "__pycache__" in base:
                continue
            for fn in files:
                if fn.endswith(".py"):
                    src = open(os.path.join(base, fn)).read()
                    flags.update(re.findall(r"NERRF_[A-Z_]+", src))
    flags.update(re.findall(r"NERRF_[A-Z_]+", open(os.path.join(root, "bench.py")).read()))
    doc = open(os.path.join(root, "docs", "flags.md")).read()
    documented = set(re.findall(r"NERRF_[A-Z_]+", doc))
    missing = flags - documented
    assert not missing, f"undocumented runtime flags: {sorted(missing)}"
