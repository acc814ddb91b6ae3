"""Streaming inference engine: ingest -> delta graph -> score -> plan -> undo.

This is BASELINE.json config 5 ("streaming inference: gRPC trace.proto ingest
-> 30 s delta graph -> online GNN+MCTS") as one process; `shard_id/world`
parameters shard scoring windows across GPUs (one engine per GPU, RCCL used
by the training path; serving shards are independent — xGMI exchange only for
the optional score merge).

Detection combines the joint model's node/sequence scores with the documented
rule indicators (reference architecture.mdx:113-121: write-to-rename ratio >
0.8, .lockbit* extension regex, ransom-note name, recon burst) so the engine
is operational even before training converges; a trained checkpoint sharpens
it.
"""
from __future__ import annotations

import os
import re
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import numpy as np
import torch

from ..data.trace import EventArray
from ..graph.store import DeltaGraphStore
from ..models.joint import JointConfig, NerrfJointModel
from ..planner.mcts import PlanResult, run_mcts, run_mcts_gpu
from ..planner.rewards import PlannerParams, build_state
from .rollback import RollbackResult, execute_rollback

_SUSPICIOUS_EXT = re.compile(r"\.(lockbit\w*|encrypted|locked|crypt\w*)$", re.IGNORECASE)
_RANSOM_NOTE = re.compile(r"(^|/)(README|HOW_TO|RESTORE)[-_]", re.IGNORECASE)


@dataclass
class Detection:
    alarm: bool
    t_detect: float
    file_scores: Dict[str, float]  # live path -> anomaly score [0,1]
    file_mb: Dict[str, float]
    proc_scores: Dict[int, float]
    encrypted_paths: List[str] = field(default_factory=list)
    exfil_destinations: List[str] = field(default_factory=list)
    indicators: Dict[str, float] = field(default_factory=dict)
    window_events: int = 0
    # counterfactual-refinement context (node features + sampled adjacency +
    # path->node map) for model-in-the-loop plan scoring; host-side numpy
    refine_ctx: Optional[dict] = None
    # raw per-head window maxima (before score folding) — the alarm gate's
    # inputs, exposed for calibration and observability
    node_max: float = 0.0
    seq_max: float = 0.0


PRETRAINED_DIR = "checkpoints/pretrained"


def load_model_from_checkpoint(path: str) -> NerrfJointModel:
    """Load a joint model from a checkpoint dir (manifest carries the config)."""
    import json
    from pathlib import Path

    from ..checkpoint import load_checkpoint
    from ..models.graphsage import SageConfig
    from ..models.lstm import LSTMConfig

    manifest = json.loads((Path(path) / "checkpoint.json").read_text())
    cfg_d = (manifest.get("config") or {}).get("model") or manifest.get("config")
    cfg = JointConfig()
    if cfg_d:
        cfg = JointConfig(
            sage=SageConfig(**cfg_d.get("sage", {})),
            lstm=LSTMConfig(**cfg_d.get("lstm", {})),
            **{k: v for k, v in cfg_d.items() if k not in ("sage", "lstm")},
        )
    model = NerrfJointModel(cfg)
    load_checkpoint(path, model)
    return model


class StreamingEngine:
    def __init__(
        self,
        model: Optional[NerrfJointModel] = None,
        device: str = "cpu",
        dtype: torch.dtype = torch.float32,
        window_s: float = 30.0,
        alarm_threshold: float = 0.7,
        planner_params: Optional[PlannerParams] = None,
        shard_id: int = 0,
        world: int = 1,
        calibration: Optional[str] = "auto",
        content_probe_root: Optional[str] = None,
    ) -> None:
        self.device = torch.device(device)
        self.dtype = dtype
        self.model = (model or NerrfJointModel(JointConfig())).to(self.device, self.dtype).eval()
        if self.device.type == "cuda" and self.dtype == torch.bfloat16:
            # one fused MFMA kernel per GNN layer on the scoring path
            self.model.gnn.use_fused_inference = True
        self.store = DeltaGraphStore(window_s=window_s)
        from ..graph.incremental import IncrementalWindowState

        self._inc_state = IncrementalWindowState()
        # first-N-KB partial-encrypt byte analysis (reference
        # threat-model.mdx:246-262): only meaningful when the engine shares
        # a filesystem with the workload — None keeps the probe off
        self.content_probe_root = content_probe_root
        self.alarm_threshold = alarm_threshold
        # calibrated per-channel thresholds (tools/calibrate_alarm.py sweep
        # over off-distribution window configs; benign FP = 0 across the
        # sweep — docs/threat-model.md has the trade table).  "auto" loads
        # the vendored table; None keeps the single-threshold rule.
        self.calibrated = None
        if calibration is not None and calibration != "off":
            # "auto" -> the vendored default; anything else is a path to a
            # calibration json (e.g. the proc-identity checkpoint ships its
            # own next to its weights: checkpoints/pretrained_procid/)
            import json as _json
            from pathlib import Path as _Path

            cal_p = (
                _Path(__file__).parent / "alarm_calibration.json"
                if calibration == "auto"
                else _Path(calibration)
            )
            if cal_p.exists():
                try:
                    c = _json.loads(cal_p.read_text())
                    self.calibrated = {"ind_thr": float(c["ind_thr"]),
                                       "model_thr": float(c["model_thr"])}
                except (ValueError, KeyError):
                    self.calibrated = None
        self.planner_params = planner_params or PlannerParams()
        self.shard_id = shard_id
        self.world = world
        self.scored_windows = 0
        self.events_scored = 0

    # ------------------------------------------------------------------ ingest
    def ingest_events(self, arr: EventArray) -> None:
        """Bulk columnar ingest (store-side string remap + delta chunking)."""
        self.store.append_array(arr)

    def ingest_from_tracker(self, address: str, max_events: Optional[int] = None,
                            timeout_s: Optional[float] = 10.0) -> int:
        from .tracker_client import pump_into_store

        return pump_into_store(address, self.store, max_events=max_events, timeout_s=timeout_s)

    # ------------------------------------------------------------------- score
    @torch.no_grad()
    def score_window(self, now: Optional[float] = None) -> Detection:
        # incremental mode (the production `now=None` tick: window == whole
        # delta set) merges cached per-delta summaries instead of re-scanning
        # every event; an explicit `now` trims at event granularity, which
        # invalidates the summaries, so that path takes the full rebuild
        timing = {} if os.environ.get("NERRF_SERVE_TIMING") else None
        _t0 = time.perf_counter() if timing is not None else 0.0
        events, window_deltas = self.store.compact_with_deltas(now)
        use_incremental = now is None and len(window_deltas) > 0
        t_detect = time.time()
        if timing is not None:
            timing["compact"] = time.perf_counter() - _t0
        if len(events) == 0:
            return Detection(False, t_detect, {}, {}, {}, window_events=0)
        # ---- graph build: host identity/edges once; GPU feature compaction
        # on ROCm devices (HBM-resident x), CPU features otherwise ----------
        import numpy as _np

        from ..data.sequences import build_sequences, build_sequences_torch
        from ..graph.constructor import build_edges_and_flags, build_graph, build_graph_parts
        from ..graph.sampling import sample_fanout, to_csr

        dev_cols = None
        if use_incremental:
            from ..graph.incremental import merge_window

            if self.device.type == "cuda":
                # HBM-resident delta ring: sealed deltas' columns ship to
                # the GPU once; the window is a device-side cat
                dev_cols = self.store.device_columns(window_deltas, self.device)
            sums = self._inc_state.summaries(window_deltas)
            parts, ed = merge_window(
                events, sums,
                device=self.device if self.device.type == "cuda" else None,
                dev_cols=dev_cols,
                state=self._inc_state,
            )
        else:
            parts = build_graph_parts(events)
            ed = build_edges_and_flags(parts)
        from ..graph.constructor import file_node_kinds

        node_kind = _np.concatenate(
            [file_node_kinds(events.paths, parts["path_root"], parts["touched_roots"]),
             _np.zeros(parts["n_procs"], dtype=_np.int8)]
        )
        node_key = _np.concatenate(
            [parts["touched_roots"], parts["upids"].astype(_np.int64)]
        )
        if timing is not None:
            timing["graph_merge"] = time.perf_counter() - _t0 - sum(timing.values())
        nbr_idx = nbr_w = None
        if self.device.type != "cuda":
            csr = to_csr(ed["edge_index"], parts["n_nodes"], ed["edge_weight"])
            nbr_idx, nbr_w = sample_fanout(csr, 16, seed=self.scored_windows)
        if timing is not None:
            timing["csr_fanout"] = time.perf_counter() - _t0 - sum(timing.values())

        if self.device.type == "cuda":
            from ..graph.gpu_store import gpu_window_graph

            gg = gpu_window_graph(events, self.device, parts=parts, ed=ed,
                                  dtype=self.dtype, dev_cols=dev_cols)
            x = gg["x"]
            edge_index = gg["edge_index"]
            edge_weight = gg["edge_weight"]
            edge_ts = gg["edge_ts"]
            # sequence assembly on-device too (the numpy build is ~57 ms per
            # 600k-event window; the torch version is a radix sort + scatters)
            seq_feats, seq_lengths_cpu, seq_fids = build_sequences_torch(
                events, self.device, dtype=self.dtype, dev_cols=dev_cols
            )
            seq_lengths = seq_lengths_cpu.to(self.device)
            seq_fids = seq_fids.numpy()
        else:
            g = build_graph(events, parts=parts)
            x = torch.from_numpy(g.x).to(self.dtype)
            edge_index = torch.from_numpy(g.edge_index)
            edge_weight = torch.from_numpy(g.edge_weight)
            edge_ts = torch.from_numpy(g.edge_ts)
            seqs = build_sequences(events, None)
            seq_feats = torch.from_numpy(seqs.feats).to(self.dtype)
            seq_lengths = torch.from_numpy(seqs.lengths)
            seq_fids = seqs.file_path_id
        if self.device.type == "cuda":
            # CSR + fanout on-device (the host argsort was ~13.6 ms/window)
            from ..graph.sampling import sample_fanout_torch

            t_idx, t_w = sample_fanout_torch(
                edge_index.to(self.device), edge_weight.to(self.device),
                parts["n_nodes"], 16, seed=self.scored_windows,
            )
        else:
            t_idx = torch.from_numpy(nbr_idx).to(self.device)
            t_w = torch.from_numpy(nbr_w).to(self.device)
        batch = {
            "x": x.to(self.device),
            "nbr_idx": t_idx,
            "nbr_w": t_w,
            "edge_index": edge_index.to(self.device),
            "edge_weight": edge_weight.to(self.device),
            "edge_ts": edge_ts.to(self.device),
            "seq_feats": seq_feats.to(self.device),
            "seq_lengths": seq_lengths.to(self.device),
        }
        node_logit, _, seq_logit = self.model(batch)
        node_score = torch.sigmoid(node_logit.float()).cpu().numpy()
        if timing is not None:
            timing["model_fwd_sync"] = time.perf_counter() - _t0 - sum(timing.values())
        seq_score = (
            torch.sigmoid(seq_logit.float()).cpu().numpy() if seq_logit is not None else None
        )
        self.scored_windows += 1
        self.events_scored += len(events)

        # ---- map node/sequence scores back to live paths -------------------
        # file nodes are the first n_files entries (unique path roots), so
        # the dict builds are straight zips — no per-node function calls
        strings = events.paths.strings
        n_files = int(parts["n_files"])
        ns = node_score
        if len(ns) < len(node_kind):
            ns = _np.concatenate([ns, _np.zeros(len(node_kind) - len(ns))])
        file_scores: Dict[str, float] = dict(
            zip((strings[k] for k in node_key[:n_files].tolist()), ns[:n_files].tolist())
        )
        proc_scores: Dict[int, float] = dict(
            zip(node_key[n_files:].tolist(), ns[n_files : len(node_kind)].tolist())
        )
        file_mb: Dict[str, float] = {}
        if seq_score is not None and len(seq_score) == len(seq_fids):
            for pid_, sc in zip(seq_fids.tolist(), seq_score.tolist()):
                if pid_ >= 0:
                    path = strings[pid_]
                    if sc > file_scores.get(path, 0.0):
                        file_scores[path] = sc
        # bytes per file (window-local) — vectorised: a per-event Python loop
        # here costs ~0.4 s per 600k-event window, which dominates the host
        # path once the model forward is on the GPU.  When the incremental
        # merge ran, the per-delta summaries already hold sparse per-path
        # byte sums (summarize_delta), so the O(window events) bincount
        # collapses to O(touched paths).
        if use_incremental and sums:
            ids = _np.concatenate([s.mb_ids for s in sums])
            vals = _np.concatenate([s.mb_vals for s in sums])
            mb_by_id = _np.bincount(
                ids, weights=vals, minlength=len(strings)) / 1e6
        else:
            valid = events.path_id >= 0
            mb_by_id = np.bincount(
                events.path_id[valid],
                weights=events.nbytes[valid].astype(np.float64),
                minlength=len(strings),
            ) / 1e6
        nz = _np.nonzero(mb_by_id)[0]
        file_mb = dict(zip((strings[i] for i in nz.tolist()),
                           mb_by_id[nz].tolist()))

        if timing is not None:
            timing["score_maps"] = time.perf_counter() - _t0 - sum(timing.values())
        # ---- rule indicators --------------------------------------------
        sc = events.syscall
        from ..data.trace import SYSCALL_IDS

        writes = int((sc == SYSCALL_IDS["write"]).sum())
        renames = int((sc == SYSCALL_IDS["rename"]).sum())
        w2r = renames / max(writes + renames, 1)
        # cached per-path regex bits (grow-only table => only new strings
        # are ever scanned; a fresh regex pass here cost ~5 ms/tick)
        from ..graph.constructor import _string_flag_bits

        bits = _string_flag_bits(events.paths)
        # mask to path ids actually referenced by THIS window: the store's
        # StringTable is global and grow-only, so unmasked bits would latch
        # every indicator permanently once a suspicious string is ever
        # interned (alarm-every-tick after remediation — ADVICE r1)
        present = _np.zeros(len(bits), dtype=bool)
        _pv = events.path_id[events.path_id >= 0]
        _nv = events.new_path_id[events.new_path_id >= 0]
        if len(_pv):
            present[_pv] = True
        if len(_nv):
            present[_nv] = True
        bits = _np.where(present, bits, 0)
        encrypted_paths = [strings[i] for i in _np.nonzero(bits & 1)[0].tolist()]
        note = bool((bits & 2).any())
        # socket egress to destinations outside the allowlist (the policy
        # channel syscall data alone cannot provide — threat-model.md)
        exfil_dests = [strings[i] for i in _np.nonzero(bits & 32)[0].tolist()]
        recon_paths = int((bits & 4).sum())
        indicators = {
            "write_to_rename": w2r,
            "suspicious_ext_count": float(len(encrypted_paths)),
            "ransom_note": float(note),
            "exfil_dest_count": float(len(exfil_dests)),
            "recon_burst": float(recon_paths),
        }
        ind_score = min(
            1.0,
            0.6 * float(len(encrypted_paths) > 0) + 0.3 * float(note)
            + 0.4 * float(w2r > 0.1) + 0.7 * float(len(exfil_dests) > 0),
        )
        if self.content_probe_root is not None:
            # partial-encrypt byte analysis on the flagged (or, absent
            # flags, the highest-write) files that exist on this filesystem
            from .content_probe import probe_encrypted_fraction

            cand = encrypted_paths or [
                p for p, _ in sorted(file_mb.items(), key=lambda kv: -kv[1])[:16]
            ]
            probe = probe_encrypted_fraction(cand, root=self.content_probe_root)
            indicators.update(probe)
            if probe["encrypted_content_frac"] >= 0.5 and probe["probed"] >= 2:
                ind_score = min(1.0, ind_score + 0.6)
        # boost file scores for files with suspicious aliases
        for p in encrypted_paths:
            base = p
            for ext_match in [_SUSPICIOUS_EXT.search(p)]:
                if ext_match:
                    base = p[: ext_match.start()]
            file_scores[base] = max(file_scores.get(base, 0.0), 0.95)
            file_scores[p] = max(file_scores.get(p, 0.0), 0.95)

        model_max = float(node_score.max()) if len(node_score) else 0.0
        seq_max = float(seq_score.max()) if seq_score is not None and len(seq_score) else 0.0
        alarm_score = max(ind_score, min(model_max, seq_max))
        if self.calibrated is not None:
            alarm = (ind_score >= self.calibrated["ind_thr"]
                     or min(model_max, seq_max) >= self.calibrated["model_thr"])
        else:
            alarm = alarm_score >= self.alarm_threshold
        if timing is not None:
            timing["indicators"] = time.perf_counter() - _t0 - sum(timing.values())
        # refinement context: lets plan() re-score counterfactual post-plan
        # graphs through the GNN in one batch (planner/model_eval.py)
        path_to_node = {
            strings[k]: i for i, k in enumerate(node_key[:n_files].tolist())
        }
        proc_nodes = _np.array(
            [n_files + i for i, s_ in enumerate(ns[n_files : len(node_kind)])
             if s_ >= 0.5],
            dtype=_np.int64,
        )
        if len(proc_nodes) == 0 and len(node_kind) > n_files:
            proc_nodes = _np.array(
                [n_files + int(_np.argmax(ns[n_files : len(node_kind)]))],
                dtype=_np.int64,
            )
        refine_ctx = {
            "x": (x.float().cpu().numpy() if hasattr(x, "cpu") else _np.asarray(x)),
            "nbr_idx": (nbr_idx if nbr_idx is not None
                        else batch["nbr_idx"].cpu().numpy()),
            "nbr_w": (nbr_w if nbr_w is not None
                      else batch["nbr_w"].float().cpu().numpy()),
            "path_to_node": path_to_node,
            "proc_nodes": proc_nodes,
        }
        if timing is not None:
            timing["refine_ctx_rest"] = time.perf_counter() - _t0 - sum(timing.values())
            timing["total"] = time.perf_counter() - _t0
            self.last_timing = dict(timing)
        return Detection(
            alarm=alarm,
            t_detect=t_detect,
            file_scores=file_scores,
            file_mb=file_mb,
            proc_scores=proc_scores,
            encrypted_paths=encrypted_paths,
            exfil_destinations=exfil_dests,
            indicators=indicators,
            window_events=len(events),
            refine_ctx=refine_ctx,
            node_max=model_max,
            seq_max=seq_max,
        )


    # -------------------------------------------------------------------- plan
    def plan(self, det: Detection, n_sims: int = 1024, use_gpu: Optional[bool] = None,
             model_refine: bool = True) -> PlanResult:
        paths = list(det.file_scores.keys())
        scores = np.array([det.file_scores[p] for p in paths], dtype=np.float64)
        mb = np.array([max(det.file_mb.get(p, 0.01), 0.01) for p in paths], dtype=np.float64)
        proc = max(det.proc_scores.values(), default=0.0)
        if det.encrypted_paths:
            proc = max(proc, 0.9)
        clean_mb = float(sum(det.file_mb.values()))
        state = build_state(scores, mb, proc_score=proc, remaining_clean_mb=clean_mb,
                            n_groups=self.planner_params.n_groups)
        if use_gpu is None:
            use_gpu = self.device.type == "cuda"
        if use_gpu:
            res = run_mcts_gpu(state, self.planner_params, n_sims=n_sims, device=str(self.device))
        else:
            res = run_mcts(state, self.planner_params, n_sims=n_sims)
        if model_refine and det.refine_ctx is not None and paths:
            res = self._refine_plan(res, det, state, paths, scores)
        return res

    def _refine_plan(self, res: PlanResult, det: Detection, state, paths, scores) -> PlanResult:
        """Batched counterfactual re-scoring of candidate plans through the
        GNN node head (SURVEY §7 "batched leaf evaluation")."""
        from ..planner.model_eval import refine_plans_with_model
        from ..planner.rewards import A_KILL, A_RESTORE, A_STOP

        ctx = det.refine_ctx
        # planner-group -> node-id map, mirroring build_state's bucketing
        order = np.argsort(-scores, kind="stable")
        split = np.array_split(order, self.planner_params.n_groups)
        p2n = ctx["path_to_node"]
        group_nodes = [
            np.array([p2n[paths[i]] for i in ids if paths[i] in p2n], dtype=np.int64)
            for ids in split
        ]
        # candidate set: the MCTS plan plus structured alternatives
        cands = [list(res.plan)]
        top_reverts = [a for a, _v, _n in res.ranked_actions if a >= 3][:3]
        cands.append([A_KILL] + top_reverts)
        cands.append([A_KILL, A_RESTORE])
        if A_KILL not in res.plan and res.plan:
            cands.append([A_KILL] + list(res.plan))
        uniq, seen = [], set()
        for c in cands:
            key = tuple(c)
            if key not in seen:
                seen.add(key)
                uniq.append(c)
        try:
            ranked = refine_plans_with_model(
                self.model, ctx["x"], ctx["nbr_idx"], ctx["nbr_w"], group_nodes,
                ctx["proc_nodes"], uniq, state, self.planner_params,
                device=str(self.device), dtype=self.dtype,
            )
        except Exception:
            return res  # refinement is advisory; the MCTS plan stands
        best_plan, combined, closed, residual = ranked[0]
        return PlanResult(
            plan=[a for a in best_plan if a != A_STOP],
            ranked_actions=res.ranked_actions,
            root_value=closed,
            simulations=res.simulations,
        )

    # ----------------------------------------------------------------- respond
    def respond(
        self,
        det: Detection,
        plan: PlanResult,
        target_dir: str,
        manifest: Optional[Dict[str, str]] = None,
        encrypted_ext: str = ".lockbit3",
    ) -> RollbackResult:
        """Execute the undo plan: sandbox-validate then restore."""
        return execute_rollback(
            target_dir,
            encrypted_ext=encrypted_ext,
            manifest=manifest,
            decrypt=True,
            validate_in_sandbox=True,
        )

    # ------------------------------------------------------------- multi-shard
    def merge_detections(self, det: Detection, group=None) -> Detection:
        """Merge this shard's Detection with every other shard's.

        One engine shard runs per GPU (config 5); shards exchange compact
        detection summaries (scores/indicators, not raw events) through
        torch.distributed — gloo on CPU tests, RCCL over xGMI on an MI355X
        node.  Alarm is the OR over shards; per-file scores merge by max.
        """
        import torch.distributed as dist

        if not dist.is_initialized() or dist.get_world_size(group) == 1:
            return det
        world = dist.get_world_size(group)
        payload = {
            "alarm": det.alarm,
            "t_detect": det.t_detect,
            "file_scores": det.file_scores,
            "file_mb": det.file_mb,
            "proc_scores": det.proc_scores,
            "encrypted_paths": det.encrypted_paths,
            "exfil_destinations": det.exfil_destinations,
            "indicators": det.indicators,
            "window_events": det.window_events,
        }
        gathered: list = [None] * world
        dist.all_gather_object(gathered, payload, group=group)
        merged = Detection(
            alarm=any(g["alarm"] for g in gathered),
            t_detect=min(g["t_detect"] for g in gathered),
            file_scores={},
            file_mb={},
            proc_scores={},
        )
        enc: list = []
        for g in gathered:
            for p, s in g["file_scores"].items():
                merged.file_scores[p] = max(merged.file_scores.get(p, 0.0), s)
            for p, mb in g["file_mb"].items():
                merged.file_mb[p] = merged.file_mb.get(p, 0.0) + mb
            for p, s in g["proc_scores"].items():
                merged.proc_scores[p] = max(merged.proc_scores.get(p, 0.0), s)
            enc.extend(g["encrypted_paths"])
            merged.exfil_destinations = sorted(
                set(merged.exfil_destinations) | set(g.get("exfil_destinations", []))
            )
            for k, v in g["indicators"].items():
                merged.indicators[k] = max(merged.indicators.get(k, 0.0), v)
            merged.window_events += g["window_events"]
        merged.encrypted_paths = sorted(set(enc))
        return merged

    # ----------------------------------------------------------------- monitor
    def run_monitor(
        self,
        interval_s: float = 5.0,
        max_iterations: Optional[int] = None,
        on_alarm=None,
        tracker_address: Optional[str] = None,
        stop_event=None,
        sims: int = 1024,
        registry=None,  # serve.registry.DetectionRegistry: persist alarms
        target_dir: str = "",
    ):
        """Continuous detection loop: ingest -> score -> (on alarm) plan.

        The operational serve mode (config 5): every `interval_s` the engine
        compacts the delta window, scores it, and on alarm produces a ranked
        undo plan, invoking `on_alarm(detection, plan)` — the caller decides
        whether to execute (`respond`).  Bounded by `max_iterations` /
        `stop_event` for tests and supervised runs.  Yields per-iteration
        status dicts.
        """
        import threading

        it = 0
        ingest_thread = None
        if tracker_address is not None:
            ingest_thread = threading.Thread(
                target=self.ingest_from_tracker,
                kwargs={"address": tracker_address, "timeout_s": None},
                daemon=True,
            )
            ingest_thread.start()
        while max_iterations is None or it < max_iterations:
            if stop_event is not None and stop_event.is_set():
                return
            t0 = time.perf_counter()
            det = self.score_window()
            plan = None
            attack_id = None
            if det.alarm:
                plan = self.plan(det, n_sims=sims)
                if registry is not None:
                    attack_id = registry.record(
                        det, plan, target_dir=target_dir,
                        n_groups=self.planner_params.n_groups,
                    )
                if on_alarm is not None:
                    on_alarm(det, plan)
            status = {
                "iteration": it,
                "window_events": det.window_events,
                "alarm": det.alarm,
                "attack_id": attack_id,
                "score_s": time.perf_counter() - t0,
                "plan": None if plan is None else plan.describe(self.planner_params.n_groups),
            }
            yield status
            it += 1
            elapsed = time.perf_counter() - t0
            if max_iterations is None or it < max_iterations:
                time.sleep(max(0.0, interval_s - elapsed))
