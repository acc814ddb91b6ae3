"""Supply-chain compromise scenario (the reference's second headline scenario).

The reference's expected-results table names two scenarios: LockBit-on-
WordPress and a supply-chain attack ("npm-style postinstall", expected
18 min MTTR / 0 MB loss — reference README.md:121-127, labeled "expected"
with no shipped implementation).  This implements it for real:

  * a victim "application" directory with dependency tree (node_modules-like
    layout) and application data,
  * the attack: a postinstall-style process overwrites a dependency's
    entrypoint with a backdoored copy, then exfil-stages application data
    (reads + writes to a staging blob) — NO mass encryption, so the LockBit
    indicators (extension regex, ransom note) stay silent and detection must
    come from the graph/sequence anomalies (one process touching the whole
    dependency tree + data reads feeding a single growing blob),
  * recovery: quarantine staging blobs + restore backdoored dependencies
    from the content-addressed manifest (sha256-verified).
"""
from __future__ import annotations

import hashlib
import json
import time
from dataclasses import dataclass, field
from pathlib import Path
from typing import Dict, List, Optional

import numpy as np


@dataclass
class SupplyChainReport:
    backdoored: List[str] = field(default_factory=list)
    staged_blob: str = ""
    bytes_staged: int = 0
    t_start: float = 0.0
    t_end: float = 0.0
    manifest_sha256: Dict[str, str] = field(default_factory=dict)
    trace_events: List[dict] = field(default_factory=list)


def seed_app(app_dir: str | Path, n_deps: int = 12, n_data: int = 8, seed: int = 0) -> Dict[str, str]:
    """Create an app tree: deps with entrypoints + application data files."""
    app = Path(app_dir)
    rng = np.random.default_rng(seed)
    manifest: Dict[str, str] = {}
    for i in range(n_deps):
        dep = app / "node_modules" / f"dep_{i:03d}"
        dep.mkdir(parents=True, exist_ok=True)
        body = (f"// dep_{i:03d}\nmodule.exports = {int(rng.integers(1 << 30))};\n").encode()
        p = dep / "index.js"
        p.write_bytes(body)
        manifest[str(p)] = hashlib.sha256(body).hexdigest()
    data = app / "data"
    data.mkdir(parents=True, exist_ok=True)
    for i in range(n_data):
        body = rng.integers(0, 256, size=24 * 1024, dtype=np.uint8).tobytes()
        p = data / f"records_{i:02d}.db"
        p.write_bytes(body)
        manifest[str(p)] = hashlib.sha256(body).hexdigest()
    return manifest


def run_supply_chain_attack(
    app_dir: str | Path,
    trace_path: Optional[str | Path] = None,
    pid: int = 7777,
) -> SupplyChainReport:
    app = Path(app_dir)
    rep = SupplyChainReport()
    rep.t_start = time.time()

    def emit(event: str, path: str, size: int = 0, new_path: str = "") -> None:
        rec = {"timestamp": time.time(), "event": event, "path": path,
               "size": size, "pid": pid}
        if new_path:
            rec["new_path"] = new_path
        rep.trace_events.append(rec)

    backdoor = b"// postinstall payload\nrequire('child_process');\n"
    # phase 1: trojanize every dependency entrypoint
    for p in sorted((app / "node_modules").glob("dep_*/index.js")):
        original = p.read_bytes()
        rep.manifest_sha256[str(p)] = hashlib.sha256(original).hexdigest()
        emit("openat", str(p))
        emit("read", str(p), len(original))
        p.write_bytes(backdoor + original)
        emit("write", str(p), len(backdoor) + len(original))
        rep.backdoored.append(str(p))
    # phase 2: stage application data into one exfil blob
    blob = app / ".cache" / "telemetry.bin"
    blob.parent.mkdir(exist_ok=True)
    rep.staged_blob = str(blob)
    with open(blob, "wb") as fh:
        for p in sorted((app / "data").glob("*.db")):
            body = p.read_bytes()
            emit("openat", str(p))
            emit("read", str(p), len(body))
            fh.write(body)
            emit("write", str(blob), len(body))
            rep.bytes_staged += len(body)
    rep.t_end = time.time()
    if trace_path is not None:
        with open(trace_path, "w") as fh:
            for rec in rep.trace_events:
                fh.write(json.dumps(rec) + "\n")
    return rep


def recover_supply_chain(
    app_dir: str | Path,
    report: SupplyChainReport,
    manifest: Dict[str, str],
) -> Dict:
    """Quarantine the staging blob; strip the backdoor from dependencies;
    sha256-verify every restored file."""
    t0 = time.perf_counter()
    restored = 0
    backdoor = b"// postinstall payload\nrequire('child_process');\n"
    for p_str in report.backdoored:
        p = Path(p_str)
        body = p.read_bytes()
        if body.startswith(backdoor):
            p.write_bytes(body[len(backdoor):])
        if hashlib.sha256(p.read_bytes()).hexdigest() == manifest.get(str(p)):
            restored += 1
    blob = Path(report.staged_blob)
    quarantined = False
    if blob.exists():
        blob.rename(blob.with_suffix(".quarantined"))
        quarantined = True
    ok = all(
        Path(p).exists() and hashlib.sha256(Path(p).read_bytes()).hexdigest() == h
        for p, h in manifest.items()
    )
    return {
        "restored_deps": restored,
        "deps_total": len(report.backdoored),
        "blob_quarantined": quarantined,
        "recovered_ok": ok,
        "duration_ms": (time.perf_counter() - t0) * 1000.0,
        "data_loss_mb": 0.0 if ok else report.bytes_staged / 1e6,
    }
