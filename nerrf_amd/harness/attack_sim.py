"""On-disk LockBit-style attack simulator (deterministic, reversible).

Mirrors the reference benchmark methodology (behavior of
benchmarks/m1/scripts/sim_lockbit_m1.py: seed enterprise files, chunked
XOR "encryption" with a SHA-256-derived keystream, rename to .lockbit3,
delete the original, ransom note, JSON trace lines) as a pure in-process
library so e2e tests run without k8s.  XOR is an involution, so a correct
rollback (decrypt + rename-back) restores content byte-for-byte — giving the
recovery loop a sha256-verifiable ground truth.
"""
from __future__ import annotations

import hashlib
import json
import os
import time
from dataclasses import dataclass, field
from pathlib import Path
from typing import Dict, List, Optional

import numpy as np


@dataclass
class AttackReport:
    target_dir: str
    encrypted_ext: str
    files_attacked: List[str] = field(default_factory=list)
    bytes_attacked: int = 0
    t_start: float = 0.0
    t_end: float = 0.0
    manifest_sha256: Dict[str, str] = field(default_factory=dict)  # pre-attack
    trace_events: List[dict] = field(default_factory=list)


def _keystream(path_name: str, length: int) -> np.ndarray:
    """Deterministic per-file keystream from SHA-256(path)."""
    key = hashlib.sha256(path_name.encode()).digest()
    reps = length // len(key) + 1
    return np.frombuffer((key * reps)[:length], dtype=np.uint8)


def seed_files(
    target_dir: str | Path,
    n_files: int = 24,
    file_kb: int = 64,
    seed: int = 0,
) -> Dict[str, str]:
    """Create victim .dat files; returns {path: sha256} manifest."""
    target = Path(target_dir)
    target.mkdir(parents=True, exist_ok=True)
    rng = np.random.default_rng(seed)
    manifest = {}
    for i in range(n_files):
        p = target / f"doc_{i:04d}.dat"
        data = rng.integers(0, 256, size=file_kb * 1024, dtype=np.uint8).tobytes()
        p.write_bytes(data)
        manifest[str(p)] = hashlib.sha256(data).hexdigest()
    return manifest


def run_attack(
    target_dir: str | Path,
    encrypted_ext: str = ".lockbit3",
    chunk_kb: int = 256,
    note_name: str = "README_LOCKBIT.txt",
    trace_path: Optional[str | Path] = None,
) -> AttackReport:
    """Encrypt every .dat file under target_dir (XOR keystream), rename,
    delete originals, drop a ransom note; emits trace events."""
    target = Path(target_dir)
    report = AttackReport(target_dir=str(target), encrypted_ext=encrypted_ext)
    report.t_start = time.time()
    pid = os.getpid()

    def emit(event: str, path: str, size: int = 0, new_path: str = "") -> None:
        rec = {"timestamp": time.time(), "event": event, "path": path,
               "size": size, "pid": pid}
        if new_path:
            rec["new_path"] = new_path
        report.trace_events.append(rec)

    for p in sorted(target.glob("*.dat")):
        data = p.read_bytes()
        report.manifest_sha256[str(p)] = hashlib.sha256(data).hexdigest()
        emit("openat", str(p))
        arr = np.frombuffer(data, dtype=np.uint8)
        ks = _keystream(p.name, len(arr))
        enc = (arr ^ ks).tobytes()
        enc_path = p.with_name(p.name + encrypted_ext)
        chunk = chunk_kb * 1024
        with open(enc_path, "wb") as fh:
            for off in range(0, len(enc), chunk):
                emit("read", str(p), min(chunk, len(enc) - off))
                fh.write(enc[off : off + chunk])
                emit("write", str(enc_path), min(chunk, len(enc) - off))
        emit("rename", str(p), 0, str(enc_path))
        p.unlink()
        emit("unlink", str(p))
        report.files_attacked.append(str(p))
        report.bytes_attacked += len(data)

    note = target / note_name
    note.write_text("ALL YOUR FILES ARE ENCRYPTED (simulation)\n")
    emit("openat", str(note))
    emit("write", str(note), 42)
    report.t_end = time.time()

    if trace_path is not None:
        with open(trace_path, "w") as fh:
            for rec in report.trace_events:
                fh.write(json.dumps(rec) + "\n")
    return report


def decrypt_file(enc_path: str | Path, encrypted_ext: str = ".lockbit3") -> Path:
    """Reverse one file: XOR back and rename to the original name."""
    enc_path = Path(enc_path)
    assert enc_path.name.endswith(encrypted_ext), enc_path
    orig = enc_path.with_name(enc_path.name[: -len(encrypted_ext)])
    arr = np.frombuffer(enc_path.read_bytes(), dtype=np.uint8)
    ks = _keystream(orig.name, len(arr))
    orig.write_bytes((arr ^ ks).tobytes())
    enc_path.unlink()
    return orig


def verify_manifest(manifest: Dict[str, str]) -> Dict[str, bool]:
    """sha256-verify files against a pre-attack manifest."""
    out = {}
    for path, digest in manifest.items():
        p = Path(path)
        out[path] = p.exists() and hashlib.sha256(p.read_bytes()).hexdigest() == digest
    return out
