"""Rollback executor + sandboxed undo validation.

Executor semantics follow the reference recovery loop (behavior of
benchmarks/m1/scripts/m1_rollback.sh: for every *.lockbit3 restore the
original name, ms-precision timing, recovery_results.json with files/s and
MB/s) extended with real decryption (the attack XOR is an involution) and a
sha256 safety gate.

Sandbox: the spec'd Firecracker microVM + OverlayFS reverse-diff + sha256
gate (reference architecture.mdx:77-87, ROADMAP.md:71-78) is modeled as a
copy-on-write staging directory: the undo plan is replayed against the
staging copy first; only if every restored file passes its sha256 check is
the plan applied to the live directory.  Same gate contract (diff == 0),
container-friendly mechanism.
"""
from __future__ import annotations

import json
import shutil
import tempfile
import time
from dataclasses import dataclass, field
from pathlib import Path
from typing import Dict, List, Optional

from ..harness.attack_sim import decrypt_file, verify_manifest


@dataclass
class RollbackResult:
    duration_ms: float
    files_restored: int
    files_failed: int
    mb_restored: float
    files_per_sec: float
    mb_per_sec: float
    sandbox_validated: bool
    sha256_ok: Optional[bool] = None
    details: List[str] = field(default_factory=list)

    def as_dict(self) -> dict:
        return {
            "duration_ms": self.duration_ms,
            "files_restored": self.files_restored,
            "files_failed": self.files_failed,
            "mb_restored": self.mb_restored,
            "files_per_sec": self.files_per_sec,
            "mb_per_sec": self.mb_per_sec,
            "sandbox_validated": self.sandbox_validated,
            "sha256_ok": self.sha256_ok,
        }


def _restore_dir(
    directory: Path,
    encrypted_ext: str,
    decrypt: bool,
) -> tuple[int, int, float, List[str]]:
    restored = failed = 0
    nbytes = 0
    details: List[str] = []
    for enc in sorted(directory.rglob(f"*{encrypted_ext}")):
        try:
            size = enc.stat().st_size
            if decrypt:
                orig = decrypt_file(enc, encrypted_ext)
            else:  # rename-back only (reference m1_rollback.sh semantics)
                orig = enc.with_name(enc.name[: -len(encrypted_ext)])
                enc.rename(orig)
            restored += 1
            nbytes += size
            details.append(f"restored {orig.name}")
        except (OSError, AssertionError) as e:
            failed += 1
            details.append(f"FAILED {enc.name}: {e}")
    return restored, failed, nbytes / 1e6, details


def sandbox_validate(
    directory: str | Path,
    encrypted_ext: str,
    manifest: Optional[Dict[str, str]] = None,
    decrypt: bool = True,
) -> bool:
    """Replay the undo against a CoW staging copy; gate = sha256 diff == 0."""
    directory = Path(directory)
    with tempfile.TemporaryDirectory(prefix="nerrf_sandbox_") as tmp:
        staging = Path(tmp) / "staging"
        shutil.copytree(directory, staging)
        restored, failed, _, _ = _restore_dir(staging, encrypted_ext, decrypt)
        if failed:
            return False
        if manifest is not None:
            # map manifest entries into the staging tree by path relative to
            # the live directory (supports nested victim layouts)
            remapped = {}
            for p, digest in manifest.items():
                try:
                    rel = Path(p).relative_to(directory)
                except ValueError:
                    rel = Path(Path(p).name)
                remapped[str(staging / rel)] = digest
            checks = verify_manifest(remapped)
            return all(checks.values())
        return restored > 0


def execute_rollback(
    directory: str | Path,
    encrypted_ext: str = ".lockbit3",
    manifest: Optional[Dict[str, str]] = None,
    decrypt: bool = True,
    validate_in_sandbox: bool = True,
    results_path: Optional[str | Path] = None,
) -> RollbackResult:
    """Validate in the sandbox (optional), then restore the live directory."""
    directory = Path(directory)
    sandbox_ok = True
    if validate_in_sandbox:
        sandbox_ok = sandbox_validate(directory, encrypted_ext, manifest, decrypt)
        if not sandbox_ok:
            return RollbackResult(
                duration_ms=0.0, files_restored=0, files_failed=0, mb_restored=0.0,
                files_per_sec=0.0, mb_per_sec=0.0, sandbox_validated=False,
                sha256_ok=False, details=["sandbox gate rejected the plan"],
            )
    t0 = time.perf_counter()
    restored, failed, mb, details = _restore_dir(directory, encrypted_ext, decrypt)
    dur_ms = (time.perf_counter() - t0) * 1000.0
    sha_ok = None
    if manifest is not None:
        sha_ok = all(verify_manifest(manifest).values())
    res = RollbackResult(
        duration_ms=dur_ms,
        files_restored=restored,
        files_failed=failed,
        mb_restored=mb,
        files_per_sec=restored / (dur_ms / 1000.0) if dur_ms > 0 else 0.0,
        mb_per_sec=mb / (dur_ms / 1000.0) if dur_ms > 0 else 0.0,
        sandbox_validated=sandbox_ok,
        sha256_ok=sha_ok,
        details=details,
    )
    if results_path is not None:
        with open(results_path, "w") as fh:
            json.dump(res.as_dict(), fh, indent=2)
    return res
