"""First-N-KB content byte analysis (partial-encryption indicator).

The reference's threat model lists "first-N-KB partial-encrypt byte
analysis" among the detection indicators (threat-model.mdx:246-262,
architecture.mdx:113-121): LockBit-class ransomware often encrypts only the
head of each file for speed, so the byte distribution of the first few KB
separates encrypted content from plaintext even when the tail is intact.

Syscall traces carry no content, so this probe reads the files themselves —
it is only meaningful where the engine shares a filesystem with the
workload (the DaemonSet deployment, the on-disk scenario harness, tests).
The engine keeps it OFF unless `content_probe_root` is set.

Shannon entropy of the first 4 KB: ASCII/text sits ~4-5.5 bits/byte,
compressed or encrypted content >7.5, and the harness's rolling-XOR
"encryption" of text lands ~6-7 (the key spreads each symbol over key-len
byte values) — the default threshold 6.0 separates plaintext from both.
"""
from __future__ import annotations

import math
import os
from typing import Dict, Iterable, List, Optional

ENTROPY_THRESHOLD = 6.0  # bits/byte over the first N KB


def first_kb_entropy(path: str, n_kb: int = 4) -> Optional[float]:
    """Shannon entropy (bits/byte) of the first `n_kb` KB, or None if the
    file is unreadable/empty (never raises — the probe must not take down
    a scoring tick)."""
    try:
        with open(path, "rb") as f:
            data = f.read(n_kb * 1024)
    except OSError:
        return None
    if not data:
        return None
    counts = [0] * 256
    for b in data:
        counts[b] += 1
    n = len(data)
    ent = 0.0
    for c in counts:
        if c:
            p = c / n
            ent -= p * math.log2(p)
    return ent


def probe_encrypted_fraction(
    paths: Iterable[str],
    root: str = "",
    n_kb: int = 4,
    max_files: int = 32,
    threshold: float = ENTROPY_THRESHOLD,
) -> Dict[str, float]:
    """Sample up to `max_files` of `paths` (joined under `root` when they
    are not absolute on this filesystem) and return
    {probed, high_entropy, encrypted_content_frac}.

    Missing files are skipped, not counted: the trace's path namespace may
    differ from the engine's (container vs host mounts)."""
    probed = 0
    high = 0
    for p in paths:
        if probed >= max_files:
            break
        cand: List[str] = [p]
        if root:
            cand.append(os.path.join(root, p.lstrip("/")))
        ent = None
        for c in cand:
            ent = first_kb_entropy(c, n_kb=n_kb)
            if ent is not None:
                break
        if ent is None:
            continue
        probed += 1
        if ent >= threshold:
            high += 1
    return {
        "probed": float(probed),
        "high_entropy": float(high),
        "encrypted_content_frac": (high / probed) if probed else 0.0,
    }
