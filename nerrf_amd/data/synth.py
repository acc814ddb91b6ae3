"""Deterministic synthetic trace generation (benign load + LockBit-style attack).

The upstream project validates against a deterministic, reversible ransomware
simulator run inside minikube (reference behavior:
/root/reference/benchmarks/m1/scripts/sim_lockbit_m1.py — 5-phase attack:
recon burst, file seeding, chunked encrypt/rename/unlink at a rate limit,
ransom note) with ground truth expressed as an attack time-window CSV
(/root/reference/benchmarks/m1/scripts/m1_minikube_bootstrap.sh:219-224).

Here the same scenario is a pure in-process generator producing columnar
:class:`~nerrf_amd.data.trace.EventArray` batches — no k8s needed — so it can
drive CPU tests, GPU benches, and multi-hour training-set synthesis at
millions of events/second (vectorised numpy, no per-event Python loop).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional, Tuple

import numpy as np

from .trace import SYSCALL_IDS, EventArray, StringTable


@dataclass
class AttackWindow:
    """Ground truth: [t_start, t_end) and the directory under attack."""

    t_start: float
    t_end: float
    target_dir: str
    encrypted_ext: str = ".lockbit3"


@dataclass
class SynthConfig:
    duration_s: float = 120.0
    # benign background
    n_benign_procs: int = 24
    n_benign_files: int = 400
    benign_rate_hz: float = 800.0  # aggregate benign events/second
    # attack
    attack: bool = True
    attack_start_frac: float = 0.35
    n_victim_files: int = 48
    victim_file_mb: float = 2.4
    encrypt_rate_mbps: float = 2.0  # reference sim rate limit (sim_lockbit_m1.py:17)
    chunk_kb: int = 256
    recon_burst: int = 40
    target_dir: str = "/app/uploads"
    encrypted_ext: str = ".lockbit3"
    seed: int = 0
    kind: str = "lockbit"  # lockbit | supply_chain | supply_chain_net | benign_rotate | benign_backup | benign_build
    # adversarial: attack process reports an allowlisted comm (defeats naive
    # trust in the process-identity channel; behavior channels unaffected)
    comm_masquerade: bool = False


_BENIGN_DIRS = ["/var/www/html", "/home/svc/data", "/app/cache", "/var/log/app"]
_BENIGN_EXTS = [".html", ".log", ".json", ".tmp", ".dat", ".css"]
_RECON_BINS = ["/bin/ps", "/bin/netstat", "/usr/bin/whoami", "/bin/df", "/bin/mount", "/bin/ss"]


def _interleave(cols: List[Tuple[np.ndarray, ...]]) -> Tuple[np.ndarray, ...]:
    """Concatenate column tuples and sort all by the first column (time)."""
    merged = [np.concatenate([c[i] for c in cols]) for i in range(len(cols[0]))]
    order = np.argsort(merged[0], kind="stable")
    return tuple(m[order] for m in merged)


def generate(cfg: SynthConfig) -> Tuple[EventArray, Optional[AttackWindow]]:
    if cfg.kind in ("supply_chain", "supply_chain_net"):
        return generate_supply_chain(cfg)
    if cfg.kind == "benign_rotate":
        return generate_benign_rotate(cfg)
    if cfg.kind == "benign_backup":
        return generate_benign_backup(cfg)
    if cfg.kind == "benign_build":
        return generate_benign_build(cfg)
    rng = np.random.default_rng(cfg.seed)
    paths = StringTable()
    comms = StringTable()

    benign_comm_ids = np.array(
        [comms.intern(c) for c in ("nginx", "python3", "node", "postgres", "cron", "rsyslogd")],
        dtype=np.int64,
    )
    benign_path_ids = np.array(
        [
            paths.intern(f"{rng.choice(_BENIGN_DIRS)}/f{j:05d}{_BENIGN_EXTS[j % len(_BENIGN_EXTS)]}")
            for j in range(cfg.n_benign_files)
        ],
        dtype=np.int64,
    )

    blocks: List[Tuple[np.ndarray, ...]] = []

    # ---- benign background: open/read/write/close micro-sessions ----------
    n_benign = max(1, int(cfg.benign_rate_hz * cfg.duration_s))
    ts = np.sort(rng.uniform(0.0, cfg.duration_s, size=n_benign))
    pid = rng.integers(100, 100 + cfg.n_benign_procs, size=n_benign).astype(np.int64)
    # syscall mix: mostly read/write/openat/close, few renames (benign
    # write-to-rename ratio stays far below the 0.8 indicator threshold).
    sys_choices = np.array(
        [
            SYSCALL_IDS["openat"],
            SYSCALL_IDS["read"],
            SYSCALL_IDS["write"],
            SYSCALL_IDS["close"],
            SYSCALL_IDS["rename"],
        ],
        dtype=np.int8,
    )
    syscall = sys_choices[
        rng.choice(len(sys_choices), p=[0.22, 0.36, 0.28, 0.13, 0.01], size=n_benign)
    ]
    path_id = benign_path_ids[rng.integers(0, len(benign_path_ids), size=n_benign)]
    new_path_id = np.full(n_benign, -1, dtype=np.int64)
    ren = syscall == SYSCALL_IDS["rename"]
    new_path_id[ren] = benign_path_ids[rng.integers(0, len(benign_path_ids), size=int(ren.sum()))]
    nbytes = np.zeros(n_benign, dtype=np.int64)
    rw = (syscall == SYSCALL_IDS["read"]) | (syscall == SYSCALL_IDS["write"])
    nbytes[rw] = np.exp(rng.normal(8.5, 1.5, size=int(rw.sum()))).astype(np.int64)  # ~5 KB median
    comm_id = benign_comm_ids[pid % len(benign_comm_ids)]
    blocks.append((ts, pid, syscall, path_id, new_path_id, nbytes, comm_id))

    window: Optional[AttackWindow] = None

    if cfg.attack:
        t0 = cfg.attack_start_frac * cfg.duration_s
        atk_pid = np.int64(6666)
        # comm-masquerade variant: the payload reports an allowlisted daemon
        # comm (process-identity adversarial case — the trusted-comm channel
        # must not become a bypass; behavioral channels still fire)
        atk_comm = comms.intern("nginx" if cfg.comm_masquerade else "lockbit")

        # phase 1 — recon burst (exec of enumeration binaries)
        n_recon = cfg.recon_burst
        rts = t0 + np.sort(rng.uniform(0.0, 1.5, size=n_recon))
        rpaths = np.array(
            [paths.intern(_RECON_BINS[i % len(_RECON_BINS)]) for i in range(n_recon)],
            dtype=np.int64,
        )
        blocks.append(
            (
                rts,
                np.full(n_recon, atk_pid),
                np.full(n_recon, SYSCALL_IDS["exec"], dtype=np.int8),
                rpaths,
                np.full(n_recon, -1, dtype=np.int64),
                np.zeros(n_recon, dtype=np.int64),
                np.full(n_recon, atk_comm, dtype=np.int64),
            )
        )

        # phase 2+3 — per-file chunked encrypt: openat, read xk, write xk,
        # rename(.dat -> .lockbit3), unlink(original)
        chunk_bytes = cfg.chunk_kb * 1024
        file_bytes = int(cfg.victim_file_mb * 1e6)
        k = max(1, file_bytes // chunk_bytes)
        per_file_s = (file_bytes / 1e6) / cfg.encrypt_rate_mbps  # rate-limited
        victim_ids = np.array(
            [paths.intern(f"{cfg.target_dir}/doc_{j:04d}.dat") for j in range(cfg.n_victim_files)],
            dtype=np.int64,
        )
        enc_ids = np.array(
            [
                paths.intern(f"{cfg.target_dir}/doc_{j:04d}.dat{cfg.encrypted_ext}")
                for j in range(cfg.n_victim_files)
            ],
            dtype=np.int64,
        )
        ev_per_file = 1 + k + k + 1 + 1  # open, k reads, k writes, rename, unlink
        n_atk = cfg.n_victim_files * ev_per_file
        f_start = t0 + 2.0 + np.arange(cfg.n_victim_files) * per_file_s
        # per-file intra-times: open at 0, read i at (i+0.25)/k, write i at
        # (i+0.75)/k of the file's slice, rename/unlink at the end
        intra = np.concatenate(
            [
                np.zeros(1),
                (np.arange(k) + 0.25) / k * per_file_s,
                (np.arange(k) + 0.75) / k * per_file_s,
                np.array([per_file_s * 1.001, per_file_s * 1.002]),
            ]
        )
        a_ts = (f_start[:, None] + intra[None, :]).reshape(-1)
        sys_pat = np.concatenate(
            [
                np.array([SYSCALL_IDS["openat"]], dtype=np.int8),
                np.full(k, SYSCALL_IDS["read"], dtype=np.int8),
                np.full(k, SYSCALL_IDS["write"], dtype=np.int8),
                np.array([SYSCALL_IDS["rename"], SYSCALL_IDS["unlink"]], dtype=np.int8),
            ]
        )
        a_sys = np.tile(sys_pat, cfg.n_victim_files)
        # reads hit the victim file, writes hit the encrypted copy
        pat_path = np.empty(ev_per_file, dtype=np.int64)
        a_path = np.empty(n_atk, dtype=np.int64)
        a_newp = np.full(n_atk, -1, dtype=np.int64)
        for j in range(cfg.n_victim_files):
            pat_path[0] = victim_ids[j]
            pat_path[1 : 1 + k] = victim_ids[j]
            pat_path[1 + k : 1 + 2 * k] = enc_ids[j]
            pat_path[1 + 2 * k] = victim_ids[j]  # rename src
            pat_path[2 + 2 * k] = victim_ids[j]  # unlink original
            a_path[j * ev_per_file : (j + 1) * ev_per_file] = pat_path
            a_newp[j * ev_per_file + 1 + 2 * k] = enc_ids[j]
        a_bytes = np.zeros(n_atk, dtype=np.int64)
        rw_pat = (a_sys == SYSCALL_IDS["read"]) | (a_sys == SYSCALL_IDS["write"])
        a_bytes[rw_pat] = chunk_bytes
        blocks.append(
            (
                a_ts,
                np.full(n_atk, atk_pid),
                a_sys,
                a_path,
                a_newp,
                a_bytes,
                np.full(n_atk, atk_comm, dtype=np.int64),
            )
        )

        # phase 4 — ransom note
        note_id = paths.intern(f"{cfg.target_dir}/README_LOCKBIT.txt")
        t_note = float(a_ts[-1]) + 0.05
        blocks.append(
            (
                np.array([t_note, t_note + 0.01]),
                np.full(2, atk_pid),
                np.array([SYSCALL_IDS["openat"], SYSCALL_IDS["write"]], dtype=np.int8),
                np.array([note_id, note_id], dtype=np.int64),
                np.full(2, -1, dtype=np.int64),
                np.array([0, 1337], dtype=np.int64),
                np.full(2, atk_comm, dtype=np.int64),
            )
        )
        window = AttackWindow(
            t_start=float(t0),
            t_end=t_note + 0.02,
            target_dir=cfg.target_dir,
            encrypted_ext=cfg.encrypted_ext,
        )

    ts_, pid_, sys_, path_, newp_, bytes_, comm_ = _interleave(blocks)
    arr = EventArray(
        ts=ts_,
        pid=pid_,
        syscall=sys_,
        path_id=path_,
        new_path_id=newp_,
        nbytes=bytes_,
        ret_val=np.zeros(len(ts_), dtype=np.int64),
        comm_id=comm_,
        paths=paths,
        comms=comms,
    )
    return arr, window


def generate_mixed_dataset(
    n_scenarios: int,
    base_seed: int = 0,
    attack_fraction: float = 0.6,
    duration_s: float = 120.0,
) -> List[Tuple[EventArray, Optional[AttackWindow]]]:
    """A set of scenario traces, some with attacks, some clean."""
    out = []
    for i in range(n_scenarios):
        cfg = SynthConfig(
            duration_s=duration_s,
            attack=(i % 100) < int(attack_fraction * 100),
            seed=base_seed + 1000 * i,
            n_victim_files=24 + (i * 7) % 40,
            attack_start_frac=0.2 + 0.5 * ((i * 13) % 10) / 10.0,
        )
        out.append(generate(cfg))
    return out


def generate_supply_chain(cfg: SynthConfig) -> Tuple[EventArray, Optional[AttackWindow]]:
    """Supply-chain compromise trace: trojanize dependency entrypoints, then
    stage application data into one exfil blob.  No renames, no ransom note,
    no suspicious extensions — detection must come from graph/sequence
    anomalies (one process fanning over the whole dependency tree, data
    reads feeding a single growing blob).  Mirrors
    harness/supply_chain.py's on-disk attack as a pure event generator."""
    rng = np.random.default_rng(cfg.seed)
    base, _ = generate(SynthConfig(
        duration_s=cfg.duration_s,
        n_benign_procs=cfg.n_benign_procs,
        n_benign_files=cfg.n_benign_files,
        benign_rate_hz=cfg.benign_rate_hz,
        attack=False,
        seed=cfg.seed,
    ))
    if not cfg.attack:
        return base, None
    paths, comms = base.paths, base.comms
    atk_pid = np.int64(7777)
    # comm_masquerade: the trojaned postinstall reports an allowlisted
    # daemon comm (see the lockbit variant above)
    atk_comm = comms.intern("node" if cfg.comm_masquerade else "postinstall")
    t0 = cfg.attack_start_frac * cfg.duration_s
    n_deps = max(cfg.n_victim_files, 4)
    dep_ids = np.array(
        [paths.intern(f"/srv/app/node_modules/dep_{j:03d}/index.js") for j in range(n_deps)],
        dtype=np.int64,
    )
    n_data = max(n_deps // 2, 2)
    data_ids = np.array(
        [paths.intern(f"/srv/app/data/records_{j:02d}.db") for j in range(n_data)],
        dtype=np.int64,
    )
    blob_id = paths.intern("/srv/app/.cache/telemetry.bin")

    ts_l, sys_l, path_l, bytes_l = [], [], [], []
    t = t0
    for j in range(n_deps):  # trojanize: openat, read, write per dep
        for sc, by in (("openat", 0), ("read", 4096), ("write", 4200)):
            ts_l.append(t)
            sys_l.append(SYSCALL_IDS[sc])
            path_l.append(dep_ids[j])
            bytes_l.append(by)
            t += 0.01 + rng.uniform(0, 0.01)
    chunk = cfg.chunk_kb * 1024
    k = max(int(cfg.victim_file_mb * 1e6) // chunk, 1)
    for j in range(n_data):  # exfil staging: read data, write blob
        ts_l.append(t); sys_l.append(SYSCALL_IDS["openat"]); path_l.append(data_ids[j]); bytes_l.append(0)
        t += 0.005
        for _ in range(k):
            ts_l.append(t); sys_l.append(SYSCALL_IDS["read"]); path_l.append(data_ids[j]); bytes_l.append(chunk)
            t += (chunk / 1e6) / cfg.encrypt_rate_mbps / 2
            ts_l.append(t); sys_l.append(SYSCALL_IDS["write"]); path_l.append(blob_id); bytes_l.append(chunk)
            t += (chunk / 1e6) / cfg.encrypt_rate_mbps / 2
    if cfg.kind == "supply_chain_net":
        # network egress: ship the staged blob to an unlisted destination —
        # the socket node (kind 2) + destination-allowlist flag light up
        dest_id = paths.intern("tcp://203.0.113.37:443")
        ts_l.append(t); sys_l.append(SYSCALL_IDS["connect"]); path_l.append(dest_id); bytes_l.append(0)
        t += 0.005
        for _ in range(k * n_data):
            ts_l.append(t); sys_l.append(SYSCALL_IDS["sendto"]); path_l.append(dest_id); bytes_l.append(chunk)
            t += (chunk / 1e6) / cfg.encrypt_rate_mbps
    n_atk = len(ts_l)
    atk_cols = (
        np.asarray(ts_l),
        np.full(n_atk, atk_pid),
        np.asarray(sys_l, dtype=np.int8),
        np.asarray(path_l, dtype=np.int64),
        np.full(n_atk, -1, dtype=np.int64),
        np.asarray(bytes_l, dtype=np.int64),
        np.full(n_atk, atk_comm, dtype=np.int64),
    )
    benign_cols = (
        base.ts, base.pid, base.syscall, base.path_id, base.new_path_id,
        base.nbytes, base.comm_id,
    )
    ts_, pid_, sys_, path_, newp_, bytes_, comm_ = _interleave([benign_cols, atk_cols])
    arr = EventArray(
        ts=ts_, pid=pid_, syscall=sys_, path_id=path_, new_path_id=newp_,
        nbytes=bytes_, ret_val=np.zeros(len(ts_), dtype=np.int64), comm_id=comm_,
        paths=paths, comms=comms,
    )
    window = AttackWindow(t_start=float(t0), t_end=float(t) + 0.01, target_dir="/srv/app")
    return arr, window


def generate_benign_rotate(cfg: SynthConfig) -> Tuple[EventArray, Optional[AttackWindow]]:
    """Hard negative: log rotation / housekeeping daemon.

    Rename-heavy benign burst (x.log -> x.log.1, truncate + rewrite, gzip
    side-writes) that stresses the write-to-rename and double-extension
    indicators without any attack.  Ground truth: clean (window None)."""
    rng = np.random.default_rng(cfg.seed)
    base, _ = generate(SynthConfig(
        duration_s=cfg.duration_s, n_benign_procs=cfg.n_benign_procs,
        n_benign_files=cfg.n_benign_files, benign_rate_hz=cfg.benign_rate_hz,
        attack=False, seed=cfg.seed,
    ))
    paths, comms = base.paths, base.comms
    rot_pid = np.int64(888)
    rot_comm = comms.intern("logrotate")
    t0 = cfg.attack_start_frac * cfg.duration_s
    n_logs = max(cfg.n_victim_files, 8)
    ts_l, sys_l, path_l, newp_l, bytes_l = [], [], [], [], []
    t = t0
    for j in range(n_logs):
        log_id = paths.intern(f"/var/log/svc/app_{j:03d}.log")
        rot_id = paths.intern(f"/var/log/svc/app_{j:03d}.log.1")
        gz_id = paths.intern(f"/var/log/svc/app_{j:03d}.log.1.gz")
        # rename current -> .1
        ts_l.append(t); sys_l.append(SYSCALL_IDS["rename"]); path_l.append(log_id); newp_l.append(rot_id); bytes_l.append(0)
        t += 0.02
        # recreate + first write
        for sc, by in (("openat", 0), ("write", 512)):
            ts_l.append(t); sys_l.append(SYSCALL_IDS[sc]); path_l.append(log_id); newp_l.append(-1); bytes_l.append(by)
            t += 0.01
        # compress the rotated file: read .1, write .gz, unlink .1
        for _ in range(4):
            ts_l.append(t); sys_l.append(SYSCALL_IDS["read"]); path_l.append(rot_id); newp_l.append(-1); bytes_l.append(65536)
            t += 0.01
            ts_l.append(t); sys_l.append(SYSCALL_IDS["write"]); path_l.append(gz_id); newp_l.append(-1); bytes_l.append(16384)
            t += 0.01
        ts_l.append(t); sys_l.append(SYSCALL_IDS["unlink"]); path_l.append(rot_id); newp_l.append(-1); bytes_l.append(0)
        t += 0.02 + rng.uniform(0, 0.02)
    n = len(ts_l)
    rot_cols = (
        np.asarray(ts_l), np.full(n, rot_pid), np.asarray(sys_l, dtype=np.int8),
        np.asarray(path_l, dtype=np.int64), np.asarray(newp_l, dtype=np.int64),
        np.asarray(bytes_l, dtype=np.int64), np.full(n, rot_comm, dtype=np.int64),
    )
    benign_cols = (base.ts, base.pid, base.syscall, base.path_id,
                   base.new_path_id, base.nbytes, base.comm_id)
    ts_, pid_, sys_, path_, newp_, bytes_, comm_ = _interleave([benign_cols, rot_cols])
    arr = EventArray(ts=ts_, pid=pid_, syscall=sys_, path_id=path_, new_path_id=newp_,
                     nbytes=bytes_, ret_val=np.zeros(len(ts_), dtype=np.int64),
                     comm_id=comm_, paths=paths, comms=comms)
    return arr, None


def generate_benign_backup(cfg: SynthConfig) -> Tuple[EventArray, Optional[AttackWindow]]:
    """Hard negative: backup daemon — reads many data files, writes one big
    archive (the supply-chain exfil lookalike).  Clean ground truth."""
    base, _ = generate(SynthConfig(
        duration_s=cfg.duration_s, n_benign_procs=cfg.n_benign_procs,
        n_benign_files=cfg.n_benign_files, benign_rate_hz=cfg.benign_rate_hz,
        attack=False, seed=cfg.seed,
    ))
    paths, comms = base.paths, base.comms
    bk_pid = np.int64(999)
    bk_comm = comms.intern("backupd")
    t0 = cfg.attack_start_frac * cfg.duration_s
    n_files = max(cfg.n_victim_files, 8)
    archive = paths.intern("/backup/daily/archive.tar")
    chunk = cfg.chunk_kb * 1024
    ts_l, sys_l, path_l, bytes_l = [], [], [], []
    t = t0
    for j in range(n_files):
        src = paths.intern(f"/srv/data/records_{j:03d}.db")
        ts_l.append(t); sys_l.append(SYSCALL_IDS["openat"]); path_l.append(src); bytes_l.append(0)
        t += 0.005
        for _ in range(3):
            ts_l.append(t); sys_l.append(SYSCALL_IDS["read"]); path_l.append(src); bytes_l.append(chunk)
            t += 0.02
            ts_l.append(t); sys_l.append(SYSCALL_IDS["write"]); path_l.append(archive); bytes_l.append(chunk)
            t += 0.02
    n = len(ts_l)
    bk_cols = (
        np.asarray(ts_l), np.full(n, bk_pid), np.asarray(sys_l, dtype=np.int8),
        np.asarray(path_l, dtype=np.int64), np.full(n, -1, dtype=np.int64),
        np.asarray(bytes_l, dtype=np.int64), np.full(n, bk_comm, dtype=np.int64),
    )
    benign_cols = (base.ts, base.pid, base.syscall, base.path_id,
                   base.new_path_id, base.nbytes, base.comm_id)
    ts_, pid_, sys_, path_, newp_, bytes_, comm_ = _interleave([benign_cols, bk_cols])
    arr = EventArray(ts=ts_, pid=pid_, syscall=sys_, path_id=path_, new_path_id=newp_,
                     nbytes=bytes_, ret_val=np.zeros(len(ts_), dtype=np.int64),
                     comm_id=comm_, paths=paths, comms=comms)
    return arr, None


def generate_benign_build(cfg: SynthConfig) -> Tuple[EventArray, Optional[AttackWindow]]:
    """Hard negative: compiler walking a dependency tree.

    One toolchain process fans reads over many sources/headers and writes
    one object per translation unit, with exec bursts (cc1/as/ld) — the
    degree/recon-burst profile of a supply-chain compromise touching a
    dependency tree, but clean ground truth (NEXT.md detection-realism
    item: "compiler-touching-dependency-tree negatives")."""
    rng = np.random.default_rng(cfg.seed)
    base, _ = generate(SynthConfig(
        duration_s=cfg.duration_s, n_benign_procs=cfg.n_benign_procs,
        n_benign_files=cfg.n_benign_files, benign_rate_hz=cfg.benign_rate_hz,
        attack=False, seed=cfg.seed,
    ))
    paths, comms = base.paths, base.comms
    cc_pid = np.int64(777)
    cc_comm = comms.intern("cc1plus")
    t0 = cfg.attack_start_frac * cfg.duration_s
    n_units = max(cfg.n_victim_files, 8)
    headers = [paths.intern(f"/src/include/h_{k:03d}.h") for k in range(16)]
    tool_bins = [paths.intern(p) for p in ("/usr/bin/cc1plus", "/usr/bin/as", "/usr/bin/ld")]
    ts_l, sys_l, path_l, bytes_l = [], [], [], []
    t = t0
    for j in range(n_units):
        src = paths.intern(f"/src/module/unit_{j:03d}.cc")
        obj = paths.intern(f"/src/build/unit_{j:03d}.o")
        # exec burst: toolchain binaries
        for b in tool_bins[:2]:
            ts_l.append(t); sys_l.append(SYSCALL_IDS["exec"]); path_l.append(b); bytes_l.append(0)
            t += 0.002
        ts_l.append(t); sys_l.append(SYSCALL_IDS["openat"]); path_l.append(src); bytes_l.append(0)
        t += 0.003
        ts_l.append(t); sys_l.append(SYSCALL_IDS["read"]); path_l.append(src); bytes_l.append(32768)
        t += 0.004
        # dependency-tree fan: read a random subset of shared headers
        for h in rng.choice(len(headers), size=6, replace=False):
            ts_l.append(t); sys_l.append(SYSCALL_IDS["read"]); path_l.append(headers[int(h)]); bytes_l.append(8192)
            t += 0.002
        for _ in range(2):
            ts_l.append(t); sys_l.append(SYSCALL_IDS["write"]); path_l.append(obj); bytes_l.append(65536)
            t += 0.004
        t += float(rng.uniform(0, 0.01))
    # link step: read all objects, write one binary
    out_bin = paths.intern("/src/build/app.bin")
    ts_l.append(t); sys_l.append(SYSCALL_IDS["exec"]); path_l.append(tool_bins[2]); bytes_l.append(0)
    t += 0.005
    for j in range(n_units):
        obj = paths.intern(f"/src/build/unit_{j:03d}.o")
        ts_l.append(t); sys_l.append(SYSCALL_IDS["read"]); path_l.append(obj); bytes_l.append(65536)
        t += 0.002
        ts_l.append(t); sys_l.append(SYSCALL_IDS["write"]); path_l.append(out_bin); bytes_l.append(65536)
        t += 0.002
    n = len(ts_l)
    cc_cols = (
        np.asarray(ts_l), np.full(n, cc_pid), np.asarray(sys_l, dtype=np.int8),
        np.asarray(path_l, dtype=np.int64), np.full(n, -1, dtype=np.int64),
        np.asarray(bytes_l, dtype=np.int64), np.full(n, cc_comm, dtype=np.int64),
    )
    benign_cols = (base.ts, base.pid, base.syscall, base.path_id,
                   base.new_path_id, base.nbytes, base.comm_id)
    ts_, pid_, sys_, path_, newp_, bytes_, comm_ = _interleave([benign_cols, cc_cols])
    arr = EventArray(ts=ts_, pid=pid_, syscall=sys_, path_id=path_, new_path_id=newp_,
                     nbytes=bytes_, ret_val=np.zeros(len(ts_), dtype=np.int64),
                     comm_id=comm_, paths=paths, comms=comms)
    return arr, None
