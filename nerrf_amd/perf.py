"""Runtime performance knobs for MI355X.

`enable_tuned_gemms()` points PyTorch's TunableOp at the vendored
hipBLASLt/rocBLAS algorithm table (nerrf_amd/tunableop_gfx950.csv — tuned on
an MI355X for this model's GEMM shapes; +7% on the training step) with
re-tuning disabled, so runs are deterministic and never pay tuning cost.
"""
from __future__ import annotations

import os
from pathlib import Path

_TABLE = Path(__file__).parent / "tunableop_gfx950.csv"


def enable_tuned_gemms(tuning: bool = False) -> bool:
    """Enable TunableOp with the vendored gfx950 result table.

    Returns True when enabled.  Safe to call on CPU-only hosts (no-op).
    """
    import torch

    if not torch.cuda.is_available() or not hasattr(torch.cuda, "tunable"):
        return False
    # allow an operator-driven re-tune via the standard env knobs
    tuning = tuning or os.environ.get("PYTORCH_TUNABLEOP_TUNING", "0") == "1"
    try:
        torch.cuda.tunable.enable(True)
        torch.cuda.tunable.tuning_enable(tuning)
        if tuning:
            fname = os.environ.get(
                "PYTORCH_TUNABLEOP_FILENAME",
                os.path.join(os.environ.get("TMPDIR", "/tmp"), "nerrf_tunableop_.csv"),
            )
            torch.cuda.tunable.set_filename(fname, insert_device_ordinal=True)
            # no warm-start: shapes already present in a loaded table are
            # never re-tuned, which would defeat a longer tuning budget
        elif _TABLE.exists():
            torch.cuda.tunable.read_file(str(_TABLE))
        return True
    except (RuntimeError, OSError):
        return False
