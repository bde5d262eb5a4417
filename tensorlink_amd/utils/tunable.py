"""PyTorch TunableOp integration for hipBLASLt/rocBLAS GEMM selection.

The plain-library GEMMs (projections, lm_head) go through torch.matmul;
hipBLASLt's default heuristic picks poor kernels for our skinny decode
shapes (measured ~1.7 TB/s on [256, 18944]x[18944, 3584] — profiles/).
TunableOp benchmarks every available algo per shape and caches the winner;
the tuned table is committed per-arch and loaded at engine start.
"""

from __future__ import annotations

import os

import torch

TUNED_FILE = os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "config", "tunableop_gfx950.csv")


def setup_tunableop(tune: bool = False, filename: str | None = None) -> bool:
    """Enable TunableOp. tune=False loads the committed results table
    (no runtime tuning); tune=True runs in tuning mode (slow first steps)
    and the caller should call save_tunableop() at the end."""
    if not torch.cuda.is_available():
        return False
    if os.environ.get("TL_TUNABLEOP", "1") == "0":
        return False
    fn = filename or TUNED_FILE
    try:
        t = torch.cuda.tunable
        t.enable(True)
        t.set_filename(fn)
        if tune:
            os.makedirs(os.path.dirname(fn), exist_ok=True)
            t.tuning_enable(True)
            t.set_max_tuning_duration(int(os.environ.get(
                "TL_TUNE_MS", "100")))
        else:
            t.tuning_enable(False)
            if os.path.exists(fn):
                t.read_file(fn)
            else:
                return False
        return True
    except Exception:
        return False


def save_tunableop(filename: str | None = None) -> None:
    # torch 2.10 writes the results file automatically at process exit
    # (set_filename + tuning_enable); write_file only exists on older
    # versions, so this is best-effort.
    fn = filename or TUNED_FILE
    wf = getattr(torch.cuda.tunable, "write_file", None)
    if wf is not None:
        wf(fn)
