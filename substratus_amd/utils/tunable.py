"""Pre-tuned hipBLASLt/rocBLAS GEMM solution tables (TunableOp).

PyTorch-ROCm's TunableOp benchmarks every hipBLASLt/rocBLAS solution for
a GEMM shape and records the winner. The library heuristic leaves real
bandwidth on the decode projections of the larger models — measured on
llama2-70b decode (M=512): o-proj 1.4 -> 1.94 TB/s, down-proj
1.5 -> 1.97 TB/s, whole-model serving 5,458 -> 5,784 tok/s.

Tables are produced offline on a target box (see docs/PERF.md; run any
bench with PYTORCH_TUNABLEOP_ENABLED=1 PYTORCH_TUNABLEOP_TUNING=1 and an
eager decode path) and shipped under substratus_amd/data/tunables/<arch>.csv (repo-root tunables/ overrides). At runtime we
load them READ-ONLY: tuning stays disabled, untuned shapes keep the
heuristic default, and a version/arch validator mismatch simply skips
the table. Explicitly setting PYTORCH_TUNABLEOP_ENABLED (either value)
bypasses this loader entirely; SUB_TUNABLEOP=0 disables it.

No reference counterpart (reference GEMMs ride cuBLAS heuristics); this
is MI355X-specific plumbing.
"""

from __future__ import annotations

import os

import torch

_loaded: bool | None = None


def tunables_dir() -> str:
    """Canonical table location is package data (ships in wheels/sdists);
    a repo-root tunables/ directory, if present, overrides it (operational
    knob for box-local retuning without touching the package)."""
    pkg = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    root = os.path.dirname(pkg)
    override = os.path.join(root, "tunables")
    if os.path.isdir(override):
        return override
    return os.path.join(pkg, "data", "tunables")


def load_tunables() -> bool:
    """Enable TunableOp in read-only mode with the shipped table for this
    GPU arch, if one exists. Idempotent; safe to call from every rank."""
    global _loaded
    if _loaded is not None:
        return _loaded
    _loaded = False
    if not torch.cuda.is_available():
        return False
    if "PYTORCH_TUNABLEOP_ENABLED" in os.environ:
        return False          # user drives TunableOp explicitly
    if os.environ.get("SUB_TUNABLEOP", "1") == "0":
        return False
    arch = torch.cuda.get_device_properties(0).gcnArchName.split(":")[0]
    path = os.path.join(tunables_dir(), f"{arch}.csv")
    if not os.path.exists(path):
        return False
    try:
        import torch.cuda.tunable as tun
        tun.enable(True)
        tun.tuning_enable(False)
        tun.read_file(path)
        _loaded = True
    except Exception:           # corrupt table / validator mismatch
        return False
    return True
