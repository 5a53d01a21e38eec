"""hipBLASLt GEMM algorithm selection via torch TunableOp.

The repo ships a pre-tuned cache for gfx950 (config/tunableop/
gfx950_<device>.csv, generated on an MI355X with
PYTORCH_TUNABLEOP_TUNING=1 — see benchmarks/tune_gemms.sh): fwd/dgrad
GEMMs of the BERT-Large shapes go from ~550-700 TF with default
hipBLASLt heuristics to 1.0-1.6 PF with the tuned algorithms.

Call :func:`enable` before the first GEMM. Tuning stays OFF by default
(the cache is read-only); re-tune with BPA_TUNE=1.
"""

from __future__ import annotations

import os

_REPO = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def enable() -> bool:
    cache = os.path.join(_REPO, "config", "tunableop", "gfx950_.csv")
    base_dir = os.path.dirname(cache)
    if not os.path.isdir(base_dir):
        return False
    os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
    os.environ.setdefault(
        "PYTORCH_TUNABLEOP_TUNING", "1" if os.environ.get("BPA_TUNE") == "1" else "0"
    )
    # torch appends the device index before .csv: gfx950_.csv -> gfx950_0.csv
    os.environ.setdefault(
        "PYTORCH_TUNABLEOP_FILENAME", os.path.join(base_dir, "gfx950_.csv")
    )
    return True
