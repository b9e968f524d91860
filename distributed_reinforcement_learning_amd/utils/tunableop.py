"""Enable PyTorch TunableOp with the shipped gfx950 GEMM tuning table.

hipBLASLt algorithm selection tuned offline on an MI355X gives +3.5% on the
flagship step (profiles/tunableop_gfx950.csv; 1.40 -> 1.35 ms/step measured).
Must run BEFORE the first GEMM; importing this module's ``enable()`` at
process start is enough. Opt out with DRLA_NO_TUNABLEOP=1. Re-tune with
PYTORCH_TUNABLEOP_TUNING=1 PYTORCH_TUNABLEOP_FILENAME=<path> python bench.py.
"""

from __future__ import annotations

import os


def enable() -> bool:
    if os.environ.get("DRLA_NO_TUNABLEOP"):
        return False
    if "PYTORCH_TUNABLEOP_ENABLED" in os.environ:
        return True  # user controls it
    here = os.path.dirname(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    table = os.path.join(here, "profiles", "tunableop_gfx950_0.csv")
    if not os.path.exists(table):
        return False
    os.environ["PYTORCH_TUNABLEOP_ENABLED"] = "1"
    os.environ["PYTORCH_TUNABLEOP_TUNING"] = "0"  # use the table, no search
    # %d expands to the device ordinal; identical GPUs share one tuning,
    # shipped as tunableop_gfx950_{0..7}.csv
    os.environ["PYTORCH_TUNABLEOP_FILENAME"] = table.replace(
        "_0.csv", "_%d.csv")
    return True
