"""hipBLASLt GEMM algorithm selection via ROCm TunableOp.

The flagship step is ~45% plain hipBLASLt GEMMs; the library's default
heuristic does not always pick the fastest solution for a given shape on
gfx950. TunableOp benchmarks every available solution per GEMM shape once
and records the winner. We tune once on an MI355X (``DALLE_AMD_TUNE=1``)
and ship the resulting table; every later run loads it and gets the tuned
kernels with zero warmup cost. A table recorded on a different
hipBLASLt/ROCm build fails TunableOp's validator and is ignored — runs
then simply use the default heuristic.
"""

import os
from pathlib import Path

_TABLE = Path(__file__).resolve().parents[2] / 'tuned' / 'tunableop_gfx950.csv'


def maybe_enable_tunableop(table_path=None):
    """Enable TunableOp if a tuned table is shipped or tuning is requested.

    Call before the first GEMM. Returns True when enabled. Set
    ``DALLE_AMD_TUNE=1`` to (re)tune and write/extend the table in place.
    """
    import torch

    if not torch.cuda.is_available():
        return False
    table = Path(table_path) if table_path else _TABLE
    tune = os.environ.get('DALLE_AMD_TUNE') == '1'
    if not tune and not table.exists():
        return False

    import torch.cuda.tunable as tunable
    table.parent.mkdir(parents=True, exist_ok=True)
    tunable.set_filename(str(table))
    tunable.enable(True)
    tunable.tuning_enable(tune)
    return True
