"""hipBLASLt GEMM autotuning via PyTorch TunableOp.

A pre-tuned table for the flagship shapes (Llama-3-70B serving on gfx950)
ships in dynamo_amd/data/tunableop_gfx950.csv: +5% on the decode step with
zero tuning cost at startup. Call before the first GEMM.
"""
from __future__ import annotations

import os
import shutil
import tempfile


def enable_tunableop(tuning: bool = False) -> bool:
    table = os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "data", "tunableop_gfx950.csv")
    os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
    os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "1" if tuning else "0")
    if "PYTORCH_TUNABLEOP_FILENAME" not in os.environ and os.path.exists(table):
        # TunableOp writes (touches) its file; keep the vendored table
        # read-only by pointing at a scratch copy.
        scratch = os.path.join(tempfile.gettempdir(), "dynamo_amd_tunableop.csv")
        try:
            shutil.copyfile(table, scratch)
            os.environ["PYTORCH_TUNABLEOP_FILENAME"] = scratch
        except OSError:
            os.environ["PYTORCH_TUNABLEOP_FILENAME"] = table
    return True
