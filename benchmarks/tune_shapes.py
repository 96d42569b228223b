"""Generate TunableOp entries for a model's serving GEMM shapes.

Pins hipBLASLt/rocBLAS algorithm selection (the round-1 Mixtral 47<->51
ms/step variance was traced to algorithm re-selection sensitivity across
allocation layouts): run each hot GEMM shape once with
PYTORCH_TUNABLEOP_TUNING=1, then merge the resulting entries into
dynamo_amd/data/tunableop_gfx950.csv (committed; loaded read-only by
enable_tunableop at engine start).

GPU box:
  python benchmarks/tune_shapes.py --model mixtral-8x7b --out gpurun_out/tune.csv
then merge locally: python benchmarks/tune_shapes.py --merge gpurun_out/tune.csv
"""
from __future__ import annotations

import argparse
import os
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def serving_shapes(model: str, tp: int = 1):
    """(out_features, in_features) weight shapes of the model's hot GEMMs;
    batch rows swept over decode (1..16) and prefill (8192) regimes."""
    from dynamo_amd.engine.config import PRESETS
    mc = PRESETS[model]
    hq = mc.num_q_heads // tp
    hkv = max(1, mc.num_kv_heads // tp)
    D = mc.hidden_size
    I = mc.intermediate_size // tp
    shapes = [
        ((hq + 2 * hkv) * mc.head_dim, D),   # qkv
        (D, hq * mc.head_dim),               # o proj
        (mc.vocab_size, D),                  # lm head
    ]
    if mc.num_experts:
        shapes += [(2 * I, D), (D, I)]       # per-expert gate_up / down
        shapes += [(mc.num_experts, D)]      # router
    else:
        shapes += [(2 * I, D), (D, I)]       # fused gate_up / down
    return shapes


def run_tuning(model: str, out: str, tp: int = 1):
    os.environ["PYTORCH_TUNABLEOP_ENABLED"] = "1"
    os.environ["PYTORCH_TUNABLEOP_TUNING"] = "1"
    os.environ["PYTORCH_TUNABLEOP_FILENAME"] = out
    import torch
    dev = "cuda:0"
    from dynamo_amd.engine.config import PRESETS
    vocab = PRESETS[model].vocab_size
    rows = [1, 2, 4, 8, 16, 32, 64, 121, 8192]
    for (o, i) in serving_shapes(model, tp):
        w = torch.randn(o, i, dtype=torch.bfloat16, device=dev)
        # the lm head only ever sees <= max_num_seqs rows (last-token
        # sampling); prefill GEMMs see up to max_batched_tokens
        for m in (r for r in rows if not (o == vocab and r > 64)):
            x = torch.randn(m, i, dtype=torch.bfloat16, device=dev)
            torch.nn.functional.linear(x, w)
        del w
        torch.cuda.empty_cache()
    torch.cuda.synchronize()
    # TunableOp appends a per-GPU suffix and dumps at interpreter exit
    print(f"tuned {model} shapes -> {out}0 (written at exit)")


def merge(src: str):
    table = os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "dynamo_amd", "data",
        "tunableop_gfx950.csv")
    have = set()
    lines = open(table).read().splitlines()
    for ln in lines:
        if ln.startswith("Gemm"):
            have.add(ln.split(",")[1])
    added = 0
    with open(table, "a") as fh:
        for ln in open(src).read().splitlines():
            if ln.startswith("Gemm") and ln.split(",")[1] not in have:
                fh.write(ln + "\n")
                have.add(ln.split(",")[1])
                added += 1
    print(f"merged {added} new entries into {table}")


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="mixtral-8x7b")
    p.add_argument("--tp", type=int, default=1)
    p.add_argument("--out", default="gpurun_out/tune.csv")
    p.add_argument("--merge", default=None)
    a = p.parse_args()
    if a.merge:
        merge(a.merge)
    else:
        os.makedirs(os.path.dirname(a.out) or ".", exist_ok=True)
        run_tuning(a.model, a.out, a.tp)
