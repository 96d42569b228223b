"""Decompose the flagship decode step into graph-replay / sampling / D2H /
host-bookkeeping components (run on a GPU box):

  python benchmarks/step_anatomy.py [--model llama-3-70b] [--conc 16]

The full-step time should match bench.py's ms_per_step; the difference
between full and (graph + sample + d2h) is host bookkeeping in
LLMEngine._fast_decode_step.
"""
import argparse
import json
import random
import sys
import time

sys.path.insert(0, ".")

import torch

from dynamo_amd.engine import EngineConfig, LLMEngine, SamplingParams
from dynamo_amd.engine.config import PRESETS
from dynamo_amd.utils import enable_tunableop


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama-3-70b")
    ap.add_argument("--conc", type=int, default=16)
    ap.add_argument("--isl", type=int, default=8192)
    ap.add_argument("--iters", type=int, default=40)
    a = ap.parse_args()
    enable_tunableop(tuning=False)
    cfg = EngineConfig(model=PRESETS[a.model], device="cuda:0",
                       max_num_seqs=max(32, a.conc), max_model_len=16384,
                       page_size=64)
    eng = LLMEngine(cfg, seed=0)
    rng = random.Random(0)
    for i in range(a.conc):
        eng.add_request(f"r{i}",
                        [rng.randrange(cfg.model.vocab_size)
                         for _ in range(a.isl)],
                        SamplingParams(max_tokens=1024, temperature=0.0,
                                       ignore_eos=True))
    while any(not r.output_tokens for r in eng.requests.values()):
        eng.step()
    torch.cuda.synchronize()
    for _ in range(8):
        eng.step()
    torch.cuda.synchronize()

    N = a.iters
    t0 = time.monotonic()
    for _ in range(N):
        eng.step()
    torch.cuda.synchronize()
    full = (time.monotonic() - t0) / N

    gr = eng.graph_runner
    last = eng._last_sampled
    logits = gr.step(last)
    torch.cuda.synchronize()
    t0 = time.monotonic()
    for _ in range(N):
        logits = gr.step(last)
    torch.cuda.synchronize()
    graph = (time.monotonic() - t0) / N

    from dynamo_amd.engine.sampling import sample_tokens
    running = eng.scheduler.running
    torch.cuda.synchronize()
    t0 = time.monotonic()
    for i in range(N):
        s = sample_tokens(logits, running, i)
    torch.cuda.synchronize()
    samp = (time.monotonic() - t0) / N

    t0 = time.monotonic()
    for _ in range(N):
        s.cpu().tolist()
    d2h = (time.monotonic() - t0) / N

    print(json.dumps({
        "model": a.model, "conc": a.conc,
        "full_step_ms": round(full * 1e3, 3),
        "graph_replay_ms": round(graph * 1e3, 3),
        "sampling_ms": round(samp * 1e3, 3),
        "d2h_ms": round(d2h * 1e3, 3),
        "host_other_ms": round((full - graph - samp - d2h) * 1e3, 3),
    }))


if __name__ == "__main__":
    main()
