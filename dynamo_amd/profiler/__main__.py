import argparse
import json

from .profile_sla import run_profile


def main():
    ap = argparse.ArgumentParser("dynamo_amd.profiler")
    ap.add_argument("--url", default="http://127.0.0.1:8000")
    ap.add_argument("--model", default="")
    ap.add_argument("--isl", type=int, default=2048)
    ap.add_argument("--osl", type=int, default=64)
    ap.add_argument("--concurrencies", default="1,2,4,8")
    ap.add_argument("--requests-per-level", type=int, default=8)
    ap.add_argument("--itl-slo-ms", type=float, default=25.0)
    ap.add_argument("--ttft-slo-s", type=float, default=2.0)
    ap.add_argument("--vocab", type=int, default=512)
    ap.add_argument("--out", default=None)
    a = ap.parse_args()
    res = run_profile(a.url, a.model, a.isl, a.osl,
                      [int(c) for c in a.concurrencies.split(",")],
                      a.requests_per_level, a.itl_slo_ms, a.ttft_slo_s,
                      vocab=a.vocab, out=a.out)
    print(json.dumps(res["perf_model"] | {"meets_slo": res["meets_slo"]}))


if __name__ == "__main__":
    main()
