#!/bin/bash
# BASELINE config #1: OPT-125m aggregated on CPU (no GPU; plumbing).
MODEL=opt-125m DEVICE=cpu exec bash "$(dirname "$0")/llama8b-agg.sh" "$@"
