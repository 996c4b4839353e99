#!/usr/bin/env bash
# Round-2 measurement battery: all BASELINE.md rows from bench.py flags.
set -x
B="python bench.py --steps 25 --warmup 5"
run() { echo "== $* =="; timeout 240 $B "$@" 2>/dev/null | tail -1; }
run
run --causal
run --causal --striped
run --config 3
run --kv-heads 2 --causal
run --seq-per-gpu 16384
run --seq-per-gpu 16384 --causal
run --seq-per-gpu 32768
run --seq-per-gpu 32768 --causal
run --seq-per-gpu 65536 --causal
run --d-head 128
run --d-head 128 --causal
run --fwd-only
run --d-head 128 --fwd-only
run --config 4
run --config 5
echo "== config4 heads at 16k =="
timeout 240 $B --heads 32 --kv-heads 4 --causal --seq-per-gpu 16384 2>/dev/null | tail -1
echo "== fwd-only 262144 =="
timeout 300 python bench.py --steps 5 --warmup 2 --fwd-only --seq-per-gpu 262144 2>/dev/null | tail -1
echo "== causal 131072 =="
timeout 300 python bench.py --steps 5 --warmup 2 --causal --seq-per-gpu 131072 2>/dev/null | tail -1
# MX-FP8 rows (beyond-reference serving path)
run --fp8
run --fp8 --seq-per-gpu 32768
run --fp8 --causal
run --fp8 --causal --seq-per-gpu 32768
run --fp8 --heads 32 --kv-heads 4 --causal
run --fp8 --d-head 128
run --config 5 --fp8
