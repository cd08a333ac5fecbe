#!/usr/bin/env bash
# Full benchmark sweep on one MI355X box — every table row in
# docs/BENCHMARKS.md in a single session. Writes a plain-text log that can
# be committed under profiles/ verbatim (the round-1 run of this sweep is
# profiles/r01_final_numbers.txt).
#
# Usage:  bash scripts/run_all_benches.sh [outfile]
set -u
out="${1:-gpurun_out/all_benches.txt}"
mkdir -p "$(dirname "$out")"
cd "$(dirname "$0")/.."
export HSA_ENABLE_IPC_MODE_LEGACY=0

run() {
  echo "=== $* ===" | tee -a "$out"
  timeout 600 "$@" 2>&1 | tail -25 | tee -a "$out"
  echo | tee -a "$out"
}

: > "$out"
echo "# all_benches $(date -u +%Y-%m-%dT%H:%M:%SZ) $(hostname)" | tee -a "$out"

run python bench.py --steps 256 --warmup 32                       # headline 8B
run python bench.py --steps 256 --warmup 32 --batch 8             # 8 sessions
run python bench.py --steps 256 --warmup 32 --weights-fp8         # fp8 mode
run python bench.py --model llama3-70b --prompt-len 256 --steps 48 --warmup 8
run python bench.py --model llama3-1b --steps 256 --warmup 32
run python scripts/bench_spec.py llama3-8b 256
run python scripts/bench_gemv.py
run python scripts/bench_sessions.py
run env SESS_N=16 python scripts/bench_sessions.py
run python scripts/bench_agent_turn.py
run env CORPUS_N=1000000 python scripts/bench_memdir.py

echo "sweep done -> $out"
