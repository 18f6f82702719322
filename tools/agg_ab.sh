#!/bin/bash
# Same-box A/B sweep of the two-phase agg pipeline knobs over the 1B-row
# north-star workload (box-to-box variance is ~2x the effects being measured,
# so variants must run interleaved on ONE box — DESIGN.md §8 variance note).
# Usage: bash tools/agg_ab.sh [rows] ; writes one line per variant.
set -u
ROWS=${1:-1000000000}
cd "$(dirname "$0")/.."
run() {
  local tag="$1"; shift
  local out
  out=$(env "$@" timeout 300 python3 bench.py --rows "$ROWS" --steps 3 \
        --warmup 1 --skip-cpu-baseline 2>/dev/null | tail -1)
  local ms
  ms=$(echo "$out" | python3 -c 'import json,sys
try: print(json.loads(sys.stdin.read())["ms_per_step"])
except Exception: print("FAIL")')
  echo "AB $tag ms_per_step=$ms"
}

# round-1 baseline geometry first, then the candidates, then baseline again
# (drift check)
run "r1base  b256  g9 c64 "  AURON_AGG2_BLOCK=256  AURON_AGG2_GRID_LOG2=9 AURON_AGG2_CHUNK_M=64
run "occ     b1024 g9 c64 "  AURON_AGG2_BLOCK=1024 AURON_AGG2_GRID_LOG2=9 AURON_AGG2_CHUNK_M=64
run "chunk   b1024 g9 c256"  AURON_AGG2_BLOCK=1024 AURON_AGG2_GRID_LOG2=9 AURON_AGG2_CHUNK_M=256
run "wrloc8  b1024 g8 c256"  AURON_AGG2_BLOCK=1024 AURON_AGG2_GRID_LOG2=8 AURON_AGG2_CHUNK_M=256
run "wrloc7  b1024 g7 c256"  AURON_AGG2_BLOCK=1024 AURON_AGG2_GRID_LOG2=7 AURON_AGG2_CHUNK_M=256
run "split   b1024 g8 c256s" AURON_AGG2_BLOCK=1024 AURON_AGG2_GRID_LOG2=8 AURON_AGG2_CHUNK_M=256 AURON_AGG2_SPLIT=1
run "r1drift b256  g9 c64 "  AURON_AGG2_BLOCK=256  AURON_AGG2_GRID_LOG2=9 AURON_AGG2_CHUNK_M=64
