#!/bin/bash
# Round-2 profiling sweep (run on the GPU box via gpurun).
# Collects: kernel-trace stats for mixed/flagship/stencil, PMC FETCH/WRITE
# passes (separate runs, per the MI355X guide), and builds the PMC traffic
# manifest keyed by live kernel hashes.
set -x
R=${GRAFT_REPO_ROOT:-/root/repo}
O=$R/gpurun_out
mkdir -p "$O"
cd /tmp && export TMPDIR=/tmp

# 1. plain benches first (host-cost + value with the staged cache)
( cd "$R" && timeout 300 python bench.py --workload mixed --steps 30 \
    --no-cpu-baseline > "$O/r02_mixed_bench.json" 2> "$O/r02_mixed_bench.err" )
( cd "$R" && timeout 300 python bench.py --workload stencil --steps 30 \
    --no-cpu-baseline > "$O/r02_stencil_bench.json" 2>/dev/null )

# 2. kernel-trace + stats
timeout 420 rocprofv3 --kernel-trace --stats -d "$O/prof_mixed" -o mixed -- \
  bash -c "cd $R && python bench.py --workload mixed --steps 6 --warmup 2 --no-cpu-baseline" \
  > "$O/r02_prof_mixed.log" 2>&1
timeout 420 rocprofv3 --kernel-trace --stats -d "$O/prof_flag" -o flag -- \
  bash -c "cd $R && python bench.py --steps 4 --warmup 1 --no-cpu-baseline" \
  > "$O/r02_prof_flag.log" 2>&1
timeout 420 rocprofv3 --kernel-trace --stats -d "$O/prof_sten" -o sten -- \
  bash -c "cd $R && python bench.py --workload stencil --steps 8 --warmup 2 --no-cpu-baseline" \
  > "$O/r02_prof_sten.log" 2>&1

# 3. PMC passes (separate runs; FETCH and WRITE separately)
for ctr in FETCH_SIZE WRITE_SIZE; do
  timeout 420 rocprofv3 --pmc $ctr -d "$O/pmc_flag_$ctr" -o f -- \
    bash -c "cd $R && python bench.py --steps 3 --warmup 1 --no-cpu-baseline" \
    > "$O/r02_pmc_flag_$ctr.log" 2>&1
  timeout 420 rocprofv3 --pmc $ctr -d "$O/pmc_sten_$ctr" -o s -- \
    bash -c "cd $R && python bench.py --workload stencil --steps 6 --warmup 2 --no-cpu-baseline" \
    > "$O/r02_pmc_sten_$ctr.log" 2>&1
  timeout 420 rocprofv3 --pmc $ctr -d "$O/pmc_mixed_$ctr" -o m -- \
    bash -c "cd $R && python bench.py --workload mixed --steps 6 --warmup 2 --no-cpu-baseline" \
    > "$O/r02_pmc_mixed_$ctr.log" 2>&1
done

# 4. manifest from the flagship PMC (validated by kernel hash at bench time)
FC=$(ls "$O"/pmc_flag_FETCH_SIZE/*counter*.csv 2>/dev/null | head -1)
WC=$(ls "$O"/pmc_flag_WRITE_SIZE/*counter*.csv 2>/dev/null | head -1)
[ -z "$FC" ] && FC=$(find "$O/pmc_flag_FETCH_SIZE" -name '*.csv' | head -1)
[ -z "$WC" ] && WC=$(find "$O/pmc_flag_WRITE_SIZE" -name '*.csv' | head -1)
( cd "$R" && python tools/make_pmc_manifest.py --fetch "$FC" --write "$WC" \
    --workload flagship --elems 1000000000 --world 1 \
    --out "$O/pmc_manifest.json" ) > "$O/r02_manifest.log" 2>&1

ls -la "$O" | head -40
