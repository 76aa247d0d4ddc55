#!/bin/bash
# Round-2 GPU validation A: full gpu test suite, rn50 batched vs per-key,
# dense + sparse sanity, rocprof of the batched rn50 step.
set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
R=gpurun_out

timeout 900 python -m pytest tests -m gpu -x -q 2>&1 | tail -8 > $R/pytest_gpu.txt
echo "pytest rc=$?" >> $R/pytest_gpu.txt

timeout 300 python bench.py --mode rn50 --steps 30 --warmup 8 > $R/rn50_batched.json 2> $R/rn50_batched.err
timeout 300 python bench.py --mode rn50 --steps 20 --warmup 5 --per-key > $R/rn50_perkey.json 2> $R/rn50_perkey.err
timeout 300 python bench.py --steps 15 --warmup 4 > $R/dense64.json 2> $R/dense64.err
timeout 300 python bench.py --mode dense --size-mb 1 --keys-per-server 40 --steps 30 --warmup 8 > $R/dense1mb.json 2> $R/dense1mb.err
timeout 300 python bench.py --mode sparse --steps 60 --warmup 15 > $R/sparse.json 2> $R/sparse.err

# per-message software overhead table for the 1 MB per-key config
XPS_TIMING=1 timeout 300 python bench.py --mode dense --size-mb 1 --keys-per-server 40 --steps 30 --warmup 8 --no-rtt > $R/dense1mb_timing.json 2> $R/dense1mb_timing.err

cd /tmp && export TMPDIR=/tmp
timeout 420 rocprofv3 --kernel-trace --stats -d "$GRAFT_REPO_ROOT/$R/prof_rn50" -o rn50 -- \
  python "$GRAFT_REPO_ROOT/bench.py" --mode rn50 --steps 10 --warmup 3 --no-rtt \
  > "$GRAFT_REPO_ROOT/$R/rn50_prof.json" 2> "$GRAFT_REPO_ROOT/$R/rn50_prof.err"
echo DONE
