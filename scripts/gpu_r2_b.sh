#!/bin/bash
# Round-2 GPU run B: measure the lane/lock/fused work.
set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
R=gpurun_out

timeout 900 python -m pytest tests -m gpu -x -q 2>&1 | tail -6 > $R/pytest_gpu_b.txt

# per-key 1 MB: lanes=2 (default) vs lanes=1, plus timing table
timeout 300 python bench.py --mode dense --size-mb 1 --keys-per-server 40 --steps 40 --warmup 10 > $R/d1mb_l2.json 2> $R/d1mb_l2.err
XPS_STREAMS_PER_PEER=1 timeout 300 python bench.py --mode dense --size-mb 1 --keys-per-server 40 --steps 40 --warmup 10 > $R/d1mb_l1.json 2> $R/d1mb_l1.err
XPS_TIMING=1 timeout 300 python bench.py --mode dense --size-mb 1 --keys-per-server 40 --steps 40 --warmup 10 --no-rtt > $R/d1mb_l2_t.json 2> $R/d1mb_l2_t.err

# dense 64 MB: lanes=2 vs 1
timeout 300 python bench.py --steps 15 --warmup 4 > $R/d64_l2.json 2> $R/d64_l2.err
XPS_STREAMS_PER_PEER=1 timeout 300 python bench.py --steps 15 --warmup 4 > $R/d64_l1.json 2> $R/d64_l1.err

# sparse: fused vs unfused
timeout 300 python bench.py --mode sparse --steps 100 --warmup 20 > $R/sparse_fused.json 2> $R/sparse_fused.err
timeout 300 python bench.py --mode sparse --steps 100 --warmup 20 --no-fused > $R/sparse_unfused.json 2> $R/sparse_unfused.err

# rn50 batched after lane changes
timeout 300 python bench.py --mode rn50 --steps 40 --warmup 10 > $R/rn50_b.json 2> $R/rn50_b.err
echo DONE
