#!/bin/bash
# Round-2 run G: SCALE rehearsal (multi-rank on one GPU) + PMC counters.
set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
R=gpurun_out
TR="python -m torch.distributed.run --nnodes=1 --master-addr 127.0.0.1"

# 8 joint procs sharing one GPU: the 17-node bootstrap + cross-process
# one-sided pushes + rings, the closest 1-GPU analog of the driver's
# 8-GPU run (small pools: 8 x ~6 GB)
timeout 580 $TR --nproc-per-node 8 --master-port 29421 bench.py --gpus 8 --steps 5 --warmup 2 --keys-per-server 2 --no-rtt > $R/g_d64_8proc.json 2> $R/g_d64_8proc.err
echo "8proc rc=$?" >> $R/g_d64_8proc.err

# 4-rank bucketed reduce rounds (multi-worker group protocol on GPU)
timeout 420 $TR --nproc-per-node 4 --master-port 29423 bench.py --gpus 4 --mode rn50 --steps 10 --warmup 3 --no-rtt > $R/g_rn50_4proc.json 2> $R/g_rn50_4proc.err
echo "rn50x4 rc=$?" >> $R/g_rn50_4proc.err

# 4-rank fused sparse (atomic scatters, multi-worker)
timeout 420 $TR --nproc-per-node 4 --master-port 29425 bench.py --gpus 4 --mode sparse --steps 30 --warmup 10 --no-rtt > $R/g_sparse_4proc.json 2> $R/g_sparse_4proc.err
echo "sparsex4 rc=$?" >> $R/g_sparse_4proc.err

# PMC: HBM read/write bytes of the assign kernel (own run, counters only)
cd /tmp && export TMPDIR=/tmp
rocprofv3 --list-avail 2>/dev/null | grep -iE "FETCH_SIZE|WRITE_SIZE" | head -6 > "$GRAFT_REPO_ROOT/$R/pmc_avail.txt"
timeout 420 rocprofv3 --pmc FETCH_SIZE WRITE_SIZE --kernel-trace -d "$GRAFT_REPO_ROOT/$R/pmc_dense" -o pmc -- \
  python "$GRAFT_REPO_ROOT/bench.py" --steps 3 --warmup 1 --keys-per-server 4 --no-rtt \
  > "$GRAFT_REPO_ROOT/$R/g_pmc.json" 2> "$GRAFT_REPO_ROOT/$R/g_pmc.err"
echo DONE
