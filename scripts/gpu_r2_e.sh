#!/bin/bash
set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
R=gpurun_out
timeout 900 python -m pytest tests -m gpu -x -q 2>&1 | tail -6 > $R/pytest_gpu_e.txt
timeout 300 python bench.py --steps 15 --warmup 4 > $R/e_d64.json 2> $R/e_d64.err
timeout 300 python bench.py --mode dense --size-mb 1 --keys-per-server 40 --steps 60 --warmup 10 > $R/e_d1mb.json 2> $R/e_d1mb.err
XPS_TIMING=1 timeout 300 python bench.py --mode dense --size-mb 1 --keys-per-server 40 --steps 60 --warmup 10 --no-rtt > $R/e_d1mb_t.json 2> $R/e_d1mb_t.err
timeout 300 python bench.py --batch-keys --size-mb 1 --keys-per-server 40 --steps 60 --warmup 10 > $R/e_d1mb_batch.json 2> $R/e_d1mb_batch.err
timeout 300 python bench.py --mode sparse --steps 200 --warmup 30 > $R/e_sparse.json 2> $R/e_sparse.err
timeout 300 python bench.py --mode rn50 --steps 60 --warmup 10 > $R/e_rn50.json 2> $R/e_rn50.err
timeout 580 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 --master-addr 127.0.0.1 --master-port 29411 bench.py --gpus 2 --steps 10 --warmup 3 --keys-per-server 8 --no-rtt > $R/e_d64_2proc.json 2> $R/e_d64_2proc.err
echo DONE
