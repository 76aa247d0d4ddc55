#!/usr/bin/env python3
"""Flagship benchmark: dense push+pull goodput + round-trip latency on
the MI355X parameter server (BASELINE.json metric).

Layout (BytePS): N ranks = N GPUs, each a JOINT process (one worker +
one co-located server sharing the GPU); rank 0 additionally hosts the
scheduler thread. Metadata rides the shm rings; payloads move GPU<->GPU
over hipIpc/xGMI (cross-process even on one GPU).

One step = ZPush of every key (wait all) + ZPull of every key (wait
all), mirroring ps-lite tests/test_benchmark.cc PUSH_PULL mode.

Run standalone (1 GPU) or under `python -m torch.distributed.run
--nnodes=1 --nproc-per-node N bench.py --gpus N ...`.
"""

import argparse
import json
import os
import statistics
import sys
import threading
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import numpy as np


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--size-mb", type=float, default=64.0,
                   help="message size per key (headline: 64 MiB)")
    p.add_argument("--keys-per-server", type=int, default=8)
    p.add_argument("--op", choices=["assign", "sum"], default="assign")
    p.add_argument("--rtt-iters", type=int, default=100)
    p.add_argument("--no-rtt", action="store_true")
    p.add_argument("--pool-gb", type=int, default=0, help="0 = auto-size")
    p.add_argument("--smoke", action="store_true", help="tiny correctness run")
    return p.parse_args()


def main():
    args = parse_args()
    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    local = int(os.environ.get("LOCAL_RANK", rank))
    n = max(args.gpus, world)
    size = int(args.size_mb * (1 << 20))
    keys_per_server = args.keys_per_server
    if args.smoke:
        args.steps, args.warmup, keys_per_server = 2, 1, 2
        size = min(size, 1 << 20)

    total_keys = keys_per_server * n
    # pool: push bufs + pull bufs (worker) + store + temps (server), per GPU
    need_gb = max(2, int(4 * total_keys * size / (1 << 30)) + 2)
    pool_gb = args.pool_gb or need_gb

    master = os.environ.get("MASTER_ADDR", "127.0.0.1")
    master_port = int(os.environ.get("MASTER_PORT", "29400"))

    import ps_lite_amd as ps

    assert ps.gpu_count() > 0, "bench.py needs an MI355X (no GPU visible)"
    ps.setup_env(n, n, root_uri=master if world > 1 else "127.0.0.1",
                 root_port=master_port + 137,
                 XPS_DEV_ID=local, XPS_POOL_GB=pool_gb)

    sched = None
    if rank == 0:
        def sched_main():
            ps.start(role="scheduler", device=-1)
            ps.finalize(role="scheduler")  # blocks in the final barrier

        sched = threading.Thread(target=sched_main, daemon=True)
        sched.start()
    ps.start(role="joint", rank=rank, device=local)

    server = ps.KVServer(0)
    server.set_gpu_dense_handle(accumulate=(args.op == "sum"))
    worker = ps.KVWorker(0, 0)
    ps.barrier("worker", ps.WORKER_GROUP)

    cmd = 2 if args.op == "sum" else 1

    # keys: server s owns [s * 2^64/n, ...); key i of server s = base + i
    step = (1 << 64) // n
    keys = []
    for s in range(n):
        for i in range(keys_per_server):
            keys.append(s * step + i)
    key_arrs = [np.array([k], dtype=np.uint64) for k in keys]
    nfloats = size // 4
    lens = np.array([nfloats], dtype=np.int32)

    push_bufs, pull_bufs = [], []
    rng = np.random.default_rng(1234 + rank)
    fill = rng.standard_normal(nfloats, dtype=np.float32)
    for _ in keys:
        b = ps.pool_alloc(size)
        b.copy_from(fill)
        push_bufs.append(b)
        pull_bufs.append(ps.pool_alloc(size))

    keys_np = np.array(keys, dtype=np.uint64)
    push_ptrs = None
    pull_ptrs = None

    def one_round(bufs, pull):
        nonlocal push_ptrs, pull_ptrs
        if pull:
            if pull_ptrs is None:
                pull_ptrs = [b.ptr for b in bufs]
            worker.round(keys_np, pull_ptrs, size, local, cmd, True)
        else:
            if push_ptrs is None:
                push_ptrs = [b.ptr for b in bufs]
            worker.round(keys_np, push_ptrs, size, local, cmd, False)

    # warmup
    for _ in range(args.warmup):
        one_round(push_bufs, pull=False)
        one_round(pull_bufs, pull=True)

    if args.smoke:
        got = pull_bufs[0].to_numpy_f32()
        assert np.allclose(got, fill, atol=1e-6), "smoke: pulled values mismatch"
        print("SMOKE_OK: pulled values match pushed values")

    try:
        import torch
        has_torch = torch.cuda.is_available()
    except Exception:
        torch, has_torch = None, False

    def sync():
        if has_torch:
            torch.cuda.synchronize(local)
        else:
            ps.device_sync(local)

    ps.barrier("worker", ps.WORKER_GROUP)
    sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        one_round(push_bufs, pull=False)
        one_round(pull_bufs, pull=True)
    sync()
    t1 = time.perf_counter()
    ps.barrier("worker", ps.WORKER_GROUP)
    elapsed = t1 - t0

    # p50 round-trip: one blocking push+pull of a single key
    rtts = {}
    if not args.no_rtt and not args.smoke:
        for j, (sz_name, sz) in enumerate((("1mb", 1 << 20), ("64mb", 64 << 20))):
            if sz > size:
                continue
            # dedicated key so the store entry matches this message size
            rtt_key = np.array([10_000_000 + 100 * rank + j], dtype=np.uint64)
            samples = []
            l = np.array([sz // 4], dtype=np.int32)
            for _ in range(args.rtt_iters):
                a = time.perf_counter()
                worker.wait(worker.zpush_ptr(rtt_key, push_bufs[0].ptr, sz, local, l,
                                             cmd=cmd))
                worker.wait(worker.zpull_ptr(rtt_key, pull_bufs[0].ptr, sz, local, l,
                                             cmd=cmd))
                samples.append((time.perf_counter() - a) * 1e6)
            rtts[sz_name] = statistics.median(samples)

    # MAX elapsed over ranks (gloo reduction when multi-rank)
    if world > 1 and has_torch:
        import torch.distributed as dist
        dist.init_process_group("gloo", rank=rank, world_size=world)
        t = torch.tensor([elapsed], dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())
        for k in ("1mb", "64mb"):
            if k in rtts:
                t = torch.tensor([rtts[k]], dtype=torch.float64)
                dist.all_reduce(t, op=dist.ReduceOp.MAX)
                rtts[k] = float(t.item())
        dist.destroy_process_group()

    # bytes per worker per step: push size*keys + pull size*keys
    bytes_per_worker_step = 2.0 * total_keys * size
    gbs_per_worker = bytes_per_worker_step * args.steps / elapsed / 1e9
    total_gbs = gbs_per_worker * n
    ms_per_step = elapsed / args.steps * 1e3

    if rank == 0:
        out = {
            "metric": "push+pull GB/s per worker + p50 round-trip µs, 1 MB & 64 MB, at 1/2/4/8 GPUs",
            "value": round(total_gbs, 3),
            "unit": "GB/s",
            "n_gpus": n,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic",
            "config": {
                "model": "dense push+pull (ps-lite test_benchmark PUSH_PULL)",
                "global_batch": total_keys,
                "seq_len": size,
                "parallelism": f"byteps-joint x{n} (worker+server per GPU)",
                "msg_bytes": size,
                "keys_per_server": keys_per_server,
                "op": args.op,
                "gbs_per_worker": round(gbs_per_worker, 3),
                "p50_rtt_us": {k: round(v, 1) for k, v in rtts.items()},
            },
        }
        print(json.dumps(out))

    ps.finalize(role="joint")
    if sched is not None:
        sched.join()


if __name__ == "__main__":
    main()
