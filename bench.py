#!/usr/bin/env python3
"""Flagship benchmark: push+pull goodput + round-trip latency on the
MI355X parameter server (BASELINE.json metric).

Layout (BytePS): N ranks = N GPUs, each a JOINT process (one worker +
one co-located server sharing the GPU); rank 0 additionally hosts the
scheduler thread. Metadata rides the shm rings; payloads move GPU<->GPU
over hipIpc/xGMI (cross-process even on one GPU).

Modes:
  dense (default, config #2/#3): one step = ZPush of every key (wait
      all) + ZPull of every key (wait all) — ps-lite
      tests/test_benchmark.cc PUSH_PULL semantics.
  rn50 (config #4): ResNet-50 gradient buckets, each key's pull issued
      right behind its push (overlapped).
  sparse (config #5): 1M-row fp32 embedding, 8K hot keys/step,
      server-side HIP scatter/gather.
  --cpu (config #1): localhost TCP van, host buffers, CPU default
      handler — the plumbing baseline; runs without a GPU.

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
    p.add_argument("--mode", choices=["dense", "rn50", "sparse"], default="dense")
    p.add_argument("--cpu", action="store_true", help="config #1: TCP van, host buffers")
    p.add_argument("--size-mb", type=float, default=64.0,
                   help="dense message size per key (headline: 64 MiB)")
    p.add_argument("--keys-per-server", type=int, default=16,
                   help="with --gpus 8 this is exact; the TOTAL key count is "
                        "keys_per_server*8 and stays FIXED as N varies (the "
                        "model does not grow with the cluster -> honest weak "
                        "scaling of per-worker work)")
    p.add_argument("--op", choices=["assign", "sum"], default="assign")
    p.add_argument("--dtype", choices=["fp32", "bf16"], default="fp32",
                   help="payload dtype for the dense sum/reduce kernels "
                        "(bf16 halves bytes moved; headline stays fp32)")
    p.add_argument("--batch-keys", action="store_true",
                   help="dense: one multi-key message per server per round. "
                        "Meant for SMALL keys (e.g. --size-mb 1): the batched "
                        "buffer must stay under the <2 GiB hipIpc slab limit "
                        "or it falls off the zero-copy path")
    p.add_argument("--per-key", action="store_true",
                   help="rn50: legacy one-message-per-bucket reduce rounds "
                        "(default is one multi-key message per server per round)")
    p.add_argument("--no-overlap", action="store_true",
                   help="dense: separate push and pull phases (the reference "
                        "PUSH_PULL loop issues ZPush+ZPull per key together, "
                        "so overlap is the default)")
    p.add_argument("--hot-keys", type=int, default=8192, help="sparse keys per step")
    p.add_argument("--no-fused", action="store_true",
                   help="sparse: separate push and pull trips (default is the "
                        "fused ZPushPull single-trip round)")
    p.add_argument("--emb-rows", type=int, default=1 << 20)
    p.add_argument("--emb-width", type=int, default=64)
    p.add_argument("--rtt-iters", type=int, default=100)
    p.add_argument("--no-rtt", action="store_true")
    p.add_argument("--pool-gb", type=int, default=0, help="0 = auto-size")
    p.add_argument("--smoke", action="store_true", help="tiny correctness run")
    return p.parse_args()


class Cluster:
    def __init__(self, ps, n, rank, local, device, root_uri, root_port, pool_gb):
        self.ps = ps
        ps.setup_env(n, n, root_uri=root_uri, root_port=root_port,
                     XPS_DEV_ID=device, XPS_POOL_GB=pool_gb)
        self.sched = None
        if rank == 0:
            def sched_main():
                ps.start(role="scheduler", device=-1)
                ps.finalize(role="scheduler")  # blocks in the final barrier

            self.sched = threading.Thread(target=sched_main, daemon=True)
            self.sched.start()
        ps.start(role="joint", rank=rank, device=device)

    def finish(self):
        self.ps.finalize(role="joint")
        if self.sched is not None:
            self.sched.join(timeout=60)


def reduce_max(values, rank, world):
    """MAX over ranks via gloo (driver launches us under torchrun)."""
    if world <= 1:
        return values
    import torch
    import torch.distributed as dist

    if not dist.is_initialized():
        # gloo prints "[Gloo] Rank ... connected ..." to C-level stdout;
        # the driver parses our stdout for ONE json line — route the
        # noise to stderr during init
        sys.stdout.flush()
        saved = os.dup(1)
        try:
            os.dup2(2, 1)
            dist.init_process_group("gloo", rank=rank, world_size=world)
        finally:
            os.dup2(saved, 1)
            os.close(saved)
    t = torch.tensor(values, dtype=torch.float64)
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    return t.tolist()


def main():
    args = parse_args()
    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    local = int(os.environ.get("LOCAL_RANK", rank))
    n = max(args.gpus, world)
    if args.smoke:
        args.steps, args.warmup = 2, 1
        args.size_mb = min(args.size_mb, 1.0)
        args.keys_per_server = 2
        args.hot_keys = 256

    master = os.environ.get("MASTER_ADDR", "127.0.0.1")
    master_port = int(os.environ.get("MASTER_PORT", "29400"))

    import ps_lite_amd as ps
    from ps_lite_amd.models import EmbeddingSpec, resnet50_grad_buckets

    if args.cpu:
        assert args.mode in ("dense", "rn50"), "--cpu supports dense and rn50"
        device = -1
    else:
        count = ps.gpu_count()
        assert count > 0, "bench.py needs an MI355X (no GPU visible); use --cpu for config #1"
        device = local % count  # fallback lets N ranks share fewer GPUs in dev runs
        if device != local and rank == 0:
            print(f"# note: {n} ranks on {count} GPU(s); sharing devices", file=sys.stderr)

    # ---- workload geometry ------------------------------------------------
    step_range = (1 << 64) // n
    cmd = 2 if args.op == "sum" else 1
    if args.mode == "dense":
        size = int(args.size_mb * (1 << 20))
        # FIXED model: keys_per_server*8 keys total, sharded over the n
        # servers (each worker pushes+pulls the whole model every step,
        # BytePS semantics) — per-worker work does not grow with n
        total_keys = args.keys_per_server * 8
        assert total_keys % n == 0
        msg_sizes = [size] * total_keys
        keys = [(j % n) * step_range + (j // n) for j in range(total_keys)]
        model_name = "dense push+pull (ps-lite test_benchmark PUSH_PULL)"
        overlap_pull = not args.no_overlap and not args.cpu and not args.batch_keys
    elif args.mode == "rn50":
        buckets = resnet50_grad_buckets()
        if args.smoke:
            buckets = buckets[:8]
        msg_sizes = [(b + 255) & ~255 for b in buckets]
        keys = [(i % n) * step_range + (i // n) + 1 for i in range(len(buckets))]
        cmd = 2 if args.op == "sum" else 1
        model_name = f"resnet50 grad buckets x{len(buckets)} (BytePS layout)"
        overlap_pull = True
    else:  # sparse
        spec = EmbeddingSpec(rows=args.emb_rows, width=args.emb_width)
        model_name = f"sparse embedding {spec.rows}x{spec.width}, {args.hot_keys} hot keys/step"
        msg_sizes = []
        overlap_pull = False

    total_msg_bytes = sum(msg_sizes)
    need_gb = max(2, int(4 * total_msg_bytes / (1 << 30)) + 2)
    pool_gb = args.pool_gb or need_gb

    def trace(msg):
        if os.environ.get("XPS_BENCH_TRACE"):
            print(f"# [rank {rank}] {msg}", file=sys.stderr, flush=True)

    cluster = Cluster(ps, n, rank, local, device,
                      master if world > 1 else "127.0.0.1", master_port + 137, pool_gb)
    trace("cluster up")

    server = ps.KVServer(0)
    if args.cpu and args.mode == "rn50":
        server.set_reduce_handle(num_workers=n)  # CPU BytePS rounds
    elif args.cpu:
        server.set_default_handle()
    elif args.mode == "sparse":
        spec = EmbeddingSpec(rows=args.emb_rows, width=args.emb_width)
        server.set_gpu_sparse_handle(spec.rows_local(n), spec.width, accumulate=True,
                                     key_shift=spec.key_shift)
    else:
        server.set_gpu_dense_handle(mode=("reduce" if args.mode == "rn50" else args.op),
                                    dtype=("bf16" if args.dtype == "bf16" else "f32"))
    trace("server handler installed")
    worker = ps.KVWorker(0, 0)
    ps.barrier("worker", ps.WORKER_GROUP)
    trace("post-handler barrier done")

    rng = np.random.default_rng(1234 + rank)

    # ---- buffers + the per-step closure ----------------------------------
    if args.mode == "sparse":
        spec = EmbeddingSpec(rows=args.emb_rows, width=args.emb_width)
        width = spec.width
        hot = args.hot_keys
        nsets = 4
        key_sets, key_dev_bufs = [], []
        for s in range(nsets):
            k = spec.hot_batch(hot, seed=100 * rank + s)
            key_sets.append(k)
            kb = ps.pool_alloc(k.nbytes) if device >= 0 else None
            if kb is not None:
                kb.copy_from(k)
            key_dev_bufs.append(kb)
        vbytes = hot * width * 4
        push_buf = ps.pool_alloc(vbytes)
        pull_buf = ps.pool_alloc(vbytes)
        push_buf.copy_from(rng.standard_normal(hot * width).astype(np.float32))
        # uniform row width: lens stays empty (8K explicit lens would blow
        # the ring-slot inline budget and force the TCP fallback)
        lens = np.array([], dtype=np.int32)
        state = {"i": 0}

        fused = not args.no_fused

        def one_step():
            i = state["i"] % nsets
            state["i"] += 1
            k, kb = key_sets[i], key_dev_bufs[i]
            if fused:
                # one trip: scatter the update and gather the post-update
                # rows in a single request per server
                ts = worker.zpushpull_ptr(k, push_buf.ptr, pull_buf.ptr, vbytes, device,
                                          lens, cmd=2, keys_dev_ptr=kb.ptr if kb else 0)
                worker.wait(ts)
            else:
                ts = worker.zpush_ptr(k, push_buf.ptr, vbytes, device, lens, cmd=2,
                                      keys_dev_ptr=kb.ptr if kb else 0)
                worker.wait(ts)
                ts = worker.zpull_ptr(k, pull_buf.ptr, vbytes, device, lens, cmd=0,
                                      keys_dev_ptr=kb.ptr if kb else 0)
                worker.wait(ts)

        bytes_per_worker_step = 2.0 * vbytes
    else:
        keys_np = np.array(keys, dtype=np.uint64)
        batch = args.batch_keys and args.mode == "dense" and device >= 0
        batch_reduce = args.mode == "rn50" and not args.per_key
        alloc = ps.pool_alloc if device >= 0 else ps.host_alloc
        if batch_reduce:
            # BytePS bucketed DenseReduce: ONE multi-key ZPush + ONE
            # multi-key ZPull of every bucket per step (the slicer splits
            # them into one message per server; the server runs one
            # batched kernel chain per round instead of 169 per-key
            # dispatches). Buckets are laid out contiguously in sorted
            # key order (the slicer requires sorted keys).
            order = np.argsort(keys_np, kind="stable")
            skeys = keys_np[order]
            ssizes = [msg_sizes[i] for i in order]
            lens_el = np.array([s // 4 for s in ssizes], dtype=np.int32)
            total_b = int(sum(ssizes))
            bpush_r = alloc(total_b)
            bpull_r = alloc(total_b)
            bpush_r.copy_from(rng.standard_normal(total_b // 4).astype(np.float32))

            def one_step():
                tsp = worker.zpush_ptr(skeys, bpush_r.ptr, total_b, device, lens_el, cmd=cmd)
                tsq = worker.zpull_ptr(skeys, bpull_r.ptr, total_b, device, lens_el, cmd=cmd)
                worker.wait(tsp)
                worker.wait(tsq)

            bytes_per_worker_step = 2.0 * total_msg_bytes
            push_bufs, pull_bufs = [], []
            push_ptrs, pull_ptrs = [], []
        else:
            push_bufs, pull_bufs = [], []
            for sz in msg_sizes:
                # host buffers come from the shm pool so the CPU config
                # rides the same-host zero-copy plane (not TCP)
                b = alloc(sz)
                b.copy_from(rng.standard_normal(sz // 4).astype(np.float32))
                push_bufs.append(b)
                pull_bufs.append(alloc(sz))
            push_ptrs = [b.ptr for b in push_bufs]
            pull_ptrs = [b.ptr for b in pull_bufs]
            uniform = len(set(msg_sizes)) == 1
            if batch:
                k = len(keys) // n
                # python-int division: step_range is 2**64 at n=1, which
                # overflows a numpy uint64 scalar
                srv_of = np.array([int(key) // step_range for key in keys])
                srv_keys = [np.sort(keys_np[srv_of == s]) for s in range(n)]
                blens = np.full(k, msg_sizes[0] // 4, dtype=np.int32)
                bpush = [ps.pool_alloc(k * msg_sizes[0]) for _ in range(n)]
                bpull = [ps.pool_alloc(k * msg_sizes[0]) for _ in range(n)]
                for b in bpush:
                    b.copy_from(rng.standard_normal(k * msg_sizes[0] // 4).astype(np.float32))

            def one_step():
                if batch:
                    tss = [worker.zpush_ptr(srv_keys[s], bpush[s].ptr, k * msg_sizes[0], device,
                                            blens, cmd=cmd) for s in range(n)]
                    for ts in tss:
                        worker.wait(ts)
                    tss = [worker.zpull_ptr(srv_keys[s], bpull[s].ptr, k * msg_sizes[0], device,
                                            blens, cmd=cmd) for s in range(n)]
                    for ts in tss:
                        worker.wait(ts)
                elif uniform and not overlap_pull:
                    worker.round(keys_np, push_ptrs, msg_sizes[0], device, cmd, False)
                    worker.round(keys_np, pull_ptrs, msg_sizes[0], device, cmd, True)
                else:
                    worker.round_mixed(keys_np, push_ptrs, pull_ptrs, msg_sizes, device, cmd,
                                       overlap_pull)

            bytes_per_worker_step = 2.0 * total_msg_bytes

    trace("buffers ready")
    for w in range(args.warmup):
        one_step()
        trace(f"warmup step {w} done")

    if args.smoke and args.mode == "dense" and device >= 0:
        got = pull_bufs[0].to_numpy_f32()
        want = push_bufs[0].to_numpy_f32()
        assert np.allclose(got, want, atol=1e-6), "smoke: pulled values mismatch"
        print("SMOKE_OK: pulled values match pushed values")

    def sync():
        if device >= 0:
            ps.device_sync(device)

    ps.barrier("worker", ps.WORKER_GROUP)
    sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        one_step()
    sync()
    t1 = time.perf_counter()
    ps.barrier("worker", ps.WORKER_GROUP)
    elapsed = t1 - t0

    # p50 round trip: one blocking push+pull of a single key
    rtts = {}
    if not args.no_rtt and not args.smoke and args.mode == "dense":
        for j, (sz_name, sz) in enumerate((("1mb", 1 << 20), ("64mb", 64 << 20))):
            if sz > max(msg_sizes):
                continue
            rtt_key = np.array([10_000_000 + 100 * rank + j], dtype=np.uint64)
            samples = []
            l = np.array([sz // 4], dtype=np.int32)
            for _ in range(args.rtt_iters):
                a = time.perf_counter()
                worker.wait(worker.zpush_ptr(rtt_key, push_ptrs[0], sz, device, l, cmd=cmd))
                worker.wait(worker.zpull_ptr(rtt_key, pull_ptrs[0], sz, device, l, cmd=cmd))
                samples.append((time.perf_counter() - a) * 1e6)
            rtts[sz_name] = statistics.median(samples)

    vals = reduce_max([elapsed] + [rtts.get("1mb", 0.0), rtts.get("64mb", 0.0)], rank, world)
    elapsed = vals[0]
    if "1mb" in rtts:
        rtts["1mb"] = vals[1]
    if "64mb" in rtts:
        rtts["64mb"] = vals[2]

    gbs_per_worker = bytes_per_worker_step * args.steps / elapsed / 1e9
    total_gbs = gbs_per_worker * n
    ms_per_step = elapsed / args.steps * 1e3

    if rank == 0:
        out = {
            "metric": ("push+pull GB/s per worker + p50 round-trip µs, "
                       "1 MB & 64 MB, at 1/2/4/8 GPUs"),
            "value": round(total_gbs, 3),
            "unit": "GB/s",
            "n_gpus": n,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": args.dtype,
            "data": "synthetic",
            "config": {
                "model": model_name,
                "global_batch": len(msg_sizes) or args.hot_keys,
                "seq_len": max(msg_sizes) if msg_sizes else args.emb_width,
                "parallelism": ("cpu-tcp" if args.cpu
                                else f"byteps-joint x{n} (worker+server per GPU)"),
                "mode": args.mode + ("-cpu" if args.cpu else ""),
                "msg_bytes": max(msg_sizes) if msg_sizes else args.hot_keys * args.emb_width * 4,
                "keys_per_server": args.keys_per_server,
                "batch_keys": args.batch_keys,
                "overlap": overlap_pull,
                "op": args.op,
                "gbs_per_worker": round(gbs_per_worker, 3),
                "p50_rtt_us": {k: round(v, 1) for k, v in rtts.items()},
            },
        }
        if not args.cpu:
            # per-peer plane traffic (rank 0's view): each peer pair is
            # its own xGMI link, so at N>1 this is per-link utilization
            try:
                pb = ps.plane_peer_bytes("worker")
                out["config"]["per_peer_gb"] = {
                    int(k): [round(v[0] / 1e9, 2), round(v[1] / 1e9, 2)]
                    for k, v in pb.items()}
            except Exception:
                pass
        print(json.dumps(out, ensure_ascii=False))

    cluster.finish()


if __name__ == "__main__":
    main()
