"""Utility parity tests: parallel ordered match / parallel sort
(ps-lite parallel_kv_match.h / parallel_sort.h)."""
import numpy as np

import ps_lite_amd as ps


def test_parallel_ordered_match():
    rng = np.random.default_rng(0)
    src_keys = np.sort(rng.choice(10_000, size=500, replace=False)).astype(np.uint64)
    k = 4
    src_vals = rng.standard_normal(len(src_keys) * k).astype(np.float32)
    dst_keys = np.sort(rng.choice(10_000, size=800, replace=False)).astype(np.uint64)
    n, out = ps._core.parallel_ordered_match(src_keys, src_vals, dst_keys, k=k, nthreads=4)
    # reference
    ref = np.zeros(len(dst_keys) * k, dtype=np.float32)
    src_map = {int(key): i for i, key in enumerate(src_keys)}
    expect_n = 0
    for i, key in enumerate(dst_keys):
        if int(key) in src_map:
            si = src_map[int(key)]
            ref[i * k:(i + 1) * k] = src_vals[si * k:(si + 1) * k]
            expect_n += 1
    assert n == expect_n
    assert np.allclose(out, ref)


def test_parallel_ordered_match_accumulate():
    keys = np.array([1, 2, 3], dtype=np.uint64)
    vals = np.array([1.0, 2.0, 3.0], dtype=np.float32)
    n, out = ps._core.parallel_ordered_match(keys, vals, keys, k=1, accumulate=True)
    assert n == 3
    assert np.allclose(out, vals)


def test_parallel_sort():
    rng = np.random.default_rng(1)
    keys = rng.integers(0, 1 << 62, size=100_000, dtype=np.uint64)
    out = ps._core.parallel_sort(keys, nthreads=4)
    assert np.array_equal(out, np.sort(keys))


def test_wire_meta_roundtrip():
    assert ps._core._test_meta_roundtrip()


def test_shm_ring_stress_full_ring():
    """8 producers x 2000 msgs through a 1024-slot ring: the full-ring
    backoff path runs constantly; nothing may be lost or corrupted and
    the consumer must never wedge."""
    received, pushed, checksum_ok = ps._core._ring_stress(8, 2000, 512)
    assert received == pushed == 8 * 2000
    assert checksum_ok


def test_shm_gc_reaps_dead_owner_segments(tmp_path):
    """A process that dies without teardown leaks its host-shm arena; the
    next plane bring-up GCs it (owner pid recorded in the arena header)."""
    import os
    import subprocess
    import sys
    uid = 0xfeed0000 | os.getpid()
    seg = f"/dev/shm/xps_hostpool_{uid:016x}"
    code = (f"import ps_lite_amd as ps, os; "
            f"ps._core._host_pool_init_for_test({uid}, 1 << 20); os._exit(0)")
    env = dict(os.environ)
    env.setdefault("PYTHONPATH", os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    subprocess.run([sys.executable, "-c", code], check=True, timeout=60, env=env)
    assert os.path.exists(seg), "leaked segment expected"
    ps._core._gc_stale_shm()
    assert not os.path.exists(seg), "GC did not reap the dead-owner segment"


def test_rank_ordering_policies():
    """BytePS placement policies (reference van.cc:126-177): ordered
    hosts honored; mixed mode gives non-colocated servers the low
    server ranks."""
    nodes = [("worker", "hostB", 1), ("server", "hostB", 2),
             ("server", "hostC", 3), ("worker", "hostA", 4),
             ("server", "hostA", 5)]
    # default: (host, port)
    out = ps._core._order_nodes(nodes)
    assert [h for _, h, _ in out] == ["hostA", "hostA", "hostB", "hostB", "hostC"]
    # ordered hosts win over name order
    ps._core.init_env({"BYTEPS_ORDERED_HOSTS": "hostC,hostB"})
    out = ps._core._order_nodes(nodes)
    assert [h for _, h, _ in out] == ["hostC", "hostB", "hostB", "hostA", "hostA"]
    # mixed mode: hostC has no worker -> its server sorts before
    # colocated servers regardless of hostname order
    ps._core.init_env({"BYTEPS_ORDERED_HOSTS": "", "BYTEPS_ENABLE_MIXED_MODE": "1"})
    out = ps._core._order_nodes(nodes)
    servers = [h for r, h, _ in out if r == "server"]
    assert servers[0] == "hostC", out
    ps._core.init_env({"BYTEPS_ENABLE_MIXED_MODE": "0"})


def test_partition_tensor():
    from ps_lite_amd.utils.partition import partition_tensor
    # 18 MB over 4 servers at 4 MB parts -> 5 parts, exact coverage
    keys, lens = partition_tensor(18 << 20, num_servers=4)
    assert len(keys) == 5 == len(lens)
    assert int(lens.astype(np.int64).sum()) * 4 == 18 << 20
    assert all(int(n) * 4 <= (4 << 20) + 4 for n in lens)
    # keys sorted + spread over all 4 server ranges
    assert np.all(np.diff(keys.astype(np.uint64)) > 0)
    step = (1 << 64) // 4
    servers = {int(k) // step for k in keys}
    assert servers == {0, 1, 2, 3}
    # a worker can push/pull with them directly (end-to-end)


def test_wire_property_roundtrip():
    """Property test: arbitrary Meta field values survive the wire
    byte-stream round trip exactly (hypothesis-driven)."""
    from hypothesis import given, settings, strategies as st

    i32 = st.integers(min_value=-(2**31), max_value=2**31 - 1)
    u64 = st.integers(min_value=0, max_value=2**64 - 1)
    i64 = st.integers(min_value=-(2**63), max_value=2**63 - 1)

    @settings(max_examples=200, deadline=None, derandomize=True)
    @given(app=i32, cust=i32, ts=i32, snd=i32, rcv=i32,
           req=st.booleans(), push=st.booleans(), pull=st.booleans(),
           sapp=st.booleans(), head=i32,
           body=st.text(max_size=64).map(lambda t: t.encode("utf-8", "ignore")[:64].decode("utf-8", "ignore")),
           key=u64, addr=u64, vlen=i64, opt=i32, sig=u64, seq=u64,
           host=st.text(alphabet="abc123.-", max_size=32), port=i32,
           uid=u64, nh=st.integers(min_value=0, max_value=40))
    def check(app, cust, ts, snd, rcv, req, push, pull, sapp, head, body,
              key, addr, vlen, opt, sig, seq, host, port, uid, nh):
        assert ps._core._meta_roundtrip_fields(
            app, cust, ts, snd, rcv, req, push, pull, sapp, head, body,
            key, addr, vlen, opt, sig, seq, host, port, uid, nh)

    check()
