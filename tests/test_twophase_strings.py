"""String group keys in the two-phase exchange (the last §7 exchange gap):
the 32-byte YtStateRow is reused with key_bits = (offset within the state's
own partition pool slice) << 24 | len, and the key-byte pool slices travel
beside the states in the all-to-all — the reference's key shuffle
(engine_api/shuffling_reader.cpp:40-42) applied to string group keys.
Partition = splitmix64(FNV-1a(key bytes)) % world, identical in the oracle
and the GPU kernels (cross-checked here)."""
import ctypes as C

import numpy as np
import pytest

import ytsaurus_amd as y
from ytsaurus_amd._abi import YtStateRow, VT_STRING, VT_DOUBLE, VT_INT64


def make_raw(seed, n=6000, nkeys=97, with_null=True, double_sum=True):
    rng = np.random.default_rng(seed)
    keys = [f"key-{i:04d}".encode() for i in range(nkeys)]
    ks = [keys[int(i)] for i in rng.integers(0, nkeys, n)]
    kn = (rng.random(n) < (0.03 if with_null else 0)).astype(np.uint8)
    ks = [None if kn[i] else ks[i] for i in range(n)]
    if double_sum:
        v = rng.random(n) * 100 - 50
    else:
        v = rng.integers(-10**6, 10**6, n, dtype=np.int64)
    vn = (rng.random(n) < 0.06).astype(np.uint8)
    return ks, v, vn


def chunk_of_raw(raw):
    ks, v, vn = raw
    if v.dtype == np.float64:
        vcol = y.encode_double(v, vn)
    else:
        vcol = y.encode_int64(v, vn)
    return y.Chunk([y.encode_string(ks), vcol], len(ks))


def make_shard(seed, n=6000, nkeys=97, with_null=True, double_sum=True):
    return chunk_of_raw(make_raw(seed, n, nkeys, with_null, double_sum))


def plan():
    return y.Plan(keys=[y.col(0)], aggs=[y.agg_sum(y.col(1)), y.agg_sum1()])


def approx_rows(got, want):
    gm = {r[0]: r for r in got}
    assert len(got) == len(want)
    for k, sv, cnt in want:
        gk, gs, gc = gm[k]
        assert gc == cnt, k
        if sv is None:
            assert gs is None, k
        elif isinstance(sv, float):
            assert gs == pytest.approx(sv, rel=1e-9, abs=1e-12), k
        else:
            assert gs == sv, k


@pytest.mark.parametrize("double_sum", [True, False])
def test_oracle_string_two_phase(double_sum):
    world = 3
    raws = [make_raw(50 + r, double_sum=double_sum) for r in range(world)]
    shards = [chunk_of_raw(r) for r in raws]
    # bottom queries
    outs = [y.oracle_partial_str(plan(), s, world) for s in shards]
    # exchange: partition p receives each rank's p-slice (states + pool)
    union = []
    for p in range(world):
        segs = []
        for r in range(world):
            states, counts, pool, pbytes = outs[r]
            rbase = sum(counts[:p])
            bbase = sum(pbytes[:p])
            seg_states = [states[rbase + i] for i in range(counts[p])]
            seg_pool = pool[bbase:bbase + pbytes[p]]
            segs.append((seg_states, counts[p], seg_pool, pbytes[p]))
        union += y.oracle_merge_str(plan(), segs)
    big = concat_raws(raws)
    want, _ = y.oracle_execute(plan(), big)
    approx_rows(sorted(union, key=rkey), sorted(want, key=rkey))
    # partitions are key-disjoint
    keysets = []
    for p in range(world):
        segs = []
        for r in range(world):
            states, counts, pool, pbytes = outs[r]
            rbase = sum(counts[:p])
            bbase = sum(pbytes[:p])
            segs.append(([states[rbase + i] for i in range(counts[p])],
                         counts[p], pool[bbase:bbase + pbytes[p]], pbytes[p]))
        keysets.append({r2[0] for r2 in y.oracle_merge_str(plan(), segs)})
    for a in range(world):
        for b in range(a + 1, world):
            assert not (keysets[a] & keysets[b])


def rkey(r):
    return (r[0] is None, r[0] or b"")


def concat_raws(raws):
    ks = [k for r in raws for k in r[0]]
    v = np.concatenate([r[1] for r in raws])
    vn = np.concatenate([r[2] for r in raws])
    return chunk_of_raw((ks, v, vn))


def test_gloo_string_exchange():
    """world-2 multiprocess CPU coverage of the exchange logic itself."""
    import torch.multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    ps = [ctx.Process(target=_gloo_rank, args=(r, 2, q)) for r in range(2)]
    for p in ps:
        p.start()
    res = {}
    for _ in range(2):
        rank, rows = q.get(timeout=300)
        res[rank] = rows
    for p in ps:
        p.join(timeout=60)
    union = res[0] + res[1]
    want, _ = y.oracle_execute(plan(), concat_raws(
        [make_raw(50 + r) for r in range(2)]))
    approx_rows(sorted(union, key=rkey), sorted(want, key=rkey))


def _gloo_rank(rank, world, q):
    import os
    import torch
    import torch.distributed as dist
    import ytsaurus_amd as y2
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29517")
    dist.init_process_group("gloo", rank=rank, world_size=world)
    shard = make_shard(50 + rank)
    states, counts, pool, pbytes = y2.oracle_partial_str(plan(), shard, world)
    # exchange sizes
    cnt_t = torch.tensor(counts + pbytes, dtype=torch.int64)
    all_cnt = [torch.zeros_like(cnt_t) for _ in range(world)]
    dist.all_gather(all_cnt, cnt_t)
    # exchange my row/pool slices: rank r sends slice p to rank p
    segs = []
    for r in range(world):
        rc_counts = all_cnt[r][:world].tolist()
        rc_bytes = all_cnt[r][world:].tolist()
        if r == rank:
            for p in range(world):
                rbase = sum(counts[:p])
                bbase = sum(pbytes[:p])
                st_bytes = bytes(
                    C.string_at(C.byref(states, rbase * C.sizeof(YtStateRow)),
                                counts[p] * C.sizeof(YtStateRow)))
                pl = pool[bbase:bbase + pbytes[p]]
                if p == rank:
                    segs.append((r, st_bytes, pl))
                else:
                    t1 = torch.frombuffer(bytearray(st_bytes),
                                          dtype=torch.uint8)
                    t2 = torch.frombuffer(bytearray(pl), dtype=torch.uint8)
                    dist.send(t1, dst=p, tag=1)
                    dist.send(t2, dst=p, tag=2)
        else:
            n1 = rc_counts[rank] * C.sizeof(YtStateRow)
            n2 = rc_bytes[rank]
            t1 = torch.zeros(n1, dtype=torch.uint8)
            t2 = torch.zeros(n2, dtype=torch.uint8)
            dist.recv(t1, src=r, tag=1)
            dist.recv(t2, src=r, tag=2)
            segs.append((r, t1.numpy().tobytes(), t2.numpy().tobytes()))
    dist.barrier()
    # merge my partition
    msegs = []
    for _, st_bytes, pl in sorted(segs):
        cnt = len(st_bytes) // C.sizeof(YtStateRow)
        arr = (YtStateRow * max(cnt, 1)).from_buffer_copy(
            st_bytes or bytes(C.sizeof(YtStateRow)))
        msegs.append((arr, cnt, pl, len(pl)))
    rows = y2.oracle_merge_str(plan(), msegs)
    q.put((rank, rows))
    dist.destroy_process_group()


@pytest.mark.gpu
@pytest.mark.parametrize("double_sum", [True, False])
def test_gpu_string_two_phase(cuda, double_sum):
    shard = make_shard(99, n=120_000, nkeys=1009, double_sum=double_sum)
    dev = shard.c_device(cuda)
    cap = 4 * 1009 + 1024
    states_t = cuda.zeros((cap, 4), dtype=cuda.int64, device="cuda")
    pool_cap = 32 * cap
    pool_t = cuda.zeros(pool_cap, dtype=cuda.uint8, device="cuda")
    counts, pbytes, st = y.gpu_partial_str(plan(), dev, 1,
                                           states_t.data_ptr(), cap,
                                           pool_t.data_ptr(), pool_cap,
                                           max_groups_hint=4096)
    got, _ = y.gpu_merge_str(plan(), states_t.data_ptr(), counts,
                             pool_t.data_ptr(), pbytes,
                             col_types=[VT_STRING,
                                        VT_DOUBLE if double_sum else VT_INT64],
                             max_groups_hint=4096)
    want, _ = y.oracle_execute(plan(), shard)
    approx_rows(sorted(got, key=rkey), sorted(want, key=rkey))


@pytest.mark.gpu
def test_gpu_string_states_merged_by_oracle(cuda):
    """cross-implementation: GPU partial states + pool merged by the ORACLE
    merge — the byte format and partition function must agree exactly."""
    world = 2
    shard_gpu = make_shard(50, double_sum=True)     # rank 0 on GPU
    shard_cpu = make_shard(51, double_sum=True)     # rank 1 on oracle
    dev = shard_gpu.c_device(cuda)
    cap = shard_gpu.row_count + 16
    states_t = cuda.zeros((cap, 4), dtype=cuda.int64, device="cuda")
    pool_cap = 32 * cap
    pool_t = cuda.zeros(pool_cap, dtype=cuda.uint8, device="cuda")
    counts0, pbytes0, _ = y.gpu_partial_str(plan(), dev, world,
                                            states_t.data_ptr(), cap,
                                            pool_t.data_ptr(), pool_cap,
                                            max_groups_hint=4096)
    h_states = states_t.cpu().numpy().tobytes()
    h_pool = bytes(pool_t.cpu().numpy().tobytes())
    states1, counts1, pool1, pbytes1 = y.oracle_partial_str(plan(), shard_cpu,
                                                            world)
    union = []
    for p in range(world):
        segs = []
        rbase0 = sum(counts0[:p]) * C.sizeof(YtStateRow)
        bbase0 = sum(pbytes0[:p])
        sb = h_states[rbase0:rbase0 + counts0[p] * C.sizeof(YtStateRow)]
        arr0 = (YtStateRow * max(counts0[p], 1)).from_buffer_copy(
            sb or bytes(C.sizeof(YtStateRow)))
        segs.append((arr0, counts0[p],
                     h_pool[bbase0:bbase0 + pbytes0[p]], pbytes0[p]))
        rbase1 = sum(counts1[:p])
        bbase1 = sum(pbytes1[:p])
        segs.append(([states1[rbase1 + i] for i in range(counts1[p])],
                     counts1[p], pool1[bbase1:bbase1 + pbytes1[p]],
                     pbytes1[p]))
        union += y.oracle_merge_str(plan(), segs)
    want, _ = y.oracle_execute(plan(), concat_raws(
        [make_raw(50), make_raw(51)]))
    approx_rows(sorted(union, key=rkey), sorted(want, key=rkey))


def test_coordinator_combine_string_partitions():
    """coordinator tail over STRING-keyed front outputs: key-disjoint
    concatenation + a global ORDER BY sum LIMIT applied by the combiner
    (the string merge emits full group sets, so the global sort is exact)"""
    world = 3
    raws = [make_raw(70 + r, n=4000, nkeys=61, with_null=False)
            for r in range(world)]
    shards = [chunk_of_raw(r) for r in raws]
    outs = [y.oracle_partial_str(plan(), s, world) for s in shards]
    results = []
    for p in range(world):
        segs = []
        for r in range(world):
            states, counts, pool, pbytes = outs[r]
            rbase = sum(counts[:p])
            bbase = sum(pbytes[:p])
            segs.append(([states[rbase + i] for i in range(counts[p])],
                         counts[p], pool[bbase:bbase + pbytes[p]], pbytes[p]))
        results.append(y.oracle_merge_str(plan(), segs))
    oplan = y.Plan(keys=[y.col(0)], aggs=[y.agg_sum(y.col(1)), y.agg_sum1()],
                   order_by=[(1, True)], limit=9)
    combined = y.coordinate_results(oplan, results)
    big = concat_raws(raws)
    want, _ = y.oracle_execute(oplan, big)
    assert [r[1] for r in combined] == pytest.approx([r[1] for r in want],
                                                     rel=1e-9)
    assert len(combined) == 9
