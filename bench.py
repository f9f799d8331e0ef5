#!/usr/bin/env python3
"""bench.py — headline benchmark for the MI355X-native YTsaurus query-path
executor.

Workload (BASELINE.json configs[2], the configuration the headline metric
"rows/sec scan+GROUP BY on 1B-row int64 chunks" is quoted on):
GROUP BY int64 key (1M distinct) with sum(v), sum(1) over N_ROWS rows of
seeded synthetic data, stored in the reference's unversioned columnar chunk
format (DirectDense int64 segments), resident in HBM when timing starts.

A step = one full execution of the hot path over the resident encoded chunks
(decode + group-by + aggregate + compact + readback). For --gpus N > 1, a
step = per-rank partial aggregation, RCCL all-to-all of hash-partitioned
state rows over xGMI, local merge; per-GPU rows fixed → weak
scaling.

Prints ONE JSON line from rank 0 per the driver contract, including:
  roofline     — dominant kernel (fused scan_group) achieved GB/s vs the
                 8 TB/s HBM peak, algorithmic bytes = encoded bytes of the
                 two scanned columns, timed with HIP events
                 inside the library on the launch stream
  cpu_baseline — the CPU oracle (restated reference algorithm, "port") timed
                 on this box's host cores on a bounded sample
"""
import argparse
import ctypes
import json
import os
import sys
import time
from concurrent.futures import ThreadPoolExecutor

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

SEED = 20260915
KEY_SPACE = 1_000_000          # 1M distinct keys (config 3)
VAL_BITS = 40                  # values uniform in [0, 2^40)
SLICE = 16 * 128 * 1024        # 2Mi rows per generation slice (16 segments)


def log(msg):
    print("[bench] " + msg, file=sys.stderr, flush=True)


def gen_encode_column(n, seed, lo, hi, nthreads, base_seed_tag):
    """Generate + encode one int64 column in parallel slices; bit-identical
    to a serial encode (cum_rows_base threads the reference's cumulative
    RowCount_ through)."""
    import ytsaurus_amd as y

    slices = [(s, min(s + SLICE, n)) for s in range(0, n, SLICE)]

    def work(item):
        i, (b, e) = item
        rng = np.random.default_rng([seed, base_seed_tag, i])
        vals = rng.integers(lo, hi, e - b, dtype=np.int64)
        return y.encode_int64(vals, cum_rows_base=b)

    with ThreadPoolExecutor(max_workers=nthreads) as ex:
        encs = list(ex.map(work, enumerate(slices)))
    return encs


STR_WINDOW = 65536             # distinct keys per generation slice (strgroup)


def gen_encode_strkey_column(n, seed, nthreads, window=STR_WINDOW):
    """String key column for the config-5 family: each 2Mi-row slice draws
    from its own disjoint window of `window` keys, so the reference's
    min-size rule picks DictionaryDense (window <= ~210K keeps per-128Ki-
    segment dictionaries smaller than direct) — global distinct ≈
    (n/SLICE)·window. window ≈ 210K at 1B rows gives the NAMED config-5
    scale: 100M global distinct keys."""
    import ytsaurus_amd as y

    slices = [(s, min(s + SLICE, n)) for s in range(0, n, SLICE)]

    wstep = SLICE // 2   # window advances every 1Mi rows (8 segments)

    def work(item):
        i, (b, e) = item
        m = e - b
        rng = np.random.default_rng([seed, 777, i])
        widx = (b + np.arange(m)) // wstep
        ids = widx * window + rng.integers(0, window, m)
        # fixed-width 10-byte keys "k%09d", built with numpy (no per-row Python)
        dig = (ids[:, None] // 10 ** np.arange(8, -1, -1)) % 10
        arr = np.empty((m, 10), dtype=np.uint8)
        arr[:, 0] = ord("k")
        arr[:, 1:] = dig + ord("0")
        begins = np.arange(m, dtype=np.uint64) * 10
        lens = np.full(m, 10, dtype=np.uint32)
        nulls = np.zeros(m, dtype=np.uint8)
        return y.encode_string_raw(arr.tobytes(), begins, lens, nulls)

    with ThreadPoolExecutor(max_workers=nthreads) as ex:
        return list(ex.map(work, enumerate(slices)))


def gen_encode_double_column(n, seed, nthreads):
    import ytsaurus_amd as y

    slices = [(s, min(s + SLICE, n)) for s in range(0, n, SLICE)]

    def work(item):
        i, (b, e) = item
        rng = np.random.default_rng([seed, 778, i])
        return y.encode_double(rng.random(e - b))

    with ThreadPoolExecutor(max_workers=nthreads) as ex:
        return list(ex.map(work, enumerate(slices)))


def build_device_chunk(enc_cols, n, torch):
    """Concatenate per-slice encodings into one device chunk."""
    from ytsaurus_amd._abi import YtChunk, YtColumn, YtSegment

    keep = []
    cols = (YtColumn * len(enc_cols))()
    total_bytes = 0
    for ci, encs in enumerate(enc_cols):
        nseg = sum(e._cenc.segment_count for e in encs)
        segs = (YtSegment * nseg)()
        keep.append(segs)
        at = 0
        for e in encs:
            blob_base = e._cenc.blob
            host = np.frombuffer(
                ctypes.string_at(blob_base, e._cenc.blob_size), dtype=np.uint8)
            t = torch.from_numpy(host.copy()).cuda()
            keep.append(t)
            for j in range(e._cenc.segment_count):
                s = e._cenc.segments[j]
                segs[at] = YtSegment(type=s.type, row_count=s.row_count,
                                     min_value=s.min_value,
                                     data=t.data_ptr() + (s.data - blob_base),
                                     data_size=s.data_size)
                total_bytes += s.data_size
                at += 1
        cols[ci] = YtColumn(value_type=encs[0].value_type, segment_count=nseg,
                            segments=segs)
    ch = YtChunk(row_count=n, column_count=len(enc_cols), columns=cols)
    ch._keep = keep
    return ch, total_bytes


def cpu_baseline_leg(plan_f, enc_cols, n, cores, out_cap, pool_bytes=0):
    """Time the oracle (CPU restatement, kind 'port') on a bounded sample of
    the same workload: enough slices for ~10-30 s of CPU work."""
    import ytsaurus_amd as y
    from ytsaurus_amd._abi import YtChunk, YtColumn, YtSegment

    sample_rows = min(n, 8 * SLICE)    # ≤ 16Mi rows
    # host chunk view over the first slices
    cols = (YtColumn * len(enc_cols))()
    keep = []
    for ci, encs in enumerate(enc_cols):
        segs_list = []
        rows = 0
        for e in encs:
            for j in range(e._cenc.segment_count):
                s = e._cenc.segments[j]
                if rows >= sample_rows:
                    break
                segs_list.append(s)
                rows += s.row_count
        nseg = len(segs_list)
        segs = (YtSegment * nseg)(*segs_list)
        keep.append(segs)
        cols[ci] = YtColumn(value_type=encs[0].value_type,
                            segment_count=nseg, segments=segs)
        sample_rows = rows
    ch = YtChunk(row_count=sample_rows, column_count=len(enc_cols), columns=cols)

    from ytsaurus_amd.api import _mk_rowset, _attach_join
    from ytsaurus_amd import _abi
    rs = _mk_rowset(out_cap, 4, pool_bytes=pool_bytes)
    st = _abi.YtStatistics()
    err = ctypes.create_string_buffer(256)
    plan_obj = plan_f()
    _j = _attach_join(plan_obj, lambda c: c.c_host())
    t0 = time.monotonic()
    rc = _abi.oracle_lib().yto_execute(
        ctypes.byref(plan_obj.c), ctypes.byref(ch), ctypes.byref(rs),
        ctypes.byref(st), cores, err, 256)
    dt = time.monotonic() - t0
    if rc != 0:
        log("cpu baseline failed: %s" % err.value)
        return None
    return {
        "value": sample_rows / dt,
        "unit": "rows/s",
        "cores": cores,
        "kind": "port",
        "sample": "first %d rows of the same encoded chunk, oracle MT group-by"
                  % sample_rows,
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--rows", type=float, default=1e9,
                    help="rows per GPU (weak scaling)")
    ap.add_argument("--keys", type=int, default=KEY_SPACE)
    ap.add_argument("--workload", default="groupby",
                    choices=["groupby", "scanfilter", "strgroup", "topk",
                             "join"])
    ap.add_argument("--limit", type=int, default=1000,
                    help="topk workload: ORDER BY ... LIMIT n")
    ap.add_argument("--no-cpu-baseline", action="store_true")
    ap.add_argument("--traffic-bytes", type=float, default=0.0,
                    help="measured per-launch HBM bytes from a rocprofv3 --pmc run")
    args = ap.parse_args()

    n = int(args.rows)
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))

    import torch
    assert torch.cuda.is_available(), "bench needs a GPU (no CPU fallback exists)"
    torch.cuda.set_device(local_rank)

    import ytsaurus_amd as y

    dist = None
    force_two = os.environ.get("YTQL_FORCE_TWOPHASE") == "1"
    if world > 1 or (force_two and "RANK" in os.environ):
        import torch.distributed as dist_mod
        dist = dist_mod
        dist.init_process_group("nccl")

    key_space = args.keys

    def make_plan():
        if args.workload in ("groupby", "strgroup"):
            return y.Plan(keys=[y.col(0)],
                          aggs=[y.agg_sum(y.col(1)), y.agg_sum1()])
        if args.workload == "topk":
            # SELECT k, v ORDER BY k LIMIT n (TopCollector / OrderOpHelper
            # shape): histogram k-selection, no scan materialization
            return y.Plan(projects=[y.col(0), y.col(1)],
                          order_by=[(0, False)], limit=args.limit)
        if args.workload == "join":
            # fact JOIN dim (unique 1M-key dimension) + GROUP BY fact key:
            # generic-path probe per row (MultiJoinOpHelper slice)
            return y.Plan(keys=[y.col(0)],
                          aggs=[y.agg_sum(y.col(2)), y.agg_sum1()],
                          join=main.join_spec)
        lo, hi = int(0.25 * 2**VAL_BITS), int(0.75 * 2**VAL_BITS)
        return y.Plan(filter=(y.col(0) >= lo).and_(y.col(0) <= hi),
                      aggs=[y.agg_sum(y.col(1)), y.agg_sum(y.col(2)),
                            y.agg_sum(y.col(3)), y.agg_sum1()])

    ncols = 4 if args.workload == "scanfilter" else 2
    if args.workload in ("topk", "join"):
        assert world == 1, "%s: single-GPU this round" % args.workload
    cores = os.cpu_count() or 8

    if args.workload == "strgroup":
        assert world == 1, "strgroup: single-GPU this round (string partial " \
                           "states not in the exchange format yet)"
        # BASELINE configs[4] family: dictionary-encoded string key GROUP BY
        # + double sum, via the global-atomic per-dictionary-id accumulate
        # path; per-1Mi-row key windows sized so global
        # distinct ≈ --keys while each 128Ki-row segment still sees a
        # dictionary-favoured density (the reference's min-size rule flips
        # to direct strings above ~94K distinct per segment)
        nwindows = (n + SLICE // 2 - 1) // (SLICE // 2)
        str_window = max(1024, min(131_072, (args.keys + nwindows - 1) // nwindows))
        key_space = nwindows * str_window

    t0 = time.monotonic()
    main.join_spec = None
    if args.workload == "join":
        # dimension: 1M unique keys (permuted), one value column
        rng = np.random.default_rng([SEED, 999])
        fn = key_space
        fkey = rng.permutation(np.arange(fn, dtype=np.int64))
        fval = rng.integers(0, 2**VAL_BITS, fn, dtype=np.int64)
        fchunk = y.Chunk([y.encode_int64(fkey), y.encode_int64(fval)], fn)
        main.join_chunk = fchunk
        main.join_spec = y.Join(fchunk, 0, 0, [1])
    if args.workload == "strgroup":
        enc_cols = [gen_encode_strkey_column(n, SEED + rank, cores, str_window),
                    gen_encode_double_column(n, SEED + rank, cores)]
    else:
        enc_cols = []
        for ci in range(ncols):
            if args.workload in ("groupby", "join") and ci == 0:
                lo, hi = 0, key_space
            else:
                lo, hi = 0, 2**VAL_BITS
            enc_cols.append(gen_encode_column(n, SEED + rank, lo, hi, cores, ci))
    log("rank %d: generated+encoded %d rows x %d cols in %.1fs"
        % (rank, n, ncols, time.monotonic() - t0))

    t0 = time.monotonic()
    dev_chunk, enc_bytes = build_device_chunk(enc_cols, n, torch)
    torch.cuda.synchronize()
    upload_s = time.monotonic() - t0
    log("rank %d: uploaded %.2f GB encoded in %.1fs (%.1f GB/s PCIe-inclusive)"
        % (rank, enc_bytes / 1e9, upload_s, enc_bytes / 1e9 / max(upload_s, 1e-9)))

    main.join_dev = None
    if args.workload == "join":
        main.join_dev = main.join_chunk.c_device(torch)
    plan = make_plan()
    hint = key_space if args.workload in ("groupby", "join", "strgroup") else 0
    pool_b = (key_space * 10 + (1 << 20)) if args.workload == "strgroup" else 0
    out_cap = (args.limit + 64) if args.workload == "topk" else key_space + 4096
    out_ncols = 2 if args.workload == "topk" else 1 + len(plan.aggs)
    out_rs = y.make_rowset(out_cap, out_ncols, pool_bytes=pool_b)

    # multi-GPU state buffers
    if dist is not None:
        cap = 2 * key_space + 1024
        states_t = torch.zeros((cap, 4), dtype=torch.int64, device="cuda")
        recv_t = torch.zeros((2 * cap, 4), dtype=torch.int64, device="cuda")
        if args.workload == "strgroup":
            # string-key exchange: key-byte pool slices travel with the states
            spool_t = torch.zeros(cap * 12, dtype=torch.uint8, device="cuda")
            rpool_t = torch.zeros(2 * cap * 12, dtype=torch.uint8, device="cuda")

    scan_ms_total = 0.0
    scan_launches = 0

    def step():
        nonlocal scan_ms_total, scan_launches
        if dist is None:
            _, st = y.gpu_execute(plan, dev_chunk, max_groups_hint=hint,
                                  rowset=out_rs, raw_rowset=True,
                                  join_foreign=main.join_dev)
            scan_ms_total += st.kernel_scan_ms
            scan_launches += st.kernel_scan_launches
            step.last = st
            return st
        # bottom query: partial aggregate + hash partition on device
        import time as _t
        t0 = _t.monotonic()
        if args.workload == "strgroup":
            return step_str()
        counts, st = y.gpu_partial(plan, dev_chunk, world,
                                   states_t.data_ptr(), cap,
                                   max_groups_hint=hint)
        scan_ms_total += st.kernel_scan_ms
        scan_launches += st.kernel_scan_launches
        t1 = _t.monotonic()
        # exchange sizes then states (RCCL all-to-all over xGMI)
        sizes = torch.tensor(counts, dtype=torch.int64, device="cuda")
        rsizes = torch.zeros(world, dtype=torch.int64, device="cuda")
        dist.all_to_all_single(rsizes, sizes)
        rs = rsizes.cpu().tolist()
        offs = np.cumsum([0] + counts[:-1]).tolist()
        send_split = [c for c in counts]
        recv_split = [int(x) for x in rs]
        total_recv = sum(recv_split)
        dist.all_to_all_single(
            recv_t[:total_recv].view(-1, 4), states_t[:sum(counts)].view(-1, 4),
            output_split_sizes=recv_split, input_split_sizes=send_split)
        t2 = _t.monotonic()
        # front query: merge + finalize
        _, mst = y.gpu_merge(plan, recv_t.data_ptr(), total_recv,
                             max_groups_hint=hint, rowset=out_rs,
                             raw_rowset=True)
        if os.environ.get("YTQL_TIMING"):
            log("2ph step: partial %.1fms a2a %.1fms merge %.1fms"
                % ((t1 - t0) * 1e3, (t2 - t1) * 1e3, (_t.monotonic() - t2) * 1e3))
        return st

    def step_str():
        # string-keyed two-phase: exchange states AND key-byte pool slices
        import time as _t
        t0 = _t.monotonic()
        counts, pbytes, st = y.gpu_partial_str(plan, dev_chunk, world,
                                               states_t.data_ptr(), cap,
                                               spool_t.data_ptr(),
                                               spool_t.numel(),
                                               max_groups_hint=hint)
        scan_ms_total_add(st)
        t1 = _t.monotonic()
        sizes = torch.tensor(counts + pbytes, dtype=torch.int64, device="cuda")
        rsizes = torch.zeros(2 * world, dtype=torch.int64, device="cuda")
        dist.all_to_all_single(rsizes.view(world, 2),
                               sizes.view(2, world).t().contiguous())
        rs = rsizes.cpu().view(world, 2)
        recv_rows = [int(x) for x in rs[:, 0]]
        recv_bytes = [int(x) for x in rs[:, 1]]
        dist.all_to_all_single(
            recv_t[:sum(recv_rows)].view(-1, 4),
            states_t[:sum(counts)].view(-1, 4),
            output_split_sizes=recv_rows, input_split_sizes=counts)
        dist.all_to_all_single(
            rpool_t[:max(sum(recv_bytes), 1)],
            spool_t[:max(sum(pbytes), 1)],
            output_split_sizes=recv_bytes, input_split_sizes=pbytes)
        t2 = _t.monotonic()
        _, mst = y.gpu_merge_str(plan, recv_t.data_ptr(), recv_rows,
                                 rpool_t.data_ptr(), recv_bytes,
                                 col_types=col_types_of(),
                                 max_groups_hint=hint,
                                 rowset=out_rs, raw_rowset=True)
        if os.environ.get("YTQL_TIMING"):
            log("2ph-str step: partial %.1fms a2a %.1fms merge %.1fms"
                % ((t1 - t0) * 1e3, (t2 - t1) * 1e3,
                   (_t.monotonic() - t2) * 1e3))
        return st

    def scan_ms_total_add(st):
        nonlocal scan_ms_total, scan_launches
        scan_ms_total += st.kernel_scan_ms
        scan_launches += st.kernel_scan_launches
        step.last = st

    def col_types_of():
        from ytsaurus_amd._abi import VT_STRING, VT_DOUBLE
        return [VT_STRING, VT_DOUBLE]

    # warmup
    for _ in range(args.warmup):
        step()
    torch.cuda.synchronize()
    if dist:
        dist.barrier()
    scan_ms_total = 0.0
    scan_launches = 0

    t0 = time.monotonic()
    for _ in range(args.steps):
        step()
    torch.cuda.synchronize()
    if dist:
        dist.barrier()
    elapsed = time.monotonic() - t0

    # max over ranks
    if dist:
        t = torch.tensor([elapsed], dtype=torch.float64, device="cuda")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    st = getattr(step, "last", None)
    if st is not None:
        log("last step: exec %.1fms scanK %.2fms otherK %.2fms setup %.2fms rows_written %d"
            % (st.execute_time_ms, st.kernel_scan_ms / max(st.kernel_scan_launches, 1),
               st.kernel_other_ms, st.decode_time_ms, st.rows_written))

    ms_per_step = elapsed / args.steps * 1000
    total_rows = n * world
    value = total_rows / (elapsed / args.steps)

    # roofline: dominant kernel = fused scan_group; algorithmic bytes =
    # encoded bytes of the scanned columns per launch (one launch per step)
    scan_ms_avg = scan_ms_total / max(scan_launches, 1)
    achieved_gbs = (enc_bytes / 1e9) / (scan_ms_avg / 1e3) if scan_ms_avg else None
    peak_gbs = 8000.0   # HBM3E spec peak, MI355X_MICROARCH.md
    traffic = args.traffic_bytes or None
    if traffic is None and args.workload == "groupby" and n == 10**9:
        # measured once at this exact config/state via separate rocprofv3
        # --pmc FETCH_SIZE / WRITE_SIZE passes (profiles/r02/pmc_1e9.json):
        # k_scan_partition fetch 8.29 GB (x2-corrected) + write 47.49 GB
        traffic = 55.78e9
    roofline = {
        "bound": "hbm",
        "achieved": achieved_gbs,
        "peak": peak_gbs,
        "unit": "GB/s",
        "frac": (achieved_gbs / peak_gbs) if achieved_gbs else None,
        "traffic": traffic,
    }

    cpu_baseline = None
    if rank == 0 and world == 1 and not args.no_cpu_baseline:
        bl_pool = 0
        bl_cap = KEY_SPACE + 1024
        bl_cores = cores
        if args.workload == "strgroup":
            bl_cap = key_space + 4096
            bl_pool = key_space * 10 + (1 << 20)
        if args.workload == "topk":
            # the oracle scan+order path materializes all rows before the
            # sort, and runs single-threaded
            bl_cap = 8 * SLICE + 1024
            bl_cores = 1
        cpu_baseline = cpu_baseline_leg(make_plan, enc_cols, n, bl_cores,
                                        bl_cap, bl_pool)

    if rank == 0:
        out = {
            "metric": "rows/sec scan+GROUP BY on 1B-row int64 chunks; achieved HBM GB/s vs peak",
            "value": value,
            "unit": "rows/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "int64",
            "data": "synthetic",
            "config": {
                "workload": {
                    "groupby": "groupby_1M_distinct@%drows" % n,
                    "scanfilter": "scan_filter_sum_4col@%drows" % n,
                    "strgroup": "strgroup_dict_%ddistinct@%drows"
                                % (key_space, n),
                    "topk": "orderby_limit%d@%drows" % (args.limit, n),
                    "join": "join_dim%d_groupby@%drows" % (key_space, n),
                }[args.workload],
                "rows_per_gpu": n,
                "distinct_keys": key_space,
                "columns": ncols,
                "chunk_format": "unversioned DictionaryDense string + double"
                                if args.workload == "strgroup"
                                else "unversioned DirectDense int64 (reference layout)",
                "encoded_gb": enc_bytes / 1e9,
            },
            "roofline": roofline,
            "cpu_baseline": cpu_baseline,
        }
        print(json.dumps(out), flush=True)

    if dist:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
