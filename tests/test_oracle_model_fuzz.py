"""Oracle-vs-model fuzz (CPU only): random filter+GROUP BY plans evaluated
by an INDEPENDENT pure-Python interpreter over the raw arrays, against the
C oracle over the encoded chunks. Pins the oracle's expression semantics
(wrapping int64 arithmetic, the reference's non-canonical null relations —
null == null is TRUE, null < any value — and Kleene AND/OR,
cg_fragment_compiler.cpp:1547-1649) beyond the transcribed goldens."""
import numpy as np
import pytest

import ytsaurus_amd as y
import test_fuzz as tf

MASK = (1 << 64) - 1


def wrap(v):
    v &= MASK
    return v - (1 << 64) if v >= (1 << 63) else v


NULL = object()


def ev(e, row):
    c = e.c
    op = c.op
    if op == 0:                      # column
        v = row[c.col]
        return NULL if v is None else v
    if op == 1:                      # int literal
        return c.lit_i64
    if op == 2:                      # null literal
        return NULL
    kids = e._keep
    a = ev(kids[0], row)
    b = ev(kids[1], row) if len(kids) > 1 else None
    if op in (10, 11, 12):           # + - *
        if a is NULL or b is NULL:
            return NULL
        r = {10: a + b, 11: a - b, 12: a * b}[op]
        r = wrap(int(r))
        # the oracle's arithmetic keeps the LHS operand's TYPE: a
        # boolean-typed lhs yields a boolean-typed result (bits != 0) —
        # outside the reference's legal type space (its typechecker rejects
        # bool arithmetic), mirrored for fuzz closure
        return (r != 0) if isinstance(a, bool) else r
    if 20 <= op <= 25:               # relations: null<any, null==null
        an, bn = a is NULL, b is NULL
        if an or bn:
            lt = bn < an
            eq = an == bn
        elif isinstance(a, bool):
            # the oracle picks SIGNED comparison only for an int64-typed
            # LHS; a boolean lhs compares raw bits unsigned (again outside
            # the reference's legal type space, mirrored for closure)
            x, z = int(a) & MASK, int(b) & MASK
            lt, eq = x < z, x == z
        else:
            lt, eq = a < b, a == b
        return {20: eq, 21: not eq, 22: lt, 23: lt or eq,
                24: not (lt or eq), 25: not lt}[op]
    if op in (30, 31):               # Kleene AND/OR
        an, bn = a is NULL, b is NULL
        av = False if an else bool(a)
        bv = False if bn else bool(b)
        if op == 30:
            if (not an and not av) or (not bn and not bv):
                return False
            return NULL if (an or bn) else True
        if (not an and av) or (not bn and bv):
            return True
        return NULL if (an or bn) else False
    if op == 32:                     # NOT
        return NULL if a is NULL else (not a)
    raise AssertionError(op)


def model(kw, arrays, n):
    rows = []
    for i in range(n):
        rows.append([None if (nl is not None and nl[i]) else int(v[i])
                     for v, nl in arrays])
    groups = {}
    for r in rows:
        if kw["filter"] is not None:
            f = ev(kw["filter"], r)
            if f is NULL or not f:
                continue
        key = tuple(None if (k := ev(e, r)) is NULL else k
                    for e in kw["keys"])
        g = groups.setdefault(key, [0] + [None] * len(kw["aggs"]))
        g[0] += 1
        for ai, agg in enumerate(kw["aggs"]):
            f, arg = (agg if isinstance(agg, tuple) else (None, None))
            from ytsaurus_amd._abi import (AGG_SUM, AGG_SUM1, AGG_MIN,
                                           AGG_MAX, AGG_AVG)
            func = agg[0]
            if func == AGG_SUM1:
                continue
            v = ev(agg[1], r)
            if v is NULL:
                continue
            cur = g[1 + ai]
            if func == AGG_SUM:
                g[1 + ai] = wrap((0 if cur is None else cur) + v)
            elif func == AGG_MIN:
                g[1 + ai] = v if cur is None else min(cur, v)
            elif func == AGG_MAX:
                g[1 + ai] = v if cur is None else max(cur, v)
            elif func == AGG_AVG:
                s, c2 = cur if cur is not None else (0, 0)
                g[1 + ai] = (wrap(s + v), c2 + 1)
    from ytsaurus_amd._abi import AGG_SUM1, AGG_AVG
    out = []
    for key, g in groups.items():
        vals = list(key)
        for ai, agg in enumerate(kw["aggs"]):
            if agg[0] == AGG_SUM1:
                vals.append(g[0])
            elif agg[0] == AGG_AVG:
                st = g[1 + ai]
                vals.append(None if st is None else st[0] / st[1])
            else:
                vals.append(g[1 + ai])
        out.append(tuple(vals))
    return out


@pytest.mark.parametrize("seed", range(40))
def test_oracle_vs_python_model(seed):
    rng = np.random.default_rng([20260916, seed])
    n = int(rng.choice([97, 1500]))
    ncols = int(rng.integers(2, 5))
    chunk, arrays = tf._rand_chunk(rng, n, ncols)
    filt = tf._rand_expr(rng, ncols) if rng.random() < 0.6 else None
    kc = int(rng.integers(1, 4))
    keycols = rng.choice(ncols, size=min(kc, ncols), replace=False)
    keys = [y.col(int(c)) for c in keycols]
    # expression keys sometimes (arith over columns)
    if rng.random() < 0.4:
        keys[0] = tf._rand_expr(rng, ncols)
    aggs = []
    from ytsaurus_amd._abi import AGG_MIN, AGG_MAX
    for _ in range(int(rng.integers(1, 4))):
        f = rng.integers(0, 5)
        c = y.col(int(rng.integers(0, ncols)))
        aggs.append([y.agg_sum(c), y.agg_sum1(), (AGG_MIN, c), (AGG_MAX, c),
                     y.agg_avg(c)][f])
    kw = dict(filter=filt, keys=keys, aggs=aggs)
    plan = y.Plan(**kw)
    # keep expr handles for the model (Plan consumes the same objects)
    try:
        got, _ = y.oracle_execute(plan, chunk)
    except RuntimeError as e:
        assert "forbidden" in str(e) or "62 bits" in str(e), str(e)
        return
    want = model(kw, arrays, n)
    def norm(rows):
        def k(r):
            return tuple((x is None, str(type(x)), x if x is not None else 0)
                         for x in r)
        out = []
        for r in rows:
            out.append(tuple(float(x) if isinstance(x, float) else x
                             for x in r))
        return sorted(out, key=k)
    g2, w2 = norm(got), norm(want)
    assert len(g2) == len(w2), (len(g2), len(w2))
    for a, b in zip(g2, w2):
        assert len(a) == len(b)
        for x, z in zip(a, b):
            if isinstance(z, float):
                assert x == pytest.approx(z, rel=1e-12), (a, b)
            else:
                assert x == z, (a, b)


def finish_model(kw, rows, kc):
    """totals(before) -> having -> totals(after) -> order/limit -> totals row
    (folding_profiler.cpp:1810-1815 Process order)"""
    from ytsaurus_amd._abi import AGG_SUM, AGG_SUM1, AGG_MIN, AGG_MAX

    def fold(rs):
        tot = []
        for ai, agg in enumerate(kw["aggs"]):
            f = agg[0]
            acc = None
            for r in rs:
                v = r[kc + ai]
                if f == AGG_SUM1:
                    acc = (acc or 0) + v
                elif v is None:
                    continue
                elif acc is None:
                    acc = v
                elif f == AGG_SUM:
                    acc = wrap(acc + v)
                elif f == AGG_MIN:
                    acc = min(acc, v)
                elif f == AGG_MAX:
                    acc = max(acc, v)
            tot.append(acc)
        return tot

    tot = None
    if kw.get("with_totals") and not kw.get("totals_after_having"):
        tot = fold(rows)
    if kw.get("having") is not None:
        keep = []
        for r in rows:
            h = ev(kw["having"], r)
            if h is not NULL and h:
                keep.append(r)
        rows = keep
    if kw.get("with_totals") and kw.get("totals_after_having"):
        tot = fold(rows)
    if kw.get("order_by"):
        def okey(r):
            out = []
            for col, desc in kw["order_by"]:
                v = r[col]
                null = v is None
                k = (0 if null else 1,
                     (int(v) if not null else 0))
                out.append((-k[0], -k[1]) if desc else k)
            return tuple(out)
        rows = sorted(rows, key=okey)
        off = kw.get("offset", 0)
        rows = rows[off:off + kw["limit"]]
    if kw.get("with_totals"):
        rows = rows + [tuple([None] * kc + tot)]
    return rows


@pytest.mark.parametrize("seed", range(35))
def test_oracle_vs_python_model_finish(seed):
    """having / WITH TOTALS (both modes) / ORDER BY..LIMIT over grouped
    output, cross-checked against the python restatement of the finish
    pipeline"""
    from ytsaurus_amd._abi import AGG_MIN, AGG_MAX
    rng = np.random.default_rng([20260918, seed])
    n = int(rng.choice([97, 2000]))
    ncols = int(rng.integers(2, 4))
    chunk, arrays = tf._rand_chunk(rng, n, ncols)
    filt = tf._rand_expr(rng, ncols) if rng.random() < 0.4 else None
    kc = int(rng.integers(1, 3))
    keycols = rng.choice(ncols, size=min(kc, ncols), replace=False)
    keys = [y.col(int(c)) for c in keycols]
    aggs = []
    for _ in range(int(rng.integers(1, 3))):
        f = rng.integers(0, 4)
        c = y.col(int(rng.integers(0, ncols)))
        aggs.append([y.agg_sum(c), y.agg_sum1(), (AGG_MIN, c),
                     (AGG_MAX, c)][f])
    out_cols = kc + len(aggs)
    having = tf._rand_expr(rng, out_cols) if rng.random() < 0.5 else None
    totals = bool(rng.integers(0, 2))
    after = bool(rng.integers(0, 2))
    order_by, limit, offset = (), 0, 0
    use_order = rng.random() < 0.5
    if use_order:
        order_by = [(int(rng.integers(0, out_cols)), bool(rng.integers(0, 2)))]
        limit = int(rng.integers(1, 40))
        offset = int(rng.integers(0, 3))
    kw = dict(filter=filt, keys=keys, aggs=aggs, having=having,
              with_totals=totals, totals_after_having=after,
              order_by=order_by, limit=limit, offset=offset)
    plan = y.Plan(**kw)
    try:
        got, _ = y.oracle_execute(plan, chunk)
    except RuntimeError as e:
        msg = str(e)
        assert ("forbidden" in msg or "62 bits" in msg or "NaN" in msg
                or "this round" in msg), msg
        return
    grp = model(dict(filter=filt, keys=keys, aggs=aggs), arrays, n)
    want = finish_model(kw, grp, kc)
    if order_by:
        # boundary ties are arbitrary: compare the order-key sequence and
        # the row count (and the totals row exactly)
        oc = order_by[0][0]
        gseq = [r[oc] for r in (got[:-1] if totals else got)]
        wseq = [r[oc] for r in (want[:-1] if totals else want)]
        assert gseq == wseq, (seed, gseq[:5], wseq[:5])
        assert len(got) == len(want)
        if totals:
            assert got[-1] == want[-1], seed
    else:
        def srt(rows):
            return sorted(rows, key=lambda r: tuple(
                (x is None, str(type(x)), 0 if x is None else x) for x in r))
        assert srt(got) == srt(want), seed
