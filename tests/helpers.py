"""Shared test helpers: golden-case loading and plan/chunk construction."""
import json
import os

import numpy as np

import ytsaurus_amd as y

GOLDEN = os.path.join(os.path.dirname(os.path.abspath(__file__)), "golden")

_OPS = {
    "add": lambda a, b: a + b,
    "sub": lambda a, b: a - b,
    "mul": lambda a, b: a * b,
    "div": lambda a, b: a // b,
    "mod": lambda a, b: a % b,
    "eq": lambda a, b: a == b,
    "ne": lambda a, b: a != b,
    "lt": lambda a, b: a < b,
    "le": lambda a, b: a <= b,
    "gt": lambda a, b: a > b,
    "ge": lambda a, b: a >= b,
    "and": lambda a, b: a.and_(b),
    "or": lambda a, b: a.or_(b),
}


def build_expr(spec):
    op = spec[0]
    if op == "col":
        return y.col(spec[1])
    if op == "int":
        return y.lit(spec[1])
    if op == "double":
        return y.litf(spec[1])
    if op == "null":
        return y.null()
    if op == "not":
        return build_expr(spec[1]).not_()
    return _OPS[op](build_expr(spec[1]), build_expr(spec[2]))


def build_plan(spec):
    filt = build_expr(spec["filter"]) if "filter" in spec else None
    keys = [build_expr(k) for k in spec.get("keys", [])]
    aggs = []
    for a in spec.get("aggs", []):
        if a[0] == "sum":
            aggs.append(y.agg_sum(build_expr(a[1])))
        elif a[0] == "sum1":
            aggs.append(y.agg_sum1())
        else:
            raise ValueError(a)
    projects = [build_expr(p) for p in spec.get("projects", [])]
    return y.Plan(filter=filt, keys=keys, aggs=aggs, projects=projects)


def build_chunk(columns, rows, max_seg=0):
    n = len(rows)
    encs = []
    for ci, (_, typ) in enumerate(columns):
        vals = [r[ci] for r in rows]
        nulls = np.array([1 if v is None else 0 for v in vals], dtype=np.uint8)
        if typ == "int64":
            arr = np.array([0 if v is None else int(v) for v in vals], dtype=np.int64)
            encs.append(y.encode_int64(arr, nulls, max_segment_values=max_seg))
        elif typ == "double":
            arr = np.array([0.0 if v is None else float(v) for v in vals])
            encs.append(y.encode_double(arr, nulls, max_segment_values=max_seg))
        else:
            raise ValueError(typ)
    return y.Chunk(encs, n)


def norm_rows(rows):
    """JSON has no int/bool distinction for expected values; normalize both
    sides to comparable tuples."""
    out = []
    for r in rows:
        out.append(tuple(int(v) if isinstance(v, bool) else v for v in r))
    return out


def load_cases():
    with open(os.path.join(GOLDEN, "cases.json")) as f:
        return json.load(f)
