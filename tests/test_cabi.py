"""C-ABI surface: the product library must load and export every symbol
declared in include/ytql_gpu.h, and must FAIL LOUDLY (YT_ERR_NO_GPU) on a
machine without a HIP device — no CPU fallback exists."""
import ctypes as C
import os
import re

import pytest

from ytsaurus_amd import _abi

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def declared_functions():
    hdr = open(os.path.join(REPO, "include", "ytql_gpu.h")).read()
    # function declarations: ret name(args...);
    names = re.findall(r"^\s*(?:int|void|int64_t)\s+(yt_\w+)\s*\(", hdr, re.M)
    assert len(names) >= 8
    return names


def test_exports_every_declared_symbol():
    lib = _abi.gpu_lib()
    for name in declared_functions():
        assert hasattr(lib, name), "missing export: %s" % name


def test_no_gpu_is_loud():
    try:
        import torch
        if torch.cuda.is_available():
            pytest.skip("GPU present")
    except Exception:
        pass
    lib = _abi.gpu_lib()
    err = C.create_string_buffer(256)
    rc = lib.yt_gpu_available(err, 256)
    assert rc == _abi.YT_ERR_NO_GPU
    # execute entries refuse too
    import numpy as np
    import ytsaurus_amd as y
    chunk = y.Chunk([y.encode_int64(np.arange(4, dtype=np.int64))], 4)
    plan = y.Plan(aggs=[y.agg_sum(y.col(0))])
    with pytest.raises(RuntimeError):
        y.gpu_execute(plan, chunk.c_host())


def test_oracle_header_marks_test_infra():
    src = open(os.path.join(REPO, "oracle", "ytql_oracle.c")).read()
    assert "TEST INFRASTRUCTURE ONLY" in src


def test_product_does_not_link_oracle():
    """The product library must not reference oracle symbols."""
    import subprocess
    out = subprocess.run(
        ["nm", "-D", os.path.join(REPO, "ytsaurus_amd", "libytql_gpu.so")],
        capture_output=True, text=True).stdout
    assert "yto_" not in out
