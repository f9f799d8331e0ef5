"""In-tree build driver: compiles the product library (hipcc, gfx950) and the
oracle (gcc). The built .so files live in-tree so the gpurun snapshot carries
them (they are git-ignored, not gpurun-ignored)."""
import os
import subprocess
import sys

HERE = os.path.dirname(os.path.abspath(__file__))
REPO = os.path.dirname(HERE)


def _run(cmd, cwd):
    print("+", " ".join(cmd), flush=True)
    subprocess.run(cmd, cwd=cwd, check=True)


def _newer(target, sources):
    if not os.path.exists(target):
        return False
    t = os.path.getmtime(target)
    return all(os.path.getmtime(s) < t for s in sources)


def build_gpu(force=False):
    srcs = [os.path.join(HERE, "csrc", f)
            for f in ("kernels.hip", "evaluator.cpp", "encoder.cpp")]
    hdrs = [os.path.join(HERE, "csrc", "common.h"),
            os.path.join(REPO, "include", "ytql_gpu.h")]
    out = os.path.join(HERE, "libytql_gpu.so")
    if not force and _newer(out, srcs + hdrs):
        return out
    cmd = ["hipcc", "--offload-arch=gfx950", "-O3", "-std=c++17",
           "-fPIC", "-shared", "-o", out] + srcs
    _run(cmd, HERE)
    return out


def build_oracle(force=False):
    src = os.path.join(REPO, "oracle", "ytql_oracle.c")
    out = os.path.join(REPO, "oracle", "libytql_oracle.so")
    if not force and _newer(out, [src, os.path.join(REPO, "include", "ytql_gpu.h")]):
        return out
    cmd = ["gcc", "-O2", "-std=c11", "-fPIC", "-shared", "-pthread",
           "-Wall", "-o", out, src]
    _run(cmd, os.path.join(REPO, "oracle"))
    return out


def build_all(force=False):
    build_oracle(force)
    build_gpu(force)


if __name__ == "__main__":
    build_all("--force" in sys.argv)
