"""Isolated GEMM microbenchmark: pertgnn MFMA kernels vs torch (rocBLAS).

    python benchmarks/gemm_bench.py [--iters 50]
"""
import argparse
import sys
import time
from pathlib import Path

import torch

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))


def bench(fn, iters, *args):
    for _ in range(5):
        fn(*args)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn(*args)
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=50)
    args = ap.parse_args()
    import pertgnn._C as C

    dev = torch.device("cuda:0")
    shapes = [
        ("qkvs fwd", 42000, 1024, 256),
        ("conv mid", 42000, 256, 256),
        ("edge-ish", 55000, 256, 512),
        ("512H qkvs", 21000, 2048, 512),
    ]
    print(f"{'name':<12} {'M':>6} {'N':>5} {'K':>4}  {'ours_f32':>9} {'ours_bf16':>9} {'rocBLAS':>9}  (µs; TF in parens)")
    for name, m, n, k in shapes:
        a = torch.randn(m, k, device=dev)
        w = torch.randn(n, k, device=dev)
        fl = 2.0 * m * n * k
        t32 = bench(C.gemm_nt, args.iters, a, w)
        t16 = bench(C.gemm_nt_bf16, args.iters, a, w)
        trb = bench(lambda a, w: a @ w.t(), args.iters, a, w)
        print(f"{name:<12} {m:>6} {n:>5} {k:>4}  "
              f"{t32 * 1e6:7.1f} ({fl / t32 / 1e12:5.1f})  "
              f"{t16 * 1e6:7.1f} ({fl / t16 / 1e12:5.1f})  "
              f"{trb * 1e6:7.1f} ({fl / trb / 1e12:5.1f})")
        # TN (wgrad) shape: [n, m] x [m, k]
        g = torch.randn(m, n, device=dev)
        ttn32 = bench(C.gemm_tn, args.iters, g, a)
        ttn16 = bench(C.gemm_tn_bf16, args.iters, g, a)
        ttnrb = bench(lambda g, a: g.t() @ a, args.iters, g, a)
        print(f"{'  wgrad':<12} {m:>6} {n:>5} {k:>4}  "
              f"{ttn32 * 1e6:7.1f} ({fl / ttn32 / 1e12:5.1f})  "
              f"{ttn16 * 1e6:7.1f} ({fl / ttn16 / 1e12:5.1f})  "
              f"{ttnrb * 1e6:7.1f} ({fl / ttnrb / 1e12:5.1f})")


if __name__ == "__main__":
    main()
