"""Isolated vs production-shape GEMM timing at the batch-1024 flagship shapes.

The step profile shows the three model GEMMs at 410/406/265 us in-context
(211/213/327 TF).  This bench times the SAME entry points the training step
uses (bf16-in fused QKVS forward, bf16-in dgrad/wgrad) on the same shapes in
isolation, to separate kernel quality from in-context cache effects.

    python benchmarks/gemm_ctx_bench.py [--iters 50] [--m 165317]
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
    ap.add_argument("--m", type=int, default=165317)
    ap.add_argument("--h", type=int, default=256)
    args = ap.parse_args()
    import pertgnn._C as C

    dev = torch.device("cuda:0")
    m, h = args.m, args.h
    n = 4 * h  # fused qkvs width

    x = torch.randn(m, h, device=dev)            # fp32 activations
    w4 = torch.randn(n, h, device=dev)           # fp32 weights
    b4 = torch.randn(n, device=dev)
    g16 = torch.randn(m, n, device=dev).bfloat16()
    x16 = x.bfloat16()

    fl = 2.0 * m * n * h

    def row(name, t, bytes_):
        print(f"{name:<28} {t*1e6:8.1f} us  {fl/t/1e12:6.1f} TF  "
              f"{bytes_/t/1e12:5.2f} TB/s")

    # forward: qkvs = x @ w4^T  (f32 in, bf16 compute, bf16 out)
    t = bench(C.linear_fwd_bf16_o16, args.iters, x, w4, b4)
    row("fwd NT  f32->bf16", t, (m*h*4 + n*h*4 + m*n*2))
    # dgrad: dx = g16 @ w4  (bf16 g, f32 w, f32 out)
    t = bench(C.linear_dgrad16, args.iters, g16, w4)
    row("dgrad NN bf16->f32", t, (m*n*2 + n*h*4 + m*h*4))
    # wgrad: dw = g16^T @ x16 (bf16 in, f32 out + dbias)
    t = bench(C.linear_wgrad16, args.iters, g16, x, True)
    row("wgrad TN g16/x32->f32", t, (m*n*2 + m*h*4 + n*h*4))

    # act16-v2 production entries (bf16 A / bf16 C)
    t = bench(C.linear_fwd_a16o16, args.iters, x16, w4, b4)
    row("fwd NT  a16->bf16", t, (m*h*2 + n*h*4 + m*n*2))
    t = bench(C.linear_dgrad16_o16, args.iters, g16, w4)
    row("dgrad NN bf16->bf16", t, (m*n*2 + n*h*4 + m*h*2))
    t = bench(C.linear_wgrad16_b16, args.iters, g16, x16, True)
    row("wgrad TN a16b16->f32", t, (m*n*2 + m*h*2 + n*h*4))

    # rocBLAS reference on bf16
    wt16 = w4.bfloat16()
    t = bench(lambda: x16 @ wt16.t(), args.iters)
    row("rocBLAS fwd NT bf16", t, (m*h*2 + n*h*2 + m*n*2))
    t = bench(lambda: g16 @ wt16, args.iters)
    row("rocBLAS dgrad NN bf16", t, (m*n*2 + n*h*2 + m*h*2))
    t = bench(lambda: g16.t() @ x16, args.iters)
    row("rocBLAS wgrad TN bf16", t, (m*n*2 + m*h*2 + n*h*2))


if __name__ == "__main__":
    main()
