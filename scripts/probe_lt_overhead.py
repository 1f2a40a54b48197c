"""Isolate per-call host overhead of each GEMM path in the BERT step."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import apex_amd._fused_dense as fd


def timeit(name, fn, iters=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters * 1000
    # host-side issue time (no sync)
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    host = (time.perf_counter() - t0) / iters * 1000
    torch.cuda.synchronize()
    print(f"{name:45s} wall {dt:8.3f} ms   host-issue {host:8.3f} ms")


def main():
    torch.manual_seed(0)
    dt = torch.bfloat16
    m = 16384
    x768 = torch.randn(m, 768, device="cuda", dtype=dt)
    w_qkv = torch.randn(2304, 768, device="cuda", dtype=dt)
    b_qkv = torch.randn(2304, device="cuda", dtype=dt)
    w_mlp1 = torch.randn(3072, 768, device="cuda", dtype=dt)
    b_mlp1 = torch.randn(3072, device="cuda", dtype=dt)
    w_mlp2 = torch.randn(768, 3072, device="cuda", dtype=dt)
    b_mlp2 = torch.randn(768, device="cuda", dtype=dt)
    emb = torch.randn(30528, 768, device="cuda", dtype=dt)

    timeit("fd.linear_bias_forward qkv [16k,768->2304]", lambda: fd.linear_bias_forward(x768, w_qkv, b_qkv))
    timeit("fd.linear_gelu_linear fwd  [768->3072->768]",
           lambda: fd.linear_gelu_linear_forward(x768, w_mlp1, b_mlp1, w_mlp2, b_mlp2))
    x3072 = torch.randn(m, 3072, device="cuda", dtype=dt)
    timeit("fd.linear_bias_backward mlp2", lambda: fd.linear_bias_backward(x3072, w_mlp2, x768))
    timeit("torch.matmul LM head [16k,768]@[768,30k]", lambda: torch.matmul(x768, emb.t()))
    q = torch.randn(32, 12, 512, 64, device="cuda", dtype=dt)
    k = torch.randn(32, 12, 512, 64, device="cuda", dtype=dt)
    timeit("torch.matmul q@kT  [32,12,512,64]", lambda: torch.matmul(q, k.transpose(-2, -1)))
    probs = torch.randn(32, 12, 512, 512, device="cuda", dtype=dt)
    timeit("torch.matmul probs@v", lambda: torch.matmul(probs, k))
    # LM head backward-shaped GEMMs
    dy = torch.randn(m, 30528, device="cuda", dtype=dt)
    timeit("torch.matmul dY@emb [16k,30k]@[30k,768]", lambda: torch.matmul(dy, emb))
    timeit("torch.matmul dY.T@x wgrad [30k,16k]@[16k,768]", lambda: torch.matmul(dy.t(), x768))


if __name__ == "__main__":
    main()
