"""Probe: flash path vs composed bmm+wave64-softmax at the bench shapes,
fwd-only and fwd+bwd separately — diagnoses the round-2 BERT regression
(flash wiring made the bench SLOWER: 534K vs 561K tok/s)."""

import math
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000.0


def main():
    from apex_amd.transformer import (flash_attention, scaled_masked_softmax,
                                      scaled_softmax,
                                      scaled_upper_triang_masked_softmax)
    from apex_amd.transformer.fmha import FlashAttentionFunction
    import apex_amd._mfma as mfma

    shapes = [(128, 12, 512, 64, False),   # BERT bench per-layer shape
              (32, 16, 1024, 64, True),    # GPT-2 345M shape
              (16, 16, 512, 64, True)]     # transformer-large shape
    for B, H, S, D, causal in shapes:
        torch.manual_seed(0)
        scale = 1.0 / math.sqrt(D)
        q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
        k = torch.randn_like(q, requires_grad=True)
        v = torch.randn_like(q, requires_grad=True)
        dout = torch.randn_like(q)

        def composed_fwd():
            s = torch.matmul(q, k.transpose(-1, -2))
            if causal:
                p = scaled_upper_triang_masked_softmax(
                    s.view(B * H, S, S), scale).view(B, H, S, S)
            else:
                p = scaled_softmax(s.contiguous(), scale)
            return torch.matmul(p, v)

        def composed_fb():
            out = composed_fwd()
            gq, gk, gv = torch.autograd.grad(out, (q, k, v), dout)
            return gq

        def flash_fwd():
            return flash_attention(q, k, v, causal=causal, scale=scale)

        def flash_fb():
            out = flash_attention(q, k, v, causal=causal, scale=scale)
            gq, gk, gv = torch.autograd.grad(out, (q, k, v), dout)
            return gq

        # raw kernels without autograd plumbing
        qc, kc, vc = q.detach(), k.detach(), v.detach()

        def raw_fwd():
            return mfma.fmha_fwd(qc, kc, vc, causal, scale)

        out_r, lse_r = raw_fwd()

        def raw_bwd():
            return mfma.fmha_bwd(dout, qc, kc, vc, out_r, lse_r, causal, scale)

        t_cf = timeit(composed_fwd)
        t_cb = timeit(composed_fb)
        t_ff = timeit(flash_fwd)
        t_fb = timeit(flash_fb)
        t_rf = timeit(raw_fwd)
        t_rb = timeit(raw_bwd)
        print(f"B{B} H{H} S{S} D{D} causal={int(causal)}:")
        print(f"  composed fwd {t_cf:7.3f} ms   fwd+bwd {t_cb:7.3f} ms")
        print(f"  flash    fwd {t_ff:7.3f} ms   fwd+bwd {t_fb:7.3f} ms")
        print(f"  raw kern fwd {t_rf:7.3f} ms   bwd(only) {t_rb:7.3f} ms")
        print(f"  -> fwd speedup {t_cf / t_ff:.2f}x, fwd+bwd speedup {t_cb / t_fb:.2f}x",
              flush=True)


if __name__ == "__main__":
    main()
