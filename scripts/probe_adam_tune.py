"""Sweep streaming-kernel configs for the Adam memory pattern (28 B/elem)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import apex_amd._tune as tune

def main():
    n = 128 * 1024 * 1024  # 512 MB per stream x4 = beyond L3
    g = torch.randn(n, device="cuda")
    p = torch.randn(n, device="cuda")
    m = torch.zeros(n, device="cuda")
    v = torch.zeros(n, device="cuda")
    names = ["256/4", "256/8", "512/4", "512/8", "1024/4", "128/8", "256/16", "512/16"]
    bytes_moved = n * 28.0
    for vid, name in enumerate(names):
        for _ in range(2):
            tune.adam_stream_probe(g, p, m, v, vid, 0)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(10):
            tune.adam_stream_probe(g, p, m, v, vid, 0)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / 10
        print(f"block/ilp {name:8s}: {dt*1e3:7.3f} ms  {bytes_moved/dt/1e12:6.2f} TB/s")
    # also: grid-stride capped at 2048 blocks (guide G11) for the best two
    for vid, name in [(1, "256/8 cap2048"), (3, "512/8 cap2048")]:
        for _ in range(2):
            tune.adam_stream_probe(g, p, m, v, vid, 2048)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(10):
            tune.adam_stream_probe(g, p, m, v, vid, 2048)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / 10
        print(f"{name:16s}: {dt*1e3:7.3f} ms  {bytes_moved/dt/1e12:6.2f} TB/s")

if __name__ == "__main__":
    main()
