"""Standalone rmsnorm/rmsnorm_residual timing at decode and prefill shapes.
Run on the GPU box: python tools/rmsnorm_micro.py"""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from llmapigateway_amd import ops

def bench(fn, iters=200):
    for _ in range(5): fn()
    torch.cuda.synchronize()
    t0 = time.monotonic()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.monotonic() - t0) / iters * 1e6

def main():
    dev = "cuda:0"
    for T in (64, 128, 256, 2048, 8192):
        H = 4096
        x = torch.randn(T, H, device=dev).bfloat16()
        res = torch.randn(T, H, device=dev).bfloat16()
        w = torch.randn(H, device=dev).bfloat16()
        t1 = bench(lambda: ops.rmsnorm(x, w))
        t2 = bench(lambda: ops.rmsnorm_residual(x, res, w))
        gb = T * H * 2 * 2 / 1e9  # plain: read+write
        gbr = T * H * 2 * 4 / 1e9
        print(f"T={T:5d}: rmsnorm {t1:6.2f} us ({gb/t1*1e6:6.1f} GB/s)  "
              f"residual {t2:6.2f} us ({gbr/t2*1e6:6.1f} GB/s)")

if __name__ == "__main__":
    main()
