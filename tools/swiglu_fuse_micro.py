"""Fused gate_up+swiglu (gemm_m256 SWIGLU epilogue) vs library GEMM +
separate swiglu kernel, cold-L3 (2 GB weight rotation). Run on the GPU box:

    python tools/swiglu_fuse_micro.py [M [N K]]
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

from llmapigateway_amd import ops

SHAPES = [(28672, 4096), (57344, 8192)]  # 8b / 70b stacked gate_up


def bench(fn, iters=30):
    for i in range(3):
        fn(i)
    torch.cuda.synchronize()
    t0 = time.monotonic()
    for i in range(iters):
        fn(i)
    torch.cuda.synchronize()
    return (time.monotonic() - t0) / iters * 1e6


def main():
    assert torch.cuda.is_available()
    dev = "cuda:0"
    M = int(sys.argv[1]) if len(sys.argv) > 1 else 256
    shapes = [tuple(int(a) for a in sys.argv[2:4])] if len(sys.argv) == 4 else SHAPES
    for (N, K) in shapes:
        x = (torch.randn(M, K, device=dev) * 0.5).bfloat16()
        ncopies = max(1, min(16, (2048 << 20) // (N * K * 2)))
        ws = [(torch.randn(N, K, device=dev) * 0.02).bfloat16() for _ in range(ncopies)]
        wi = [ops.swizzle_weight_frag(ops.interleave_gate_up(w)) for w in ws]
        g, u = (x.float() @ ws[0].float().T).chunk(2, dim=-1)
        ref = F.silu(g) * u
        t_lib = bench(lambda i: ops.swiglu(F.linear(x, ws[i % ncopies])))
        print(f"--- M={M} N={N} K={K}  library+swiglu {t_lib:.1f} us ---")
        for nf in (4, 8):
            pipes = {4: [0, 1, 4, 5], 8: [0, 2, 3]}[nf]
            for variant, pipe in [(1, 0), (1, 1)] + [(0, p) for p in pipes]:
                try:
                    got = ops.gemm_m256_swiglu(x, wi[0], nf=nf, variant=variant, pipe=pipe)
                except Exception as e:
                    print(f"  fused v{variant} nf{nf} p{pipe}: FAIL {e}")
                    continue
                err = (got.float() - ref).abs().max().item() / (ref.abs().max().item() + 1e-3)
                if err > 0.02:
                    print(f"  fused v{variant} nf{nf} p{pipe}: WRONG rel={err}")
                    continue
                t = bench(lambda i: ops.gemm_m256_swiglu(
                    x, wi[i % ncopies], nf=nf, variant=variant, pipe=pipe))
                print(f"  fused v{variant} nf{nf} p{pipe}: {t:8.1f} us  (lib/x {t_lib/t:.2f})")


if __name__ == "__main__":
    main()
