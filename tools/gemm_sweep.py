"""Exhaustive config sweep for gemm_m256 on the decode projection shapes,
cold-L3 (2 GB weight rotation). Prints every (variant, nf, pipe, nsk)
combo's time so the production dispatch table (ops/__init__.py) can be
set from measurements, not theory. Run on the GPU box:

    python tools/gemm_sweep.py [M N K]
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

from llmapigateway_amd import ops

SHAPES = [
    (256, 6144, 4096),
    (256, 4096, 4096),
    (256, 28672, 4096),
    (256, 4096, 14336),
]


def bench(fn, iters=30):
    # warmup OUTSIDE the timed region: first-call hipBLASLt algo selection /
    # TunableOp tuning on a never-seen shape costs 100s of ms and would
    # otherwise dominate the 30-iter average.
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
    global SHAPES
    if len(sys.argv) == 4:
        SHAPES = [tuple(int(a) for a in sys.argv[1:4])]
    for (M, N, K) in SHAPES:
        x = (torch.randn(M, K, device=dev) * 0.5).bfloat16()
        ncopies = max(1, min(16, (2048 << 20) // (N * K * 2)))
        ws = [(torch.randn(N, K, device=dev) * 0.02).bfloat16() for _ in range(ncopies)]
        wf = [ops.swizzle_weight_frag(w) for w in ws]
        ref = x.float() @ ws[0].float().T
        t_l = bench(lambda i: F.linear(x, ws[i % ncopies]))
        print(f"--- M={M} N={N} K={K}  library {t_l:.1f} us ---")
        tiles4, tiles8 = N // 64, N // 128 if N % 128 == 0 else 0
        combos = []
        for nf, tiles in ((4, tiles4), (8, tiles8)):
            if not tiles:
                continue
            nsks = sorted({1, max(1, -(-256 // tiles)), max(1, -(-512 // tiles))})
            nsks = [s for s in nsks if s <= max(1, (K // 64) // 2) and s <= 8]
            for nsk in nsks:
                pipes = [0] + ([1, 4, 5] if nf == 4 else [2, 3])
                for pipe in pipes:
                    combos.append((0, nf, pipe, nsk))
                combos.append((1, nf, 0, nsk))
                combos.append((1, nf, 1, nsk))  # non-temporal W stream
        # producer/consumer variant (BN=96, nf ignored)
        nsks2 = sorted({1, max(1, -(-256 // max(1, N // 96)))})
        for nsk in nsks2:
            if nsk <= max(1, (K // 64) // 2) and nsk <= 8:
                combos.append((2, 4, 0, nsk))
        best = (None, 1e18)
        for (v, nf, pipe, nsk) in combos:
            try:
                got = ops.gemm_m256(x, wf[0], nf=nf, nsk=nsk, variant=v, pipe=pipe).float()
            except Exception as e:
                print(f"  v{v} nf{nf} p{pipe} nsk{nsk}: LAUNCH FAIL {e}")
                continue
            err = (got - ref).abs().max().item() / (ref.abs().max().item() + 1e-3)
            if err > 0.02:
                print(f"  v{v} nf{nf} p{pipe} nsk{nsk}: WRONG rel={err}")
                continue
            t = bench(lambda i: ops.gemm_m256(x, wf[i % ncopies], nf=nf, nsk=nsk, variant=v, pipe=pipe))
            tag = f"v{v} nf{nf} p{pipe} nsk{nsk}"
            print(f"  {tag:20s} {t:8.1f} us   ({N*K*2/t/1e6:.2f} TB/s, lib/x {t_l/t:.2f})")
            if t < best[1]:
                best = (tag, t)
        print(f"  BEST: {best[0]} {best[1]:.1f} us vs library {t_l:.1f}")


if __name__ == "__main__":
    main()
