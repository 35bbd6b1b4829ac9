"""Time ops.linear (custom skinny-M streaming GEMM) vs F.linear (tuned
library) for the decode projection shapes. Run on the GPU box."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

from llmapigateway_amd import ops

# force the custom kernel for the "custom" column regardless of the
# production dispatch threshold (without this, M > _SKINNY_MAX_M silently
# measures the library against itself — which fooled us once)
ops._SKINNY_MAX_M = 1_000_000

SHAPES = [
    (256, 6144, 4096),    # qkv
    (256, 4096, 4096),    # o
    (256, 28672, 4096),   # gate_up
    (256, 4096, 14336),   # down
    (256, 128256, 4096),  # lm_head
    (64, 6144, 4096),
    (64, 4096, 14336),
    (64, 28672, 4096),
    (1, 4096, 14336),
]


def bench(fn, iters=50):
    # rotate weight copies so W is cold in cache. CAVEAT: the 256 MB
    # rotation cap leaves >100 MB weights resident in the 256 MB Infinity
    # Cache (L3) — such rows measure the L3-warm regime, NOT the engine's
    # L3-cold weight cycle. Cross-check any dispatch decision against an
    # in-engine bench (this bit us once: an L3-warm "parity" result
    # regressed batch-256 decode 2x in situ).
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
    if len(sys.argv) == 4:  # probe a single shape (for rocprofv3 --pmc runs)
        SHAPES = [tuple(int(a) for a in sys.argv[1:4])]
    print(f"{'M':>4} {'N':>7} {'K':>6} {'custom us':>10} {'swz us':>9} {'library us':>11} {'lib/swz':>8}")
    for (M, N, K) in SHAPES:
        x = (torch.randn(M, K, device=dev) * 0.5).bfloat16()
        ncopies = max(1, min(16, (2048 << 20) // (N * K * 2)))  # rotate >2 GB: defeat the 256 MB L3
        ws = [(torch.randn(N, K, device=dev) * 0.02).bfloat16() for _ in range(ncopies)]
        # correctness spot check
        ref = x.float() @ ws[0].float().T
        got = ops.linear(x, ws[0]).float()
        err = (got - ref).abs().max().item() / (ref.abs().max().item() + 1e-3)
        assert err < 0.02, f"skinny GEMM wrong for {(M,N,K)}: rel {err}"
        wz = [ops.swizzle_weight(w) for w in ws]
        got_s = ops.linear(x, ws[0], wz[0]).float()
        err_s = (got_s - ref).abs().max().item() / (ref.abs().max().item() + 1e-3)
        assert err_s < 0.02, f"swizzled GEMM wrong for {(M,N,K)}: rel {err_s}"
        for _ in range(5):
            ops.linear(x, ws[0]); F.linear(x, ws[0]); ops.linear(x, ws[0], wz[0])
        t_c = bench(lambda i: ops.linear(x, ws[i % ncopies]))
        t_s = bench(lambda i: ops.linear(x, ws[i % ncopies], wz[i % ncopies]))
        t_l = bench(lambda i: F.linear(x, ws[i % ncopies]))
        print(f"{M:>4} {N:>7} {K:>6} {t_c:>10.1f} {t_s:>9.1f} {t_l:>11.1f} {t_l / t_s:>8.2f}")


if __name__ == "__main__":
    main()
