"""Time the custom decode GEMMs vs F.linear (tuned library) for the decode
projection shapes, cold-L3 (2 GB weight rotation). Run on the GPU box.

Arms:
  m256    — macro-tile LDS-staged kernel (csrc/gemm_m256.hip, fragment twin)
  library — TunableOp-tuned hipBLASLt/rocBLAS
  skinny  — round-1 streaming kernel (only with --skinny; historical)

Usage: python tools/gemm_probe.py [M N K] [--nf 4|8] [--nsk N] [--skinny]
"""

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
    (16, 6144, 4096),
    (1, 4096, 14336),
]


def bench(fn, iters=50):
    # rotate weight copies so W is cold in cache. CAVEAT: the rotation cap
    # must exceed the 256 MB Infinity Cache (L3) or rows measure the
    # L3-warm regime, NOT the engine's L3-cold weight cycle (this bit us
    # once: an L3-warm "parity" result regressed batch-256 decode 2x in
    # situ).
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
    args = [a for a in sys.argv[1:] if not a.startswith("--")]
    flags = [a for a in sys.argv[1:] if a.startswith("--")]
    nf_force = nsk_force = None
    with_skinny = "--skinny" in flags
    for f in flags:
        if f.startswith("--nf"):
            nf_force = int(f.split("=")[1]) if "=" in f else None
        if f.startswith("--nsk"):
            nsk_force = int(f.split("=")[1]) if "=" in f else None
    if len(args) == 3:  # probe a single shape (for rocprofv3 --pmc runs)
        SHAPES = [tuple(int(a) for a in args)]
    hdr = (f"{'M':>4} {'N':>7} {'K':>6} {'p0 us':>8} {'p3 us':>8} {'reg us':>8} "
           f"{'library us':>11} {'lib/best':>9} {'TB/s':>6}")
    if with_skinny:
        hdr += f" {'skinny us':>10}"
    print(hdr)
    for (M, N, K) in SHAPES:
        x = (torch.randn(M, K, device=dev) * 0.5).bfloat16()
        ncopies = max(1, min(16, (2048 << 20) // (N * K * 2)))  # rotate >2 GB
        ws = [(torch.randn(N, K, device=dev) * 0.02).bfloat16() for _ in range(ncopies)]
        ref = x.float() @ ws[0].float().T
        wf = [ops.swizzle_weight_frag(w) for w in ws]
        checks = [dict(variant=0, pipe=0), dict(variant=1)]
        if N % 128 == 0:
            checks += [dict(variant=0, nf=8, pipe=2), dict(variant=0, nf=8, pipe=3)]
        for kw in checks:
            kw.setdefault("nf", nf_force)
            got = ops.gemm_m256(x, wf[0], nsk=nsk_force, **kw).float()
            err = (got - ref).abs().max().item() / (ref.abs().max().item() + 1e-3)
            assert err < 0.02, f"gemm_m256 {kw} wrong for {(M,N,K)}: rel {err}"
        for _ in range(5):
            ops.gemm_m256(x, wf[0], nf=nf_force, nsk=nsk_force); F.linear(x, ws[0])
        t_g = bench(lambda i: ops.gemm_m256(x, wf[i % ncopies], nf=nf_force, nsk=nsk_force, variant=0, pipe=0))
        if N % 128 == 0:
            t_p = bench(lambda i: ops.gemm_m256(x, wf[i % ncopies], nf=8, nsk=nsk_force, variant=0, pipe=3))
        else:
            t_p = float("nan")
        t_r = bench(lambda i: ops.gemm_m256(x, wf[i % ncopies], nf=nf_force, nsk=nsk_force, variant=1))
        t_l = bench(lambda i: F.linear(x, ws[i % ncopies]))
        cand = [t for t in (t_g, t_p, t_r) if t == t]
        t_m = min(cand)
        tbs = N * K * 2 / t_m / 1e6  # effective W-stream rate
        line = (f"{M:>4} {N:>7} {K:>6} {t_g:>8.1f} {t_p:>8.1f} {t_r:>8.1f} {t_l:>11.1f} "
                f"{t_l / t_m:>9.2f} {tbs:>6.2f}")
        if with_skinny:
            got_s = ops.linear(x, ws[0]).float()
            err_s = (got_s - ref).abs().max().item() / (ref.abs().max().item() + 1e-3)
            assert err_s < 0.02, f"skinny GEMM wrong for {(M,N,K)}: rel {err_s}"
            t_s = bench(lambda i: ops.linear(x, ws[i % ncopies]))
            line += f" {t_s:>10.1f}"
        print(line)


if __name__ == "__main__":
    main()
