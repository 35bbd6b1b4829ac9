"""Standalone decode-attention microbench: kernel time + effective KV
stream rate at the llama-3-8b head config. Usage: python tools/decode_attn_micro.py [ctx] [B]"""
import sys, time, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from llmapigateway_amd import ops

def run(ctx, B=256, ver=None):
    dev = "cuda:0"
    Hq, Hkv, BS, D = 32, 8, 64, 128
    maxb = (ctx + BS - 1) // BS
    NB = B * maxb + 8
    kc = torch.randn(NB, Hkv, BS, D, dtype=torch.bfloat16, device=dev)
    vc = torch.randn_like(kc)
    bt = torch.randperm(NB)[: B * maxb].view(B, maxb).int().to(dev)
    q = torch.randn(B, Hq, D, dtype=torch.bfloat16, device=dev)
    lens = torch.full((B,), ctx, dtype=torch.int32, device=dev)
    if ver is not None:
        ops._DECODE_VER = ver
    for _ in range(5):
        ops.attention_decode(q, kc, vc, bt, lens)
    torch.cuda.synchronize()
    t0 = time.monotonic()
    iters = 200
    for _ in range(iters):
        ops.attention_decode(q, kc, vc, bt, lens)
    torch.cuda.synchronize()
    us = (time.monotonic() - t0) / iters * 1e6
    kv_bytes = B * ctx * Hkv * D * 2 * 2
    print(f"B={B} ctx={ctx} v{ops._DECODE_VER}: {us:7.1f} us  {kv_bytes/us/1e6:5.2f} TB/s effective")

if __name__ == "__main__":
    if len(sys.argv) > 1:
        run(int(sys.argv[1]), int(sys.argv[2]) if len(sys.argv) > 2 else 256)
    else:
        for v in (2, 3):
            for c in (192, 512, 2048, 8192):
                run(c, ver=v)
