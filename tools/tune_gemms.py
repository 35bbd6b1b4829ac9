"""Offline hipBLASLt/rocBLAS GEMM tuning sweep for the engine's hot shapes.

Runs on a GPU box (via gpurun), with PyTorch TunableOp in *tuning* mode:
every F.linear shape the engine can emit — each decode batch bucket and the
prefill token-budget sizes, for every projection of every served model — is
executed once so TunableOp benchmarks all backend solutions and records the
winner. The resulting table is copied to
llmapigateway_amd/ops/tunableop_gfx950<ordinal>.csv for all 8 ordinals
(merged back into the repo by gpurun, committed, and shipped with the
snapshot; ops/__init__.py activates it at import).

Usage (on the GPU box):
    python tools/tune_gemms.py [--models llama-3-8b mistral-7b] [--out DIR]
"""

from __future__ import annotations

import argparse
import os
import shutil
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

OUT_TMP = os.environ.get(
    "TUNE_OUT", os.path.join(os.getcwd(), "gpurun_out", "tunableop_tune.csv")
)
os.makedirs(os.path.dirname(OUT_TMP), exist_ok=True)

# Tuning mode must be configured before the first GEMM.
os.environ["PYTORCH_TUNABLEOP_ENABLED"] = "1"
os.environ["PYTORCH_TUNABLEOP_TUNING"] = "1"
os.environ["PYTORCH_TUNABLEOP_FILENAME"] = OUT_TMP
os.environ.setdefault("PYTORCH_TUNABLEOP_MAX_TUNING_DURATION_MS", "100")
os.environ.setdefault("PYTORCH_TUNABLEOP_MAX_TUNING_ITERATIONS", "30")
# Rotate through a >L2-sized pool of A/B/C buffers while benchmarking each
# solution: the engine streams 0.5 GB of cold weights per layer, so a
# hot-cache micro-bench picks cache-exploiting solutions that lose in situ.
os.environ.setdefault("PYTORCH_TUNABLEOP_ROTATING_BUFFER_SIZE", "1024")
# do not read any pre-existing table: tune from scratch
os.environ["PYTORCH_TUNABLEOP_TUNING_AFTER_READ"] = "1"

import torch  # noqa: E402
import torch.nn.functional as F  # noqa: E402

from llmapigateway_amd.models.configs import get_model_config  # noqa: E402

# llmapigateway_amd.ops may have flipped TunableOp into lookup mode at
# import; force tuning mode back on for this sweep.
import torch.cuda.tunable as tunable  # noqa: E402

tunable.enable(True)
tunable.tuning_enable(True)
tunable.set_filename(OUT_TMP, insert_device_ordinal=True)

# decode batch buckets (engine/graph_runner.py) + prefill flat-token sizes
# (engine budget_tokens=8192 full waves, plus common partial waves)
DECODE_MS = [1, 2, 4, 8, 16, 24, 32, 48, 64, 96, 128, 192, 256]
PREFILL_MS = [512, 1024, 2048, 4096, 8192]


def shapes_for(model_name: str, tp: int = 1):
    c = get_model_config(model_name)
    if tp > 1:
        c = c.scaled_for_tp(tp)
    h = c.hidden_size
    proj = [
        (c.q_size + 2 * c.kv_size, h),   # qkv (column-sharded under TP)
        (h, c.q_size),                   # o (row-sharded)
        (2 * c.intermediate_size, h),    # gate_up
        (h, c.intermediate_size),        # down
    ]
    head = [(c.vocab_size, h)]           # lm_head (decode buckets + prefill last-rows)
    out = []
    for m in DECODE_MS + PREFILL_MS:
        for (n, k) in proj:
            out.append((m, n, k))
    for m in DECODE_MS:
        for (n, k) in head:
            out.append((m, n, k))
    return out


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--models", nargs="+", default=["llama-3-8b", "mistral-7b"])
    ap.add_argument(
        "--tp-models", nargs="+", default=[],
        help="model:tp entries to tune per-rank sharded shapes, e.g. llama-3-70b:8",
    )
    ap.add_argument("--out", default=os.path.join(REPO, "llmapigateway_amd", "ops"))
    args = ap.parse_args()

    assert torch.cuda.is_available(), "tuning needs a GPU"
    dev = torch.device("cuda:0")
    seen = set()
    todo = []
    for m in args.models:
        for s in shapes_for(m):
            if s not in seen:
                seen.add(s)
                todo.append(s)
    for ent in args.tp_models:
        m, _, tp = ent.partition(":")
        for s in shapes_for(m, int(tp or 1)):
            if s not in seen:
                seen.add(s)
                todo.append(s)
    print(f"tuning {len(todo)} unique (M,N,K) bf16 TN shapes")

    for i, (M, N, K) in enumerate(todo):
        x = torch.randn(M, K, dtype=torch.bfloat16, device=dev)
        w = torch.randn(N, K, dtype=torch.bfloat16, device=dev)
        F.linear(x, w)  # first call triggers tuning for this shape
        torch.cuda.synchronize()
        if (i + 1) % 10 == 0:
            print(f"  {i + 1}/{len(todo)}")

    # results are flushed to OUT_TMP (with device ordinal inserted) by
    # TunableOp's exit handler; this version of torch has no write_file().
    print(f"{len(tunable.get_results())} tuning results recorded; "
          f"table written at exit to {OUT_TMP.replace('.csv', '0.csv')}")
    _ = shutil  # keep import for future use


if __name__ == "__main__":
    main()
