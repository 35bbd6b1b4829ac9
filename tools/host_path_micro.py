"""Host-side engine bookkeeping cost per decode step: the model forward
and sampling are stubbed with cached GPU tensors, so what remains is the
scheduler/bookkeeping python (numpy gathers, delivery loops, flushes).
Run on the GPU box: python tools/host_path_micro.py"""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from llmapigateway_amd.engine import LLMEngine, EngineRequest, SamplingParams

eng = LLMEngine(model="tiny-llama", device="cuda:0", dtype=torch.bfloat16,
                max_batch_size=256, max_model_len=512, num_blocks=4096,
                use_hipgraph=False)
B = 256
logits = torch.randn(B, eng.config.vocab_size, device="cuda:0")

class FakeModel:
    def forward(self, batch, *a, **k):
        n = batch.token_ids.shape[0]
        return logits[:n]

eng.model.forward = FakeModel().forward
reqs = [EngineRequest(list(range(5, 37)), SamplingParams(max_tokens=10_000, ignore_eos=True))
        for _ in range(B)]
for r in reqs:
    eng.add_request(r)
eng.step()  # prefill
for _ in range(10):
    eng.step()
torch.cuda.synchronize()
t0 = time.monotonic()
N = 200
for _ in range(N):
    eng.step()
torch.cuda.synchronize()
us = (time.monotonic() - t0) / N * 1e6
print(f"host-side decode step cost at b{B}: {us:.0f} us/step")
for r in reqs:
    eng.abort_request(r)
