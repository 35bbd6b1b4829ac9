"""llmapigateway_amd — MI355X-native fault-tolerant LLM API gateway.

A ground-up rebuild of the capabilities of fabiojbg/LLMApiGateway (an
OpenAI-compatible fault-tolerant HTTP LLM gateway) where each configured
"provider" may resolve either to an upstream HTTP endpoint or to a local
GPU-resident inference engine running on AMD Instinct MI355X (gfx950):

- control plane: FastAPI app with the reference's API surface
  (/v1/chat/completions, /v1/models, config editor, usage stats) and config
  schema (providers.json / models_fallback_rules.json with comments);
- data plane: hand-written CDNA4 HIP kernels (RMSNorm, RoPE, prefill/decode
  attention on MFMA, sampling), paged KV cache sized for 288 GB HBM3E,
  HIP-stream scheduling with hipGraph-captured decode, and RCCL-over-xGMI
  tensor parallelism.

Reference layer map: see SURVEY.md in the repo root.
"""

__version__ = "0.1.0"
