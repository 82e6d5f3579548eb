"""dts_amd — MI355X-native Dialogue Tree Search framework.

A from-scratch reimplementation of the capability surface of MVPandey/DTS
(reference layer map: SURVEY.md §1) with all LLM inference served locally on
AMD Instinct MI355X GPUs: PyTorch-ROCm weights + hand-written CDNA4 HIP
kernels, paged shared-prefix KV cache, continuous batching, TP over
RCCL/xGMI and DP branch sharding.

Layers (mirroring reference /root/reference layering):
  dts_amd.utils    — config + logging                (ref backend/utils/)
  dts_amd.llm      — backend-agnostic LLM interface  (ref backend/llm/)
  dts_amd.search   — the DTS search engine           (ref backend/core/dts/)
  dts_amd.serving  — MI355X serving engine           (new: replaces OpenRouter HTTP)
  dts_amd.models   — model definitions (Llama, GPT-2, Mixtral)
  dts_amd.ops      — HIP/CDNA4 kernels + torch references
  dts_amd.parallel — RCCL/xGMI TP + DP branch sharding
  dts_amd.server   — FastAPI WS/REST server          (ref backend/api/)
"""

__version__ = "0.1.0"
