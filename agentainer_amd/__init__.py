"""agentainer_amd — MI355X-native multi-tenant LLM-agent runtime.

A from-scratch rebuild of the capabilities of oso95/Agentainer-lab (a Go
container-runtime-for-LLM-agents control plane) as an in-process inference
runtime for AMD Instinct MI355X (gfx950, CDNA4):

  * The Docker-container-per-agent data plane (reference: pkg/docker,
    internal/agent) is replaced by an in-process multi-tenant inference
    engine: each deployed "agent" is a model-shard binding plus a
    conversation KV-cache resident in HBM3E.
  * The request write-ahead log and crash-replay semantics
    (reference: internal/requests) are preserved: pending/completed/failed,
    retry<=3, dead-letter, 24h TTL, at-least-once replay.
  * Hot ops (RMSNorm, RoPE, paged attention prefill/decode, sampling,
    KV-cache append) are hand-written CDNA4 HIP kernels (agentainer_amd.ops).
  * Tensor-parallel all-reduce and MoE all-to-all run on RCCL over xGMI via
    torch.distributed (backend "nccl" == RCCL on ROCm).

Reference layer map and behavioral contract: /root/repo/SURVEY.md.
"""

__version__ = "0.1.0"
