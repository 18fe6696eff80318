"""mlx_sharding_amd — MI355X-native pipeline-parallel LLM inference engine.

A from-scratch framework with the capabilities of mzbac/mlx_sharding
(reference layout: /root/reference), re-designed for AMD Instinct MI355X:
PyTorch-ROCm host runtime, hand-written HIP/CDNA4 kernels (MFMA + LDS
tiling) for the hot ops, and RCCL point-to-point over xGMI for the
stage-to-stage hidden-state hops (the reference used gRPC unary RPC;
a wire-compatible gRPC mode is retained for CPU plumbing and the
control plane).

Layout:
  models/    sharded model wrappers (llama, gemma2, deepseek_v2)
  ops/       compute ops: torch reference impls + HIP/CDNA4 kernels
  parallel/  transports (gRPC-compat, RCCL) + pipeline engine
  utils/     loading, sampling, detokenizer
  server/    shard server + OpenAI-compatible API + web UI
  cli/       console entry points
"""

__version__ = "0.1.0"
