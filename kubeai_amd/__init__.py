"""kubeai_amd — an MI355X-native Kubernetes-style AI serving stack.

Two halves, mirroring the capability surface of the reference
(kubeai-project/kubeai):

- ``kubeai_amd.controlplane``: Model spec + controller + OpenAI gateway/proxy +
  prefix-aware load balancer (CHWBL) + concurrency autoscaler + messenger —
  the operator logic of the reference (reference: internal/*), re-built as a
  process-level operator (the reference's Pods become engine processes; the
  same manifests ship under deploy/ for cluster use).
- ``kubeai_amd.engine``: the in-house per-Model inference server the reference
  delegates to vLLM — continuous batching, paged KV cache with prefix reuse,
  hand-written CDNA4 (gfx950) HIP kernels for the hot ops, TP over RCCL/xGMI.
"""

__version__ = "0.1.0"
