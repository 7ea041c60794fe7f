"""arks_amd — an MI355X-native LLM serving stack.

A from-scratch rebuild of the capabilities of scitix/arks (a Kubernetes-native
LLM serving control plane; see /root/reference) with the pluggable
vLLM/SGLang runtime slot replaced by a first-party inference engine:

- hand-written CDNA4 (gfx950) HIP kernels for the hot ops (paged attention
  prefill+decode, RMSNorm, RoPE, SiLU, sampling) — ``arks_amd.ops``
- a paged-KV continuous-batching engine — ``arks_amd.engine``
- tensor parallelism over RCCL/xGMI — ``arks_amd.parallel``
- an OpenAI-compatible HTTP server — ``arks_amd.server``
- the Arks control plane re-implemented natively (CRD types, reconcilers,
  ext-proc-equivalent gateway) — ``arks_amd.crd``, ``arks_amd.controlplane``,
  ``arks_amd.gateway``
"""

__version__ = "0.1.0"
