"""FusionInfer-AMD: an MI355X-native LLM inference framework.

A from-scratch build with the capabilities of fusioninfer/fusioninfer
(reference: a Kubernetes control plane delegating execution to vLLM —
see SURVEY.md). This framework is both halves, MI355X-first:

* a first-party serving engine (paged KV cache, continuous batching,
  hand-written CDNA4 HIP kernels, RCCL over xGMI for TP and PD KV
  handoff) — the part the reference delegated to vLLM containers, and
* a control plane with the reference's InferenceService semantics
  (``fusioninfer_amd.controlplane``) rendering LeaderWorkerSet /
  Volcano PodGroup / EPP / InferencePool / HTTPRoute resources for
  ``amd.com/gpu`` nodes.
"""

__version__ = "0.2.0"
