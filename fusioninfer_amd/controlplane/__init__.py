"""FusionInfer-AMD control plane.

A from-scratch reimplementation of the reference operator's semantics
(fusioninfer/fusioninfer: InferenceService CRD -> LeaderWorkerSet +
Volcano PodGroup + EPP/InferencePool/HTTPRoute) for
``amd.com/gpu`` nodes, with the engine pods running the first-party
FusionInfer-AMD engine instead of vLLM images, and torchrun/TCPStore
rendezvous (consuming the same LWS_LEADER_ADDRESS contract) instead of Ray.

Language note: the reference is Go (kubebuilder). This environment has no
Go toolchain and no network for Go module fetch, so the control plane is
implemented in Python with the same resource-rendering and reconcile
semantics; resources are plain dicts serializable to the exact YAML shapes
the reference produces. The reconciler runs against any client with the
kube verbs (get/create/update/delete/list) — tests use the in-memory
apiserver in ``fake/``.
"""
