"""llm_d_inference_scheduler_amd — MI355X-native disaggregated-inference router.

A from-scratch re-grounding of the llm-d Router's capabilities
(llm-d/llm-d-inference-scheduler, a Kubernetes Envoy ext-proc Endpoint Picker
plus P/D sidecar) onto one 8xMI355X node:

  * GPU roles (prefill / decode / encode) replace model-server pods
  * a C++ core (`_router_core`) + gfx950 HIP kernels (`_hip_ops`) replace the
    Go EPP binary's hot paths (chained prefix hashing, block-table match,
    filter/score/pick, flow queues)
  * KV-cache / embedding movement runs over xGMI (RCCL P2P +
    MFMA-tile-aligned block layout) instead of delegated NIXL/RDMA
  * per-GPU PyTorch-ROCm worker engines (paged KV, continuous batching)
    replace external vLLM pods

Layer map mirrors SURVEY.md §1; every module docstring cites the reference
component (file:line into /root/reference) it provides parity with.
"""

__version__ = "0.1.0"
