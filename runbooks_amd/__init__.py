"""runbooks_amd — an MI355X-native ML platform.

A from-scratch rebuild of the capabilities of substratusai/runbooks
(reference: /root/reference) for AMD Instinct MI355X (gfx950, CDNA4):

* Control plane: Model / Dataset / Server / Notebook custom resources with
  the same spec/status shape as the reference's CRDs
  (reference api/v1/*_types.go), reconciled onto ``amd.com/gpu`` nodes.
* In-pod runtime: PyTorch-ROCm train/serve with hand-written CDNA4 HIP
  kernels (RMSNorm, RoPE, fused AdamW, paged/flash attention, sampling)
  and RCCL-over-xGMI collectives (DP gradient all-reduce, TP
  all-reduce/all-gather) — the parts the reference delegates to external
  container images via its container contract
  (reference docs/container-contract.md).
"""

__version__ = "0.1.0"
