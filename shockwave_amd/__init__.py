"""shockwave_amd — an MI355X-native elastic-training cluster scheduler.

A from-scratch rebuild of the capabilities of uw-mad-dash/shockwave (NSDI'23)
designed for AMD Instinct MI355X clusters:

* head-node round-based lease scheduler with Shockwave's predictive-market
  (dynamic Eisenberg-Gale) planner, solved with scipy/HiGHS MILP
  (the reference used cvxpy+Gurobi; no Gurobi exists here and none is needed),
* a gRPC control plane (msgpack-serialized messages — no protoc step),
* per-GPU worker runtime launching PyTorch-ROCm training jobs under a
  lease-preemptible iterator (``LeaseIterator``, API-compatible with the
  reference's GavelIterator),
* dynamic batch-size adaptation (Accordion critical-regime detection and
  gradient-noise-scale) backed by hand-written CDNA4 HIP kernels
  (fused SGD/Adam, multi-tensor grad-accumulate + L2 norm, GNS estimator),
* RCCL-over-xGMI data parallelism and checkpoint streaming.

Reference layer map: /root/reference SURVEY.md §1.
"""

__version__ = "0.1.0"
