"""machin_amd — an MI355X-native reinforcement-learning framework.

A from-scratch rebuild of the capability surface of iffiX/machin
(reference layer map: SURVEY.md §1) designed for one node of 8×MI355X
GPUs: PyTorch-ROCm as the tensor substrate, hand-written CDNA4 (gfx950)
HIP kernels for the hot data-path ops (PER sum-tree, GAE / V-trace
scans, categorical projection, fused polyak updates, on-device noise),
and RCCL collectives over xGMI for all bulk multi-GPU traffic.

Layers (bottom → top):
  machin_amd.ops       — HIP kernels + CPU fallbacks
  machin_amd.parallel  — processes, pools, world (RCCL/gloo), servers
  machin_amd.frame     — transitions, replay buffers, 18 algorithm classes
  machin_amd.model     — network bases and shipped nets
  machin_amd.env       — environment wrappers + built-in classic control
  machin_amd.auto      — config generation / launch CLI
  machin_amd.utils     — checkpointing, config, logging, helpers
"""

__version__ = "0.1.0"
