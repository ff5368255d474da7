"""harmony_amd — an MI355X-native multi-tenant parameter-server training framework.

A from-scratch rebuild of the capabilities of snuspl/harmony (Java/REEF/YARN
parameter-server system) designed for a single node of 8 AMD Instinct MI355X
GPUs: elastic distributed tables live in HBM3E, pull/push are RCCL collectives
over xGMI, app hot loops are hand-written CDNA4 HIP kernels, and a long-running
jobserver co-schedules concurrent jobs on the same GPUs via HIP streams.

Layer map (mirrors reference SURVEY.md §1):
  et/        — elastic table: block partitioning, device shards, ownership,
               collective pull/push data plane, migration, checkpoint
               (reference: services/et)
  runtime/   — executor bootstrap, control plane (TCPStore), tasklets,
               task-unit scheduling (reference: evaluator side of et + REEF)
  dolphin/   — PS training runtime: Trainer SPI, worker loop, SSP clock,
               model accessor (reference: jobserver/src/.../dolphin/core)
  mlapps/    — NMF, MLR, LDA, GBT, Lasso + example verification apps
  pregel/    — BSP graph engine (PageRank, shortest path)
  jobserver/ — long-running job server + pluggable global scheduler
  optimizer/ — cost-model elasticity optimizer + plan compiler/executor
  ops/       — CDNA4 HIP kernels + fp32 torch reference implementations
"""

__version__ = "0.1.0"
