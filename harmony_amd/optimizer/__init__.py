"""Elasticity: cost-model optimizer + plan compiler/executor + orchestrator.

Reference: dolphin/optimizer (ETOptimizationOrchestrator, HomogeneousOptimizer,
SampleOptimizers, MetricProcessor) + dolphin/plan (PlanCompiler) +
services/et plan engine (ETPlan op-DAG, PlanExecutorImpl).
"""

from harmony_amd.optimizer.plan import (DropTableOp, MoveOp, Plan,
                                        PlanExecutor, SetBatchShareOp,
                                        compile_switch,
                                        StartWorkerOp, StopWorkerOp)
from harmony_amd.optimizer.optimizers import (HomogeneousCostOptimizer,
                                              Optimizer, SampleOptimizers)
from harmony_amd.optimizer.orchestrator import OptimizationOrchestrator

__all__ = ["Plan", "MoveOp", "SetBatchShareOp", "StartWorkerOp",
           "compile_switch",
           "StopWorkerOp", "DropTableOp", "PlanExecutor", "Optimizer",
           "HomogeneousCostOptimizer", "SampleOptimizers",
           "OptimizationOrchestrator"]
