"""Elasticity: cost-model optimizer + plan compiler/executor + orchestrator.

Reference: dolphin/optimizer (ETOptimizationOrchestrator, HomogeneousOptimizer,
SampleOptimizers, MetricProcessor) + dolphin/plan (PlanCompiler) +
services/et plan engine (ETPlan op-DAG, PlanExecutorImpl).
"""

from harmony_amd.optimizer.plan import MoveOp, Plan, PlanExecutor, SetBatchShareOp
from harmony_amd.optimizer.optimizers import (HomogeneousCostOptimizer,
                                              Optimizer, SampleOptimizers)
from harmony_amd.optimizer.orchestrator import OptimizationOrchestrator

__all__ = ["Plan", "MoveOp", "SetBatchShareOp", "PlanExecutor", "Optimizer",
           "HomogeneousCostOptimizer", "SampleOptimizers",
           "OptimizationOrchestrator"]
