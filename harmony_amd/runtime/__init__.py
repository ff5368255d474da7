"""Executor runtime: process bootstrap, control plane, tasklets.

The reference runs on Apache REEF: a driver JVM allocates evaluator JVMs and
wires an Avro/TCP NetworkConnectionService between them. The MI355X-native
runtime is one process per GPU launched by torch.distributed.run (or a single
process for local mode): the RCCL process group is the data plane, and a
TCPStore on rank 0 is the control plane (barriers, SSP clock, task-unit
ordering, job submission fan-out).
"""

from harmony_amd.runtime.bootstrap import ExecutorContext, init_executor

__all__ = ["ExecutorContext", "init_executor"]
