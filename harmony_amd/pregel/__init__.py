"""Pregel: BSP graph processing on the elastic table.

Reference: jobserver/src/.../pregel — vertex table + TWO message tables
double-buffered per superstep (graph/impl/MessageManager.java:42-110),
PregelWorkerTask supersteps with COMP/SEND/SYNC task units, PregelMaster
ANDing per-worker {allVerticesHalt, noOngoingMsgs} votes (PregelMaster.java:48).
"""

from harmony_amd.pregel.engine import Computation, PregelEngine

__all__ = ["Computation", "PregelEngine"]
