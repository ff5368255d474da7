"""Elastic Table (ET): a sharded, GPU-resident key-value store.

MI355X-native rebuild of the reference's elastic distributed table
(services/et): model/input tables are partitioned into blocks over a dense
integer keyspace; each executor (one process per GPU) owns a set of blocks
held in one contiguous HBM tensor; pull/push are RCCL collectives over xGMI;
blocks migrate live between GPUs with an ownership-first protocol.
"""

from harmony_amd.et.partitioner import OrderingBasedPartitioner, HashBasedPartitioner
from harmony_amd.et.table import Table
from harmony_amd.et.ownership import Ownership

__all__ = [
    "OrderingBasedPartitioner",
    "HashBasedPartitioner",
    "Table",
    "Ownership",
]
