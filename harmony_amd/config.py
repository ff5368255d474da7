"""Typed configuration system.

The reference wires everything through Tang @NamedParameter classes and ships
whole serialized configurations between processes (reference:
dolphin/DolphinParameters.java, jobserver/Parameters.java,
et/configuration/TableConfiguration.java:36-82). Here the same roles are
covered by plain dataclasses that serialize to JSON: a job spec is a JSON
document built by the client, sent to the jobserver over a socket, and handed
to every executor.
"""

from __future__ import annotations

import dataclasses
import json
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

# Default keyspace partitioning: the reference uses 1024 blocks per table
# (et/configuration/parameters/NumTotalBlocks.java:23). On MI355X fewer,
# larger blocks amortize collective launches better; tables may override.
DEFAULT_NUM_BLOCKS = 1024

# Jobserver command socket (reference: jobserver/Parameters.java:29).
DEFAULT_JOBSERVER_PORT = 7008


@dataclass
class TableConfig:
    """Configuration of one elastic table.

    Mirrors reference et/configuration/TableConfiguration.java:36-82 with the
    codec machinery dropped: values are device tensors, so the wire format is
    raw device buffers, not Avro-coded byte arrays.
    """

    table_id: str
    num_keys: int                      # dense integer keyspace [0, num_keys)
    value_dim: int = 1                 # row width of the value tensor
    dtype: str = "float32"             # torch dtype name
    num_blocks: int = DEFAULT_NUM_BLOCKS
    is_mutable: bool = True
    is_ordered: bool = True            # ordering-based (range) partitioner
    update_fn: str = "add"             # name in et.update_functions registry
    init_fn: str = "zeros"             # name in et.update_functions registry
    storage: str = "dense"             # "dense" (device tensor) | "object" (host)
    init_args: Dict[str, Any] = field(default_factory=dict)
    update_args: Dict[str, Any] = field(default_factory=dict)
    # Optional bulk-load input path (reference: TableConfiguration file input)
    input_path: Optional[str] = None
    # Restore-from-checkpoint id (reference: ETMaster.createTable(chkpId, ...))
    chkp_id: Optional[str] = None

    def __post_init__(self) -> None:
        if self.num_blocks > self.num_keys > 0:
            self.num_blocks = max(1, self.num_keys)

    @property
    def block_size(self) -> int:
        """Keys per block (last block may be padded)."""
        return (self.num_keys + self.num_blocks - 1) // self.num_blocks

    @property
    def padded_num_keys(self) -> int:
        return self.block_size * self.num_blocks


@dataclass
class ExecutorConfig:
    """Per-executor resources (reference: ExecutorConfiguration.java:30-64).

    An executor is one GPU-bound process (one rank). num_tasklets bounds the
    number of concurrently running job tasklets on this executor.
    """

    num_tasklets: int = 4
    device: str = "cuda"               # "cuda" (ROCm) or "cpu" for tests


@dataclass
class JobConfig:
    """One PS job submission (reference: DolphinJobLauncher serialized conf)."""

    job_id: str
    app: str                           # "nmf" | "mlr" | "lda" | "gbt" | "lasso" | ...
    max_num_epochs: int = 1
    num_mini_batches: int = 1          # mini-batches per epoch
    num_worker_blocks: int = 0         # input-table blocks (0 -> num_mini_batches)
    clock_slack: int = -1              # SSP slack (reference: ClockSlack);
                                       # -1 disables the store-side throttle
                                       # (collectives already enforce lockstep)
    num_trainer_threads: int = 1
    app_args: Dict[str, Any] = field(default_factory=dict)
    model_is_local: bool = False       # also create a local-model table
    optimizer: Optional[str] = None    # elasticity optimizer ("homogeneous",
                                       # "hetero_ilp" or "module:Class")
    optimizer_period: int = 8          # batches between optimization windows
    dashboard_url: Optional[str] = None  # POST epoch metrics here
    trace_path: Optional[str] = None   # JSONL span output (rocTX on GPU)
    model_chkp_per_epoch: bool = False # snapshot model tables every epoch
    offline_model_eval: bool = False   # replay epoch snapshots after training
    chkp_path: str = "/tmp/harmony_chkp_temp"  # reference ChkpTempPath
    chkp_commit_path: str = "/tmp/harmony_chkp_commit"  # ChkpCommitPath:
                                       # temp checkpoints move here when the
                                       # executor closes (two-phase commit)
    restore_chkp: Optional[str] = None # start with model tables restored from
                                       # this checkpoint id (reference
                                       # ETMaster.createTable(chkpId, ...))

    def to_json(self) -> str:
        return json.dumps(dataclasses.asdict(self))

    @staticmethod
    def from_json(s: str) -> "JobConfig":
        return JobConfig(**json.loads(s))


@dataclass
class RuntimeConfig:
    """Node/runtime level settings."""

    world_size: int = 1
    master_addr: str = "127.0.0.1"
    master_port: int = 29500
    control_port: int = 0              # 0 -> master_port + 7
    backend: str = "auto"              # "nccl" (=RCCL) | "gloo" | "auto"
    device: str = "auto"               # "cuda" | "cpu" | "auto"

    def resolved_control_port(self) -> int:
        return self.control_port or self.master_port + 7


def dtype_of(name: str):
    import torch

    return {
        "float32": torch.float32,
        "float64": torch.float64,
        "bfloat16": torch.bfloat16,
        "float16": torch.float16,
        "int32": torch.int32,
        "int64": torch.int64,
    }[name]
