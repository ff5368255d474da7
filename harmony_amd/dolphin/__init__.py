"""Dolphin: the parameter-server training runtime.

Reference: jobserver/src/.../dolphin/core — worker tasklet loop, Trainer SPI,
model accessor, SSP mini-batch controller, master-side barriers.
"""

from harmony_amd.dolphin.trainer import Trainer, TrainerContext
from harmony_amd.dolphin.worker import WorkerTasklet

__all__ = ["Trainer", "TrainerContext", "WorkerTasklet"]
