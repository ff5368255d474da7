"""Offline model evaluation over per-epoch checkpoints.

Reference: dolphin/core/master/ModelChkpManager.java:46 (snapshots the model
table at epoch boundaries via table.checkpoint()) + core/worker/
ModelEvaluator.java:39 / ModelEvaluationTasklet (after training, iterate the
checkpoints, load each into the table, call trainer.evaluateModel).
"""

from __future__ import annotations

from typing import Dict, List

from harmony_amd.et.checkpoint import CheckpointManager


class ModelChkpManager:
    """Snapshots the job's model tables at epoch boundaries (collective)."""

    def __init__(self, cm: CheckpointManager, app_id: str, tables: dict,
                 ratio: float = 1.0):
        self.cm = cm
        self.app_id = app_id
        self.tables = {t.cfg.table_id: t for t in tables.values()
                       if hasattr(t, "cfg")}
        self.ratio = ratio
        self.chkp_ids: List[str] = []

    def on_epoch_finished(self, epoch: int) -> None:
        from harmony_amd.et.checkpoint import register_pending_commit

        cid = f"epoch{epoch}"
        for t in self.tables.values():
            sub = f"{cid}/{_safe(t.cfg.table_id)}"
            self.cm.checkpoint(t, self.app_id, sub, ratio=self.ratio)
            # committed on executor close (reference ChkpManagerSlave:226)
            register_pending_commit(self.cm, self.app_id, sub)
        if cid not in self.chkp_ids:
            self.chkp_ids.append(cid)


class ModelEvaluator:
    """Replays checkpoints: for each epoch snapshot, restore the model tables
    and run trainer.evaluate_model() over the training data."""

    def __init__(self, cm: CheckpointManager, app_id: str, tables: dict,
                 trainer, provider):
        self.cm = cm
        self.app_id = app_id
        self.tables = {t.cfg.table_id: t for t in tables.values()
                       if hasattr(t, "cfg")}
        self.trainer = trainer
        self.provider = provider

    def evaluate_all(self, chkp_ids: List[str]) -> Dict[str, dict]:
        out: Dict[str, dict] = {}
        for cid in chkp_ids:
            for t in self.tables.values():
                self.cm.load_into(t, self.app_id,
                                  f"{cid}/{_safe(t.cfg.table_id)}")
            # one pass over the data: pull + eval per batch
            for batch in self.provider.epoch_iter(0):
                self.trainer.set_batch_data(batch)
                self.trainer.pull_model()
                self.trainer.local_compute()
            out[cid] = self.trainer.evaluate_model() or {}
        return out


def _safe(s: str) -> str:
    return s.replace("/", "_")
