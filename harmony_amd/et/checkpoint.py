"""Table checkpoint / restore — reference-compatible layout.

Reference: et/evaluator/impl/ChkpManagerSlave.java:64 (per-block files under
`<ChkpTempPath>/<appId>/<chkpId>/<blockIdx>` + a serialized table conf,
two-phase temp->commit on close, sampling-ratio snapshots) and
driver/impl/ChkpManagerMaster.java:62 (start/done aggregation, restore
planning). `ETMaster.createTable(chkpId, executors)` restores with
re-partitioning if the executor set changed.

MI355X shape: a block is one dense device tensor — checkpoint is a
device->host DMA + one file per block (same `<root>/<appId>/<chkpId>/<blockIdx>`
layout), restore is file -> HBM for whichever rank NOW owns the block (block
files are rank-independent, so restore re-partitions for free). Writes happen
per rank in parallel (each rank writes only blocks it owns).
"""

from __future__ import annotations

import json
import os
import shutil
from dataclasses import asdict
from pathlib import Path
from typing import Optional

import torch

from harmony_amd.config import TableConfig


# checkpoints written during a server session, committed on executor close
# (reference ChkpManagerSlave.commitAllLocalChkps, ChkpManagerSlave.java:226):
# (temp_root, commit_root, app_id, chkp_id) tuples, process-local
_pending_commits: "set[tuple]" = set()


def register_pending_commit(mgr: "CheckpointManager", app_id: str,
                            chkp_id: str) -> None:
    _pending_commits.add((str(mgr.temp_root), str(mgr.commit_root),
                          app_id, chkp_id))


def commit_all_pending() -> int:
    """Two-phase temp->commit of every registered checkpoint; returns how
    many were committed. Call once per node on executor close (one mover:
    commit is a directory move on the shared FS)."""
    n = 0
    for temp, commit, app_id, chkp_id in sorted(_pending_commits):
        mgr = CheckpointManager(temp_root=temp, commit_root=commit)
        if (mgr.temp_root / app_id / chkp_id).exists():
            mgr.commit(app_id, chkp_id)
            n += 1
    _pending_commits.clear()
    return n


class CheckpointManager:
    def __init__(self, temp_root: str = "/tmp/harmony_chkp_temp",
                 commit_root: str = "/tmp/harmony_chkp_commit"):
        self.temp_root = Path(temp_root)
        self.commit_root = Path(commit_root)

    def _dir(self, root: Path, app_id: str, chkp_id: str) -> Path:
        return root / app_id / chkp_id

    # ------------------------------------------------------------ checkpoint

    def checkpoint(self, table, app_id: str, chkp_id: str,
                   ratio: float = 1.0, seed: int = 0) -> None:
        """Write this rank's blocks (call on every rank; mutual exclusion with
        migration is the caller's job, as in AllocatedTable.java:127-161)."""
        d = self._dir(self.temp_root, app_id, chkp_id)
        d.mkdir(parents=True, exist_ok=True)
        if table.rank == 0:
            with open(d / "table_conf.json", "w") as f:
                json.dump(asdict(table.cfg), f)
        for b in table.owned_blocks:
            data = table.local_block_view(b)
            if ratio < 1.0:
                # sampled snapshot (reference samplingRatio,
                # elastictable.avsc:306-317): keep a deterministic row subset
                n = data.shape[0]
                k = max(1, int(n * ratio))
                g = torch.Generator().manual_seed(seed * 1000003 + b)
                rows = torch.randperm(n, generator=g)[:k].sort().values
                payload = {"rows": rows, "values": data[rows.to(data.device)].cpu(),
                           "n": n}
            else:
                payload = {"rows": None, "values": data.cpu(), "n": data.shape[0]}
            torch.save(payload, d / str(b))

    # ---------------------------------------------------------- temp->commit

    def commit(self, app_id: str, chkp_id: str) -> Path:
        """Two-phase commit: move the temp checkpoint to the commit root
        (reference commitAllLocalChkps on executor close)."""
        src = self._dir(self.temp_root, app_id, chkp_id)
        dst = self._dir(self.commit_root, app_id, chkp_id)
        dst.parent.mkdir(parents=True, exist_ok=True)
        if dst.exists():
            shutil.rmtree(dst)
        shutil.move(str(src), str(dst))
        return dst

    def exists(self, app_id: str, chkp_id: str) -> Optional[Path]:
        for root in (self.temp_root, self.commit_root):
            d = self._dir(root, app_id, chkp_id)
            if d.exists():
                return d
        return None

    # --------------------------------------------------------------- restore

    def load_into(self, table, app_id: str, chkp_id: str) -> int:
        """Load blocks this rank owns NOW (restore re-partitions freely —
        reference ChkpLoadMsg.blockOwners planning). Call on every rank.
        Returns the number of blocks loaded (0 = nothing matched — callers
        that require a real restore must check)."""
        d = self.exists(app_id, chkp_id)
        if d is None:
            raise FileNotFoundError(f"no checkpoint {app_id}/{chkp_id}")
        n = 0
        for b in table.owned_blocks:
            f = d / str(b)
            if not f.exists():
                continue  # block missing from a sampled/partial checkpoint
            payload = torch.load(f, weights_only=True)
            view = table.local_block_view(b)
            if payload["rows"] is None:
                view.copy_(payload["values"].to(view.device, view.dtype))
            else:
                view[payload["rows"].to(view.device)] = \
                    payload["values"].to(view.device, view.dtype)
            n += 1
        return n

    def saved_table_config(self, app_id: str, chkp_id: str) -> TableConfig:
        d = self.exists(app_id, chkp_id)
        if d is None:
            raise FileNotFoundError(f"no checkpoint {app_id}/{chkp_id}")
        with open(d / "table_conf.json") as f:
            return TableConfig(**json.load(f))
