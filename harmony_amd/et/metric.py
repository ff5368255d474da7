"""ET metric service: executor-side collector flush -> driver-side manager
with pluggable receivers.

Reference: et/metric — executor MetricCollector (custom metrics + built-in
table/block + network stats, flushed as MetricReportMsg,
metric/MetricCollector.java:14) and driver MetricManager (start/stop
collection per executor via MetricControlMsg, pluggable MetricReceiver,
metric/MetricManager.java:6, driver/api/MetricReceiver.java, default
LoggingMetricReceiver).

MI355X shape: reports travel through the control store as JSON records
keyed per (job, rank, seq); the manager (rank 0) polls and fans out to
receivers. Built-in fields mirror the reference's MetricReportMsg:
table -> numBlocks, sent get requests, received bytes.
"""

from __future__ import annotations

import json
import logging
import time
from typing import Callable, Dict, List, Optional

logger = logging.getLogger("harmony.metric")


class MetricReceiver:
    """SPI (reference driver/api/MetricReceiver.java)."""

    def on_metric_msg(self, src_rank: int, report: dict) -> None:
        raise NotImplementedError


class LoggingMetricReceiver(MetricReceiver):
    def on_metric_msg(self, src_rank: int, report: dict) -> None:
        logger.info("metrics from rank %d: %s", src_rank, json.dumps(report))


class ExecutorMetricCollector:
    """Per-rank metric source. Apps/tables add custom metrics; flush() ships
    a report through the store when collection is enabled."""

    def __init__(self, cp, rank: int, enabled_key: str = "met/enabled"):
        self.cp = cp
        self.rank = rank
        self.enabled_key = enabled_key
        self.custom: Dict[str, float] = {}
        self.tables: Dict[str, dict] = {}   # table_id -> builtin stats
        self._seq = 0

    def enabled(self) -> bool:
        return self.cp.flag_set(self.enabled_key)

    def set_table_stats(self, table_id: str, num_blocks: int,
                        sent_get_reqs: int = 0, recv_bytes: int = 0) -> None:
        self.tables[table_id] = {"numBlocks": num_blocks,
                                 "countSentGetReq": sent_get_reqs,
                                 "bytesReceivedGetResp": recv_bytes}

    def add_custom(self, key: str, value: float) -> None:
        self.custom[key] = value

    def flush(self) -> None:
        if not self.enabled():
            return
        self._seq += 1
        report = {"rank": self.rank, "time": time.time(),
                  "tableToStats": self.tables, "customMetrics": self.custom}
        self.cp.store.set(f"met/report/{self.rank}/{self._seq}",
                          json.dumps(report))
        self.cp.store.set(f"met/latest/{self.rank}", str(self._seq))


class MetricManager:
    """Driver-side (rank 0): start/stop collection, poll reports, fan out."""

    def __init__(self, cp, world_size: int,
                 receivers: Optional[List[MetricReceiver]] = None,
                 enabled_key: str = "met/enabled"):
        self.cp = cp
        self.world_size = world_size
        self.receivers = receivers or [LoggingMetricReceiver()]
        self.enabled_key = enabled_key
        self._consumed = [0] * world_size

    def start_collection(self) -> None:
        self.cp.set_flag(self.enabled_key)

    def stop_collection(self) -> None:
        # flags are one-way in the store; a distinct key marks stop
        self.cp.store.set(self.enabled_key + "/stopped", "1")

    def poll(self) -> int:
        """Drain new reports; returns how many were delivered."""
        n = 0
        for r in range(self.world_size):
            try:
                if not self.cp.flag_set(f"met/latest/{r}"):
                    continue
                latest = int(self.cp.store.get(f"met/latest/{r}"))
            except Exception:  # noqa: BLE001
                continue
            while self._consumed[r] < latest:
                self._consumed[r] += 1
                raw = self.cp.store.get(f"met/report/{r}/{self._consumed[r]}")
                report = json.loads(raw)
                for recv in self.receivers:
                    recv.on_metric_msg(r, report)
                n += 1
        return n
