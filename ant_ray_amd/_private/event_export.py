"""Export-event pipeline: file writer + HTTP aggregator publisher.

Role parity with the reference's event export backbone:
- `RAY_enable_export_api_write` writing `export_*.proto`-shaped JSON
  lines into the session's `export_events/` directory (reference
  src/ray/protobuf/export_*.proto + the export-API file writers), and
- the dashboard aggregator agent publishing buffered task/actor/job
  events to a configurable HTTP endpoint (reference
  python/ray/dashboard/modules/aggregator/aggregator_agent.py:78).

MI355X-first reduction: both sinks run as ONE asyncio task inside the
GCS process (no agent subprocess), draining the GCS's existing
cluster-event and task-event rings by sequence number. Records are
normalized to the export schema: {event_id, source_type, timestamp,
event_data}. The HTTP publisher batches, retries with backoff, and
drops (with a counter) rather than block the control plane.

Env:
  RAY_enable_export_api_write=1    enable the file sink
  RAY_export_events_dir=PATH       file sink directory (default: beside
                                   the GCS persist file, else tmp)
  RAY_export_event_http_target=URL enable the HTTP publisher
"""
from __future__ import annotations

import asyncio
import json
import logging
import os
import tempfile
import time
import uuid
from typing import Dict, List, Optional

logger = logging.getLogger(__name__)

# cluster-event source -> export source_type (export_*.proto names)
_SOURCE_TYPES = {
    "NODE": "EXPORT_NODE",
    "ACTOR": "EXPORT_ACTOR",
    "JOB": "EXPORT_DRIVER_JOB",
    "WORKER": "EXPORT_WORKER",
}


def _normalize_cluster_event(ev: dict) -> dict:
    return {
        "event_id": uuid.uuid4().hex,
        "source_type": _SOURCE_TYPES.get(ev.get("source"), "EXPORT_EVENT"),
        "timestamp": ev.get("timestamp"),
        "event_data": {k: v for k, v in ev.items() if k != "seq"},
    }


def _normalize_task_event(ev: dict) -> dict:
    return {
        "event_id": uuid.uuid4().hex,
        "source_type": "EXPORT_TASK",
        "timestamp": ev.get("end_ts") or ev.get("start_ts"),
        "event_data": {k: v for k, v in ev.items() if k != "seq"},
    }


class ExportEventAggregator:
    """Drains the GCS event rings into the enabled sinks."""

    def __init__(self, gcs, out_dir: Optional[str] = None,
                 http_target: Optional[str] = None, period: float = 1.0,
                 max_buffer: int = 10000):
        self.gcs = gcs
        self.out_dir = out_dir
        self.http_target = http_target
        self.period = period
        # per-ring high-water marks: the rings share one GCS seq counter,
        # so a single mark would skip a task event whose seq is lower
        # than a cluster event drained in the same pass
        self._last_cluster_seq = 0
        self._last_task_seq = 0
        self._files: Dict[str, object] = {}
        self._http_buf: List[dict] = []
        self._max_buffer = max_buffer
        self.published = 0
        self.written = 0
        self.dropped = 0
        if self.out_dir:
            os.makedirs(self.out_dir, exist_ok=True)

    # ------------------------------------------------------------- drain
    def _collect_new(self) -> List[dict]:
        out = []
        for ev in getattr(self.gcs, "cluster_events", []):
            if ev.get("seq", 0) > self._last_cluster_seq:
                out.append(_normalize_cluster_event(ev))
                self._last_cluster_seq = ev["seq"]
        for ev in getattr(self.gcs, "task_events", []):
            if ev.get("seq", 0) > self._last_task_seq:
                out.append(_normalize_task_event(ev))
                self._last_task_seq = ev["seq"]
        out.sort(key=lambda r: r["timestamp"] or 0)
        return out

    # -------------------------------------------------------------- file
    def _write_files(self, records: List[dict]) -> None:
        for rec in records:
            st = rec["source_type"]
            f = self._files.get(st)
            if f is None:
                path = os.path.join(self.out_dir,
                                    f"event_{st}.log")
                f = self._files[st] = open(path, "a", buffering=1)
            f.write(json.dumps(rec) + "\n")
            self.written += 1

    # -------------------------------------------------------------- http
    async def _publish_http(self, records: List[dict]) -> None:
        self._http_buf.extend(records)
        if len(self._http_buf) > self._max_buffer:
            self.dropped += len(self._http_buf) - self._max_buffer
            self._http_buf = self._http_buf[-self._max_buffer:]
        if not self._http_buf:
            return
        batch, self._http_buf = self._http_buf, []
        body = json.dumps(batch).encode()
        loop = asyncio.get_running_loop()
        for attempt in range(3):
            try:
                await loop.run_in_executor(None, self._post, body)
                self.published += len(batch)
                return
            except Exception as e:
                if attempt == 2:
                    # keep for the next round rather than lose them;
                    # bounded by _max_buffer above
                    self._http_buf = batch + self._http_buf
                    logger.debug("export publish failed (%s); will retry",
                                 e)
                else:
                    await asyncio.sleep(0.2 * (attempt + 1))

    def _post(self, body: bytes) -> None:
        import urllib.request

        req = urllib.request.Request(
            self.http_target, data=body,
            headers={"Content-Type": "application/json"})
        with urllib.request.urlopen(req, timeout=5) as resp:
            resp.read()

    # --------------------------------------------------------------- run
    async def run(self) -> None:
        shutdown = self.gcs._shutdown
        while True:
            try:
                await asyncio.wait_for(shutdown.wait(), timeout=self.period)
                stopping = True
            except asyncio.TimeoutError:
                stopping = False
            try:
                records = self._collect_new()
                if records and self.out_dir:
                    self._write_files(records)
                if self.http_target:
                    await self._publish_http(records)
            except Exception:
                logger.exception("export-event drain failed")
            if stopping:
                break
        for f in self._files.values():
            try:
                f.close()
            except Exception:
                pass

    def stats(self) -> dict:
        return {"written": self.written, "published": self.published,
                "dropped": self.dropped, "buffered": len(self._http_buf),
                "last_cluster_seq": self._last_cluster_seq,
                "last_task_seq": self._last_task_seq}


def maybe_start(gcs) -> Optional[ExportEventAggregator]:
    """Called from GCS.start(): wires the aggregator when enabled."""
    file_on = os.environ.get("RAY_enable_export_api_write", "0") == "1"
    target = os.environ.get("RAY_export_event_http_target") or None
    if not file_on and not target:
        return None
    out_dir = None
    if file_on:
        out_dir = os.environ.get("RAY_export_events_dir")
        if not out_dir:
            if gcs.persist_path:
                out_dir = os.path.join(os.path.dirname(gcs.persist_path),
                                       "export_events")
            else:
                out_dir = os.path.join(tempfile.gettempdir(),
                                       f"antray_export_{os.getpid()}")
    agg = ExportEventAggregator(gcs, out_dir=out_dir, http_target=target,
                                period=float(os.environ.get(
                                    "RAY_export_event_period_s", "1.0")))
    asyncio.get_running_loop().create_task(agg.run())
    logger.info("export-event pipeline on (dir=%s, http=%s)", out_dir,
                target)
    return agg
