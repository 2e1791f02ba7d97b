"""Metrics buffering + rollups.

Reference analogs: services/metrics_buffer_service.py:84-131 (60 s flush /
1000-entry force-flush), services/metrics_rollup_service.py (hourly rollups),
services/metrics_query_service.py. On multi-GPU runs the per-rank buffers
are aggregated with a RCCL all-reduce instead of per-row DB writes from
every worker — see parallel/collectives.py.
"""

from __future__ import annotations

import datetime
import threading
import time
from collections import defaultdict
from typing import Any, Dict, List, Optional

from ..db.engine import Database
from ..db.models import DbMetricRollup, DbToolMetric


class MetricsBuffer:
    def __init__(self, db: Optional[Database] = None, flush_interval: float = 60.0, max_size: int = 1000):
        self.db = db
        self.flush_interval = flush_interval
        self.max_size = max_size
        self._rows: List[Dict[str, Any]] = []
        self._agg: Dict[str, list] = {}   # tool_id -> [count, errors, latency_sum_ms]
        self._lock = threading.Lock()
        self._last_flush = time.monotonic()
        # live counters for /metrics + admin dashboards
        self.counters: Dict[str, float] = defaultdict(float)
        self.latency_sum_ms: Dict[str, float] = defaultdict(float)
        self.latency_count: Dict[str, int] = defaultdict(int)

    def record_tool_metric(self, tool_id: str, response_time_ms: float, success: bool, error: Optional[str] = None) -> None:
        with self._lock:
            self._rows.append(
                {"tool_id": tool_id, "response_time_ms": response_time_ms, "is_success": success, "error_message": error}
            )
            self.counters["tool_invocations_total"] += 1
            if not success:
                self.counters["tool_errors_total"] += 1
            self.latency_sum_ms[tool_id] += response_time_ms
            self.latency_count[tool_id] += 1
            need_flush = len(self._rows) >= self.max_size
        if need_flush:
            self.flush()

    def record_batch(self, tool_ids: List[str], response_time_ms: float, successes: List[bool]) -> None:
        """Batched record: one lock, N rows."""
        with self._lock:
            for tid, ok in zip(tool_ids, successes):
                self._rows.append({"tool_id": tid, "response_time_ms": response_time_ms, "is_success": ok, "error_message": None})
                self.latency_sum_ms[tid] += response_time_ms
                self.latency_count[tid] += 1
            self.counters["tool_invocations_total"] += len(tool_ids)
            self.counters["tool_errors_total"] += sum(1 for s in successes if not s)
            need_flush = len(self._rows) >= self.max_size
        if need_flush:
            self.flush()

    def record_aggregate(self, tool_id: str, count: int, errors: int, response_time_ms: float) -> None:
        """Aggregated record from the GPU pipeline: one row per (tool, batch)
        carrying `count` (schema revision 0002) instead of N rows — the
        batched analog of the reference's per-row metric buffering."""
        if count <= 0:
            return
        with self._lock:
            self._rows.append({"tool_id": tool_id, "response_time_ms": response_time_ms,
                               "is_success": errors == 0, "error_message": None, "count": count})
            self.counters["tool_invocations_total"] += count
            self.counters["tool_errors_total"] += errors
            self.latency_sum_ms[tool_id] += response_time_ms * count
            self.latency_count[tool_id] += count
            need_flush = len(self._rows) >= self.max_size
        if need_flush:
            self.flush()

    def record_aggregate_many(self, tool_ids: List[str], counts, errors, response_time_ms: float) -> None:
        """One-lock batched aggregate (GPU pipeline: one row per tool per
        batch). Accumulates in-memory per tool; rows materialize at flush
        time, so the hot path never touches the DB."""
        with self._lock:
            total = 0
            total_err = 0
            agg = self._agg
            for tid, cnt, err in zip(tool_ids, counts, errors):
                cnt = int(cnt)
                err = int(err)
                if cnt <= 0:
                    continue
                lat = response_time_ms * cnt
                a = agg.get(tid)
                if a is None:
                    agg[tid] = [cnt, err, lat]
                else:
                    a[0] += cnt
                    a[1] += err
                    a[2] += lat
                self.latency_sum_ms[tid] += lat
                self.latency_count[tid] += cnt
                total += cnt
                total_err += err
            self.counters["tool_invocations_total"] += total
            self.counters["tool_errors_total"] += total_err

    def maybe_flush(self) -> None:
        if time.monotonic() - self._last_flush >= self.flush_interval:
            self.flush()

    def flush(self) -> int:
        with self._lock:
            rows, self._rows = self._rows, []
            for tid, (cnt, err, lat) in self._agg.items():
                avg = lat / max(cnt, 1)
                if err:
                    rows.append({"tool_id": tid, "response_time_ms": avg,
                                 "is_success": False, "error_message": None, "count": err})
                if cnt - err:
                    rows.append({"tool_id": tid, "response_time_ms": avg,
                                 "is_success": True, "error_message": None, "count": cnt - err})
            self._agg.clear()
            self._last_flush = time.monotonic()
        if not rows or self.db is None:
            return len(rows)
        with self.db.session() as s:
            s.bulk_insert_mappings(DbToolMetric, rows)
        return len(rows)

    def snapshot(self) -> Dict[str, Any]:
        with self._lock:
            top = sorted(self.latency_count.items(), key=lambda kv: -kv[1])[:25]
            return {
                "counters": dict(self.counters),
                "pending_rows": len(self._rows) + len(self._agg),
                "top_tools": [
                    {
                        "tool_id": t,
                        "count": c,
                        "avg_ms": self.latency_sum_ms[t] / max(c, 1),
                    }
                    for t, c in top
                ],
            }


def rollup_hourly(db: Database, entity_type: str = "tool") -> int:
    """Raw rows → hourly rollups, then prune raw (reference: metrics_rollup_service)."""
    from sqlalchemy import delete, func, select

    created = 0
    with db.session() as s:
        rows = s.execute(
            select(
                DbToolMetric.tool_id,
                func.strftime("%Y-%m-%d %H:00:00", DbToolMetric.timestamp).label("hour")
                if db.url.startswith("sqlite")
                else func.date_trunc("hour", DbToolMetric.timestamp).label("hour"),
                func.count(),
                func.sum(1 - func.cast(DbToolMetric.is_success, __import__("sqlalchemy").Integer)),
                func.sum(DbToolMetric.response_time_ms),
                func.min(DbToolMetric.response_time_ms),
                func.max(DbToolMetric.response_time_ms),
            ).group_by(DbToolMetric.tool_id, "hour")
        ).all()
        for tool_id, hour, count, errs, total, mn, mx in rows:
            if isinstance(hour, str):
                hour = datetime.datetime.fromisoformat(hour)
            s.add(
                DbMetricRollup(
                    entity_type=entity_type,
                    entity_id=tool_id,
                    hour=hour,
                    count=count,
                    error_count=int(errs or 0),
                    total_ms=float(total or 0.0),
                    min_ms=float(mn or 0.0),
                    max_ms=float(mx or 0.0),
                )
            )
            created += 1
        s.execute(delete(DbToolMetric))
    return created


class TokenUsageTracker:
    """Per-credential request accounting, hourly-bucketed (reference:
    TokenUsageLog db.py:5584 + TokenUsageMiddleware). record() is a dict
    increment under a lock — safe on the hot paths; flush() upserts the
    current hour's rows (same buffered shape as MetricsBuffer)."""

    def __init__(self, db: Optional[Database] = None):
        self.db = db
        self._counts: Dict[tuple, int] = {}
        self._lock = threading.Lock()

    def record(self, credential: str, user: Optional[str] = None, n: int = 1) -> None:
        key = (credential or "anonymous", user)
        with self._lock:
            self._counts[key] = self._counts.get(key, 0) + n

    def flush(self) -> int:
        with self._lock:
            counts, self._counts = self._counts, {}
        if not counts or self.db is None:
            return 0
        from sqlalchemy import select

        from ..db.models import DbTokenUsage, utcnow

        bucket = utcnow().replace(minute=0, second=0, microsecond=0)
        with self.db.session() as s:
            for (credential, user), n in counts.items():
                row = s.execute(select(DbTokenUsage).where(
                    DbTokenUsage.bucket == bucket,
                    DbTokenUsage.credential == credential)).scalar_one_or_none()
                if row is None:
                    s.add(DbTokenUsage(bucket=bucket, credential=credential,
                                       user_email=user, requests=n))
                else:
                    row.requests += n
        return len(counts)

    def query(self, limit: int = 200) -> List[Dict[str, Any]]:
        self.flush()
        if self.db is None:
            return []
        from sqlalchemy import select

        from ..db.models import DbTokenUsage

        with self.db.session() as s:
            rows = s.execute(select(DbTokenUsage)
                             .order_by(DbTokenUsage.bucket.desc(),
                                       DbTokenUsage.requests.desc())
                             .limit(limit)).scalars().all()
            return [{"bucket": str(r.bucket), "credential": r.credential,
                     "user": r.user_email, "requests": r.requests} for r in rows]
