"""Engine construction + native migration chain.

Reference analog: mcpgateway/db.py:122 `build_engine` and
mcpgateway/bootstrap_db.py (alembic `upgrade head` at startup). Alembic is
not in this image, so the migration chain is implemented natively: a
``forge_schema_version`` table records applied revisions; `run_migrations`
applies the linear chain in order, exactly like `alembic upgrade head`.
Revision 0001 materializes the full current schema (models.Base.metadata);
later schema changes MUST be appended as new revisions, never by editing
0001 — same discipline as the reference's 117-migration chain.
"""

from __future__ import annotations

import threading
from contextlib import contextmanager
from typing import Callable, Iterator, List, Tuple

from sqlalchemy import Connection, create_engine, text
from sqlalchemy.engine import Engine
from sqlalchemy.orm import Session, sessionmaker
from sqlalchemy.pool import StaticPool

from .models import Base


def build_engine(database_url: str, pool_size: int = 16, max_overflow: int = 8) -> Engine:
    """Build a SQLAlchemy engine with pool-error resilience (reference: db.py:617-716)."""
    kwargs: dict = {"pool_pre_ping": True, "future": True}
    if database_url.startswith("sqlite"):
        kwargs["connect_args"] = {"check_same_thread": False, "timeout": 30}
        if ":memory:" in database_url or database_url == "sqlite://":
            kwargs["poolclass"] = StaticPool
    else:
        kwargs["pool_size"] = pool_size
        kwargs["max_overflow"] = max_overflow
    engine = create_engine(database_url, **kwargs)
    if database_url.startswith("sqlite"):
        from sqlalchemy import event

        @event.listens_for(engine, "connect")
        def _set_sqlite_pragma(dbapi_conn, _rec):  # pragma: no cover - driver hook
            cur = dbapi_conn.cursor()
            cur.execute("PRAGMA journal_mode=WAL")
            cur.execute("PRAGMA synchronous=NORMAL")
            cur.execute("PRAGMA foreign_keys=ON")
            cur.close()

    return engine


# ---------------------------------------------------------------------------
# Migration chain (alembic-equivalent)
# ---------------------------------------------------------------------------

def _rev_0001_initial(conn: Connection) -> None:
    Base.metadata.create_all(conn)


def _rev_0002_tool_metrics_count(conn: Connection) -> None:
    """Batched metric rows: add tool_metrics.count (GPU pipeline aggregates
    one row per tool per micro-batch). No-op for fresh DBs (0001 already
    creates the column via the current model)."""
    cols = {r[1] for r in conn.exec_driver_sql("PRAGMA table_info(tool_metrics)")} \
        if conn.engine.dialect.name == "sqlite" else set()
    if conn.engine.dialect.name == "sqlite":
        if "count" not in cols:
            conn.exec_driver_sql("ALTER TABLE tool_metrics ADD COLUMN count INTEGER DEFAULT 1")
    else:  # pragma: no cover - postgres path
        conn.exec_driver_sql("ALTER TABLE tool_metrics ADD COLUMN IF NOT EXISTS count INTEGER DEFAULT 1")


def _rev_0003_plugin_bindings(conn: Connection) -> None:
    """Per-tool plugin bindings table. No-op for fresh DBs (0001 create_all
    already built it from the current model)."""
    Base.metadata.tables["plugin_bindings"].create(conn, checkfirst=True)


def _rev_0004_gateway_lifecycle(conn: Connection) -> None:
    """Async registration lifecycle columns on gateways (retry backoff +
    failure classification — reference: gateway_service.py:4077-4362,:7469).
    No-op for fresh DBs (0001 create_all builds the current model)."""
    if conn.engine.dialect.name == "sqlite":
        cols = {r[1] for r in conn.exec_driver_sql("PRAGMA table_info(gateways)")}
        if not cols:  # legacy DB without the table: build it from the model
            Base.metadata.tables["gateways"].create(conn, checkfirst=True)
            return
        ddl = [("retry_count", "INTEGER DEFAULT 0"), ("next_retry_at", "FLOAT"),
               ("last_error", "TEXT"), ("failure_class", "VARCHAR(32)")]
        for name, typ in ddl:
            if name not in cols:
                conn.exec_driver_sql(f"ALTER TABLE gateways ADD COLUMN {name} {typ}")
    else:  # pragma: no cover - postgres path
        conn.exec_driver_sql("ALTER TABLE gateways ADD COLUMN IF NOT EXISTS retry_count INTEGER DEFAULT 0")
        conn.exec_driver_sql("ALTER TABLE gateways ADD COLUMN IF NOT EXISTS next_retry_at FLOAT")
        conn.exec_driver_sql("ALTER TABLE gateways ADD COLUMN IF NOT EXISTS last_error TEXT")
        conn.exec_driver_sql("ALTER TABLE gateways ADD COLUMN IF NOT EXISTS failure_class VARCHAR(32)")


def _rev_0005_oauth_tokens(conn: Connection) -> None:
    """Upstream OAuth token storage (reference: token_storage_service.py)."""
    Base.metadata.tables["oauth_tokens"].create(conn, checkfirst=True)


def _rev_0006_leader_leases(conn: Connection) -> None:
    """DB leader lease for shared-DB multi-process deployments
    (reference: Redis SET NX leader, gateway_service.py:1254/:5272)."""
    conn.exec_driver_sql(
        "CREATE TABLE IF NOT EXISTS leader_leases ("
        "name VARCHAR(128) PRIMARY KEY, holder VARCHAR(255), expires_at FLOAT)")


def _rev_0007_token_usage(conn: Connection) -> None:
    """Hourly-bucketed credential usage (reference: TokenUsageLog)."""
    Base.metadata.tables["token_usage"].create(conn, checkfirst=True)


# Linear chain: (revision_id, apply_fn). Append-only.
MIGRATIONS: List[Tuple[str, Callable[[Connection], None]]] = [
    ("0001_initial_registry", _rev_0001_initial),
    ("0002_tool_metrics_count", _rev_0002_tool_metrics_count),
    ("0003_plugin_bindings", _rev_0003_plugin_bindings),
    ("0004_gateway_lifecycle", _rev_0004_gateway_lifecycle),
    ("0005_oauth_tokens", _rev_0005_oauth_tokens),
    ("0006_leader_leases", _rev_0006_leader_leases),
    ("0007_token_usage", _rev_0007_token_usage),
]


def run_migrations(engine: Engine, retries: int = 10) -> List[str]:
    """Apply pending revisions; returns the list applied (bootstrap_db analog).

    Concurrency-safe for multi-rank startup against one shared database:
    several processes may race `upgrade head` (exactly like N gunicorn
    workers racing alembic in the reference) — losers retry and observe the
    winner's revisions.
    """
    import time as _time

    last_exc: Exception = RuntimeError("unreachable")
    for attempt in range(retries):
        applied: List[str] = []
        try:
            with engine.begin() as conn:
                conn.execute(
                    text(
                        "CREATE TABLE IF NOT EXISTS forge_schema_version ("
                        "revision VARCHAR(64) PRIMARY KEY, applied_at TIMESTAMP DEFAULT CURRENT_TIMESTAMP)"
                    )
                )
                done = {r[0] for r in conn.execute(text("SELECT revision FROM forge_schema_version"))}
                for rev, fn in MIGRATIONS:
                    if rev in done:
                        continue
                    fn(conn)
                    conn.execute(text("INSERT INTO forge_schema_version (revision) VALUES (:r)"), {"r": rev})
                    applied.append(rev)
            return applied
        except Exception as exc:  # racing peer won — back off and re-read
            last_exc = exc
            _time.sleep(0.1 * (attempt + 1))
    raise last_exc


class Database:
    """Session factory + lifecycle wrapper (reference: db.py SessionLocal + ResilientSession :422)."""

    def __init__(self, database_url: str, pool_size: int = 16, max_overflow: int = 8):
        self.url = database_url
        self.engine = build_engine(database_url, pool_size, max_overflow)
        self._sessionmaker = sessionmaker(bind=self.engine, expire_on_commit=False, future=True)
        self._lock = threading.Lock()

    def migrate(self) -> List[str]:
        return run_migrations(self.engine)

    @contextmanager
    def session(self) -> Iterator[Session]:
        s = self._sessionmaker()
        try:
            yield s
            s.commit()
        except Exception:
            s.rollback()
            raise
        finally:
            s.close()

    def close(self) -> None:
        self.engine.dispose()
