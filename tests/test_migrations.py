"""Migration-chain upgrade paths (reference analog: tests/migration/ —
upgrade-path tests across schema versions; our chain: db/engine.py
MIGRATIONS, forge_schema_version, append-only)."""

import sqlalchemy as sa

from mcp_context_forge_amd.db.engine import MIGRATIONS, build_engine, run_migrations


def test_fresh_db_applies_full_chain(tmp_path):
    eng = build_engine(f"sqlite:///{tmp_path}/m.db")
    applied = run_migrations(eng)
    assert applied == [rev for rev, _ in MIGRATIONS]
    with eng.connect() as c:
        rows = [r[0] for r in c.exec_driver_sql(
            "SELECT revision FROM forge_schema_version ORDER BY applied_at, revision")]
    assert rows == [rev for rev, _ in MIGRATIONS]
    # idempotent: second run applies nothing
    assert run_migrations(eng) == []


def test_upgrade_from_older_schema(tmp_path):
    """A DB stopped at revision 0002 picks up only the newer revisions —
    the upgrade path a deployed gateway takes across releases."""
    url = f"sqlite:///{tmp_path}/old.db"
    eng = build_engine(url)
    run_migrations(eng)
    with eng.begin() as c:
        # roll the DB back to "release N-1": drop the newest revision's
        # artifacts and its version row
        c.exec_driver_sql("DELETE FROM forge_schema_version WHERE revision LIKE '0003%'")
        c.exec_driver_sql("DROP TABLE plugin_bindings")
    eng.dispose()

    eng2 = build_engine(url)
    applied = run_migrations(eng2)
    assert applied == ["0003_plugin_bindings"]
    with eng2.connect() as c:
        tables = {r[0] for r in c.exec_driver_sql(
            "SELECT name FROM sqlite_master WHERE type='table'")}
    assert "plugin_bindings" in tables


def test_migration_0002_adds_count_to_legacy_table(tmp_path):
    """Pre-0002 tool_metrics (no `count` column) gains it on upgrade."""
    url = f"sqlite:///{tmp_path}/legacy.db"
    eng = build_engine(url)
    with eng.begin() as c:
        # build a legacy 0001-era table by hand (old shape: no count column)
        c.exec_driver_sql("""CREATE TABLE tool_metrics (
            id INTEGER PRIMARY KEY, tool_id VARCHAR(36), timestamp DATETIME,
            response_time_ms FLOAT, is_success BOOLEAN, error_message TEXT)""")
        c.exec_driver_sql(
            "CREATE TABLE forge_schema_version (revision VARCHAR(64) PRIMARY KEY, "
            "applied_at TIMESTAMP DEFAULT CURRENT_TIMESTAMP)")
        c.exec_driver_sql(
            "INSERT INTO forge_schema_version (revision) VALUES ('0001_initial_registry')")
    eng.dispose()

    eng2 = build_engine(url)
    applied = run_migrations(eng2)
    assert applied[0] == "0002_tool_metrics_count"
    with eng2.connect() as c:
        cols = {r[1] for r in c.exec_driver_sql("PRAGMA table_info(tool_metrics)")}
    assert "count" in cols
