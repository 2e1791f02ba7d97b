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


def test_upgrade_across_release_boundary_preserves_data(tmp_path):
    """A deployed DB at revision 0003 (pre-lifecycle) upgrades through
    0004-0007 with its gateway rows intact and the new columns usable."""
    url = f"sqlite:///{tmp_path}/rel.db"
    eng = build_engine(url)
    run_migrations(eng)
    with eng.begin() as c:
        # roll back to "release with 0003": drop the 0004+ version rows and
        # the columns/tables they added (sqlite: rebuild gateways narrow),
        # then seed a legacy row in the OLD shape
        for rev in ("0004%", "0005%", "0006%", "0007%"):
            c.exec_driver_sql(f"DELETE FROM forge_schema_version WHERE revision LIKE '{rev}'")
        c.exec_driver_sql("DROP TABLE gateways")
        c.exec_driver_sql(
            "CREATE TABLE gateways (id VARCHAR PRIMARY KEY, name VARCHAR, url VARCHAR, "
            "transport VARCHAR, enabled BOOLEAN, reachable BOOLEAN)")
        c.exec_driver_sql(
            "INSERT INTO gateways VALUES ('g1', 'legacy', 'http://u', 'streamablehttp', 1, 1)")
        c.exec_driver_sql("DROP TABLE IF EXISTS oauth_tokens")
        c.exec_driver_sql("DROP TABLE IF EXISTS leader_leases")
        c.exec_driver_sql("DROP TABLE IF EXISTS token_usage")
    eng.dispose()

    eng2 = build_engine(url)
    applied = run_migrations(eng2)
    assert applied == ["0004_gateway_lifecycle", "0005_oauth_tokens",
                       "0006_leader_leases", "0007_token_usage"]
    with eng2.connect() as c:
        row = c.exec_driver_sql(
            "SELECT name, retry_count FROM gateways WHERE id='g1'").first()
        assert row[0] == "legacy"          # data preserved
        # new columns/tables usable
        c.exec_driver_sql("SELECT next_retry_at, failure_class FROM gateways")
        for t in ("oauth_tokens", "leader_leases", "token_usage"):
            c.exec_driver_sql(f"SELECT * FROM {t} LIMIT 1")


def test_migration_0004_builds_gateways_on_legacy_db(tmp_path):
    """A legacy DB that never had a gateways table at all (pre-federation
    deployment) gets it built from the current model."""
    url = f"sqlite:///{tmp_path}/nogw.db"
    eng = build_engine(url)
    run_migrations(eng)
    with eng.begin() as c:
        for rev in ("0004%", "0005%", "0006%", "0007%"):
            c.exec_driver_sql(f"DELETE FROM forge_schema_version WHERE revision LIKE '{rev}'")
        c.exec_driver_sql("DROP TABLE gateways")
        c.exec_driver_sql("DROP TABLE IF EXISTS oauth_tokens")
        c.exec_driver_sql("DROP TABLE IF EXISTS leader_leases")
        c.exec_driver_sql("DROP TABLE IF EXISTS token_usage")
    eng.dispose()
    eng2 = build_engine(url)
    assert "0004_gateway_lifecycle" in run_migrations(eng2)
    with eng2.connect() as c:
        cols = {r[1] for r in c.exec_driver_sql("PRAGMA table_info(gateways)")}
    assert {"id", "name", "url", "retry_count", "failure_class"} <= cols


def test_concurrent_migrations_single_winner(tmp_path):
    """Two processes racing run_migrations on a shared DB: the chain is
    applied exactly once (the retry path absorbs sqlite lock errors)."""
    import threading

    url = f"sqlite:///{tmp_path}/race.db"
    results = {}

    def runner(tag):
        eng = build_engine(url)
        try:
            results[tag] = run_migrations(eng)
        finally:
            eng.dispose()

    ts = [threading.Thread(target=runner, args=(i,)) for i in range(2)]
    for t in ts:
        t.start()
    for t in ts:
        t.join()
    all_applied = sum(results.values(), [])
    assert sorted(all_applied) == sorted(rev for rev, _ in MIGRATIONS), results
    eng = build_engine(url)
    with eng.connect() as c:
        rows = [r[0] for r in c.exec_driver_sql("SELECT revision FROM forge_schema_version")]
    assert sorted(rows) == sorted(rev for rev, _ in MIGRATIONS)
