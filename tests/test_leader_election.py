"""DB leader lease (services/leader.py) — the shared-DB multi-process
analog of the reference's Redis SET NX leader + TTL re-election
(gateway_service.py:1254, :5272)."""

import asyncio
import time

from mcp_context_forge_amd.db.engine import Database
from mcp_context_forge_amd.services.leader import DbLeaderElector


def _db(tmp_path):
    db = Database(f"sqlite:///{tmp_path}/shared.db")
    db.migrate()
    return db


def test_single_holder(tmp_path):
    db = _db(tmp_path)
    a = DbLeaderElector(db, ttl_s=5.0, holder_id="A")
    b = DbLeaderElector(db, ttl_s=5.0, holder_id="B")
    assert a.try_acquire() is True
    assert b.try_acquire() is False
    assert a.is_leader and not b.is_leader
    # renewal by the holder succeeds
    assert a.try_acquire() is True
    db.close()


def test_takeover_on_expiry(tmp_path):
    db = _db(tmp_path)
    a = DbLeaderElector(db, ttl_s=0.2, holder_id="A")
    b = DbLeaderElector(db, ttl_s=0.2, holder_id="B")
    assert a.try_acquire()
    assert not b.try_acquire()
    time.sleep(0.3)  # A "crashed" — no heartbeat, lease expires
    assert b.try_acquire() is True
    # A comes back but the lease is B's now
    assert a.try_acquire() is False
    assert not a.is_leader
    db.close()


def test_release_hands_over(tmp_path):
    db = _db(tmp_path)
    a = DbLeaderElector(db, ttl_s=60.0, holder_id="A")
    b = DbLeaderElector(db, ttl_s=60.0, holder_id="B")
    assert a.try_acquire()
    a.release()
    assert b.try_acquire() is True
    db.close()


def test_heartbeat_loop_and_engine_gating(tmp_path, run):
    """Engine wiring: with leader_election_enabled, a follower engine's
    health/lifecycle ticks are no-ops while the lease is held elsewhere."""
    from mcp_context_forge_amd.config import Settings
    from mcp_context_forge_amd.engine import GatewayEngine

    async def go():
        url = f"sqlite:///{tmp_path}/gw.db"
        pre = Database(url)
        pre.migrate()
        blocker = DbLeaderElector(pre, ttl_s=0.4, holder_id="other-process")
        assert blocker.try_acquire()

        e = GatewayEngine(Settings(database_url=url, federation_enabled=False,
                                   auth_required=False, plugins_enabled=False,
                                   leader_election_enabled=True, leader_lease_ttl_s=0.3))
        await e.startup()
        assert e.leader_elector is not None
        assert e.leader_elector.is_leader is False
        assert await e.gateway_service.check_health_once() == {}
        assert (await e.gateway_service.lifecycle_tick())["activated"] == 0
        # the other process dies: its lease expires, the engine's heartbeat
        # (ttl/3 = 100 ms) takes over
        for _ in range(100):
            if e.leader_elector.is_leader:
                break
            await asyncio.sleep(0.05)
        assert e.leader_elector.is_leader is True
        await e.shutdown()
        pre.close()

    run(go())


def test_mutation_kills_initial_and_release_state(tmp_path):
    """Mutation-sweep survivors pinned down: a fresh elector is NOT leader
    until it acquires, and release actually relinquishes."""
    db = _db(tmp_path)
    a = DbLeaderElector(db, ttl_s=5.0, holder_id="A")
    assert a.is_leader is False           # pre-acquire state
    assert a.try_acquire()
    assert a.is_leader is True
    a.release()
    assert a.is_leader is False           # release flips local state too
    db.close()


def test_leader_lease_across_processes(tmp_path):
    """Two real PROCESSES share one sqlite file: the subprocess holds the
    lease, this process is denied, and after the holder dies + TTL expiry
    this process takes over — the deployment shape (shared-DB workers)."""
    import subprocess
    import sys
    import textwrap

    db_path = tmp_path / "xproc.db"
    script = textwrap.dedent(f"""
        import sys, time
        sys.path.insert(0, {repr(str(__import__('pathlib').Path(__file__).resolve().parents[1]))})
        from mcp_context_forge_amd.db.engine import Database
        from mcp_context_forge_amd.services.leader import DbLeaderElector
        db = Database("sqlite:///{db_path}")
        db.migrate()
        el = DbLeaderElector(db, ttl_s=1.0, holder_id="SUBPROC")
        assert el.try_acquire()
        print("HELD", flush=True)
        time.sleep(30)   # hold until killed
    """)
    proc = subprocess.Popen([sys.executable, "-c", script],
                            stdout=subprocess.PIPE, text=True)
    try:
        line = proc.stdout.readline().strip()
        assert line == "HELD", line
        db2 = Database(f"sqlite:///{db_path}")
        db2.migrate()
        me = DbLeaderElector(db2, ttl_s=1.0, holder_id="MAIN")
        assert me.try_acquire() is False, "other process holds the lease"
    finally:
        proc.kill()
        proc.wait(timeout=10)
    # holder crashed without releasing: lease expires after TTL
    time.sleep(1.2)
    assert me.try_acquire() is True
    assert me.is_leader
    db2.close()
