"""Meta layer: φ-accrual failure detection, procedures, region migration."""

import pytest

from greptimedb_amd.meta.failure_detector import PhiAccrualFailureDetector
from greptimedb_amd.meta.procedure import Procedure, ProcedureManager, Status
from greptimedb_amd.meta.supervisor import RegionSupervisor


def test_phi_low_while_heartbeating():
    det = PhiAccrualFailureDetector()
    t = 0.0
    for _ in range(50):
        det.heartbeat(t)
        t += 1000
    assert det.phi(t + 500) < 1.0
    assert det.is_available(t + 500)


def test_phi_rises_after_silence():
    det = PhiAccrualFailureDetector(acceptable_heartbeat_pause_ms=0)
    t = 0.0
    for _ in range(50):
        det.heartbeat(t)
        t += 1000
    # 60s of silence with 1s cadence → clearly dead
    assert det.phi(t + 60_000) > 8.0
    assert not det.is_available(t + 60_000)


def test_supervisor_failover_once():
    fired = []
    sup = RegionSupervisor(on_failover=fired.append)
    sup.detectors_threshold = 8
    t = 0.0
    for _ in range(30):
        sup.heartbeat("gpu0", t)
        sup.heartbeat("gpu1", t)
        t += 1000
    # gpu1 goes silent
    for _ in range(30):
        sup.heartbeat("gpu0", t)
        t += 1000
    failed = sup.check(t)
    assert failed == ["gpu1"] and fired == ["gpu1"]
    assert sup.check(t + 1000) == []  # fires once


class CountingProc(Procedure):
    TYPE = "counting"

    def initial_state(self):
        return {"i": 0}

    def step(self, state):
        state["i"] += 1
        if state["i"] >= 3:
            return Status.DONE, state
        return Status.EXECUTING, state


class CrashingProc(Procedure):
    TYPE = "crashing"
    crash = True

    def initial_state(self):
        return {"i": 0}

    def step(self, state):
        state["i"] += 1
        if state["i"] == 2 and CrashingProc.crash:
            raise RuntimeError("simulated crash")
        if state["i"] >= 3:
            return Status.DONE, state
        return Status.EXECUTING, state


def test_procedure_runs_to_done(tmp_path):
    pm = ProcedureManager(str(tmp_path / "proc"))
    pm.register(CountingProc)
    pid = pm.submit(CountingProc())
    assert pm.store.load_all() == []  # cleaned up
    assert pid


def test_procedure_crash_resume(tmp_path):
    pm = ProcedureManager(str(tmp_path / "proc"))
    pm.register(CrashingProc)
    CrashingProc.crash = True
    with pytest.raises(RuntimeError):
        pm.submit(CrashingProc())
    # state persisted at i=1
    recs = pm.store.load_all()
    assert len(recs) == 1 and recs[0]["state"]["i"] == 1
    # restart: recover() resumes from persisted state
    CrashingProc.crash = False
    pm2 = ProcedureManager(str(tmp_path / "proc"))
    pm2.register(CrashingProc)
    resumed = pm2.recover()
    assert len(resumed) == 1
    assert pm2.store.load_all() == []


def test_region_migration_cpu(tmp_engine):
    from greptimedb_amd.meta.migration import migrate_region
    from greptimedb_amd.query.executor import Executor
    ex = Executor(tmp_engine)
    ex.execute("CREATE TABLE t (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (h))")
    ex.execute("INSERT INTO t (h, ts, v) VALUES ('a', 1000, 1.0), ('b', 2000, 2.0)")
    before = ex.execute("SELECT h, ts, v FROM t ORDER BY ts").rows()
    for i in range(len(tmp_engine.table("t").regions)):
        migrate_region(tmp_engine, "t", i, "cpu")  # cpu→cpu exercises the path
    after = ex.execute("SELECT h, ts, v FROM t ORDER BY ts").rows()
    assert before == after


def test_meta_kv_cas_and_range(tmp_path):
    from greptimedb_amd.meta.kv import MetaKV
    kv = MetaKV(str(tmp_path / "meta.json"))
    kv.put("t/a", {"x": 1})
    kv.put("t/b", 2)
    assert kv.get("t/a") == {"x": 1}
    assert kv.range("t/") == {"t/a": {"x": 1}, "t/b": 2}
    assert kv.cas("t/a", {"x": 1}, {"x": 2})
    assert not kv.cas("t/a", {"x": 1}, {"x": 3})
    assert kv.get("t/a") == {"x": 2}
    assert kv.delete("t/b") and not kv.delete("t/b")
    # cas with expect None = create-if-absent
    assert kv.cas("t/new", None, 7) and not kv.cas("t/new", None, 8)


def test_meta_kv_leases_expire(tmp_path):
    import time
    from greptimedb_amd.meta.kv import MetaKV
    kv = MetaKV(str(tmp_path / "meta.json"))
    lid = kv.grant_lease(0.15)
    kv.put("lease/k", "v", lease=lid)
    assert kv.get("lease/k") == "v"
    assert kv.keepalive(lid)
    time.sleep(0.25)
    assert kv.get("lease/k") is None       # lease expired, key gone
    assert not kv.keepalive(lid)


def test_election(tmp_path):
    import time
    from greptimedb_amd.meta.kv import Election, MetaKV
    kv = MetaKV(str(tmp_path / "meta.json"))
    a = Election(kv, "election/leader", "node-a", ttl_s=0.2)
    b = Election(kv, "election/leader", "node-b", ttl_s=0.2)
    assert a.campaign()
    assert not b.campaign()                 # seat taken
    assert a.leader() == "node-a"
    assert a.campaign()                     # keepalive path
    a.resign()
    assert b.campaign()                     # seat free → b wins
    assert kv.get("election/leader") == "node-b"
    time.sleep(0.3)                         # b stops heartbeating → expiry
    assert a.campaign()
    assert a.leader() == "node-a"


def test_event_recorder(tmp_path):
    """DDL + migration events land in the greptime_events system table
    (reference src/common/event-recorder recorder.rs:311)."""
    from greptimedb_amd.engine.engine import EngineConfig, MitoEngine
    from greptimedb_amd.query.executor import Executor
    eng = MitoEngine(EngineConfig(data_dir=str(tmp_path / "ev"), device="cpu",
                                  background_flush=False, record_events=True))
    ex = Executor(eng)
    ex.execute("CREATE TABLE evt_src (h STRING, ts TIMESTAMP TIME INDEX,"
               " v DOUBLE, PRIMARY KEY (h))")
    ex.execute("DROP TABLE evt_src")
    r = ex.execute("SELECT event_type, payload FROM greptime_events"
                   " ORDER BY ts")
    types = [row[0] for row in r.rows()]
    assert "create_table" in types and "drop_table" in types
    import json as _json
    payloads = [_json.loads(row[1]) for row in r.rows()]
    assert any(p.get("table") == "evt_src" for p in payloads)
    eng.close()


def test_wire_heartbeat_and_mailbox(tmp_path):
    """Heartbeat/mailbox over the wire (reference datanode heartbeat.rs +
    mailbox_handler.rs): stats flow up, instructions piggyback down, and a
    stopped node trips the φ detector."""
    import socket
    import time as _time

    from greptimedb_amd.engine.engine import EngineConfig, MitoEngine
    from greptimedb_amd.engine.ingest import Ingestor
    from greptimedb_amd.meta.heartbeat import HeartbeatTask, MetaServer
    from greptimedb_amd.models.tsbs import CpuWorkload

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    failed = []
    srv = MetaServer(port, on_failover=failed.append, threshold=1.5,
                     acceptable_pause_ms=150.0)
    eng = MitoEngine(EngineConfig(data_dir=str(tmp_path / "d"), device="cpu",
                                  background_flush=False, default_regions=2))
    Ingestor(eng).ingest_lines(CpuWorkload(scale=4).next_batch(100))
    hb = HeartbeatTask("gpu0", eng, srv.port, interval_s=0.05)
    # regular beats → stats visible, no failover
    for _ in range(8):
        hb.beat_once()
        _time.sleep(0.02)
    assert "gpu0" in srv.stats
    assert sum(r["rows"] for r in srv.stats["gpu0"]["regions"]) == 100
    assert srv.check() == []
    # mailbox: downgrade instruction applies on the next beat
    srv.send_instruction("gpu0", {"kind": "downgrade_region",
                                  "table": "cpu", "region_idx": 0})
    got = hb.beat_once()
    assert got and got[0]["kind"] == "downgrade_region"
    assert not eng.table("cpu").regions[0].writable
    srv.send_instruction("gpu0", {"kind": "upgrade_region",
                                  "table": "cpu", "region_idx": 0})
    hb.beat_once()
    assert eng.table("cpu").regions[0].writable
    # silence → φ exceeds threshold → failover fires exactly once
    # (the seed samples from first_heartbeat_estimate keep the variance
    # wide early on, so give the pause a few multiples of the estimate)
    _time.sleep(2.5)
    newly = srv.check()
    assert newly == ["gpu0"] and failed == ["gpu0"]
    assert srv.check() == []
    hb.stop()
    srv.close()
    eng.close()


def test_maintenance_mode_suppresses_failover():
    """Maintenance mode: φ detection keeps running but no failover fires
    (reference: metasrv maintenance mode for planned restarts)."""
    from greptimedb_amd.meta.supervisor import RegionSupervisor
    failed = []
    sup = RegionSupervisor(on_failover=failed.append, threshold=1.5,
                           acceptable_pause_ms=100.0)
    t = 0.0
    for _ in range(20):
        sup.heartbeat("node-a", t)
        t += 100.0
    sup.maintenance = True
    assert sup.check(t + 60_000.0) == []
    assert failed == []
    sup.maintenance = False
    assert sup.check(t + 60_000.0) == ["node-a"]
    assert failed == ["node-a"]
