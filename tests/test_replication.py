"""Leader/follower replication + catchup + promotion (P7; ref store-api
region roles + mito2 handle_catchup WAL replay)."""

import numpy as np

from greptimedb_amd.engine.engine import EngineConfig, MitoEngine
from greptimedb_amd.engine.ingest import Ingestor
from greptimedb_amd.meta.replication import FollowerReplica
from greptimedb_amd.models.tsbs import CpuWorkload
from greptimedb_amd.query.executor import Executor


def _mk(tmp_path, name):
    return MitoEngine(EngineConfig(data_dir=str(tmp_path / name), device="cpu",
                                   background_flush=False))


def test_follower_catchup_and_promote(tmp_path):
    leader = _mk(tmp_path, "leader")
    ing = Ingestor(leader)
    w = CpuWorkload(scale=13, seed=2)
    for _ in range(3):
        ing.ingest_lines(w.next_batch(500))
    leader.commit_wal()

    follower = _mk(tmp_path, "follower")
    rep = FollowerReplica(str(tmp_path / "leader"), follower)
    rep.sync_catalog()
    assert "cpu" in follower.tables
    # region ids must match for WAL routing
    assert [r.region_id for r in follower.table("cpu").regions] == \
           [r.region_id for r in leader.table("cpu").regions]
    applied = rep.catchup()
    assert applied == 1500

    q = "SELECT count(*), sum(usage_user) FROM cpu"
    le, fe = Executor(leader), Executor(follower)
    lrow = list(le.execute(q).rows())[0]
    frow = list(fe.execute(q).rows())[0]
    assert int(lrow[0]) == int(frow[0]) == 1500
    assert abs(float(lrow[1]) - float(frow[1])) < 1e-9

    # incremental: more writes, catchup only applies the delta
    for _ in range(2):
        ing.ingest_lines(w.next_batch(500))
    leader.commit_wal()
    assert rep.catchup() == 1000
    assert rep.catchup() == 0          # idempotent
    assert int(list(fe.execute("SELECT count(*) FROM cpu").rows())[0][0]) == 2500

    # promotion flips the role with no data movement
    rep.promote()
    assert rep.role == "leader"
    assert rep.catchup() == 0          # leaders do not tail
    leader.close()
    follower.close()


def test_follower_survives_leader_flush_purge(tmp_path):
    """Entries already applied may be purged from the leader WAL after a
    leader flush; the follower keeps its copy and stays consistent."""
    leader = _mk(tmp_path, "leader")
    ing = Ingestor(leader)
    w = CpuWorkload(scale=7, seed=9)
    ing.ingest_lines(w.next_batch(400))
    leader.commit_wal()

    follower = _mk(tmp_path, "follower")
    rep = FollowerReplica(str(tmp_path / "leader"), follower)
    rep.sync_catalog()
    assert rep.catchup() == 400

    leader.flush_all()                  # purges applied WAL segments
    ing.ingest_lines(w.next_batch(300))
    leader.commit_wal()
    assert rep.catchup() == 300
    fe = Executor(follower)
    assert int(list(fe.execute("SELECT count(*) FROM cpu").rows())[0][0]) == 700
    leader.close()
    follower.close()
