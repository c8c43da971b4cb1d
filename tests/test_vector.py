"""Vector column type + brute-force GPU kNN (vector index v0)."""

import numpy as np
import pytest

from greptimedb_amd.engine.engine import EngineConfig, MitoEngine
from greptimedb_amd.query.executor import Executor


@pytest.fixture
def ex(tmp_engine):
    ex = Executor(tmp_engine)
    ex.execute("CREATE TABLE docs (bucket STRING, ts TIMESTAMP TIME INDEX, "
               "title STRING, emb VECTOR(4), PRIMARY KEY (bucket)) "
               "WITH ('append_mode'='true')")
    rows = []
    vecs = [(1, 0, 0, 0), (0, 1, 0, 0), (0.9, 0.1, 0, 0), (0, 0, 1, 0), (0.5, 0.5, 0, 0)]
    for i, v in enumerate(vecs):
        rows.append(f"('b{i%2}', {1000+i}, 'doc{i}', '[{','.join(map(str,v))}]')")
    ex.execute("INSERT INTO docs (bucket, ts, title, emb) VALUES " + ",".join(rows))
    return ex


def test_knn_l2(ex):
    r = ex.execute("SELECT title, vec_l2sq_distance(emb, '[1,0,0,0]') AS d "
                   "FROM docs ORDER BY d LIMIT 3")
    assert list(r.columns[0]) == ["doc0", "doc2", "doc4"]
    assert r.columns[1][0] == 0.0


def test_knn_cos_with_filter(ex):
    r = ex.execute("SELECT title, vec_cos_distance(emb, '[1,0,0,0]') AS d "
                   "FROM docs WHERE bucket = 'b0' ORDER BY d LIMIT 2")
    # b0 holds doc0, doc2, doc4
    assert list(r.columns[0]) == ["doc0", "doc2"]


def test_knn_after_flush_and_reopen(ex):
    eng = ex.engine
    eng.flush_all()
    r = ex.execute("SELECT title, vec_l2sq_distance(emb, '[0,0,1,0]') AS d "
                   "FROM docs ORDER BY d LIMIT 1")
    assert list(r.columns[0]) == ["doc3"]
    d = eng.config.data_dir
    eng.close()
    eng2 = MitoEngine(EngineConfig(data_dir=d, device="cpu", background_flush=False))
    ex2 = Executor(eng2)
    r = ex2.execute("SELECT title, vec_l2sq_distance(emb, '[0,1,0,0]') AS d "
                    "FROM docs ORDER BY d LIMIT 2")
    assert list(r.columns[0]) == ["doc1", "doc4"]
    eng2.close()


def test_dot_product_desc(ex):
    r = ex.execute("SELECT title, vec_dot_product(emb, '[1,0,0,0]') AS s "
                   "FROM docs ORDER BY s DESC LIMIT 1")
    assert list(r.columns[0]) == ["doc0"]
    assert r.columns[1][0] == 1.0
