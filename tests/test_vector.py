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


def test_ivf_index_recall(tmp_engine):
    """IVF-flat kNN (ADMIN build_vector_index) vs exact brute force."""
    import torch
    ex = Executor(tmp_engine)
    ex.execute("CREATE TABLE vx (b STRING, ts TIMESTAMP TIME INDEX, "
               "emb VECTOR(8), PRIMARY KEY (b)) WITH ('append_mode'='true')")
    rng = np.random.RandomState(3)
    vecs = rng.randn(2000, 8).astype(np.float32)
    rows = []
    for i, v in enumerate(vecs):
        rows.append(f"('b', {1000 + i}, '[{','.join(f'{x:.5f}' for x in v)}]')")
    for s in range(0, len(rows), 500):
        ex.execute("INSERT INTO vx (b, ts, emb) VALUES " + ",".join(rows[s:s + 500]))
    tmp_engine.flush_all()     # vectors into SST sources (indexable)
    q = "[" + ",".join(f"{x:.5f}" for x in vecs[7]) + "]"
    sql = (f"SELECT ts, vec_l2sq_distance(emb, '{q}') AS d FROM vx "
           f"ORDER BY d LIMIT 10")
    exact = [int(t[0]) for t in ex.execute(sql).rows()]
    assert exact[0] == 1007     # the query vector itself
    r = ex.execute("ADMIN build_vector_index('vx', 'emb', 16, 16)")
    assert int(list(r.rows())[0][0]) >= 1   # at least one SST indexed
    # nprobe == nlist probes every list → identical to exact
    approx = [int(t[0]) for t in ex.execute(sql).rows()]
    assert approx == exact
    # small nprobe still finds the exact-match vector
    ex.execute("ADMIN build_vector_index('vx', 'emb', 32, 4)")
    approx4 = [int(t[0]) for t in ex.execute(sql).rows()]
    assert approx4[0] == 1007
    assert len(set(approx4) & set(exact)) >= 5   # decent recall@10


def test_ivf_kmeans_cpu():
    import torch
    from greptimedb_amd.vector import build_ivf, ivf_candidates
    x = torch.randn(500, 16, generator=torch.Generator().manual_seed(1))
    ivf = build_ivf(x, 8)
    assert ivf["centroids"].shape == (8, 16)
    assert int(ivf["offsets"][-1]) == 500
    # probing all lists returns every row exactly once
    rows = ivf_candidates(ivf, x[0], 8)
    assert sorted(rows.tolist()) == list(range(500))
