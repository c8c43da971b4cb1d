"""Golden SQL cases (reference parity: tests/cases sqlness runner).

Each tests/cases/*.sql runs statement-by-statement against a fresh engine;
outputs are compared to the committed .result golden. Regenerate with:
  python tests/test_sqlness.py --update
"""

import glob
import math
import os

import numpy as np
import pytest

CASES_DIR = os.path.join(os.path.dirname(__file__), "cases")


def _fmt_value(v):
    if v is None:
        return "NULL"
    if isinstance(v, (float, np.floating)):
        if math.isnan(v):
            return "NaN"
        return f"{float(v):g}"
    return str(v)


def run_case(path, engine):
    from greptimedb_amd.query.executor import Executor
    from greptimedb_amd.utils.errors import GreptimeError
    ex = Executor(engine)
    out = []
    for stmt in open(path).read().split(";"):
        stmt = stmt.strip()
        if not stmt or stmt.startswith("--"):
            continue
        out.append(f"-- {stmt}")
        try:
            r = ex.execute(stmt)
            out.append("| " + " | ".join(r.names) + " |")
            for row in r.rows():
                out.append("| " + " | ".join(_fmt_value(v) for v in row) + " |")
        except GreptimeError as e:
            out.append(f"ERROR: {type(e).__name__}")
        out.append("")
    return "\n".join(out)


@pytest.mark.parametrize("case", sorted(glob.glob(f"{CASES_DIR}/*.sql")),
                         ids=lambda p: os.path.basename(p))
def test_golden_case(case, tmp_engine):
    got = run_case(case, tmp_engine)
    golden = case.replace(".sql", ".result")
    assert os.path.exists(golden), f"missing golden {golden}; run --update"
    assert got == open(golden).read(), f"golden mismatch for {case}"


if __name__ == "__main__":
    import sys
    import tempfile
    repo_root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    sys.path.insert(0, repo_root)
    os.chdir(repo_root)  # engines resolve relative paths against the root
    from greptimedb_amd.engine.engine import EngineConfig, MitoEngine
    for case in sorted(glob.glob(f"{CASES_DIR}/*.sql")):
        eng = MitoEngine(EngineConfig(data_dir=tempfile.mkdtemp(), device="cpu",
                                      background_flush=False))
        open(case.replace(".sql", ".result"), "w").write(run_case(case, eng))
        eng.close()
        print("updated", case)
