"""YAML log-ETL pipelines (ref src/pipeline/src/etl: processors + transform
+ dispatcher/table-suffix routing)."""

import numpy as np
import pytest

from greptimedb_amd.pipeline import Pipeline, PipelineError, PipelineStore

NGINX = """
version: 2
processors:
  - dissect:
      fields: [message]
      patterns:
        - '%{ip} - - [%{ts}] "%{method} %{path}" %{status} %{size}'
  - date:
      fields: [ts]
      formats: ['%d/%b/%Y:%H:%M:%S %z']
  - letter:
      fields: [method]
      method: lower
transform:
  - fields: [status, size]
    type: int32
  - fields: [method, ip]
    type: string
    index: tag
  - fields: [message]
    type: string
    index: fulltext
  - field: ts
    type: time
    index: timestamp
"""

LINE = ('10.0.0.7 - - [25/May/2024:20:16:37 +0000] '
        '"GET /api/v1/items" 200 512')


def test_dissect_date_transform():
    p = Pipeline.from_yaml(NGINX)
    row, sfx = p.run_row({"message": LINE})
    assert row["ip"] == "10.0.0.7" and row["method"] == "get"
    assert row["status"] == 200 and row["size"] == 512
    assert row["ts"] == 1716668197000 and sfx == ""
    assert p.tag_keys == ["method", "ip"]
    assert p.fulltext_keys == ["message"]
    assert p.ts_key == "ts"


def test_processors_gsub_letter_csv_json():
    p = Pipeline.from_yaml("""
processors:
  - gsub:
      fields: [m]
      pattern: '\\d+'
      replacement: 'N'
  - letter: {fields: [u], method: upper}
  - csv:
      fields: [c]
      target_fields: [a, b]
  - json_parse: {fields: [j]}
  - simple_extract:
      fields:
        - 'j, jv'
      key: k.v
  - join: {fields: [arr], separator: '-'}
  - urlencoding: {fields: [url], method: decode}
  - decolorize: {fields: [col]}
""")
    row, _ = p.run_row({
        "m": "err 42 at 7", "u": "abc", "c": 'x,"y,z"',
        "j": '{"k": {"v": 3}}', "arr": [1, 2, 3],
        "url": "a%20b%2Fc", "col": "\x1b[31mred\x1b[0m"})
    assert row["m"] == "err N at N"
    assert row["u"] == "ABC"
    assert row["a"] == "x" and row["b"] == "y,z"
    assert row["jv"] == 3
    assert row["arr"] == "1-2-3"
    assert row["url"] == "a b/c"
    assert row["col"] == "red"


def test_regex_named_groups_and_digest():
    p = Pipeline.from_yaml("""
processors:
  - regex:
      fields: [m]
      patterns: ['conn from (?<src>[\\d.]+):(?<port>\\d+)']
  - digest: {fields: [m]}
""")
    row, _ = p.run_row({"m": "conn from 10.1.2.3:443 attempt 7"})
    assert row["m_src"] == "10.1.2.3" and row["m_port"] == "443"
    assert row["m_digest"] == "conn from attempt"


def test_filter_drops_rows():
    p = Pipeline.from_yaml("""
processors:
  - filter:
      fields: [level]
      match_op: in
      targets: [debug]
""")
    assert p.run_row({"level": "DEBUG", "m": "x"}) is None
    row, _ = p.run_row({"level": "info", "m": "x"})
    assert row["m"] == "x"


def test_dispatcher_and_table_suffix():
    p = Pipeline.from_yaml("""
dispatcher:
  field: app
  rules:
    - value: web
      table_suffix: frontend
    - value: db
      table_suffix: database
""")
    groups = p.run([{"app": "web", "v": 1}, {"app": "db", "v": 2},
                    {"app": "other", "v": 3}])
    assert set(groups) == {"_frontend", "_database", ""}
    p2 = Pipeline.from_yaml("table_suffix: _${service}\n")
    _, sfx = p2.run_row({"service": "auth"})
    assert sfx == "_auth"


def test_epoch_and_select_and_v1():
    p = Pipeline.from_yaml("""
version: 1
processors:
  - epoch:
      fields: [t]
      resolution: s
transform:
  - field: t
    type: epoch
    index: timestamp
  - field: v
    type: float64
""")
    row, _ = p.run_row({"t": "1716668197", "v": "2.5", "junk": "drop-me"})
    assert row == {"t": 1716668197000, "v": 2.5}   # v1: only transformed kept
    p2 = Pipeline.from_yaml("""
processors:
  - select:
      type: exclude
      fields: [secret]
""")
    row, _ = p2.run_row({"a": 1, "secret": "x"})
    assert row == {"a": 1}


def test_pipeline_store_roundtrip(tmp_path):
    store = PipelineStore(str(tmp_path))
    store.put("nginx", NGINX)
    p = store.get("nginx")
    assert p.tag_keys == ["method", "ip"]
    assert store.list() == ["nginx"]
    store.delete("nginx")
    with pytest.raises(PipelineError):
        store.get("nginx")
    with pytest.raises(PipelineError):
        store.put("bad", "processors:\n  - nosuch: {}\n")


def test_ingest_with_pipeline(tmp_engine):
    from greptimedb_amd.engine.logstore import LogStore
    from greptimedb_amd.query.executor import Executor
    ls = LogStore(tmp_engine)
    p = Pipeline.from_yaml(NGINX)
    entries = [{"message": LINE},
               {"message": '10.0.0.8 - - [25/May/2024:20:16:38 +0000] '
                           '"POST /login" 401 64'}]
    n = ls.ingest_with_pipeline("nginx_logs", entries, p)
    assert n == 2
    ex = Executor(tmp_engine)
    r = ex.execute("SELECT ip, status, size FROM nginx_logs ORDER BY ts")
    assert [tuple(t) for t in r.rows()] == [("10.0.0.7", 200.0, 512.0),
                                            ("10.0.0.8", 401.0, 64.0)]
    # message is fulltext-indexed → MATCHES works
    r = ex.execute("SELECT count(*) FROM nginx_logs WHERE matches(message, 'login')")
    assert list(r.rows())[0][0] == 1


def test_cmcd_processor():
    """CMCD key=value media telemetry (reference processor/cmcd.rs)."""
    p = Pipeline.from_yaml("""
processors:
  - cmcd:
      fields:
        - data
transform:
  - fields:
      - data_sid
    type: string
  - fields:
      - data_br
    type: int64
""")
    row, _sfx = p.run_row({"data": 'br=3200,bs,d=4004,sid="abc-123"'})
    assert row["data_br"] == 3200
    assert row["data_sid"] == "abc-123"


def test_vrl_processor():
    """VRL remap subset (reference processor/vrl.rs)."""
    p = Pipeline.from_yaml("""
processors:
  - vrl:
      source: |
        .service = upcase(.svc)
        .latency_ms = to_float(.lat) * 1000
        if .latency_ms > 500 { .slow = "yes" } else { .slow = "no" }
        .msg = trim(.msg) + " [" + .service + "]"
        del(.svc)
transform:
  - fields:
      - service
      - slow
      - msg
    type: string
  - fields:
      - latency_ms
    type: float64
""")
    row, _sfx = p.run_row({"svc": "api", "lat": "0.75", "msg": "  hello "})
    assert row["service"] == "API"
    assert row["latency_ms"] == 750.0
    assert row["slow"] == "yes"
    assert row["msg"] == "hello [API]"
    assert "svc" not in row


def test_vrl_functions():
    from greptimedb_amd.pipeline.vrl import VrlProgram
    p = VrlProgram('''
.parts = split(.csv, ",")
.n = length(.parts)
.joined = join(.parts, "-")
.has = contains(.csv, "b")
.hash = sha256("x")
''')
    row = p.run({"csv": "a,b,c"})
    assert row["parts"] == ["a", "b", "c"]
    assert row["n"] == 3 and row["joined"] == "a-b-c" and row["has"]
    assert len(row["hash"]) == 64
