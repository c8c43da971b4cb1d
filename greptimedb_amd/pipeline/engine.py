"""YAML log-ETL pipeline engine.

Reference parity: src/pipeline/src/etl (YAML-defined pipelines; processor
chain + transform section + dispatcher/table-suffix routing — see
processor/{dissect,date,epoch,regex,gsub,letter,csv,json_parse,json_path,
simple_extract,join,filter,select,urlencoding,decolorize,digest}.rs and
etl/transform). MI355X redesign note: processors run host-side on the
ingest path (string work), producing typed columns that land in the GPU
memtable via the normal LogStore path; fulltext/tag/timestamp index hints
from the transform section drive table schema creation.

Version semantics (GreptimeDB pipeline versions):
  v1 — if a `transform` section exists, ONLY transformed fields are kept;
  v2 — transformed fields are typed/indexed, untouched fields pass through
       with identity semantics.
"""

from __future__ import annotations

import json
import re
import urllib.parse

import yaml

from greptimedb_amd.utils.errors import InvalidSyntax


class PipelineError(InvalidSyntax):
    pass


_ANSI = re.compile(r"\x1b\[[0-9;]*m")
_DIGEST_PATTERNS = [
    (re.compile(r"'[^']*'|\"[^\"]*\""), ""),            # quoted strings
    (re.compile(r"[0-9a-fA-F]{8}-[0-9a-fA-F]{4}-[0-9a-fA-F]{4}-"
                r"[0-9a-fA-F]{4}-[0-9a-fA-F]{12}"), ""),  # uuid
    (re.compile(r"\b(?:\d{1,3}\.){3}\d{1,3}(?::\d+)?\b"), ""),  # ipv4[:port]
    (re.compile(r"\([^)]*\)|\[[^\]]*\]|\{[^}]*\}|<[^>]*>"), ""),  # brackets
    (re.compile(r"\b\d+\b"), ""),                        # numbers
]


def _listify(v):
    if v is None:
        return []
    return v if isinstance(v, list) else [v]


def _fields_of(cfg: dict) -> list[tuple[str, str]]:
    """fields/field entries; 'src, dst' renames (reference field syntax)."""
    out = []
    for f in _listify(cfg.get("fields") or cfg.get("field")):
        if isinstance(f, str) and "," in f:
            src, dst = (p.strip() for p in f.split(",", 1))
            out.append((src, dst))
        else:
            out.append((str(f), str(f)))
    return out


class _Dissect:
    """%{key} patterns with literal separators; modifiers: %{} / %{?skip}
    ignored keys, %{+key} append (append_separator), %{key->} pad-skip."""

    def __init__(self, cfg):
        self.fields = _fields_of(cfg)
        self.ignore_missing = bool(cfg.get("ignore_missing"))
        self.append_sep = cfg.get("append_separator", " ")
        self.patterns = []
        for pat in _listify(cfg.get("patterns") or cfg.get("pattern")):
            self.patterns.append(self._compile(pat))

    @staticmethod
    def _compile(pat: str):
        # split into [(literal, key|None, append, pad)]
        parts = []
        i = 0
        rx = re.compile(r"%\{([^}]*)\}")
        last = 0
        for m in rx.finditer(pat):
            lit = pat[last:m.start()]
            key = m.group(1)
            last = m.end()
            pad = key.endswith("->")
            if pad:
                key = key[:-2]
            append = key.startswith("+")
            if append:
                key = key[1:]
            if key.startswith("?") or key == "":
                key = None  # named-skip / skip
            parts.append((lit, key, append, pad))
        tail = pat[last:]
        return parts, tail

    def _match(self, parts, tail, text):
        out = {}
        pos = 0
        n = len(text)
        for idx, (lit, key, append, pad) in enumerate(parts):
            if lit:
                if not text.startswith(lit, pos):
                    return None
                pos += len(lit)
            # find the next literal to bound this key's value
            nxt = None
            for lit2, _k, _a, _p in parts[idx + 1:]:
                if lit2:
                    nxt = lit2
                    break
            else:
                nxt = tail if tail else None
            if nxt:
                end = text.find(nxt, pos)
                if end < 0:
                    return None
            else:
                end = n
            val = text[pos:end]
            pos = end
            if pad:
                val = val.rstrip()
            if key is not None:
                if append and key in out:
                    out[key] = out[key] + self.append_sep + val
                else:
                    out[key] = val
        if tail and not text.startswith(tail, pos):
            return None
        return out

    def __call__(self, row):
        for src, _dst in self.fields:
            v = row.get(src)
            if v is None:
                if self.ignore_missing:
                    continue
                raise PipelineError(f"dissect: field {src} missing")
            for parts, tail in self.patterns:
                got = self._match(parts, tail, str(v))
                if got is not None:
                    row.update(got)
                    break
        return row


class _Regex:
    def __init__(self, cfg):
        self.fields = _fields_of(cfg)
        self.ignore_missing = bool(cfg.get("ignore_missing"))
        self.patterns = [re.compile(p.replace("(?<", "(?P<"))
                         for p in _listify(cfg.get("patterns") or cfg.get("pattern"))]

    def __call__(self, row):
        for src, _dst in self.fields:
            v = row.get(src)
            if v is None:
                if self.ignore_missing:
                    continue
                raise PipelineError(f"regex: field {src} missing")
            for rx in self.patterns:
                m = rx.search(str(v))
                if m:
                    for k, g in m.groupdict().items():
                        if g is not None:
                            row[f"{src}_{k}"] = g
        return row


class _Date:
    def __init__(self, cfg):
        self.fields = _fields_of(cfg)
        self.formats = _listify(cfg.get("formats") or cfg.get("format")) or None
        self.ignore_missing = bool(cfg.get("ignore_missing"))
        tz = cfg.get("timezone")
        self.tz_off_s = 0
        if tz and isinstance(tz, str) and re.fullmatch(r"[+-]\d{2}:?\d{2}", tz):
            sign = -1 if tz[0] == "-" else 1
            hh, mm = int(tz[1:3]), int(tz[-2:])
            self.tz_off_s = sign * (hh * 3600 + mm * 60)

    def _parse(self, s: str) -> int:
        import datetime as dt
        fmts = self.formats or ["%Y-%m-%dT%H:%M:%S%z", "%Y-%m-%d %H:%M:%S",
                                "%Y-%m-%dT%H:%M:%S", "%d/%b/%Y:%H:%M:%S %z"]
        for f in fmts:
            try:
                d = dt.datetime.strptime(s, f)
            except ValueError:
                continue
            if d.tzinfo is None:
                d = d.replace(tzinfo=dt.timezone.utc)
                return int(d.timestamp() * 1000) - self.tz_off_s * 1000
            return int(d.timestamp() * 1000)
        raise PipelineError(f"date: cannot parse {s!r}")

    def __call__(self, row):
        for src, dst in self.fields:
            v = row.get(src)
            if v is None:
                if self.ignore_missing:
                    continue
                raise PipelineError(f"date: field {src} missing")
            row[dst] = self._parse(str(v))
        return row


class _Epoch:
    _DIV = {"s": 0.001, "second": 0.001, "sec": 0.001,
            "ms": 1, "millisecond": 1, "milli": 1,
            "us": 1000, "microsecond": 1000, "micro": 1000,
            "ns": 1_000_000, "nanosecond": 1_000_000, "nano": 1_000_000}

    def __init__(self, cfg):
        self.fields = _fields_of(cfg)
        self.ignore_missing = bool(cfg.get("ignore_missing"))
        res = str(cfg.get("resolution", "ms"))
        if res not in self._DIV:
            raise PipelineError(f"epoch: bad resolution {res}")
        self.div = self._DIV[res]

    def __call__(self, row):
        for src, dst in self.fields:
            v = row.get(src)
            if v is None:
                if self.ignore_missing:
                    continue
                raise PipelineError(f"epoch: field {src} missing")
            x = float(v)
            row[dst] = int(x / self.div) if self.div >= 1 else int(x * 1000)
        return row


class _Gsub:
    def __init__(self, cfg):
        self.fields = _fields_of(cfg)
        self.rx = re.compile(cfg.get("pattern", ""))
        self.replacement = cfg.get("replacement", "")
        self.ignore_missing = bool(cfg.get("ignore_missing"))

    def __call__(self, row):
        for src, dst in self.fields:
            v = row.get(src)
            if v is None:
                continue
            row[dst] = self.rx.sub(self.replacement, str(v))
        return row


class _Letter:
    def __init__(self, cfg):
        self.fields = _fields_of(cfg)
        self.method = cfg.get("method", "lower")

    def __call__(self, row):
        for src, dst in self.fields:
            v = row.get(src)
            if v is None:
                continue
            s = str(v)
            row[dst] = (s.upper() if self.method == "upper" else
                        s.capitalize() if self.method == "capital" else s.lower())
        return row


class _Csv:
    def __init__(self, cfg):
        self.fields = _fields_of(cfg)
        self.sep = cfg.get("separator", ",")
        self.quote = cfg.get("quote", '"')
        self.targets = _listify(cfg.get("target_fields"))
        if isinstance(self.targets, str):
            self.targets = [t.strip() for t in self.targets.split(",")]

    def __call__(self, row):
        import csv as _csv
        import io
        for src, _dst in self.fields:
            v = row.get(src)
            if v is None:
                continue
            r = next(_csv.reader(io.StringIO(str(v)), delimiter=self.sep,
                                 quotechar=self.quote))
            for name, val in zip(self.targets, r):
                row[name] = val
        return row


class _JsonParse:
    def __init__(self, cfg):
        self.fields = _fields_of(cfg)
        self.ignore_missing = bool(cfg.get("ignore_missing"))

    def __call__(self, row):
        for src, dst in self.fields:
            v = row.get(src)
            if v is None:
                if self.ignore_missing:
                    continue
                raise PipelineError(f"json_parse: field {src} missing")
            try:
                row[dst] = json.loads(v) if isinstance(v, str) else v
            except json.JSONDecodeError as e:
                raise PipelineError(f"json_parse: {e}") from None
        return row


def _json_path_get(obj, path: str):
    """Minimal JSONPath: $.a.b[0].c (reference json_path uses jsonpath_rust)."""
    cur = obj
    for part in re.findall(r"\.([A-Za-z_][\w]*)|\[(\d+)\]|\['([^']+)'\]",
                           path.lstrip("$")):
        key = part[0] or part[2]
        if key:
            if not isinstance(cur, dict):
                return None
            cur = cur.get(key)
        else:
            i = int(part[1])
            if not isinstance(cur, list) or i >= len(cur):
                return None
            cur = cur[i]
        if cur is None:
            return None
    return cur


class _JsonPath:
    def __init__(self, cfg):
        self.fields = _fields_of(cfg)
        self.path = cfg.get("json_path") or cfg.get("path")
        self.result_index = cfg.get("result_index")

    def __call__(self, row):
        for src, dst in self.fields:
            v = row.get(src)
            if v is None:
                continue
            if isinstance(v, str):
                try:
                    v = json.loads(v)
                except json.JSONDecodeError:
                    continue
            got = _json_path_get(v, self.path)
            if isinstance(got, list) and self.result_index is not None:
                got = got[self.result_index] if self.result_index < len(got) else None
            row[dst] = got
        return row


class _SimpleExtract:
    def __init__(self, cfg):
        self.fields = _fields_of(cfg)
        self.key = cfg.get("key", "")

    def __call__(self, row):
        for src, dst in self.fields:
            v = row.get(src)
            if isinstance(v, str):
                try:
                    v = json.loads(v)
                except json.JSONDecodeError:
                    continue
            cur = v
            for part in self.key.split("."):
                if not isinstance(cur, dict):
                    cur = None
                    break
                cur = cur.get(part)
            row[dst] = cur
        return row


class _Join:
    def __init__(self, cfg):
        self.fields = _fields_of(cfg)
        self.sep = cfg.get("separator", ",")

    def __call__(self, row):
        for src, dst in self.fields:
            v = row.get(src)
            if isinstance(v, list):
                row[dst] = self.sep.join(str(x) for x in v)
        return row


class _UrlEncoding:
    def __init__(self, cfg):
        self.fields = _fields_of(cfg)
        self.method = cfg.get("method", "decode")

    def __call__(self, row):
        for src, dst in self.fields:
            v = row.get(src)
            if v is None:
                continue
            row[dst] = (urllib.parse.quote(str(v)) if self.method == "encode"
                        else urllib.parse.unquote(str(v)))
        return row


class _Decolorize:
    def __init__(self, cfg):
        self.fields = _fields_of(cfg)

    def __call__(self, row):
        for src, dst in self.fields:
            v = row.get(src)
            if v is not None:
                row[dst] = _ANSI.sub("", str(v))
        return row


class _Digest:
    """Log-template digest: strips variable parts (numbers, uuids, ips,
    quoted/bracketed text) into `{field}_digest` (ref digest.rs presets)."""

    def __init__(self, cfg):
        self.fields = _fields_of(cfg)

    def __call__(self, row):
        for src, _dst in self.fields:
            v = row.get(src)
            if v is None:
                continue
            s = str(v)
            for rx, rep in _DIGEST_PATTERNS:
                s = rx.sub(rep, s)
            row[f"{src}_digest"] = re.sub(r"\s+", " ", s).strip()
        return row


class _Filter:
    """Drop the whole row when the field matches (ref filter.rs)."""

    def __init__(self, cfg):
        self.fields = _fields_of(cfg)
        self.op = cfg.get("match_op", "in")
        self.ci = cfg.get("case_insensitive", True)
        tg = [str(t) for t in _listify(cfg.get("targets"))]
        self.targets = {t.lower() for t in tg} if self.ci else set(tg)

    def __call__(self, row):
        for src, _dst in self.fields:
            v = row.get(src)
            if v is None:
                continue
            s = str(v).lower() if self.ci else str(v)
            hit = s in self.targets
            if (hit and self.op == "in") or (not hit and self.op == "not_in"):
                return None
        return row


class _Select:
    def __init__(self, cfg):
        self.type = cfg.get("type", "include")
        self.keys = [d for _s, d in _fields_of(cfg)]

    def __call__(self, row):
        if self.type == "include":
            return {k: v for k, v in row.items() if k in self.keys}
        return {k: v for k, v in row.items() if k not in self.keys}


class _Cmcd:
    """CMCD (Common Media Client Data, CTA-5004): comma-separated key=value
    pairs, bare keys = boolean true, quoted strings unquoted (reference:
    src/pipeline processor/cmcd.rs)."""

    def __init__(self, cfg):
        self.fields = _fields_of(cfg)

    def __call__(self, row):
        for src, dst in self.fields:
            v = row.get(src)
            if v is None:
                continue
            prefix = (dst or src)
            for part in str(v).split(","):
                part = part.strip()
                if not part:
                    continue
                if "=" in part:
                    k, val = part.split("=", 1)
                    val = val.strip()
                    if val.startswith('"') and val.endswith('"'):
                        out = val[1:-1]
                    else:
                        try:
                            out = int(val)
                        except ValueError:
                            try:
                                out = float(val)
                            except ValueError:
                                out = val
                else:
                    k, out = part, True
                row[f"{prefix}_{k.strip()}"] = out
        return row


class _Vrl:
    """VRL remap program over the row (reference processor/vrl.rs; subset
    interpreter in pipeline/vrl.py, compiled once per pipeline)."""

    def __init__(self, cfg):
        from greptimedb_amd.pipeline.vrl import VrlProgram
        src = cfg.get("source") if isinstance(cfg, dict) else str(cfg)
        self.program = VrlProgram(src or "")

    def __call__(self, row):
        return self.program.run(row)


_PROCESSORS = {
    "dissect": _Dissect, "regex": _Regex, "date": _Date, "epoch": _Epoch,
    "gsub": _Gsub, "letter": _Letter, "csv": _Csv, "json_parse": _JsonParse,
    "json_path": _JsonPath, "simple_extract": _SimpleExtract, "join": _Join,
    "urlencoding": _UrlEncoding, "decolorize": _Decolorize, "digest": _Digest,
    "filter": _Filter, "select": _Select, "cmcd": _Cmcd, "vrl": _Vrl,
}

_NUM_TYPES = {"int8", "int16", "int32", "int64", "uint8", "uint16", "uint32",
              "uint64", "float32", "float64", "boolean"}
_TIME_TYPES = {"time", "timestamp", "epoch"}


class TransformRule:
    def __init__(self, cfg: dict):
        self.fields = _fields_of(cfg)
        self.type = str(cfg.get("type", "string")).split(",")[0].strip()
        self.index = cfg.get("index")
        self.tag = self.index == "tag" or cfg.get("tag") is True
        self.fulltext = self.index == "fulltext"
        self.timestamp = self.index == "timestamp" or self.type.startswith("epoch") \
            or self.type in ("time", "timestamp")
        self.on_failure = cfg.get("on_failure", "ignore")
        self.default = cfg.get("default")

    def convert(self, v):
        if v is None:
            return self.default
        try:
            if self.type in _NUM_TYPES:
                if self.type == "boolean":
                    return bool(v) if not isinstance(v, str) else \
                        v.lower() in ("true", "1", "yes")
                return float(v) if self.type.startswith("float") else int(float(v))
            if self.timestamp:
                return int(v)
            return str(v)
        except (TypeError, ValueError):
            if self.on_failure == "default":
                return self.default
            if self.on_failure == "ignore":
                return None
            raise PipelineError(f"transform: cannot convert {v!r} to {self.type}")


class Pipeline:
    """Compiled pipeline: processor chain + transform rules + routing."""

    def __init__(self, spec: dict, name: str = ""):
        self.name = name
        self.spec = spec
        self.version = int(spec.get("version", 2))
        self.processors = []
        for p in spec.get("processors") or []:
            if not isinstance(p, dict) or len(p) != 1:
                raise PipelineError(f"bad processor entry {p!r}")
            (kind, cfg), = p.items()
            cls = _PROCESSORS.get(kind)
            if cls is None:
                raise PipelineError(f"unknown processor {kind!r}")
            self.processors.append(cls(cfg or {}))
        self.transforms = [TransformRule(t) for t in
                           (spec.get("transform") or spec.get("transforms") or [])]
        # dispatcher: route rows to table suffixes on a field value
        d = spec.get("dispatcher")
        self.dispatch_field = d.get("field") if d else None
        self.dispatch_rules = {str(r["value"]): str(r.get("table_suffix", r["value"]))
                               for r in (d.get("rules", []) if d else [])}
        # table_suffix template: "_${field}" (ref tablesuffix.rs)
        self.table_suffix = spec.get("table_suffix")

    @classmethod
    def from_yaml(cls, text: str, name: str = "") -> "Pipeline":
        try:
            spec = yaml.safe_load(text)
        except yaml.YAMLError as e:
            raise PipelineError(f"pipeline yaml: {e}") from None
        if not isinstance(spec, dict):
            raise PipelineError("pipeline yaml must be a mapping")
        return cls(spec, name)

    # ------------------------------------------------------------ hints

    @property
    def tag_keys(self) -> list[str]:
        out = []
        for t in self.transforms:
            if t.tag:
                out.extend(d for _s, d in t.fields)
        return out

    @property
    def fulltext_keys(self) -> list[str]:
        out = []
        for t in self.transforms:
            if t.fulltext:
                out.extend(d for _s, d in t.fields)
        return out

    @property
    def ts_key(self) -> str:
        for t in self.transforms:
            if t.timestamp:
                for _s, d in t.fields:
                    return d
        return "timestamp"

    # ------------------------------------------------------------- run

    def run_row(self, row: dict):
        """→ (row, table_suffix) or None when filtered out."""
        row = dict(row)
        for p in self.processors:
            row = p(row)
            if row is None:
                return None
        if self.transforms:
            keep = self.version >= 2
            out = dict(row) if keep else {}
            for t in self.transforms:
                for src, dst in t.fields:
                    v = t.convert(row.get(src))
                    if v is not None or not keep:
                        out[dst] = v
                    if src != dst and src in out:
                        del out[src]
            row = out
        suffix = ""
        if self.dispatch_field is not None:
            v = str(row.get(self.dispatch_field))
            s = self.dispatch_rules.get(v)
            if s is not None:
                suffix = "_" + s
        elif self.table_suffix:
            def rep(m):
                return str(row.get(m.group(1), ""))
            suffix = re.sub(r"\$\{(\w+)\}", rep, self.table_suffix)
        return row, suffix

    def run(self, rows: list[dict]):
        """→ {table_suffix: [rows]} (filtered rows dropped)."""
        by_suffix: dict[str, list[dict]] = {}
        for r in rows:
            got = self.run_row(r)
            if got is None:
                continue
            row, suffix = got
            by_suffix.setdefault(suffix, []).append(row)
        return by_suffix


class PipelineStore:
    """Named pipelines persisted under <data_dir>/pipelines/ (reference
    stores them in a greptime_private table; files keep this node-local)."""

    def __init__(self, data_dir: str):
        import os
        self.dir = os.path.join(data_dir, "pipelines")
        os.makedirs(self.dir, exist_ok=True)
        self._cache: dict[str, Pipeline] = {}

    def _path(self, name: str) -> str:
        import os
        if not re.fullmatch(r"[\w.-]+", name):
            raise PipelineError(f"bad pipeline name {name!r}")
        return os.path.join(self.dir, f"{name}.yaml")

    def put(self, name: str, text: str) -> Pipeline:
        p = Pipeline.from_yaml(text, name)   # validate before persisting
        with open(self._path(name), "w") as f:
            f.write(text)
        self._cache[name] = p
        return p

    def get(self, name: str) -> Pipeline:
        p = self._cache.get(name)
        if p is None:
            import os
            path = self._path(name)
            if not os.path.exists(path):
                raise PipelineError(f"pipeline {name!r} not found")
            with open(path) as f:
                p = Pipeline.from_yaml(f.read(), name)
            self._cache[name] = p
        return p

    def delete(self, name: str):
        import os
        self._cache.pop(name, None)
        path = self._path(name)
        if os.path.exists(path):
            os.remove(path)

    def list(self) -> list[str]:
        import os
        return sorted(f[:-5] for f in os.listdir(self.dir) if f.endswith(".yaml"))
