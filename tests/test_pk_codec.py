"""Memcomparable pk codec properties (reference: mito-codec row_converter)."""

import random

from greptimedb_amd.engine import pk_codec


def test_roundtrip():
    cases = [
        ("a", "b", "c"),
        ("", "x", None),
        (None, None, None),
        ("exactly8", "nine char", "a" * 17),
        ("host_0", "us-east-1", "prod"),
    ]
    for tags in cases:
        enc = pk_codec.encode_pk(tags)
        assert pk_codec.decode_pk(enc, len(tags)) == tags


def test_order_preserving():
    rng = random.Random(3)
    vals = []
    for _ in range(300):
        n = rng.randint(0, 12)
        vals.append("".join(rng.choice("abcxyz01") for _ in range(n)))
    vals = sorted(set(vals))
    encoded = [pk_codec.encode_string(v.encode()) for v in vals]
    assert encoded == sorted(encoded), "byte order must match string order"


def test_null_sorts_first():
    a = pk_codec.encode_pk((None,))
    b = pk_codec.encode_pk(("",))
    c = pk_codec.encode_pk(("a",))
    assert a < b < c


def test_tuple_order():
    tuples = [("a", "b"), ("a", "c"), ("ab", "a"), ("b", None), ("b", "a")]
    enc = [pk_codec.encode_pk(t) for t in tuples]
    order = sorted(range(len(enc)), key=lambda i: enc[i])
    keyed = sorted(range(len(tuples)),
                   key=lambda i: tuple((v is not None, v or "") for v in tuples[i]))
    assert order == keyed


def test_tsid_stable():
    from greptimedb_amd.engine.series import tsid_hash
    pk = pk_codec.encode_pk(("h1", "r1"))
    assert tsid_hash(pk) == tsid_hash(pk)
    assert tsid_hash(pk) != tsid_hash(pk_codec.encode_pk(("h1", "r2")))
