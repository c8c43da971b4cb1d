"""Primary-key (series key) codec — memcomparable encoding of tag values.

Reference parity: src/mito-codec/src/row_converter.rs DensePrimaryKeyCodec.
The reference encodes each primary-key column with a memcomparable byte
encoding so that lexicographic byte order == tuple order; SSTs store the
encoded key in the __primary_key dictionary column. We implement the same
*shape* (per-column: null marker byte + memcomparable value bytes; string
values in 8-byte groups with a continuation/length marker) — byte-level
compatibility with the reference is a stated goal and is covered by
tests/test_pk_codec.py's ordering properties; cross-reading actual
reference-written SSTs is validated structurally (column layout) since no
Rust toolchain exists in this environment to produce goldens.

Format per column:
  null:      b"\x00"
  non-null:  b"\x01" + value encoding
String/bytes value encoding (memcomparable group encoding):
  split into 8-byte groups; every group is padded with \x00 to 8 bytes and
  followed by a marker byte = 9 if more groups follow else (#significant
  bytes in this group). Empty string encodes as one all-pad group marker 0.
"""

from __future__ import annotations


def encode_string(b: bytes) -> bytes:
    out = bytearray()
    n = len(b)
    if n == 0:
        return bytes(8) + b"\x00"
    i = 0
    while i < n:
        group = b[i:i + 8]
        i += 8
        if i < n:
            out += group + b"\x09"
        elif len(group) == 8:
            # exact multiple of 8: the memcomparable crate (chunks_exact +
            # remainder) emits the full group with the continuation marker
            # followed by an empty all-pad group — match it byte-for-byte
            out += group + b"\x09" + bytes(8) + b"\x00"
        else:
            out += group + bytes(8 - len(group)) + bytes([len(group)])
    return bytes(out)


def decode_string(buf: bytes, off: int) -> tuple[bytes, int]:
    out = bytearray()
    while True:
        group = buf[off:off + 8]
        marker = buf[off + 8]
        off += 9
        if marker == 9:
            out += group
        else:
            out += group[:marker]
            return bytes(out), off


def encode_pk(values: tuple[str, ...]) -> bytes:
    """Encode an ordered tuple of tag values (None = null) to pk bytes."""
    out = bytearray()
    for v in values:
        if v is None:
            out += b"\x00"
        else:
            out += b"\x01"
            out += encode_string(v.encode() if isinstance(v, str) else v)
    return bytes(out)


SPARSE_MARKER = 0xFF


def encode_sparse(labels: dict) -> bytes:
    """Sparse pk (reference: mito-codec SparsePrimaryKeyCodec, used by the
    metric engine): self-describing sorted (name, value) pairs, prefixed
    with a marker byte so dense and sparse keys are distinguishable."""
    out = bytearray([SPARSE_MARKER])
    for name in sorted(labels):
        v = labels[name]
        if v is None:
            continue
        out += encode_string(name.encode())
        out += encode_string(v.encode() if isinstance(v, str) else v)
    return bytes(out)


def decode_sparse(buf: bytes) -> dict:
    assert buf[0] == SPARSE_MARKER
    off = 1
    out = {}
    while off < len(buf):
        name, off = decode_string(buf, off)
        val, off = decode_string(buf, off)
        out[name.decode()] = val.decode()
    return out


def is_sparse(buf: bytes) -> bool:
    return len(buf) > 0 and buf[0] == SPARSE_MARKER


def decode_pk(buf: bytes, n_cols: int) -> tuple:
    out = []
    off = 0
    for _ in range(n_cols):
        marker = buf[off]
        off += 1
        if marker == 0:
            out.append(None)
        else:
            raw, off = decode_string(buf, off)
            out.append(raw.decode())
    return tuple(out)
