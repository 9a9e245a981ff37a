"""Avro object-container-file reader/writer (no external deps).

Capability analogue of the reference's Avro example reader
(dataset/avro_example.cc / utils/avro.cc): container format is
  magic "Obj\\x01" | file-metadata map (avro.schema JSON, avro.codec)
  | 16-byte sync marker | blocks of {count, byte-size, data, sync}.
Supported schema types: null, boolean, int, long, float, double, string,
bytes, and ["null", T] unions; codecs: null and deflate.
"""
from __future__ import annotations

import json
import struct
import zlib
from typing import Dict, List, Tuple

import numpy as np


# --- zigzag varints --------------------------------------------------------
def _read_long(buf: bytes, p: int) -> Tuple[int, int]:
    r = 0
    s = 0
    while True:
        b = buf[p]
        p += 1
        r |= (b & 0x7F) << s
        if not b & 0x80:
            break
        s += 7
    return (r >> 1) ^ -(r & 1), p


def _write_long(v: int) -> bytes:
    v = (v << 1) ^ (v >> 63)
    out = bytearray()
    v &= (1 << 64) - 1
    while True:
        b = v & 0x7F
        v >>= 7
        if v:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def _read_bytes(buf: bytes, p: int) -> Tuple[bytes, int]:
    n, p = _read_long(buf, p)
    return buf[p:p + n], p + n


# --- schema-driven value codecs -------------------------------------------
def _read_value(buf: bytes, p: int, schema):
    if isinstance(schema, dict):
        schema = schema.get("type", schema)
    if isinstance(schema, list):  # union
        idx, p = _read_long(buf, p)
        return _read_value(buf, p, schema[idx])
    if schema == "null":
        return None, p
    if schema == "boolean":
        return bool(buf[p]), p + 1
    if schema in ("int", "long"):
        return _read_long(buf, p)
    if schema == "float":
        return struct.unpack_from("<f", buf, p)[0], p + 4
    if schema == "double":
        return struct.unpack_from("<d", buf, p)[0], p + 8
    if schema in ("string", "bytes"):
        raw, p = _read_bytes(buf, p)
        return (raw.decode() if schema == "string" else raw), p
    raise NotImplementedError(f"avro type {schema!r}")


def _write_value(v, schema) -> bytes:
    if isinstance(schema, dict):
        schema = schema.get("type", schema)
    if isinstance(schema, list):
        if v is None:
            return _write_long(schema.index("null"))
        idx = next(i for i, s in enumerate(schema) if s != "null")
        return _write_long(idx) + _write_value(v, schema[idx])
    if schema == "null":
        return b""
    if schema == "boolean":
        return bytes([1 if v else 0])
    if schema in ("int", "long"):
        return _write_long(int(v))
    if schema == "float":
        return struct.pack("<f", float(v))
    if schema == "double":
        return struct.pack("<d", float(v))
    if schema == "string":
        raw = str(v).encode()
        return _write_long(len(raw)) + raw
    if schema == "bytes":
        return _write_long(len(v)) + v
    raise NotImplementedError(f"avro type {schema!r}")


# --- container file --------------------------------------------------------
def read_avro(path: str) -> Tuple[dict, List[dict]]:
    """Returns (schema, records)."""
    with open(path, "rb") as f:
        buf = f.read()
    if buf[:4] != b"Obj\x01":
        raise ValueError(f"{path}: not an Avro container file")
    p = 4
    meta = {}
    while True:
        count, p = _read_long(buf, p)
        if count == 0:
            break
        if count < 0:  # size-prefixed block form
            _, p = _read_long(buf, p)
            count = -count
        for _ in range(count):
            k, p = _read_bytes(buf, p)
            v, p = _read_bytes(buf, p)
            meta[k.decode()] = v
    sync = buf[p:p + 16]
    p += 16
    schema = json.loads(meta["avro.schema"].decode())
    codec = meta.get("avro.codec", b"null").decode()
    fields = schema["fields"]
    records: List[dict] = []
    n = len(buf)
    while p < n:
        count, p = _read_long(buf, p)
        size, p = _read_long(buf, p)
        block = buf[p:p + size]
        p += size
        if buf[p:p + 16] != sync:
            raise ValueError(f"{path}: sync marker mismatch")
        p += 16
        if codec == "deflate":
            block = zlib.decompress(block, -15)
        elif codec != "null":
            raise NotImplementedError(f"avro codec {codec!r}")
        q = 0
        for _ in range(count):
            row = {}
            for fld in fields:
                row[fld["name"]], q = _read_value(block, q, fld["type"])
            records.append(row)
    return schema, records


def write_avro(path: str, schema: dict, records: List[dict],
               codec: str = "null") -> None:
    body = bytearray()
    for row in records:
        for fld in schema["fields"]:
            body += _write_value(row.get(fld["name"]), fld["type"])
    block = bytes(body)
    if codec == "deflate":
        co = zlib.compressobj(9, zlib.DEFLATED, -15)
        block = co.compress(block) + co.flush()
    sync = b"\x00\x01\x02\x03\x04\x05\x06\x07" * 2
    meta = {"avro.schema": json.dumps(schema).encode(),
            "avro.codec": codec.encode()}
    out = bytearray(b"Obj\x01")
    out += _write_long(len(meta))
    for k, v in meta.items():
        kk = k.encode()
        out += _write_long(len(kk)) + kk
        out += _write_long(len(v)) + v
    out += _write_long(0)
    out += sync
    out += _write_long(len(records))
    out += _write_long(len(block))
    out += block
    out += sync
    with open(path, "wb") as f:
        f.write(bytes(out))


def read_avro_columns(paths: List[str]) -> Dict[str, np.ndarray]:
    """Reads Avro shards into a column dict (null -> NaN / "")."""
    all_records: List[dict] = []
    schema = None
    for path in paths:
        schema, recs = read_avro(path)
        all_records.extend(recs)
    if schema is None or not all_records:
        return {}
    cols: Dict[str, np.ndarray] = {}
    for fld in schema["fields"]:
        name = fld["name"]
        t = fld["type"]
        if isinstance(t, list):
            t = next((s for s in t if s != "null"), "null")
        if isinstance(t, dict):
            t = t.get("type", "string")
        vals = [r.get(name) for r in all_records]
        if t in ("string", "bytes"):
            cols[name] = np.array(
                ["" if v is None else str(v) for v in vals], dtype=object)
        elif t == "boolean":
            cols[name] = np.array(
                [float("nan") if v is None else float(v) for v in vals],
                dtype=np.float32)
        else:
            cols[name] = np.array(
                [float("nan") if v is None else float(v) for v in vals],
                dtype=np.float32)
    return cols
