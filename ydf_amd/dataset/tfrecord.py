"""TFRecord (+ tf.Example) reader/writer without TensorFlow.

Capability analogue of the reference's tensorflow_no_dep example reader
(dataset/tensorflow_no_dep/tf_record.cc): TFRecord framing is
  u64-LE length | u32 masked-crc32c(length) | payload | u32 masked-crc32c
and each payload is a tf.Example protobuf:
  Example{features=1 Features{feature=1 map<string, Feature>}}
  Feature{bytes_list=1{value=1}, float_list=2{value=1 packed},
          int64_list=3{value=1 packed}}.
CRCs are verified on read; gzip containers supported ("tfrecord+gzip" /
auto-detected 0x1f8b magic).
"""
from __future__ import annotations

import gzip
import struct
from typing import Dict, Iterator, List

import numpy as np

# ---------------------------------------------------------------------------
# crc32c (Castagnoli), table-driven — needed for the masked record CRCs
# ---------------------------------------------------------------------------
_CRC_TABLE = None


def _crc_table():
    global _CRC_TABLE
    if _CRC_TABLE is None:
        poly = 0x82F63B78
        table = []
        for i in range(256):
            c = i
            for _ in range(8):
                c = (c >> 1) ^ poly if c & 1 else c >> 1
            table.append(c)
        _CRC_TABLE = table
    return _CRC_TABLE


def crc32c(data: bytes) -> int:
    table = _crc_table()
    crc = 0xFFFFFFFF
    for b in data:
        crc = table[(crc ^ b) & 0xFF] ^ (crc >> 8)
    return crc ^ 0xFFFFFFFF


def _masked_crc(data: bytes) -> int:
    crc = crc32c(data)
    return ((crc >> 15 | crc << 17) + 0xA282EAD8) & 0xFFFFFFFF


# ---------------------------------------------------------------------------
# record framing
# ---------------------------------------------------------------------------
def read_tfrecords(path: str, verify_crc: bool = True) -> Iterator[bytes]:
    with open(path, "rb") as f:
        data = f.read()
    if data[:2] == b"\x1f\x8b":
        data = gzip.decompress(data)
    p = 0
    n = len(data)
    while p + 12 <= n:
        (length,) = struct.unpack_from("<Q", data, p)
        (len_crc,) = struct.unpack_from("<I", data, p + 8)
        if verify_crc and _masked_crc(data[p:p + 8]) != len_crc:
            raise ValueError(f"{path}: bad length crc at offset {p}")
        p += 12
        payload = data[p:p + length]
        (data_crc,) = struct.unpack_from("<I", data, p + length)
        if verify_crc and _masked_crc(payload) != data_crc:
            raise ValueError(f"{path}: bad data crc at offset {p}")
        p += length + 4
        yield payload


def write_tfrecords(path: str, records: List[bytes],
                    compress: bool = False) -> None:
    out = bytearray()
    for r in records:
        hdr = struct.pack("<Q", len(r))
        out += hdr + struct.pack("<I", _masked_crc(hdr))
        out += r + struct.pack("<I", _masked_crc(r))
    blob = gzip.compress(bytes(out)) if compress else bytes(out)
    with open(path, "wb") as f:
        f.write(blob)


# ---------------------------------------------------------------------------
# tf.Example proto
# ---------------------------------------------------------------------------
def parse_example(raw: bytes) -> Dict[str, object]:
    """One tf.Example -> {name: bytes-list | float-list | int-list}."""
    from ydf_amd.model.import_ydf import (Wire, _msg, _packed_floats,
                                          _packed_varints)

    out: Dict[str, object] = {}
    ex = _msg(raw)
    if 1 not in ex:
        return out
    feats = _msg(ex[1][0])
    for entry in feats.get(1, []):
        e = _msg(entry)
        name = e.get(1, [b""])[0].decode()
        feature = _msg(e.get(2, [b""])[0])
        if 1 in feature:  # bytes_list
            bl = _msg(feature[1][0])
            out[name] = [v.decode("utf-8", "replace")
                         for v in bl.get(1, [])]
        elif 2 in feature:  # float_list (packed or repeated)
            fl = _msg(feature[2][0])
            vals: List[float] = []
            for v in fl.get(1, []):
                if isinstance(v, bytes):
                    vals.extend(_packed_floats(v))
                else:
                    vals.append(float(v))
            out[name] = vals
        elif 3 in feature:  # int64_list
            il = _msg(feature[3][0])
            vals = []
            for v in il.get(1, []):
                if isinstance(v, bytes):
                    vals.extend(_packed_varints(v))
                else:
                    vals.append(int(v))
            out[name] = [_signed(v) for v in vals]
    return out


def _signed(v: int) -> int:
    return v - (1 << 64) if v >= (1 << 63) else v


def encode_example(row: Dict[str, object]) -> bytes:
    from ydf_amd.model.export_ydf import (_varint, f_bytes, f_float, f_msg,
                                          f_varint)

    entries = b""
    for name, value in row.items():
        if isinstance(value, (list, tuple, np.ndarray)):
            vals = list(value)
        else:
            vals = [value]
        if len(vals) and isinstance(vals[0], (str, bytes)):
            inner = b"".join(
                f_bytes(1, v.encode() if isinstance(v, str) else v)
                for v in vals)
            feature = f_msg(1, inner)
        elif len(vals) and (isinstance(vals[0], (int, np.integer))):
            packed = b"".join(_varint(int(v) & ((1 << 64) - 1))
                              for v in vals)
            feature = f_msg(3, f_bytes(1, packed))
        else:
            inner = b"".join(f_float(1, float(v)) for v in vals)
            feature = f_msg(2, inner)
        entry = f_bytes(1, name.encode()) + f_msg(2, feature)
        entries += f_msg(1, entry)
    return f_msg(1, entries)


# ---------------------------------------------------------------------------
# column-dict bridge
# ---------------------------------------------------------------------------
def read_tfrecord_columns(paths: List[str]) -> Dict[str, np.ndarray]:
    """Reads tf.Example shards into a column dict (scalar features; the
    first value of each list is taken, missing -> NaN/empty)."""
    rows: List[Dict[str, object]] = []
    names: Dict[str, str] = {}  # name -> kind: s(tring) | i(nt) | f(loat)
    for path in paths:
        for rec in read_tfrecords(path):
            row = parse_example(rec)
            rows.append(row)
            for k, v in row.items():
                if k not in names and len(v):
                    names[k] = ("s" if isinstance(v[0], str)
                                else "i" if isinstance(v[0], int)
                                else "f")
    cols: Dict[str, np.ndarray] = {}
    for name, kind in names.items():
        if kind == "s":
            cols[name] = np.array(
                [(row.get(name) or [""])[0] for row in rows], dtype=object)
        elif kind == "i":
            # integers keep their dtype: categorical-integer vocabularies
            # match on the "13" (not "13.0") string form
            vals = [row.get(name) for row in rows]
            if any(v is None or not v for v in vals):
                cols[name] = np.array(
                    [float(v[0]) if v else np.nan for v in vals],
                    dtype=np.float32)
            else:
                cols[name] = np.array([int(v[0]) for v in vals],
                                      dtype=np.int64)
        else:
            cols[name] = np.array(
                [float((row.get(name) or [np.nan])[0]) for row in rows],
                dtype=np.float32)
    return cols


def write_tfrecord_columns(path: str, cols: Dict[str, np.ndarray],
                           compress: bool = False) -> None:
    n = len(next(iter(cols.values())))
    records = []
    for i in range(n):
        row = {}
        for k, v in cols.items():
            x = v[i]
            if isinstance(x, (str, np.str_)):
                row[k] = [str(x)]
            elif np.issubdtype(np.asarray(x).dtype, np.integer):
                row[k] = [int(x)]
            else:
                row[k] = [float(x)]
        records.append(encode_example(row))
    write_tfrecords(path, records, compress=compress)
