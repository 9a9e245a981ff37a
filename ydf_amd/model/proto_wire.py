"""Schema-driven protobuf wire decoder for export verification.

Decodes bytes against `ydf_schema.json` — a machine-extracted map of the
reference's .proto field numbers/types (tools/extract_proto_schema.py).
This is deliberately INDEPENDENT of `import_ydf.py`'s hand-written
reader: a test that decodes `export_ydf.py` output through this module
checks the wire format against the reference's own schema, so a shared
misreading of the format in importer+exporter cannot pass.

Reference format pinned: model/model_library.cc:92-107 (directory
layout), utils/blob_sequence.h:125-150 (node shards container),
model/decision_tree/decision_tree.proto:202 (Node records).
"""
from __future__ import annotations

import json
import os
import struct
from typing import Any, Dict, List, Tuple

_SCHEMA = None


def schema() -> dict:
    global _SCHEMA
    if _SCHEMA is None:
        p = os.path.join(os.path.dirname(__file__), "ydf_schema.json")
        with open(p) as f:
            _SCHEMA = json.load(f)
    return _SCHEMA


_SCALAR_WIRETYPE = {
    "double": 1, "float": 5, "int32": 0, "int64": 0, "uint32": 0,
    "uint64": 0, "sint32": 0, "sint64": 0, "bool": 0, "fixed64": 1,
    "sfixed64": 1, "fixed32": 5, "sfixed32": 5, "string": 2, "bytes": 2,
}


class WireError(ValueError):
    pass


def _read_varint(data: bytes, pos: int) -> Tuple[int, int]:
    result = 0
    shift = 0
    while True:
        if pos >= len(data):
            raise WireError("truncated varint")
        b = data[pos]
        result |= (b & 0x7F) << shift
        pos += 1
        if not b & 0x80:
            return result, pos
        shift += 7
        if shift > 70:
            raise WireError("varint too long")


def _zigzag(v: int) -> int:
    return (v >> 1) ^ -(v & 1)


def _signed64(v: int) -> int:
    return v - (1 << 64) if v >= (1 << 63) else v


def _signed32(v: int) -> int:
    v &= (1 << 64) - 1
    v &= (1 << 32) - 1 if v < (1 << 32) else (1 << 64) - 1
    if v >= (1 << 63):
        v -= 1 << 64
    elif (1 << 31) <= v < (1 << 32):
        v -= 1 << 32
    return v


def resolve_type(scope: str, tname: str) -> str:
    """Protobuf name resolution: try `scope.tname`, then walk outward."""
    s = schema()
    parts = scope.split(".")
    for i in range(len(parts), -1, -1):
        cand = ".".join(parts[:i] + [tname])
        if cand in s["messages"] or cand in s["enums"]:
            return cand
    # fully-qualified or package-prefixed name
    for key in list(s["messages"]) + list(s["enums"]):
        if key.endswith("." + tname) or key == tname:
            return key
    raise WireError(f"cannot resolve type {tname!r} in scope {scope!r}")


def _convert_scalar(ftype: str, wt: int, raw: Any) -> Any:
    if ftype == "double":
        return struct.unpack("<d", raw)[0]
    if ftype == "float":
        return struct.unpack("<f", raw)[0]
    if ftype in ("fixed64", "sfixed64"):
        v = struct.unpack("<q" if ftype == "sfixed64" else "<Q", raw)[0]
        return v
    if ftype in ("fixed32", "sfixed32"):
        return struct.unpack("<i" if ftype == "sfixed32" else "<I", raw)[0]
    if ftype in ("sint32", "sint64"):
        return _zigzag(raw)
    if ftype in ("int32", "int64"):
        return _signed64(raw)
    if ftype in ("uint32", "uint64"):
        return raw
    if ftype == "bool":
        return bool(raw)
    if ftype == "string":
        return raw.decode("utf-8")
    if ftype == "bytes":
        return raw
    raise WireError(f"unknown scalar type {ftype}")


_PACKED_FMT = {"float": ("<f", 4), "double": ("<d", 8),
               "fixed32": ("<I", 4), "sfixed32": ("<i", 4),
               "fixed64": ("<Q", 8), "sfixed64": ("<q", 8)}


def _decode_packed(ftype: str, raw: bytes) -> List[Any]:
    out = []
    if ftype in _PACKED_FMT:
        fmt, size = _PACKED_FMT[ftype]
        if len(raw) % size:
            raise WireError(f"packed {ftype} length {len(raw)} % {size}")
        for i in range(0, len(raw), size):
            out.append(struct.unpack(fmt, raw[i:i + size])[0])
        return out
    pos = 0
    while pos < len(raw):
        v, pos = _read_varint(raw, pos)
        if ftype in ("sint32", "sint64"):
            v = _zigzag(v)
        elif ftype in ("int32", "int64"):
            v = _signed64(v)
        elif ftype == "bool":
            v = bool(v)
        out.append(v)
    return out


def decode(msg_name: str, data: bytes, strict: bool = True
           ) -> Dict[str, Any]:
    """Decodes `data` as message `msg_name` (full or suffix-unique name).

    Returns {field_name: value} with sub-messages as nested dicts,
    repeated fields as lists and enums as their NAME string. In strict
    mode, raises WireError on unknown field numbers or wire-type
    mismatches — the check that catches wrong-field-number bugs.
    """
    s = schema()
    if msg_name not in s["messages"]:
        msg_name = resolve_type("", msg_name)
    fields = s["messages"][msg_name]
    out: Dict[str, Any] = {}
    pos = 0
    while pos < len(data):
        key, pos = _read_varint(data, pos)
        fnum, wt = key >> 3, key & 7
        fdesc = fields.get(str(fnum))
        if fdesc is None:
            if strict:
                raise WireError(
                    f"{msg_name}: unknown field number {fnum} "
                    f"(wire type {wt}) at byte {pos}")
            # skip
            if wt == 0:
                _, pos = _read_varint(data, pos)
            elif wt == 1:
                pos += 8
            elif wt == 2:
                ln, pos = _read_varint(data, pos)
                pos += ln
            elif wt == 5:
                pos += 4
            else:
                raise WireError(f"unsupported wire type {wt}")
            continue
        ftype = fdesc["type"]
        fname = fdesc["name"]
        repeated = fdesc["label"] == "repeated"
        is_scalar = ftype in _SCALAR_WIRETYPE
        if not is_scalar:
            full = resolve_type(msg_name, ftype)
            is_enum = full in s["enums"]
        else:
            is_enum = False
            full = None

        if wt == 2 and (is_scalar and ftype not in ("string", "bytes")
                        or is_enum) and repeated:
            ln, pos = _read_varint(data, pos)
            raw = data[pos:pos + ln]
            pos += ln
            vals = _decode_packed("int64" if is_enum else ftype, raw)
            if is_enum:
                vals = [s["enums"][full].get(str(v), v) for v in vals]
            out.setdefault(fname, []).extend(vals)
            continue

        # expected wire type check
        if is_scalar:
            exp_wt = _SCALAR_WIRETYPE[ftype]
        elif is_enum:
            exp_wt = 0
        else:
            exp_wt = 2
        if wt != exp_wt:
            raise WireError(
                f"{msg_name}.{fname} (#{fnum}): wire type {wt}, "
                f"schema says {exp_wt} ({ftype})")

        if wt == 0:
            raw, pos = _read_varint(data, pos)
        elif wt == 1:
            raw = data[pos:pos + 8]
            pos += 8
            if len(raw) != 8:
                raise WireError("truncated fixed64")
        elif wt == 5:
            raw = data[pos:pos + 4]
            pos += 4
            if len(raw) != 4:
                raise WireError("truncated fixed32")
        else:  # wt == 2
            ln, pos = _read_varint(data, pos)
            raw = data[pos:pos + ln]
            pos += ln
            if len(raw) != ln:
                raise WireError("truncated length-delimited field")

        if is_enum:
            val = s["enums"][full].get(str(raw), raw)
        elif is_scalar:
            val = _convert_scalar(ftype, wt, raw)
        else:
            val = decode(full, raw, strict=strict)

        if repeated:
            out.setdefault(fname, []).append(val)
        else:
            if fname in out and strict and not isinstance(val, dict):
                # last-one-wins is legal protobuf; keep it but don't flag
                pass
            out[fname] = val
    return out


def read_blob_sequence(path: str) -> List[bytes]:
    """Reads a reference blob-sequence file: 2-byte magic "BS", version
    u16, reserved, then per-record u32 length + payload
    (utils/blob_sequence.h:125-150)."""
    with open(path, "rb") as f:
        data = f.read()
    if data[:2] != b"BS":
        raise WireError("bad blob-sequence magic")
    version, compression = struct.unpack("<HB", data[2:5])
    if version > 1:
        raise WireError(f"unsupported blob-sequence version {version}")
    if compression == 1:
        import zlib
        data = data[:8] + zlib.decompress(data[8:], wbits=31)
    elif compression:
        raise WireError(f"unknown compression byte {compression}")
    pos = 8
    records = []
    while pos < len(data):
        if pos + 4 > len(data):
            raise WireError("truncated record header")
        ln = struct.unpack("<I", data[pos:pos + 4])[0]
        pos += 4
        rec = data[pos:pos + ln]
        if len(rec) != ln:
            raise WireError("truncated record payload")
        records.append(rec)
        pos += ln
    return records
