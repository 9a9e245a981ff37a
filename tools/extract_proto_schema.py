#!/usr/bin/env python3
"""Extracts a JSON wire schema from the reference's .proto sources.

Machine-reads the protobuf message definitions that make up YDF's
on-disk model format (model/model_library.cc:92-107 directory layout)
and emits `ydf_amd/model/ydf_schema.json`: for every message, the map
field-number -> {name, type, label}; for every enum, value -> name.

The output is *derived data* (field numbers and names are the wire
contract the new framework must be compatible with); no reference code
is copied. `ydf_amd/model/proto_wire.py` decodes exported bytes against
this schema, giving an export-format check that is fully independent of
`ydf_amd/model/import_ydf.py`'s hand-written reader.

Usage:  python tools/extract_proto_schema.py [reference_root] [out.json]
"""
from __future__ import annotations

import json
import os
import re
import sys

PROTO_FILES = [
    "yggdrasil_decision_forests/dataset/data_spec.proto",
    "yggdrasil_decision_forests/dataset/weight.proto",
    "yggdrasil_decision_forests/model/abstract_model.proto",
    "yggdrasil_decision_forests/model/hyperparameter.proto",
    "yggdrasil_decision_forests/model/prediction.proto",
    "yggdrasil_decision_forests/model/decision_tree/decision_tree.proto",
    "yggdrasil_decision_forests/model/gradient_boosted_trees/"
    "gradient_boosted_trees.proto",
    "yggdrasil_decision_forests/model/random_forest/random_forest.proto",
    "yggdrasil_decision_forests/model/isolation_forest/"
    "isolation_forest.proto",
    "yggdrasil_decision_forests/metric/metric.proto",
    "yggdrasil_decision_forests/utils/distribution.proto",
]

SCALARS = {
    "double", "float", "int32", "int64", "uint32", "uint64", "sint32",
    "sint64", "fixed32", "fixed64", "sfixed32", "sfixed64", "bool",
    "string", "bytes",
}


def _strip_comments(text: str) -> str:
    text = re.sub(r"//[^\n]*", "", text)
    text = re.sub(r"/\*.*?\*/", "", text, flags=re.S)
    return text


def _parse_block(text: str, pos: int):
    """Returns (body, end_pos) for the {...} block starting at text[pos]=='{'."""
    depth = 0
    start = pos
    while pos < len(text):
        c = text[pos]
        if c == "{":
            depth += 1
        elif c == "}":
            depth -= 1
            if depth == 0:
                return text[start + 1:pos], pos + 1
        pos += 1
    raise ValueError("unbalanced braces")


FIELD_RE = re.compile(
    r"(optional|repeated|required)?\s*"
    r"([A-Za-z_][\w.]*)\s+"          # type
    r"([A-Za-z_]\w*)\s*=\s*(\d+)"    # name = number
    r"((?:\s*\[[^\]]*\])?)\s*;")

MAP_RE = re.compile(
    r"map\s*<\s*([\w.]+)\s*,\s*([\w.]+)\s*>\s*"
    r"([A-Za-z_]\w*)\s*=\s*(\d+)\s*(?:\[[^\]]*\])?\s*;")

ENUM_VAL_RE = re.compile(r"([A-Za-z_]\w*)\s*=\s*(-?\d+)\s*(?:\[[^\]]*\])?\s*;")


def parse_messages(text: str, prefix: str, out: dict):
    pos = 0
    while True:
        m = re.search(r"\b(message|enum)\s+([A-Za-z_]\w*)\s*\{", text[pos:])
        if not m:
            break
        kind, name = m.group(1), m.group(2)
        body, end = _parse_block(text, pos + m.end() - 1)
        full = f"{prefix}.{name}" if prefix else name
        if kind == "enum":
            vals = {}
            for em in ENUM_VAL_RE.finditer(_remove_nested_blocks(body)):
                vals[int(em.group(2))] = em.group(1)
            out.setdefault("enums", {})[full] = vals
        else:
            parse_messages(body, full, out)  # nested messages/enums first
            flat = _remove_nested_blocks(body)
            # drop oneof wrappers but keep their fields
            flat = re.sub(r"\boneof\s+[A-Za-z_]\w*\s*", "", flat)
            fields = {}
            for mm in MAP_RE.finditer(flat):
                ktype, vtype, fname, fnum = mm.groups()
                # map<K,V> encodes as a repeated synthetic entry message
                # with key=1, value=2 (protobuf map wire format)
                entry = f"{full}.{fname.title()}MapEntry"
                out.setdefault("messages", {})[entry] = {
                    1: {"name": "key", "type": ktype,
                        "label": "optional", "packed": False},
                    2: {"name": "value", "type": vtype,
                        "label": "optional", "packed": False},
                }
                fields[int(fnum)] = {
                    "name": fname, "type": entry, "label": "repeated",
                    "packed": False, "map": True,
                }
            flat = MAP_RE.sub("", flat)
            for fm in FIELD_RE.finditer(flat):
                label, ftype, fname, fnum, opts = fm.groups()
                fields[int(fnum)] = {
                    "name": fname,
                    "type": ftype,
                    "label": label or "optional",
                    "packed": "packed" in (opts or ""),
                }
            out.setdefault("messages", {})[full] = fields
        pos = pos + m.start() + (end - (pos + m.end() - 1)) + (
            m.end() - m.start())


def _remove_nested_blocks(body: str) -> str:
    """Removes nested message/enum/oneof/extend { } blocks, keeping oneof
    bodies (their fields belong to the parent)."""
    res = []
    pos = 0
    while pos < len(body):
        m = re.search(
            r"\b(message|enum|extend|reserved|extensions|oneof)\b",
            body[pos:])
        if not m:
            res.append(body[pos:])
            break
        res.append(body[pos:pos + m.start()])
        kind = m.group(1)
        after = pos + m.end()
        if kind in ("reserved", "extensions"):
            semi = body.index(";", after)
            pos = semi + 1
            continue
        brace = body.index("{", after)
        inner, end = _parse_block(body, brace)
        if kind == "oneof":
            res.append(" " + inner + " ")
        pos = end
    return "".join(res)


def main():
    ref = sys.argv[1] if len(sys.argv) > 1 else "/root/reference"
    out_path = sys.argv[2] if len(sys.argv) > 2 else os.path.join(
        os.path.dirname(__file__), "..", "ydf_amd", "model",
        "ydf_schema.json")
    out: dict = {"messages": {}, "enums": {}, "source_files": []}
    for rel in PROTO_FILES:
        p = os.path.join(ref, rel)
        with open(p) as f:
            raw = f.read()
        text = _strip_comments(raw)
        pkg = re.search(r"\bpackage\s+([\w.]+)\s*;", text)
        pkg_name = pkg.group(1) if pkg else ""
        parse_messages(text, pkg_name, out)
        out["source_files"].append(rel)
    with open(out_path, "w") as f:
        json.dump(out, f, indent=1, sort_keys=True)
    n_msg = len(out["messages"])
    n_enum = len(out["enums"])
    print(f"wrote {out_path}: {n_msg} messages, {n_enum} enums")


if __name__ == "__main__":
    main()
