"""Model -> dependency-free C++ / Java source code generation.

Capability analogue of the reference's embed codegen (serving/embed/:
cpp/cpp_embed.cc and java/java_embed.cc; if-else routing variant). The
emitted C++ translation unit has no dependencies beyond
<cstdint>/<cmath> and exposes
  float <name>_predict(const float* features);
(binary classification returns the positive-class probability;
regression returns the value; multi-class models emit
  void <name>_predict_multi(const float* features, float* out);).
to_java emits an equivalent standalone Java class with
  public static float predict(float[] f).
"""
from __future__ import annotations

from typing import List

import numpy as np



def _emit_node(f, forest, n: int, indent: str, lines: List[str]):
    if forest.feat[n] < 0:
        lines.append(f"{indent}acc += {float(forest.thr[n])!r}f;")
        return
    fi = int(forest.feat[n])
    left = int(forest.left[n])
    ci = int(forest.cat_idx[n])
    if ci <= -2:  # oblique sparse projection
        oi = -(ci + 2)
        s0 = int(forest.obl_ranges[oi, 0])
        nn = int(forest.obl_ranges[oi, 1])
        dot = " + ".join(
            f"{float(forest.obl_w[s0 + q])!r}f * f{int(forest.obl_attr[s0 + q])}"
            for q in range(nn))
        lines.append(
            f"{indent}if ({dot} > {float(forest.thr[n])!r}f) {{")
    elif ci >= 0:
        words = ", ".join(f"0x{int(w):016x}ull"
                          for w in np.asarray(forest.masks[ci],
                                              dtype=np.uint64))
        lines.append(
            f"{indent}{{ static const uint64_t m[4] = {{{words}}};")
        lines.append(
            f"{indent}  const int c = f{fi} < 0 ? 0 : (f{fi} > 255 ? 255 : "
            f"(int)f{fi});")
        lines.append(f"{indent}  if ((m[c >> 6] >> (c & 63)) & 1ull) {{")
    else:
        if forest.has_na_routing:
            na = "true" if forest.na_right[n] else "false"
            lines.append(
                f"{indent}if (std::isnan(f{fi}) ? {na} : "
                f"(f{fi} > {float(forest.thr[n])!r}f)) {{")
        else:
            lines.append(
                f"{indent}if (f{fi} > {float(forest.thr[n])!r}f) {{")
    _emit_node(f, forest, left + 1, indent + "  ", lines)
    lines.append(f"{indent}}} else {{")
    _emit_node(f, forest, left, indent + "  ", lines)
    lines.append(f"{indent}}}" + ("}" if ci >= 0 else ""))


def _check_embeddable(forest):
    if forest.has_set_conditions:
        raise NotImplementedError(
            "embed codegen does not support categorical-set / "
            "large-vocab set conditions yet")


def _routing_tables(model, prefix: str) -> str:
    """Emits the model as node TABLES + a fixed traversal loop
    (reference serving/embed Algorithm=ROUTING, the default:
    cpp_target_lowering.cc). O(nodes) data instead of O(nodes) code, so
    1000-tree forests compile fast and small."""
    f = model.forest
    n = f.n_nodes
    feat = ", ".join(str(int(v)) for v in f.feat)
    left = ", ".join(str(int(v)) for v in f.left)
    thr = ", ".join(f"{float(v)!r}f" for v in f.thr)
    cat = ", ".join(str(int(v)) for v in f.cat_idx)
    roots = ", ".join(str(int(v)) for v in f.roots)
    out = [
        f"static const int32_t {prefix}_feat[{n}] = {{{feat}}};",
        f"static const float {prefix}_thr[{n}] = {{{thr}}};",
        f"static const uint32_t {prefix}_left[{n}] = {{{left}}};",
        f"static const int32_t {prefix}_cat[{n}] = {{{cat}}};",
        f"static const uint32_t {prefix}_roots[{f.n_trees}] = {{{roots}}};",
    ]
    if len(f.masks):
        words = ", ".join(
            f"0x{int(w):016x}ull"
            for w in np.asarray(f.masks, dtype=np.uint64).reshape(-1))
        out.append(f"static const uint64_t {prefix}_masks"
                   f"[{len(f.masks) * 4}] = {{{words}}};")
    if f.has_na_routing:
        bits = np.zeros((n + 63) // 64, dtype=np.uint64)
        for i in range(n):
            if f.na_right[i]:
                bits[i >> 6] |= np.uint64(1 << (i & 63))
        words = ", ".join(f"0x{int(w):016x}ull" for w in bits)
        out.append(f"static const uint64_t {prefix}_na[{len(bits)}] = "
                   f"{{{words}}};")
    if len(f.obl_ranges):
        rng = ", ".join(f"{int(a)}, {int(b)}" for a, b in f.obl_ranges)
        attr = ", ".join(str(int(v)) for v in f.obl_attr)
        w = ", ".join(f"{float(v)!r}f" for v in f.obl_w)
        out += [
            f"static const int32_t {prefix}_orng"
            f"[{2 * len(f.obl_ranges)}] = {{{rng}}};",
            f"static const int32_t {prefix}_oattr"
            f"[{max(1, len(f.obl_attr))}] = {{{attr or '0'}}};",
            f"static const float {prefix}_ow"
            f"[{max(1, len(f.obl_w))}] = {{{w or '0.f'}}};",
        ]
    return "\n".join(out)


def _routing_walk(model, prefix: str, acc_expr: str) -> str:
    """The traversal loop over the node tables."""
    f = model.forest
    scale = model._leaf_scale()
    body = [
        f"  for (uint32_t t = 0; t < {f.n_trees}u; ++t) {{",
        f"    uint32_t n = {prefix}_roots[t];",
        "    int32_t fi;",
        f"    while ((fi = {prefix}_feat[n]) >= 0) {{",
        f"      const int32_t c = {prefix}_cat[n];",
        "      bool right;",
    ]
    if len(f.masks):
        body += [
            "      if (c >= 0) {",
            "        const float x = f[fi];",
            "        const int v = x < 0 ? 0 : (x > 255 ? 255 : (int)x);",
            f"        right = ({prefix}_masks[4 * c + (v >> 6)] >> "
            "(v & 63)) & 1ull;",
            "      } else",
        ]
    if len(f.obl_ranges):
        body += [
            "      if (c <= -2) {",
            "        const int32_t r = -(c + 2);",
            f"        const int32_t s0 = {prefix}_orng[2 * r];",
            f"        const int32_t nn = {prefix}_orng[2 * r + 1];",
            "        float dot = 0.f;",
            "        for (int32_t q = 0; q < nn; ++q)",
            f"          dot += {prefix}_ow[s0 + q] * "
            f"f[{prefix}_oattr[s0 + q]];",
            f"        right = dot > {prefix}_thr[n];",
            "      } else",
        ]
    if f.has_na_routing:
        body += [
            "      if (std::isnan(f[fi])) {",
            f"        right = ({prefix}_na[n >> 6] >> (n & 63)) & 1ull;",
            "      } else",
        ]
    body += [
        f"      right = f[fi] > {prefix}_thr[n];",
        f"      n = {prefix}_left[n] + (right ? 1u : 0u);",
        "    }",
        f"    {acc_expr} += {prefix}_thr[n] * {scale!r}f;",
        "  }",
    ]
    return "\n".join(body)


def to_cpp_routing(model, function_name: str = "ydf_model") -> str:
    """Table-driven (ROUTING) C++ codegen — the default algorithm."""
    forest = model.forest
    _check_embeddable(forest)
    feats = model.dataspec.feature_columns
    C = model._n_outputs()
    lines = [
        "// Generated by ydf_amd (MI355X-native decision forests).",
        "// Algorithm: ROUTING (node tables + fixed traversal loop).",
        "// Feature order: " + ", ".join(
            f"{i}:{c.name}" for i, c in enumerate(feats)),
        "#include <cstdint>",
        "#include <cmath>",
        "",
        _routing_tables(model, function_name),
        "",
    ]
    if C == 1:
        lines.append(f"float {function_name}_predict(const float* f) {{")
        lines.append(
            f"  float acc = {float(model.init_predictions[0])!r}f;")
        lines.append(_routing_walk(model, function_name, "acc"))
        if model.activation == "sigmoid":
            lines.append("  return 1.0f / (1.0f + std::exp(-acc));")
        elif model.activation == "exp":
            lines.append("  return std::exp(acc);")
        else:
            lines.append("  return acc;")
        lines.append("}")
    else:
        lines.append(f"void {function_name}_predict_multi("
                     "const float* f, float* out) {")
        for c in range(C):
            init = float(model.init_predictions[c]
                         if c < len(model.init_predictions) else 0.0)
            lines.append(f"  out[{c}] = {init!r}f;")
        lines.append(_routing_walk(model, function_name,
                                   f"out[t % {C}u]"))
        if model.activation == "softmax":
            lines.append(
                "  float m = out[0];\n"
                f"  for (int c = 1; c < {C}; ++c) m = out[c] > m ? out[c] "
                ": m;\n"
                "  float s = 0.f;\n"
                f"  for (int c = 0; c < {C}; ++c) {{ out[c] = "
                "std::exp(out[c] - m); s += out[c]; }\n"
                f"  for (int c = 0; c < {C}; ++c) out[c] /= s;")
        lines.append("}")
    return "\n".join(lines) + "\n"


def to_cpp(model, function_name: str = "ydf_model",
           algorithm: str = "ROUTING") -> str:
    """Generates a standalone C++ source string for `model`.

    algorithm="ROUTING" (default, like the reference embed.proto:38):
    node tables + fixed walk loop, O(nodes) data — scales to
    1000-tree forests. "IF_ELSE": per-node branch code (small models,
    fully branch-predictable)."""
    if algorithm == "ROUTING":
        return to_cpp_routing(model, function_name)
    forest = model.forest
    _check_embeddable(forest)
    feats = model.dataspec.feature_columns
    C = model._n_outputs()
    lines = [
        "// Generated by ydf_amd (MI355X-native decision forests).",
        "// Feature order: " + ", ".join(
            f"{i}:{c.name}" for i, c in enumerate(feats)),
        "#include <cstdint>",
        "#include <cmath>",
        "",
    ]
    scale = model._leaf_scale()

    def tree_fn(t: int) -> str:
        body: List[str] = []
        _emit_node(None, forest, int(forest.roots[t]), "  ", body)
        name = f"{function_name}_tree{t}"
        args = ", ".join(f"float f{i}" for i in range(len(feats)))
        lo = int(forest.roots[t])
        hi = int(forest.roots[t + 1]) if t + 1 < forest.n_trees \
            else forest.n_nodes
        used_set = {int(v) for v in forest.feat[lo:hi] if v >= 0}
        for n in range(lo, hi):
            ci = int(forest.cat_idx[n])
            if ci <= -2:
                oi = -(ci + 2)
                s0 = int(forest.obl_ranges[oi, 0])
                nn = int(forest.obl_ranges[oi, 1])
                used_set.update(int(a)
                                for a in forest.obl_attr[s0:s0 + nn])
        used = sorted(used_set)
        decl = (f"static inline float {name}(const float* f) {{\n"
                + "".join(f"  const float f{i} = f[{i}];\n" for i in used)
                + "  float acc = 0.f;\n"
                + "\n".join(body) + "\n  return acc;\n}\n")
        return decl

    for t in range(forest.n_trees):
        lines.append(tree_fn(t))
    if C == 1:
        lines.append(f"float {function_name}_predict(const float* f) {{")
        lines.append(f"  float acc = {float(model.init_predictions[0])!r}f;")
        for t in range(forest.n_trees):
            lines.append(f"  acc += {function_name}_tree{t}(f) * "
                         f"{scale!r}f;")
        if model.activation == "sigmoid":
            lines.append("  return 1.0f / (1.0f + std::exp(-acc));")
        elif model.activation == "exp":
            lines.append("  return std::exp(acc);")
        else:
            lines.append("  return acc;")
        lines.append("}")
    else:
        lines.append(f"void {function_name}_predict_multi("
                     "const float* f, float* out) {")
        for c in range(C):
            init = float(model.init_predictions[c]
                         if c < len(model.init_predictions) else 0.0)
            lines.append(f"  out[{c}] = {init!r}f;")
        lines.append(f"  for (int c = 0; c < {C}; ++c) {{}}")
        for t in range(forest.n_trees):
            lines.append(f"  out[{t % C}] += {function_name}_tree{t}(f) * "
                         f"{scale!r}f;")
        if model.activation == "softmax":
            lines.append(
                "  float m = out[0];\n"
                f"  for (int c = 1; c < {C}; ++c) m = out[c] > m ? out[c] "
                ": m;\n"
                "  float s = 0.f;\n"
                f"  for (int c = 0; c < {C}; ++c) {{ out[c] = "
                "std::exp(out[c] - m); s += out[c]; }\n"
                f"  for (int c = 0; c < {C}; ++c) out[c] /= s;")
        lines.append("}")
    return "\n".join(lines) + "\n"


def _emit_node_java(forest, n: int, indent: str, lines: List[str]):
    if forest.feat[n] < 0:
        lines.append(f"{indent}acc += {float(forest.thr[n])!r}f;")
        return
    fi = int(forest.feat[n])
    left = int(forest.left[n])
    ci = int(forest.cat_idx[n])
    if ci <= -2:  # oblique
        oi = -(ci + 2)
        s0 = int(forest.obl_ranges[oi, 0])
        nn = int(forest.obl_ranges[oi, 1])
        dot = " + ".join(
            f"{float(forest.obl_w[s0 + q])!r}f * f[{int(forest.obl_attr[s0 + q])}]"
            for q in range(nn))
        lines.append(f"{indent}if ({dot} > {float(forest.thr[n])!r}f) {{")
    elif ci >= 0:
        words = ", ".join(
            f"0x{int(w):016x}L"
            for w in np.asarray(forest.masks[ci], dtype=np.uint64))
        lines.append(f"{indent}{{ final long[] m = {{{words}}};")
        lines.append(
            f"{indent}  final int c = f[{fi}] < 0 ? 0 : (f[{fi}] > 255 "
            f"? 255 : (int) f[{fi}]);")
        lines.append(
            f"{indent}  if (((m[c >> 6] >>> (c & 63)) & 1L) != 0) {{")
    else:
        if forest.has_na_routing:
            na = "true" if forest.na_right[n] else "false"
            lines.append(
                f"{indent}if (Float.isNaN(f[{fi}]) ? {na} : "
                f"(f[{fi}] > {float(forest.thr[n])!r}f)) {{")
        else:
            lines.append(
                f"{indent}if (f[{fi}] > {float(forest.thr[n])!r}f) {{")
    _emit_node_java(forest, left + 1, indent + "  ", lines)
    lines.append(f"{indent}}} else {{")
    _emit_node_java(forest, left, indent + "  ", lines)
    lines.append(f"{indent}}}" + ("}" if ci >= 0 else ""))


def to_java(model, class_name: str = "YdfModel") -> str:
    """Generates a standalone Java class for `model` (reference
    serving/embed/java/java_embed.cc). Binary classification returns the
    positive-class probability; multi-class emits predictMulti."""
    forest = model.forest
    _check_embeddable(forest)
    feats = model.dataspec.feature_columns
    C = model._n_outputs()
    scale = model._leaf_scale()
    lines = [
        "// Generated by ydf_amd (MI355X-native decision forests).",
        "// Feature order: " + ", ".join(
            f"{i}:{c.name}" for i, c in enumerate(feats)),
        f"public final class {class_name} {{",
        f"  private {class_name}() {{}}",
        "",
    ]
    for t in range(forest.n_trees):
        body: List[str] = []
        _emit_node_java(forest, int(forest.roots[t]), "    ", body)
        lines.append(f"  private static float tree{t}(float[] f) {{")
        lines.append("    float acc = 0.f;")
        lines.extend(body)
        lines.append("    return acc;")
        lines.append("  }")
        lines.append("")
    if C == 1:
        lines.append("  public static float predict(float[] f) {")
        lines.append(
            f"    float acc = {float(model.init_predictions[0])!r}f;")
        for t in range(forest.n_trees):
            lines.append(f"    acc += tree{t}(f) * {scale!r}f;")
        if model.activation == "sigmoid":
            lines.append(
                "    return (float) (1.0 / (1.0 + Math.exp(-acc)));")
        elif model.activation == "exp":
            lines.append("    return (float) Math.exp(acc);")
        else:
            lines.append("    return acc;")
        lines.append("  }")
    else:
        lines.append("  public static float[] predictMulti(float[] f) {")
        lines.append(f"    float[] out = new float[{C}];")
        for c in range(C):
            init = float(model.init_predictions[c]
                         if c < len(model.init_predictions) else 0.0)
            lines.append(f"    out[{c}] = {init!r}f;")
        for t in range(forest.n_trees):
            lines.append(
                f"    out[{t % C}] += tree{t}(f) * {scale!r}f;")
        if model.activation == "softmax":
            lines.append(
                "    float m = out[0];\n"
                f"    for (int c = 1; c < {C}; ++c) m = out[c] > m ? "
                "out[c] : m;\n"
                "    float s = 0.f;\n"
                f"    for (int c = 0; c < {C}; ++c) {{ out[c] = (float) "
                "Math.exp(out[c] - m); s += out[c]; }\n"
                f"    for (int c = 0; c < {C}; ++c) out[c] /= s;")
        lines.append("    return out;")
        lines.append("  }")
    lines.append("}")
    return "\n".join(lines) + "\n"


def to_js(model, function_name: str = "ydfModel") -> str:
    """Generates a standalone JavaScript module for `model` — the
    capability analogue of the reference's JavaScript port
    (port/javascript: WASM inference over the same model format; here
    the model is compiled to dependency-free JS with routing tables).

    Exports (CommonJS + browser global):
      <name>_predict(features)        binary/regression, features =
                                      array in data-spec feature order
                                      (categorical = vocab index)
      <name>_predictMulti(features)   multi-class -> array of probs
      <name>_features                 feature name/type metadata
    """
    f = model.forest
    _check_embeddable(f)
    feats = model.dataspec.feature_columns
    C = model._n_outputs()
    scale = model._leaf_scale()
    n = f.n_nodes

    def arr(name, vals, typ):
        body = ",".join(str(v) for v in vals)
        return f"const {name} = new {typ}([{body}]);"

    lines = [
        "// Generated by ydf_amd (MI355X-native decision forests).",
        "// Feature order: " + ", ".join(
            f"{i}:{c.name}" for i, c in enumerate(feats)),
        "(function (root, factory) {",
        "  if (typeof module === 'object' && module.exports) "
        "{ module.exports = factory(); }",
        f"  else {{ root.{function_name} = factory(); }}",
        "}(typeof self !== 'undefined' ? self : this, function () {",
        arr("FEAT", (int(v) for v in f.feat), "Int32Array"),
        arr("LEFT", (int(v) for v in f.left), "Uint32Array"),
        arr("THR", (float(v) for v in f.thr), "Float32Array"),
        arr("CAT", (int(v) for v in f.cat_idx), "Int32Array"),
        arr("ROOTS", (int(v) for v in f.roots), "Uint32Array"),
    ]
    if len(f.masks):
        # JS lacks u64 literals in arrays pre-BigUint64; use 8 x u32
        words = []
        for m in np.asarray(f.masks, dtype=np.uint64).reshape(-1):
            words.append(int(m) & 0xFFFFFFFF)
            words.append(int(m) >> 32)
        lines.append(arr("MASKS", words, "Uint32Array"))
    if f.has_na_routing:
        lines.append(arr("NA", (int(v) for v in f.na_right),
                         "Uint8Array"))
    if len(f.obl_ranges):
        lines.append(arr("ORNG",
                         (int(v) for v in
                          np.asarray(f.obl_ranges).reshape(-1)),
                         "Int32Array"))
        lines.append(arr("OATTR", (int(v) for v in f.obl_attr),
                         "Int32Array"))
        lines.append(arr("OW", (float(v) for v in f.obl_w),
                         "Float32Array"))
    walk = [
        "  function margin(fv, cls) {",
        "    let acc = 0.0;",
        f"    for (let t = cls; t < {f.n_trees}; t += "
        f"{C if C > 1 else 1}) {{",
        "      let node = ROOTS[t];",
        "      for (;;) {",
        "        const fi = FEAT[node];",
        "        if (fi < 0) break;",
        "        const c = CAT[node];",
        "        let right;",
    ]
    if len(f.masks):
        walk += [
            "        if (c >= 0) {",
            "          const v = fv[fi] < 0 ? 0 : (fv[fi] > 255 ? 255 "
            ": Math.floor(fv[fi]));",
            "          const w = MASKS[8 * c + ((v >> 5) | 0)];",
            "          right = ((w >>> (v & 31)) & 1) !== 0;",
            "        } else",
        ]
    if len(f.obl_ranges):
        walk += [
            "        if (c <= -2) {",
            "          const r = -(c + 2);",
            "          let dot = 0.0;",
            "          for (let q = 0; q < ORNG[2*r+1]; ++q)",
            "            dot += OW[ORNG[2*r] + q] * "
            "fv[OATTR[ORNG[2*r] + q]];",
            "          right = dot > THR[node];",
            "        } else",
        ]
    if f.has_na_routing:
        walk += [
            "        if (Number.isNaN(fv[fi])) {",
            "          right = NA[node] !== 0;",
            "        } else",
        ]
    walk += [
        "        right = fv[fi] > THR[node];",
        "        node = LEFT[node] + (right ? 1 : 0);",
        "      }",
        f"      acc += THR[node] * {scale!r};",
        "    }",
        "    return acc;",
        "  }",
    ]
    lines += walk
    init0 = float(model.init_predictions[0])
    if C == 1:
        lines.append("  function predict(fv) {")
        lines.append(f"    const m = {init0!r} + margin(fv, 0);")
        if model.activation == "sigmoid":
            lines.append("    return 1.0 / (1.0 + Math.exp(-m));")
        elif model.activation == "exp":
            lines.append("    return Math.exp(m);")
        else:
            lines.append("    return m;")
        lines.append("  }")
    else:
        inits = ", ".join(
            repr(float(model.init_predictions[c]
                       if c < len(model.init_predictions) else 0.0))
            for c in range(C))
        lines += [
            "  function predict(fv) {",
            f"    const out = [{inits}];",
            f"    for (let c = 0; c < {C}; ++c) out[c] += margin(fv, c);",
        ]
        if model.activation == "softmax":
            lines += [
                "    const mx = Math.max.apply(null, out);",
                "    let s = 0.0;",
                f"    for (let c = 0; c < {C}; ++c) "
                "{ out[c] = Math.exp(out[c] - mx); s += out[c]; }",
                f"    for (let c = 0; c < {C}; ++c) out[c] /= s;",
            ]
        lines.append("    return out;")
        lines.append("  }")
    meta = ", ".join(
        f"{{name: {c.name!r}, type: {c.semantic.name!r}}}"
        for c in feats)
    lines += [
        f"  return {{ predict: predict, features: [{meta}] }};",
        "}));",
    ]
    return "\n".join(lines) + "\n"
