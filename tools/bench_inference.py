#!/usr/bin/env python3
"""Serving benchmark (BASELINE config #5): batched inference of a
1000-tree depth-6 GBT over 10M synthetic rows on 1 MI355X.

Random-structure trees (as BASELINE.json specifies for the inference
bench) + synthetic feature matrix; measures examples/sec end-to-end
through the flat-forest HIP kernel (capability analogue of the reference's
cli/benchmark_inference.cc + serving/decision_forest engines).
"""
import argparse
import json
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from ydf_amd import ops  # noqa: E402


def random_forest_struct(n_trees: int, depth: int, n_features: int,
                         rng: np.random.RandomState):
    """Complete binary trees with random features/thresholds/leaves."""
    n_nodes = (1 << (depth + 1)) - 1
    n_internal = (1 << depth) - 1
    feat = np.empty(n_trees * n_nodes, dtype=np.int32)
    thr = np.empty(n_trees * n_nodes, dtype=np.float32)
    left = np.zeros(n_trees * n_nodes, dtype=np.int32)
    roots = np.arange(n_trees, dtype=np.int32) * n_nodes
    for t in range(n_trees):
        base = t * n_nodes
        feat[base:base + n_internal] = rng.randint(0, n_features, n_internal)
        feat[base + n_internal:base + n_nodes] = -1
        thr[base:base + n_internal] = rng.randn(n_internal)
        thr[base + n_internal:base + n_nodes] = \
            rng.randn(n_nodes - n_internal) * 0.1
        # BFS complete layout: left child of local node k is 2k+1
        k = np.arange(n_internal)
        left[base + k] = base + 2 * k + 1
    return feat, thr, left, roots


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=10_000_000)
    ap.add_argument("--trees", type=int, default=1000)
    ap.add_argument("--depth", type=int, default=6)
    ap.add_argument("--features", type=int, default=28)
    ap.add_argument("--runs", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--device", default=None)
    ap.add_argument("--engine", default="flat",
                    choices=["flat", "qs", "both", "binned8", "binned4",
                             "all"])
    args = ap.parse_args()
    device = torch.device(args.device) if args.device else (
        torch.device("cuda") if torch.cuda.is_available()
        else torch.device("cpu"))

    rng = np.random.RandomState(7)
    feat, thr, left, roots = random_forest_struct(
        args.trees, args.depth, args.features, rng)
    g = torch.Generator(device=device)
    g.manual_seed(42)
    X = torch.randn((args.features, args.rows), generator=g, device=device)
    featd = torch.from_numpy(feat).to(device)
    thrd = torch.from_numpy(thr).to(device)
    leftd = torch.from_numpy(left).to(device)
    rootsd = torch.from_numpy(roots).to(device)
    out = torch.empty(args.rows, dtype=torch.float32, device=device)
    act = torch.empty_like(out)

    b8 = None
    if args.engine in ("binned8", "binned4", "all") \
            and device.type == "cuda":
        from ydf_amd.model.forest import (FlatForest, pack_binned8_nodes,
                                          padded_boundaries)

        # quantile cut table from a sample; node thresholds map to bins
        sample, _ = torch.sort(X[:, :1 << 18], dim=1)
        m = sample.shape[1]
        qi = torch.linspace(0, m - 1, 257, device=device)[1:-1].long()
        bnd = sample[:, qi].contiguous()
        bins = torch.empty(X.shape, dtype=torch.uint8, device=device)
        from ydf_amd import ops as _ops

        ff8 = FlatForest(feat=feat, thr=thr, left=left, roots=roots)
        packed8 = torch.from_numpy(pack_binned8_nodes(
            ff8, bnd.cpu().numpy())).to(device)
        from ydf_amd.model.forest import pack_binned4_nodes

        n4, lv4 = pack_binned4_nodes(ff8, bnd.cpu().numpy())
        b8 = (bins, bnd, packed8,
              torch.from_numpy(n4).to(device),
              torch.from_numpy(lv4).to(device))

    qs = None
    if args.engine in ("qs", "both", "all"):
        from ydf_amd.model.forest import FlatForest, build_quickscorer

        ff = FlatForest(feat=feat, thr=thr, left=left, roots=roots)
        conds, offs, lv = build_quickscorer(ff)
        qs = (torch.from_numpy(conds).to(device),
              torch.from_numpy(offs).to(device),
              torch.from_numpy(lv).to(device))

    def run_flat():
        ops.predict_forest(X, featd, thrd, leftd, rootsd, out)
        ops.sigmoid(out, act)

    def run_qs():
        ops.predict_forest_qs(X, qs[0], qs[1], qs[2], out)
        ops.sigmoid(out, act)

    def run_b8():
        # binning included in the timed region (reads raw X per batch)
        ops.bin_data(X, b8[1], b8[0])
        ops.predict_forest_binned8(b8[0], b8[2], roots_d8, out)
        ops.sigmoid(out, act)

    roots_d8 = rootsd
    engines = {}
    if args.engine in ("flat", "both", "all"):
        engines["flat"] = run_flat
    if args.engine in ("qs", "both", "all"):
        engines["qs"] = run_qs
    if args.engine in ("binned8", "all") and b8 is not None:
        engines["binned8"] = run_b8
    if args.engine in ("binned4", "all") and b8 is not None:
        def run_b4():
            ops.bin_data(X, b8[1], b8[0])
            ops.predict_forest_binned4(b8[0], b8[3], b8[4], roots_d8,
                                       out)
            ops.sigmoid(out, act)
        engines["binned4"] = run_b4
    results = {}
    for name, run in engines.items():
        for _ in range(args.warmup):
            run()
        if device.type == "cuda":
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.runs):
            run()
        if device.type == "cuda":
            torch.cuda.synchronize()
        results[name] = (time.perf_counter() - t0) / args.runs
    dt = min(results.values())
    print(json.dumps({
        "metric": "gbt1000_d6_inference_examples_per_s",
        "engines": {k: args.rows / v for k, v in results.items()},
        "value": args.rows / dt,
        "unit": "examples/s",
        "rows": args.rows,
        "trees": args.trees,
        "depth": args.depth,
        "features": args.features,
        "ms_per_batch": dt * 1000,
        "us_per_example": dt / args.rows * 1e6,
        "device": str(device),
        "data": "synthetic",
    }))


if __name__ == "__main__":
    main()
