#!/bin/bash
cd /root/repo
python -m pytest tests/test_gpu_kernels.py -q -m gpu 2>&1 | tail -1
bash -c 'python - <<PY
import time, numpy as np, torch
import ydf_amd as ydf
rng = np.random.RandomState(0)
n = 200000; F = 28
d = {f"x{i}": rng.randn(n).astype(np.float32) for i in range(F)}
d["label"] = np.where(d["x0"] + d["x1"]*d["x2"] > 0, "a", "b")
m = ydf.GradientBoostedTreesLearner(label="label", num_trees=1000, max_depth=6,
                                    validation_ratio=0.0, device="cuda:0").train(d)
for B in (100, 1000, 10000):
    batch = {k: v[:B] for k, v in d.items() if k != "label"}
    X = m._encode_features(batch)
    sess = m.serving_session(B)
    p1, p2 = sess.predict(X), m.predict(batch, device="cuda:0")
    np.testing.assert_allclose(p1, p2, rtol=1e-5, atol=1e-6)
    for _ in range(5): sess.predict(X); m.predict(batch, device="cuda:0")
    t0 = time.perf_counter()
    for _ in range(50): sess.predict(X)
    tg = (time.perf_counter()-t0)/50
    t0 = time.perf_counter()
    for _ in range(50): m.predict(batch, device="cuda:0")
    te = (time.perf_counter()-t0)/50
    print(f"B={B}: session {tg*1e6:.0f}us vs eager predict {te*1e6:.0f}us")
PY'
timeout 300 python tools/bench_inference.py --rows 10000000 --engine binned4 2>/dev/null | tail -1 | python -c "import json,sys; d=json.load(sys.stdin); print('10M binned4', round(d['value']/1e6,1),'M ex/s')"
