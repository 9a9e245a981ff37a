#!/bin/bash
cd /root/repo
echo "=== full GPU pytest ==="
python -m pytest tests/ -q -m gpu 2>&1 | tail -1
echo "=== smoke ==="
python -c "import __graft_entry__ as g; g.smoke(); print('smoke OK')" 2>&1 | tail -1
echo "=== honest-GBT on GPU (sanity) ==="
python - <<'PY' 2>&1 | tail -1
import numpy as np, ydf_amd as ydf
rng = np.random.RandomState(0); n = 200000
d = {"x1": rng.randn(n).astype(np.float32), "x2": rng.randn(n).astype(np.float32)}
d["label"] = np.where(d["x1"] + 0.5*d["x2"] + 0.3*rng.randn(n) > 0, "a", "b")
m = ydf.GradientBoostedTreesLearner(label="label", num_trees=30, validation_ratio=0.0,
                                    honest=True, device="cuda:0").train(d)
print("honest acc", round(m.evaluate(d).accuracy, 4))
PY
echo "=== bench spot ==="
python bench.py --steps 50 --warmup 8 2>/dev/null | python -c "import json,sys; d=json.load(sys.stdin); print('11M', round(d['value'],1),'trees/s')"
