"""Debug helper: dump per-(slot,feature) best gains/bins CPU vs GPU."""
import numpy as np
import torch

import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from ydf_amd import ops

rng = np.random.RandomState(3)
F, B, n_slots = 7, 256, 5
hist = rng.randint(0, 50, size=(n_slots, F, B, 3)).astype(np.float32)
ht = torch.from_numpy(hist)
abs_of_slot = torch.arange(7, 7 + n_slots, dtype=torch.int32)


def run(device):
    h = ht.to(device)
    aos = abs_of_slot.to(device)
    ns = torch.zeros((63, 3), device=device)
    bg = torch.empty((n_slots, F), device=device)
    bb = torch.empty((n_slots, F), dtype=torch.int32, device=device)
    bf = torch.empty(n_slots, dtype=torch.int32, device=device)
    bbin = torch.empty(n_slots, dtype=torch.int32, device=device)
    bgain = torch.empty(n_slots, device=device)
    ops.split_scan(h, aos, ns, bg, bb, bf, bbin, bgain, 0, n_slots,
                   1.0, 0.0, 5, 0.0)
    if device != "cpu":
        torch.cuda.synchronize()
    return (bg.cpu().numpy(), bb.cpu().numpy(), bf.cpu().numpy(),
            bbin.cpu().numpy(), bgain.cpu().numpy())


bg_c, bb_c, bf_c, bbin_c, bgain_c = run("cpu")
bg_g, bb_g, bf_g, bbin_g, bgain_g = run("cuda")
print("best_feat cpu:", bf_c, "gpu:", bf_g)
print("best_gain cpu:", bgain_c)
print("best_gain gpu:", bgain_g)
for s in range(n_slots):
    for f in range(F):
        if bb_c[s, f] != bb_g[s, f] or abs(bg_c[s, f] - bg_g[s, f]) > 1e-3:
            print(f"slot{s} f{f}: cpu gain={bg_c[s,f]:.6f} bin={bb_c[s,f]} "
                  f"| gpu gain={bg_g[s,f]:.6f} bin={bb_g[s,f]}")
print("max |bg diff|:", np.abs(bg_c - bg_g).max())
