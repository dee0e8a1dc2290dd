import numpy as np, sys, os
sys.path.insert(0, "/root/repo")
from dpark_amd.engine import Engine
g = np.load("/root/repo/tests/golden/synth_cfg2_shape.npz")
keys, vals = g["keys"], g["vals"]
print("n =", len(keys), "uniques =", len(np.unique(keys)),
      "min", keys.min(), "max", keys.max())
e = Engine(device=0)
e.reserve(len(keys), with_vals=True)
e.put(keys, vals)
print("packed:", e.is_packed())
e.sort()
k, v = e.get()
w = np.sort(keys)
bad = np.nonzero(k != w)[0]
print("misplaced:", len(bad), bad[:20])
if len(bad):
    i = int(bad[0]); lo = max(0, i-3)
    print("got ", k[lo:i+5])
    print("want", w[lo:i+5])
    # which input index held the misplaced value?
    # count multiset diff
    from collections import Counter
    cg, cw = Counter(k.tolist()), Counter(w.tolist())
    print("multiset equal:", cg == cw)
e.close()
