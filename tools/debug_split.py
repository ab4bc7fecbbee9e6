"""Debug harness: split_argmax HIP kernel vs torch reference, verbose."""
import sys
import torch
sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))
from spark_ensemble_amd.ops import dispatch as hip, reference as ref

def run(n, f, b, c, d, mig, mcw, seed=21):
    g = torch.Generator().manual_seed(seed)
    hist = torch.rand(n, f, b, c, generator=g)
    hist[..., :d] -= 0.5
    hg = hist.to("cuda:0")
    got = hip.split_search(hg, 1e-6, mcw, 1.0, mig, d_dims=d)
    want = ref.split_search(hist, 1e-6, mcw, 1.0, mig, d_dims=d)
    gg, gf, gb, gls = [t.cpu() for t in got]
    wg, wf, wb, wls = want
    bad = []
    for i in range(n):
        ok = (torch.isfinite(gg[i]) == torch.isfinite(wg[i]))
        if ok and torch.isfinite(wg[i]):
            ok = (abs(float(gg[i] - wg[i])) < 1e-3 + 1e-3*abs(float(wg[i]))
                  and int(gf[i]) == int(wf[i]) and int(gb[i]) == int(wb[i]))
        if not ok:
            bad.append(i)
    print(f"n={n} F={f} B={b} C={c} D={d} mig={mig} mcw={mcw}: "
          f"{len(bad)} bad nodes / {n}")
    for i in bad[:3]:
        print(f"  node {i}: got gain={float(gg[i]):.6f} f={int(gf[i])} b={int(gb[i])}"
              f" | want gain={float(wg[i]):.6f} f={int(wf[i])} b={int(wb[i])}")
        # reference per-candidate gain at both choices
        h1 = hist[i:i+1]
        cum = h1.cumsum(dim=2)
        total = h1.sum(dim=2)
        def gain_at(ff, bb):
            left = cum[0, ff, bb]
            right = total[0, ff] - left
            par = total[0, 0]
            def sc(s):
                gv = s[:d]; hh = s[d]
                return float((gv*gv).sum()) / (float(hh) + 1e-6)
            return sc(left) + sc(right) - sc(par)
        if int(gf[i]) >= 0:
            print(f"    ref gain at mine: {gain_at(int(gf[i]), int(gb[i])):.6f}")
        print(f"    ref gain at want: {gain_at(int(wf[i]), int(wb[i])):.6f}")

for (n, f, b, c, d) in [(3,40,32,4,2),(8,64,256,2,1),(1,256,256,3,1)]:
    for mig, mcw in [(0.0,0.0),(0.05,0.3)]:
        run(n, f, b, c, d, mig, mcw)
print("done")
