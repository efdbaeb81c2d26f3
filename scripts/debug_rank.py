"""One-off: diagnose sorted centered-rank mismatch at n=131072."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from fiber_amd import ops

torch.manual_seed(3)
n = 131072
f = (torch.randn(n, device="cuda") * 4).round().contiguous()
got = ops.centered_rank(f)
want = ops.centered_rank_ref(f)
bad = (got != want).nonzero().flatten()
print("mismatches:", bad.numel(), "of", n)
for i in bad[:8].cpu().tolist():
    print("idx", i, "f", float(f[i]), "got", float(got[i]),
          "want", float(want[i]))
if bad.numel():
    i0 = int(bad[0])
    v = f[i0]
    grp = (f == v).nonzero().flatten()
    print("tie group for f=%r size %d" % (float(v), grp.numel()))
    g = (got[grp] + 0.5) * (n - 1)
    w = (want[grp] + 0.5) * (n - 1)
    print("rank sets equal:", torch.equal(g.sort().values, w.sort().values))
    print("grp idx[:10]", grp[:10].cpu().tolist())
    print("got ranks[:10]", g[:10].cpu().tolist())
    print("want ranks[:10]", w[:10].cpu().tolist())
    # is torch's stable argsort itself stable here?
    order = torch.argsort(f, stable=True)
    pos_in_sorted = torch.empty_like(order)
    pos_in_sorted[order] = torch.arange(n, device=f.device)
    tie_pos = pos_in_sorted[grp]
    print("torch tie positions monotone in index:",
          bool((tie_pos[1:] > tie_pos[:-1]).all()))
