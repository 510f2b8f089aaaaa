import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from quda_amd import GaugeField, LatticeGeometry, SpinorField
from quda_amd.ops import reference as ref
from quda_amd.ops.dispatch import dslash_wilson, hip_ext

geo = LatticeGeometry((8, 8, 8, 8))
g = GaugeField(geo, "double").random_su3_(seed=31)
psi = SpinorField(geo, "double").gaussian_(seed=32)
gd = GaugeField(geo, "half", "cuda", reconstruct="twelve").from_complex(
    g.to_complex().cuda())
sd = SpinorField(geo, "half", "cuda", n_parity=2).from_complex(
    psi.to_complex().cuda())
out0 = SpinorField(geo, "half", "cuda", n_parity=1)
out1 = SpinorField(geo, "half", "cuda", n_parity=1)
ext = hip_ext()
ext.set_dslash_lds(0)
dslash_wilson(out0, sd.parity_view(1), gd, 0)
ext.set_dslash_lds(1)
dslash_wilson(out1, sd.parity_view(1), gd, 0)
ext.set_dslash_lds(0)
a = out0.to_complex().cpu()[0]
b = out1.to_complex().cpu()[0]
d = (a - b).abs().amax(dim=(1, 2))
bad = (d > 1e-4 * a.abs().max()).nonzero().flatten()
print("total sites", d.numel(), "bad", bad.numel(), "max err",
      d.max().item(), "rel", (d.max() / a.abs().max()).item())
# coords of bad sites (parity 0)
X = geo.dims
lex_of_cb = geo.lex_of_cb[0]
for i in bad[:12].tolist():
    lex = lex_of_cb[i].item()
    x = lex % X[0]; r = lex // X[0]
    y = r % X[1]; r //= X[1]
    z = r % X[2]; t = r // X[2]
    print("bad site", i, (x, y, z, t), d[i].item())
# histogram by t and by x
if bad.numel():
    import collections
    ct = collections.Counter()
    for i in bad.tolist():
        lex = lex_of_cb[i].item()
        x = lex % X[0]; r = lex // X[0]
        y = r % X[1]; r //= X[1]
        z = r % X[2]; t = r // X[2]
        ct[("x", x)] += 1; ct[("t", t)] += 1
    print(sorted(ct.items()))
