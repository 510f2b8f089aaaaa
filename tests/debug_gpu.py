"""Standalone GPU debug script (not a pytest test). Run via gpurun."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import quda_amd_hip as ext
from quda_amd import LatticeGeometry, SpinorField, GaugeField
from quda_amd.ops import blas
from quda_amd.ops.dispatch import dslash_wilson

print("arch:", torch.cuda.get_device_properties(0).gcnArchName)

geo = LatticeGeometry((4, 4, 4, 4))
x = SpinorField(geo, "double", "cuda", n_parity=1)
y = SpinorField(geo, "double", "cuda", n_parity=1)
x.data.fill_(1.0)
y.data.fill_(2.0)
blas.axpy(3.0, x, y)
torch.cuda.synchronize()
print("axpy (expect 5):", y.data.flatten()[:4].tolist())
print("norm2 (expect", 25 * 24 * geo.volume_cb, "):", blas.norm2(y))

# dslash on unit gauge, constant spinor => D psi = 4 psi
g = GaugeField(geo, "double", "cuda").unit_()
s = SpinorField(geo, "double", "cuda", n_parity=2)
c = torch.full((2, geo.volume_cb, 4, 3), 1.0, dtype=torch.complex128, device="cuda")
s.from_complex(c)
out = SpinorField(geo, "double", "cuda", n_parity=1)
dslash_wilson(out, s.parity_view(1), g, 0)
torch.cuda.synchronize()
oc = out.to_complex().cpu()
print("dslash const unit gauge (expect 4+0j):", oc.flatten()[:4].tolist())
print("max dev from 4:", (oc - 4).abs().max().item())
