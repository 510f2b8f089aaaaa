import time, torch
from quda_amd.fields.gauge import GaugeField
from quda_amd.fields.geometry import LatticeGeometry
from quda_amd.gauge import heatbath_sweep, overrelax_sweep, plaquette
from quda_amd.gauge.heatbath import set_device_rng
from quda_amd.gauge.ops import wilson_flow_measure, flow_scale_t0, flow_scale_w0
set_device_rng(True)
geo = LatticeGeometry((16, 16, 16, 32))
u = GaugeField(geo, "double", "cuda").unit_().to_complex()
t = time.time()
for it in range(400):
    u = heatbath_sweep(u, geo, 6.0, seed=100 + 7 * it)
    for j in range(3):
        u = overrelax_sweep(u, geo, 6.0, seed=9000 + 13 * it + j)
p, _, _ = plaquette(u, geo)
print(f"thermalized beta=6.0: plaq={p:.4f} ({time.time()-t:.0f}s)")
t = time.time()
uf, hist = wilson_flow_measure(u, geo, 0.02, 200)
torch.cuda.synchronize()
t0 = flow_scale_t0(hist)
w0 = flow_scale_w0(hist)
print(f"flow 200 steps eps 0.02 in {time.time()-t:.1f}s; "
      f"t0/a^2 = {t0:.3f} (lit ~2.79 at beta 6.0), w0^2/a^2 = {w0}")
