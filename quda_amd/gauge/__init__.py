"""Gauge-sector tools (ref: SURVEY.md 2.9 — observables, smearing, HMC).

Torch-tensor implementations over the oracle gauge layout
[4, 2, V_cb, 3, 3]; they run on CPU or GPU (torch-ROCm dispatches to HIP
elementwise/rocBLAS batched kernels). Single-rank for now: these are
setup/analysis utilities, not the solver hot path."""

from .ops import (ape_smear, exp_su3, gauge_action, gauge_force,
                  plaquette, polyakov_loop, project_ta, staple_sum,
                  stout_smear, topological_charge, wilson_flow, wilson_loop,
                  hyp_smear, loop_trace, path_product, det_trace,
                  over_improved_stout_smear, topological_charge_density,
                  improved_gauge_action, improved_gauge_force,
                  energy_density, wilson_flow_measure, flow_scale_t0,
                  flow_scale_w0)
from .hmc import (hmc_trajectory, leapfrog, mom_action, nested_leapfrog,
                  omelyan, random_momentum)
from .fix import gauge_fix_ovr, gauge_fix_quality
from .heatbath import heatbath_sweep, overrelax_sweep
from .fermion_force import (fermion_action_and_force, hmc_trajectory_2f,
                            pseudofermion_refresh, wilson_fermion_force)

__all__ = ["plaquette", "gauge_action", "staple_sum", "gauge_force",
           "project_ta", "exp_su3", "ape_smear", "stout_smear",
           "wilson_flow", "polyakov_loop", "topological_charge",
           "energy_density", "wilson_flow_measure", "flow_scale_t0",
           "hyp_smear", "loop_trace", "path_product", "det_trace",
           "over_improved_stout_smear", "topological_charge_density",
           "improved_gauge_action", "improved_gauge_force",
           "flow_scale_w0",
           "leapfrog", "hmc_trajectory", "mom_action", "random_momentum",
           "omelyan", "nested_leapfrog",
           "heatbath_sweep", "overrelax_sweep", "wilson_fermion_force",
           "fermion_action_and_force", "hmc_trajectory_2f",
           "pseudofermion_refresh", "gauge_fix_ovr", "gauge_fix_quality", "wilson_loop"]
