// quarter-precision (fp8-e4m3 block-float) Wilson dslash TU (recon 12+18)
// — the BASELINE config-5 precision class, gauge + spinor both fp8 with
// per-site fp32 norms on the spinors.
#include "dslash_wilson_impl.h"

void launch_dslash_wilson_quarter(const DslashCall &c, hipStream_t st) {
  if (c.recon == 12) dslash_launch_all<PrecQuarter, 12>(c, st);
  else if (c.recon == 8) dslash_launch_all<PrecQuarter, 8>(c, st);
  else dslash_launch_all<PrecQuarter, 18>(c, st);
}

void launch_pack_face_quarter(const PackCall &c, hipStream_t st) {
  pack_launch<PrecQuarter>(c, st);
}
