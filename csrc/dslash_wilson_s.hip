// single-precision Wilson dslash TU (recon 18 + 12)
#include "dslash_wilson_impl.h"

void launch_dslash_wilson_single(const DslashCall &c, hipStream_t st) {
  if (c.recon == 12) dslash_launch_all<PrecSingle, 12>(c, st);
  else if (c.recon == 8) dslash_launch_all<PrecSingle, 8>(c, st);
  else dslash_launch_all<PrecSingle, 18>(c, st);
}

void launch_pack_face_single(const PackCall &c, hipStream_t st) {
  pack_launch<PrecSingle>(c, st);
}
