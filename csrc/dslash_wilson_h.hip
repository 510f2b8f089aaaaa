// half-precision Wilson dslash TU (recon 12 + 18)
#include "dslash_wilson_impl.h"

void launch_dslash_wilson_half(const DslashCall &c, hipStream_t st) {
  if (c.recon == 12) dslash_launch_all<PrecHalf, 12>(c, st);
  else if (c.recon == 8) dslash_launch_all<PrecHalf, 8>(c, st);
  else dslash_launch_all<PrecHalf, 18>(c, st);
}

void launch_pack_face_half(const PackCall &c, hipStream_t st) {
  pack_launch<PrecHalf>(c, st);
}
