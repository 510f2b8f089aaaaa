// double-precision Wilson dslash TU (recon 18 only at fp64)
#include "dslash_wilson_impl.h"

void launch_dslash_wilson_double(const DslashCall &c, hipStream_t st) {
  dslash_launch_all<PrecDouble, 18>(c, st);
}

void launch_pack_face_double(const PackCall &c, hipStream_t st) {
  pack_launch<PrecDouble>(c, st);
}
