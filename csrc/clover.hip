// standalone clover apply TU
#include "dslash_wilson.h"
#include "launchers.h"

template <typename Prec>
static void clover_apply_t(const CloverApplyCall &c, hipStream_t st) {
  SpinorAcc<Prec> out{(typename Prec::Store *)c.out.data, (float *)c.out.norm, c.Vcb};
  SpinorAcc<Prec> in{(typename Prec::Store *)c.in.data, (float *)c.in.norm, c.Vcb};
  CloverAcc<Prec> cl{(const typename Prec::Store *)c.clover, c.Vcb};
  int blk = 256;
  int grid = (int)((c.sites + blk - 1) / blk);
  hipLaunchKernelGGL((k_clover_apply<Prec>), dim3(grid), dim3(blk), 0, st, out,
                     in, cl, c.parity, c.sites);
}

void launch_clover_apply(const CloverApplyCall &c, hipStream_t st) {
  switch (c.prec) {
    case 0: clover_apply_t<PrecDouble>(c, st); break;
    case 1: clover_apply_t<PrecSingle>(c, st); break;
    case 2: clover_apply_t<PrecHalf>(c, st); break;
  }
}
