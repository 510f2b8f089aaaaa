// pybind11/torch bindings for the quda_amd HIP kernels (in-tree extension).
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include "launchers.h"

namespace {

int prec_of(const at::Tensor &t) {
  switch (t.scalar_type()) {
    case at::kDouble: return 0;
    case at::kFloat: return 1;
    case at::kHalf: return 2;
    case at::kFloat8_e4m3fn: return 3;
    default: TORCH_CHECK(false, "unsupported dtype");
  }
}

void *ptr_or_null(const at::Tensor &t) {
  return t.numel() ? t.data_ptr() : nullptr;
}

BlasField field_of(const at::Tensor &data, const at::Tensor &norm, long Vcb) {
  return BlasField{ptr_or_null(data), ptr_or_null(norm), Vcb};
}

int chunk_w_of(const at::Tensor &t) {
  switch (t.scalar_type()) {  // 16B chunks for 24-real sites
    case at::kDouble: return 2;
    case at::kFloat: return 4;
    default: return 8;
  }
}

// field view with a site offset into the chunk-stride dimension
BlasField field_of_off(const at::Tensor &data, const at::Tensor &norm,
                       long v_stride, long s_offset) {
  BlasField f{ptr_or_null(data), ptr_or_null(norm), v_stride};
  if (s_offset && f.data) {
    f.data = (char *)f.data + (long)s_offset * chunk_w_of(data) * data.element_size();
    if (f.norm) f.norm = (float *)f.norm + s_offset;
  }
  return f;
}

hipStream_t stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

void check_launch(const char *what) {
  hipError_t e = hipGetLastError();
  TORCH_CHECK(e == hipSuccess, what, ": ", hipGetErrorString(e));
}

}  // namespace

static void dslash_wilson(at::Tensor out, at::Tensor out_n, at::Tensor in,
                          at::Tensor in_n, at::Tensor gauge, at::Tensor clover,
                          at::Tensor x, at::Tensor x_n,
                          std::vector<int64_t> dims, int64_t parity_offset,
                          int64_t Vcb, int64_t parity, bool dagger,
                          int64_t mode, bool xpay, double a, int64_t recon,
                          std::vector<at::Tensor> ghost,
                          std::vector<at::Tensor> ghost_nrm,
                          std::vector<int64_t> face_cb, int64_t comm_mask,
                          int64_t kt, double b_re, double b_im,
                          int64_t v_stride, int64_t s_offset) {
  // v_stride: chunk stride of the spinor fields (Ls*Vcb for 5-d fields);
  // s_offset: site offset of the 4-d slice being operated on (s*Vcb)
  TORCH_CHECK(out.is_contiguous() && in.is_contiguous() && gauge.is_contiguous());
  if (v_stride == 0) v_stride = Vcb;
  DslashCall c{};
  c.comm_mask = (int)comm_mask;
  c.kt = (int)kt;
  if (comm_mask) {
    TORCH_CHECK(ghost.size() == 8 && ghost_nrm.size() == 8 && face_cb.size() == 4);
    for (int k = 0; k < 8; ++k) {
      c.ghost[k] = ptr_or_null(ghost[k]);
      c.ghost_nrm[k] = (const float *)ptr_or_null(ghost_nrm[k]);
    }
    for (int k = 0; k < 4; ++k) c.face_cb[k] = face_cb[k];
  }
  c.out = field_of_off(out, out_n, v_stride, s_offset);
  c.in = field_of_off(in, in_n, v_stride, s_offset);
  c.x = field_of_off(x, x_n, v_stride, s_offset);
  c.gauge = gauge.data_ptr();
  c.clover = ptr_or_null(clover);
  for (int i = 0; i < 4; ++i) c.Xdim[i] = (int)dims[i];
  c.parity_offset = (int)parity_offset;
  c.Vcb = Vcb;
  c.parity = (int)parity;
  c.dagger = dagger;
  c.mode = (int)mode;
  c.xpay = xpay;
  c.a = a;
  c.b_re = b_re;
  c.b_im = b_im;
  c.recon = (int)recon;
  switch (prec_of(out)) {
    case 0: launch_dslash_wilson_double(c, stream()); break;
    case 1: launch_dslash_wilson_single(c, stream()); break;
    case 2: launch_dslash_wilson_half(c, stream()); break;
    case 3: launch_dslash_wilson_quarter(c, stream()); break;
  }
  check_launch("dslash_wilson");
}

static void dslash_wilson_mrhs(
    std::vector<at::Tensor> out, std::vector<at::Tensor> out_n,
    std::vector<at::Tensor> in, std::vector<at::Tensor> in_n,
    at::Tensor gauge, at::Tensor clover, std::vector<at::Tensor> x,
    std::vector<at::Tensor> x_n, std::vector<int64_t> dims,
    int64_t parity_offset, int64_t Vcb, int64_t parity, bool dagger,
    int64_t mode, bool xpay, double a, int64_t recon,
    std::vector<at::Tensor> ghost, std::vector<at::Tensor> ghost_nrm,
    std::vector<int64_t> face_cb, int64_t comm_mask, int64_t kt,
    int64_t v_stride, std::vector<int64_t> s_offsets) {
  // v_stride/s_offsets: address 4-d slices of 5-d fields (domain-wall
  // s-batching — the same tensor appears n times with different offsets)
  int n = (int)out.size();
  TORCH_CHECK(n == 2 || n == 4, "mrhs kernel supports 2 or 4 RHS");
  TORCH_CHECK((int)in.size() == n);
  if (v_stride == 0) v_stride = Vcb;
  DslashMrhsCall c{};
  c.nrhs = n;
  for (int r = 0; r < n; ++r) {
    TORCH_CHECK(out[r].is_contiguous() && in[r].is_contiguous());
    long so = s_offsets.empty() ? 0 : s_offsets[r];
    c.out[r] = field_of_off(out[r], out_n[r], v_stride, so);
    c.in[r] = field_of_off(in[r], in_n[r], v_stride, so);
    if (!x.empty()) c.x[r] = field_of_off(x[r], x_n[r], v_stride, so);
  }
  c.comm_mask = (int)comm_mask;
  c.kt = (int)kt;
  if (comm_mask) {
    TORCH_CHECK(ghost.size() == 8 && ghost_nrm.size() == 8 && face_cb.size() == 4);
    for (int k = 0; k < 8; ++k) {
      c.ghost[k] = ptr_or_null(ghost[k]);
      c.ghost_nrm[k] = (const float *)ptr_or_null(ghost_nrm[k]);
    }
    for (int k = 0; k < 4; ++k) c.face_cb[k] = face_cb[k];
  }
  c.gauge = gauge.data_ptr();
  c.clover = ptr_or_null(clover);
  for (int i = 0; i < 4; ++i) c.Xdim[i] = (int)dims[i];
  c.parity_offset = (int)parity_offset;
  c.Vcb = Vcb;
  c.parity = (int)parity;
  c.dagger = dagger;
  c.mode = (int)mode;
  c.xpay = xpay;
  c.a = a;
  c.recon = (int)recon;
  c.prec = prec_of(out[0]);
  launch_dslash_wilson_mrhs(c, stream());
  check_launch("dslash_wilson_mrhs");
}

static void pack_face(at::Tensor dst, at::Tensor dst_nrm, at::Tensor in,
                      at::Tensor in_n, std::vector<int64_t> dims,
                      int64_t parity_offset, int64_t Vcb, int64_t parity,
                      int64_t mu, int64_t s01, int64_t edge, int64_t Fcb,
                      int64_t v_stride = 0, int64_t s_offset = 0) {
  PackCall c{};
  c.in = field_of_off(in, in_n, v_stride ? v_stride : Vcb, s_offset);
  c.dst = dst.data_ptr();
  c.dst_nrm = (float *)ptr_or_null(dst_nrm);
  for (int i = 0; i < 4; ++i) c.Xdim[i] = (int)dims[i];
  c.parity_offset = (int)parity_offset;
  c.Vcb = Vcb;
  c.parity = (int)parity;
  c.mu = (int)mu;
  c.s01 = (int)s01;
  c.edge = (int)edge;
  c.Fcb = Fcb;
  c.prec = prec_of(in);
  switch (c.prec) {
    case 0: launch_pack_face_double(c, stream()); break;
    case 1: launch_pack_face_single(c, stream()); break;
    case 2: launch_pack_face_half(c, stream()); break;
    case 3: launch_pack_face_quarter(c, stream()); break;
  }
  check_launch("pack_face");
}

static void pack_face_stag(at::Tensor dst, at::Tensor dst_nrm, at::Tensor in,
                           at::Tensor in_n, std::vector<int64_t> dims,
                           int64_t parity_offset, int64_t Vcb, int64_t parity,
                           int64_t mu, int64_t edge, int64_t Fcb,
                           int64_t depth) {
  PackCall c{};
  c.in = field_of(in, in_n, Vcb);
  c.dst = dst.data_ptr();
  c.dst_nrm = (float *)ptr_or_null(dst_nrm);
  for (int i = 0; i < 4; ++i) c.Xdim[i] = (int)dims[i];
  c.parity_offset = (int)parity_offset;
  c.Vcb = Vcb;
  c.parity = (int)parity;
  c.mu = (int)mu;
  c.edge = (int)edge;
  c.Fcb = Fcb;
  c.prec = prec_of(in);
  c.depth = (int)depth;
  launch_pack_face_stag(c, stream());
  check_launch("pack_face_stag");
}

static void dslash_staggered(at::Tensor out, at::Tensor out_n, at::Tensor in,
                             at::Tensor in_n, at::Tensor gauge,
                             at::Tensor long_gauge, at::Tensor x,
                             at::Tensor x_n, std::vector<int64_t> dims,
                             int64_t parity_offset, int64_t Vcb,
                             int64_t parity, bool xpay, double a, double b,
                             int64_t recon, std::vector<at::Tensor> ghost,
                             std::vector<at::Tensor> ghost_nrm,
                             std::vector<int64_t> face_cb, int64_t comm_mask,
                             int64_t kt, int64_t ghost_depth) {
  TORCH_CHECK(out.is_contiguous() && in.is_contiguous() && gauge.is_contiguous());
  StagDslashCall c{};
  c.out = field_of(out, out_n, Vcb);
  c.in = field_of(in, in_n, Vcb);
  c.x = field_of(x, x_n, Vcb);
  c.gauge = gauge.data_ptr();
  c.long_gauge = ptr_or_null(long_gauge);
  for (int i = 0; i < 4; ++i) c.Xdim[i] = (int)dims[i];
  c.parity_offset = (int)parity_offset;
  c.Vcb = Vcb;
  c.parity = (int)parity;
  c.xpay = xpay;
  c.a = a;
  c.b = b;
  c.recon = (int)recon;
  c.comm_mask = (int)comm_mask;
  c.kt = (int)kt;
  c.prec = prec_of(out);
  c.ghost_depth = (int)ghost_depth;
  if (comm_mask) {
    TORCH_CHECK(ghost.size() == 8 && ghost_nrm.size() == 8 && face_cb.size() == 4);
    for (int k = 0; k < 8; ++k) {
      c.ghost[k] = ptr_or_null(ghost[k]);
      c.ghost_nrm[k] = (const float *)ptr_or_null(ghost_nrm[k]);
    }
    for (int k = 0; k < 4; ++k) c.face_cb[k] = face_cb[k];
  }
  launch_dslash_staggered(c, stream());
  check_launch("dslash_staggered");
}

static at::Tensor blas_op(int64_t op, double a, double b, at::Tensor x,
                          at::Tensor x_n, at::Tensor y, at::Tensor y_n,
                          int64_t Vcb, int64_t sites, double c2 = 0.0,
                          double d2 = 0.0, int64_t ncomp = 24,
                          bool deterministic = false,
                          c10::optional<at::Tensor> z = c10::nullopt,
                          c10::optional<at::Tensor> z_n = c10::nullopt,
                          c10::optional<at::Tensor> w = c10::nullopt,
                          c10::optional<at::Tensor> w_n = c10::nullopt) {
  BlasCall c{};
  c.op = (int)op;
  c.prec = prec_of(x);
  c.a = a;
  c.b = b;
  c.c = c2;
  c.d = d2;
  c.ncomp = (int)ncomp;
  c.det = deterministic;
  c.x = field_of(x, x_n, Vcb);
  c.y = field_of(y, y_n, Vcb);
  if (z) c.z = field_of(*z, *z_n, Vcb);
  if (w) c.w = field_of(*w, *w_n, Vcb);
  c.sites = sites;
  at::Tensor result;
  bool reduction = (op == BLAS_AXPY_NORM2 || op == BLAS_XMY_NORM2 ||
                    op == BLAS_NORM2 || op == BLAS_REDOT || op == BLAS_CDOT ||
                    op == BLAS_TRIPLE_CG);
  if (reduction) {
    if (deterministic) {
      long g = (sites + 255) / 256;
      if (g > 2048) g = 2048;
      result = at::zeros({g, 2}, x.options().dtype(at::kDouble));
    } else {
      result = at::zeros({2}, x.options().dtype(at::kDouble));
    }
    c.result = result.data_ptr<double>();
  }
  launch_blas(c, stream());
  check_launch("blas_op");
  if (reduction && deterministic) result = result.sum(0);
  return result;
}

static void convert(at::Tensor dst, at::Tensor dst_n, at::Tensor src,
                    at::Tensor src_n, int64_t Vcb, int64_t sites,
                    int64_t ncomp = 24) {
  launch_convert(field_of(dst, dst_n, Vcb), prec_of(dst),
                 field_of(src, src_n, Vcb), prec_of(src), sites, (int)ncomp,
                 stream());
  check_launch("convert");
}

static void clover_apply(at::Tensor out, at::Tensor out_n, at::Tensor in,
                         at::Tensor in_n, at::Tensor clover, int64_t parity,
                         int64_t Vcb, int64_t v_stride, int64_t s_offset) {
  CloverApplyCall c{};
  c.out = field_of_off(out, out_n, v_stride ? v_stride : Vcb, s_offset);
  c.in = field_of_off(in, in_n, v_stride ? v_stride : Vcb, s_offset);
  c.clover = clover.data_ptr();
  c.parity = (int)parity;
  c.Vcb = Vcb;
  c.sites = Vcb;
  c.prec = prec_of(out);
  launch_clover_apply(c, stream());
  check_launch("clover_apply");
}

static void twist_apply(at::Tensor out, at::Tensor out_n, at::Tensor in,
                        at::Tensor in_n, double b_re, double b_im,
                        int64_t Vcb, int64_t sites, int64_t tau3_vcb = 0,
                        bool acc = false) {
  TwistApplyCall c{};
  c.out = field_of(out, out_n, Vcb);
  c.in = field_of(in, in_n, Vcb);
  c.b_re = b_re;
  c.b_im = b_im;
  c.sites = sites;
  c.tau3_vcb = tau3_vcb;
  c.acc = acc;
  c.prec = prec_of(out);
  launch_twist_apply(c, stream());
  check_launch("twist_apply");
}

static void dwf5(at::Tensor out, at::Tensor out_n, at::Tensor in,
                 at::Tensor in_n, at::Tensor x, at::Tensor x_n,
                 int64_t Vcb4, int64_t Ls, bool xpay, bool dagger, double a,
                 double alpha, double beta, double mf, int64_t kind) {
  Dwf5Call c{};
  long stride = Vcb4 * Ls;
  c.out = field_of(out, out_n, stride);
  c.in = field_of(in, in_n, stride);
  c.x = field_of(x, x_n, stride);
  c.Vcb4 = Vcb4;
  c.Ls = (int)Ls;
  c.xpay = xpay;
  c.dagger = dagger;
  c.a = a;
  c.alpha = alpha;
  c.beta = beta;
  c.mf = mf;
  c.prec = prec_of(out);
  c.kind = (int)kind;
  launch_dwf5(c, stream());
  check_launch("dwf5");
}

static void zdwf5(at::Tensor out, at::Tensor out_n, at::Tensor in,
                  at::Tensor in_n, at::Tensor x, at::Tensor x_n,
                  int64_t Vcb4, int64_t Ls, bool xpay, double a_re,
                  double a_im, int64_t kind,
                  std::vector<double> au, std::vector<double> al,
                  std::vector<double> wu, std::vector<double> wl,
                  std::vector<int64_t> su, std::vector<int64_t> sl,
                  std::vector<int64_t> ord_u, std::vector<int64_t> ord_l,
                  std::vector<double> diu, std::vector<double> eu,
                  std::vector<double> dil, std::vector<double> el,
                  double cwu_re, double cwu_im, double cwl_re, double cwl_im) {
  TORCH_CHECK(Ls <= QA_ZMAX, "zMobius Ls > ", QA_ZMAX);
  ZCoef zc{};
  auto put2 = [&](double dst[][2], const std::vector<double> &v) {
    for (int s = 0; s < (int)Ls && 2 * s + 1 < (int)v.size(); ++s) {
      dst[s][0] = v[2 * s];
      dst[s][1] = v[2 * s + 1];
    }
  };
  put2(zc.au, au); put2(zc.al, al); put2(zc.wu, wu); put2(zc.wl, wl);
  put2(zc.diu, diu); put2(zc.eu, eu); put2(zc.dil, dil); put2(zc.el, el);
  for (int s = 0; s < (int)Ls; ++s) {
    if (s < (int)su.size()) zc.su[s] = (int)su[s];
    if (s < (int)sl.size()) zc.sl[s] = (int)sl[s];
    if (s < (int)ord_u.size()) zc.ord_u[s] = (int)ord_u[s];
    if (s < (int)ord_l.size()) zc.ord_l[s] = (int)ord_l[s];
  }
  zc.cwu[0] = cwu_re; zc.cwu[1] = cwu_im;
  zc.cwl[0] = cwl_re; zc.cwl[1] = cwl_im;
  ZDwf5Call c{};
  long stride = Vcb4 * Ls;
  c.out = field_of(out, out_n, stride);
  c.in = field_of(in, in_n, stride);
  c.x = field_of(x, x_n, stride);
  c.Vcb4 = Vcb4;
  c.Ls = (int)Ls;
  c.xpay = xpay;
  c.a_re = a_re;
  c.a_im = a_im;
  c.prec = prec_of(out);
  c.kind = (int)kind;
  TORCH_CHECK(c.prec != 2, "zdwf5: half precision unsupported");
  c.zc = &zc;
  launch_zdwf5(c, stream());
  check_launch("zdwf5");
}

static void eofa5(at::Tensor out, at::Tensor out_n, at::Tensor in,
                  at::Tensor in_n, at::Tensor x, at::Tensor x_n,
                  int64_t Vcb4, int64_t Ls, bool xpay, bool dagger, double a,
                  double alpha, double beta, double mf,
                  std::vector<double> u, std::vector<double> w, double sh,
                  int64_t pm, int64_t kind) {
  TORCH_CHECK(Ls <= 32, "eofa5: Ls > 32");
  Eofa5Call c{};
  long stride = Vcb4 * Ls;
  c.out = field_of(out, out_n, stride);
  c.in = field_of(in, in_n, stride);
  c.x = field_of(x, x_n, stride);
  c.Vcb4 = Vcb4;
  c.Ls = (int)Ls;
  c.xpay = xpay;
  c.dagger = dagger;
  c.a = a;
  c.alpha = alpha;
  c.beta = beta;
  c.mf = mf;
  for (int s = 0; s < (int)Ls; ++s) {
    if (s < (int)u.size()) c.u[s] = u[s];
    if (s < (int)w.size()) c.w[s] = w[s];
  }
  c.sh = sh;
  c.pm = (int)pm;
  c.prec = prec_of(out);
  c.kind = (int)kind;
  launch_eofa5(c, stream());
  check_launch("eofa5");
}

static void set_dslash_block(int64_t b) {
  if (b == 64 || b == 128 || b == 256) qa_dslash_block_ref() = (int)b;
}

static void set_dslash_waves(int64_t w) {
  if (w == 0 || w == 3) qa_dslash_waves_ref() = (int)w;
}

static void set_dslash_lds(int64_t v) { qa_dslash_lds_ref() = (int)v; }

// ---------------------------------------------------------------------------
// HIP-IPC remote-write halos (role of comm_target.cpp:41-134 P2P
// remote-write): export/import device-buffer handles so the pack kernel
// writes directly into the PEER rank's recv buffer over xGMI.
static py::bytes ipc_get_handle(at::Tensor t) {
  // torch's caching allocator suballocates: the IPC handle refers to the
  // BASE allocation, so export (handle, offset-of-tensor-within-it)
  TORCH_CHECK(t.is_cuda() && t.is_contiguous());
  void *base = nullptr;
  size_t sz = 0;
  hipError_t e = hipMemGetAddressRange((hipDeviceptr_t *)&base, &sz,
                                       (hipDeviceptr_t)t.data_ptr());
  TORCH_CHECK(e == hipSuccess, "hipMemGetAddressRange: ",
              hipGetErrorString(e));
  hipIpcMemHandle_t h;
  e = hipIpcGetMemHandle(&h, base);
  TORCH_CHECK(e == hipSuccess, "hipIpcGetMemHandle: ", hipGetErrorString(e));
  uint64_t off = (uint64_t)((char *)t.data_ptr() - (char *)base);
  std::string out(sizeof(h) + sizeof(off), '\0');
  memcpy(&out[0], &h, sizeof(h));
  memcpy(&out[sizeof(h)], &off, sizeof(off));
  return py::bytes(out);
}

static py::tuple ipc_open_handle(py::bytes handle) {
  // returns (tensor_ptr, mapped_base) — close with the BASE
  std::string s = handle;
  TORCH_CHECK(s.size() == sizeof(hipIpcMemHandle_t) + sizeof(uint64_t));
  hipIpcMemHandle_t h;
  uint64_t off;
  memcpy(&h, s.data(), sizeof(h));
  memcpy(&off, s.data() + sizeof(h), sizeof(off));
  void *ptr = nullptr;
  hipError_t e =
      hipIpcOpenMemHandle(&ptr, h, hipIpcMemLazyEnablePeerAccess);
  TORCH_CHECK(e == hipSuccess, "hipIpcOpenMemHandle: ", hipGetErrorString(e));
  return py::make_tuple((int64_t)((uintptr_t)ptr + off),
                        (int64_t)(uintptr_t)ptr);
}

static void ipc_close_handle(int64_t ptr) {
  hipIpcCloseMemHandle((void *)(uintptr_t)ptr);
}

// pack a face directly into a raw (peer) destination pointer
static void pack_face_ptr(int64_t dst, int64_t dst_nrm, at::Tensor in,
                          at::Tensor in_n, std::vector<int64_t> dims,
                          int64_t parity_offset, int64_t Vcb, int64_t parity,
                          int64_t mu, int64_t s01, int64_t edge, int64_t Fcb,
                          int64_t v_stride, int64_t s_offset) {
  PackCall c{};
  c.in = field_of_off(in, in_n, v_stride ? v_stride : Vcb, s_offset);
  c.dst = (void *)(uintptr_t)dst;
  c.dst_nrm = (float *)(uintptr_t)dst_nrm;
  for (int i = 0; i < 4; ++i) c.Xdim[i] = (int)dims[i];
  c.parity_offset = (int)parity_offset;
  c.Vcb = Vcb;
  c.parity = (int)parity;
  c.mu = (int)mu;
  c.s01 = (int)s01;
  c.edge = (int)edge;
  c.Fcb = Fcb;
  c.prec = prec_of(in);
  switch (c.prec) {
    case 0: launch_pack_face_double(c, stream()); break;
    case 1: launch_pack_face_single(c, stream()); break;
    case 2: launch_pack_face_half(c, stream()); break;
    case 3: launch_pack_face_quarter(c, stream()); break;
  }
  check_launch("pack_face_ptr");
}

static void heatbath_sweep_dir(at::Tensor u, std::vector<int64_t> dims,
                               int64_t parity_offset, int64_t Vcb,
                               int64_t parity, int64_t mu, double beta_eff,
                               int64_t seed, int64_t mode) {
  TORCH_CHECK(u.is_contiguous() && u.scalar_type() == at::kComplexDouble);
  HeatbathCall c{};
  c.u = u.data_ptr();
  for (int i = 0; i < 4; ++i) c.Xdim[i] = (int)dims[i];
  c.parity_offset = (int)parity_offset;
  c.Vcb = Vcb;
  c.parity = (int)parity;
  c.mu = (int)mu;
  c.beta_eff = beta_eff;
  c.seed = (unsigned long long)seed;
  c.mode = (int)mode;
  launch_heatbath(c, stream());
  check_launch("heatbath");
}

static void stout_smear_dir(at::Tensor out, at::Tensor in,
                            std::vector<int64_t> dims,
                            int64_t parity_offset, int64_t Vcb, int64_t mu,
                            double rho) {
  TORCH_CHECK(out.is_contiguous() && in.is_contiguous() &&
              in.scalar_type() == at::kComplexDouble);
  StoutCall c{};
  c.out = out.data_ptr();
  c.in = in.data_ptr();
  for (int i = 0; i < 4; ++i) c.Xdim[i] = (int)dims[i];
  c.parity_offset = (int)parity_offset;
  c.Vcb = Vcb;
  c.mu = (int)mu;
  c.rho = rho;
  launch_stout(c, stream());
  check_launch("stout");
}

static void flow_zmat_dir(at::Tensor Z, at::Tensor in,
                          std::vector<int64_t> dims, int64_t parity_offset,
                          int64_t Vcb, int64_t mu, double eps) {
  TORCH_CHECK(Z.is_contiguous() && in.is_contiguous());
  StoutCall c{};
  c.out = Z.data_ptr();
  c.in = in.data_ptr();
  for (int i = 0; i < 4; ++i) c.Xdim[i] = (int)dims[i];
  c.parity_offset = (int)parity_offset;
  c.Vcb = Vcb;
  c.mu = (int)mu;
  c.rho = eps;
  launch_zmat(c, stream());
  check_launch("zmat");
}

static void flow_expmul_dir(at::Tensor out, at::Tensor in, at::Tensor Zc,
                            std::vector<int64_t> dims,
                            int64_t parity_offset, int64_t Vcb, int64_t mu) {
  TORCH_CHECK(out.is_contiguous() && in.is_contiguous() &&
              Zc.is_contiguous());
  StoutCall c{};
  c.out = out.data_ptr();
  c.in = in.data_ptr();
  c.aux = Zc.data_ptr();
  for (int i = 0; i < 4; ++i) c.Xdim[i] = (int)dims[i];
  c.parity_offset = (int)parity_offset;
  c.Vcb = Vcb;
  c.mu = (int)mu;
  launch_expmul(c, stream());
  check_launch("expmul");
}

static void coarse_dslash_mfma(at::Tensor mats, at::Tensor nbr9,
                               at::Tensor c, at::Tensor out, int64_t Na,
                               int64_t Nc, int64_t NR) {
  TORCH_CHECK(mats.scalar_type() == at::kComplexFloat &&
              c.scalar_type() == at::kComplexFloat &&
              out.scalar_type() == at::kComplexFloat);
  TORCH_CHECK(nbr9.scalar_type() == at::kLong);
  TORCH_CHECK(mats.is_contiguous() && nbr9.is_contiguous() &&
              c.is_contiguous() && out.is_contiguous());
  TORCH_CHECK(Nc % 16 == 0 && NR >= 1 && NR <= 16);
  CoarseMfmaCall cc{mats.data_ptr(), nbr9.data_ptr(), c.data_ptr(),
                    out.data_ptr(), Na, (int)Nc, (int)NR};
  launch_coarse_dslash_mfma(cc, stream());
  check_launch("coarse_dslash_mfma");
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("dslash_wilson", &dslash_wilson, "Wilson(-clover/-twisted) dslash",
        py::arg("out"), py::arg("out_n"), py::arg("in"), py::arg("in_n"),
        py::arg("gauge"), py::arg("clover"), py::arg("x"), py::arg("x_n"),
        py::arg("dims"), py::arg("parity_offset"), py::arg("Vcb"),
        py::arg("parity"), py::arg("dagger"), py::arg("mode"),
        py::arg("xpay"), py::arg("a"), py::arg("recon"), py::arg("ghost"),
        py::arg("ghost_nrm"), py::arg("face_cb"), py::arg("comm_mask"),
        py::arg("kt"), py::arg("b_re") = 0.0, py::arg("b_im") = 0.0,
        py::arg("v_stride") = 0, py::arg("s_offset") = 0);
  m.def("pack_face", &pack_face, "halo face pack (spin-projected)",
        py::arg("dst"), py::arg("dst_nrm"), py::arg("in"), py::arg("in_n"),
        py::arg("dims"), py::arg("parity_offset"), py::arg("Vcb"),
        py::arg("parity"), py::arg("mu"), py::arg("s01"), py::arg("edge"),
        py::arg("Fcb"), py::arg("v_stride") = 0, py::arg("s_offset") = 0);
  m.def("pack_face_stag", &pack_face_stag, "staggered halo face pack");
  m.def("dslash_wilson_mrhs", &dslash_wilson_mrhs,
        "multi-RHS Wilson(-clover) dslash: NRHS sides per gauge load",
        py::arg("out"), py::arg("out_n"), py::arg("in"), py::arg("in_n"),
        py::arg("gauge"), py::arg("clover"), py::arg("x"), py::arg("x_n"),
        py::arg("dims"), py::arg("parity_offset"), py::arg("Vcb"),
        py::arg("parity"), py::arg("dagger"), py::arg("mode"),
        py::arg("xpay"), py::arg("a"), py::arg("recon"), py::arg("ghost"),
        py::arg("ghost_nrm"), py::arg("face_cb"), py::arg("comm_mask"),
        py::arg("kt"), py::arg("v_stride") = 0,
        py::arg("s_offsets") = std::vector<int64_t>{});
  m.def("flow_zmat_dir", &flow_zmat_dir, "wilson-flow Z = eps TA[S U^d]");
  m.def("flow_expmul_dir", &flow_expmul_dir, "U' = exp(Zc) U");
  m.def("stout_smear_dir", &stout_smear_dir,
        "native stout smear of one direction (exp via scale-and-square)");
  m.def("heatbath_sweep_dir", &heatbath_sweep_dir,
        "SU(2)-subgroup heatbath/overrelax update of one (parity, mu)");
  m.def("coarse_dslash_mfma", &coarse_dslash_mfma,
        "coarse-grid 9-matrix dslash on f32 MFMA tiles");
  m.def("dslash_staggered", &dslash_staggered, "naive staggered dslash");
  m.def("set_dslash_block", &set_dslash_block, "autotuner: dslash workgroup size");
  m.def("set_dslash_waves", &set_dslash_waves,
        "occupancy experiment: 0 default, 3 = 64-thread/3-wave variant");
  m.def("set_dslash_lds", &set_dslash_lds,
        "LDS-tiled dslash policy: 1 = stage the in-spinor halo tile in LDS "
        "(half/quarter, local, tile-divisible dims)");
  m.def("ipc_get_handle", &ipc_get_handle, "hipIpcGetMemHandle of a tensor");
  m.def("ipc_open_handle", &ipc_open_handle, "open a peer IPC handle");
  m.def("ipc_close_handle", &ipc_close_handle, "close a peer IPC mapping");
  m.def("pack_face_ptr", &pack_face_ptr,
        "pack a halo face into a raw (IPC peer) pointer");
  m.def("dwf5", &dwf5, "DWF/Moebius 5th-dim ops (Ds apply / M5 inverse)");
  m.def("zdwf5", &zdwf5, "zMobius per-slice-complex 5th-dim ops");
  m.def("eofa5", &eofa5, "EOFA rank-1 extended M5 ops");
  m.def("blas_op", &blas_op, "fused blas/reduction",
        py::arg("op"), py::arg("a"), py::arg("b"), py::arg("x"),
        py::arg("x_n"), py::arg("y"), py::arg("y_n"), py::arg("Vcb"),
        py::arg("sites"), py::arg("c2") = 0.0, py::arg("d2") = 0.0,
        py::arg("ncomp") = 24, py::arg("deterministic") = false,
        py::arg("z") = py::none(), py::arg("z_n") = py::none(),
        py::arg("w") = py::none(), py::arg("w_n") = py::none());
  m.def("convert", &convert, "precision conversion copy", py::arg("dst"),
        py::arg("dst_n"), py::arg("src"), py::arg("src_n"), py::arg("Vcb"),
        py::arg("sites"), py::arg("ncomp") = 24);
  m.def("clover_apply", &clover_apply, "clover term apply",
        py::arg("out"), py::arg("out_n"), py::arg("in"), py::arg("in_n"),
        py::arg("clover"), py::arg("parity"), py::arg("Vcb"),
        py::arg("v_stride") = 0, py::arg("s_offset") = 0);
  m.def("twist_apply", &twist_apply, "twisted-mass T(b) apply",
        py::arg("out"), py::arg("out_n"), py::arg("in"), py::arg("in_n"),
        py::arg("b_re"), py::arg("b_im"), py::arg("Vcb"), py::arg("sites"),
        py::arg("tau3_vcb") = 0, py::arg("acc") = false);
  m.attr("BLAS_AXPY") = (int)BLAS_AXPY;
  m.attr("BLAS_AXPY_NORM2") = (int)BLAS_AXPY_NORM2;
  m.attr("BLAS_XPAY") = (int)BLAS_XPAY;
  m.attr("BLAS_AXPBY") = (int)BLAS_AXPBY;
  m.attr("BLAS_CAXPY") = (int)BLAS_CAXPY;
  m.attr("BLAS_XMY_NORM2") = (int)BLAS_XMY_NORM2;
  m.attr("BLAS_SCAL") = (int)BLAS_SCAL;
  m.attr("BLAS_NORM2") = (int)BLAS_NORM2;
  m.attr("BLAS_TRIPLE_CG") = (int)BLAS_TRIPLE_CG;
  m.attr("BLAS_REDOT") = (int)BLAS_REDOT;
  m.attr("BLAS_CDOT") = (int)BLAS_CDOT;
  m.attr("BLAS_CAXPBY") = (int)BLAS_CAXPBY;
}
