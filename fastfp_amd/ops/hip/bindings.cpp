// PyTorch bindings for the fastfp_amd fp64 HIP kernels (MI355X/gfx950).
//
// Thin shape/contiguity-checked wrappers; all launches go onto the
// current torch HIP stream so the ops compose with torch eager code.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

extern "C" {
void launch_sigdots(const double*, const double*, const double*,
                    const double*, int, int, double*, double*, hipStream_t);
void launch_sbgemm(const double*, const double*, const double*,
                   const double*, int, int, int, int, double*, long, long,
                   int, hipStream_t);
void launch_chol_batch(const double*, const double*, int, int, int, int,
                       double*, double*, hipStream_t);
void launch_trsm_fp(const double*, const double*, const double*,
                    const double*, const double*, int, int, int, int,
                    double, double*, hipStream_t);
void launch_sigdots_block(const double*, const double*, const double*,
                          const double*, const double*, const long*,
                          const long*, int, int, int, double*, double*,
                          hipStream_t);
void launch_diag_inv(const double*, int, long, double*, hipStream_t);
}

namespace {

void check_f64(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on the GPU");
  TORCH_CHECK(t.scalar_type() == torch::kFloat64, name, " must be fp64");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

hipStream_t stream() {
  return at::hip::getCurrentHIPStream().stream();
}

}  // namespace

// sigdots: (toas, ninv, nr (ntoa,), freqs (F,)) -> sNs (3,F), sNr (2,F)
std::vector<torch::Tensor> sigdots(torch::Tensor toas, torch::Tensor ninv,
                                   torch::Tensor nr, torch::Tensor freqs) {
  check_f64(toas, "toas");
  check_f64(ninv, "ninv");
  check_f64(nr, "nr");
  check_f64(freqs, "freqs");
  const int ntoa = toas.size(0);
  const int F = freqs.size(0);
  auto opts = toas.options();
  auto sNs = torch::empty({3, F}, opts);
  auto sNr = torch::empty({2, F}, opts);
  launch_sigdots(toas.data_ptr<double>(), ninv.data_ptr<double>(),
                 nr.data_ptr<double>(), freqs.data_ptr<double>(), ntoa, F,
                 sNs.data_ptr<double>(), sNr.data_ptr<double>(), stream());
  return {sNs, sNr};
}

// sbgemm into `out` (plane-strided), out[j, col0 + c] over F2 columns.
void sbgemm(torch::Tensor T, torch::Tensor toas, torch::Tensor ninv,
            torch::Tensor freqs, torch::Tensor out, int64_t plane_stride,
            int64_t ldo, int64_t mp, int64_t ksplit) {
  check_f64(T, "T");
  check_f64(toas, "toas");
  check_f64(ninv, "ninv");
  check_f64(freqs, "freqs");
  check_f64(out, "out");
  const int ntoa = T.size(0);
  const int m = T.size(1);
  const int F2 = 2 * freqs.size(0);
  TORCH_CHECK(mp % 16 == 0 && mp >= m, "bad mp");  // M-tiled: no upper cap
  launch_sbgemm(T.data_ptr<double>(), toas.data_ptr<double>(),
                ninv.data_ptr<double>(), freqs.data_ptr<double>(), ntoa, m,
                (int)mp, F2, out.data_ptr<double>(), plane_stride, ldo,
                (int)ksplit, stream());
}

// chol_batch: TNT (m,m) or (P,m,m), phiinv (D,m) or (P,D,m)
// -> L (P*D,mp,mp), invd (P*D,mp/16,16,16)
std::vector<torch::Tensor> chol_batch(torch::Tensor TNT, torch::Tensor phiinv,
                                      int64_t mp) {
  check_f64(TNT, "TNT");
  check_f64(phiinv, "phiinv");
  const bool batched = TNT.dim() == 3;
  const int P = batched ? TNT.size(0) : 1;
  const int m = TNT.size(batched ? 1 : 0);
  const int D = phiinv.size(batched ? 1 : 0);
  TORCH_CHECK(phiinv.dim() == (batched ? 3 : 2), "phiinv rank");
  TORCH_CHECK(phiinv.size(batched ? 2 : 1) == m, "phiinv width != m");
  TORCH_CHECK(mp % 16 == 0 && mp <= 128 && mp >= m,
              "chol_batch requires m <= 128 (solve dimension); got m=", m);
  auto opts = TNT.options();
  auto L = torch::empty({(long)P * D, mp, mp}, opts);
  auto invd = torch::empty({(long)P * D, mp / 16, 16, 16}, opts);
  launch_chol_batch(TNT.data_ptr<double>(), phiinv.data_ptr<double>(), m,
                    (int)mp, D, P, L.data_ptr<double>(),
                    invd.data_ptr<double>(), stream());
  return {L, invd};
}

// chol_batch_into: same as chol_batch but writing into caller-owned
// L/invd (the engine's ping-pong pipeline buffers — avoids per-chunk
// allocator churn across streams).
void chol_batch_into(torch::Tensor TNT, torch::Tensor phiinv, int64_t mp,
                     torch::Tensor L, torch::Tensor invd) {
  check_f64(TNT, "TNT");
  check_f64(phiinv, "phiinv");
  check_f64(L, "L");
  check_f64(invd, "invd");
  const bool batched = TNT.dim() == 3;
  const int P = batched ? TNT.size(0) : 1;
  const int m = TNT.size(batched ? 1 : 0);
  const int D = phiinv.size(batched ? 1 : 0);
  TORCH_CHECK(phiinv.dim() == (batched ? 3 : 2), "phiinv rank");
  TORCH_CHECK(phiinv.size(batched ? 2 : 1) == m, "phiinv width != m");
  TORCH_CHECK(mp % 16 == 0 && mp <= 128 && mp >= m, "mp");
  TORCH_CHECK(L.size(0) == (long)P * D && L.size(1) == mp && L.size(2) == mp,
              "L shape");
  TORCH_CHECK(invd.size(0) == (long)P * D && invd.size(1) == mp / 16,
              "invd shape");
  launch_chol_batch(TNT.data_ptr<double>(), phiinv.data_ptr<double>(), m,
                    (int)mp, D, P, L.data_ptr<double>(),
                    invd.data_ptr<double>(), stream());
}

// trsm_fp_accum: single pulsar (L (D,mp,mp), RHS (mp,2F+1), sNs (3,F),
// fp (D,F)) or pulsar-batched (L (P*D,mp,mp), RHS (P,mp,2F+1),
// sNs (P,3,F), fp (P,D,F)); accumulates in place.
void trsm_fp_accum(torch::Tensor L, torch::Tensor invd, torch::Tensor RHS,
                   torch::Tensor sNs, torch::Tensor sNr, torch::Tensor fp,
                   double gsign) {
  check_f64(L, "L");
  check_f64(invd, "invd");
  check_f64(RHS, "RHS");
  check_f64(sNs, "sNs");
  check_f64(sNr, "sNr");
  check_f64(fp, "fp");
  const bool batched = RHS.dim() == 3;
  const int P = batched ? RHS.size(0) : 1;
  const int mp = L.size(1);
  const int F = sNs.size(batched ? 2 : 1);
  const int D = fp.size(batched ? 1 : 0);
  TORCH_CHECK(L.size(0) == (long)P * D, "L batch != P*D");
  TORCH_CHECK(RHS.size(batched ? 1 : 0) == mp &&
                  RHS.size(batched ? 2 : 1) == 2 * F + 1,
              "RHS shape");
  TORCH_CHECK(fp.dim() == (batched ? 3 : 2) && fp.size(batched ? 2 : 1) == F,
              "fp shape");
  TORCH_CHECK(mp % 16 == 0 && mp <= 256, "mp must be <=256, multiple of 16");
  launch_trsm_fp(L.data_ptr<double>(), invd.data_ptr<double>(),
                 RHS.data_ptr<double>(), sNs.data_ptr<double>(),
                 sNr.data_ptr<double>(), mp, F, D, P, gsign,
                 fp.data_ptr<double>(), stream());
}

// diag_inv: invert the 16x16 diagonal blocks of batched lower-tri L
// (B, mp, mp) -> invd (B, mp/16, 16, 16); the m > 128 direct path.
torch::Tensor diag_inv(torch::Tensor L) {
  check_f64(L, "L");
  TORCH_CHECK(L.dim() == 3 && L.size(1) == L.size(2), "L must be (B,mp,mp)");
  const long B = L.size(0);
  const int mp = L.size(1);
  TORCH_CHECK(mp % 16 == 0, "mp multiple of 16");
  auto invd = torch::empty({B, mp / 16, 16, 16}, L.options());
  launch_diag_inv(L.data_ptr<double>(), mp, B * (mp / 16),
                  invd.data_ptr<double>(), stream());
  return invd;
}

// sigdots_block: Sherman-Morrison block-diagonal-N dots.
// toas/uvec/nr (ntoa, permuted), beta (nblk), offsets/sizes (int64).
std::vector<torch::Tensor> sigdots_block(torch::Tensor toas, torch::Tensor uvec,
                                         torch::Tensor nr, torch::Tensor freqs,
                                         torch::Tensor beta,
                                         torch::Tensor offsets,
                                         torch::Tensor sizes) {
  check_f64(toas, "toas");
  check_f64(uvec, "uvec");
  check_f64(nr, "nr");
  check_f64(freqs, "freqs");
  check_f64(beta, "beta");
  const int ntoa = toas.size(0);
  const int F = freqs.size(0);
  const int nblk = sizes.size(0);
  auto opts = toas.options();
  auto sNs = torch::empty({3, F}, opts);
  auto sNr = torch::empty({2, F}, opts);
  launch_sigdots_block(toas.data_ptr<double>(), uvec.data_ptr<double>(),
                       nr.data_ptr<double>(), freqs.data_ptr<double>(),
                       beta.data_ptr<double>(), offsets.data_ptr<long>(),
                       sizes.data_ptr<long>(), nblk, ntoa, F,
                       sNs.data_ptr<double>(), sNr.data_ptr<double>(),
                       stream());
  return {sNs, sNr};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("sigdots_block", &sigdots_block,
        "fused sincos dots with block-diagonal N (Sherman-Morrison)");
  m.def("diag_inv", &diag_inv,
        "invert 16x16 diagonal blocks of batched lower-triangular L");
  m.def("sigdots", &sigdots, "fused sincos signal-basis dots");
  m.def("sbgemm", &sbgemm, "fused signal-basis MFMA fp64 DGEMM");
  m.def("chol_batch", &chol_batch, "batched LDS-resident fp64 Cholesky");
  m.def("chol_batch_into", &chol_batch_into,
        "chol_batch into caller-owned L/invd (pipeline buffers)");
  m.def("trsm_fp_accum", &trsm_fp_accum,
        "batched TRSM + fused 2x2 Fp reduction");
}
