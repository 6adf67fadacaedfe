// fastfp_amd fp64 HIP kernels for MI355X (gfx950, CDNA4).
//
// Hand-written from scratch for the restructured Fp-statistic math
// (docs/DESIGN.md) — NOT a port of the reference's JAX
// graph (the reference has zero native code).  The
// implicit XLA op surface being replaced is catalogued in SURVEY.md
// §2.2; each kernel below names the reference call sites it subsumes.
//
// Kernel inventory:
//   sigdots_kernel     — fused sin/cos signal basis + diagonal-weighted
//                        dots sNs/sNr (subsumes fastfp/fastfp.py:78-79 +
//                        utils.py:49-53 elementwise/dot ops)
//   sbgemm_kernel      — MFMA fp64 DGEMM B = T^T (N^-1 S) with the
//                        sin/cos S-panel GENERATED in LDS (never
//                        materialized in HBM) and LDS-staged T panels
//                        (subsumes utils.py:51-52 GEMV batched over all
//                        frequencies)
//   chol_batch_kernel  — batched Cholesky of Sigma = TNT + diag(phi^-1),
//                        assembled in LDS (never materialized in HBM),
//                        MFMA trailing updates, inverted 16x16 diagonal
//                        blocks exported for the solver (subsumes
//                        utils.py:76 + the LU in utils.py:54)
//   trsm_fp_kernel     — batched blocked forward-substitution
//                        W = L^-1 [B | TNr] fused with the per-frequency
//                        2x2 Gram/solve/Fp reduction; W lives entirely
//                        in LDS and never touches HBM (subsumes
//                        utils.py:54 + fastfp.py:81-90)
//
// Conventions:
//   wave = 64 lanes (CDNA);  MFMA v_mfma_f64_16x16x4f64:
//     A[i][k]: lane l holds A[i = l&15][k = l>>4]
//     B[k][j]: lane l holds B[k = l>>4][j = l&15]
//     D/C    : lane l holds rows 4*v + (l>>4) (v=0..3), col l&15 (probed: tools/mfma_probe.hip)
//   mp = m padded to a multiple of 16, mp <= 128.  Pad rows of Sigma
//   are identity (L pad = I) and pad rows of RHS are zero, so padding
//   never changes results.
//   All reductions have fixed order -> bitwise-deterministic fp64.

#include <hip/hip_runtime.h>
#include <math.h>
#include <stdlib.h>

#define FASTFP_MAXMP 128
#define NB 16  // Cholesky/TRSM block size (one MFMA tile)

typedef double f64x4 __attribute__((ext_vector_type(4)));

#define MFMA_F64(a, b, c) __builtin_amdgcn_mfma_f64_16x16x4f64((a), (b), (c), 0, 0, 0)

static __device__ __forceinline__ double wave_reduce_sum(double v) {
  // fixed-order binary tree over the 64-lane wave
  for (int off = 32; off > 0; off >>= 1)
    v += __shfl_down(v, off, 64);
  return v;
}

// ---------------------------------------------------------------------
// sigdots: per-frequency dots  s^T N^-1 s, c^T N^-1 c, s^T N^-1 c,
//          s^T N^-1 r, c^T N^-1 r   (amplitude f^-1/3 omitted: it
//          cancels in Fp — docs/DESIGN.md §2)
// grid.x = F, block = 256
// ---------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(256) void sigdots_kernel(
    const double* __restrict__ toas, const double* __restrict__ ninv,
    const double* __restrict__ nr, const double* __restrict__ freqs,
    int ntoa, int F, double* __restrict__ sNs /*(3,F)*/,
    double* __restrict__ sNr /*(2,F)*/) {
  const int f = blockIdx.x;
  if (f >= F) return;
  const double w = 2.0 * M_PI * freqs[f];
  double ss = 0, cc = 0, sc = 0, sr = 0, cr = 0;
  for (int i = threadIdx.x; i < ntoa; i += blockDim.x) {
    double s, c;
    sincos(w * toas[i], &s, &c);
    const double ni = ninv[i];
    const double ri = nr[i];
    ss = fma(s * s, ni, ss);
    cc = fma(c * c, ni, cc);
    sc = fma(s * c, ni, sc);
    sr = fma(s, ri, sr);
    cr = fma(c, ri, cr);
  }
  __shared__ double red[4][5];
  ss = wave_reduce_sum(ss);
  cc = wave_reduce_sum(cc);
  sc = wave_reduce_sum(sc);
  sr = wave_reduce_sum(sr);
  cr = wave_reduce_sum(cr);
  const int lane = threadIdx.x & 63, wv = threadIdx.x >> 6;
  if (lane == 0) {
    red[wv][0] = ss; red[wv][1] = cc; red[wv][2] = sc;
    red[wv][3] = sr; red[wv][4] = cr;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    double a0 = 0, a1 = 0, a2 = 0, a3 = 0, a4 = 0;
    for (int v = 0; v < 4; ++v) {
      a0 += red[v][0]; a1 += red[v][1]; a2 += red[v][2];
      a3 += red[v][3]; a4 += red[v][4];
    }
    sNs[f] = a0; sNs[F + f] = a1; sNs[2 * F + f] = a2;
    sNr[f] = a3; sNr[F + f] = a4;
  }
}

// ---------------------------------------------------------------------
// sbgemm: out[j, c] = sum_k T[k, j] * trig_c(toas[k]) * ninv[k]
//   trig_c = sin(2 pi f_{c/2} t) for even c, cos for odd c.
// The S-panel is generated on the fly into LDS (fused signal basis);
// the T-panel is LDS-staged.  Output is the (mp x 2F) RHS block,
// written directly (ksplit==1, ld = ldo) or into per-split partial
// planes (deterministic torch reduction afterwards).
// grid.x = ceil(2F / 64) col tiles, grid.y = ksplit;  block = 512 (8 waves)
// ---------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(512) void sbgemm_kernel(
    const double* __restrict__ T /*(ntoa, m) row-major*/,
    const double* __restrict__ toas, const double* __restrict__ ninv,
    const double* __restrict__ freqs, int ntoa, int m, int mp, int F2,
    double* __restrict__ out, long plane_stride, long ldo) {
  // 8 waves: 4 column strips x 2 row halves.  The 4-wave variant held
  // 8 accumulator tiles per wave (64+ AGPR) which capped occupancy at
  // 3 waves/SIMD; splitting the M dim halves the accumulator and lets
  // more waves cover the sincos + LDS latency.
  // grid.z tiles the M dimension in blocks of 128 so the basis size is
  // unbounded (GP-ECORR models run to many hundreds of columns).
  __shared__ double lT[16][FASTFP_MAXMP + 1];
  __shared__ double lS[64][17];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wv = tid >> 6;
  const int c0 = blockIdx.x * 64;       // first output column of this tile
  const int mbase = blockIdx.z * FASTFP_MAXMP;  // M-tile origin
  const int mp_loc = min(FASTFP_MAXMP, mp - mbase);
  const int jw = (wv & 3) * 16;         // column strip
  const int rh = wv >> 2;               // row half (0: rows 0-63, 1: 64-127)
  const int rowbase = rh * 64;
  const int nrt_tot = mp_loc >> 4;
  // row tiles this wave owns (half 1 may be empty for small mp)
  const int rt_lo = min(rh * 4, nrt_tot);
  const int rt_hi = min(rt_lo + 4, nrt_tot);

  // K range of this split
  const int ks = gridDim.y;
  const int kchunk = (ntoa + ks - 1) / ks;
  const int kbeg = blockIdx.y * kchunk;
  const int kend = min(ntoa, kbeg + kchunk);
  double* outp = out + (long)blockIdx.y * plane_stride;

  f64x4 acc[4];
#pragma unroll
  for (int q = 0; q < 4; ++q) acc[q] = f64x4{0, 0, 0, 0};

  for (int k0 = kbeg; k0 < kend; k0 += 16) {
    // stage T panel rows k0..k0+15, cols mbase.. (zero-padded)
    for (int idx = tid; idx < 16 * mp_loc; idx += 512) {
      const int k = idx / mp_loc, j = idx % mp_loc;
      const int gk = k0 + k;
      const int gj = mbase + j;
      lT[k][j] = (gk < kend && gj < m) ? T[(long)gk * m + gj] : 0.0;
    }
    // stage trig panel: 32 freq-pairs x 16 toas, one sincos each
    for (int idx = tid; idx < 32 * 16; idx += 512) {
      const int p = idx / 16, k = idx % 16;
      const int gk = k0 + k;
      const int ceven = c0 + 2 * p;
      double sv = 0.0, cv = 0.0;
      if (gk < kend && ceven < F2) {
        const double wf = 2.0 * M_PI * freqs[ceven >> 1];
        sincos(wf * toas[gk], &sv, &cv);
        const double ni = ninv[gk];
        sv *= ni; cv *= ni;
      }
      lS[2 * p][k] = sv;
      lS[2 * p + 1][k] = cv;
    }
    __syncthreads();

#pragma unroll
    for (int kk = 0; kk < 4; ++kk) {
      const double b = lS[jw + (lane & 15)][kk * 4 + (lane >> 4)];
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        if (rt_lo + q >= rt_hi) break;
        const double a =
            lT[kk * 4 + (lane >> 4)][rowbase + q * 16 + (lane & 15)];
        acc[q] = MFMA_F64(a, b, acc[q]);
      }
    }
    __syncthreads();
  }

  // epilogue: store (row j, col c) — cols contiguous, coalesced
#pragma unroll
  for (int q = 0; q < 4; ++q) {
    if (rt_lo + q >= rt_hi) break;
#pragma unroll
    for (int v = 0; v < 4; ++v) {
      const int row = mbase + rowbase + q * 16 + 4 * v + (lane >> 4);
      const int col = c0 + jw + (lane & 15);
      if (col < F2) outp[(long)row * ldo + col] = acc[q][v];
    }
  }
}

// ---------------------------------------------------------------------
// sbgemm_db: DOUBLE-BUFFERED variant of sbgemm — the next K-tile's T
// panel + generated trig panel are staged while the current tile's
// MFMAs run, so there is ONE barrier per K-tile instead of two and
// the global loads + the ~100-cycle fp64 sincos hide under the MFMA
// phase.  LDS doubles to ~50 KB (still 2+ workgroups/CU at the
// 112-VGPR occupancy).  A/B arm: FASTFP_SBGEMM_ALGO=db.
// ---------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(512) void sbgemm_db_kernel(
    const double* __restrict__ T /*(ntoa, m) row-major*/,
    const double* __restrict__ toas, const double* __restrict__ ninv,
    const double* __restrict__ freqs, int ntoa, int m, int mp, int F2,
    double* __restrict__ out, long plane_stride, long ldo) {
  __shared__ double lT[2][16][FASTFP_MAXMP + 1];
  __shared__ double lS[2][64][17];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wv = tid >> 6;
  const int c0 = blockIdx.x * 64;
  const int mbase = blockIdx.z * FASTFP_MAXMP;
  const int mp_loc = min(FASTFP_MAXMP, mp - mbase);
  const int jw = (wv & 3) * 16;
  const int rh = wv >> 2;
  const int rowbase = rh * 64;
  const int nrt_tot = mp_loc >> 4;
  const int rt_lo = min(rh * 4, nrt_tot);
  const int rt_hi = min(rt_lo + 4, nrt_tot);

  const int ks = gridDim.y;
  const int kchunk = (ntoa + ks - 1) / ks;
  const int kbeg = blockIdx.y * kchunk;
  const int kend = min(ntoa, kbeg + kchunk);
  double* outp = out + (long)blockIdx.y * plane_stride;

  f64x4 acc[4];
#pragma unroll
  for (int q = 0; q < 4; ++q) acc[q] = f64x4{0, 0, 0, 0};

  // staging of K-tile k0 into buffer b
#define SB_STAGE(k0, b)                                                    \
  {                                                                        \
    for (int idx = tid; idx < 16 * mp_loc; idx += 512) {                   \
      const int k = idx / mp_loc, j = idx % mp_loc;                        \
      const int gk = (k0) + k;                                             \
      const int gj = mbase + j;                                            \
      lT[b][k][j] = (gk < kend && gj < m) ? T[(long)gk * m + gj] : 0.0;    \
    }                                                                      \
    for (int idx = tid; idx < 32 * 16; idx += 512) {                       \
      const int p = idx / 16, k = idx % 16;                                \
      const int gk = (k0) + k;                                             \
      const int ceven = c0 + 2 * p;                                        \
      double sv = 0.0, cv = 0.0;                                           \
      if (gk < kend && ceven < F2) {                                       \
        const double wf = 2.0 * M_PI * freqs[ceven >> 1];                  \
        sincos(wf * toas[gk], &sv, &cv);                                   \
        const double ni = ninv[gk];                                        \
        sv *= ni; cv *= ni;                                                \
      }                                                                    \
      lS[b][2 * p][k] = sv;                                                \
      lS[b][2 * p + 1][k] = cv;                                            \
    }                                                                      \
  }

  SB_STAGE(kbeg, 0);
  __syncthreads();
  int buf = 0;
  for (int k0 = kbeg; k0 < kend; k0 += 16) {
    if (k0 + 16 < kend) SB_STAGE(k0 + 16, buf ^ 1);
#pragma unroll
    for (int kk = 0; kk < 4; ++kk) {
      const double b = lS[buf][jw + (lane & 15)][kk * 4 + (lane >> 4)];
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        if (rt_lo + q >= rt_hi) break;
        const double a =
            lT[buf][kk * 4 + (lane >> 4)][rowbase + q * 16 + (lane & 15)];
        acc[q] = MFMA_F64(a, b, acc[q]);
      }
    }
    buf ^= 1;
    __syncthreads();
  }
#undef SB_STAGE

#pragma unroll
  for (int q = 0; q < 4; ++q) {
    if (rt_lo + q >= rt_hi) break;
#pragma unroll
    for (int v = 0; v < 4; ++v) {
      const int row = mbase + rowbase + q * 16 + 4 * v + (lane >> 4);
      const int col = c0 + jw + (lane & 15);
      if (col < F2) outp[(long)row * ldo + col] = acc[q][v];
    }
  }
}

// ---------------------------------------------------------------------
// chol_batch: per draw d, assemble Sigma = TNT + diag(phiinv[d]) (+
// identity padding to mp) in LDS, factor L L^T = Sigma (right-looking,
// NB=16 blocks, MFMA trailing updates), invert each 16x16 diagonal
// block, write L (D, mp, mp) and invdiag (D, mp/16, 16, 16).
// grid.x = D, block = 512 (8 waves).  The 16x16 diagonal factor and its
// inversion run REGISTER-resident in wave 0 with cross-lane __shfl
// (width 16) -- the LDS read-modify-write chains of the naive version
// were the kernel's dominant cost (profiles/r01_initial_stats.md).
// ---------------------------------------------------------------------
// (A (+i)-rotation column swizzle in place of the +1 row pad would
// shave Ash from 33.3 to 32.0 KB at mp=64 — one more workgroup per CU
// — but the extra address arithmetic measured +16 VGPR, dropping a
// waves/SIMD tier: net zero.  Padded layout kept.)
//
// TRI (FASTFP_CHOL_TRI=1, NBT <= 4): LOWER-TRIANGLE-packed Sigma —
// row i holds i+1 entries (+1 pad), halving LDS (33.3 -> 17.4 KB at
// mp=64) so ~7 workgroups fit per CU instead of 4; concurrency is the
// suspected chol bottleneck (82% parked waves resist every other
// explanation tested).  All factor-phase accesses are at-or-below the
// diagonal; the diagonal tile's upper half is read through the
// symmetric mirror AS_, and trailing stores into diagonal tiles are
// lower-guarded.
#define A_(i, j)                                                        \
  Ash[TRI ? ((i) * ((i) + 3) / 2 + (j)) : ((i) * (NBT * 16 + 1) + (j))]
#define AS_(i, j) ((j) <= (i) ? A_(i, j) : A_(j, i))

// (Forcing a 6-waves/SIMD allocation target on the small shapes got
// there only by spilling 44 B/lane and measured neutral — reverted;
// the no-spill allocation already reaches 5 waves with TRI.)
template <int NBT, bool TRI = false>
__global__ __launch_bounds__(512, 4) void chol_batch_kernel(
    const double* __restrict__ TNT_all /*(P,m,m)*/,
    const double* __restrict__ phiinv_all /*(P,D,m)*/, int m, int D,
    double* __restrict__ L_all /*(P*D,mp,mp)*/,
    double* __restrict__ invd_all /*(P*D, mp/16, 16, 16)*/) {
  // grid.y batches PULSARS: one launch factors every (pulsar, draw)
  const int pp = blockIdx.y;
  const double* TNT = TNT_all + (long)pp * m * m;
  const double* phiinv = phiinv_all + (long)pp * D * m;
  double* L = L_all + (long)pp * D * (NBT * 16) * (NBT * 16);
  double* invd = invd_all + (long)pp * D * NBT * 256;
  // LDS is sized by the template so small matrices keep multiple
  // workgroups per CU (nb=4: 33 KB -> 4 WG/CU vs one at 134 KB; the
  // serial diagonal phases then overlap ACROSS workgroups -- the PMC
  // profile showed 84% of wave cycles parked, profiles/).
  constexpr int mp = NBT * 16;
  __shared__ double Ash[TRI ? mp * (mp + 3) / 2 : mp * (mp + 1)];
  __shared__ double inv16[2][16][17];

  const int d = blockIdx.x;
  if (d >= D) return;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wv = tid >> 6;
  constexpr int nb = NBT;

  const int nthr = blockDim.x;
  const int nwv = nthr >> 6;
  // assemble Sigma in LDS (lower triangle only under TRI)
  for (int idx = tid; idx < mp * mp; idx += nthr) {
    const int i = idx / mp, j = idx % mp;
    if (TRI && j > i) continue;
    double v = (i < m && j < m) ? TNT[(long)i * m + j] : 0.0;
    if (i == j) v += (i < m) ? phiinv[(long)d * m + i] : 1.0;
    A_(i, j) = v;
  }
  __syncthreads();

  // Look-ahead right-looking Cholesky: the serial 16x16 diagonal
  // factor of column kb+1 (wave 0, register/shfl) runs CONCURRENTLY
  // with the MFMA trailing updates of columns kb+2.. (waves 1..7) --
  // the PMC profile showed 84-88% of wave cycles parked waiting on the
  // serial diagonal.  Two inverted-diagonal buffers ping-pong by kb
  // parity.
#define CHOL_DIAG(KB, DST)                                                   \
  {                                                                          \
    const int dk0 = (KB)*NB;                                                 \
    const int i = lane & 15;                                                 \
    double row[16];                                                          \
    _Pragma("unroll") for (int c = 0; c < 16; ++c) row[c] =                  \
        AS_(dk0 + i, dk0 + c);                                               \
    _Pragma("unroll") for (int t = 0; t < 16; ++t) {                         \
      const double dv = sqrt(__shfl(row[t], t, 16));                         \
      const double rdv = 1.0 / dv;                                           \
      if (i > t) row[t] *= rdv;                                              \
      else if (i == t) row[t] = dv;                                          \
      _Pragma("unroll") for (int j = t + 1; j < 16; ++j) {                   \
        const double ljt = __shfl(row[t], j, 16);                            \
        if (i >= j) row[j] = fma(-row[t], ljt, row[j]);                      \
      }                                                                      \
    }                                                                        \
    _Pragma("unroll") for (int c = 0; c < 16; ++c) if (c <= i)               \
        A_(dk0 + i, dk0 + c) = row[c];                                       \
    const int c = i;                                                         \
    double x[16];                                                            \
    /* lane i computes only 1/L_ii; the rest arrive by shfl (16x less   */   \
    /* serial f64 division work than every lane inverting all 16).      */   \
    /* predicated select avoids a dynamic register index (scratch!)     */   \
    /* the reciprocal broadcast is inlined per r (a diag[16] register   */   \
    /* array cost ~16 VGPR and a waves/SIMD tier)                       */   \
    double dii = 0.0;                                                        \
    _Pragma("unroll") for (int r = 0; r < 16; ++r) if (r == i) dii = row[r]; \
    const double myrcp = 1.0 / dii;                                          \
    _Pragma("unroll") for (int r = 0; r < 16; ++r) {                         \
      const double dr = __shfl(myrcp, r, 16);                                \
      double acc2 = 0.0;                                                     \
      _Pragma("unroll") for (int t = 0; t < 16; ++t) {                       \
        const double lrt = __shfl(row[t], r, 16);                            \
        if (t >= c && t < r) acc2 = fma(lrt, x[t], acc2);                    \
      }                                                                      \
      x[r] = (r < c) ? 0.0 : (r == c) ? dr : -acc2 * dr;                     \
    }                                                                        \
    _Pragma("unroll") for (int r = 0; r < 16; ++r) (DST)[r][c] = x[r];       \
    if (lane < 16)                                                           \
      _Pragma("unroll") for (int r = 0; r < 16; ++r)                         \
          invd[((long)d * nb + (KB)) * 256 + r * 16 + c] = x[r];             \
  }

  // prologue: factor + invert diagonal block 0
  if (wv == 0) CHOL_DIAG(0, inv16[0]);
  __syncthreads();

  for (int kb = 0; kb < nb; ++kb) {
    const int k0 = kb * NB;

    // panel TRSM: row tiles below the diagonal, P <- P * inv(L_kk)^T
    for (int rt = kb + 1 + wv; rt < nb; rt += nwv) {
      const int r0 = rt * NB;
      f64x4 pacc = {0, 0, 0, 0};
#pragma unroll
      for (int kk = 0; kk < 4; ++kk) {
        const double a = A_(r0 + (lane & 15), k0 + kk * 4 + (lane >> 4));
        const double b = inv16[kb & 1][lane & 15][kk * 4 + (lane >> 4)];
        pacc = MFMA_F64(a, b, pacc);
      }
#pragma unroll
      for (int v = 0; v < 4; ++v)
        A_(r0 + 4 * v + (lane >> 4), k0 + (lane & 15)) = pacc[v];
    }
    __syncthreads();
    if (kb + 1 >= nb) break;

    // trailing phase A1: ONLY column kb+1 tiles, so its diagonal can
    // factor next (all 8 waves)
    const int j1 = (kb + 1) * NB;
    for (int ib = kb + 1 + wv; ib < nb; ib += nwv) {
      const int i0 = ib * NB;
      f64x4 uacc;
      // diagonal tile (i0 == j1): upper half read via the symmetric
      // mirror under TRI, and its stores lower-guarded
#pragma unroll
      for (int v = 0; v < 4; ++v)
        uacc[v] = AS_(i0 + 4 * v + (lane >> 4), j1 + (lane & 15));
#pragma unroll
      for (int kk = 0; kk < 4; ++kk) {
        const double a = -A_(i0 + (lane & 15), k0 + kk * 4 + (lane >> 4));
        const double b = A_(j1 + (lane & 15), k0 + kk * 4 + (lane >> 4));
        uacc = MFMA_F64(a, b, uacc);
      }
#pragma unroll
      for (int v = 0; v < 4; ++v)
        if (!TRI || i0 + 4 * v + (lane >> 4) >= j1 + (lane & 15))
          A_(i0 + 4 * v + (lane >> 4), j1 + (lane & 15)) = uacc[v];
    }
    __syncthreads();

    // phase A2: wave 0 factors diagonal kb+1 while the other waves do
    // the remaining trailing tiles (jb >= kb+2).  (Rotating the factor
    // wave across kb to spread the serial phase over SIMDs was
    // measured NEUTRAL — the kernel's parking is not SIMD-0 contention
    // — and was reverted.)
    if (wv == 0) {
      CHOL_DIAG(kb + 1, inv16[(kb + 1) & 1]);
    } else {
      const int t = nb - kb - 2;
      const int ntile = t * (t + 1) / 2;
      for (int q = wv - 1; q < ntile; q += nwv - 1) {
        int ib = kb + 2, rem = q;
        while (rem > ib - kb - 2) { rem -= (ib - kb - 1); ++ib; }
        const int jb = kb + 2 + rem;
        const int i0 = ib * NB, j0 = jb * NB;
        f64x4 uacc;
#pragma unroll
        for (int v = 0; v < 4; ++v)
          uacc[v] = AS_(i0 + 4 * v + (lane >> 4), j0 + (lane & 15));
#pragma unroll
        for (int kk = 0; kk < 4; ++kk) {
          const double a = -A_(i0 + (lane & 15), k0 + kk * 4 + (lane >> 4));
          const double b = A_(j0 + (lane & 15), k0 + kk * 4 + (lane >> 4));
          uacc = MFMA_F64(a, b, uacc);
        }
#pragma unroll
        for (int v = 0; v < 4; ++v)
          if (!TRI || i0 + 4 * v + (lane >> 4) >= j0 + (lane & 15))
            A_(i0 + 4 * v + (lane >> 4), j0 + (lane & 15)) = uacc[v];
      }
    }
    __syncthreads();
  }

  // write back L (full rows; upper-triangle junk is never read — TRI
  // writes deterministic zeros there)
  for (int idx = tid; idx < mp * mp; idx += nthr) {
    const int i = idx / mp, j = idx % mp;
    L[((long)d * mp + i) * mp + j] = (TRI && j > i) ? 0.0 : A_(i, j);
  }
}

// ---------------------------------------------------------------------
// chol_wave: fully IN-WAVE Cholesky for mp <= 64 — one DRAW per wave,
// rows as lanes, the whole factor in registers with width-64 __shfl
// broadcasts.  No LDS, no barriers, no cross-wave dependencies: every
// wave is busy on its own draw, attacking the blocked kernel's
// 82%-parked-waves structure at its root.  The diagonal-block
// inverses come from the separate diag_inv kernel.  The t/j loops are
// fully unrolled so every a[] index is compile-time (runtime indices
// demote the register array to scratch).
//
// MEASURED NEGATIVE (kept as the record): 4.90 ms vs the blocked
// kernel's 3.04 ms at the bench shape.  The ~2,080 serial-ish __shfl
// broadcasts per draw (DS-pipe) plus 209 VGPR -> 2 waves/SIMD swamp
// the parking win; the blocked kernel's MFMA trailing updates do the
// same O(mp^3) work in far fewer issue slots.  Numerics validated
// (7 GPU tests green under FASTFP_CHOL_ALGO=wave).
// grid = (ceil(D/4), P), block = 256 (4 waves).  A/B arm only.
// ---------------------------------------------------------------------
template <int MP>
__global__ __launch_bounds__(256) void chol_wave_kernel(
    const double* __restrict__ TNT_all /*(P,m,m)*/,
    const double* __restrict__ phiinv_all /*(P,D,m)*/, int m, int D,
    double* __restrict__ L_all /*(P*D,mp,mp)*/) {
  const int pp = blockIdx.y;
  const int d = blockIdx.x * 4 + (threadIdx.x >> 6);
  if (d >= D) return;
  const int i = threadIdx.x & 63;  // row owned by this lane
  const double* TNT = TNT_all + (long)pp * m * m;
  const double* phiinv = phiinv_all + ((long)pp * D + d) * m;
  double* L = L_all + ((long)pp * D + d) * MP * MP;

  // row i of Sigma (lower triangle only; padding rows are identity)
  double a[MP];
#pragma unroll
  for (int j = 0; j < MP; ++j) {
    double v = 0.0;
    if (i < m && j < m && j <= i) v = TNT[(long)i * m + j];
    if (j == i) v += (i < m) ? phiinv[i] : 1.0;
    a[j] = (i < MP) ? v : 0.0;
  }

  // unblocked right-looking factor, rows-as-lanes
#pragma unroll
  for (int t = 0; t < MP; ++t) {
    const double att = __shfl(a[t], t, 64);
    const double rdv = 1.0 / sqrt(att);
    if (i == t)
      a[t] = att * rdv;  // = sqrt(att)
    else if (i > t)
      a[t] *= rdv;
#pragma unroll
    for (int j = t + 1; j < MP; ++j) {
      const double ljt = __shfl(a[t], j, 64);
      if (i >= j) a[j] = fma(-a[t], ljt, a[j]);
    }
  }

  // write row i (upper half as deterministic zeros)
  if (i < MP) {
#pragma unroll
    for (int j = 0; j < MP; ++j)
      L[(long)i * MP + j] = (j <= i) ? a[j] : 0.0;
  }
}

// ---------------------------------------------------------------------
// trsm_fp: per (draw, frequency-tile) forward-substitute
//   W = L^-1 [B_cols | u]   (u = the T^T N^-1 r column, solved
//   redundantly per tile: one column vs 126)
// ENTIRELY IN REGISTERS, fused with the per-frequency reduction
//   M = sNs - [Ws.Ws, Ws.Wc; ., Wc.Wc],  N = sNr - [Ws.wu, Wc.wu]
//   Fp[d,f] += 0.5 * N^T M^-1 N   (closed-form 2x2)
//
// Key identity (probed layout, tools/mfma_probe.hip): the MFMA D/C
// accumulator layout of a 16x16 tile IS the B-fragment layout across
// k-steps -- value (row 4v+g, col j) lives in lane (g<<4)|j reg v, and
// the b-operand for k-step kk wants (row kk*4 + (l>>4), col l&15),
// i.e. reg kk of the SAME lane.  So the blocked forward substitution
// runs with W held in registers (acc layout), zero cross-lane moves:
//   acc  = -RHS_rb;  acc += sum_cb L[rb,cb] (x) W[cb]   (a from LDS Lp)
//   W[rb] = Iv[rb] (x) (-acc)                           (a from LDS Iv)
// The reduction is also register/shfl-resident (a frequency's sin/cos
// columns are ADJACENT lanes); LDS holds only the double-buffered L
// row-panel, the inverted diagonal blocks and the tiny solved-u column
// (~18 KB at the compressed shape -> 4 WG/CU, occupancy bounded by the
// 61-VGPR budget at 8 waves/SIMD).
// grid = (ceil(F/63), ceil(D/DPG), P);  block = 512 (8 waves)
// cols layout: [s0 c0 s1 c1 ... s62 c62 | u | pad]
// ---------------------------------------------------------------------
#define FPT_COLS 128
#define FPT_FREQS 63
#define NBMAX (FASTFP_MAXMP / 16)

// SWAP: launch with grid (D, ftiles, P) instead of (ftiles, D, P) so
// consecutive workgroups share one RHS column strip (L2-resident
// across ~D consecutive blocks) instead of streaming 16 different
// strips — A/B arm for the measured 2.9 ms RHS-load attribution
// (FASTFP_TRSM_GRID=d).
template <int NBT, int DPG, bool SWAP = false>
__global__ __launch_bounds__(512, 4) void trsm_fp_kernel(
    const double* __restrict__ L_all /*(P*D,mp,mp)*/,
    const double* __restrict__ invd_all /*(P*D, mp/16, 16, 16)*/,
    const double* __restrict__ RHS_all /*(P, mp, 2F+1)*/,
    const double* __restrict__ sNs_all /*(P,3,F)*/,
    const double* __restrict__ sNr_all /*(P,2,F)*/, int F, int D,
    double gsign, double* __restrict__ fp_all /*(P,D,F)*/) {
  // grid.z batches PULSARS (one launch per draw chunk); each pulsar
  // accumulates into its own fp plane, summed deterministically by the
  // caller (no cross-pulsar races, no atomics).
  // gsign: +1 for the direct path (M = sNs - W.W), -1 for the
  // Schur-compressed draw path (M = M0 + W.W) -- docs/DESIGN.md.
  // NBT = mp/16 is a template parameter so every W[] index below is
  // compile-time: with a runtime index the register array is demoted to
  // scratch (288 B/lane measured) and every MFMA b-operand becomes a
  // memory load -- the whole point of the register-resident design.
  // DPG (1 or 2) = draws per workgroup: at the compressed path's small
  // NBT the solve is short, so two draws share the Lp/Iv staging and
  // barriers, and their MFMA chains interleave on the pipe.
  constexpr int mp = NBT * 16;
  __shared__ double Lp[2][DPG][16][NBT * 16 + 1];  // double-buffered
  __shared__ double Iv[DPG][NBT][16][17];
  __shared__ double Wu[DPG][NBT * 16];  // the solved u column

  const int pp = blockIdx.z;
  const double* L = L_all + (long)pp * D * mp * mp;
  const double* invd = invd_all + (long)pp * D * NBT * 256;
  const double* RHS = RHS_all + (long)pp * mp * (2L * F + 1);
  const double* sNs = sNs_all + (long)pp * 3 * F;
  const double* sNr = sNr_all + (long)pp * 2 * F;
  double* fp = fp_all + (long)pp * D * F;
  const int d0 = (SWAP ? blockIdx.x : blockIdx.y) * DPG;
  const int f0 = (SWAP ? blockIdx.y : blockIdx.x) * FPT_FREQS;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wv = tid >> 6;
  const long ldr = 2L * F + 1;
  const int jw = wv * 16;      // this wave's 16-column strip
  const int li = lane & 15;    // a-frag row / strip column
  const int lk = lane >> 4;    // k index / acc row group
  const int ndr = min(DPG, D - d0);  // draws handled here (edge: D odd)

  // stage ALL inverted diagonal blocks once (both draws)
  for (int idx = tid; idx < ndr * NBT * 256; idx += 512) {
    const int e = idx / (NBT * 256);
    const int r = idx % (NBT * 256);
    Iv[e][r >> 8][(r >> 4) & 15][r & 15] =
        invd[((long)(d0 + e) * NBT) * 256 + r];
  }

  // load this wave's RHS strip into registers (acc layout):
  // W[e][rt][v] = RHS[row = rt*16 + 4v + lk][block-col jw + li]
  // (same RHS for every draw)
  const int bc = jw + li;
  f64x4 W[DPG][NBT];
#pragma unroll
  for (int rt = 0; rt < NBT; ++rt) {
#pragma unroll
    for (int v = 0; v < 4; ++v) {
      const long row = rt * 16 + 4 * v + lk;
      double val = 0.0;
      if (bc < 126) {
        const long gc = 2L * f0 + bc;
        if (gc < 2L * F) val = RHS[row * ldr + gc];
      } else if (bc == 126) {
        val = RHS[row * ldr + 2L * F];  // the u column
      }
#pragma unroll
      for (int e = 0; e < DPG; ++e) W[e][rt][v] = val;
    }
  }
  __syncthreads();  // Iv staged

  // blocked forward substitution, W in registers, DPG draws
  // interleaved.  The L row-panel is DOUBLE-BUFFERED: each iteration
  // stages the NEXT row block's panel while computing on the current
  // one, so there is ONE barrier per row block instead of two and the
  // staging loads hide under the MFMA phase (the PMC profile charged
  // ~30% of wave cycles to the staging barriers).
#pragma unroll
  for (int rb = 0; rb < NBT; ++rb) {
    if (rb + 1 < NBT) {
      const int ncols = (rb + 1) * 16;
      for (int idx = tid; idx < ndr * 16 * ncols; idx += 512) {
        const int e = idx / (16 * ncols);
        const int q = idx % (16 * ncols);
        const int r = q / ncols, c = q % ncols;
        Lp[(rb + 1) & 1][e][r][c] =
            L[((long)(d0 + e) * mp + (rb + 1) * 16 + r) * mp + c];
      }
    }
    // two accumulator chains per draw (even/odd cb) deepen the MFMA
    // pipeline; folded together before the diagonal solve
    f64x4 acc[DPG], acc2[DPG];
#pragma unroll
    for (int e = 0; e < DPG; ++e) {
      acc[e] = -W[e][rb];  // -RHS_rb
      acc2[e] = f64x4{0, 0, 0, 0};
    }
#pragma unroll
    for (int cb = 0; cb < rb; ++cb) {
#pragma unroll
      for (int kk = 0; kk < 4; ++kk) {
#pragma unroll
        for (int e = 0; e < DPG; ++e) {
          const double a = Lp[rb & 1][e][li][cb * 16 + kk * 4 + lk];
          if (cb & 1)
            acc2[e] = MFMA_F64(a, W[e][cb][kk], acc2[e]);
          else
            acc[e] = MFMA_F64(a, W[e][cb][kk], acc[e]);
        }
      }
    }
#pragma unroll
    for (int e = 0; e < DPG; ++e) acc[e] += acc2[e];
    // W[rb] = Iv[rb] * (RHS - sum) = Iv[rb] * (-acc)
    // (a split even/odd-kk chain costs 7 extra VGPRs -> occupancy
    // 8 -> 7 waves/SIMD and measures SLOWER; keep the single chain)
    f64x4 sol[DPG];
#pragma unroll
    for (int e = 0; e < DPG; ++e) sol[e] = f64x4{0, 0, 0, 0};
#pragma unroll
    for (int kk = 0; kk < 4; ++kk) {
#pragma unroll
      for (int e = 0; e < DPG; ++e) {
        const double a = Iv[e][rb][li][kk * 4 + lk];
        sol[e] = MFMA_F64(a, -acc[e][kk], sol[e]);
      }
    }
#pragma unroll
    for (int e = 0; e < DPG; ++e) W[e][rb] = sol[e];
    __syncthreads();  // panel rb+1 staged for all; Lp[rb&1] reads done
  }

  // fused reduction, register-resident: a frequency's sin/cos columns
  // are ADJACENT lanes (li even/odd), so the per-frequency dots come
  // from __shfl_xor(w, 1) products accumulated in registers across
  // (rt, v); only the solved u column (wave 7, li == 14) goes through a
  // tiny LDS stage.  No barriers inside the accumulation loop (the
  // LDS-staged 16-row variant cost 2*NBT barriers per draw and
  // dominated the kernel at small NBT).
  if (wv == 7) {
#pragma unroll
    for (int rt = 0; rt < NBT; ++rt)
#pragma unroll
      for (int v = 0; v < 4; ++v)
#pragma unroll
        for (int e = 0; e < DPG; ++e)
          if (li == 14) Wu[e][rt * 16 + 4 * v + lk] = W[e][rt][v];
  }
  __syncthreads();

#pragma unroll
  for (int e = 0; e < DPG; ++e) {
    if (e >= ndr) break;
    // lane-split accumulation: a frequency's sin column is an even
    // lane, its cos column the adjacent odd lane, and the quadratics
    // pair up — w*w accumulates ss on even lanes and cc on odd lanes
    // in the SAME register (likewise w*wu -> su/cu; w*wp is sc on
    // both).  3 fma chains instead of 5; the odd-lane halves arrive
    // by two shfl_xor after the row-group reduction.
    double a1 = 0, a2 = 0, a3 = 0;
#pragma unroll
    for (int rt = 0; rt < NBT; ++rt) {
#pragma unroll
      for (int v = 0; v < 4; ++v) {
        const double w = W[e][rt][v];
        const double wp = __shfl_xor(w, 1, 64);  // partner column
        const double wu = Wu[e][rt * 16 + 4 * v + lk];
        a1 = fma(w, w, a1);
        a2 = fma(w, wu, a2);
        a3 = fma(w, wp, a3);
      }
    }
    // sum the 4 row groups (lanes lk = 0..3 share li): +16, +32 lanes
    a1 += __shfl_down(a1, 32, 64); a1 += __shfl_down(a1, 16, 64);
    a2 += __shfl_down(a2, 32, 64); a2 += __shfl_down(a2, 16, 64);
    a3 += __shfl_down(a3, 32, 64); a3 += __shfl_down(a3, 16, 64);
    const double b1 = __shfl_xor(a1, 1, 64);  // cc (for even lanes)
    const double b2 = __shfl_xor(a2, 1, 64);  // cu
    const int q = (jw + li) >> 1;  // frequency index within the block
    if (lk == 0 && (li & 1) == 0 && q < FPT_FREQS && f0 + q < F) {
      const int f = f0 + q;
      const double M11 = sNs[f] - gsign * a1;
      const double M22 = sNs[F + f] - gsign * b1;
      const double M12 = sNs[2 * F + f] - gsign * a3;
      const double N1 = sNr[f] - gsign * a2;
      const double N2 = sNr[F + f] - gsign * b2;
      const double det = fma(M11, M22, -M12 * M12);
      const double num =
          fma(N1 * N1, M22, fma(-2.0 * N1, N2 * M12, N2 * N2 * M11));
      fp[(long)(d0 + e) * F + f] += 0.5 * num / det;
    }
  }
}

// ---------------------------------------------------------------------
// trsm_fp_rl: RIGHT-LOOKING variant of trsm_fp.  The left-looking form
// accumulates each row block's correction in one (two, even/odd-split)
// dependent MFMA chain; at 64-cycle f64 MFMA dependent latency that
// caps the pipe near 55% (docs/TUNING_NOTES.md).  Right-looking
// applies each solved block's update to ALL remaining row blocks
// immediately:
//   solve  W[cb] = Iv[cb] (x) W[cb]          (4-MFMA chain)
//   update W[rb] -= L[rb][cb] (x) W[cb]      (independent chains, one
//                                             per remaining row block)
// Same MFMA count, same register state (W only — the separate acc
// chains vanish, saving 8 VGPRs), but most MFMAs now sit on
// independent accumulators, so the pipe can fill while the critical
// solve chain waits.  L is staged as COLUMN panels (double-buffered,
// row stride 19: conflict-free for the a-fragment access pattern
// 19*li + 4*kk + lk mod 32 banks).  Selected by FASTFP_TRSM_ALGO=rl.
// ---------------------------------------------------------------------
template <int NBT>
__global__ __launch_bounds__(512, 4) void trsm_fp_rl_kernel(
    const double* __restrict__ L_all /*(P*D,mp,mp)*/,
    const double* __restrict__ invd_all /*(P*D, mp/16, 16, 16)*/,
    const double* __restrict__ RHS_all /*(P, mp, 2F+1)*/,
    const double* __restrict__ sNs_all /*(P,3,F)*/,
    const double* __restrict__ sNr_all /*(P,2,F)*/, int F, int D,
    double gsign, double* __restrict__ fp_all /*(P,D,F)*/) {
  constexpr int mp = NBT * 16;
  constexpr int PROWS = NBT > 1 ? (NBT - 1) * 16 : 1;
  __shared__ double Lc[2][PROWS][19];  // column panel, double-buffered
  __shared__ double Iv[NBT][16][19];
  __shared__ double Wu[NBT * 16];  // the solved u column

  const int pp = blockIdx.z;
  const double* L = L_all + (long)pp * D * mp * mp;
  const double* invd = invd_all + (long)pp * D * NBT * 256;
  const double* RHS = RHS_all + (long)pp * mp * (2L * F + 1);
  const double* sNs = sNs_all + (long)pp * 3 * F;
  const double* sNr = sNr_all + (long)pp * 2 * F;
  double* fp = fp_all + (long)pp * D * F;
  const int d0 = blockIdx.y;
  const int f0 = blockIdx.x * FPT_FREQS;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wv = tid >> 6;
  const long ldr = 2L * F + 1;
  const int jw = wv * 16;
  const int li = lane & 15;
  const int lk = lane >> 4;

  // stage inverted diagonal blocks
  for (int idx = tid; idx < NBT * 256; idx += 512)
    Iv[idx >> 8][(idx >> 4) & 15][idx & 15] =
        invd[(long)d0 * NBT * 256 + idx];

  // load RHS strip into registers (acc layout)
  const int bc = jw + li;
  f64x4 W[NBT];
#pragma unroll
  for (int rt = 0; rt < NBT; ++rt) {
#pragma unroll
    for (int v = 0; v < 4; ++v) {
      const long row = rt * 16 + 4 * v + lk;
      double val = 0.0;
      if (bc < 126) {
        const long gc = 2L * f0 + bc;
        if (gc < 2L * F) val = RHS[row * ldr + gc];
      } else if (bc == 126) {
        val = RHS[row * ldr + 2L * F];  // the u column
      }
      W[rt][v] = val;
    }
  }

  // stage column panel 0 (rows 16..mp of column block 0)
  if (NBT > 1) {
    for (int idx = tid; idx < (NBT - 1) * 16 * 16; idx += 512) {
      const int r = idx >> 4, c = idx & 15;
      Lc[0][r][c] = L[((long)d0 * mp + 16 + r) * mp + c];
    }
  }
  __syncthreads();

#pragma unroll
  for (int cb = 0; cb < NBT; ++cb) {
    // solve W[cb] = Iv[cb] (x) W[cb]
    f64x4 sol = f64x4{0, 0, 0, 0};
#pragma unroll
    for (int kk = 0; kk < 4; ++kk) {
      const double a = Iv[cb][li][kk * 4 + lk];
      sol = MFMA_F64(a, W[cb][kk], sol);
    }
    W[cb] = sol;
    // independent-accumulator updates of every remaining row block
#pragma unroll
    for (int rb = cb + 1; rb < NBT; ++rb) {
#pragma unroll
      for (int kk = 0; kk < 4; ++kk) {
        const double a = -Lc[cb & 1][(rb - cb - 1) * 16 + li][kk * 4 + lk];
        W[rb] = MFMA_F64(a, sol[kk], W[rb]);
      }
    }
    // prefetch the NEXT column panel (issued after the updates so its
    // address arithmetic doesn't inflate the MFMA region's live
    // registers; the loads still overlap the updates' latency since
    // nothing waits on them until the barrier)
    if (cb + 1 < NBT) {
      const int nrows2 = (NBT - 2 - cb) * 16;  // rows (cb+2)*16..mp
      for (int idx = tid; idx < nrows2 * 16; idx += 512) {
        const int r = idx >> 4, c = idx & 15;
        Lc[(cb + 1) & 1][r][c] =
            L[((long)d0 * mp + (cb + 2) * 16 + r) * mp + (cb + 1) * 16 + c];
      }
    }
    __syncthreads();
  }

  // fused register/shfl reduction — identical to trsm_fp_kernel
  if (wv == 7) {
#pragma unroll
    for (int rt = 0; rt < NBT; ++rt)
#pragma unroll
      for (int v = 0; v < 4; ++v)
        if (li == 14) Wu[rt * 16 + 4 * v + lk] = W[rt][v];
  }
  __syncthreads();

  // lane-split accumulation (see trsm_fp_kernel): 3 fma chains, odd
  // halves fetched by shfl_xor after the row-group reduction
  double a1 = 0, a2 = 0, a3 = 0;
#pragma unroll
  for (int rt = 0; rt < NBT; ++rt) {
#pragma unroll
    for (int v = 0; v < 4; ++v) {
      const double w = W[rt][v];
      const double wp = __shfl_xor(w, 1, 64);
      const double wu = Wu[rt * 16 + 4 * v + lk];
      a1 = fma(w, w, a1);
      a2 = fma(w, wu, a2);
      a3 = fma(w, wp, a3);
    }
  }
  a1 += __shfl_down(a1, 32, 64); a1 += __shfl_down(a1, 16, 64);
  a2 += __shfl_down(a2, 32, 64); a2 += __shfl_down(a2, 16, 64);
  a3 += __shfl_down(a3, 32, 64); a3 += __shfl_down(a3, 16, 64);
  const double b1 = __shfl_xor(a1, 1, 64);
  const double b2 = __shfl_xor(a2, 1, 64);
  const int q = (jw + li) >> 1;
  if (lk == 0 && (li & 1) == 0 && q < FPT_FREQS && f0 + q < F) {
    const int f = f0 + q;
    const double M11 = sNs[f] - gsign * a1;
    const double M22 = sNs[F + f] - gsign * b1;
    const double M12 = sNs[2 * F + f] - gsign * a3;
    const double N1 = sNr[f] - gsign * a2;
    const double N2 = sNr[F + f] - gsign * b2;
    const double det = fma(M11, M22, -M12 * M12);
    const double num =
        fma(N1 * N1, M22, fma(-2.0 * N1, N2 * M12, N2 * N2 * M11));
    fp[(long)d0 * F + f] += 0.5 * num / det;
  }
}

// ---------------------------------------------------------------------
// trsm_fp_res: fully LDS-RESIDENT variant.  The r02 PMC profile showed
// the left-looking and right-looking kernels both stall ~40% of wave
// cycles on the per-iteration L-panel staging barriers (waitANY 25% +
// issue waits), NOT on the MFMA dependency chain (the RL experiment
// measured the chain hypothesis dead: +-2%).  At the compressed shape
// (NBT<=5) the strictly-lower triangle of L is only NBT*(NBT-1)/2
// 16x16 tiles (12-24 KB), so the whole factor is staged ONCE at kernel
// start and the solve loop runs with ZERO barriers and zero memory
// waits: 2 barriers total per kernel (post-stage, pre-reduction)
// instead of NBT+1.  Row stride 19 keeps every a-fragment read
// bank-conflict-free (19*li + 4*kk + lk distinct mod 32).
// Dispatched for NBT <= 5 when FASTFP_TRSM_ALGO=res (A/B arm).
// ---------------------------------------------------------------------
// MODE: 0 = full kernel; 1 = skip the reduction/epilogue (timing
// attribution only — writes nothing); 2 = additionally skip the RHS
// global loads (pure stage+solve).  Modes 1/2 exist for the phase
// cost attribution in tools/trsm_diag.py and are never used in
// production (FASTFP_TRSM_MODE).
template <int NBT, int MODE = 0>
__global__ __launch_bounds__(512, 4) void trsm_fp_res_kernel(
    const double* __restrict__ L_all /*(P*D,mp,mp)*/,
    const double* __restrict__ invd_all /*(P*D, mp/16, 16, 16)*/,
    const double* __restrict__ RHS_all /*(P, mp, 2F+1)*/,
    const double* __restrict__ sNs_all /*(P,3,F)*/,
    const double* __restrict__ sNr_all /*(P,2,F)*/, int F, int D,
    double gsign, double* __restrict__ fp_all /*(P,D,F)*/) {
  constexpr int mp = NBT * 16;
  constexpr int NTILE = NBT * (NBT - 1) / 2;  // strictly-lower tiles
  __shared__ double Lt[NTILE > 0 ? NTILE : 1][16][19];
  __shared__ double Iv[NBT][16][19];
  __shared__ double Wu[NBT * 16];

  const int pp = blockIdx.z;
  const double* L = L_all + (long)pp * D * mp * mp;
  const double* invd = invd_all + (long)pp * D * NBT * 256;
  const double* RHS = RHS_all + (long)pp * mp * (2L * F + 1);
  const double* sNs = sNs_all + (long)pp * 3 * F;
  const double* sNr = sNr_all + (long)pp * 2 * F;
  double* fp = fp_all + (long)pp * D * F;
  const int d0 = blockIdx.y;
  const int f0 = blockIdx.x * FPT_FREQS;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wv = tid >> 6;
  const long ldr = 2L * F + 1;
  const int jw = wv * 16;
  const int li = lane & 15;
  const int lk = lane >> 4;

  // stage EVERYTHING once: inverted diagonal blocks + the strictly-
  // lower L tiles (tile t(rb,cb) = rb*(rb-1)/2 + cb)
  for (int idx = tid; idx < NBT * 256; idx += 512)
    Iv[idx >> 8][(idx >> 4) & 15][idx & 15] =
        invd[(long)d0 * NBT * 256 + idx];
  if (NTILE > 0) {
    for (int idx = tid; idx < NTILE * 256; idx += 512) {
      const int t = idx >> 8, r = (idx >> 4) & 15, c = idx & 15;
      // invert t -> (rb, cb): rb = largest with rb*(rb-1)/2 <= t
      int rb = 1;
      while (rb * (rb + 1) / 2 <= t) ++rb;
      const int cb = t - rb * (rb - 1) / 2;
      Lt[t][r][c] = L[((long)d0 * mp + rb * 16 + r) * mp + cb * 16 + c];
    }
  }

  // load RHS strip into registers (acc layout)
  const int bc = jw + li;
  f64x4 W[NBT];
#pragma unroll
  for (int rt = 0; rt < NBT; ++rt) {
#pragma unroll
    for (int v = 0; v < 4; ++v) {
      double val = 1.0;
      if (MODE < 2) {
        const long row = rt * 16 + 4 * v + lk;
        val = 0.0;
        if (bc < 126) {
          const long gc = 2L * f0 + bc;
          if (gc < 2L * F) val = RHS[row * ldr + gc];
        } else if (bc == 126) {
          val = RHS[row * ldr + 2L * F];  // the u column
        }
      }
      W[rt][v] = val;
    }
  }
  __syncthreads();  // everything staged; solve loop is barrier-free

#pragma unroll
  for (int cb = 0; cb < NBT; ++cb) {
    f64x4 sol = f64x4{0, 0, 0, 0};
#pragma unroll
    for (int kk = 0; kk < 4; ++kk) {
      const double a = Iv[cb][li][kk * 4 + lk];
      sol = MFMA_F64(a, W[cb][kk], sol);
    }
    W[cb] = sol;
#pragma unroll
    for (int rb = cb + 1; rb < NBT; ++rb) {
#pragma unroll
      for (int kk = 0; kk < 4; ++kk) {
        const double a = -Lt[rb * (rb - 1) / 2 + cb][li][kk * 4 + lk];
        W[rb] = MFMA_F64(a, sol[kk], W[rb]);
      }
    }
  }

  if (MODE >= 1) {
    // attribution mode: keep the solve live, skip reduction/epilogue
    double s = 0.0;
#pragma unroll
    for (int rt = 0; rt < NBT; ++rt) s += W[rt][0];
    if (s == 1.2345678e300) fp[0] = s;  // never true for real data
    return;
  }

  // fused register/shfl reduction — identical to trsm_fp_kernel
  if (wv == 7) {
#pragma unroll
    for (int rt = 0; rt < NBT; ++rt)
#pragma unroll
      for (int v = 0; v < 4; ++v)
        if (li == 14) Wu[rt * 16 + 4 * v + lk] = W[rt][v];
  }
  __syncthreads();

  // lane-split accumulation (see trsm_fp_kernel): 3 fma chains, odd
  // halves fetched by shfl_xor after the row-group reduction
  double a1 = 0, a2 = 0, a3 = 0;
#pragma unroll
  for (int rt = 0; rt < NBT; ++rt) {
#pragma unroll
    for (int v = 0; v < 4; ++v) {
      const double w = W[rt][v];
      const double wp = __shfl_xor(w, 1, 64);
      const double wu = Wu[rt * 16 + 4 * v + lk];
      a1 = fma(w, w, a1);
      a2 = fma(w, wu, a2);
      a3 = fma(w, wp, a3);
    }
  }
  a1 += __shfl_down(a1, 32, 64); a1 += __shfl_down(a1, 16, 64);
  a2 += __shfl_down(a2, 32, 64); a2 += __shfl_down(a2, 16, 64);
  a3 += __shfl_down(a3, 32, 64); a3 += __shfl_down(a3, 16, 64);
  const double b1 = __shfl_xor(a1, 1, 64);
  const double b2 = __shfl_xor(a2, 1, 64);
  const int q = (jw + li) >> 1;
  if (lk == 0 && (li & 1) == 0 && q < FPT_FREQS && f0 + q < F) {
    const int f = f0 + q;
    const double M11 = sNs[f] - gsign * a1;
    const double M22 = sNs[F + f] - gsign * b1;
    const double M12 = sNs[2 * F + f] - gsign * a3;
    const double N1 = sNr[f] - gsign * a2;
    const double N2 = sNr[F + f] - gsign * b2;
    const double det = fma(M11, M22, -M12 * M12);
    const double num =
        fma(N1 * N1, M22, fma(-2.0 * N1, N2 * M12, N2 * N2 * M11));
    fp[(long)d0 * F + f] += 0.5 * num / det;
  }
}

// ---------------------------------------------------------------------
// diag_inv: invert the 16x16 diagonal blocks of a batch of lower-
// triangular L (B, mp, mp) -> invd (B, mp/16, 16, 16).  Used by the
// m > 128 direct path, where L comes from batched rocSOLVER Cholesky
// (the LDS-resident chol_batch kernel caps at mp = 128: Sigma no
// longer fits LDS above that).  One wave per (batch, block); register
// rows + width-16 shfl — same technique as chol_batch's CHOL_DIAG.
// grid.x = ceil(B * mp/16 / 8), block = 512 (8 waves)
// ---------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(512, 4) void diag_inv_kernel(
    const double* __restrict__ L, int mp, long nblk_total,
    double* __restrict__ invd) {
  const long w = (long)blockIdx.x * 8 + (threadIdx.x >> 6);
  if (w >= nblk_total) return;
  const int nbt = mp >> 4;
  const long b = w / nbt;
  const int kb = (int)(w % nbt);
  const int lane = threadIdx.x & 63;
  const int i = lane & 15;  // row; the 4 sub-groups compute redundantly
  const double* Lb = L + b * (long)mp * mp + (long)kb * 16 * mp + kb * 16;
  double row[16];
#pragma unroll
  for (int c = 0; c < 16; ++c) row[c] = (c <= i) ? Lb[(long)i * mp + c] : 0.0;
  const int c = i;  // lane owns output column c
  double x[16];
  double dii = 1.0;
#pragma unroll
  for (int r = 0; r < 16; ++r)
    if (r == i) dii = row[r];
  const double myrcp = 1.0 / dii;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const double dr = __shfl(myrcp, r, 16);
    double acc = 0.0;
#pragma unroll
    for (int t = 0; t < 16; ++t) {
      const double lrt = __shfl(row[t], r, 16);
      if (t >= c && t < r) acc = fma(lrt, x[t], acc);
    }
    x[r] = (r < c) ? 0.0 : (r == c) ? dr : -acc * dr;
  }
  if (lane < 16)
#pragma unroll
    for (int r = 0; r < 16; ++r)
      invd[(b * nbt + kb) * 256 + r * 16 + c] = x[r];
}

// ---------------------------------------------------------------------
// Block-diagonal white noise (EcorrKernelNoise, BASELINE config 4):
// N is block-diagonal per observing epoch.  The reference never
// implemented this case (/root/reference/fastfp/utils.py:30-31,
// README.md:22).  Blocks are contiguous after the BlockNoise TOA
// permutation (fastfp_amd/blocknoise.py).
//
// Each block is diag(nvec) + e2*J (diagonal + rank-1), so the inverse
// is the Sherman-Morrison closed form
//   N_b^{-1} = D^{-1} - beta_b d d^T,  d = D^{-1} 1,
//   beta_b = e2 / (1 + e2 * sum(1/nvec_b))
// — no factorization, O(sz) per block, ANY epoch size (the round-1
// per-block Cholesky kernel and its 32-TOA cap are gone).
// ---------------------------------------------------------------------

// sigdots_block: the five per-frequency dots with BLOCK-diagonal N.
// sr/cr use the precomputed nr = N^{-1} r vector (host S-M solve); the
// quadratics are the diagonal part (uvec-weighted, like sigdots) minus
// the per-block rank-1 corrections beta_b * (d.s)_b (d.c)_b.
// grid.x = F, block = 256.
extern "C" __global__ __launch_bounds__(256) void sigdots_block_kernel(
    const double* __restrict__ toas, const double* __restrict__ uvec,
    const double* __restrict__ nr, const double* __restrict__ freqs,
    const double* __restrict__ beta, const long* __restrict__ offsets,
    const long* __restrict__ sizes, int nblk, int ntoa, int F,
    double* __restrict__ sNs, double* __restrict__ sNr) {
  const int f = blockIdx.x;
  if (f >= F) return;
  const double w = 2.0 * M_PI * freqs[f];
  double ss = 0, cc = 0, sc = 0, sr = 0, cr = 0;
  // diagonal part + the N^-1 r dots
  for (int i = threadIdx.x; i < ntoa; i += blockDim.x) {
    double s, c;
    sincos(w * toas[i], &s, &c);
    const double ui = uvec[i];
    const double ri = nr[i];
    ss = fma(s * s, ui, ss);
    cc = fma(c * c, ui, cc);
    sc = fma(s * c, ui, sc);
    sr = fma(s, ri, sr);
    cr = fma(c, ri, cr);
  }
  // rank-1 corrections: one weighted sum per block (any size)
  for (int b = threadIdx.x; b < nblk; b += blockDim.x) {
    const double bb = beta[b];
    if (bb == 0.0) continue;  // singleton / no-ecorr block
    const long o = offsets[b];
    const int sz = (int)sizes[b];
    double us = 0.0, uc = 0.0;
    for (int i = 0; i < sz; ++i) {
      double s, c;
      sincos(w * toas[o + i], &s, &c);
      const double ui = uvec[o + i];
      us = fma(s, ui, us);
      uc = fma(c, ui, uc);
    }
    ss = fma(-bb * us, us, ss);
    cc = fma(-bb * uc, uc, cc);
    sc = fma(-bb * us, uc, sc);
  }
  __shared__ double red[4][5];
  ss = wave_reduce_sum(ss);
  cc = wave_reduce_sum(cc);
  sc = wave_reduce_sum(sc);
  sr = wave_reduce_sum(sr);
  cr = wave_reduce_sum(cr);
  const int lane = threadIdx.x & 63, wv = threadIdx.x >> 6;
  if (lane == 0) {
    red[wv][0] = ss; red[wv][1] = cc; red[wv][2] = sc;
    red[wv][3] = sr; red[wv][4] = cr;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    double a0 = 0, a1 = 0, a2 = 0, a3 = 0, a4 = 0;
    for (int v = 0; v < 4; ++v) {
      a0 += red[v][0]; a1 += red[v][1]; a2 += red[v][2];
      a3 += red[v][3]; a4 += red[v][4];
    }
    sNs[f] = a0; sNs[F + f] = a1; sNs[2 * F + f] = a2;
    sNr[f] = a3; sNr[F + f] = a4;
  }
}

// ---------------------------------------------------------------------
// host-side launchers (called from bindings.cpp)
// ---------------------------------------------------------------------
extern "C" {

void launch_sigdots(const double* toas, const double* ninv, const double* nr,
                    const double* freqs, int ntoa, int F, double* sNs,
                    double* sNr, hipStream_t stream) {
  hipLaunchKernelGGL(sigdots_kernel, dim3(F), dim3(256), 0, stream, toas,
                     ninv, nr, freqs, ntoa, F, sNs, sNr);
}

void launch_sbgemm(const double* T, const double* toas, const double* ninv,
                   const double* freqs, int ntoa, int m, int mp, int F2,
                   double* out, long plane_stride, long ldo, int ksplit,
                   hipStream_t stream) {
  const int ctiles = (F2 + 63) / 64;
  const int mtiles = (mp + FASTFP_MAXMP - 1) / FASTFP_MAXMP;
  static const char* algo = getenv("FASTFP_SBGEMM_ALGO");
  static const bool use_db = algo && 0 == __builtin_strcmp(algo, "db");
  if (use_db) {
    hipLaunchKernelGGL(sbgemm_db_kernel, dim3(ctiles, ksplit, mtiles),
                       dim3(512), 0, stream, T, toas, ninv, freqs, ntoa, m,
                       mp, F2, out, plane_stride, ldo);
    return;
  }
  hipLaunchKernelGGL(sbgemm_kernel, dim3(ctiles, ksplit, mtiles), dim3(512),
                     0, stream, T, toas, ninv, freqs, ntoa, m, mp, F2, out,
                     plane_stride, ldo);
}

void launch_diag_inv(const double* L, int mp, long nblk_total, double* invd,
                     hipStream_t stream);

void launch_chol_batch(const double* TNT, const double* phiinv, int m, int mp,
                       int D, int P, double* L, double* invd,
                       hipStream_t stream) {
  // small matrices: 256 threads -> 4 workgroups/CU despite the 115-VGPR
  // diagonal-factor pressure; large: 512 threads for MFMA coverage
  const dim3 grid(D, P), blk(mp <= 64 ? 256 : 512);
  // FASTFP_CHOL_ALGO=wave: fully in-wave factor (one draw per wave,
  // no LDS/barriers) for mp <= 64; the diagonal-block inverses come
  // from the diag_inv kernel
  static const char* walgo = getenv("FASTFP_CHOL_ALGO");
  static const bool use_wave = walgo && 0 == __builtin_strcmp(walgo, "wave");
  if (use_wave && mp <= 64) {
    const dim3 wgrid((D + 3) / 4, P);
    switch (mp >> 4) {
#define CHOL_WAVE_CASE(NBT) \
      case NBT: hipLaunchKernelGGL((chol_wave_kernel<NBT * 16>), wgrid, \
                    dim3(256), 0, stream, TNT, phiinv, m, D, L); break;
      CHOL_WAVE_CASE(1) CHOL_WAVE_CASE(2) CHOL_WAVE_CASE(3) CHOL_WAVE_CASE(4)
#undef CHOL_WAVE_CASE
    }
    launch_diag_inv(L, mp, (long)P * D * (mp >> 4), invd, stream);
    return;
  }
  // Lower-triangle-packed Sigma is the DEFAULT (measured -13% on chol
  // at the bench shape from the extra workgroup of concurrency; at
  // mp=128 the square layout is 132 KB LDS = ONE workgroup per CU, so
  // packing doubles concurrency).  FASTFP_CHOL_TRI=0 restores the
  // square layout for A/B.
  static const char* tri_env = getenv("FASTFP_CHOL_TRI");
  static const bool tri = !(tri_env && tri_env[0] == '0');
  if (tri) {
    switch (mp >> 4) {
#define CHOL_TRI_CASE(NBT) \
      case NBT: hipLaunchKernelGGL((chol_batch_kernel<NBT, true>), grid, \
                    blk, 0, stream, TNT, phiinv, m, D, L, invd); break;
      CHOL_TRI_CASE(1) CHOL_TRI_CASE(2) CHOL_TRI_CASE(3) CHOL_TRI_CASE(4)
      CHOL_TRI_CASE(5) CHOL_TRI_CASE(6) CHOL_TRI_CASE(7) CHOL_TRI_CASE(8)
#undef CHOL_TRI_CASE
    }
    return;
  }
  switch (mp >> 4) {
#define CHOL_CASE(NBT) \
    case NBT: hipLaunchKernelGGL(chol_batch_kernel<NBT>, grid, blk, 0, \
                                 stream, TNT, phiinv, m, D, L, invd); break;
    CHOL_CASE(1) CHOL_CASE(2) CHOL_CASE(3) CHOL_CASE(4)
    CHOL_CASE(5) CHOL_CASE(6) CHOL_CASE(7) CHOL_CASE(8)
#undef CHOL_CASE
  }
}

void launch_trsm_fp(const double* L, const double* invd, const double* RHS,
                    const double* sNs, const double* sNr, int mp, int F,
                    int D, int P, double gsign, double* fp,
                    hipStream_t stream) {
  const int ftiles = (F + FPT_FREQS - 1) / FPT_FREQS;
  const dim3 blk(512);
  const int nb = mp >> 4;
  // FASTFP_TRSM_ALGO selects the A/B arm: "rl" = right-looking staged
  // panels; "res" = fully LDS-resident barrier-free (NBT <= 5, falls
  // back to rl above that)
  static const char* algo_env = getenv("FASTFP_TRSM_ALGO");
  static const bool use_res = algo_env && 0 == __builtin_strcmp(algo_env, "res");
  static const bool use_rl = algo_env && 0 == __builtin_strcmp(algo_env, "rl");
  if (use_res && nb <= 5) {
    const dim3 grid(ftiles, D, P);
    // FASTFP_TRSM_MODE=1/2: phase-attribution debug arms (NBT=4 only)
    static const char* mode_env = getenv("FASTFP_TRSM_MODE");
    static const int mode = mode_env ? atoi(mode_env) : 0;
    if (mode == 1 && nb == 4) {
      hipLaunchKernelGGL((trsm_fp_res_kernel<4, 1>), grid, blk, 0, stream,
                         L, invd, RHS, sNs, sNr, F, D, gsign, fp);
      return;
    }
    if (mode == 2 && nb == 4) {
      hipLaunchKernelGGL((trsm_fp_res_kernel<4, 2>), grid, blk, 0, stream,
                         L, invd, RHS, sNs, sNr, F, D, gsign, fp);
      return;
    }
    switch (nb) {
#define TRSM_RES_CASE(NBT) \
      case NBT: hipLaunchKernelGGL((trsm_fp_res_kernel<NBT>), grid, blk, 0, \
                    stream, L, invd, RHS, sNs, sNr, F, D, gsign, fp); break;
      TRSM_RES_CASE(1) TRSM_RES_CASE(2) TRSM_RES_CASE(3)
      TRSM_RES_CASE(4) TRSM_RES_CASE(5)
#undef TRSM_RES_CASE
    }
    return;
  }
  if ((use_rl || (use_res && nb > 5)) && nb <= 8) {
    const dim3 grid(ftiles, D, P);
    switch (nb) {
#define TRSM_RL_CASE(NBT) \
      case NBT: hipLaunchKernelGGL((trsm_fp_rl_kernel<NBT>), grid, blk, 0, \
                    stream, L, invd, RHS, sNs, sNr, F, D, gsign, fp); break;
      TRSM_RL_CASE(1) TRSM_RL_CASE(2) TRSM_RL_CASE(3) TRSM_RL_CASE(4)
      TRSM_RL_CASE(5) TRSM_RL_CASE(6) TRSM_RL_CASE(7) TRSM_RL_CASE(8)
#undef TRSM_RL_CASE
    }
    return;
  }
  // FASTFP_TRSM_GRID=d: draws-innermost grid (RHS L2-locality A/B arm)
  static const char* grid_env = getenv("FASTFP_TRSM_GRID");
  static const bool swap_grid = grid_env && grid_env[0] == 'd';
  if (swap_grid && nb <= 8) {
    const dim3 grid(D, ftiles, P);
    switch (nb) {
#define TRSM_SW_CASE(NBT) \
      case NBT: hipLaunchKernelGGL((trsm_fp_kernel<NBT, 1, true>), grid, \
                    blk, 0, stream, L, invd, RHS, sNs, sNr, F, D, gsign, \
                    fp); break;
      TRSM_SW_CASE(1) TRSM_SW_CASE(2) TRSM_SW_CASE(3) TRSM_SW_CASE(4)
      TRSM_SW_CASE(5) TRSM_SW_CASE(6) TRSM_SW_CASE(7) TRSM_SW_CASE(8)
#undef TRSM_SW_CASE
    }
    return;
  }
  // DPG selects draws per workgroup for small solves; the measured
  // default is 1 (8 waves/SIMD at 61 VGPR beats DPG=2's 2 chains at
  // 4 waves/SIMD by ~5%); FASTFP_TRSM_DPG=2 re-enables the A/B arm
  static const char* dpg_env = getenv("FASTFP_TRSM_DPG");
  static const int dpg_want = dpg_env ? atoi(dpg_env) : 1;  // measured: DPG1 (8 waves/SIMD, 61 VGPR) beats DPG2 (4 waves/SIMD) by ~5%
  if (nb <= 4 && dpg_want >= 2) {
    const dim3 grid(ftiles, (D + 1) / 2, P);
    switch (nb) {
#define TRSM_CASE2(NBT) \
      case NBT: hipLaunchKernelGGL((trsm_fp_kernel<NBT, 2>), grid, blk, 0, \
                    stream, L, invd, RHS, sNs, sNr, F, D, gsign, fp); break;
      TRSM_CASE2(1) TRSM_CASE2(2) TRSM_CASE2(3) TRSM_CASE2(4)
#undef TRSM_CASE2
    }
  } else {
    const dim3 grid(ftiles, D, P);
    switch (nb) {
#define TRSM_CASE(NBT) \
      case NBT: hipLaunchKernelGGL((trsm_fp_kernel<NBT, 1>), grid, blk, 0, \
                    stream, L, invd, RHS, sNs, sNr, F, D, gsign, fp); break;
      TRSM_CASE(1) TRSM_CASE(2) TRSM_CASE(3) TRSM_CASE(4)
      TRSM_CASE(5) TRSM_CASE(6) TRSM_CASE(7) TRSM_CASE(8)
#undef TRSM_CASE
      // m > 128 (GP-ECORR direct path) dispatches the RIGHT-LOOKING
      // variant: its single in-place accumulator set (no acc/acc2
      // chains) keeps W = NBT*4 f64 regs per lane within budget at
      // NBT up to 16.  The FACTOR for these sizes comes from rocSOLVER
      // (chol_batch's LDS-resident Sigma caps at 128) + diag_inv.
#define TRSM_BIG_CASE(NBT) \
      case NBT: hipLaunchKernelGGL((trsm_fp_rl_kernel<NBT>), grid, blk, 0, \
                    stream, L, invd, RHS, sNs, sNr, F, D, gsign, fp); break;
      TRSM_BIG_CASE(9) TRSM_BIG_CASE(10) TRSM_BIG_CASE(11) TRSM_BIG_CASE(12)
      TRSM_BIG_CASE(13) TRSM_BIG_CASE(14) TRSM_BIG_CASE(15) TRSM_BIG_CASE(16)
#undef TRSM_BIG_CASE
    }
  }
}

void launch_diag_inv(const double* L, int mp, long nblk_total, double* invd,
                     hipStream_t stream) {
  hipLaunchKernelGGL(diag_inv_kernel, dim3((nblk_total + 7) / 8), dim3(512),
                     0, stream, L, mp, nblk_total, invd);
}

void launch_sigdots_block(const double* toas, const double* uvec,
                          const double* nr, const double* freqs,
                          const double* beta, const long* offsets,
                          const long* sizes, int nblk, int ntoa, int F,
                          double* sNs, double* sNr, hipStream_t stream) {
  hipLaunchKernelGGL(sigdots_block_kernel, dim3(F), dim3(256), 0, stream,
                     toas, uvec, nr, freqs, beta, offsets, sizes,
                     nblk, ntoa, F, sNs, sNr);
}

}  // extern "C"
