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
// grid.x = ceil(2F / 64) col tiles, grid.y = ksplit;  block = 256 (4 waves)
// ---------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(256) void sbgemm_kernel(
    const double* __restrict__ T /*(ntoa, m) row-major*/,
    const double* __restrict__ toas, const double* __restrict__ ninv,
    const double* __restrict__ freqs, int ntoa, int m, int mp, int F2,
    double* __restrict__ out, long plane_stride, long ldo) {
  __shared__ double lT[16][FASTFP_MAXMP + 1];
  __shared__ double lS[64][17];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wv = tid >> 6;
  const int c0 = blockIdx.x * 64;       // first output column of this tile
  const int nrt = mp >> 4;              // row tiles

  // K range of this split
  const int ks = gridDim.y;
  const int kchunk = (ntoa + ks - 1) / ks;
  const int kbeg = blockIdx.y * kchunk;
  const int kend = min(ntoa, kbeg + kchunk);
  double* outp = out + (long)blockIdx.y * plane_stride;

  f64x4 acc[FASTFP_MAXMP / 16];
#pragma unroll
  for (int rt = 0; rt < FASTFP_MAXMP / 16; ++rt) acc[rt] = f64x4{0, 0, 0, 0};

  for (int k0 = kbeg; k0 < kend; k0 += 16) {
    // stage T panel rows k0..k0+15 (zero-padded)
    for (int idx = tid; idx < 16 * mp; idx += 256) {
      const int k = idx / mp, j = idx % mp;
      const int gk = k0 + k;
      lT[k][j] = (gk < kend && j < m) ? T[(long)gk * m + j] : 0.0;
    }
    // stage trig panel: 32 freq-pairs x 16 toas, one sincos each
    for (int idx = tid; idx < 32 * 16; idx += 256) {
      const int p = idx / 16, k = idx % 16;
      const int gk = k0 + k;
      const int ceven = c0 + 2 * p;
      double s = 0.0, c = 0.0;
      if (gk < kend && ceven < F2) {
        const double wf = 2.0 * M_PI * freqs[ceven >> 1];
        sincos(wf * toas[gk], &s, &c);
        const double ni = ninv[gk];
        s *= ni; c *= ni;
      }
      lS[2 * p][k] = s;
      if (2 * p + 1 < 64) lS[2 * p + 1][k] = c;
    }
    __syncthreads();

    const int jw = wv * 16;  // this wave's 16-column strip
#pragma unroll
    for (int kk = 0; kk < 4; ++kk) {
      const double b = lS[jw + (lane & 15)][kk * 4 + (lane >> 4)];
#pragma unroll
      for (int rt = 0; rt < FASTFP_MAXMP / 16; ++rt) {
        if (rt >= nrt) break;
        const double a = lT[kk * 4 + (lane >> 4)][rt * 16 + (lane & 15)];
        acc[rt] = MFMA_F64(a, b, acc[rt]);
      }
    }
    __syncthreads();
  }

  // epilogue: store (row j, col c) — cols contiguous, coalesced
  const int jw = wv * 16;
#pragma unroll
  for (int rt = 0; rt < FASTFP_MAXMP / 16; ++rt) {
    if (rt >= nrt) break;
#pragma unroll
    for (int v = 0; v < 4; ++v) {
      const int row = rt * 16 + 4 * v + (lane >> 4);
      const int col = c0 + jw + (lane & 15);
      if (col < F2) outp[(long)row * ldo + col] = acc[rt][v];
    }
  }
}

// ---------------------------------------------------------------------
// chol_batch: per draw d, assemble Sigma = TNT + diag(phiinv[d]) (+
// identity padding to mp) in LDS, factor L L^T = Sigma (right-looking,
// NB=16 blocks, MFMA trailing updates), invert each 16x16 diagonal
// block, write L (D, mp, mp) and invdiag (D, mp/16, 16, 16).
// grid.x = D, block = 256 (4 waves)
// ---------------------------------------------------------------------
#define A_(i, j) Ash[(i) * (FASTFP_MAXMP + 1) + (j)]

extern "C" __global__ __launch_bounds__(256) void chol_batch_kernel(
    const double* __restrict__ TNT /*(m,m)*/,
    const double* __restrict__ phiinv /*(D,m)*/, int m, int mp, int D,
    double* __restrict__ L /*(D,mp,mp)*/,
    double* __restrict__ invd /*(D, mp/16, 16, 16)*/) {
  __shared__ double Ash[FASTFP_MAXMP * (FASTFP_MAXMP + 1)];
  __shared__ double inv16[16][17];

  const int d = blockIdx.x;
  if (d >= D) return;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wv = tid >> 6;
  const int nb = mp >> 4;

  // assemble Sigma in LDS
  for (int idx = tid; idx < mp * mp; idx += 256) {
    const int i = idx / mp, j = idx % mp;
    double v = (i < m && j < m) ? TNT[(long)i * m + j] : 0.0;
    if (i == j) v += (i < m) ? phiinv[(long)d * m + i] : 1.0;
    A_(i, j) = v;
  }
  __syncthreads();

  for (int kb = 0; kb < nb; ++kb) {
    const int k0 = kb * NB;

    // 1) unblocked Cholesky of the 16x16 diagonal block (wave 0,
    //    lanes 0..15 own rows; in-wave lockstep ordering)
    if (wv == 0 && lane < 16) {
      const int i = lane;
      for (int t = 0; t < 16; ++t) {
        const double att = A_(k0 + t, k0 + t);
        const double dv = sqrt(att);
        double lit = 0.0;
        if (i > t) {
          lit = A_(k0 + i, k0 + t) / dv;
          A_(k0 + i, k0 + t) = lit;
        }
        if (i == t) A_(k0 + t, k0 + t) = dv;
        // rank-1 update of the remaining lower part of the block
        for (int j = t + 1; j <= i; ++j)
          A_(k0 + i, k0 + j) -= lit * A_(k0 + j, k0 + t);
      }
      // 2) invert the diagonal block: X = L_kk^-1, column c per lane
      const int c = lane;
      for (int r = 0; r < 16; ++r) {
        double x;
        if (r < c) x = 0.0;
        else if (r == c) x = 1.0 / A_(k0 + r, k0 + r);
        else {
          double acc = 0.0;
          for (int t = c; t < r; ++t)
            acc = fma(A_(k0 + r, k0 + t), inv16[t][c], acc);
          x = -acc / A_(k0 + r, k0 + r);
        }
        inv16[r][c] = x;
      }
    }
    __syncthreads();

    // export invdiag (coalesced)
    for (int idx = tid; idx < 256; idx += 256) {
      const int r = idx / 16, c = idx % 16;
      invd[((long)d * nb + kb) * 256 + idx] = inv16[r][c];
    }

    // 3) panel TRSM: row tiles below the diagonal, P <- P * inv(L_kk)^T
    for (int rt = kb + 1 + wv; rt < nb; rt += 4) {
      const int r0 = rt * NB;
      f64x4 pacc = {0, 0, 0, 0};
#pragma unroll
      for (int kk = 0; kk < 4; ++kk) {
        const double a = A_(r0 + (lane & 15), k0 + kk * 4 + (lane >> 4));
        const double b = inv16[lane & 15][kk * 4 + (lane >> 4)];  // inv^T[k][j]
        pacc = MFMA_F64(a, b, pacc);
      }
      // in-wave: all reads above complete before these writes
#pragma unroll
      for (int v = 0; v < 4; ++v)
        A_(r0 + 4 * v + (lane >> 4), k0 + (lane & 15)) = pacc[v];
    }
    __syncthreads();

    // 4) trailing update: tiles (ib, jb), kb < jb <= ib < nb
    const int t = nb - kb - 1;
    const int ntile = t * (t + 1) / 2;
    for (int q = wv; q < ntile; q += 4) {
      // triangular index -> (ib, jb), row-major over the lower wedge
      int ib = kb + 1, rem = q;
      while (rem > ib - kb - 1) { rem -= (ib - kb); ++ib; }
      const int jb = kb + 1 + rem;
      const int i0 = ib * NB, j0 = jb * NB;
      f64x4 uacc;
#pragma unroll
      for (int v = 0; v < 4; ++v)
        uacc[v] = A_(i0 + 4 * v + (lane >> 4), j0 + (lane & 15));
#pragma unroll
      for (int kk = 0; kk < 4; ++kk) {
        const double a = -A_(i0 + (lane & 15), k0 + kk * 4 + (lane >> 4));
        const double b = A_(j0 + (lane & 15), k0 + kk * 4 + (lane >> 4));  // P[jb]^T
        uacc = MFMA_F64(a, b, uacc);
      }
#pragma unroll
      for (int v = 0; v < 4; ++v)
        A_(i0 + 4 * v + (lane >> 4), j0 + (lane & 15)) = uacc[v];
    }
    __syncthreads();
  }

  // write back L (full rows; upper-triangle junk is never read)
  for (int idx = tid; idx < mp * mp; idx += 256) {
    const int i = idx / mp, j = idx % mp;
    L[((long)d * mp + i) * mp + j] = A_(i, j);
  }
}

// ---------------------------------------------------------------------
// trsm_fp: per (draw, frequency-tile) forward-substitute
//   W = L^-1 [B_cols | u]   (u = the T^T N^-1 r column, solved
//   redundantly per tile: one column vs 126)
// entirely in LDS, then the fused per-frequency reduction
//   M = sNs - [Ws.Ws, Ws.Wc; ., Wc.Wc],  N = sNr - [Ws.wu, Wc.wu]
//   Fp[d,f] += 0.5 * N^T M^-1 N   (closed-form 2x2)
// grid.x = ceil(F / 63), grid.y = D;  block = 512 (8 waves)
// cols layout in LDS W: [s0 c0 s1 c1 ... s62 c62 | u | pad]
// ---------------------------------------------------------------------
#define FPT_COLS 128
#define FPT_FREQS 63
#define W_(r, c) Wsh[(r) * (FPT_COLS + 1) + (c)]

extern "C" __global__ __launch_bounds__(512) void trsm_fp_kernel(
    const double* __restrict__ L /*(D,mp,mp)*/,
    const double* __restrict__ invd /*(D, mp/16, 16, 16)*/,
    const double* __restrict__ RHS /*(mp, 2F+1)*/,
    const double* __restrict__ sNs /*(3,F)*/,
    const double* __restrict__ sNr /*(2,F)*/, int mp, int F, int D,
    double* __restrict__ fp /*(D,F)*/) {
  __shared__ double Wsh[FASTFP_MAXMP * (FPT_COLS + 1)];
  __shared__ double Lp[16][FASTFP_MAXMP + 1];
  __shared__ double Iv[16][17];

  const int d = blockIdx.y;
  const int f0 = blockIdx.x * FPT_FREQS;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wv = tid >> 6;
  const int nb = mp >> 4;
  const long ldr = 2L * F + 1;
  const double* Ld = L + (long)d * mp * mp;
  const double* Ivd = invd + (long)d * nb * 256;

  // stage RHS columns [2*f0 .. 2*f0+125] + u into W (zero-fill tail)
  for (int idx = tid; idx < mp * 126; idx += 512) {
    const int r = idx / 126, c = idx % 126;
    const int gc = 2 * f0 + c;
    W_(r, c) = (gc < 2 * F) ? RHS[r * ldr + gc] : 0.0;
  }
  for (int r = tid; r < mp; r += 512) {
    W_(r, 126) = RHS[r * ldr + 2 * F];
    W_(r, 127) = 0.0;
  }
  __syncthreads();

  // blocked forward substitution, row-tile rb at a time
  const int jw = wv * 16;  // this wave's column strip
  for (int rb = 0; rb < nb; ++rb) {
    const int r0 = rb * NB;
    // stage L row-panel (cols 0..r0) and the inverted diag block
    for (int idx = tid; idx < 16 * (r0 > 0 ? r0 : 1); idx += 512) {
      if (r0 == 0) break;
      const int r = idx / r0, c = idx % r0;
      Lp[r][c] = Ld[(long)(r0 + r) * mp + c];
    }
    for (int idx = tid; idx < 256; idx += 512) {
      Iv[idx / 16][idx % 16] = Ivd[rb * 256 + idx];
    }
    __syncthreads();

    // acc = sum_cb L[rb,cb] W[cb]  -  RHS_rb
    f64x4 acc;
#pragma unroll
    for (int v = 0; v < 4; ++v)
      acc[v] = -W_(r0 + 4 * v + (lane >> 4), jw + (lane & 15));
    for (int cb = 0; cb < rb; ++cb) {
#pragma unroll
      for (int kk = 0; kk < 4; ++kk) {
        const double a = Lp[lane & 15][cb * 16 + kk * 4 + (lane >> 4)];
        const double b = W_(cb * 16 + kk * 4 + (lane >> 4), jw + (lane & 15));
        acc = MFMA_F64(a, b, acc);
      }
    }
    __syncthreads();  // everyone done READING W[rb] (acc init) before overwrite
#pragma unroll
    for (int v = 0; v < 4; ++v)
      W_(r0 + 4 * v + (lane >> 4), jw + (lane & 15)) = acc[v];
    __syncthreads();
    // W[rb] <- inv(L_rb,rb) * (RHS - sum) = Iv * (-Wtmp)
    f64x4 sol = {0, 0, 0, 0};
#pragma unroll
    for (int kk = 0; kk < 4; ++kk) {
      const double a = Iv[lane & 15][kk * 4 + (lane >> 4)];
      const double b = -W_(r0 + kk * 4 + (lane >> 4), jw + (lane & 15));
      sol = MFMA_F64(a, b, sol);
    }
    __syncthreads();
#pragma unroll
    for (int v = 0; v < 4; ++v)
      W_(r0 + 4 * v + (lane >> 4), jw + (lane & 15)) = sol[v];
    __syncthreads();
  }

  // fused per-frequency 2x2 reduction; wave w owns freqs w*8 .. w*8+7
  for (int q = wv * 8; q < wv * 8 + 8; ++q) {
    if (q >= FPT_FREQS || f0 + q >= F) continue;
    const int cs = 2 * q, cc_ = 2 * q + 1;
    double gss = 0, gcc = 0, gsc = 0, gsu = 0, gcu = 0;
#pragma unroll
    for (int h = 0; h < FASTFP_MAXMP / 64; ++h) {
      const int r = h * 64 + lane;
      if (r < mp) {
        const double ws = W_(r, cs);
        const double wc = W_(r, cc_);
        const double wu = W_(r, 126);
        gss = fma(ws, ws, gss);
        gcc = fma(wc, wc, gcc);
        gsc = fma(ws, wc, gsc);
        gsu = fma(ws, wu, gsu);
        gcu = fma(wc, wu, gcu);
      }
    }
    gss = wave_reduce_sum(gss);
    gcc = wave_reduce_sum(gcc);
    gsc = wave_reduce_sum(gsc);
    gsu = wave_reduce_sum(gsu);
    gcu = wave_reduce_sum(gcu);
    if (lane == 0) {
      const int f = f0 + q;
      const double M11 = sNs[f] - gss;
      const double M22 = sNs[F + f] - gcc;
      const double M12 = sNs[2 * F + f] - gsc;
      const double N1 = sNr[f] - gsu;
      const double N2 = sNr[F + f] - gcu;
      const double det = fma(M11, M22, -M12 * M12);
      const double num =
          fma(N1 * N1, M22, fma(-2.0 * N1, N2 * M12, N2 * N2 * M11));
      fp[(long)d * F + f] += 0.5 * num / det;
    }
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
  hipLaunchKernelGGL(sbgemm_kernel, dim3(ctiles, ksplit), dim3(256), 0,
                     stream, T, toas, ninv, freqs, ntoa, m, mp, F2, out,
                     plane_stride, ldo);
}

void launch_chol_batch(const double* TNT, const double* phiinv, int m, int mp,
                       int D, double* L, double* invd, hipStream_t stream) {
  hipLaunchKernelGGL(chol_batch_kernel, dim3(D), dim3(256), 0, stream, TNT,
                     phiinv, m, mp, D, L, invd);
}

void launch_trsm_fp(const double* L, const double* invd, const double* RHS,
                    const double* sNs, const double* sNr, int mp, int F,
                    int D, double* fp, hipStream_t stream) {
  const int ftiles = (F + FPT_FREQS - 1) / FPT_FREQS;
  hipLaunchKernelGGL(trsm_fp_kernel, dim3(ftiles, D), dim3(512), 0, stream,
                     L, invd, RHS, sNs, sNr, mp, F, D, fp);
}

}  // extern "C"
