// Standalone probe: determine the exact lane->element mapping of
// v_mfma_f64_16x16x4f64 on gfx950.  Prints, for each (lane, reg) of the
// D accumulator, the output (row, col), and checks the assumed A/B
// index roles.  Build & run on a GPU box:
//   hipcc --offload-arch=gfx950 -O2 tools/mfma_probe.hip -o /tmp/probe && /tmp/probe
#include <hip/hip_runtime.h>
#include <cstdio>

typedef double f64x4 __attribute__((ext_vector_type(4)));

// hypothesis under test: a holds A[i=l&15][k=l>>4], b holds B[k=l>>4][j=l&15]
__global__ void probe(double* d1, double* d2, double* d3, double* d4) {
  const int l = threadIdx.x;
  const double i_h = (double)(l & 15);
  const double k_h = (double)(l >> 4);
  f64x4 z = {0, 0, 0, 0};
  // D1: a = i (hyp), b = 1  ->  D1[r][c] = 4*r  if a's i-index = l&15 = row
  f64x4 r1 = __builtin_amdgcn_mfma_f64_16x16x4f64(i_h, 1.0, z, 0, 0, 0);
  // D2: a = 1, b = j (hyp)  ->  D2[r][c] = 4*c  if b's j-index = l&15 = col
  f64x4 r2 = __builtin_amdgcn_mfma_f64_16x16x4f64(1.0, i_h, z, 0, 0, 0);
  // D3: a = k (hyp), b = 1  ->  D3 = 0+1+2+3 = 6 everywhere if a's k = l>>4
  f64x4 r3 = __builtin_amdgcn_mfma_f64_16x16x4f64(k_h, 1.0, z, 0, 0, 0);
  // D4: a = 1, b = k (hyp)  ->  6 everywhere if b's k = l>>4
  f64x4 r4 = __builtin_amdgcn_mfma_f64_16x16x4f64(1.0, k_h, z, 0, 0, 0);
  for (int v = 0; v < 4; ++v) {
    d1[l * 4 + v] = r1[v];
    d2[l * 4 + v] = r2[v];
    d3[l * 4 + v] = r3[v];
    d4[l * 4 + v] = r4[v];
  }
}

int main() {
  double *d1, *d2, *d3, *d4;
  hipMalloc(&d1, 256 * 8); hipMalloc(&d2, 256 * 8);
  hipMalloc(&d3, 256 * 8); hipMalloc(&d4, 256 * 8);
  hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, d1, d2, d3, d4);
  double h1[256], h2[256], h3[256], h4[256];
  hipMemcpy(h1, d1, 2048, hipMemcpyDeviceToHost);
  hipMemcpy(h2, d2, 2048, hipMemcpyDeviceToHost);
  hipMemcpy(h3, d3, 2048, hipMemcpyDeviceToHost);
  hipMemcpy(h4, d4, 2048, hipMemcpyDeviceToHost);
  hipDeviceSynchronize();
  printf("lane reg : row(D1/4) col(D2/4) aK(D3==6) bK(D4==6)\n");
  int okA = 1, okB = 1;
  for (int l = 0; l < 64; ++l)
    for (int v = 0; v < 4; ++v) {
      double row = h1[l * 4 + v] / 4.0, col = h2[l * 4 + v] / 4.0;
      if (h3[l * 4 + v] != 6.0) okA = 0;
      if (h4[l * 4 + v] != 6.0) okB = 0;
      printf("%2d %d : %5.2f %5.2f %4.1f %4.1f\n", l, v, row, col,
             h3[l * 4 + v], h4[l * 4 + v]);
    }
  printf("A k-index hypothesis (l>>4): %s\n", okA ? "OK" : "WRONG");
  printf("B k-index hypothesis (l>>4): %s\n", okB ? "OK" : "WRONG");
  return 0;
}
