// Empirical fragment-layout probe for v_mfma_f64_16x16x4f64 on gfx950.
// Computes D = A*B (16x16x4) with the ASSUMED staging
//   a(lane) = A[lane%16][lane/16],  b(lane) = B[lane/16][lane%16]
// then, against a host reference, searches for the (lane, reg) -> (row, col)
// mapping of D. If every raw value matches exactly one reference entry the
// assumed A/B layouts are consistent and the printed map is the full answer.
// Build: hipcc --offload-arch=gfx950 -O2 probe_mfma_f64.hip -o probe && ./probe
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>

typedef double v4d __attribute__((ext_vector_type(4)));

__global__ void probe(const double* A, const double* B, double* Dout) {
  int l = threadIdx.x;
  double a = A[(l % 16) * 4 + (l / 16)];   // A[i][k], row-major 16x4
  double b = B[(l / 16) * 16 + (l % 16)];  // B[k][j], row-major 4x16
  v4d acc = {0, 0, 0, 0};
  acc = __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, acc, 0, 0, 0);
  for (int r = 0; r < 4; ++r) Dout[l * 4 + r] = acc[r];
}

int main() {
  double hA[16 * 4], hB[4 * 16], hD[16 * 16], hRaw[64 * 4];
  srand(12345);
  for (int i = 0; i < 64; ++i) hA[i] = (double)(rand() % 997);
  for (int i = 0; i < 64; ++i) hB[i] = (double)(rand() % 983);
  for (int i = 0; i < 16; ++i)
    for (int j = 0; j < 16; ++j) {
      double s = 0;
      for (int k = 0; k < 4; ++k) s += hA[i * 4 + k] * hB[k * 16 + j];
      hD[i * 16 + j] = s;
    }
  double *dA, *dB, *dD;
  hipMalloc(&dA, sizeof(hA));
  hipMalloc(&dB, sizeof(hB));
  hipMalloc(&dD, sizeof(hRaw));
  hipMemcpy(dA, hA, sizeof(hA), hipMemcpyHostToDevice);
  hipMemcpy(dB, hB, sizeof(hB), hipMemcpyHostToDevice);
  probe<<<1, 64>>>(dA, dB, dD);
  hipMemcpy(hRaw, dD, sizeof(hRaw), hipMemcpyDeviceToHost);

  int unmatched = 0;
  // try to express the found map as row = f(l, r), col = g(l, r)
  int row_of[64][4], col_of[64][4];
  for (int l = 0; l < 64; ++l)
    for (int r = 0; r < 4; ++r) {
      row_of[l][r] = col_of[l][r] = -1;
      int hits = 0;
      for (int i = 0; i < 16; ++i)
        for (int j = 0; j < 16; ++j)
          if (hRaw[l * 4 + r] == hD[i * 16 + j]) {
            row_of[l][r] = i;
            col_of[l][r] = j;
            ++hits;
          }
      if (hits != 1) {
        ++unmatched;
        if (unmatched < 5)
          printf("lane %d reg %d: %d matches (raw=%f)\n", l, r, hits,
                 hRaw[l * 4 + r]);
      }
    }
  printf("unmatched: %d / 256\n", unmatched);
  if (!unmatched) {
    // check candidate formulas
    int ok1 = 1, ok2 = 1, ok3 = 1, ok4 = 1;
    for (int l = 0; l < 64; ++l)
      for (int r = 0; r < 4; ++r) {
        if (!(col_of[l][r] == l % 16 && row_of[l][r] == (l / 16) * 4 + r)) ok1 = 0;
        if (!(col_of[l][r] == l % 16 && row_of[l][r] == (l / 16) + 4 * r)) ok2 = 0;
        if (!(row_of[l][r] == l % 16 && col_of[l][r] == (l / 16) * 4 + r)) ok3 = 0;
        if (!(row_of[l][r] == l % 16 && col_of[l][r] == (l / 16) + 4 * r)) ok4 = 0;
      }
    printf("formula f32-style (col=l%%16, row=4*(l/16)+r): %s\n", ok1 ? "MATCH" : "no");
    printf("formula strided   (col=l%%16, row=(l/16)+4*r): %s\n", ok2 ? "MATCH" : "no");
    printf("formula T f32     (row=l%%16, col=4*(l/16)+r): %s\n", ok3 ? "MATCH" : "no");
    printf("formula T strided (row=l%%16, col=(l/16)+4*r): %s\n", ok4 ? "MATCH" : "no");
    if (!ok1 && !ok2 && !ok3 && !ok4) {
      for (int l = 0; l < 64; ++l)
        for (int r = 0; r < 4; ++r)
          printf("l=%d r=%d -> (%d,%d)\n", l, r, row_of[l][r], col_of[l][r]);
    }
  }
  return 0;
}
