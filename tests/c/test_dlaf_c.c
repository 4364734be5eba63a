/* C-ABI smoke test: POTRF + POTRI + SYEVD through libdlaf_c.so. */
#include <math.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include "dlaf_c.h"

static void make_spd(double* a, int n, int ld) {
  for (int j = 0; j < n; ++j)
    for (int i = 0; i < n; ++i) {
      double v = ((i * 31 + j * 17) % 13) / 13.0;
      double w = ((j * 31 + i * 17) % 13) / 13.0;
      a[i + (long)j * ld] = 0.5 * (v + w) + (i == j ? 2.0 * n : 0.0);
    }
}

int main(void) {
  const int n = 96, nb = 32, ld = n;
  if (dlaf_initialize(0, NULL) != 0) return 1;
  int ctx = dlaf_create_grid(1, 1, 'R');
  if (ctx < 0) return 2;

  struct DLAF_descriptor d = {n, n, nb, nb, 0, 0, 1, 1, ld};
  double* a0 = malloc((size_t)n * ld * sizeof(double));
  double* a = malloc((size_t)n * ld * sizeof(double));
  make_spd(a0, n, ld);
  memcpy(a, a0, (size_t)n * ld * sizeof(double));

  if (dlaf_cholesky_factorization_d(ctx, 'L', a, d) != 0) return 3;
  /* check ||L L^T - A||_max on the lower triangle */
  double err = 0.0;
  for (int j = 0; j < n; ++j)
    for (int i = j; i < n; ++i) {
      double s = 0.0;
      for (int k = 0; k <= j; ++k)
        s += a[i + (long)k * ld] * a[j + (long)k * ld];
      double e = fabs(s - a0[i + (long)j * ld]);
      if (e > err) err = e;
    }
  printf("potrf residual %.3e\n", err);
  if (err > 1e-10 * n) return 4;

  /* POTRI: continue to the inverse, check A * Ainv = I (lower stored) */
  if (dlaf_inverse_from_cholesky_factor_d(ctx, 'L', a, d) != 0) return 5;
  double ierr = 0.0;
  for (int j = 0; j < n; j += 7)
    for (int i = 0; i < n; i += 5) {
      double s = 0.0;
      for (int k = 0; k < n; ++k) {
        double inv_ik = (i >= k) ? a[i + (long)k * ld] : a[k + (long)i * ld];
        s += a0[k + (long)j * ld] * inv_ik;
      }
      double e = fabs(s - (i == j ? 1.0 : 0.0));
      if (e > ierr) ierr = e;
    }
  printf("potri residual %.3e\n", ierr);
  if (ierr > 1e-8 * n) return 6;

  /* SYEVD: A = Z diag(w) Z^T */
  double* w = malloc(n * sizeof(double));
  double* z = malloc((size_t)n * ld * sizeof(double));
  memcpy(a, a0, (size_t)n * ld * sizeof(double));
  if (dlaf_symmetric_eigensolver_d(ctx, 'L', a, d, w, z, d) != 0) return 7;
  double eerr = 0.0;
  for (int j = 0; j < n; j += 7)
    for (int i = 0; i < n; i += 5) {
      double s = 0.0;
      for (int k = 0; k < n; ++k)
        s += z[i + (long)k * ld] * w[k] * z[j + (long)k * ld];
      double e = fabs(s - a0[i + (long)j * ld]);
      if (e > eerr) eerr = e;
    }
  printf("syevd residual %.3e  w[0]=%.6f w[n-1]=%.6f\n", eerr, w[0], w[n - 1]);
  if (eerr > 1e-9 * n * n) return 8;
  for (int k = 1; k < n; ++k)
    if (w[k] < w[k - 1]) return 9;

  dlaf_free_grid(ctx);
  dlaf_finalize();
  printf("OK\n");
  return 0;
}
