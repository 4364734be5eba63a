/* C-ABI distributed test: 1x2 grid POTRF through libdlaf_c.so.
 *
 * Launched as two processes with the torchrun-style rendezvous env
 * (RANK / WORLD_SIZE / MASTER_ADDR / MASTER_PORT) by tests/test_capi_c.py;
 * the embedded runtime initializes torch.distributed (gloo on CPU).
 * Each rank owns the block-cyclic column panels of a small SPD matrix,
 * factorizes through dlaf_cholesky_factorization_d, and checks its OWNED
 * entries against a full reference Cholesky computed locally in C.
 * Reference counterpart: the multi-rank grids of src/c_api/grid.cpp +
 * test/unit/c_api/. */
#include <math.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include "dlaf_c.h"

static double elem(int i, int j, int n) {
  double v = ((i * 31 + j * 17) % 13) / 13.0;
  double w = ((j * 31 + i * 17) % 13) / 13.0;
  return 0.5 * (v + w) + (i == j ? 2.0 * n : 0.0);
}

int main(void) {
  const int n = 96, nb = 16;
  const char* rank_s = getenv("RANK");
  const char* world_s = getenv("WORLD_SIZE");
  if (!rank_s || !world_s) {
    fprintf(stderr, "needs RANK/WORLD_SIZE env\n");
    return 10;
  }
  const int rank = atoi(rank_s);
  const int npcol = atoi(world_s); /* 1 x WORLD_SIZE grid */

  if (dlaf_initialize(0, NULL) != 0) return 1;
  int ctx = dlaf_create_grid(1, npcol, 'R');
  if (ctx < 0) return 2;

  /* rank-local panel: all rows, the column tiles j with (j/nb) % npcol == rank */
  int ntc = (n + nb - 1) / nb;
  int lcols = 0;
  for (int tj = 0; tj < ntc; ++tj)
    if (tj % npcol == rank) {
      int w = (tj + 1) * nb <= n ? nb : n - tj * nb;
      lcols += w;
    }
  const int ld = n;
  double* a = calloc((size_t)ld * lcols, sizeof(double));
  int lc = 0;
  for (int tj = 0; tj < ntc; ++tj) {
    if (tj % npcol != rank) continue;
    int w = (tj + 1) * nb <= n ? nb : n - tj * nb;
    for (int c = 0; c < w; ++c, ++lc)
      for (int i = 0; i < n; ++i)
        a[i + (long)lc * ld] = elem(i, tj * nb + c, n);
  }

  struct DLAF_descriptor d = {n, n, nb, nb, 0, 0, 1, 1, ld};
  int rc = dlaf_cholesky_factorization_d(ctx, 'L', a, d);
  if (rc != 0) return 3;

  /* full reference Cholesky in C */
  double* f = malloc((size_t)n * n * sizeof(double));
  for (int j = 0; j < n; ++j)
    for (int i = 0; i < n; ++i) f[i + (long)j * n] = elem(i, j, n);
  for (int k = 0; k < n; ++k) {
    f[k + (long)k * n] = sqrt(f[k + (long)k * n]);
    for (int i = k + 1; i < n; ++i) f[i + (long)k * n] /= f[k + (long)k * n];
    for (int j = k + 1; j < n; ++j)
      for (int i = j; i < n; ++i)
        f[i + (long)j * n] -= f[i + (long)k * n] * f[j + (long)k * n];
  }

  double err = 0.0;
  lc = 0;
  for (int tj = 0; tj < ntc; ++tj) {
    if (tj % npcol != rank) continue;
    int w = (tj + 1) * nb <= n ? nb : n - tj * nb;
    for (int c = 0; c < w; ++c, ++lc) {
      int gj = tj * nb + c;
      for (int i = gj; i < n; ++i) { /* lower triangle only */
        double e = fabs(a[i + (long)lc * ld] - f[i + (long)gj * n]);
        if (e > err) err = e;
      }
    }
  }
  printf("rank %d dist potrf err %.3e\n", rank, err);
  if (err > 1e-11 * n) return 4;
  printf("OK rank %d\n", rank);
  /* NOTE: skip dlaf_finalize-side teardown races by exiting promptly */
  fflush(stdout);
  return 0;
}
