/* Oracle driver: run the REFERENCE calculate_residuals_multifreq
 * (residual.c:940 — per-channel re-predict with solutions applied,
 * spectral-index flux scaling, optional MMSE correction by cluster
 * ccid) on a layout + solutions read from a file, printing the
 * residuals. Pins sagecal_amd's residual path AND the 8-reals-per-
 * station solution parameter ordering to the reference.
 *
 * Input (whitespace separated):
 *   N Nbase tilesz M Nchan fdelta tdelta dec0 ccid rho
 *   Nchan x freq
 *   Nbase*tilesz x (sta1 sta2 u v w)
 *   M x { id nsrc
 *         nsrc x (type ll mm nn sI sQ sU sV eX eY eP
 *                 cxi sxi cphi sphi use_proj
 *                 f0 spec_idx spec_idx1 spec_idx2 sI0 sQ0 sU0 sV0) }
 *   M*8N x p           (solutions, cluster-major, nchunk=1)
 *   Nbase*8*tilesz*Nchan x data
 * Output: the residual vector, one value per line.
 */
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <complex.h>
#include <Dirac.h>
#include <Dirac_radio.h>

int main(int argc, char **argv) {
  if (argc < 2) { fprintf(stderr, "usage: %s layout.txt\n", argv[0]);
                  return 1; }
  FILE *f = fopen(argv[1], "r");
  if (!f) { perror("layout"); return 1; }
  int N, Nbase, tilesz, M, Nchan, ccid;
  double fdelta, tdelta, dec0, rho;
  if (fscanf(f, "%d %d %d %d %d %lf %lf %lf %d %lf", &N, &Nbase,
             &tilesz, &M, &Nchan, &fdelta, &tdelta, &dec0, &ccid,
             &rho) != 10) return 2;
  double *freqs = malloc(Nchan * sizeof(double));
  for (int i = 0; i < Nchan; i++)
    if (fscanf(f, "%lf", &freqs[i]) != 1) return 2;
  int rows = Nbase * tilesz;
  double *u = malloc(rows * sizeof(double));
  double *v = malloc(rows * sizeof(double));
  double *w = malloc(rows * sizeof(double));
  baseline_t *barr = calloc(rows, sizeof(baseline_t));
  for (int b = 0; b < rows; b++)
    if (fscanf(f, "%d %d %lf %lf %lf", &barr[b].sta1, &barr[b].sta2,
               &u[b], &v[b], &w[b]) != 5) return 3;
  clus_source_t *carr = calloc(M, sizeof(clus_source_t));
  for (int ci = 0; ci < M; ci++) {
    int ns;
    if (fscanf(f, "%d %d", &carr[ci].id, &ns) != 2) return 4;
    carr[ci].N = ns;
    carr[ci].nchunk = 1;
    carr[ci].p = malloc(sizeof(int));
    carr[ci].p[0] = ci * 8 * N;
    carr[ci].ll = malloc(ns * sizeof(double));
    carr[ci].mm = malloc(ns * sizeof(double));
    carr[ci].nn = malloc(ns * sizeof(double));
    carr[ci].sI = malloc(ns * sizeof(double));
    carr[ci].sQ = malloc(ns * sizeof(double));
    carr[ci].sU = malloc(ns * sizeof(double));
    carr[ci].sV = malloc(ns * sizeof(double));
    carr[ci].sI0 = malloc(ns * sizeof(double));
    carr[ci].sQ0 = malloc(ns * sizeof(double));
    carr[ci].sU0 = malloc(ns * sizeof(double));
    carr[ci].sV0 = malloc(ns * sizeof(double));
    carr[ci].f0 = malloc(ns * sizeof(double));
    carr[ci].spec_idx = malloc(ns * sizeof(double));
    carr[ci].spec_idx1 = malloc(ns * sizeof(double));
    carr[ci].spec_idx2 = malloc(ns * sizeof(double));
    carr[ci].stype = malloc(ns);
    carr[ci].ex = calloc(ns, sizeof(void *));
    for (int s = 0; s < ns; s++) {
      int ty, up;
      double eX, eY, eP, cxi, sxi, cphi, sphi;
      if (fscanf(f,
                 "%d %lf %lf %lf %lf %lf %lf %lf %lf %lf %lf "
                 "%lf %lf %lf %lf %d %lf %lf %lf %lf %lf %lf %lf %lf",
                 &ty, &carr[ci].ll[s], &carr[ci].mm[s], &carr[ci].nn[s],
                 &carr[ci].sI[s], &carr[ci].sQ[s], &carr[ci].sU[s],
                 &carr[ci].sV[s], &eX, &eY, &eP, &cxi, &sxi, &cphi,
                 &sphi, &up, &carr[ci].f0[s], &carr[ci].spec_idx[s],
                 &carr[ci].spec_idx1[s], &carr[ci].spec_idx2[s],
                 &carr[ci].sI0[s], &carr[ci].sQ0[s], &carr[ci].sU0[s],
                 &carr[ci].sV0[s]) != 24) return 5;
      carr[ci].stype[s] = (unsigned char)ty;
      if (ty == STYPE_GAUSSIAN) {
        exinfo_gaussian *g = calloc(1, sizeof(exinfo_gaussian));
        g->eX = eX; g->eY = eY; g->eP = eP;
        g->cxi = cxi; g->sxi = sxi; g->cphi = cphi; g->sphi = sphi;
        g->use_projection = up;
        carr[ci].ex[s] = g;
      }
    }
  }
  double *p = malloc((size_t)M * 8 * N * sizeof(double));
  for (int i = 0; i < M * 8 * N; i++)
    if (fscanf(f, "%lf", &p[i]) != 1) return 6;
  size_t xn = (size_t)rows * 8 * Nchan;
  double *x = malloc(xn * sizeof(double));
  for (size_t i = 0; i < xn; i++)
    if (fscanf(f, "%lf", &x[i]) != 1) return 7;
  fclose(f);
  calculate_residuals_multifreq(u, v, w, p, x, N, Nbase, tilesz, barr,
                                carr, M, freqs, Nchan, fdelta, tdelta,
                                dec0, 2, ccid, rho, 0);
  for (size_t i = 0; i < xn; i++) printf("%.15e\n", x[i]);
  return 0;
}
