/* Oracle driver for the REFERENCE consensus-polynomial machinery
 * (consensus_poly.c): the frequency basis functions (types 0-3 incl.
 * Bernstein) and the Barzilai-Borwein penalty update, so
 * consensus/poly.py can be cross-validated.
 *
 * Usage:
 *   oracle_poly basis <Npoly> <Nf> <type> <freq0>  < freqs.txt
 *       -> Nf lines of Npoly basis values (freq-major, B[f*Npoly+p])
 *   oracle_poly bii <Npoly> <Nf> <M> <type> <freq0> < input.txt
 *       input: Nf freqs, then Nf*M rho values (rho[k + f*M])
 *       -> M blocks of Npoly lines x Npoly values (cluster pinverses)
 *   oracle_poly rhobb <N> <M> <rho_upper> < input.txt
 *       input: M lines "rho_ci nchunk_ci", then 4 blocks of
 *       8*N*Mt values: Yhat, Yhat_prev, J, J_prev
 *       -> M updated rho values
 */
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <Dirac.h>

int main(int argc, char **argv) {
  if (argc < 2) return 1;
  if (!strcmp(argv[1], "basis")) {
    int Npoly = atoi(argv[2]), Nf = atoi(argv[3]), ty = atoi(argv[4]);
    double f0 = atof(argv[5]);
    double *freqs = malloc(Nf * sizeof(double));
    for (int i = 0; i < Nf; i++)
      if (scanf("%lf", &freqs[i]) != 1) return 2;
    double *B = calloc((size_t)Npoly * Nf, sizeof(double));
    setup_polynomials(B, Npoly, Nf, freqs, f0, ty);
    for (int f = 0; f < Nf; f++) {
      for (int p = 0; p < Npoly; p++) printf("%.15e ", B[f * Npoly + p]);
      printf("\n");
    }
  } else if (!strcmp(argv[1], "bii")) {
    int Npoly = atoi(argv[2]), Nf = atoi(argv[3]), M = atoi(argv[4]);
    int ty = atoi(argv[5]);
    double f0 = atof(argv[6]);
    double *freqs = malloc(Nf * sizeof(double));
    for (int i = 0; i < Nf; i++)
      if (scanf("%lf", &freqs[i]) != 1) return 2;
    double *rho = malloc((size_t)Nf * M * sizeof(double));
    for (int i = 0; i < Nf * M; i++)
      if (scanf("%lf", &rho[i]) != 1) return 2;
    double *B = calloc((size_t)Npoly * Nf, sizeof(double));
    double *Bi = calloc((size_t)Npoly * Npoly * M, sizeof(double));
    setup_polynomials(B, Npoly, Nf, freqs, f0, ty);
    find_prod_inverse_full(B, Bi, Npoly, Nf, M, rho, 1);
    for (int k = 0; k < M; k++)
      for (int r = 0; r < Npoly; r++) {
        for (int c = 0; c < Npoly; c++)
          printf("%.15e ", Bi[k * Npoly * Npoly + r * Npoly + c]);
        printf("\n");
      }
  } else if (!strcmp(argv[1], "rhobb")) {
    int N = atoi(argv[2]), M = atoi(argv[3]);
    double rup = atof(argv[4]);
    double *rho = malloc(M * sizeof(double));
    double *rhoupper = malloc(M * sizeof(double));
    clus_source_t *carr = calloc(M, sizeof(clus_source_t));
    int Mt = 0;
    for (int ci = 0; ci < M; ci++) {
      if (scanf("%lf %d", &rho[ci], &carr[ci].nchunk) != 2) return 3;
      rhoupper[ci] = rup;
      Mt += carr[ci].nchunk;
    }
    size_t L = (size_t)8 * N * Mt;
    double *Yh = malloc(L * sizeof(double));
    double *Yh0 = malloc(L * sizeof(double));
    double *J = malloc(L * sizeof(double));
    double *J0 = malloc(L * sizeof(double));
    double *bufs[4] = {Yh, Yh0, J, J0};
    for (int b = 0; b < 4; b++)
      for (size_t i = 0; i < L; i++)
        if (scanf("%lf", &bufs[b][i]) != 1) return 4;
    update_rho_bb(rho, rhoupper, N, M, Mt, carr, Yh, Yh0, J, J0, 1);
    for (int ci = 0; ci < M; ci++) printf("%.15e ", rho[ci]);
    printf("\n");
  } else {
    return 1;
  }
  return 0;
}
