/* Oracle driver for three more REFERENCE components:
 *   coords:   jd2gmst / radec2azel / precession (transforms.c)
 *   nu:       update_nu AECM grid search (updatenu.c:  Dirac.h:816)
 *   shapelet: shapelet_contrib uv envelope (shapelet.c:141)
 * so coords.py, ops.reference.update_nu_aecm and shapelet.py can be
 * cross-validated against the reference implementation.
 *
 * Usage:
 *   oracle_misc gmst <jd>
 *   oracle_misc azel <ra> <dec> <lon> <lat> <jd>
 *   oracle_misc precess <ra0> <dec0> <jd_tdb>
 *   oracle_misc nu <sumlogw> <Nd> <nulow> <nuhigh> <p> <nu0>
 *   oracle_misc shapelet <n0> <beta> <eX> <eY> <eP> <modes-file> \
 *       < points.txt          (lines: u v w; modes-file: n0*n0 values)
 */
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <complex.h>
#include <Dirac.h>
#include <Dirac_radio.h>

int main(int argc, char **argv) {
  if (argc < 2) return 1;
  if (!strcmp(argv[1], "gmst")) {
    double g;
    jd2gmst(atof(argv[2]), &g);
    printf("%.15e\n", g);
  } else if (!strcmp(argv[1], "azel")) {
    /* radec2azel() has a debug printf in the reference; go through
       jd2gmst + radec2azel_gmst (same math, clean stdout) */
    double az, el, g;
    jd2gmst(atof(argv[6]), &g);
    radec2azel_gmst(atof(argv[2]), atof(argv[3]), atof(argv[4]),
                    atof(argv[5]), g, &az, &el);
    printf("%.15e %.15e\n", az, el);
  } else if (!strcmp(argv[1], "pmatrix")) {
    double Tr[9];
    get_precession_params(atof(argv[2]), Tr);
    for (int i = 0; i < 9; i++) printf("%.15e ", Tr[i]);
    printf("\n");
  } else if (!strcmp(argv[1], "precess")) {
    double Tr[9], ra, dec;
    get_precession_params(atof(argv[4]), Tr);
    precession(atof(argv[2]), atof(argv[3]), Tr, &ra, &dec);
    printf("%.15e %.15e\n", ra, dec);
  } else if (!strcmp(argv[1], "nu")) {
    double nu = update_nu(atof(argv[2]), atoi(argv[3]), 1,
                          atof(argv[4]), atof(argv[5]), atoi(argv[6]),
                          atof(argv[7]));
    printf("%.15e\n", nu);
  } else if (!strcmp(argv[1], "shapelet")) {
    exinfo_shapelet ex;
    memset(&ex, 0, sizeof(ex));
    ex.n0 = atoi(argv[2]);
    ex.beta = atof(argv[3]);
    ex.eX = atof(argv[4]);
    ex.eY = atof(argv[5]);
    ex.eP = atof(argv[6]);
    ex.use_projection = 0;
    int M = ex.n0 * ex.n0;
    ex.modes = malloc(M * sizeof(double));
    FILE *f = fopen(argv[7], "r");
    if (!f) { perror("modes"); return 2; }
    for (int i = 0; i < M; i++)
      if (fscanf(f, "%lf", &ex.modes[i]) != 1) return 3;
    fclose(f);
    double u, v, w;
    while (scanf("%lf %lf %lf", &u, &v, &w) == 3) {
      complex double c = shapelet_contrib(&ex, u, v, w);
      printf("%.15e %.15e\n", creal(c), cimag(c));
    }
  } else {
    fprintf(stderr, "unknown mode %s\n", argv[1]);
    return 1;
  }
  return 0;
}
