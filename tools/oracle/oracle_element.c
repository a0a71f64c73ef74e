/* Oracle driver: evaluate the REFERENCE element beam
 * (set_elementcoeffs + eval_elementcoeffs, elementbeam.c:40/384) at
 * given (zenith-angle, basis-azimuth) points so the ported coefficient
 * tables + Laguerre-Gaussian basis (beams.LofarElementCoeffs) can be
 * cross-validated value-by-value.
 * Usage: oracle_element <lba|hba|alo> <freq_hz> < points.txt
 *   points.txt: lines of "r theta"
 * Output: per line "re(theta) im(theta) re(phi) im(phi)". */
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <complex.h>
#include <Dirac.h>
#include <Dirac_radio.h>

int main(int argc, char **argv) {
  if (argc < 3) { fprintf(stderr, "usage: %s type freq_hz\n", argv[0]);
                  return 1; }
  int ty = ELEM_LBA;
  if (!strcmp(argv[1], "hba")) ty = ELEM_HBA;
  else if (!strcmp(argv[1], "alo")) ty = ELEM_ALO;
  double f = atof(argv[2]);
  elementcoeff ec;
  set_elementcoeffs(ty, f, &ec);
  double r, t;
  while (scanf("%lf %lf", &r, &t) == 2) {
    elementval v = eval_elementcoeffs(r, t, &ec);
    printf("%.15e %.15e %.15e %.15e\n", creal(v.theta), cimag(v.theta),
           creal(v.phi), cimag(v.phi));
  }
  free_elementcoeffs(ec);
  return 0;
}
