/* Oracle driver: run the REFERENCE lbfgs_fit (libdirac lbfgs.c) on the
 * extended Rosenbrock problem (the reference's own library demo,
 * test/Dirac/demo.c:25-56) and print the solution, so the framework's
 * LBFGS can be cross-validated against the reference implementation.
 * Usage: oracle_lbfgs [m] [itmax]; prints "p0 p1 ... cost". */
#include <stdio.h>
#include <stdlib.h>
#include <Dirac.h>

typedef struct { double alpha; } rb_t;

static double cost(double *p, int m, void *adata) {
  rb_t *t = (rb_t *)adata;
  double f = 0.0;
  for (int i = 0; i < m / 2; i++) {
    double a = p[2 * i + 1] - p[2 * i] * p[2 * i];
    double b = 1.0 - p[2 * i];
    f += t->alpha * a * a + b * b;
  }
  return f;
}

static void grad(double *p, double *g, int m, void *adata) {
  rb_t *t = (rb_t *)adata;
  for (int i = 0; i < m / 2; i++) {
    double a = p[2 * i + 1] - p[2 * i] * p[2 * i];
    g[2 * i] = -4.0 * t->alpha * p[2 * i] * a - 2.0 * (1.0 - p[2 * i]);
    g[2 * i + 1] = 2.0 * t->alpha * a;
  }
}

int main(int argc, char **argv) {
  int m = argc > 1 ? atoi(argv[1]) : 8;
  int itmax = argc > 2 ? atoi(argv[2]) : 200;
  rb_t rt; rt.alpha = 100.0;
  double *p = malloc(m * sizeof(double));
  for (int i = 0; i < m; i++) p[i] = i % 2 ? 1.0 : -1.2;
  lbfgs_fit(cost, grad, p, m, itmax, 7, &rt, NULL);
  for (int i = 0; i < m; i++) printf("%.12e ", p[i]);
  printf("%.12e\n", cost(p, m, &rt));
  free(p);
  return 0;
}
