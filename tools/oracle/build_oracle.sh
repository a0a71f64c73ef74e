#!/bin/bash
# Build the reference-CPU oracle binaries (SURVEY.md §6 / VERDICT r1
# item 6): compiles the reference's lbfgs.c, myblas.c and predict.c
# unmodified against a minimal local BLAS (miniblas.c) and links two
# tiny drivers. Usage: bash tools/oracle/build_oracle.sh [outdir]
set -e
REF=${REF:-/root/reference}
HERE=$(cd "$(dirname "$0")" && pwd)
OUT=${1:-/tmp/sagecal_oracle}
mkdir -p "$OUT"
CFLAGS="-O2 -fcommon -I$REF/src/lib/Dirac -I$REF/src/lib/Radio"
gcc $CFLAGS -c "$REF/src/lib/Dirac/lbfgs.c" -o "$OUT/lbfgs.o"
gcc $CFLAGS -c "$REF/src/lib/Dirac/myblas.c" -o "$OUT/myblas.o"
gcc $CFLAGS -c "$REF/src/lib/Radio/predict.c" -o "$OUT/predict.o"
gcc $CFLAGS -c "$HERE/miniblas.c" -o "$OUT/miniblas.o"
gcc $CFLAGS "$HERE/oracle_lbfgs.c" "$OUT/lbfgs.o" "$OUT/myblas.o" \
    "$OUT/miniblas.o" -lpthread -lm -o "$OUT/oracle_lbfgs"
gcc $CFLAGS "$HERE/oracle_predict.c" "$OUT/predict.o" "$OUT/myblas.o" \
    "$OUT/miniblas.o" -lpthread -lm -o "$OUT/oracle_predict"
gcc $CFLAGS -c "$REF/src/lib/Radio/elementbeam.c" -o "$OUT/elementbeam.o"
gcc $CFLAGS -c "$REF/src/lib/Radio/transforms.c" -o "$OUT/transforms.o"
gcc $CFLAGS -c "$REF/src/lib/Dirac/updatenu.c" -o "$OUT/updatenu.o"
gcc $CFLAGS -c "$REF/src/lib/Radio/shapelet.c" -o "$OUT/shapelet.o"
gcc $CFLAGS "$HERE/oracle_element.c" "$OUT/elementbeam.o" \
    "$OUT/myblas.o" "$OUT/miniblas.o" -lpthread -lm \
    -o "$OUT/oracle_element"
gcc $CFLAGS -c "$REF/src/lib/Dirac/consensus_poly.c" -o "$OUT/consensus_poly.o"
gcc $CFLAGS -c "$REF/src/lib/Radio/residual.c" -o "$OUT/residual.o"
gcc $CFLAGS -c "$REF/src/lib/Dirac/manifold_average.c" -o "$OUT/manifold_average.o" 2>/dev/null || true
gcc $CFLAGS "$HERE/oracle_poly.c" "$OUT/consensus_poly.o" \
    "$OUT/myblas.o" "$OUT/miniblas.o" -lpthread -lm -o "$OUT/oracle_poly"
gcc $CFLAGS "$HERE/oracle_residual.c" "$OUT/residual.o" \
    "$OUT/predict.o" "$OUT/shapelet.o" "$OUT/elementbeam.o" \
    "$OUT/manifold_average.o" "$OUT/myblas.o" "$OUT/miniblas.o" \
    -lpthread -lm -o "$OUT/oracle_residual" 2>/dev/null || \
  gcc $CFLAGS "$HERE/oracle_residual.c" "$OUT/residual.o" \
    "$OUT/predict.o" "$OUT/shapelet.o" "$OUT/elementbeam.o" \
    "$OUT/myblas.o" "$OUT/miniblas.o" \
    -lpthread -lm -o "$OUT/oracle_residual"
gcc $CFLAGS "$HERE/oracle_misc.c" "$OUT/transforms.o" "$OUT/updatenu.o" \
    "$OUT/shapelet.o" "$OUT/elementbeam.o" "$OUT/myblas.o" \
    "$OUT/miniblas.o" -lpthread -lm -o "$OUT/oracle_misc"
echo "built: $OUT/oracle_lbfgs $OUT/oracle_predict $OUT/oracle_element $OUT/oracle_misc $OUT/oracle_poly $OUT/oracle_residual"
