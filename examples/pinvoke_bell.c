/* pinvoke-compat example: the reference simulator's C API served by
 * qrack_amd. Build (the extension .so IS the library):
 *
 *   gcc examples/pinvoke_bell.c -Iinclude \
 *       -L qrack_amd -l:_qrack.cpython-310-x86_64-linux-gnu.so \
 *       -Wl,-rpath,'$ORIGIN/../qrack_amd' -lpython3.10 -o pinvoke_bell
 *
 * Prepares a GHZ state, checks probabilities and the <Z x Z x Z>-adjacent
 * product expectations, measures shots — using only reference-named calls.
 */
#include "qrack_pinvoke_compat.h"

#include <math.h>
#include <stdio.h>
#include <stdlib.h>

int main(void)
{
    const uintq sid = init_count(3, false, false);
    seed(sid, 42);

    H(sid, 0);
    uintq c0[1] = { 0 };
    MCX(sid, 1, c0, 1);
    uintq c1[1] = { 1 };
    MCX(sid, 1, c1, 2);

    const double p2 = Prob(sid, 2);
    printf("P(q2=1) = %.6f\n", p2);
    if (fabs(p2 - 0.5) > 1e-6) return 1;

    uintq qs[2] = { 0, 2 };
    uintq zz[2] = { 2, 2 };
    const double e = PauliExpectation(sid, 2, qs, zz);
    printf("<Z0 Z2>  = %.6f\n", e);
    if (fabs(e - 1.0) > 1e-5) return 2;

    uintq all[3] = { 0, 1, 2 };
    uintq shots[64];
    MeasureShots(sid, 3, all, 64, shots);
    int zeros = 0, sevens = 0;
    for (int i = 0; i < 64; ++i) {
        if (shots[i] == 0) ++zeros;
        if (shots[i] == 7) ++sevens;
    }
    printf("shots: %d x |000>, %d x |111>\n", zeros, sevens);
    if (zeros + sevens != 64) return 3;

    if (get_error(sid)) return 4;
    destroy(sid);
    printf("OK\n");
    return 0;
}
