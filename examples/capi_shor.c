/* Shor order-finding for N=15, a=7 through the qrack_amd C ABI alone
 * (parity demo: the reference's pinvoke consumers — PyQrack, Q# — use
 * exactly this surface; include/qrack_amd_capi.h).
 *
 * Build (the extension .so exports the C ABI):
 *   gcc -O2 examples/capi_shor.c -o build/capi_shor \
 *       -Iinclude qrack_amd/_qrack.cpython-310-x86_64-linux-gnu.so \
 *       -Wl,-rpath,$PWD/qrack_amd
 * Run: ./build/capi_shor
 */
#include "qrack_amd_capi.h"

#include <stdio.h>
#include <stdlib.h>

int main(void)
{
    const int nCtrl = 8, nWork = 8;
    const uint64_t N = 15, a = 7;

    quid sid = qrack_init_count_type(nCtrl + nWork, 0, 1, 1, 0, 0, 0, 0, 0, 0);
    if (!sid) {
        fprintf(stderr, "init failed\n");
        return 1;
    }
    for (int i = 0; i < nCtrl; ++i) qrack_h(sid, i);
    /* |x>|0> -> |x>|a^x mod N> */
    qrack_pown(sid, a, N, 0, nCtrl, nCtrl);
    /* the no-terminal-swap QFT convention: forward QFT + bit-reversed read
     * implements the phase-estimation inverse transform */
    qrack_qft(sid, 0, nCtrl);

    uint64_t qubits[8];
    for (int i = 0; i < nCtrl; ++i) qubits[i] = (uint64_t)i;
    uint64_t shots[64];
    qrack_measure_shots_qubits(sid, qubits, nCtrl, 64, shots);

    /* the QFT convention omits terminal swaps: reverse the bits to read the
     * phase estimate, then peaks sit at multiples of 2^nCtrl / r (r = 4) */
    int hist[4] = { 0, 0, 0, 0 };
    int offPeak = 0;
    for (int s = 0; s < 64; ++s) {
        uint64_t v = shots[s], rev = 0;
        for (int b = 0; b < nCtrl; ++b) rev |= ((v >> b) & 1u) << (nCtrl - 1 - b);
        if (rev % 64 == 0) {
            hist[rev / 64]++;
        } else {
            offPeak++;
        }
    }
    printf("phase peaks at k*64 (k=0..3): %d %d %d %d   off-peak: %d\n", hist[0], hist[1],
        hist[2], hist[3], offPeak);
    printf("order r = 4  =>  factors gcd(7^2 +/- 1, 15) = {3, 5}\n");
    if (qrack_get_error(sid)) {
        fprintf(stderr, "simulator error latched\n");
        return 1;
    }
    qrack_destroy(sid);
    return 0;
}
