/* qrack_amd — C ABI (parity target: /root/reference/include/pinvoke_api.hpp,
 * the 202-function `quid`-handle surface PyQrack / the Q# runtime bind).
 * This compact ABI covers the same capability areas: lifecycle, the gate
 * set, measurement/sampling, ALU, separability, fidelity, and file I/O.
 * Exported from the qrack_amd/_qrack*.so python extension (plain C symbols;
 * load with dlopen/ctypes). All functions are synchronous; errors latch
 * per-simulator and are polled with qrack_get_error(). */
#ifndef QRACK_AMD_CAPI_H
#define QRACK_AMD_CAPI_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef uint64_t quid;

/* lifecycle; layer flags mirror init_count_type (pinvoke_api.cpp:661):
 * tn = tensor-network wrapper, sd = schmidt decomposition (QUnit),
 * sh = stabilizer hybrid, bdt = binary decision tree, pg = pager,
 * nw = noisy wrapper, hy = cpu/gpu hybrid, gpu = HIP engine,
 * dbl = fp64 amplitudes */
quid qrack_init_count_type(uint64_t qubits, int tn, int sd, int sh, int bdt, int pg, int nw,
    int hy, int gpu, int dbl);
quid qrack_init_count(uint64_t qubits, int gpu); /* canonical optimal stack */
quid qrack_init_clone(quid sid);
void qrack_destroy(quid sid);
void qrack_seed(quid sid, uint64_t s);
uint64_t qrack_num_qubits(quid sid);
int qrack_get_error(quid sid);

/* state */
void qrack_set_permutation(quid sid, uint64_t perm);
void qrack_reset_all(quid sid);
double qrack_prob(quid sid, uint64_t q);
double qrack_prob_perm(quid sid, const uint64_t* qs, const int* vals, uint64_t n);
void qrack_get_amplitude(quid sid, uint64_t perm, double* re, double* im);

/* single-qubit gates */
void qrack_x(quid sid, uint64_t q);
void qrack_y(quid sid, uint64_t q);
void qrack_z(quid sid, uint64_t q);
void qrack_h(quid sid, uint64_t q);
void qrack_s(quid sid, uint64_t q);
void qrack_t(quid sid, uint64_t q);
void qrack_adjs(quid sid, uint64_t q);
void qrack_adjt(quid sid, uint64_t q);
void qrack_u(quid sid, uint64_t q, double theta, double phi, double lambda);
void qrack_mtrx(quid sid, const double* m8, uint64_t q); /* row-major re,im x4 */

/* Pauli rotation: b in {1=X, 2=Z, 3=Y} */
void qrack_r(quid sid, int b, double phi, uint64_t q);

/* controlled gates */
void qrack_mcx(quid sid, const uint64_t* c, uint64_t nc, uint64_t q);
void qrack_mcy(quid sid, const uint64_t* c, uint64_t nc, uint64_t q);
void qrack_mcz(quid sid, const uint64_t* c, uint64_t nc, uint64_t q);
void qrack_mch(quid sid, const uint64_t* c, uint64_t nc, uint64_t q);
void qrack_mcu(quid sid, const uint64_t* c, uint64_t nc, uint64_t q, double theta, double phi,
    double lambda);
void qrack_mcmtrx(quid sid, const uint64_t* c, uint64_t nc, const double* m8, uint64_t q);
void qrack_macx(quid sid, const uint64_t* c, uint64_t nc, uint64_t q);
void qrack_macmtrx(quid sid, const uint64_t* c, uint64_t nc, const double* m8, uint64_t q);
void qrack_mcr(quid sid, int b, double phi, const uint64_t* c, uint64_t nc, uint64_t q);

/* swaps */
void qrack_swap(quid sid, uint64_t a, uint64_t b);
void qrack_iswap(quid sid, uint64_t a, uint64_t b);
void qrack_adjiswap(quid sid, uint64_t a, uint64_t b);
void qrack_fsim(quid sid, double theta, double phi, uint64_t a, uint64_t b);
void qrack_cswap(quid sid, const uint64_t* c, uint64_t nc, uint64_t a, uint64_t b);

/* measurement */
int qrack_m(quid sid, uint64_t q);
int qrack_force_m(quid sid, uint64_t q, int result);
uint64_t qrack_m_all(quid sid);
/* shotsArray receives `shots` packed results over the `qs` bit order */
void qrack_measure_shots(quid sid, const uint64_t* qs, uint64_t nq, uint64_t shots,
    uint64_t* shotsArray);
/* packed >64-qubit paths (two uint64 limbs per 128-bit value) */
void qrack_m_all_wide(quid sid, uint64_t* lo, uint64_t* hi);
void qrack_set_permutation_wide(quid sid, uint64_t lo, uint64_t hi);
/* wide-safe sampling: qubit INDICES (any width); results pack list positions */
void qrack_measure_shots_qubits(quid sid, const uint64_t* qubits, uint64_t nq, uint64_t shots,
    uint64_t* shotsArray);
double qrack_joint_ensemble_probability(quid sid, const int* paulis, const uint64_t* qs, uint64_t n);

/* QFT */
void qrack_qft(quid sid, uint64_t start, uint64_t length);
void qrack_iqft(quid sid, uint64_t start, uint64_t length);

/* ALU */
void qrack_add(quid sid, uint64_t a, uint64_t start, uint64_t length);
void qrack_sub(quid sid, uint64_t a, uint64_t start, uint64_t length);
void qrack_mul(quid sid, uint64_t a, uint64_t inOutStart, uint64_t carryStart, uint64_t length);
void qrack_div(quid sid, uint64_t a, uint64_t inOutStart, uint64_t carryStart, uint64_t length);
void qrack_muln(quid sid, uint64_t a, uint64_t modN, uint64_t inStart, uint64_t outStart,
    uint64_t length);
void qrack_pown(quid sid, uint64_t a, uint64_t modN, uint64_t inStart, uint64_t outStart,
    uint64_t length);
void qrack_mcpown(quid sid, uint64_t a, uint64_t modN, uint64_t inStart, uint64_t outStart,
    uint64_t length, const uint64_t* c, uint64_t nc);
void qrack_hash(quid sid, uint64_t start, uint64_t length, const unsigned char* table);

/* separability / fidelity */
int qrack_try_separate_1qb(quid sid, uint64_t q);
int qrack_try_separate_2qb(quid sid, uint64_t q1, uint64_t q2);
double qrack_get_unitary_fidelity(quid sid);
void qrack_reset_unitary_fidelity(quid sid);
// Schmidt-decomposition rounding parameter (0 = exact; >0 = approximate)
void qrack_set_sdrp(quid sid, double sdrp);
// near-Clifford rounding parameter (0 = exact)
void qrack_set_ncrp(quid sid, double ncrp);

/* compose / decompose */
quid qrack_compose(quid sid, quid other);

/* approximation / separability controls (reference SetSdrp/SetNcrp family) */
void qrack_set_ace_max_qb(quid sid, uint64_t maxQb);
void qrack_set_reactive_separate(quid sid, int on);
int qrack_try_separate_tol(quid sid, const uint64_t* qs, uint64_t n, double tol);
int qrack_are_factorized(quid sid, const uint64_t* qa, uint64_t na, const uint64_t* qb,
    uint64_t nb);

/* multiplexer (reference Multiplex1Mtrx): m8 holds 2^nc row-major 2x2s,
   interleaved re/im doubles */
void qrack_multiplex_1mtrx(quid sid, const uint64_t* cs, uint64_t nc, uint64_t q,
    const double* m8);

/* boolean logic + parity phase */
void qrack_and(quid sid, uint64_t qi1, uint64_t qi2, uint64_t qo);
void qrack_or(quid sid, uint64_t qi1, uint64_t qi2, uint64_t qo);
void qrack_xor(quid sid, uint64_t qi1, uint64_t qi2, uint64_t qo);
void qrack_nand(quid sid, uint64_t qi1, uint64_t qi2, uint64_t qo);
void qrack_nor(quid sid, uint64_t qi1, uint64_t qi2, uint64_t qo);
void qrack_cland(quid sid, uint64_t qi, int ci, uint64_t qo);
void qrack_clor(quid sid, uint64_t qi, int ci, uint64_t qo);
void qrack_clxor(quid sid, uint64_t qi, int ci, uint64_t qo);
void qrack_phase_parity(quid sid, double lambda, const uint64_t* qs, uint64_t n);

/* modular arithmetic (Shor building blocks; reference MULN/DIVN/MCMULN/MCDIVN) */
void qrack_divn(quid sid, uint64_t a, uint64_t m, uint64_t inStart, uint64_t outStart,
    uint64_t len);
void qrack_mcmuln(quid sid, uint64_t a, const uint64_t* cs, uint64_t nc, uint64_t m,
    uint64_t inStart, uint64_t outStart, uint64_t len);
void qrack_mcdivn(quid sid, uint64_t a, const uint64_t* cs, uint64_t nc, uint64_t m,
    uint64_t inStart, uint64_t outStart, uint64_t len);

/* quantum neuron sub-API (reference init_qneuron/qneuron_* family) */
quid qrack_init_qneuron(quid sid, const uint64_t* inputs, uint64_t n, uint64_t output,
    int activationFn, double alpha, double tolerance);
void qrack_destroy_qneuron(quid nid);
uint64_t qrack_get_qneuron_qubit_count(quid nid);
void qrack_set_qneuron_angles(quid nid, const double* angles, uint64_t n);
void qrack_get_qneuron_angles(quid nid, double* angles, uint64_t n);
void qrack_set_qneuron_alpha(quid nid, double alpha);
void qrack_set_qneuron_activation_fn(quid nid, int fn);
double qrack_qneuron_predict(quid nid, int expected, int resetInit);
double qrack_qneuron_unpredict(quid nid, int expected);
void qrack_qneuron_learn(quid nid, double eta, int expected, int resetInit);
void qrack_qneuron_learn_cycle(quid nid, double eta, int expected);
void qrack_qneuron_learn_permutation(quid nid, double eta, int expected, uint64_t perm);

/* serializable circuit sub-API (reference init_qcircuit/qcircuit_* family) */
quid qrack_init_qcircuit(uint64_t qubits);
void qrack_destroy_qcircuit(quid cid);
uint64_t qrack_qcircuit_qubit_count(quid cid);
void qrack_qcircuit_append_1qb(quid cid, const double* m8, uint64_t q);
void qrack_qcircuit_append_mc(quid cid, const double* m8, const uint64_t* cs, uint64_t nc,
    uint64_t q, uint64_t perm);
void qrack_qcircuit_swap(quid cid, uint64_t q1, uint64_t q2);
void qrack_qcircuit_run(quid cid, quid sid);
quid qrack_qcircuit_inverse(quid cid);
quid qrack_qcircuit_past_light_cone(quid cid, const uint64_t* qs, uint64_t n);
int qrack_qcircuit_out_to_file(quid cid, const char* path);
quid qrack_qcircuit_in_from_file(const char* path);
uint64_t qrack_qcircuit_out_to_string_length(quid cid);
void qrack_qcircuit_out_to_string(quid cid, char* out, uint64_t cap);

/* file I/O (SURVEY.md §5 checkpoint formats) */
int qrack_qstabilizer_out_to_file(quid sid, const char* path);
quid qrack_qstabilizer_in_from_file(const char* path);
int qrack_lossy_out_to_file(quid sid, const char* path);
int qrack_lossy_in_from_file(quid sid, const char* path);

#ifdef __cplusplus
}
#endif

#endif /* QRACK_AMD_CAPI_H */
