/* qrack_amd — pinvoke-compatible C ABI.
 *
 * Drop-in surface for programs written against the reference simulator's
 * pinvoke C API (unitaryfoundation/qrack, include/pinvoke_api.hpp): the
 * same exported symbol names and signatures, served by the qrack_amd
 * MI355X engine stack. All 193 reference exports are provided; simulator,
 * neuron and circuit handles are interchangeable with the qrack_* ABI
 * (qrack_amd_capi.h).
 *
 * Reference-default build convention: FPPOW=5, so real1_s is float.
 * Implementation: csrc/pinvoke_compat.inc / csrc/pinvoke_compat2.inc.
 */
#ifndef QRACK_PINVOKE_COMPAT_H
#define QRACK_PINVOKE_COMPAT_H

#include <stdbool.h>
#include <stddef.h>

typedef unsigned long long uintq;
typedef long long intq;
typedef float real1_s;
typedef void (*IdCallback)(uintq);
typedef bool (*ProbAmpCallback)(size_t, double, double);

struct _QrackTimeEvolveOpHeader {
    unsigned target;
    unsigned controlLen;
    unsigned controls[32];
};

#ifdef __cplusplus
extern "C" {
#endif

/* lifecycle / meta */
int get_error(uintq sid);
uintq init_count_type(uintq q, bool tn, bool md, bool sd, bool sh, bool bdt, bool pg, bool nw,
    bool hy, bool oc, bool hp, bool sp);
uintq init_count(uintq q, bool hp, bool sp);
uintq init_count_pager(uintq q, bool hp, bool sp);
uintq init_count_stabilizer(uintq q);
uintq init(void);
uintq init_clone(uintq sid);
void destroy(uintq sid);
void seed(uintq sid, uintq s);
void set_concurrency(uintq sid, uintq p);
void set_device(uintq sid, intq did);
void set_device_list(uintq sid, uintq n, intq* dids);
void qstabilizer_out_to_file(uintq sid, char* f);
void qstabilizer_in_from_file(uintq sid, char* f);
void lossy_out_to_file(uintq sid, char* f, int p, int b);
void lossy_in_from_file(uintq sid, char* f);
void allocateQubit(uintq sid, uintq qid);
bool release(uintq sid, uintq q);
uintq num_qubits(uintq sid);
void ResetAll(uintq sid);

/* probability / expectation / variance */
void HighestProbAll(uintq sid, uintq* r);
void HighestProbAllN(uintq sid, uintq n, uintq* r);
void ProbAll(uintq sid, uintq n, uintq* q, real1_s* p);
double Prob(uintq sid, uintq q);
double ProbRdm(uintq sid, uintq q);
double PermutationProb(uintq sid, uintq n, uintq* q, bool* c);
double PermutationProbRdm(uintq sid, uintq n, uintq* q, bool* c, bool r);
double PermutationExpectation(uintq sid, uintq n, uintq* q);
double PermutationExpectationRdm(uintq sid, uintq n, uintq* q, bool r);
double FactorizedExpectation(uintq sid, uintq n, uintq* q, uintq m, uintq* c);
double FactorizedExpectationRdm(uintq sid, uintq n, uintq* q, uintq m, uintq* c, bool r);
double Variance(uintq sid, uintq n, uintq* q);
double VarianceRdm(uintq sid, uintq n, uintq* q, bool r);
double FactorizedVariance(uintq sid, uintq n, uintq* q, uintq m, uintq* c);
double FactorizedVarianceRdm(uintq sid, uintq n, uintq* q, uintq m, uintq* c, bool r);
double FactorizedExpectationFp(uintq sid, uintq n, uintq* q, real1_s* c);
double FactorizedExpectationFpRdm(uintq sid, uintq n, uintq* q, real1_s* c, bool r);
double FactorizedVarianceFp(uintq sid, uintq n, uintq* q, real1_s* c);
double FactorizedVarianceFpRdm(uintq sid, uintq n, uintq* q, real1_s* c, bool r);
double UnitaryExpectation(uintq sid, uintq n, uintq* q, real1_s* b);
double MatrixExpectation(uintq sid, uintq n, uintq* q, real1_s* b);
double UnitaryExpectationEigenVal(uintq sid, uintq n, uintq* q, real1_s* b, real1_s* e);
double MatrixExpectationEigenVal(uintq sid, uintq n, uintq* q, real1_s* b, real1_s* e);
double UnitaryVariance(uintq sid, uintq n, uintq* q, real1_s* b);
double MatrixVariance(uintq sid, uintq n, uintq* q, real1_s* b);
double UnitaryVarianceEigenVal(uintq sid, uintq n, uintq* q, real1_s* b, real1_s* e);
double MatrixVarianceEigenVal(uintq sid, uintq n, uintq* q, real1_s* b, real1_s* e);
double PauliExpectation(uintq sid, uintq n, uintq* q, uintq* b);
double PauliVariance(uintq sid, uintq n, uintq* q, uintq* b);
double JointEnsembleProbability(uintq sid, uintq n, int* b, uintq* q);

/* state I/O */
void DumpIds(uintq sid, IdCallback callback);
void Dump(uintq sid, ProbAmpCallback callback);
void InKet(uintq sid, real1_s* ket);
void OutKet(uintq sid, real1_s* ket);
void OutProbs(uintq sid, real1_s* p);
void OutReducedDensityMatrix(uintq sid, uintq n, uintq* q, real1_s* rdm);
size_t random_choice(uintq sid, size_t n, double* p);

/* phase helpers */
void PhaseParity(uintq sid, double lambda, uintq n, uintq* q);
void PhaseRootN(uintq sid, uintq p, uintq n, uintq* q);

/* single-qubit gates */
void X(uintq sid, uintq q);
void Y(uintq sid, uintq q);
void Z(uintq sid, uintq q);
void H(uintq sid, uintq q);
void S(uintq sid, uintq q);
void SX(uintq sid, uintq q);
void SY(uintq sid, uintq q);
void T(uintq sid, uintq q);
void AdjS(uintq sid, uintq q);
void AdjSX(uintq sid, uintq q);
void AdjSY(uintq sid, uintq q);
void AdjT(uintq sid, uintq q);
void U(uintq sid, uintq q, double theta, double phi, double lambda);
void Mtrx(uintq sid, double* m, uintq q);

/* (anti-)controlled single-qubit gates */
void MCX(uintq sid, uintq n, uintq* c, uintq q);
void MCY(uintq sid, uintq n, uintq* c, uintq q);
void MCZ(uintq sid, uintq n, uintq* c, uintq q);
void MCH(uintq sid, uintq n, uintq* c, uintq q);
void MCS(uintq sid, uintq n, uintq* c, uintq q);
void MCT(uintq sid, uintq n, uintq* c, uintq q);
void MCAdjS(uintq sid, uintq n, uintq* c, uintq q);
void MCAdjT(uintq sid, uintq n, uintq* c, uintq q);
void MCU(uintq sid, uintq n, uintq* c, uintq q, double theta, double phi, double lambda,
    double gamma);
void MCMtrx(uintq sid, uintq n, uintq* c, double* m, uintq q);
void MACX(uintq sid, uintq n, uintq* c, uintq q);
void MACY(uintq sid, uintq n, uintq* c, uintq q);
void MACZ(uintq sid, uintq n, uintq* c, uintq q);
void MACH(uintq sid, uintq n, uintq* c, uintq q);
void MACS(uintq sid, uintq n, uintq* c, uintq q);
void MACT(uintq sid, uintq n, uintq* c, uintq q);
void MACAdjS(uintq sid, uintq n, uintq* c, uintq q);
void MACAdjT(uintq sid, uintq n, uintq* c, uintq q);
void MACU(uintq sid, uintq n, uintq* c, uintq q, double theta, double phi, double lambda,
    double gamma);
void MACMtrx(uintq sid, uintq n, uintq* c, double* m, uintq q);
void UCMtrx(uintq sid, uintq n, uintq* c, double* m, uintq q, uintq p);
void Multiplex1Mtrx(uintq sid, uintq n, uintq* c, uintq q, double* m);

/* coalesced gates, rotations, Pauli exponentials */
void MX(uintq sid, uintq n, uintq* q);
void MY(uintq sid, uintq n, uintq* q);
void MZ(uintq sid, uintq n, uintq* q);
void R(uintq sid, uintq b, double phi, uintq q);
void MCR(uintq sid, uintq b, double phi, uintq n, uintq* c, uintq q);
void Exp(uintq sid, uintq n, int* b, double phi, uintq* q);
void MCExp(uintq sid, uintq n, int* b, double phi, uintq nc, uintq* cs, uintq* q);

/* measurement */
uintq M(uintq sid, uintq q);
uintq ForceM(uintq sid, uintq q, bool r);
uintq MAll(uintq sid);
void MAllLong(uintq sid, uintq* r);
uintq Measure(uintq sid, uintq n, int* b, uintq* q);
void MeasureShots(uintq sid, uintq n, uintq* q, uintq s, uintq* m);

/* swap family */
void SWAP(uintq sid, uintq qi1, uintq qi2);
void ISWAP(uintq sid, uintq qi1, uintq qi2);
void AdjISWAP(uintq sid, uintq qi1, uintq qi2);
void FSim(uintq sid, double theta, double phi, uintq qi1, uintq qi2);
void CSWAP(uintq sid, uintq n, uintq* c, uintq qi1, uintq qi2);
void ACSWAP(uintq sid, uintq n, uintq* c, uintq qi1, uintq qi2);

/* Schmidt decomposition */
void Compose(uintq sid1, uintq sid2, uintq* q);
uintq Decompose(uintq sid, uintq n, uintq* q);
void Dispose(uintq sid, uintq n, uintq* q);

/* boolean logic */
void AND(uintq sid, uintq qi1, uintq qi2, uintq qo);
void OR(uintq sid, uintq qi1, uintq qi2, uintq qo);
void XOR(uintq sid, uintq qi1, uintq qi2, uintq qo);
void NAND(uintq sid, uintq qi1, uintq qi2, uintq qo);
void NOR(uintq sid, uintq qi1, uintq qi2, uintq qo);
void XNOR(uintq sid, uintq qi1, uintq qi2, uintq qo);
void CLAND(uintq sid, bool ci, uintq qi, uintq qo);
void CLOR(uintq sid, bool ci, uintq qi, uintq qo);
void CLXOR(uintq sid, bool ci, uintq qi, uintq qo);
void CLNAND(uintq sid, bool ci, uintq qi, uintq qo);
void CLNOR(uintq sid, bool ci, uintq qi, uintq qo);
void CLXNOR(uintq sid, bool ci, uintq qi, uintq qo);

/* QFT over an arbitrary qubit list (low-first) */
void QFT(uintq sid, uintq n, uintq* c);
void IQFT(uintq sid, uintq n, uintq* c);

/* ALU (registers are arbitrary qubit lists, low-first; single-word values) */
void ADD(uintq sid, uintq na, uintq* a, uintq n, uintq* q);
void SUB(uintq sid, uintq na, uintq* a, uintq n, uintq* q);
void ADDS(uintq sid, uintq na, uintq* a, uintq s, uintq n, uintq* q);
void SUBS(uintq sid, uintq na, uintq* a, uintq s, uintq n, uintq* q);
void MCADD(uintq sid, uintq na, uintq* a, uintq nc, uintq* c, uintq nq, uintq* q);
void MCSUB(uintq sid, uintq na, uintq* a, uintq nc, uintq* c, uintq nq, uintq* q);
void MUL(uintq sid, uintq na, uintq* a, uintq n, uintq* q, uintq* o);
void DIV(uintq sid, uintq na, uintq* a, uintq n, uintq* q, uintq* o);
void MULN(uintq sid, uintq na, uintq* a, uintq* m, uintq n, uintq* q, uintq* o);
void DIVN(uintq sid, uintq na, uintq* a, uintq* m, uintq n, uintq* q, uintq* o);
void POWN(uintq sid, uintq na, uintq* a, uintq* m, uintq n, uintq* q, uintq* o);
void MCMUL(uintq sid, uintq na, uintq* a, uintq nc, uintq* c, uintq n, uintq* q, uintq* o);
void MCDIV(uintq sid, uintq na, uintq* a, uintq nc, uintq* c, uintq n, uintq* q, uintq* o);
void MCMULN(uintq sid, uintq na, uintq* a, uintq nc, uintq* c, uintq* m, uintq n, uintq* q,
    uintq* o);
void MCDIVN(uintq sid, uintq na, uintq* a, uintq nc, uintq* c, uintq* m, uintq n, uintq* q,
    uintq* o);
void MCPOWN(uintq sid, uintq na, uintq* a, uintq nc, uintq* c, uintq* m, uintq n, uintq* q,
    uintq* o);
void LDA(uintq sid, uintq ni, uintq* qi, uintq nv, uintq* qv, unsigned char* t);
void ADC(uintq sid, uintq s, uintq ni, uintq* qi, uintq nv, uintq* qv, unsigned char* t);
void SBC(uintq sid, uintq s, uintq ni, uintq* qi, uintq nv, uintq* qv, unsigned char* t);
void Hash(uintq sid, uintq n, uintq* q, unsigned char* t);

/* separability / fidelity / approximation */
bool TrySeparate1Qb(uintq sid, uintq qi1);
bool TrySeparate2Qb(uintq sid, uintq qi1, uintq qi2);
bool TrySeparateTol(uintq sid, uintq n, uintq* q, double tol);
void Separate(uintq sid, uintq n, uintq* q);
bool AreFactorized(uintq sid, uintq n1, uintq* a, uintq n2, uintq* b, bool fc);
double GetUnitaryFidelity(uintq sid);
void ResetUnitaryFidelity(uintq sid);
void SetSdrp(uintq sid, double sdrp);
void SetNcrp(uintq sid, double ncrp);
void SetSprp(uintq sid, double sprp);
void SetReactiveSeparate(uintq sid, bool irs);
void SetTInjection(uintq sid, bool iti);
void SetUseExactNearClifford(uintq sid, bool enc);
void SetNoiseParameter(uintq sid, double np);
void SetAceMaxQb(uintq sid, uintq qb);
void SetSparseAceMaxMb(uintq sid, size_t mb);
void SetStochastic(uintq sid, bool s);
void SetMajorQuadrant(uintq sid, bool q);
void SetQuadrant(uintq sid, uintq t, bool b);
void FlipQuadrant(uintq sid, uintq t);
void Normalize(uintq sid);

/* Trotterized time evolution (uniform controlled Hamiltonian terms) */
void TimeEvolve(uintq sid, double t, uintq n, struct _QrackTimeEvolveOpHeader* teos, uintq mn,
    double* mtrx);

/* quantum neuron */
uintq init_qneuron(uintq sid, uintq n, uintq* c, uintq q);
uintq clone_qneuron(uintq nid);
void destroy_qneuron(uintq nid);
void set_qneuron_angles(uintq nid, real1_s* angles);
void get_qneuron_angles(uintq nid, real1_s* angles);
void set_qneuron_sim(uintq nid, uintq sid, uintq n, uintq* c, uintq q);
double qneuron_predict(uintq nid, real1_s* angles, bool e, bool r, uintq f, double a);
double qneuron_unpredict(uintq nid, real1_s* angles, bool e, uintq f, double a);
double qneuron_learn_cycle(uintq nid, real1_s* angles, bool e, uintq f, double a);
void qneuron_learn(uintq nid, real1_s* angles, double eta, bool e, bool r, uintq f, double a);
void qneuron_learn_permutation(
    uintq nid, real1_s* angles, double eta, bool e, bool r, uintq f, double a);

/* serializable circuit */
uintq init_qcircuit(bool collapse, bool clifford);
uintq init_qcircuit_clone(uintq cid);
uintq qcircuit_inverse(uintq cid);
uintq qcircuit_past_light_cone(uintq cid, uintq n, uintq* q);
void destroy_qcircuit(uintq cid);
uintq get_qcircuit_qubit_count(uintq cid);
void qcircuit_swap(uintq cid, uintq q1, uintq q2);
void qcircuit_append_1qb(uintq cid, double* m, uintq q);
void qcircuit_append_mc(uintq cid, double* m, uintq n, uintq* c, uintq q, uintq p);
void qcircuit_run(uintq cid, uintq sid);
void qcircuit_out_to_file(uintq cid, char* f);
void qcircuit_in_from_file(uintq cid, char* f);
size_t qcircuit_out_to_string_length(uintq cid);
void qcircuit_out_to_string(uintq cid, char* f);

#ifdef __cplusplus
}
#endif

#endif /* QRACK_PINVOKE_COMPAT_H */
