// qrack_amd — C ABI implementation (see include/qrack_amd_capi.h).
// Parity target: /root/reference/src/pinvoke_api.cpp (quid handle table,
// per-simulator error latch polled by get_error).
#include "../include/qrack_amd_capi.h"

#include "qcircuit.hpp"
#include "qfactory.hpp"
#include "qneuron.hpp"
#include "serialize.hpp"

#include <fstream>
#include <mutex>

using namespace qrack_amd;

namespace {

struct SimSlot {
    QInterfacePtr<float> f;
    QInterfacePtr<double> d;
    int error = 0;
    // Per-simulator operation lock: held across every op so a concurrent
    // qrack_destroy() cannot free state mid-operation (the shared_ptr keeps
    // the slot itself alive; this mutex serializes ops on one handle, same
    // contract as the reference pinvoke per-simulator locks).
    std::mutex op;
    // pinvoke-compat qubit-id indirection (reference shards map): identity
    // until allocateQubit/release diverges logical ids from indices
    bool useQidMap = false;
    std::map<uint64_t, bitLenInt> qidMap;
    bitLenInt Qubits() const { return f ? f->GetQubitCount() : (d ? d->GetQubitCount() : 0); }
};
using SimSlotPtr = std::shared_ptr<SimSlot>;

std::mutex g_mtx;
std::map<quid, SimSlotPtr> g_sims;
quid g_next = 1;

SimSlotPtr slot(quid sid)
{
    std::lock_guard<std::mutex> lk(g_mtx);
    auto it = g_sims.find(sid);
    return (it == g_sims.end()) ? nullptr : it->second;
}

quid registerSlot(SimSlotPtr s)
{
    std::lock_guard<std::mutex> lk(g_mtx);
    const quid sid = g_next++;
    g_sims[sid] = std::move(s);
    return sid;
}

template <typename F> void guarded(quid sid, F&& fn)
{
    SimSlotPtr s = slot(sid);
    if (!s) return;
    std::lock_guard<std::mutex> lk(s->op);
    try {
        fn(*s);
    } catch (const std::bad_alloc&) {
        s->error = 2;
    } catch (const std::exception& e) {
        if (std::getenv("QRACK_CAPI_DEBUG")) fprintf(stderr, "capi error: %s\n", e.what());
        s->error = 1;
    }
}

template <typename F> double guardedD(quid sid, F&& fn, double dflt = 0.0)
{
    SimSlotPtr s = slot(sid);
    if (!s) return dflt;
    std::lock_guard<std::mutex> lk(s->op);
    try {
        return fn(*s);
    } catch (const std::exception&) {
        s->error = 1;
        return dflt;
    }
}

// apply op to whichever precision is active
#define FOR_SIM(s, expr)                                                                           \
    do {                                                                                           \
        if ((s).f) {                                                                               \
            auto& q = *(s).f;                                                                      \
            expr;                                                                                  \
        } else if ((s).d) {                                                                        \
            auto& q = *(s).d;                                                                      \
            expr;                                                                                  \
        }                                                                                          \
    } while (0)

template <typename R> void mtrxFrom(const double* m8, cplx<R>* out)
{
    for (int i = 0; i < 4; ++i) out[i] = cplx<R>((R)m8[2 * i], (R)m8[2 * i + 1]);
}

std::vector<bitLenInt> ctrlVec(const uint64_t* c, uint64_t nc)
{
    std::vector<bitLenInt> v;
    for (uint64_t i = 0; i < nc; ++i) v.push_back((bitLenInt)c[i]);
    return v;
}

} // namespace

extern "C" {

quid qrack_init_count_type(
    uint64_t qubits, int tn, int sd, int sh, int bdt, int pg, int nw, int hy, int gpu, int dbl)
{
    std::vector<std::string> layers;
    if (nw) layers.push_back("noisy");
    if (tn) layers.push_back("tensor_network");
    if (sd) layers.push_back("qunit");
    if (sh) layers.push_back("stabilizer_hybrid");
    if (bdt) layers.push_back("bdt_hybrid");
    if (pg) layers.push_back("pager");
    if (hy) {
        layers.push_back("hybrid");
    } else if (gpu && HipDeviceCount() > 0) {
        layers.push_back("hip");
    } else {
        layers.push_back("cpu");
    }
    SimSlotPtr s = std::make_shared<SimSlot>();
    try {
        if (dbl) {
            s->d = CreateStack<double>((bitLenInt)qubits, layers, 0u, -1, -1, 1u);
        } else {
            s->f = CreateStack<float>((bitLenInt)qubits, layers, 0u, -1, -1, 1u);
        }
    } catch (const std::exception&) {
        return 0;
    }
    return registerSlot(std::move(s));
}

quid qrack_init_count(uint64_t qubits, int gpu)
{
    return qrack_init_count_type(qubits, 0, 1, 1, 0, 0, 0, 0, gpu, 0);
}

quid qrack_init_clone(quid sid)
{
    SimSlotPtr s = slot(sid);
    if (!s) return 0;
    SimSlotPtr c = std::make_shared<SimSlot>();
    std::lock_guard<std::mutex> lk(s->op);
    try {
        if (s->f) c->f = s->f->Clone();
        if (s->d) c->d = s->d->Clone();
    } catch (const std::exception&) {
        return 0;
    }
    return registerSlot(std::move(c));
}

void qrack_destroy(quid sid)
{
    std::lock_guard<std::mutex> lk(g_mtx);
    g_sims.erase(sid);
}

void qrack_seed(quid sid, uint64_t sd)
{
    guarded(sid, [&](SimSlot& s) { FOR_SIM(s, q.SetRandomSeed(sd)); });
}

uint64_t qrack_num_qubits(quid sid)
{
    SimSlotPtr s = slot(sid);
    if (!s) return 0;
    std::lock_guard<std::mutex> lk(s->op);
    return s->Qubits();
}

int qrack_get_error(quid sid)
{
    SimSlotPtr s = slot(sid);
    return s ? s->error : -1;
}

void qrack_set_permutation(quid sid, uint64_t perm)
{
    guarded(sid, [&](SimSlot& s) { FOR_SIM(s, q.SetPermutation(perm)); });
}

void qrack_reset_all(quid sid) { qrack_set_permutation(sid, 0u); }

double qrack_prob(quid sid, uint64_t qb)
{
    return guardedD(sid, [&](SimSlot& s) -> double {
        double r = 0;
        FOR_SIM(s, r = (double)q.Prob((bitLenInt)qb));
        return r;
    });
}

double qrack_prob_perm(quid sid, const uint64_t* qs, const int* vals, uint64_t n)
{
    return guardedD(sid, [&](SimSlot& s) -> double {
        bitCapInt mask = 0, perm = 0;
        for (uint64_t i = 0; i < n; ++i) {
            mask |= pow2((bitLenInt)qs[i]);
            if (vals[i]) perm |= pow2((bitLenInt)qs[i]);
        }
        double r = 0;
        FOR_SIM(s, r = (double)q.ProbMask(mask, perm));
        return r;
    });
}

void qrack_get_amplitude(quid sid, uint64_t perm, double* re, double* im)
{
    guarded(sid, [&](SimSlot& s) {
        FOR_SIM(s, {
            auto a = q.GetAmplitude(perm);
            *re = (double)a.re;
            *im = (double)a.im;
        });
    });
}

#define GATE1(name, call)                                                                          \
    void name(quid sid, uint64_t qb)                                                               \
    {                                                                                              \
        guarded(sid, [&](SimSlot& s) { FOR_SIM(s, q.call((bitLenInt)qb)); });                      \
    }

GATE1(qrack_x, X)
GATE1(qrack_y, Y)
GATE1(qrack_z, Z)
GATE1(qrack_h, H)
GATE1(qrack_s, S)
GATE1(qrack_t, T)
GATE1(qrack_adjs, IS)
GATE1(qrack_adjt, IT)

void qrack_u(quid sid, uint64_t qb, double theta, double phi, double lambda)
{
    guarded(sid, [&](SimSlot& s) {
        if (s.f) s.f->U((bitLenInt)qb, (float)theta, (float)phi, (float)lambda);
        if (s.d) s.d->U((bitLenInt)qb, theta, phi, lambda);
    });
}

void qrack_mtrx(quid sid, const double* m8, uint64_t qb)
{
    guarded(sid, [&](SimSlot& s) {
        if (s.f) {
            cplx<float> m[4];
            mtrxFrom(m8, m);
            s.f->Mtrx(m, (bitLenInt)qb);
        }
        if (s.d) {
            cplx<double> m[4];
            mtrxFrom(m8, m);
            s.d->Mtrx(m, (bitLenInt)qb);
        }
    });
}

void qrack_r(quid sid, int b, double phi, uint64_t qb)
{
    guarded(sid, [&](SimSlot& s) {
        if (s.f) {
            if (b == 1) s.f->RX((float)phi, (bitLenInt)qb);
            if (b == 3) s.f->RY((float)phi, (bitLenInt)qb);
            if (b == 2) s.f->RZ((float)phi, (bitLenInt)qb);
        }
        if (s.d) {
            if (b == 1) s.d->RX(phi, (bitLenInt)qb);
            if (b == 3) s.d->RY(phi, (bitLenInt)qb);
            if (b == 2) s.d->RZ(phi, (bitLenInt)qb);
        }
    });
}

#define MCGATE(name, callF, callD)                                                                 \
    void name(quid sid, const uint64_t* c, uint64_t nc, uint64_t qb)                               \
    {                                                                                              \
        guarded(sid, [&](SimSlot& s) {                                                             \
            auto cv = ctrlVec(c, nc);                                                              \
            if (s.f) s.f->callF;                                                                   \
            if (s.d) s.d->callD;                                                                   \
        });                                                                                        \
    }

MCGATE(qrack_mcx, MCInvert(cv, cplx<float>(1, 0), cplx<float>(1, 0), (bitLenInt)qb),
    MCInvert(cv, cplx<double>(1, 0), cplx<double>(1, 0), (bitLenInt)qb))
MCGATE(qrack_mcy, MCInvert(cv, cplx<float>(0, -1), cplx<float>(0, 1), (bitLenInt)qb),
    MCInvert(cv, cplx<double>(0, -1), cplx<double>(0, 1), (bitLenInt)qb))
MCGATE(qrack_mcz, MCPhase(cv, cplx<float>(1, 0), cplx<float>(-1, 0), (bitLenInt)qb),
    MCPhase(cv, cplx<double>(1, 0), cplx<double>(-1, 0), (bitLenInt)qb))
MCGATE(qrack_macx, MACInvert(cv, cplx<float>(1, 0), cplx<float>(1, 0), (bitLenInt)qb),
    MACInvert(cv, cplx<double>(1, 0), cplx<double>(1, 0), (bitLenInt)qb))

void qrack_mch(quid sid, const uint64_t* c, uint64_t nc, uint64_t qb)
{
    guarded(sid, [&](SimSlot& s) {
        auto cv = ctrlVec(c, nc);
        if (s.f) {
            const float v = 0.70710678f;
            const cplx<float> m[4] = { { v, 0 }, { v, 0 }, { v, 0 }, { -v, 0 } };
            s.f->MCMtrx(cv, m, (bitLenInt)qb);
        }
        if (s.d) {
            const double v = 0.7071067811865476;
            const cplx<double> m[4] = { { v, 0 }, { v, 0 }, { v, 0 }, { -v, 0 } };
            s.d->MCMtrx(cv, m, (bitLenInt)qb);
        }
    });
}

void qrack_mcu(
    quid sid, const uint64_t* c, uint64_t nc, uint64_t qb, double theta, double phi, double lambda)
{
    guarded(sid, [&](SimSlot& s) {
        auto cv = ctrlVec(c, nc);
        if (s.f) {
            const float ct = std::cos((float)theta / 2), st = std::sin((float)theta / 2);
            const cplx<float> m[4] = { { ct, 0 },
                (float)(-st) * polar<float>(1, (float)lambda), st * polar<float>(1, (float)phi),
                ct * polar<float>(1, (float)(phi + lambda)) };
            s.f->MCMtrx(cv, m, (bitLenInt)qb);
        }
        if (s.d) {
            const double ct = std::cos(theta / 2), st = std::sin(theta / 2);
            const cplx<double> m[4] = { { ct, 0 }, (-st) * polar<double>(1, lambda),
                st * polar<double>(1, phi), ct * polar<double>(1, phi + lambda) };
            s.d->MCMtrx(cv, m, (bitLenInt)qb);
        }
    });
}

void qrack_mcmtrx(quid sid, const uint64_t* c, uint64_t nc, const double* m8, uint64_t qb)
{
    guarded(sid, [&](SimSlot& s) {
        auto cv = ctrlVec(c, nc);
        if (s.f) {
            cplx<float> m[4];
            mtrxFrom(m8, m);
            s.f->MCMtrx(cv, m, (bitLenInt)qb);
        }
        if (s.d) {
            cplx<double> m[4];
            mtrxFrom(m8, m);
            s.d->MCMtrx(cv, m, (bitLenInt)qb);
        }
    });
}

void qrack_macmtrx(quid sid, const uint64_t* c, uint64_t nc, const double* m8, uint64_t qb)
{
    guarded(sid, [&](SimSlot& s) {
        auto cv = ctrlVec(c, nc);
        if (s.f) {
            cplx<float> m[4];
            mtrxFrom(m8, m);
            s.f->MACMtrx(cv, m, (bitLenInt)qb);
        }
        if (s.d) {
            cplx<double> m[4];
            mtrxFrom(m8, m);
            s.d->MACMtrx(cv, m, (bitLenInt)qb);
        }
    });
}

void qrack_mcr(quid sid, int b, double phi, const uint64_t* c, uint64_t nc, uint64_t qb)
{
    guarded(sid, [&](SimSlot& s) {
        auto cv = ctrlVec(c, nc);
        if (s.f) {
            if (b == 2) {
                s.f->MCPhase(cv, polar<float>(1, (float)(-phi / 2)),
                    polar<float>(1, (float)(phi / 2)), (bitLenInt)qb);
            } else {
                const float ct = std::cos((float)phi / 2), st = std::sin((float)phi / 2);
                cplx<float> m[4];
                if (b == 1) {
                    m[0] = { ct, 0 };
                    m[1] = { 0, -st };
                    m[2] = { 0, -st };
                    m[3] = { ct, 0 };
                } else {
                    m[0] = { ct, 0 };
                    m[1] = { -st, 0 };
                    m[2] = { st, 0 };
                    m[3] = { ct, 0 };
                }
                s.f->MCMtrx(cv, m, (bitLenInt)qb);
            }
        }
        if (s.d) {
            if (b == 2) {
                s.d->MCPhase(cv, polar<double>(1, -phi / 2), polar<double>(1, phi / 2),
                    (bitLenInt)qb);
            } else {
                const double ct = std::cos(phi / 2), st = std::sin(phi / 2);
                cplx<double> m[4];
                if (b == 1) {
                    m[0] = { ct, 0 };
                    m[1] = { 0, -st };
                    m[2] = { 0, -st };
                    m[3] = { ct, 0 };
                } else {
                    m[0] = { ct, 0 };
                    m[1] = { -st, 0 };
                    m[2] = { st, 0 };
                    m[3] = { ct, 0 };
                }
                s.d->MCMtrx(cv, m, (bitLenInt)qb);
            }
        }
    });
}

void qrack_swap(quid sid, uint64_t a, uint64_t b)
{
    guarded(sid, [&](SimSlot& s) { FOR_SIM(s, q.Swap((bitLenInt)a, (bitLenInt)b)); });
}
void qrack_iswap(quid sid, uint64_t a, uint64_t b)
{
    guarded(sid, [&](SimSlot& s) { FOR_SIM(s, q.ISwap((bitLenInt)a, (bitLenInt)b)); });
}
void qrack_adjiswap(quid sid, uint64_t a, uint64_t b)
{
    guarded(sid, [&](SimSlot& s) { FOR_SIM(s, q.IISwap((bitLenInt)a, (bitLenInt)b)); });
}
void qrack_fsim(quid sid, double theta, double phi, uint64_t a, uint64_t b)
{
    guarded(sid, [&](SimSlot& s) {
        if (s.f) s.f->FSim((float)theta, (float)phi, (bitLenInt)a, (bitLenInt)b);
        if (s.d) s.d->FSim(theta, phi, (bitLenInt)a, (bitLenInt)b);
    });
}
void qrack_cswap(quid sid, const uint64_t* c, uint64_t nc, uint64_t a, uint64_t b)
{
    guarded(sid, [&](SimSlot& s) {
        auto cv = ctrlVec(c, nc);
        FOR_SIM(s, q.CSwap(cv, (bitLenInt)a, (bitLenInt)b));
    });
}

int qrack_m(quid sid, uint64_t qb)
{
    return (int)guardedD(sid, [&](SimSlot& s) -> double {
        bool r = false;
        FOR_SIM(s, r = q.M((bitLenInt)qb));
        return r ? 1.0 : 0.0;
    });
}

int qrack_force_m(quid sid, uint64_t qb, int result)
{
    return (int)guardedD(sid, [&](SimSlot& s) -> double {
        bool r = false;
        FOR_SIM(s, r = q.ForceM((bitLenInt)qb, result != 0, true, true));
        return r ? 1.0 : 0.0;
    });
}

uint64_t qrack_m_all(quid sid)
{
    return (uint64_t)guardedD(sid, [&](SimSlot& s) -> double {
        bitCapInt r = 0;
        FOR_SIM(s, r = q.MAll());
        return (double)r;
    });
}

void qrack_measure_shots(
    quid sid, const uint64_t* qs, uint64_t nq, uint64_t shots, uint64_t* shotsArray)
{
    guarded(sid, [&](SimSlot& s) {
        std::vector<bitCapInt> powers;
        for (uint64_t i = 0; i < nq; ++i) powers.push_back(pow2((bitLenInt)qs[i]));
        FOR_SIM(s, {
            auto res = q.MultiShotMeasureMask(powers, (unsigned)shots);
            uint64_t idx = 0;
            for (auto& kv : res) {
                for (int k = 0; k < kv.second && idx < shots; ++k) shotsArray[idx++] = kv.first;
            }
        });
    });
}

void qrack_m_all_wide(quid sid, uint64_t* lo, uint64_t* hi)
{
    guarded(sid, [&](SimSlot& s) {
        BigCap r;
        FOR_SIM(s, r = q.MAllWide());
        if (lo) *lo = r.lo;
        if (hi) *hi = r.hi;
    });
}

void qrack_set_permutation_wide(quid sid, uint64_t lo, uint64_t hi)
{
    guarded(sid, [&](SimSlot& s) { FOR_SIM(s, q.SetPermutationWide(BigCap(lo, hi))); });
}

void qrack_measure_shots_qubits(
    quid sid, const uint64_t* qubits, uint64_t nq, uint64_t shots, uint64_t* shotsArray)
{
    guarded(sid, [&](SimSlot& s) {
        std::vector<bitLenInt> qs;
        for (uint64_t i = 0; i < nq; ++i) qs.push_back((bitLenInt)qubits[i]);
        FOR_SIM(s, {
            auto res = q.MultiShotMeasureQubits(qs, (unsigned)shots);
            uint64_t idx = 0;
            for (auto& kv : res) {
                for (int k = 0; k < kv.second && idx < shots; ++k) shotsArray[idx++] = kv.first;
            }
        });
    });
}

double qrack_joint_ensemble_probability(
    quid sid, const int* paulis, const uint64_t* qs, uint64_t n)
{
    return guardedD(sid, [&](SimSlot& s) -> double {
        std::vector<bitLenInt> bits;
        std::vector<Pauli> ps;
        for (uint64_t i = 0; i < n; ++i) {
            bits.push_back((bitLenInt)qs[i]);
            ps.push_back((Pauli)paulis[i]);
        }
        double e = 0;
        FOR_SIM(s, e = q.PauliExpectation(bits, ps));
        return (1.0 - e) / 2.0; /* probability of odd parity */
    });
}

void qrack_qft(quid sid, uint64_t start, uint64_t length)
{
    guarded(sid, [&](SimSlot& s) { FOR_SIM(s, q.QFT((bitLenInt)start, (bitLenInt)length)); });
}
void qrack_iqft(quid sid, uint64_t start, uint64_t length)
{
    guarded(sid, [&](SimSlot& s) { FOR_SIM(s, q.IQFT((bitLenInt)start, (bitLenInt)length)); });
}

void qrack_add(quid sid, uint64_t a, uint64_t start, uint64_t length)
{
    guarded(sid, [&](SimSlot& s) { FOR_SIM(s, q.INC(a, (bitLenInt)start, (bitLenInt)length)); });
}
void qrack_sub(quid sid, uint64_t a, uint64_t start, uint64_t length)
{
    guarded(sid, [&](SimSlot& s) { FOR_SIM(s, q.DEC(a, (bitLenInt)start, (bitLenInt)length)); });
}
void qrack_mul(quid sid, uint64_t a, uint64_t io, uint64_t cs, uint64_t length)
{
    guarded(sid, [&](SimSlot& s) {
        FOR_SIM(s, q.MUL(a, (bitLenInt)io, (bitLenInt)cs, (bitLenInt)length));
    });
}
void qrack_div(quid sid, uint64_t a, uint64_t io, uint64_t cs, uint64_t length)
{
    guarded(sid, [&](SimSlot& s) {
        FOR_SIM(s, q.DIV(a, (bitLenInt)io, (bitLenInt)cs, (bitLenInt)length));
    });
}
void qrack_muln(quid sid, uint64_t a, uint64_t modN, uint64_t in, uint64_t out, uint64_t length)
{
    guarded(sid, [&](SimSlot& s) {
        FOR_SIM(s, q.MULModNOut(a, modN, (bitLenInt)in, (bitLenInt)out, (bitLenInt)length));
    });
}
void qrack_pown(quid sid, uint64_t a, uint64_t modN, uint64_t in, uint64_t out, uint64_t length)
{
    guarded(sid, [&](SimSlot& s) {
        FOR_SIM(s, q.POWModNOut(a, modN, (bitLenInt)in, (bitLenInt)out, (bitLenInt)length));
    });
}
void qrack_mcpown(quid sid, uint64_t a, uint64_t modN, uint64_t in, uint64_t out, uint64_t length,
    const uint64_t* c, uint64_t nc)
{
    guarded(sid, [&](SimSlot& s) {
        auto cv = ctrlVec(c, nc);
        FOR_SIM(s, q.CPOWModNOut(a, modN, (bitLenInt)in, (bitLenInt)out, (bitLenInt)length, cv));
    });
}
void qrack_hash(quid sid, uint64_t start, uint64_t length, const unsigned char* table)
{
    guarded(sid, [&](SimSlot& s) { FOR_SIM(s, q.Hash((bitLenInt)start, (bitLenInt)length, table)); });
}

int qrack_try_separate_1qb(quid sid, uint64_t qb)
{
    return (int)guardedD(sid, [&](SimSlot& s) -> double {
        bool r = false;
        FOR_SIM(s, r = q.TrySeparate((bitLenInt)qb));
        return r ? 1.0 : 0.0;
    });
}
int qrack_try_separate_2qb(quid sid, uint64_t q1, uint64_t q2)
{
    return (int)guardedD(sid, [&](SimSlot& s) -> double {
        bool r = false;
        FOR_SIM(s, r = q.TrySeparate((bitLenInt)q1, (bitLenInt)q2));
        return r ? 1.0 : 0.0;
    });
}
double qrack_get_unitary_fidelity(quid sid)
{
    return guardedD(sid, [&](SimSlot& s) -> double {
        double r = 1.0;
        FOR_SIM(s, r = q.GetUnitaryFidelity());
        return r;
    }, 1.0);
}
void qrack_reset_unitary_fidelity(quid sid)
{
    guarded(sid, [&](SimSlot& s) { FOR_SIM(s, q.ResetUnitaryFidelity()); });
}

void qrack_set_sdrp(quid sid, double sdrp)
{
    guarded(sid, [&](SimSlot& s) { FOR_SIM(s, q.SetSdrp(sdrp)); });
}

void qrack_set_ncrp(quid sid, double ncrp)
{
    guarded(sid, [&](SimSlot& s) { FOR_SIM(s, q.SetNcrp(ncrp)); });
}

quid qrack_compose(quid sid, quid other)
{
    SimSlotPtr a = slot(sid);
    SimSlotPtr b = slot(other);
    if (!a || !b) return 0;
    // Lock both slots deadlock-free (same handle twice: single lock).
    std::unique_lock<std::mutex> la(a->op, std::defer_lock);
    std::unique_lock<std::mutex> lb;
    if (a.get() == b.get()) {
        la.lock();
    } else {
        lb = std::unique_lock<std::mutex>(b->op, std::defer_lock);
        std::lock(la, lb);
    }
    try {
        if (a->f && b->f) a->f->Compose(b->f);
        if (a->d && b->d) a->d->Compose(b->d);
    } catch (const std::exception&) {
        a->error = 1;
        return 0;
    }
    return sid;
}

int qrack_qstabilizer_out_to_file(quid sid, const char* path)
{
    SimSlotPtr s = slot(sid);
    if (!s) return -1;
    std::lock_guard<std::mutex> lk(s->op);
    try {
        std::string text;
        if (s->f) text = SaveStabilizerText<float>(s->f);
        if (s->d) text = SaveStabilizerText<double>(s->d);
        std::ofstream f(path);
        f << text;
        return 0;
    } catch (const std::exception&) {
        s->error = 1;
        return -1;
    }
}

quid qrack_qstabilizer_in_from_file(const char* path)
{
    try {
        std::ifstream f(path);
        std::string text((std::istreambuf_iterator<char>(f)), std::istreambuf_iterator<char>());
        SimSlotPtr s = std::make_shared<SimSlot>();
        s->f = LoadStabilizerText<float>(text, nullptr);
        return registerSlot(std::move(s));
    } catch (const std::exception&) {
        return 0;
    }
}

int qrack_lossy_out_to_file(quid sid, const char* path)
{
    SimSlotPtr s = slot(sid);
    if (!s) return -1;
    std::lock_guard<std::mutex> lk(s->op);
    try {
        if (s->f) LossySaveState<float>(s->f, path);
        if (s->d) LossySaveState<double>(s->d, path);
        return 0;
    } catch (const std::exception&) {
        s->error = 1;
        return -1;
    }
}

int qrack_lossy_in_from_file(quid sid, const char* path)
{
    SimSlotPtr s = slot(sid);
    if (!s) return -1;
    std::lock_guard<std::mutex> lk(s->op);
    try {
        if (s->f) LossyLoadState<float>(s->f, path);
        if (s->d) LossyLoadState<double>(s->d, path);
        return 0;
    } catch (const std::exception&) {
        s->error = 1;
        return -1;
    }
}

/* ---- approximation / separability controls ---- */

void qrack_set_ace_max_qb(quid sid, uint64_t maxQb)
{
    guarded(sid, [&](SimSlot& s) { FOR_SIM(s, q.SetAceMaxQubits((bitLenInt)maxQb)); });
}

void qrack_set_reactive_separate(quid sid, int on)
{
    guarded(sid, [&](SimSlot& s) { FOR_SIM(s, q.SetReactiveSeparate(on != 0)); });
}

int qrack_try_separate_tol(quid sid, const uint64_t* qs, uint64_t n, double tol)
{
    int out = 0;
    guarded(sid, [&](SimSlot& s) {
        std::vector<bitLenInt> v = ctrlVec(qs, n);
        FOR_SIM(s, out = q.TrySeparate(v, (decltype(q.Prob(0)))tol) ? 1 : 0);
    });
    return out;
}

int qrack_are_factorized(
    quid sid, const uint64_t* qa, uint64_t na, const uint64_t* qb, uint64_t nb)
{
    int out = 0;
    guarded(sid, [&](SimSlot& s) {
        std::vector<bitLenInt> a = ctrlVec(qa, na), b = ctrlVec(qb, nb);
        FOR_SIM(s, out = q.AreFactorized(a, b) ? 1 : 0);
    });
    return out;
}

/* ---- multiplexer ---- */

void qrack_multiplex_1mtrx(
    quid sid, const uint64_t* cs, uint64_t nc, uint64_t qb, const double* m8)
{
    guarded(sid, [&](SimSlot& s) {
        std::vector<bitLenInt> ctrls = ctrlVec(cs, nc);
        const size_t nMats = (size_t)1 << nc;
        if (s.f) {
            std::vector<cplx<float>> m(4u * nMats);
            for (size_t i = 0; i < 4u * nMats; ++i)
                m[i] = cplx<float>((float)m8[2 * i], (float)m8[2 * i + 1]);
            s.f->UniformlyControlledSingleBit(ctrls, (bitLenInt)qb, m.data());
        } else if (s.d) {
            std::vector<cplx<double>> m(4u * nMats);
            for (size_t i = 0; i < 4u * nMats; ++i)
                m[i] = cplx<double>(m8[2 * i], m8[2 * i + 1]);
            s.d->UniformlyControlledSingleBit(ctrls, (bitLenInt)qb, m.data());
        }
    });
}

/* ---- boolean logic + parity phase ---- */

#define LOGIC3(name, call)                                                                         \
    void name(quid sid, uint64_t a, uint64_t b, uint64_t o)                                        \
    {                                                                                              \
        guarded(sid,                                                                               \
            [&](SimSlot& s) { FOR_SIM(s, q.call((bitLenInt)a, (bitLenInt)b, (bitLenInt)o)); });    \
    }

LOGIC3(qrack_and, AND)
LOGIC3(qrack_or, OR)
LOGIC3(qrack_xor, XOR)
LOGIC3(qrack_nand, NAND)
LOGIC3(qrack_nor, NOR)

#define CLOGIC(name, call)                                                                         \
    void name(quid sid, uint64_t qi, int ci, uint64_t qo)                                          \
    {                                                                                              \
        guarded(sid,                                                                               \
            [&](SimSlot& s) { FOR_SIM(s, q.call((bitLenInt)qi, ci != 0, (bitLenInt)qo)); });       \
    }

CLOGIC(qrack_cland, CLAND)
CLOGIC(qrack_clor, CLOR)
CLOGIC(qrack_clxor, CLXOR)

void qrack_phase_parity(quid sid, double lambda, const uint64_t* qs, uint64_t n)
{
    guarded(sid, [&](SimSlot& s) {
        bitCapInt mask = 0;
        for (uint64_t i = 0; i < n; ++i) mask |= pow2((bitLenInt)qs[i]);
        if (s.f) s.f->PhaseParity((float)lambda, mask);
        if (s.d) s.d->PhaseParity(lambda, mask);
    });
}

/* ---- modular arithmetic (Shor) ---- */

void qrack_divn(quid sid, uint64_t a, uint64_t m, uint64_t in, uint64_t out, uint64_t len)
{
    guarded(sid, [&](SimSlot& s) {
        FOR_SIM(s, q.IMULModNOut(a, m, (bitLenInt)in, (bitLenInt)out, (bitLenInt)len));
    });
}

void qrack_mcmuln(quid sid, uint64_t a, const uint64_t* cs, uint64_t nc, uint64_t m, uint64_t in,
    uint64_t out, uint64_t len)
{
    guarded(sid, [&](SimSlot& s) {
        std::vector<bitLenInt> ctrls = ctrlVec(cs, nc);
        FOR_SIM(s, q.CMULModNOut(a, m, (bitLenInt)in, (bitLenInt)out, (bitLenInt)len, ctrls));
    });
}

void qrack_mcdivn(quid sid, uint64_t a, const uint64_t* cs, uint64_t nc, uint64_t m, uint64_t in,
    uint64_t out, uint64_t len)
{
    guarded(sid, [&](SimSlot& s) {
        std::vector<bitLenInt> ctrls = ctrlVec(cs, nc);
        FOR_SIM(s, q.CIMULModNOut(a, m, (bitLenInt)in, (bitLenInt)out, (bitLenInt)len, ctrls));
    });
}

} // extern "C"

/* ---- quantum neuron sub-API ---- */

namespace {

struct NeuronSlot {
    std::shared_ptr<QNeuron<float>> f;
    std::shared_ptr<QNeuron<double>> d;
    SimSlotPtr sim; // keeps the simulator alive
    std::mutex op;
};
using NeuronSlotPtr = std::shared_ptr<NeuronSlot>;
std::mutex g_nmtx;
std::map<quid, NeuronSlotPtr> g_neurons;
quid g_nnext = 1;

NeuronSlotPtr neuronSlot(quid nid)
{
    std::lock_guard<std::mutex> lk(g_nmtx);
    auto it = g_neurons.find(nid);
    return (it == g_neurons.end()) ? nullptr : it->second;
}

template <typename F> void nguarded(quid nid, F&& fn)
{
    NeuronSlotPtr s = neuronSlot(nid);
    if (!s) return;
    std::lock_guard<std::mutex> lk(s->op);
    try {
        fn(*s);
    } catch (const std::exception&) {
        if (s->sim) s->sim->error = 1;
    }
}

} // namespace

extern "C" {

quid qrack_init_qneuron(quid sid, const uint64_t* inputs, uint64_t n, uint64_t output,
    int activationFn, double alpha, double tolerance)
{
    SimSlotPtr sim = slot(sid);
    if (!sim) return 0;
    NeuronSlotPtr ns = std::make_shared<NeuronSlot>();
    ns->sim = sim;
    std::vector<bitLenInt> ins = ctrlVec(inputs, n);
    try {
        if (sim->f) {
            ns->f = std::make_shared<QNeuron<float>>(sim->f, ins, (bitLenInt)output,
                (QNeuronActivationFn)activationFn, (float)alpha, (float)tolerance);
        }
        if (sim->d) {
            ns->d = std::make_shared<QNeuron<double>>(sim->d, ins, (bitLenInt)output,
                (QNeuronActivationFn)activationFn, alpha, tolerance);
        }
    } catch (const std::exception&) {
        return 0;
    }
    std::lock_guard<std::mutex> lk(g_nmtx);
    const quid nid = g_nnext++;
    g_neurons[nid] = std::move(ns);
    return nid;
}

void qrack_destroy_qneuron(quid nid)
{
    std::lock_guard<std::mutex> lk(g_nmtx);
    g_neurons.erase(nid);
}

uint64_t qrack_get_qneuron_qubit_count(quid nid)
{
    uint64_t out = 0;
    nguarded(nid, [&](NeuronSlot& s) {
        if (s.f) out = (uint64_t)s.f->GetAngles().size();
        if (s.d) out = (uint64_t)s.d->GetAngles().size();
    });
    return out;
}

void qrack_set_qneuron_angles(quid nid, const double* angles, uint64_t n)
{
    nguarded(nid, [&](NeuronSlot& s) {
        if (s.f) {
            std::vector<float> a(angles, angles + n);
            s.f->SetAngles(a);
        }
        if (s.d) {
            std::vector<double> a(angles, angles + n);
            s.d->SetAngles(a);
        }
    });
}

void qrack_get_qneuron_angles(quid nid, double* angles, uint64_t n)
{
    nguarded(nid, [&](NeuronSlot& s) {
        if (s.f) {
            const auto& a = s.f->GetAngles();
            for (uint64_t i = 0; i < n && i < a.size(); ++i) angles[i] = (double)a[i];
        } else if (s.d) {
            const auto& a = s.d->GetAngles();
            for (uint64_t i = 0; i < n && i < a.size(); ++i) angles[i] = a[i];
        }
    });
}

void qrack_set_qneuron_alpha(quid nid, double alpha)
{
    nguarded(nid, [&](NeuronSlot& s) {
        if (s.f) s.f->SetAlpha((float)alpha);
        if (s.d) s.d->SetAlpha(alpha);
    });
}

void qrack_set_qneuron_activation_fn(quid nid, int fn)
{
    nguarded(nid, [&](NeuronSlot& s) {
        if (s.f) s.f->SetActivationFn((QNeuronActivationFn)fn);
        if (s.d) s.d->SetActivationFn((QNeuronActivationFn)fn);
    });
}

double qrack_qneuron_predict(quid nid, int expected, int resetInit)
{
    double out = 0;
    nguarded(nid, [&](NeuronSlot& s) {
        if (s.f) out = (double)s.f->Predict(expected != 0, resetInit != 0);
        if (s.d) out = s.d->Predict(expected != 0, resetInit != 0);
    });
    return out;
}

double qrack_qneuron_unpredict(quid nid, int expected)
{
    double out = 0;
    nguarded(nid, [&](NeuronSlot& s) {
        if (s.f) out = (double)s.f->Unpredict(expected != 0);
        if (s.d) out = s.d->Unpredict(expected != 0);
    });
    return out;
}

void qrack_qneuron_learn(quid nid, double eta, int expected, int resetInit)
{
    nguarded(nid, [&](NeuronSlot& s) {
        if (s.f) s.f->Learn((float)eta, expected != 0, resetInit != 0);
        if (s.d) s.d->Learn(eta, expected != 0, resetInit != 0);
    });
}

void qrack_qneuron_learn_cycle(quid nid, double eta, int expected)
{
    nguarded(nid, [&](NeuronSlot& s) {
        if (s.f) s.f->Learn((float)eta, expected != 0, false);
        if (s.d) s.d->Learn(eta, expected != 0, false);
    });
}

void qrack_qneuron_learn_permutation(quid nid, double eta, int expected, uint64_t perm)
{
    nguarded(nid, [&](NeuronSlot& s) {
        if (s.f) s.f->LearnPermutation((float)eta, expected != 0, perm);
        if (s.d) s.d->LearnPermutation(eta, expected != 0, perm);
    });
}

} // extern "C"

/* ---- serializable circuit sub-API ---- */

namespace {

struct CircSlot {
    QCircuitPtr<float> f;
    QCircuitPtr<double> d;
    std::mutex op;
};
using CircSlotPtr = std::shared_ptr<CircSlot>;
std::mutex g_cmtx;
std::map<quid, CircSlotPtr> g_circs;
quid g_cnext = 1;

CircSlotPtr circSlot(quid cid)
{
    std::lock_guard<std::mutex> lk(g_cmtx);
    auto it = g_circs.find(cid);
    return (it == g_circs.end()) ? nullptr : it->second;
}

quid registerCirc(CircSlotPtr c)
{
    std::lock_guard<std::mutex> lk(g_cmtx);
    const quid cid = g_cnext++;
    g_circs[cid] = std::move(c);
    return cid;
}

} // namespace

extern "C" {

quid qrack_init_qcircuit(uint64_t qubits)
{
    CircSlotPtr c = std::make_shared<CircSlot>();
    c->f = std::make_shared<QCircuit<float>>((bitLenInt)qubits);
    c->d = std::make_shared<QCircuit<double>>((bitLenInt)qubits);
    return registerCirc(std::move(c));
}

void qrack_destroy_qcircuit(quid cid)
{
    std::lock_guard<std::mutex> lk(g_cmtx);
    g_circs.erase(cid);
}

uint64_t qrack_qcircuit_qubit_count(quid cid)
{
    CircSlotPtr c = circSlot(cid);
    return c ? (uint64_t)c->f->GetQubitCount() : 0;
}

void qrack_qcircuit_append_1qb(quid cid, const double* m8, uint64_t q)
{
    CircSlotPtr c = circSlot(cid);
    if (!c) return;
    std::lock_guard<std::mutex> lk(c->op);
    cplx<float> mf[4];
    cplx<double> md[4];
    mtrxFrom<float>(m8, mf);
    mtrxFrom<double>(m8, md);
    c->f->AppendMtrx(mf, (bitLenInt)q);
    c->d->AppendMtrx(md, (bitLenInt)q);
}

void qrack_qcircuit_append_mc(
    quid cid, const double* m8, const uint64_t* cs, uint64_t nc, uint64_t q, uint64_t perm)
{
    CircSlotPtr c = circSlot(cid);
    if (!c) return;
    std::lock_guard<std::mutex> lk(c->op);
    std::vector<bitLenInt> ctrls = ctrlVec(cs, nc);
    cplx<float> mf[4];
    cplx<double> md[4];
    mtrxFrom<float>(m8, mf);
    mtrxFrom<double>(m8, md);
    c->f->AppendControlled(mf, (bitLenInt)q, ctrls, perm);
    c->d->AppendControlled(md, (bitLenInt)q, ctrls, perm);
}

void qrack_qcircuit_swap(quid cid, uint64_t q1, uint64_t q2)
{
    CircSlotPtr c = circSlot(cid);
    if (!c) return;
    std::lock_guard<std::mutex> lk(c->op);
    c->f->Swap((bitLenInt)q1, (bitLenInt)q2);
    c->d->Swap((bitLenInt)q1, (bitLenInt)q2);
}

void qrack_qcircuit_run(quid cid, quid sid)
{
    CircSlotPtr c = circSlot(cid);
    if (!c) return;
    std::lock_guard<std::mutex> clk(c->op);
    guarded(sid, [&](SimSlot& s) {
        if (s.f) c->f->Run(s.f);
        if (s.d) c->d->Run(s.d);
    });
}

quid qrack_qcircuit_inverse(quid cid)
{
    CircSlotPtr c = circSlot(cid);
    if (!c) return 0;
    std::lock_guard<std::mutex> lk(c->op);
    CircSlotPtr o = std::make_shared<CircSlot>();
    o->f = c->f->Inverse();
    o->d = c->d->Inverse();
    return registerCirc(std::move(o));
}

quid qrack_qcircuit_past_light_cone(quid cid, const uint64_t* qs, uint64_t n)
{
    CircSlotPtr c = circSlot(cid);
    if (!c) return 0;
    std::lock_guard<std::mutex> lk(c->op);
    std::set<bitLenInt> m;
    for (uint64_t i = 0; i < n; ++i) m.insert((bitLenInt)qs[i]);
    CircSlotPtr o = std::make_shared<CircSlot>();
    o->f = c->f->PastLightCone(m);
    o->d = c->d->PastLightCone(m);
    return registerCirc(std::move(o));
}

int qrack_qcircuit_out_to_file(quid cid, const char* path)
{
    CircSlotPtr c = circSlot(cid);
    if (!c) return -1;
    std::lock_guard<std::mutex> lk(c->op);
    try {
        std::ofstream f(path);
        f << c->d->Serialize();
        return 0;
    } catch (const std::exception&) {
        return -1;
    }
}

quid qrack_qcircuit_in_from_file(const char* path)
{
    try {
        std::ifstream f(path);
        std::string text((std::istreambuf_iterator<char>(f)), std::istreambuf_iterator<char>());
        CircSlotPtr c = std::make_shared<CircSlot>();
        c->f = QCircuit<float>::Deserialize(text);
        c->d = QCircuit<double>::Deserialize(text);
        return registerCirc(std::move(c));
    } catch (const std::exception&) {
        return 0;
    }
}

uint64_t qrack_qcircuit_out_to_string_length(quid cid)
{
    CircSlotPtr c = circSlot(cid);
    if (!c) return 0;
    std::lock_guard<std::mutex> lk(c->op);
    return (uint64_t)c->d->Serialize().size() + 1u;
}

void qrack_qcircuit_out_to_string(quid cid, char* out, uint64_t cap)
{
    CircSlotPtr c = circSlot(cid);
    if (!c || !cap) return;
    std::lock_guard<std::mutex> lk(c->op);
    const std::string s = c->d->Serialize();
    const uint64_t n = std::min<uint64_t>(cap - 1u, s.size());
    std::memcpy(out, s.data(), n);
    out[n] = 0;
}

} // extern "C"

#include "pinvoke_compat.inc"
