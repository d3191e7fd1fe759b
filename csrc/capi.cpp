// qrack_amd — C ABI implementation (see include/qrack_amd_capi.h).
// Parity target: /root/reference/src/pinvoke_api.cpp (quid handle table,
// per-simulator error latch polled by get_error).
#include "../include/qrack_amd_capi.h"

#include "qfactory.hpp"
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
    } catch (const std::exception&) {
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

} // extern "C"
