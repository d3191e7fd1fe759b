// pybind11 bindings for qrack_amd.
//
// The product API surface parallels the reference C ABI
// (/root/reference/src/pinvoke_api.cpp) but is exposed as a proper Python
// class per precision (QSimF / QSimD) instead of integer simulator handles;
// the pure-Python package layer (qrack_amd/__init__.py) adds the stack
// factory and torch interop.
#include <pybind11/complex.h>
#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "common/dlpack_min.hpp"
#include "qcircuit.hpp"
#include "qengine_cpu.hpp"
#include "qfactory.hpp"
#include "qhybrid.hpp"
#include "qneuron.hpp"
#include "qunitmulti.hpp"
#include "serialize.hpp"
#ifdef QRACK_AMD_HIP_ENGINE
#include "hip/qengine_hip.hpp"
#endif

namespace py = pybind11;
using namespace qrack_amd;

template <typename R> static const char* dtypeName();
template <> const char* dtypeName<float>() { return "complex64"; }
template <> const char* dtypeName<double>() { return "complex128"; }

template <typename R> static void bindQInterface(py::module_& m, const char* name)
{
    using QI = QInterface<R>;
    using Ptr = QInterfacePtr<R>;
    using C = std::complex<R>;

    auto cls = py::class_<QI, Ptr>(m, name);
    cls.def_property_readonly("num_qubits", &QI::GetQubitCount)
        .def_property_readonly("max_q_power", &QI::GetMaxQPower)
        .def("set_random_seed", &QI::SetRandomSeed)
        // ---- state access ----
        .def("set_permutation",
            [](QI& q, bitCapInt perm) { q.SetPermutation(perm); })
        .def("get_amplitude", [](QI& q, bitCapInt i) { return to_std(q.GetAmplitude(i)); })
        .def("set_amplitude", [](QI& q, bitCapInt i, C a) { q.SetAmplitude(i, from_std<R>(a)); })
        .def("get_state_vector",
            [](QI& q) {
                py::array_t<C> out((py::ssize_t)q.GetMaxQPower());
                q.GetQuantumState(reinterpret_cast<cplx<R>*>(out.mutable_data()));
                return out;
            })
        .def("set_state_vector",
            [](QI& q, py::array_t<C, py::array::c_style | py::array::forcecast> in) {
                if ((bitCapInt)in.size() != q.GetMaxQPower())
                    throw QrackError("state vector size mismatch");
                q.SetQuantumState(reinterpret_cast<const cplx<R>*>(in.data()));
            })
        .def("get_probs",
            [](QI& q) {
                py::array_t<R> out((py::ssize_t)q.GetMaxQPower());
                q.GetProbs(out.mutable_data());
                return out;
            })
        // ---- single-qubit gates ----
        .def("mtrx",
            [](QI& q, std::vector<C> m, bitLenInt t) {
                if (m.size() != 4) throw QrackError("mtrx: need 4 entries");
                const cplx<R> mm[4] = { from_std<R>(m[0]), from_std<R>(m[1]), from_std<R>(m[2]),
                    from_std<R>(m[3]) };
                q.Mtrx(mm, t);
            })
        .def("mtrx_2q",
            [](QI& q, std::vector<C> m, bitLenInt a, bitLenInt b) {
                if (m.size() != 16) throw QrackError("mtrx_2q: need 16 entries");
                std::vector<cplx<R>> mm(16);
                for (int i = 0; i < 16; ++i) mm[i] = from_std<R>(m[i]);
                q.Mtrx2q(mm.data(), a, b);
            })
        .def("mtrx_2q_batch",
            [](QI& q, std::vector<C> ms, std::vector<bitLenInt> q1s, std::vector<bitLenInt> q2s) {
                std::vector<cplx<R>> mm(ms.size());
                for (size_t i = 0; i < ms.size(); ++i) mm[i] = from_std<R>(ms[i]);
                q.Mtrx2qBatch(mm, q1s, q2s);
            })
        .def("fsim_batch",
            [](QI& q, std::vector<R> thetas, std::vector<R> phis, std::vector<bitLenInt> q1s,
                std::vector<bitLenInt> q2s) { q.FSimBatch(thetas, phis, q1s, q2s); })
        .def("cphase_pairs",
            [](QI& q, std::vector<bitLenInt> c, std::vector<bitLenInt> t, std::vector<double> a) {
                q.CPhasePairs(c, t, a);
            })
        .def("cz_batch",
            [](QI& q, std::vector<bitLenInt> c, std::vector<bitLenInt> t) { q.CzBatch(c, t); })
        .def("cnot_batch",
            [](QI& q, std::vector<bitLenInt> controls, std::vector<bitLenInt> targets) {
                q.CnotBatch(controls, targets);
            })
        .def("mtrx_1q_batch",
            [](QI& q, std::vector<bitLenInt> targets, std::vector<C> ms) {
                if (ms.size() != 4u * targets.size())
                    throw QrackError("mtrx_1q_batch: need 4 entries per target");
                std::vector<cplx<R>> mm(ms.size());
                for (size_t i = 0; i < ms.size(); ++i) mm[i] = from_std<R>(ms[i]);
                q.Mtrx1qBatch(targets, mm);
            })
        .def("phase",
            [](QI& q, C tl, C br, bitLenInt t) { q.Phase(from_std<R>(tl), from_std<R>(br), t); })
        .def("invert",
            [](QI& q, C tr, C bl, bitLenInt t) { q.Invert(from_std<R>(tr), from_std<R>(bl), t); })
        .def("x", [](QI& q, bitLenInt t) { q.X(t); })
        .def("y", [](QI& q, bitLenInt t) { q.Y(t); })
        .def("z", [](QI& q, bitLenInt t) { q.Z(t); })
        .def("h", [](QI& q, bitLenInt t) { q.H(t); })
        .def("s", [](QI& q, bitLenInt t) { q.S(t); })
        .def("is_", [](QI& q, bitLenInt t) { q.IS(t); })
        .def("t", [](QI& q, bitLenInt t) { q.T(t); })
        .def("it", [](QI& q, bitLenInt t) { q.IT(t); })
        .def("sqrt_x", [](QI& q, bitLenInt t) { q.SqrtX(t); })
        .def("isqrt_x", [](QI& q, bitLenInt t) { q.ISqrtX(t); })
        .def("rx", [](QI& q, R th, bitLenInt t) { q.RX(th, t); })
        .def("ry", [](QI& q, R th, bitLenInt t) { q.RY(th, t); })
        .def("rz", [](QI& q, R th, bitLenInt t) { q.RZ(th, t); })
        .def("rt", [](QI& q, R th, bitLenInt t) { q.RT(th, t); })
        .def("u", [](QI& q, bitLenInt t, R th, R ph, R lm) { q.U(t, th, ph, lm); })
        .def("phase_root_n", [](QI& q, bitLenInt n, bitLenInt t) { q.PhaseRootN(n, t); })
        // ---- controlled gates ----
        .def("mcmtrx",
            [](QI& q, std::vector<bitLenInt> c, std::vector<C> m, bitLenInt t) {
                const cplx<R> mm[4] = { from_std<R>(m[0]), from_std<R>(m[1]), from_std<R>(m[2]),
                    from_std<R>(m[3]) };
                q.MCMtrx(c, mm, t);
            })
        .def("macmtrx",
            [](QI& q, std::vector<bitLenInt> c, std::vector<C> m, bitLenInt t) {
                const cplx<R> mm[4] = { from_std<R>(m[0]), from_std<R>(m[1]), from_std<R>(m[2]),
                    from_std<R>(m[3]) };
                q.MACMtrx(c, mm, t);
            })
        .def("ucmtrx",
            [](QI& q, std::vector<bitLenInt> c, std::vector<C> m, bitLenInt t, bitCapInt perm) {
                const cplx<R> mm[4] = { from_std<R>(m[0]), from_std<R>(m[1]), from_std<R>(m[2]),
                    from_std<R>(m[3]) };
                q.UCMtrx(c, mm, t, perm);
            })
        .def("uniformly_controlled_single_bit",
            [](QI& q, std::vector<bitLenInt> c, bitLenInt t,
                py::array_t<C, py::array::c_style | py::array::forcecast> mtrxs) {
                if ((size_t)mtrxs.size() != (size_t)(4u * pow2((bitLenInt)c.size())))
                    throw QrackError("multiplexer: need 4*2^len(controls) entries");
                q.UniformlyControlledSingleBit(
                    c, t, reinterpret_cast<const cplx<R>*>(mtrxs.data()));
            })
        .def("mcphase",
            [](QI& q, std::vector<bitLenInt> c, C tl, C br, bitLenInt t) {
                q.MCPhase(c, from_std<R>(tl), from_std<R>(br), t);
            })
        .def("mcinvert",
            [](QI& q, std::vector<bitLenInt> c, C tr, C bl, bitLenInt t) {
                q.MCInvert(c, from_std<R>(tr), from_std<R>(bl), t);
            })
        .def("macphase",
            [](QI& q, std::vector<bitLenInt> c, C tl, C br, bitLenInt t) {
                q.MACPhase(c, from_std<R>(tl), from_std<R>(br), t);
            })
        .def("macinvert",
            [](QI& q, std::vector<bitLenInt> c, C tr, C bl, bitLenInt t) {
                q.MACInvert(c, from_std<R>(tr), from_std<R>(bl), t);
            })
        .def("cnot", [](QI& q, bitLenInt c, bitLenInt t) { q.CNOT(c, t); })
        .def("anti_cnot", [](QI& q, bitLenInt c, bitLenInt t) { q.AntiCNOT(c, t); })
        .def("ccnot", [](QI& q, bitLenInt c1, bitLenInt c2, bitLenInt t) { q.CCNOT(c1, c2, t); })
        .def("cy", [](QI& q, bitLenInt c, bitLenInt t) { q.CY(c, t); })
        .def("cz", [](QI& q, bitLenInt c, bitLenInt t) { q.CZ(c, t); })
        .def("ccz", [](QI& q, bitLenInt c1, bitLenInt c2, bitLenInt t) { q.CCZ(c1, c2, t); })
        .def("ch", [](QI& q, bitLenInt c, bitLenInt t) { q.CH(c, t); })
        .def("cs", [](QI& q, bitLenInt c, bitLenInt t) { q.CS(c, t); })
        .def("cphase_root_n", [](QI& q, bitLenInt n, bitLenInt c, bitLenInt t) { q.CPhaseRootN(n, c, t); })
        .def("crz", [](QI& q, R th, bitLenInt c, bitLenInt t) { q.CRZ(th, c, t); })
        .def("crx", [](QI& q, R th, bitLenInt c, bitLenInt t) { q.CRX(th, c, t); })
        .def("cry", [](QI& q, R th, bitLenInt c, bitLenInt t) { q.CRY(th, c, t); })
        .def("crt", [](QI& q, R th, bitLenInt c, bitLenInt t) { q.CRT(th, c, t); })
        .def("ct", &QI::CT)
        .def("cit", &QI::CIT)
        .def("ccy", &QI::CCY)
        .def("anti_cy", &QI::AntiCY)
        .def("anti_ccy", &QI::AntiCCY)
        .def("anti_ccz", &QI::AntiCCZ)
        .def("anti_ch", &QI::AntiCH)
        .def("anti_cs", &QI::AntiCS)
        .def("anti_cis", &QI::AntiCIS)
        .def("anti_ct", &QI::AntiCT)
        .def("anti_cit", &QI::AntiCIT)
        .def("anti_cphase_root_n", &QI::AntiCPhaseRootN)
        .def("anti_ciphase_root_n", &QI::AntiCIPhaseRootN)
        .def("cu", [](QI& q, std::vector<bitLenInt> c, bitLenInt t, R th, R ph, R lm) {
            q.CU(c, t, th, ph, lm);
        })
        .def("anti_cu", [](QI& q, std::vector<bitLenInt> c, bitLenInt t, R th, R ph, R lm) {
            q.AntiCU(c, t, th, ph, lm);
        })
        .def("u2", [](QI& q, bitLenInt t, R ph, R lm) { q.U2(t, ph, lm); })
        .def("iu2", [](QI& q, bitLenInt t, R ph, R lm) { q.IU2(t, ph, lm); })
        .def("ai", [](QI& q, bitLenInt t, R az, R incl) { q.AI(t, az, incl); })
        .def("iai", [](QI& q, bitLenInt t, R az, R incl) { q.IAI(t, az, incl); })
        .def("cai", [](QI& q, bitLenInt c, bitLenInt t, R az, R incl) { q.CAI(c, t, az, incl); })
        .def("anti_cai", [](QI& q, bitLenInt c, bitLenInt t, R az, R incl) { q.AntiCAI(c, t, az, incl); })
        .def("ciai", [](QI& q, bitLenInt c, bitLenInt t, R az, R incl) { q.CIAI(c, t, az, incl); })
        .def("anti_ciai", [](QI& q, bitLenInt c, bitLenInt t, R az, R incl) { q.AntiCIAI(c, t, az, incl); })
        .def("sqrt_h", &QI::SqrtH)
        .def("sh", &QI::SH)
        .def("his", &QI::HIS)
        .def("sqrt_w", &QI::SqrtW)
        .def("isqrt_w", &QI::ISqrtW)
        .def("sqrt_y", &QI::SqrtY)
        .def("isqrt_y", &QI::ISqrtY)
        .def("phase_root_n_mask", &QI::PhaseRootNMask)
        .def("uc_phase", [](QI& q, std::vector<bitLenInt> c, C tl, C br, bitLenInt t, bitCapInt perm) {
            q.UCPhase(c, from_std<R>(tl), from_std<R>(br), t, perm);
        })
        .def("uc_invert", [](QI& q, std::vector<bitLenInt> c, C tr, C bl, bitLenInt t, bitCapInt perm) {
            q.UCInvert(c, from_std<R>(tr), from_std<R>(bl), t, perm);
        })
        .def("exp_", [](QI& q, R r, bitLenInt t) { q.Exp(r, t); })
        .def("exp_x", [](QI& q, R r, bitLenInt t) { q.ExpX(r, t); })
        .def("exp_y", [](QI& q, R r, bitLenInt t) { q.ExpY(r, t); })
        .def("exp_z", [](QI& q, R r, bitLenInt t) { q.ExpZ(r, t); })
        .def("rx_dyad", &QI::RXDyad)
        .def("ry_dyad", &QI::RYDyad)
        .def("rz_dyad", &QI::RZDyad)
        .def("rt_dyad", &QI::RTDyad)
        .def("exp_dyad", &QI::ExpDyad)
        .def("exp_x_dyad", &QI::ExpXDyad)
        .def("exp_y_dyad", &QI::ExpYDyad)
        .def("exp_z_dyad", &QI::ExpZDyad)
        .def("crx_dyad", &QI::CRXDyad)
        .def("cry_dyad", &QI::CRYDyad)
        .def("crz_dyad", &QI::CRZDyad)
        .def("crt_dyad", &QI::CRTDyad)
        .def("uniformly_controlled_ry",
            [](QI& q, std::vector<bitLenInt> c, bitLenInt t, std::vector<R> angles) {
                q.UniformlyControlledRY(c, t, angles);
            })
        .def("uniformly_controlled_rz",
            [](QI& q, std::vector<bitLenInt> c, bitLenInt t, std::vector<R> angles) {
                q.UniformlyControlledRZ(c, t, angles);
            })
        .def("cisqrt_swap", [](QI& q, std::vector<bitLenInt> c, bitLenInt a, bitLenInt b) {
            q.CISqrtSwap(c, a, b);
        })
        .def("anti_cisqrt_swap", [](QI& q, std::vector<bitLenInt> c, bitLenInt a, bitLenInt b) {
            q.AntiCISqrtSwap(c, a, b);
        })
        // boolean logic, shifts, adders, classical assignment
        .def("and_", &QI::AND)
        .def("or_", &QI::OR)
        .def("xor_", &QI::XOR)
        .def("nand", &QI::NAND)
        .def("nor", &QI::NOR)
        .def("xnor", &QI::XNOR)
        .def("cland", &QI::CLAND)
        .def("clor", &QI::CLOR)
        .def("clxor", &QI::CLXOR)
        .def("clnand", &QI::CLNAND)
        .def("clnor", &QI::CLNOR)
        .def("clxnor", &QI::CLXNOR)
        .def("asl", &QI::ASL)
        .def("asr", &QI::ASR)
        .def("lsl", &QI::LSL)
        .def("lsr", &QI::LSR)
        .def("adc", &QI::ADC)
        .def("iadc", &QI::IADC)
        .def("cadc", &QI::CADC)
        .def("ciadc", &QI::CIADC)
        .def("cfull_add", &QI::CFullAdd)
        .def("cifull_add", &QI::CIFullAdd)
        .def("set_bit", &QI::SetBit)
        .def("set_reg", &QI::SetReg)
        .def("reverse", &QI::Reverse)
        // ---- swaps ----
        .def("swap", [](QI& q, bitLenInt a, bitLenInt b) { q.Swap(a, b); })
        .def("iswap", [](QI& q, bitLenInt a, bitLenInt b) { q.ISwap(a, b); })
        .def("iiswap", [](QI& q, bitLenInt a, bitLenInt b) { q.IISwap(a, b); })
        .def("sqrt_swap", [](QI& q, bitLenInt a, bitLenInt b) { q.SqrtSwap(a, b); })
        .def("isqrt_swap", [](QI& q, bitLenInt a, bitLenInt b) { q.ISqrtSwap(a, b); })
        .def("fsim", [](QI& q, R th, R ph, bitLenInt a, bitLenInt b) { q.FSim(th, ph, a, b); })
        .def("cswap",
            [](QI& q, std::vector<bitLenInt> c, bitLenInt a, bitLenInt b) { q.CSwap(c, a, b); })
        .def("csqrt_swap",
            [](QI& q, std::vector<bitLenInt> c, bitLenInt a, bitLenInt b) { q.CSqrtSwap(c, a, b); })
        // ---- mask gates ----
        .def("x_mask", [](QI& q, bitCapInt m) { q.XMask(m); })
        .def("y_mask", [](QI& q, bitCapInt m) { q.YMask(m); })
        .def("z_mask", [](QI& q, bitCapInt m) { q.ZMask(m); })
        .def("phase_parity", [](QI& q, R r, bitCapInt m) { q.PhaseParity(r, m); })
        .def("uniform_parity_rz", [](QI& q, bitCapInt m, R a) { q.UniformParityRZ(m, a); })
        .def("cuniform_parity_rz",
            [](QI& q, std::vector<bitLenInt> c, bitCapInt m, R a) { q.CUniformParityRZ(c, m, a); })
        // ---- QFT ----
        .def("qft", [](QI& q, bitLenInt s, bitLenInt l) { q.QFT(s, l); }, py::arg("start") = 0,
            py::arg("length") = 0)
        .def("iqft", [](QI& q, bitLenInt s, bitLenInt l) { q.IQFT(s, l); }, py::arg("start") = 0,
            py::arg("length") = 0)
        .def("qftr", [](QI& q, std::vector<bitLenInt> qs) { q.QFTR(qs); })
        .def("iqftr", [](QI& q, std::vector<bitLenInt> qs) { q.IQFTR(qs); })
        // ---- structural ----
        .def("compose", [](QI& q, Ptr other) { return q.Compose(other); })
        .def("compose_at", [](QI& q, Ptr other, bitLenInt start) { return q.Compose(other, start); })
        .def("decompose", [](QI& q, bitLenInt start, Ptr dest) { q.Decompose(start, dest); })
        .def("dispose", [](QI& q, bitLenInt start, bitLenInt len) { q.Dispose(start, len); })
        .def("dispose_perm",
            [](QI& q, bitLenInt start, bitLenInt len, bitCapInt perm) { q.Dispose(start, len, perm); })
        .def("allocate", [](QI& q, bitLenInt len) { return q.Allocate(len); })
        .def("allocate",
            [](QI& q, bitLenInt start, bitLenInt len) { return q.Allocate(start, len); })
        .def("clone", [](QI& q) { return q.Clone(); })
        .def("try_separate_1", [](QI& q, bitLenInt qb) { return q.TrySeparate(qb); })
        .def("try_separate_2", [](QI& q, bitLenInt q1, bitLenInt q2) { return q.TrySeparate(q1, q2); })
        .def("try_separate", [](QI& q, bitLenInt x) { return q.TrySeparate(x); })
        .def("try_separate",
            [](QI& q, bitLenInt x, bitLenInt y) { return q.TrySeparate(x, y); })
        .def("try_separate",
            [](QI& q, std::vector<bitLenInt> qs, R tol) { return q.TrySeparate(qs, tol); })
        // ---- probability / measurement ----
        .def("prob", &QI::Prob)
        .def("prob_all", &QI::ProbAll)
        .def("prob_mask", &QI::ProbMask)
        .def("prob_reg", &QI::ProbReg)
        .def("prob_parity", &QI::ProbParity)
        .def("force_m", &QI::ForceM, py::arg("qubit"), py::arg("result"), py::arg("do_force") = true,
            py::arg("do_apply") = true)
        .def("m", [](QI& q, bitLenInt t) { return q.M(t); })
        .def("m_all", [](QI& q) { return q.MAll(); })
        .def("m_all_big",
            [](QI& q) {
                // >64-qubit terminal measurement via the packed BigCap path
                // (C++ MAllWide; Python int only for >128q tails)
                if (q.GetQubitCount() <= 128u) {
                    const BigCap r = q.MAllWide();
                    return py::int_(
                        (py::int_((uint64_t)r.hi) << py::int_(64)) | py::int_((uint64_t)r.lo));
                }
                py::int_ out(0);
                const py::int_ one(1);
                for (bitLenInt i = 0; i < q.GetQubitCount(); ++i) {
                    if (q.M(i)) {
                        out = py::int_(out | (one << py::int_(i)));
                    }
                }
                return out;
            })
        .def("set_permutation_big",
            [](QI& q, py::object perm) {
                // packed wide permutation init (BigCap SetPermutationWide)
                py::int_ p(perm);
                const uint64_t lo =
                    py::cast<uint64_t>(p & py::int_(0xFFFFFFFFFFFFFFFFull));
                const uint64_t hi = py::cast<uint64_t>(
                    (p >> py::int_(64)) & py::int_(0xFFFFFFFFFFFFFFFFull));
                q.SetPermutationWide(BigCap(lo, hi));
            })
        .def("multi_shot_measure_qubits",
            [](QI& q, std::vector<bitLenInt> qubits, unsigned shots) {
                // wide-safe sampling: qubit INDICES in, position-packed
                // values out (valid for any logical width)
                return q.MultiShotMeasureQubits(qubits, shots);
            })
        .def("sample_clone_big",
            [](Ptr q) {
                QInterfacePtr<R> c = q->Clone();
                py::int_ out(0);
                const py::int_ one(1);
                for (bitLenInt i = 0; i < c->GetQubitCount(); ++i) {
                    if (c->M(i)) {
                        out = py::int_(out | (one << py::int_(i)));
                    }
                }
                return out;
            })
        .def("m_reg", [](QI& q, bitLenInt s, bitLenInt l) { return q.MReg(s, l); })
        .def("force_m_reg", &QI::ForceMReg, py::arg("start"), py::arg("length"), py::arg("result"),
            py::arg("do_force") = true, py::arg("do_apply") = true)
        .def("force_m_parity", &QI::ForceMParity, py::arg("mask"), py::arg("result"),
            py::arg("do_force") = true)
        .def("multi_shot_measure_mask",
            [](QI& q, std::vector<bitCapInt> qPowers, unsigned shots) {
                return q.MultiShotMeasureMask(qPowers, shots);
            })
        .def("expectation_bits_all",
            [](QI& q, std::vector<bitLenInt> bits) { return q.ExpectationBitsAll(bits); })
        .def("variance_bits_all",
            [](QI& q, std::vector<bitLenInt> bits) { return q.VarianceBitsAll(bits); })
        .def("pauli_expectation",
            [](QI& q, std::vector<bitLenInt> bits, std::vector<int> paulis) {
                std::vector<Pauli> ps;
                for (int p : paulis) ps.push_back((Pauli)p);
                return q.PauliExpectation(bits, ps);
            })
        .def("expectation_pauli_all",
            [](QI& q, std::vector<bitLenInt> bits, std::vector<int> paulis) {
                std::vector<Pauli> ps;
                for (int p : paulis) ps.push_back((Pauli)p);
                return q.ExpectationPauliAll(bits, ps);
            })
        .def("variance_pauli_all",
            [](QI& q, std::vector<bitLenInt> bits, std::vector<int> paulis) {
                std::vector<Pauli> ps;
                for (int p : paulis) ps.push_back((Pauli)p);
                return q.VariancePauliAll(bits, ps);
            })
        .def("prob_bits_all",
            [](QI& q, std::vector<bitLenInt> bits) {
                py::array_t<double> out((py::ssize_t)pow2((bitLenInt)bits.size()));
                q.ProbBitsAll(bits, out.mutable_data());
                return out;
            })
        .def("prob_mask_all",
            [](QI& q, bitCapInt mask) {
                int k = 0;
                for (bitCapInt m = mask; m; m &= m - 1u) ++k;
                py::array_t<double> out((py::ssize_t)pow2((bitLenInt)k));
                q.ProbMaskAll(mask, out.mutable_data());
                return out;
            })
        .def("expectation_bits_factorized",
            [](QI& q, std::vector<bitLenInt> bits, std::vector<bitCapInt> perms, bitCapInt offset) {
                return q.ExpectationBitsFactorized(bits, perms, offset);
            }, py::arg("bits"), py::arg("perms"), py::arg("offset") = 0)
        .def("variance_bits_factorized",
            [](QI& q, std::vector<bitLenInt> bits, std::vector<bitCapInt> perms, bitCapInt offset) {
                return q.VarianceBitsFactorized(bits, perms, offset);
            }, py::arg("bits"), py::arg("perms"), py::arg("offset") = 0)
        .def("expectation_floats_factorized",
            [](QI& q, std::vector<bitLenInt> bits, std::vector<double> w) {
                return q.ExpectationFloatsFactorized(bits, w);
            })
        .def("variance_floats_factorized",
            [](QI& q, std::vector<bitLenInt> bits, std::vector<double> w) {
                return q.VarianceFloatsFactorized(bits, w);
            })
        .def("expectation_unitary_all",
            [](QI& q, std::vector<bitLenInt> bits, std::vector<C> ops, std::vector<double> ev) {
                std::vector<cplx<R>> mm(ops.size());
                for (size_t i = 0; i < ops.size(); ++i) mm[i] = from_std<R>(ops[i]);
                return q.ExpectationUnitaryAll(bits, mm, ev);
            }, py::arg("bits"), py::arg("basis_ops"), py::arg("eigen_vals") = std::vector<double>())
        .def("variance_unitary_all",
            [](QI& q, std::vector<bitLenInt> bits, std::vector<C> ops, std::vector<double> ev) {
                std::vector<cplx<R>> mm(ops.size());
                for (size_t i = 0; i < ops.size(); ++i) mm[i] = from_std<R>(ops[i]);
                return q.VarianceUnitaryAll(bits, mm, ev);
            }, py::arg("bits"), py::arg("basis_ops"), py::arg("eigen_vals") = std::vector<double>())
        // ---- ALU ----
        .def("inc", &QI::INC)
        .def("dec", &QI::DEC)
        .def("cinc", &QI::CINC)
        .def("incc", &QI::INCC)
        .def("decc", &QI::DECC)
        .def("incs", &QI::INCS)
        .def("incbcd", &QI::INCBCD)
        .def("decbcd", &QI::DECBCD)
        .def("mul", &QI::MUL)
        .def("div", &QI::DIV)
        .def("mul_mod_n_out", &QI::MULModNOut)
        .def("imul_mod_n_out", &QI::IMULModNOut)
        .def("pow_mod_n_out", &QI::POWModNOut)
        .def("cmul", &QI::CMUL)
        .def("cdiv", &QI::CDIV)
        .def("cmul_mod_n_out", &QI::CMULModNOut)
        .def("cpow_mod_n_out", &QI::CPOWModNOut)
        .def("indexed_lda",
            [](QI& q, bitLenInt is, bitLenInt il, bitLenInt vs, bitLenInt vl,
                py::bytes values, bool reset) {
                std::string v = values;
                return q.IndexedLDA(is, il, vs, vl, (const unsigned char*)v.data(), reset);
            },
            py::arg("index_start"), py::arg("index_length"), py::arg("value_start"),
            py::arg("value_length"), py::arg("values"), py::arg("reset_value") = true)
        .def("indexed_adc",
            [](QI& q, bitLenInt is, bitLenInt il, bitLenInt vs, bitLenInt vl, bitLenInt c,
                py::bytes values) {
                std::string v = values;
                return q.IndexedADC(is, il, vs, vl, c, (const unsigned char*)v.data());
            })
        .def("indexed_sbc",
            [](QI& q, bitLenInt is, bitLenInt il, bitLenInt vs, bitLenInt vl, bitLenInt c,
                py::bytes values) {
                std::string v = values;
                return q.IndexedSBC(is, il, vs, vl, c, (const unsigned char*)v.data());
            })
        .def("hash",
            [](QI& q, bitLenInt s, bitLenInt l, py::bytes values) {
                std::string v = values;
                q.Hash(s, l, (const unsigned char*)v.data());
            })
        .def("full_add", &QI::FullAdd)
        .def("ifull_add", &QI::IFullAdd)
        .def("phase_flip_if_less", &QI::PhaseFlipIfLess)
        .def("cphase_flip_if_less", &QI::CPhaseFlipIfLess)
        .def("zero_phase_flip", &QI::ZeroPhaseFlip)
        .def("phase_flip", &QI::PhaseFlip)
        .def("rol", &QI::ROL)
        .def("ror", &QI::ROR)
        // ---- time evolution ----
        .def("time_evolve",
            [](QI& q, py::list ops, R t) {
                std::vector<HamiltonianOp<R>> h;
                for (auto item : ops) {
                    py::dict d = item.cast<py::dict>();
                    HamiltonianOp<R> op;
                    op.target = d["target"].cast<bitLenInt>();
                    if (d.contains("controls"))
                        op.controls = d["controls"].cast<std::vector<bitLenInt>>();
                    if (d.contains("anti")) op.anti = d["anti"].cast<bool>();
                    if (d.contains("uniform")) op.uniform = d["uniform"].cast<bool>();
                    auto mat = d["matrix"].cast<std::vector<C>>();
                    for (C c : mat) op.matrix.push_back(from_std<R>(c));
                    h.push_back(std::move(op));
                }
                q.TimeEvolve(h, t);
            })
        // ---- norm / compare ----
        .def("update_running_norm", [](QI& q) { q.UpdateRunningNorm(); })
        .def("normalize_state", [](QI& q) { q.NormalizeState(); })
        .def("sum_sqr_diff", &QI::SumSqrDiff)
        .def("approx_compare", &QI::ApproxCompare, py::arg("other"), py::arg("error_tol") = (R)1e-4)
        .def("finish", &QI::Finish)
        .def("is_finished", &QI::isFinished)
        .def("get_unitary_fidelity", &QI::GetUnitaryFidelity)
        .def("set_sdrp", &QI::SetSdrp)
        .def("get_sdrp", &QI::GetSdrp)
        .def("set_ncrp", &QI::SetNcrp)
        .def("get_ncrp", &QI::GetNcrp)
        .def("set_reactive_separate", &QI::SetReactiveSeparate)
        .def("set_stochastic", &QI::SetStochastic)
        .def("get_reactive_separate", &QI::GetReactiveSeparate)
        .def("set_noise_parameter", &QI::SetNoiseParameter)
        .def("get_noise_parameter", &QI::GetNoiseParameter)
        .def("set_ace_max_qubits", &QI::SetAceMaxQubits)
        .def("get_ace_max_qubits", &QI::GetAceMaxQubits)
        .def("set_concurrency", &QI::SetConcurrency)
        .def("set_t_injection", &QI::SetTInjection)
        .def("get_t_injection", &QI::GetTInjection)
        .def("first_nonzero_phase", &QI::FirstNonzeroPhase)
        .def("highest_prob_all", &QI::HighestProbAll)
        .def("sample_clone", &QI::SampleClone)
        .def("try_decompose",
            [](QI& q, bitLenInt start, Ptr dest) { return q.TryDecompose(start, dest); })
        .def("get_device_list", &QI::GetDeviceList)
        .def("are_factorized",
            [](QI& q, std::vector<bitLenInt> a, std::vector<bitLenInt> b, bool flush) {
                return q.AreFactorized(a, b, flush);
            }, py::arg("a"), py::arg("b"), py::arg("flush_cache") = false)
        .def("get_amplitude_count", &QI::GetAmplitudeCount)
        .def("set_sparse_ace_max_mb", &QI::SetSparseAceMaxMb)
        .def("set_sparse_probability_floor", &QI::SetSparseProbabilityFloor)
        .def("reset_unitary_fidelity", &QI::ResetUnitaryFidelity)
        .def("set_device", &QI::SetDevice)
        .def("get_device", &QI::GetDevice)
        // ---- engine-level primitives (QPager / distributed pager support) ----
        .def("apply_m",
            [](Ptr q, bitCapInt mask, bitCapInt result, C nrm) {
                auto eng = std::dynamic_pointer_cast<QEngine<R>>(q);
                if (!eng) throw QrackError("apply_m requires a state-vector engine");
                eng->ApplyM(mask, result, from_std<R>(nrm));
            })
        .def("zero_amplitudes",
            [](Ptr q) {
                auto eng = std::dynamic_pointer_cast<QEngine<R>>(q);
                if (!eng) throw QrackError("zero_amplitudes requires a state-vector engine");
                eng->ZeroAmplitudes();
            })
        .def("global_phase", [](QI& q, C f) { q.Phase(from_std<R>(f), from_std<R>(f), 0); })
        .def("phase_ramp",
            [](Ptr q, R scale, bitLenInt rampStart, bitLenInt rampBits, bitCapInt condPower) {
                auto eng = std::dynamic_pointer_cast<QEngine<R>>(q);
                if (!eng) throw QrackError("phase_ramp requires a state-vector engine");
                eng->PhaseRamp(scale, rampStart, rampBits, condPower);
            })
        .def("phase_ramp_general",
            [](Ptr q, R scale, bitLenInt rampStart, bitCapInt inPlaceRelMask,
                std::vector<bitCapInt> sPows, std::vector<uint64_t> sWeights,
                bitCapInt condPower) {
                auto eng = std::dynamic_pointer_cast<QEngine<R>>(q);
                if (!eng) throw QrackError("phase_ramp_general requires an engine");
                eng->PhaseRampGeneral(scale, rampStart, inPlaceRelMask, sPows, sWeights, condPower);
            })
        .def("qft_column_general",
            [](Ptr q, bitLenInt target, double scale, bitLenInt rampStart,
                bitCapInt inPlaceRelMask, std::vector<bitCapInt> sPows,
                std::vector<uint64_t> sWeights, double phase0, bool pre) {
                auto eng = std::dynamic_pointer_cast<QEngine<R>>(q);
                if (!eng) throw QrackError("qft_column_general requires an engine");
                eng->QftColumnGeneral(
                    target, scale, rampStart, inPlaceRelMask, sPows, sWeights, phase0, pre);
            },
            py::arg("target"), py::arg("scale"), py::arg("ramp_start"),
            py::arg("in_place_mask"), py::arg("pows"), py::arg("weights"),
            py::arg("phase0") = 0.0, py::arg("pre") = false)
#ifdef QRACK_AMD_HIP_ENGINE
        .def("qft_column2_general",
            [](Ptr q, bitLenInt targetHi, bitLenInt targetLo, double scale, bitLenInt rampStart,
                bitCapInt inPlaceRelMask, std::vector<bitCapInt> sPows,
                std::vector<uint64_t> sWeights, double phase0Hi, double phase0Lo, bool pre) {
                auto eng = std::dynamic_pointer_cast<QEngineHIP<R>>(q);
                if (!eng) throw QrackError("qft_column2_general requires the HIP engine");
                eng->QftColumn2General(targetHi, targetLo, scale, rampStart, inPlaceRelMask,
                    sPows, sWeights, phase0Hi, phase0Lo, pre);
            },
            py::arg("target_hi"), py::arg("target_lo"), py::arg("scale"), py::arg("ramp_start"),
            py::arg("in_place_mask"), py::arg("pows"), py::arg("weights"), py::arg("phase0_hi"),
            py::arg("phase0_lo"), py::arg("pre") = false)
        .def("qft_column_top_range",
            [](Ptr q, double scale, bitLenInt rampStart, bitCapInt inPlaceRelMask,
                std::vector<bitCapInt> sPows, std::vector<uint64_t> sWeights, double phase0,
                bool pre, uint64_t itLo, uint64_t itHi, uintptr_t recvPtr, bool recvIsLow,
                uintptr_t extStream) {
                auto eng = std::dynamic_pointer_cast<QEngineHIP<R>>(q);
                if (!eng) throw QrackError("qft_column_top_range requires the HIP engine");
                eng->QftColumnTopRange(scale, rampStart, inPlaceRelMask, sPows, sWeights, phase0,
                    pre, itLo, itHi, recvPtr, recvIsLow, extStream);
            },
            py::arg("scale"), py::arg("ramp_start"), py::arg("in_place_mask"), py::arg("pows"),
            py::arg("weights"), py::arg("phase0"), py::arg("pre"), py::arg("it_lo"),
            py::arg("it_hi"), py::arg("recv_ptr"), py::arg("recv_is_low"),
            py::arg("ext_stream") = 0)
#endif
        .def("norm_total",
            [](Ptr q) {
                auto eng = std::dynamic_pointer_cast<QEngine<R>>(q);
                if (!eng) throw QrackError("norm_total requires a state-vector engine");
                eng->UpdateRunningNorm((R)0);
                return (double)eng->GetRunningNorm();
            })
        .def("get_amplitude_page",
            [](Ptr q, bitCapInt offset, bitCapInt length) {
                auto eng = std::dynamic_pointer_cast<QEngine<R>>(q);
                if (!eng) throw QrackError("get_amplitude_page requires an engine");
                py::array_t<C> out((py::ssize_t)length);
                eng->GetAmplitudePage(reinterpret_cast<cplx<R>*>(out.mutable_data()), offset, length);
                return out;
            })
        .def("set_amplitude_page",
            [](Ptr q, py::array_t<C, py::array::c_style | py::array::forcecast> in, bitCapInt offset) {
                auto eng = std::dynamic_pointer_cast<QEngine<R>>(q);
                if (!eng) throw QrackError("set_amplitude_page requires an engine");
                eng->SetAmplitudePage(
                    reinterpret_cast<const cplx<R>*>(in.data()), offset, (bitCapInt)in.size());
            })
        .def("dlpack_view",
            [](Ptr q, bitCapInt offset, bitCapInt length) {
                // zero-copy 1-D complex view of the amplitude buffer, consumed by
                // torch.from_dlpack for RCCL exchanges. The engine must stay
                // alive and un-resized while the view is in use; the holder
                // keeps a shared_ptr so teardown order is safe.
                struct Holder {
                    DLManagedTensor mt;
                    int64_t shape[1];
                    QInterfacePtr<R> keep;
                };
                void* data = nullptr;
                DLDevice dev{};
#ifdef QRACK_AMD_HIP_ENGINE
                if (auto hip = std::dynamic_pointer_cast<QEngineHIP<R>>(q)) {
                    hip->Finish();
                    data = (void*)(hip->DeviceBuffer() + offset);
                    dev = { kDLROCM, hip->DeviceId() };
                }
#endif
                if (!data) {
                    if (auto cpu = std::dynamic_pointer_cast<QEngineCPU<R>>(q)) {
                        data = (void*)(cpu->Amplitudes() + offset);
                        dev = { kDLCPU, 0 };
                    } else {
                        throw QrackError("dlpack_view requires a state-vector engine");
                    }
                }
                Holder* h = new Holder{};
                h->keep = q;
                h->shape[0] = (int64_t)length;
                h->mt.dl_tensor.data = data;
                h->mt.dl_tensor.device = dev;
                h->mt.dl_tensor.ndim = 1;
                h->mt.dl_tensor.dtype = { kDLComplex, (uint8_t)(sizeof(cplx<R>) * 8), 1 };
                h->mt.dl_tensor.shape = h->shape;
                h->mt.dl_tensor.strides = nullptr;
                h->mt.dl_tensor.byte_offset = 0;
                h->mt.manager_ctx = h;
                h->mt.deleter = [](DLManagedTensor* t) { delete (Holder*)t->manager_ctx; };
                return py::capsule(&h->mt, "dltensor", [](PyObject* cap) {
                    if (PyCapsule_IsValid(cap, "dltensor")) {
                        DLManagedTensor* mt =
                            (DLManagedTensor*)PyCapsule_GetPointer(cap, "dltensor");
                        if (mt && mt->deleter) mt->deleter(mt);
                    }
                });
            })
        .def("is_clifford", [](QI& q) { return q.isClifford(); })
        .def("unit_placement",
            [](Ptr q) {
                // (unit width, device id) per distinct QUnitMulti unit
                std::vector<std::pair<int, long long>> out;
                if (auto um = std::dynamic_pointer_cast<QUnitMulti<R>>(q)) {
                    for (auto& p : um->UnitPlacement()) {
                        out.push_back({ (int)p.first, (long long)p.second });
                    }
                }
                return out;
            })
        .def("hybrid_mode",
            [](Ptr q) -> std::string {
                if (auto hy = std::dynamic_pointer_cast<QHybrid<R>>(q)) return hy->ModeName();
                return "n/a";
            })
        .def("ancilla_count",
            [](Ptr q) -> int {
                // T-gadget ancillae pending on a stabilizer-hybrid layer
                if (auto hy = std::dynamic_pointer_cast<QStabilizerHybrid<R>>(q)) {
                    return (int)hy->GetAncillaCount();
                }
                return 0;
            })
        .def("depolarizing_channel_weak_1qb", &QI::DepolarizingChannelWeak1Qb)
        .def("reduced_density_matrix", [](QI& q, bitLenInt qb) {
            cplx<R> rho[4];
            q.GetReducedDensityMatrix(qb, rho);
            py::array_t<C> out({ 2, 2 });
            auto* p = reinterpret_cast<cplx<R>*>(out.mutable_data());
            for (int i = 0; i < 4; ++i) p[i] = rho[i];
            return out;
        });
}

template <typename R> static void bindExtras(py::module_& m, const char* suffix)
{
    using QI = QInterface<R>;
    using Ptr = QInterfacePtr<R>;
    using C = std::complex<R>;

    // ---- QCircuit (parity: include/qcircuit.hpp) ----
    py::class_<QCircuit<R>, QCircuitPtr<R>>(m, (std::string("QCircuit") + suffix).c_str())
        .def(py::init<bitLenInt>())
        .def_property_readonly("num_qubits", &QCircuit<R>::GetQubitCount)
        .def_property_readonly("gate_count", &QCircuit<R>::GetGateCount)
        .def("append_mtrx",
            [](QCircuit<R>& c, std::vector<C> mv, bitLenInt t) {
                cplx<R> m[4];
                for (int i = 0; i < 4; ++i) m[i] = from_std<R>(mv[i]);
                c.AppendMtrx(m, t);
            })
        .def("append_controlled",
            [](QCircuit<R>& c, std::vector<C> mv, bitLenInt t, std::vector<bitLenInt> ctrls,
                bitCapInt perm) {
                cplx<R> m[4];
                for (int i = 0; i < 4; ++i) m[i] = from_std<R>(mv[i]);
                c.AppendControlled(m, t, ctrls, perm);
            })
        .def("swap", &QCircuit<R>::Swap)
        .def("run", [](QCircuit<R>& c, Ptr q) { c.Run(q); })
        .def("inverse", &QCircuit<R>::Inverse)
        .def("past_light_cone",
            [](QCircuit<R>& c, std::vector<bitLenInt> qs) {
                return c.PastLightCone(std::set<bitLenInt>(qs.begin(), qs.end()));
            })
        .def("serialize", &QCircuit<R>::Serialize)
        .def_static("deserialize", &QCircuit<R>::Deserialize);

    // ---- QNeuron (parity: include/qneuron.hpp) ----
    py::class_<QNeuron<R>, std::shared_ptr<QNeuron<R>>>(
        m, (std::string("QNeuron") + suffix).c_str())
        .def(py::init([](Ptr reg, std::vector<bitLenInt> inputs, bitLenInt output, int fn,
                          R alpha) {
            return std::make_shared<QNeuron<R>>(
                reg, inputs, output, (QNeuronActivationFn)fn, alpha);
        }),
            py::arg("reg"), py::arg("inputs"), py::arg("output"), py::arg("activation_fn") = 0,
            py::arg("alpha") = (R)1)
        .def("predict", &QNeuron<R>::Predict, py::arg("expected") = true,
            py::arg("reset_init") = true)
        .def("unpredict", &QNeuron<R>::Unpredict, py::arg("expected") = true)
        .def("learn", &QNeuron<R>::Learn, py::arg("eta"), py::arg("expected"),
            py::arg("reset_init") = true)
        .def("learn_permutation", &QNeuron<R>::LearnPermutation)
        .def("set_angles", &QNeuron<R>::SetAngles)
        .def("get_angles", &QNeuron<R>::GetAngles)
        .def("set_activation_fn", [](QNeuron<R>& n, int fn) {
            n.SetActivationFn((QNeuronActivationFn)fn);
        });

    // ---- serialization ----
    m.def((std::string("save_stabilizer_") + suffix).c_str(),
        [](Ptr q) { return SaveStabilizerText<R>(q); });
    m.def((std::string("load_stabilizer_") + suffix).c_str(),
        [](const std::string& s, int64_t seed) {
            RngPtr rng = (seed < 0) ? std::make_shared<Rng>() : std::make_shared<Rng>((uint64_t)seed);
            return LoadStabilizerText<R>(s, rng);
        },
        py::arg("text"), py::arg("seed") = (int64_t)-1);
    m.def((std::string("lossy_save_") + suffix).c_str(),
        [](Ptr q, const std::string& path, bitLenInt blockBits, int bits, bool rotate) {
            LossySaveState<R>(q, path, blockBits, bits, rotate);
        },
        py::arg("sim"), py::arg("path"), py::arg("block_bits") = (bitLenInt)12,
        py::arg("bits") = 16, py::arg("rotate") = true);
    m.def((std::string("lossy_load_") + suffix).c_str(),
        [](Ptr q, const std::string& path) { LossyLoadState<R>(q, path); });
}

PYBIND11_MODULE(_qrack, m)
{
    m.doc() = "qrack_amd native core (MI355X / HIP)";

    bindQInterface<float>(m, "QInterfaceF");
    bindQInterface<double>(m, "QInterfaceD");
    bindExtras<float>(m, "F");
    bindExtras<double>(m, "D");

    m.def("create", &CreateStack<float>, py::arg("qubits"), py::arg("layers") = std::vector<std::string>{ "cpu" },
        py::arg("init_perm") = (bitCapInt)0, py::arg("seed") = (int64_t)-1,
        py::arg("device_id") = (int64_t)-1, py::arg("pages_per_device") = (bitLenInt)1,
        py::arg("devices") = std::vector<int64_t>{});
    m.def("create_d", &CreateStack<double>, py::arg("qubits"),
        py::arg("layers") = std::vector<std::string>{ "cpu" }, py::arg("init_perm") = (bitCapInt)0,
        py::arg("seed") = (int64_t)-1, py::arg("device_id") = (int64_t)-1,
        py::arg("pages_per_device") = (bitLenInt)1, py::arg("devices") = std::vector<int64_t>{});

    m.def("hip_device_count", &HipDeviceCount);
#ifdef QRACK_AMD_HIP_ENGINE
    m.def("profile_report", []() {
        py::dict d;
        for (auto& kv : HipProfiler::Report()) {
            d[py::str(kv.first)] = py::make_tuple(kv.second.first, kv.second.second);
        }
        return d;
    });
    m.def("profile_reset", []() { HipProfiler::Reset(); });
#else
    m.def("profile_report", []() { return py::dict(); });
    m.def("profile_reset", []() {});
#endif
    m.attr("__version__") = "0.1.0";
}
