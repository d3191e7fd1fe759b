// Sanitizer lane (the reference has no ASAN/TSAN hooks; this
// build adds one): a C++-level battery over the layer stacks, built with
// -fsanitize=address,undefined by `make asan` and run on CPU.
#include "../csrc/qbdt.hpp"
#include "../csrc/qengine_cpu.hpp"
#include "../csrc/qengine_sparse.hpp"
#include "../csrc/qfactory.hpp"
#include "../csrc/qpager.hpp"
#include "../csrc/qstabilizerhybrid.hpp"
#include "../csrc/qunit.hpp"

#include <cstdio>

using namespace qrack_amd;

static void battery(QInterfacePtr<float> q)
{
    const bitLenInt n = q->GetQubitCount();
    q->H(0);
    for (bitLenInt i = 0; i + 1 < n; ++i) q->CNOT(i, i + 1);
    q->T(1);
    q->RY(0.3f, 2);
    q->Swap(0, n - 1);
    q->QFT(0, n);
    q->IQFT(0, n);
    (void)q->Prob(n - 1);
    auto res = q->MultiShotMeasureMask({ 1u, 2u, 4u }, 50);
    (void)q->MAll();
    q->SetPermutation(3);
    try {
        q->INC(2, 0, n - 1);
    } catch (const QrackError&) {
        // some stacks (pure BDT) have no ALU: expected capability hole
    }
    // batched 1q gates + SDRP approximate mode
    const float s2 = 0.70710678f;
    const cplx<float> hh[4] = { { s2, 0 }, { s2, 0 }, { s2, 0 }, { -s2, 0 } };
    std::vector<cplx<float>> ms;
    for (int g = 0; g < 3; ++g) ms.insert(ms.end(), hh, hh + 4);
    q->Mtrx1qBatch({ 0, 2, 4 }, ms);
    q->SetSdrp(0.3);
    q->H(1);
    q->CNOT(1, 3);
    q->RY(0.05f, 3);
    q->CZ(1, 3);
    (void)q->GetUnitaryFidelity();
    q->SetSdrp(0.0);
    // round-2 paths: T-gadget ancillae + buffered cross-unit CX + wide masks
    q->SetPermutation(0);
    q->H(0);
    q->T(0);
    q->CNOT(0, 1);
    q->H(2);
    q->CNOT(2, 3);
    q->CNOT(2, 3);
    (void)q->Prob(1);
    BigCap wide = q->MAllWide();
    (void)wide;
    q->SetPermutationWide(BigCap(5, 0));
    (void)q->MultiShotMeasureQubits({ 0, 3, 5 }, 8);
    (void)res;
}

int main()
{
    const std::vector<std::vector<std::string>> stacks = {
        { "cpu" },
        { "sparse" },
        { "bdt" },
        { "stabilizer_hybrid", "cpu" },
        { "qunit", "cpu" },
        { "qunit", "stabilizer_hybrid", "cpu" },
        { "pager", "cpu" },
        { "turboquant" },
        { "qunit", "stabilizer_hybrid", "turboquant" },
    };
    for (const auto& layers : stacks) {
        auto q = CreateStack<float>(6, layers, 0, 42, -1, 2);
        battery(q);
        std::printf("ok:");
        for (auto& l : layers) std::printf(" %s", l.c_str());
        std::printf("\n");
    }
    // exercise clone/compose/decompose lifecycles under the sanitizer
    auto a = CreateStack<float>(3, { "cpu" }, 0, 1, -1, 1);
    auto b = CreateStack<float>(2, { "cpu" }, 0, 2, -1, 1);
    a->H(0);
    b->X(0);
    a->Compose(b);
    auto c = a->Clone();
    auto dest = CreateStack<float>(2, { "cpu" }, 0, 3, -1, 1);
    c->Decompose(3, dest);
    std::printf("ok: lifecycle\n");
    return 0;
}
