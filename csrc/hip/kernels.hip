// qrack_amd — CDNA4 (gfx950) device kernels for the HIP state-vector engine.
//
// Parity target: the reference kernel inventory in
// /root/reference/src/common/qengine.cl + qheader_alu.cl,
// re-designed for MI355X rather than translated:
//  - wave64 blocks (256 threads), grid-stride loops capped at ~8 blocks/CU;
//  - fp32 amplitude streams vectorized as float4 (two complex amps per lane,
//    16 B/lane — G13 of the CDNA HIP guide); fp64 is naturally 16 B/lane;
//  - reductions: wave shuffle (64-wide) -> LDS across the block's 4 waves ->
//    one partial per block, finished on host (no global atomics);
//  - the entire ALU family is ONE uniform-opcode permutation kernel (the
//    branch is wave-uniform) instead of 26 near-identical kernels.
#include "kernels.hpp"

#include <cstdlib>

namespace qrack_amd {

#ifndef QA_BLOCK_OVERRIDE
constexpr int QA_BLOCK = 256;
#else
constexpr int QA_BLOCK = QA_BLOCK_OVERRIDE;
#endif
// Streaming gate kernels launch an EXACT grid (one iteration per thread):
// the block-count A/B on the 30q QFT is monotone all the way up —
// 2048 = 112.1 ms, 8192 = 108.0, 32768 = 104.4, exact (~0.5-1M blocks) =
// 96.0 ms. Consecutive blocks touch consecutive tiles, which keeps each
// XCD's L2 on a contiguous slice; the grid-stride remainder loop only
// handles env-capped experiments (QRACK_GPU_BLOCKS). Reductions stay
// capped at QA_REDUCE_MAX_BLOCKS to bound their partials buffers.
constexpr int QA_MAX_BLOCKS = QA_REDUCE_MAX_BLOCKS;
// the HSA dispatch packet's grid size is a 32-bit WORK-ITEM count: blocks
// * QA_BLOCK must stay below 2^32 or the launch silently wraps (caught at
// 34 qubits, where the exact column grid needs exactly 2^32 threads)
constexpr bitCapInt QA_GRID_HW_MAX = ((bitCapInt)1 << 24) - 1u;

static inline int maxBlocks()
{
    static int v = [] {
        if (const char* env = std::getenv("QRACK_GPU_BLOCKS")) return std::atoi(env);
        return 0; // 0 = exact grid
    }();
    return v;
}

static inline bool useNontemporal()
{
    static bool v = [] {
        if (const char* env = std::getenv("QRACK_GPU_NT")) return std::atoi(env) != 0;
        return false;
    }();
    return v;
}

static inline int gridFor(bitCapInt n)
{
    bitCapInt b = (n + QA_BLOCK - 1) / QA_BLOCK;
    const int cap = maxBlocks();
    if (cap > 0 && b > (bitCapInt)cap) b = cap;
    if (b > QA_GRID_HW_MAX) b = QA_GRID_HW_MAX;
    if (b < 1) b = 1;
    return (int)b;
}

// nontemporal (streaming) load/store option for the pure-stream kernels:
// state-vector gate traffic has zero reuse, so bypassing L2/LLC can help
typedef float nat_float4 __attribute__((ext_vector_type(4)));
template <bool NT> __device__ __forceinline__ float4 ld4(const float4* p)
{
    if constexpr (NT) {
        const nat_float4 v = __builtin_nontemporal_load(reinterpret_cast<const nat_float4*>(p));
        return make_float4(v.x, v.y, v.z, v.w);
    }
    return *p;
}
template <bool NT> __device__ __forceinline__ void st4(float4* p, float4 v)
{
    if constexpr (NT) {
        nat_float4 nv = { v.x, v.y, v.z, v.w };
        __builtin_nontemporal_store(nv, reinterpret_cast<nat_float4*>(p));
    } else {
        *p = v;
    }
}

// reductions/argmax/inner write one partial PER BLOCK: their grids must
// stay within the fixed partials buffers regardless of the exact-grid
// default for streaming kernels
static inline int gridForReduce(bitCapInt n)
{
    bitCapInt b = (n + QA_BLOCK - 1) / QA_BLOCK;
    if (b > (bitCapInt)QA_REDUCE_MAX_BLOCKS) b = QA_REDUCE_MAX_BLOCKS;
    if (b < 1) b = 1;
    return (int)b;
}

int reduceGridSize(bitCapInt n) { return gridForReduce(n); }

__device__ __forceinline__ bitCapInt expandBits(bitCapInt j, const bitCapInt* pows, int n)
{
    for (int k = 0; k < n; ++k) {
        j = ((j & ~(pows[k] - 1u)) << 1u) | (j & (pows[k] - 1u));
    }
    return j;
}

// ---- gate apply -------------------------------------------------------------

// KIND: 0 = generic 2x2, 1 = phase (diagonal), 2 = invert (antidiagonal)
template <typename R, int KIND> __global__ void k_apply2x2(cplx<R>* sv, GateArgs<R> a)
{
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt j = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; j < a.maxI; j += stride) {
        const bitCapInt i = expandBits(j, a.qPowers, a.nPowers);
        const bitCapInt i1 = i | a.offset1;
        const bitCapInt i2 = i | a.offset2;
        if (KIND == 1) {
            sv[i1] = a.m[0] * sv[i1];
            sv[i2] = a.m[3] * sv[i2];
        } else if (KIND == 2) {
            const cplx<R> t = sv[i1];
            sv[i1] = a.m[1] * sv[i2];
            sv[i2] = a.m[2] * t;
        } else {
            const cplx<R> x = sv[i1];
            const cplx<R> y = sv[i2];
            sv[i1] = a.m[0] * x + a.m[1] * y;
            sv[i2] = a.m[2] * x + a.m[3] * y;
        }
    }
}

// fp32 vectorized single-target path: each lane handles TWO adjacent pairs
// through float4 (16 B) loads/stores. Requires nPowers == 1 and even maxI.
template <int KIND, bool NT> __global__ void k_apply2x2_1v(cplx<float>* sv, GateArgs<float> a)
{
    float4* sv4 = reinterpret_cast<float4*>(sv);
    const bitCapInt p = a.qPowers[0];
    const bitCapInt half = a.maxI >> 1u; // float4-pair iterations
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    const cplx<float> m0 = a.m[0], m1 = a.m[1], m2 = a.m[2], m3 = a.m[3];
    if (p == 1u) {
        // target bit 0: each pair is adjacent = one float4
        for (bitCapInt k = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; k < half; k += stride) {
            // two pairs: float4s at 2k and 2k+1
            for (int h = 0; h < 2; ++h) {
                const bitCapInt idx = 2u * k + h;
                float4 v = ld4<NT>(&sv4[idx]);
                const cplx<float> x{ v.x, v.y }, y{ v.z, v.w };
                cplx<float> nx, ny;
                if (KIND == 1) {
                    nx = m0 * x;
                    ny = m3 * y;
                } else if (KIND == 2) {
                    nx = m1 * y;
                    ny = m2 * x;
                } else {
                    nx = m0 * x + m1 * y;
                    ny = m2 * x + m3 * y;
                }
                st4<NT>(&sv4[idx], make_float4(nx.re, nx.im, ny.re, ny.im));
            }
        }
    } else {
        // target bit >= 1: pairs (i, i+p); two consecutive j share the block
        for (bitCapInt k = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; k < half; k += stride) {
            const bitCapInt j = 2u * k;
            const bitCapInt i = ((j & ~(p - 1u)) << 1u) | (j & (p - 1u));
            const bitCapInt lo4 = i >> 1u;       // float4 index of (i, i+1)
            const bitCapInt hi4 = (i + p) >> 1u; // float4 index of (i+p, i+p+1)
            float4 vlo = ld4<NT>(&sv4[lo4]);
            float4 vhi = ld4<NT>(&sv4[hi4]);
            const cplx<float> x0{ vlo.x, vlo.y }, x1{ vlo.z, vlo.w };
            const cplx<float> y0{ vhi.x, vhi.y }, y1{ vhi.z, vhi.w };
            cplx<float> a0, a1, b0, b1;
            if (KIND == 1) {
                a0 = m0 * x0; a1 = m0 * x1;
                b0 = m3 * y0; b1 = m3 * y1;
            } else if (KIND == 2) {
                a0 = m1 * y0; a1 = m1 * y1;
                b0 = m2 * x0; b1 = m2 * x1;
            } else {
                a0 = m0 * x0 + m1 * y0; a1 = m0 * x1 + m1 * y1;
                b0 = m2 * x0 + m3 * y0; b1 = m2 * x1 + m3 * y1;
            }
            st4<NT>(&sv4[lo4], make_float4(a0.re, a0.im, a1.re, a1.im));
            st4<NT>(&sv4[hi4], make_float4(b0.re, b0.im, b1.re, b1.im));
        }
    }
}

// wide fp32 single-target variant: FOUR adjacent pairs (two float4 per side,
// 32 B/lane/side) — A/B candidate for the HBM-bound H stream
template <int KIND> __global__ void k_apply2x2_1w(cplx<float>* sv, GateArgs<float> a)
{
    float4* sv4 = reinterpret_cast<float4*>(sv);
    const bitCapInt p = a.qPowers[0];
    const bitCapInt quarter = a.maxI >> 2u;
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    const cplx<float> m0 = a.m[0], m1 = a.m[1], m2 = a.m[2], m3 = a.m[3];
    for (bitCapInt k = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; k < quarter; k += stride) {
        const bitCapInt j = 4u * k;
        const bitCapInt i = ((j & ~(p - 1u)) << 1u) | (j & (p - 1u));
        const bitCapInt lo4 = i >> 1u;
        const bitCapInt hi4 = (i + p) >> 1u;
#pragma unroll
        for (int h = 0; h < 2; ++h) {
            float4 vlo = sv4[lo4 + h];
            float4 vhi = sv4[hi4 + h];
            const cplx<float> x0{ vlo.x, vlo.y }, x1{ vlo.z, vlo.w };
            const cplx<float> y0{ vhi.x, vhi.y }, y1{ vhi.z, vhi.w };
            cplx<float> a0, a1, b0, b1;
            if (KIND == 1) {
                a0 = m0 * x0; a1 = m0 * x1;
                b0 = m3 * y0; b1 = m3 * y1;
            } else if (KIND == 2) {
                a0 = m1 * y0; a1 = m1 * y1;
                b0 = m2 * x0; b1 = m2 * x1;
            } else {
                a0 = m0 * x0 + m1 * y0; a1 = m0 * x1 + m1 * y1;
                b0 = m2 * x0 + m3 * y0; b1 = m2 * x1 + m3 * y1;
            }
            sv4[lo4 + h] = make_float4(a0.re, a0.im, a1.re, a1.im);
            sv4[hi4 + h] = make_float4(b0.re, b0.im, b1.re, b1.im);
        }
    }
}

static inline int wideMode()
{
    static int v = [] {
        if (const char* env = std::getenv("QRACK_GPU_WIDE")) return std::atoi(env);
        return 0;
    }();
    return v;
}

// one-sided diagonal scale: sv[i|offset] *= f for every expanded i.
// The dominant QFT kernel: CPhase(topLeft=1) touches only the
// control=1 & target=1 quarter of the state (the reference's phasesingle
// touches half; this is the MI355X-native improvement).
template <typename R> __global__ void k_scale_side(cplx<R>* sv, GateArgs<R> a, cplx<R> f)
{
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt j = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; j < a.maxI; j += stride) {
        const bitCapInt i = expandBits(j, a.qPowers, a.nPowers) | a.offset1;
        sv[i] = f * sv[i];
    }
}

// fp32 vectorized one-sided scale: two adjacent expanded indices per lane
// (requires qPowers[0] >= 2 so consecutive j stay adjacent, and even maxI)
__global__ void k_scale_side_v(cplx<float>* sv, GateArgs<float> a, cplx<float> f)
{
    float4* sv4 = reinterpret_cast<float4*>(sv);
    const bitCapInt half = a.maxI >> 1u;
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt k = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; k < half; k += stride) {
        const bitCapInt i = (expandBits(2u * k, a.qPowers, a.nPowers) | a.offset1) >> 1u;
        float4 v = sv4[i];
        const cplx<float> x{ v.x, v.y }, y{ v.z, v.w };
        const cplx<float> nx = f * x, ny = f * y;
        sv4[i] = make_float4(nx.re, nx.im, ny.re, ny.im);
    }
}

// fp32 vectorized general/phase/invert pair kernel for ANY skip set with
// qPowers[0] >= 2: two adjacent pairs per lane via float4
template <int KIND> __global__ void k_apply2x2_v(cplx<float>* sv, GateArgs<float> a)
{
    float4* sv4 = reinterpret_cast<float4*>(sv);
    const bitCapInt half = a.maxI >> 1u;
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    const cplx<float> m0 = a.m[0], m1 = a.m[1], m2 = a.m[2], m3 = a.m[3];
    for (bitCapInt k = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; k < half; k += stride) {
        const bitCapInt i = expandBits(2u * k, a.qPowers, a.nPowers);
        const bitCapInt lo4 = (i | a.offset1) >> 1u;
        const bitCapInt hi4 = (i | a.offset2) >> 1u;
        float4 vlo = sv4[lo4];
        float4 vhi = sv4[hi4];
        const cplx<float> x0{ vlo.x, vlo.y }, x1{ vlo.z, vlo.w };
        const cplx<float> y0{ vhi.x, vhi.y }, y1{ vhi.z, vhi.w };
        cplx<float> a0, a1, b0, b1;
        if (KIND == 1) {
            a0 = m0 * x0; a1 = m0 * x1;
            b0 = m3 * y0; b1 = m3 * y1;
        } else if (KIND == 2) {
            a0 = m1 * y0; a1 = m1 * y1;
            b0 = m2 * x0; b1 = m2 * x1;
        } else {
            a0 = m0 * x0 + m1 * y0; a1 = m0 * x1 + m1 * y1;
            b0 = m2 * x0 + m3 * y0; b1 = m2 * x1 + m3 * y1;
        }
        sv4[lo4] = make_float4(a0.re, a0.im, a1.re, a1.im);
        sv4[hi4] = make_float4(b0.re, b0.im, b1.re, b1.im);
    }
}

template <typename R> static bool isOne(cplx<R> c) { return c.re == (R)1 && c.im == (R)0; }

template <typename R> static int matrixKind(const cplx<R>* m)
{
    const bool isPhase = (norm(m[1]) <= 0) && (norm(m[2]) <= 0);
    const bool isInvert = (norm(m[0]) <= 0) && (norm(m[3]) <= 0);
    return isPhase ? 1 : (isInvert ? 2 : 0);
}

__global__ void k_apply2x2_1mfma(cplx<float>* sv, GateArgs<float> a);

template <typename R>
void launchApply2x2(cplx<R>* sv, const GateArgs<R>& a, hipStream_t stream)
{
    const int kind = matrixKind(a.m);
    const int grid = gridFor(a.maxI);
    if constexpr (std::is_same_v<R, float>) {
        static const bool useMfma = []() {
            if (const char* env = std::getenv("QRACK_GPU_MFMA")) return std::atoi(env) != 0;
            return false;
        }();
        if (useMfma && a.nPowers == 1) {
            hipLaunchKernelGGL((k_apply2x2_1mfma), dim3(gridFor(a.maxI >> 2u)), dim3(QA_BLOCK), 0,
                stream, sv, a);
            return;
        }
    }
    // fp32 pairs can vectorize as float4 whenever consecutive iteration
    // indices stay memory-adjacent: lowest skip power >= 2
    const bool vecOk = (a.nPowers > 0) && (a.qPowers[0] >= 2u) && ((a.maxI & 1u) == 0u);

    if (kind == 1) {
        // diagonal: skip untouched sides entirely
        const bool tl1 = isOne(a.m[0]);
        const bool br1 = isOne(a.m[3]);
        if (tl1 && br1) return; // identity
        if (tl1 || br1) {
            GateArgs<R> s = a;
            s.offset1 = tl1 ? a.offset2 : a.offset1;
            const cplx<R> f = tl1 ? a.m[3] : a.m[0];
            if constexpr (std::is_same_v<R, float>) {
                if (vecOk) {
                    hipLaunchKernelGGL((k_scale_side_v), dim3(gridFor(a.maxI >> 1u)),
                        dim3(QA_BLOCK), 0, stream, sv, s, f);
                    return;
                }
            }
            hipLaunchKernelGGL(
                (k_scale_side<R>), dim3(grid), dim3(QA_BLOCK), 0, stream, sv, s, f);
            return;
        }
    }
    if constexpr (std::is_same_v<R, float>) {
        if (a.nPowers == 1 && (a.maxI & 1u) == 0u && a.maxI >= 2u && a.offset1 == 0u) {
            // single-target fast path (handles target bit 0 too)
            if (wideMode() && a.qPowers[0] >= 4u && (a.maxI & 3u) == 0u) {
                const int gridw = gridFor(a.maxI >> 2u);
                switch (kind) {
                case 1:
                    hipLaunchKernelGGL((k_apply2x2_1w<1>), dim3(gridw), dim3(QA_BLOCK), 0, stream, sv, a);
                    return;
                case 2:
                    hipLaunchKernelGGL((k_apply2x2_1w<2>), dim3(gridw), dim3(QA_BLOCK), 0, stream, sv, a);
                    return;
                default:
                    hipLaunchKernelGGL((k_apply2x2_1w<0>), dim3(gridw), dim3(QA_BLOCK), 0, stream, sv, a);
                    return;
                }
            }
            const int gridv = gridFor(a.maxI >> 1u);
            if (useNontemporal()) {
                switch (kind) {
                case 1:
                    hipLaunchKernelGGL((k_apply2x2_1v<1, true>), dim3(gridv), dim3(QA_BLOCK), 0, stream, sv, a);
                    return;
                case 2:
                    hipLaunchKernelGGL((k_apply2x2_1v<2, true>), dim3(gridv), dim3(QA_BLOCK), 0, stream, sv, a);
                    return;
                default:
                    hipLaunchKernelGGL((k_apply2x2_1v<0, true>), dim3(gridv), dim3(QA_BLOCK), 0, stream, sv, a);
                    return;
                }
            }
            switch (kind) {
            case 1:
                hipLaunchKernelGGL((k_apply2x2_1v<1, false>), dim3(gridv), dim3(QA_BLOCK), 0, stream, sv, a);
                return;
            case 2:
                hipLaunchKernelGGL((k_apply2x2_1v<2, false>), dim3(gridv), dim3(QA_BLOCK), 0, stream, sv, a);
                return;
            default:
                hipLaunchKernelGGL((k_apply2x2_1v<0, false>), dim3(gridv), dim3(QA_BLOCK), 0, stream, sv, a);
                return;
            }
        }
        if (vecOk) {
            const int gridv = gridFor(a.maxI >> 1u);
            switch (kind) {
            case 1:
                hipLaunchKernelGGL((k_apply2x2_v<1>), dim3(gridv), dim3(QA_BLOCK), 0, stream, sv, a);
                return;
            case 2:
                hipLaunchKernelGGL((k_apply2x2_v<2>), dim3(gridv), dim3(QA_BLOCK), 0, stream, sv, a);
                return;
            default:
                hipLaunchKernelGGL((k_apply2x2_v<0>), dim3(gridv), dim3(QA_BLOCK), 0, stream, sv, a);
                return;
            }
        }
    }
    switch (kind) {
    case 1:
        hipLaunchKernelGGL((k_apply2x2<R, 1>), dim3(grid), dim3(QA_BLOCK), 0, stream, sv, a);
        break;
    case 2:
        hipLaunchKernelGGL((k_apply2x2<R, 2>), dim3(grid), dim3(QA_BLOCK), 0, stream, sv, a);
        break;
    default:
        hipLaunchKernelGGL((k_apply2x2<R, 0>), dim3(grid), dim3(QA_BLOCK), 0, stream, sv, a);
        break;
    }
}

// ---- multiplexer ------------------------------------------------------------

template <typename R>
__global__ void k_uniformly_ctrl(cplx<R>* sv, bitCapInt maxI, bitCapInt targetPower,
    const bitCapInt* ctrlPowers, int nCtrls, const cplx<R>* mtrxs)
{
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt j = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; j < maxI; j += stride) {
        const bitCapInt i = ((j & ~(targetPower - 1u)) << 1u) | (j & (targetPower - 1u));
        bitCapInt sel = 0;
        for (int b = 0; b < nCtrls; ++b) {
            if (i & ctrlPowers[b]) sel |= (ONE_BCI << b);
        }
        const cplx<R>* m = mtrxs + 4u * sel;
        const cplx<R> x = sv[i];
        const cplx<R> y = sv[i | targetPower];
        sv[i] = m[0] * x + m[1] * y;
        sv[i | targetPower] = m[2] * x + m[3] * y;
    }
}

template <typename R>
void launchUniformlyControlled(cplx<R>* sv, bitCapInt maxI, bitCapInt targetPower,
    const bitCapInt* ctrlPowersDev, int nCtrls, const cplx<R>* mtrxsDev, hipStream_t stream)
{
    hipLaunchKernelGGL((k_uniformly_ctrl<R>), dim3(gridFor(maxI)), dim3(QA_BLOCK), 0, stream, sv,
        maxI, targetPower, ctrlPowersDev, nCtrls, mtrxsDev);
}

// ---- mask gates -------------------------------------------------------------

template <typename R> __global__ void k_xmask(cplx<R>* sv, bitCapInt maxQPower, bitCapInt mask)
{
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt i = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; i < maxQPower; i += stride) {
        const bitCapInt j = i ^ mask;
        if (i < j) {
            const cplx<R> t = sv[i];
            sv[i] = sv[j];
            sv[j] = t;
        }
    }
}

template <typename R>
void launchXMask(cplx<R>* sv, bitCapInt maxQPower, bitCapInt mask, hipStream_t stream)
{
    hipLaunchKernelGGL(
        (k_xmask<R>), dim3(gridFor(maxQPower)), dim3(QA_BLOCK), 0, stream, sv, maxQPower, mask);
}

template <typename R>
__global__ void k_parityphase(cplx<R>* sv, bitCapInt maxQPower, bitCapInt mask, cplx<R> even, cplx<R> odd)
{
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt i = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; i < maxQPower; i += stride) {
        sv[i] = (__popcll(i & mask) & 1 ? odd : even) * sv[i];
    }
}

template <typename R>
void launchParityPhase(
    cplx<R>* sv, bitCapInt maxQPower, bitCapInt mask, cplx<R> even, cplx<R> odd, hipStream_t stream)
{
    hipLaunchKernelGGL((k_parityphase<R>), dim3(gridFor(maxQPower)), dim3(QA_BLOCK), 0, stream, sv,
        maxQPower, mask, even, odd);
}

// ---- reductions --------------------------------------------------------------

__device__ __forceinline__ double waveReduceSum(double v)
{
    for (int off = 32; off; off >>= 1) v += __shfl_down(v, off);
    return v;
}

template <typename R> __global__ void k_reduce(const cplx<R>* sv, ReduceArgs a, int op, double* partials)
{
    double s = 0;
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt i = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; i < a.maxI; i += stride) {
        const double n = (double)norm(sv[i]);
        switch ((ReduceOp)op) {
        case ReduceOp::NORM_ALL:
            s += n;
            break;
        case ReduceOp::PROB_BITSET:
            if ((i & a.mask) == a.mask) s += n;
            break;
        case ReduceOp::PROB_MASK:
            if ((i & a.mask) == a.perm) s += n;
            break;
        case ReduceOp::PROB_PARITY:
            if (__popcll(i & a.mask) & 1) s += n;
            break;
        case ReduceOp::EXP_PERM:
        case ReduceOp::EXP_PERM_SQ: {
            double val = a.offset;
            for (int b = 0; b < a.nBits; ++b) {
                if ((i >> a.bitsArr[b]) & 1u) val += (double)a.permsArr[b];
            }
            s += ((ReduceOp)op == ReduceOp::EXP_PERM) ? val * n : val * val * n;
            break;
        }
        case ReduceOp::NORM_FLOOR:
            if (n >= a.normThresh) s += n;
            break;
        }
    }
    s = waveReduceSum(s);
    __shared__ double waveSums[QA_BLOCK / 64];
    const int lane = threadIdx.x & 63;
    const int wid = threadIdx.x >> 6;
    if (lane == 0) waveSums[wid] = s;
    __syncthreads();
    if (wid == 0) {
        double t = (lane < QA_BLOCK / 64) ? waveSums[lane] : 0.0;
        t = waveReduceSum(t);
        if (lane == 0) partials[blockIdx.x] = t;
    }
}

template <typename R>
int launchReduce(const cplx<R>* sv, const ReduceArgs& a, int op, double* partialsDev,
    hipStream_t stream)
{
    const int grid = gridForReduce(a.maxI);
    hipLaunchKernelGGL(
        (k_reduce<R>), dim3(grid), dim3(QA_BLOCK), 0, stream, sv, a, op, partialsDev);
    return grid;
}

template <typename R>
__global__ void k_argmax(const cplx<R>* sv, bitCapInt maxI, double* vals, bitCapInt* idxs)
{
    double best = -1.0;
    bitCapInt bestI = 0;
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt i = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; i < maxI; i += stride) {
        const double n = (double)norm(sv[i]);
        if (n > best) {
            best = n;
            bestI = i;
        }
    }
    for (int off = 32; off; off >>= 1) {
        const double ov = __shfl_down(best, off);
        const bitCapInt oi = (bitCapInt)__shfl_down((unsigned long long)bestI, off);
        if (ov > best) {
            best = ov;
            bestI = oi;
        }
    }
    __shared__ double wv[QA_BLOCK / 64];
    __shared__ bitCapInt wi[QA_BLOCK / 64];
    const int lane = threadIdx.x & 63;
    const int wid = threadIdx.x >> 6;
    if (lane == 0) {
        wv[wid] = best;
        wi[wid] = bestI;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
        for (int k = 1; k < QA_BLOCK / 64; ++k) {
            if (wv[k] > wv[0]) {
                wv[0] = wv[k];
                wi[0] = wi[k];
            }
        }
        vals[blockIdx.x] = wv[0];
        idxs[blockIdx.x] = wi[0];
    }
}

template <typename R>
int launchArgMax(const cplx<R>* sv, bitCapInt maxI, double* valsDev, bitCapInt* idxDev,
    hipStream_t stream)
{
    const int grid = gridForReduce(maxI);
    hipLaunchKernelGGL(
        (k_argmax<R>), dim3(grid), dim3(QA_BLOCK), 0, stream, sv, maxI, valsDev, idxDev);
    return grid;
}

// ---- normalize / projection --------------------------------------------------

template <typename R>
__global__ void k_normalize(cplx<R>* sv, bitCapInt maxQPower, cplx<R> factor, R normThresh)
{
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt i = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; i < maxQPower; i += stride) {
        const cplx<R> v = sv[i];
        sv[i] = (norm(v) < normThresh) ? cplx<R>(0, 0) : factor * v;
    }
}

template <typename R>
void launchNormalize(cplx<R>* sv, bitCapInt maxQPower, cplx<R> factor, R normThresh, hipStream_t stream)
{
    hipLaunchKernelGGL((k_normalize<R>), dim3(gridFor(maxQPower)), dim3(QA_BLOCK), 0, stream, sv,
        maxQPower, factor, normThresh);
}

template <typename R>
__global__ void k_applym(cplx<R>* sv, bitCapInt maxQPower, bitCapInt mask, bitCapInt result, cplx<R> nrm)
{
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt i = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; i < maxQPower; i += stride) {
        sv[i] = ((i & mask) == result) ? nrm * sv[i] : cplx<R>(0, 0);
    }
}

template <typename R>
void launchApplyM(cplx<R>* sv, bitCapInt maxQPower, bitCapInt mask, bitCapInt result, cplx<R> nrm,
    hipStream_t stream)
{
    hipLaunchKernelGGL((k_applym<R>), dim3(gridFor(maxQPower)), dim3(QA_BLOCK), 0, stream, sv,
        maxQPower, mask, result, nrm);
}

template <typename R>
__global__ void k_applyparity(cplx<R>* sv, bitCapInt maxQPower, bitCapInt mask, int odd, cplx<R> nrm)
{
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt i = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; i < maxQPower; i += stride) {
        sv[i] = ((__popcll(i & mask) & 1) == odd) ? nrm * sv[i] : cplx<R>(0, 0);
    }
}

template <typename R>
void launchApplyParity(cplx<R>* sv, bitCapInt maxQPower, bitCapInt mask, bool odd, cplx<R> nrm,
    hipStream_t stream)
{
    hipLaunchKernelGGL((k_applyparity<R>), dim3(gridFor(maxQPower)), dim3(QA_BLOCK), 0, stream, sv,
        maxQPower, mask, odd ? 1 : 0, nrm);
}

// ---- structural ---------------------------------------------------------------

template <typename R>
__global__ void k_compose(const cplx<R>* a, const cplx<R>* b, cplx<R>* out, bitCapInt nMaxQPower,
    bitLenInt start, bitLenInt oQubits)
{
    const bitCapInt lowMask = (ONE_BCI << start) - 1u;
    const bitCapInt midMask = (ONE_BCI << oQubits) - 1u;
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt i = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; i < nMaxQPower; i += stride) {
        const bitCapInt low = i & lowMask;
        const bitCapInt mid = (i >> start) & midMask;
        const bitCapInt high = i >> (start + oQubits);
        out[i] = a[low | (high << start)] * b[mid];
    }
}

template <typename R>
void launchCompose(const cplx<R>* a, const cplx<R>* b, cplx<R>* out, bitCapInt nMaxQPower,
    bitLenInt start, bitLenInt oQubits, hipStream_t stream)
{
    hipLaunchKernelGGL((k_compose<R>), dim3(gridFor(nMaxQPower)), dim3(QA_BLOCK), 0, stream, a, b,
        out, nMaxQPower, start, oQubits);
}

template <typename R>
__global__ void k_dispose_slice(const cplx<R>* in, cplx<R>* out, bitCapInt remPower, bitLenInt start,
    bitLenInt length, bitCapInt slicePerm, cplx<R> scale)
{
    const bitCapInt lowMask = (ONE_BCI << start) - 1u;
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt r = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; r < remPower; r += stride) {
        const bitCapInt low = r & lowMask;
        const bitCapInt high = (r >> start) << (start + length);
        out[r] = scale * in[low | (slicePerm << start) | high];
    }
}

template <typename R>
void launchDisposeSlice(const cplx<R>* in, cplx<R>* out, bitCapInt remPower, bitLenInt start,
    bitLenInt length, bitCapInt slicePerm, cplx<R> scale, hipStream_t stream)
{
    hipLaunchKernelGGL((k_dispose_slice<R>), dim3(gridFor(remPower)), dim3(QA_BLOCK), 0, stream, in,
        out, remPower, start, length, slicePerm, scale);
}

template <typename R>
__global__ void k_gather_part(const cplx<R>* in, cplx<R>* dest, bitCapInt partPower, bitLenInt start,
    bitLenInt length, bitCapInt remIndex, cplx<R> scale)
{
    const bitCapInt lowMask = (ONE_BCI << start) - 1u;
    const bitCapInt low = remIndex & lowMask;
    const bitCapInt high = (remIndex >> start) << (start + length);
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt p = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; p < partPower; p += stride) {
        dest[p] = scale * in[low | (p << start) | high];
    }
}

template <typename R>
void launchGatherPart(const cplx<R>* in, cplx<R>* dest, bitCapInt partPower, bitLenInt start,
    bitLenInt length, bitCapInt remIndex, cplx<R> scale, hipStream_t stream)
{
    hipLaunchKernelGGL((k_gather_part<R>), dim3(gridFor(partPower)), dim3(QA_BLOCK), 0, stream, in,
        dest, partPower, start, length, remIndex, scale);
}

template <typename R>
__global__ void k_allocate_expand(const cplx<R>* in, cplx<R>* out, bitCapInt nMaxQPower,
    bitLenInt start, bitLenInt length)
{
    const bitCapInt lowMask = (ONE_BCI << start) - 1u;
    const bitCapInt midMask = (ONE_BCI << length) - 1u;
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt i = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; i < nMaxQPower; i += stride) {
        const bitCapInt low = i & lowMask;
        const bitCapInt mid = (i >> start) & midMask;
        const bitCapInt high = i >> (start + length);
        out[i] = mid ? cplx<R>(0, 0) : in[low | (high << start)];
    }
}

template <typename R>
void launchAllocateExpand(const cplx<R>* in, cplx<R>* out, bitCapInt nMaxQPower, bitLenInt start,
    bitLenInt length, hipStream_t stream)
{
    hipLaunchKernelGGL((k_allocate_expand<R>), dim3(gridFor(nMaxQPower)), dim3(QA_BLOCK), 0, stream,
        in, out, nMaxQPower, start, length);
}

template <typename R> __global__ void k_shuffle_swap(cplx<R>* aHigh, cplx<R>* bLow, bitCapInt half)
{
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt i = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; i < half; i += stride) {
        const cplx<R> t = aHigh[i];
        aHigh[i] = bLow[i];
        bLow[i] = t;
    }
}

template <typename R>
void launchShuffleSwap(cplx<R>* aHigh, cplx<R>* bLow, bitCapInt half, hipStream_t stream)
{
    hipLaunchKernelGGL(
        (k_shuffle_swap<R>), dim3(gridFor(half)), dim3(QA_BLOCK), 0, stream, aHigh, bLow, half);
}

// ---- ALU / permutation ---------------------------------------------------------

__device__ __forceinline__ bitCapInt devModMul(bitCapInt a, bitCapInt b, bitCapInt m)
{
    return (bitCapInt)(((__uint128_t)a * b) % m);
}

__device__ __forceinline__ bitCapInt devModPow(bitCapInt base, bitCapInt e, bitCapInt m)
{
    bitCapInt result = 1u % m;
    base %= m;
    while (e) {
        if (e & 1u) result = devModMul(result, base, m);
        base = devModMul(base, base, m);
        e >>= 1u;
    }
    return result;
}

__device__ __forceinline__ bitCapInt tableRead(
    const unsigned char* table, bitCapInt entry, int bytes)
{
    bitCapInt v = 0;
    for (int b = 0; b < bytes; ++b) v |= ((bitCapInt)table[entry * bytes + b]) << (8 * b);
    return v;
}

template <typename R> __global__ void k_permute(const cplx<R>* sv, cplx<R>* nsv, PermArgs a)
{
    const bitCapInt lenMask = (ONE_BCI << a.length) - 1u;
    const bitCapInt regMask = lenMask << a.start;
    const bitCapInt lenPower = ONE_BCI << a.length;
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt j = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; j < a.maxI; j += stride) {
        bitCapInt i = expandBits(j, a.qPowers, a.nPowers) | a.controlMask;
        const bitCapInt reg = (i & regMask) >> a.start;
        switch ((PermOp)a.op) {
        case PermOp::INC: {
            nsv[(i & ~regMask) | (((reg + a.operand) & lenMask) << a.start)] = sv[i];
            break;
        }
        case PermOp::INCDECC: {
            const bitCapInt out = reg + a.operand;
            const bitCapInt res = (out < lenPower)
                ? ((i & ~regMask) | (out << a.start))
                : ((i & ~regMask) | ((out - lenPower) << a.start) | a.carryMask);
            nsv[res] = sv[i];
            break;
        }
        case PermOp::INCS: {
            const bitCapInt out = (reg + a.operand) & lenMask;
            const bitCapInt signBit = ONE_BCI << (a.length - 1u);
            const bool ovf = (~(reg ^ a.operand) & (reg ^ out) & signBit) != 0;
            bitCapInt res = (i & ~regMask) | (out << a.start);
            if (ovf) res ^= a.carryMask;
            nsv[res] = sv[i];
            break;
        }
        case PermOp::MUL:
        case PermOp::DIV: {
            const bitCapInt carryRegMask = lenMask << a.start2;
            const bitCapInt out = reg * a.operand;
            const bitCapInt mapped = (i & ~(regMask | carryRegMask)) |
                ((out & lenMask) << a.start) | (((out >> a.length) & lenMask) << a.start2);
            if ((PermOp)a.op == PermOp::MUL) {
                nsv[mapped] = sv[i];
            } else {
                nsv[i] = sv[mapped];
            }
            break;
        }
        case PermOp::MULMODN:
        case PermOp::IMULMODN: {
            const bitCapInt outMask = ((ONE_BCI << a.length2) - 1u) << a.start2;
            const bitCapInt out = devModMul(reg, a.operand, a.modN);
            const bitCapInt mapped = (i & ~outMask) | (out << a.start2);
            if ((PermOp)a.op == PermOp::MULMODN) {
                nsv[mapped] = sv[i];
            } else {
                nsv[i] = sv[mapped];
            }
            break;
        }
        case PermOp::POWMODN: {
            const bitCapInt outMask = ((ONE_BCI << a.length2) - 1u) << a.start2;
            const bitCapInt out = devModPow(a.operand, reg, a.modN);
            nsv[(i & ~outMask) | (out << a.start2)] = sv[i];
            break;
        }
        case PermOp::HASH: {
            const bitCapInt val = tableRead(a.table, reg, a.tableBytes) & lenMask;
            nsv[(i & ~regMask) | (val << a.start)] = sv[i];
            break;
        }
        case PermOp::LDA: {
            // a.start/a.length = index register; a.start2/length2 = value register
            const bitCapInt idx = reg;
            const bitCapInt valMask = ((ONE_BCI << a.length2) - 1u) << a.start2;
            const bitCapInt val =
                tableRead(a.table, idx, a.tableBytes) & ((ONE_BCI << a.length2) - 1u);
            nsv[(i & ~valMask) | (val << a.start2)] = sv[i];
            break;
        }
        case PermOp::ADC:
        case PermOp::SBC: {
            const bitCapInt idx = reg;
            const bitCapInt valPower = ONE_BCI << a.length2;
            const bitCapInt valMask = (valPower - 1u) << a.start2;
            const bitCapInt tval = tableRead(a.table, idx, a.tableBytes) & (valPower - 1u);
            const bitCapInt val = (i & valMask) >> a.start2;
            bitCapInt out;
            if ((PermOp)a.op == PermOp::ADC) {
                out = val + tval + a.extra;
            } else {
                out = val + valPower - tval + a.extra;
            }
            const bitCapInt wrapped = out & (valPower - 1u);
            bitCapInt res = (i & ~valMask) | (wrapped << a.start2);
            if (out >= valPower) res |= a.carryMask;
            nsv[res] = sv[i];
            break;
        }
        case PermOp::INCBCD: {
            const int digits = (int)(a.length / 4u);
            bitCapInt value = 0, mul = 1, tenPow = 1;
            bool valid = true;
            for (int d = 0; d < digits; ++d) {
                const bitCapInt digit = (reg >> (4 * d)) & 0xFu;
                if (digit > 9u) valid = false;
                value += digit * mul;
                mul *= 10u;
                tenPow *= 10u;
            }
            if (!valid) {
                nsv[i] = sv[i];
                break;
            }
            bitCapInt out = (value + a.operand) % tenPow;
            bitCapInt enc = 0;
            for (int d = 0; d < digits; ++d) {
                enc |= (out % 10u) << (4 * d);
                out /= 10u;
            }
            nsv[(i & ~regMask) | (enc << a.start)] = sv[i];
            break;
        }
        case PermOp::ROL: {
            const bitLenInt shift = (bitLenInt)a.operand;
            const bitCapInt nreg = ((reg << shift) | (reg >> (a.length - shift))) & lenMask;
            nsv[(i & ~regMask) | (nreg << a.start)] = sv[i];
            break;
        }
        }
    }
}

template <typename R>
void launchPermute(const cplx<R>* sv, cplx<R>* nsv, const PermArgs& a, hipStream_t stream)
{
    hipLaunchKernelGGL((k_permute<R>), dim3(gridFor(a.maxI)), dim3(QA_BLOCK), 0, stream, sv, nsv, a);
}

template <typename R>
__global__ void k_phaseflipless(cplx<R>* sv, bitCapInt maxQPower, bitCapInt greaterPerm,
    bitLenInt start, bitCapInt regMask, bitCapInt flagMask)
{
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt i = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; i < maxQPower; i += stride) {
        if ((!flagMask || (i & flagMask)) && (((i & regMask) >> start) < greaterPerm)) {
            sv[i] = cplx<R>(-1, 0) * sv[i];
        }
    }
}

template <typename R>
void launchPhaseFlipIfLess(cplx<R>* sv, bitCapInt maxQPower, bitCapInt greaterPerm, bitLenInt start,
    bitCapInt regMask, bitCapInt flagMask, hipStream_t stream)
{
    hipLaunchKernelGGL((k_phaseflipless<R>), dim3(gridFor(maxQPower)), dim3(QA_BLOCK), 0, stream, sv,
        maxQPower, greaterPerm, start, regMask, flagMask);
}

// ---- fused QFT phase ramp ------------------------------------------------------

__device__ __forceinline__ bitCapInt insertZeroBitDev(bitCapInt i, bitCapInt p)
{
    return ((i & ~(p - 1u)) << 1u) | (i & (p - 1u));
}

template <typename R> __device__ __forceinline__ void devSinCos(R t, R* s, R* c);
template <> __device__ __forceinline__ void devSinCos<float>(float t, float* s, float* c)
{
    __sincosf(t, s, c);
}
template <> __device__ __forceinline__ void devSinCos<double>(double t, double* s, double* c)
{
    sincos(t, s, c);
}

template <typename R>
__global__ void k_phase_ramp(
    cplx<R>* sv, bitCapInt maxI, bitCapInt condPow, bitLenInt rampStart, bitCapInt rampMask, R scale)
{
    // condPow != 0: iterate the condPow-set half (maxI = maxQPower/2 with the
    // bit inserted); condPow == 0: iterate everything (maxI = maxQPower)
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt j = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; j < maxI; j += stride) {
        bitCapInt i = j;
        if (condPow) i = (((j & ~(condPow - 1u)) << 1u) | (j & (condPow - 1u))) | condPow;
        const bitCapInt frac = (i >> rampStart) & rampMask;
        const R theta = scale * (R)frac;
        R s, c;
        devSinCos<R>(theta, &s, &c);
        sv[i] = cplx<R>{ c, s } * sv[i];
    }
}

// fp32 vectorized: two adjacent indices per lane via float4
__global__ void k_phase_ramp_v(cplx<float>* sv, bitCapInt maxI, bitCapInt condPow,
    bitLenInt rampStart, bitCapInt rampMask, float scale)
{
    float4* sv4 = reinterpret_cast<float4*>(sv);
    const bitCapInt half = maxI >> 1u;
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt k = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; k < half; k += stride) {
        const bitCapInt j = 2u * k;
        bitCapInt i = j;
        if (condPow) i = (((j & ~(condPow - 1u)) << 1u) | (j & (condPow - 1u))) | condPow;
        const bitCapInt i4 = i >> 1u;
        float4 v = sv4[i4];
        const bitCapInt f0 = (i >> rampStart) & rampMask;
        const bitCapInt f1 = ((i + 1u) >> rampStart) & rampMask;
        float s0, c0, s1, c1;
        __sincosf(scale * (float)f0, &s0, &c0);
        __sincosf(scale * (float)f1, &s1, &c1);
        const cplx<float> a{ v.x, v.y }, b{ v.z, v.w };
        const cplx<float> na = cplx<float>{ c0, s0 } * a;
        const cplx<float> nb = cplx<float>{ c1, s1 } * b;
        sv4[i4] = make_float4(na.re, na.im, nb.re, nb.im);
    }
}

template <typename R>
void launchPhaseRamp(cplx<R>* sv, bitCapInt maxQPower, bitLenInt rampStart, bitLenInt rampBits,
    bitCapInt condPower, double scale, hipStream_t stream)
{
    const bitCapInt maxI = condPower ? (maxQPower >> 1u) : maxQPower;
    const bitCapInt rampMask = (ONE_BCI << rampBits) - 1u;
    if constexpr (std::is_same_v<R, float>) {
        if ((condPower == 0u || condPower >= 2u) && (maxI & 1u) == 0u && maxI >= 2u) {
            hipLaunchKernelGGL((k_phase_ramp_v), dim3(gridFor(maxI >> 1u)), dim3(QA_BLOCK), 0,
                stream, sv, maxI, condPower, rampStart, rampMask, (float)scale);
            return;
        }
    }
    hipLaunchKernelGGL((k_phase_ramp<R>), dim3(gridFor(maxI)), dim3(QA_BLOCK), 0, stream, sv,
        maxI, condPower, rampStart, rampMask, (R)scale);
}

// fully fused QFT column: H on bit t AND the column's phase ramp in ONE
// pass. POST (QFT): out1 *= e^{i*scale*frac} after H; PRE (IQFT): in1 *=
// e^{i*scale*frac} before H. frac depends only on the pair base's low bits,
// so both elements of a float4 share their own frac values.
template <typename R, bool PRE>
__global__ void k_qft_col(
    cplx<R>* sv, bitCapInt maxI, bitCapInt tPow, bitLenInt rampStart, bitCapInt rampMask, R scale)
{
    const R s = (R)0.70710678118654752440;
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt j = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; j < maxI; j += stride) {
        const bitCapInt i = ((j & ~(tPow - 1u)) << 1u) | (j & (tPow - 1u));
        const bitCapInt frac = (i >> rampStart) & rampMask;
        R sn, cs;
        devSinCos<R>(scale * (R)frac, &sn, &cs);
        const cplx<R> f{ cs, sn };
        cplx<R> x = sv[i];
        cplx<R> y = sv[i | tPow];
        if (PRE) y = f * y;
        cplx<R> o0 = s * (x + y);
        cplx<R> o1 = s * (x - y);
        if (!PRE) o1 = f * o1;
        sv[i] = o0;
        sv[i | tPow] = o1;
    }
}

// fp32 vectorized: two adjacent pairs per lane (requires tPow >= 2)
template <bool PRE>
__global__ void k_qft_col_v(
    cplx<float>* sv, bitCapInt maxI, bitCapInt tPow, bitLenInt rampStart, bitCapInt rampMask,
    float scale)
{
    float4* sv4 = reinterpret_cast<float4*>(sv);
    const float s = 0.70710678f;
    const bitCapInt half = maxI >> 1u;
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt k = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; k < half; k += stride) {
        const bitCapInt j = 2u * k;
        const bitCapInt i = ((j & ~(tPow - 1u)) << 1u) | (j & (tPow - 1u));
        const bitCapInt lo4 = i >> 1u;
        const bitCapInt hi4 = (i | tPow) >> 1u;
        float4 vlo = sv4[lo4];
        float4 vhi = sv4[hi4];
        float s0, c0, s1, c1;
        __sincosf(scale * (float)((i >> rampStart) & rampMask), &s0, &c0);
        __sincosf(scale * (float)(((i + 1u) >> rampStart) & rampMask), &s1, &c1);
        const cplx<float> f0{ c0, s0 }, f1{ c1, s1 };
        cplx<float> x0{ vlo.x, vlo.y }, x1{ vlo.z, vlo.w };
        cplx<float> y0{ vhi.x, vhi.y }, y1{ vhi.z, vhi.w };
        if (PRE) {
            y0 = f0 * y0;
            y1 = f1 * y1;
        }
        cplx<float> a0 = s * (x0 + y0), a1 = s * (x1 + y1);
        cplx<float> b0 = s * (x0 - y0), b1 = s * (x1 - y1);
        if (!PRE) {
            b0 = f0 * b0;
            b1 = f1 * b1;
        }
        sv4[lo4] = make_float4(a0.re, a0.im, a1.re, a1.im);
        sv4[hi4] = make_float4(b0.re, b0.im, b1.re, b1.im);
    }
}

// generalized fused column: the ramp's bits may be relocated (distributed
// pager lazy qubit maps) — frac(i) = ((i >> rampStart) & inPlaceRelMask) +
// sum_k (i & sPow[k] ? sWeight[k] : 0), plus a constant phase0 (the
// meta-page scalar term folded into the same pass).
template <typename R, bool PRE>
__global__ void k_qft_col_gen(cplx<R>* sv, bitCapInt maxI, bitCapInt tPow, RampArgs a, R phase0)
{
    const R s = (R)0.70710678118654752440;
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt j = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; j < maxI; j += stride) {
        const bitCapInt i = ((j & ~(tPow - 1u)) << 1u) | (j & (tPow - 1u));
        uint64_t frac = (uint64_t)((i >> a.rampStart) & a.inPlaceRelMask);
        for (int k = 0; k < a.nScattered; ++k) {
            if (i & a.sPow[k]) frac += a.sWeight[k];
        }
        R sn, cs;
        devSinCos<R>((R)a.scale * (R)frac + phase0, &sn, &cs);
        const cplx<R> f{ cs, sn };
        cplx<R> x = sv[i];
        cplx<R> y = sv[i | tPow];
        if (PRE) y = f * y;
        cplx<R> o0 = s * (x + y);
        cplx<R> o1 = s * (x - y);
        if (!PRE) o1 = f * o1;
        sv[i] = o0;
        sv[i | tPow] = o1;
    }
}

template <bool PRE>
__global__ void k_qft_col_gen_v(
    cplx<float>* sv, bitCapInt maxI, bitCapInt tPow, RampArgs a, float phase0)
{
    float4* sv4 = reinterpret_cast<float4*>(sv);
    const float s = 0.70710678f;
    const bitCapInt half = maxI >> 1u;
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt k = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; k < half; k += stride) {
        const bitCapInt j = 2u * k;
        const bitCapInt i = ((j & ~(tPow - 1u)) << 1u) | (j & (tPow - 1u));
        const bitCapInt lo4 = i >> 1u;
        const bitCapInt hi4 = (i | tPow) >> 1u;
        float4 vlo = sv4[lo4];
        float4 vhi = sv4[hi4];
        uint64_t fr0 = (uint64_t)((i >> a.rampStart) & a.inPlaceRelMask);
        uint64_t fr1 = (uint64_t)(((i + 1u) >> a.rampStart) & a.inPlaceRelMask);
        for (int t = 0; t < a.nScattered; ++t) {
            if (i & a.sPow[t]) fr0 += a.sWeight[t];
            if ((i + 1u) & a.sPow[t]) fr1 += a.sWeight[t];
        }
        float s0, c0, s1, c1;
        __sincosf((float)a.scale * (float)fr0 + phase0, &s0, &c0);
        __sincosf((float)a.scale * (float)fr1 + phase0, &s1, &c1);
        const cplx<float> f0{ c0, s0 }, f1{ c1, s1 };
        cplx<float> x0{ vlo.x, vlo.y }, x1{ vlo.z, vlo.w };
        cplx<float> y0{ vhi.x, vhi.y }, y1{ vhi.z, vhi.w };
        if (PRE) {
            y0 = f0 * y0;
            y1 = f1 * y1;
        }
        cplx<float> a0 = s * (x0 + y0), a1 = s * (x1 + y1);
        cplx<float> b0 = s * (x0 - y0), b1 = s * (x1 - y1);
        if (!PRE) {
            b0 = f0 * b0;
            b1 = f1 * b1;
        }
        sv4[lo4] = make_float4(a0.re, a0.im, a1.re, a1.im);
        sv4[hi4] = make_float4(b0.re, b0.im, b1.re, b1.im);
    }
}

// A/B experiment (QRACK_GPU_QFT_PIPE=1): software-pipelined column — loads
// for iteration i+1 issue before iteration i's stores, hiding the RMW
// turnaround on the two streams. Same math as k_qft_col_v<false>.
template <bool PRE>
__global__ void k_qft_col_v_pipe(
    cplx<float>* sv, bitCapInt maxI, bitCapInt tPow, bitLenInt rampStart, bitCapInt rampMask,
    float scale)
{
    float4* sv4 = reinterpret_cast<float4*>(sv);
    const float s = 0.70710678f;
    const bitCapInt half = maxI >> 1u;
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    bitCapInt kIt = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x;
    if (kIt >= half) return;
    auto idx = [&](bitCapInt kk, bitCapInt& lo4, bitCapInt& hi4, bitCapInt& i) {
        const bitCapInt j = 2u * kk;
        i = ((j & ~(tPow - 1u)) << 1u) | (j & (tPow - 1u));
        lo4 = i >> 1u;
        hi4 = (i | tPow) >> 1u;
    };
    bitCapInt lo4, hi4, i;
    idx(kIt, lo4, hi4, i);
    float4 vlo = sv4[lo4];
    float4 vhi = sv4[hi4];
    while (true) {
        const bitCapInt kNext = kIt + stride;
        bitCapInt nlo4 = 0, nhi4 = 0, ni = 0;
        float4 nlo{}, nhi{};
        const bool more = kNext < half;
        if (more) {
            idx(kNext, nlo4, nhi4, ni);
            nlo = sv4[nlo4]; // prefetch next pair before this one's stores
            nhi = sv4[nhi4];
        }
        float s0, c0, s1, c1;
        __sincosf(scale * (float)((i >> rampStart) & rampMask), &s0, &c0);
        __sincosf(scale * (float)(((i + 1u) >> rampStart) & rampMask), &s1, &c1);
        const cplx<float> f0{ c0, s0 }, f1{ c1, s1 };
        cplx<float> x0{ vlo.x, vlo.y }, x1{ vlo.z, vlo.w };
        cplx<float> y0{ vhi.x, vhi.y }, y1{ vhi.z, vhi.w };
        if (PRE) {
            y0 = f0 * y0;
            y1 = f1 * y1;
        }
        cplx<float> a0 = s * (x0 + y0), a1 = s * (x1 + y1);
        cplx<float> b0 = s * (x0 - y0), b1 = s * (x1 - y1);
        if (!PRE) {
            b0 = f0 * b0;
            b1 = f1 * b1;
        }
        sv4[lo4] = make_float4(a0.re, a0.im, a1.re, a1.im);
        sv4[hi4] = make_float4(b0.re, b0.im, b1.re, b1.im);
        if (!more) break;
        kIt = kNext;
        lo4 = nlo4;
        hi4 = nhi4;
        i = ni;
        vlo = nlo;
        vhi = nhi;
    }
}

template <typename R>
void launchQftColumnGeneral(cplx<R>* sv, bitCapInt maxQPower, bitCapInt tPow, const RampArgs& a,
    double phase0, bool pre, hipStream_t stream)
{
    const bitCapInt maxI = maxQPower >> 1u;
    if constexpr (std::is_same_v<R, float>) {
        if (tPow >= 2u && (maxI & 1u) == 0u) {
            if (pre) {
                hipLaunchKernelGGL((k_qft_col_gen_v<true>), dim3(gridFor(maxI >> 1u)),
                    dim3(QA_BLOCK), 0, stream, sv, maxI, tPow, a, (float)phase0);
            } else {
                hipLaunchKernelGGL((k_qft_col_gen_v<false>), dim3(gridFor(maxI >> 1u)),
                    dim3(QA_BLOCK), 0, stream, sv, maxI, tPow, a, (float)phase0);
            }
            return;
        }
    }
    if (pre) {
        hipLaunchKernelGGL((k_qft_col_gen<R, true>), dim3(gridFor(maxI)), dim3(QA_BLOCK), 0,
            stream, sv, maxI, tPow, a, (R)phase0);
    } else {
        hipLaunchKernelGGL((k_qft_col_gen<R, false>), dim3(gridFor(maxI)), dim3(QA_BLOCK), 0,
            stream, sv, maxI, tPow, a, (R)phase0);
    }
}

// TWO disjoint 4x4 gates per full-state pass (see kernels.hpp
// Gate4x4Pair2Args): each thread owns a 16-amplitude orbit over the two
// gates' four bit positions, applies gate A across its axis pair then
// gate B across the other — one state read+write for two SU(4)s.
template <typename R>
__global__ void __launch_bounds__(256, 3) k_mtrx2q_pair2(cplx<R>* sv, Gate4x4Pair2Args<R> a)
{
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt k = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; k < a.orbits;
         k += stride) {
        bitCapInt r = k;
        for (int b = 0; b < 4; ++b) r = insertZeroBitDev(r, a.sorted4[b]);
        bitCapInt idx[16];
        for (int ib = 0; ib < 4; ++ib) {
            for (int ia = 0; ia < 4; ++ia) {
                bitCapInt x = r;
                if (ia & 1) x |= a.pA1;
                if (ia & 2) x |= a.pA2;
                if (ib & 1) x |= a.pB1;
                if (ib & 2) x |= a.pB2;
                idx[4 * ib + ia] = x;
            }
        }
        cplx<R> v[16];
        for (int b = 0; b < 16; ++b) v[b] = sv[idx[b]];
        // gate A mixes the ia axis for each ib
        for (int ib = 0; ib < 4; ++ib) {
            cplx<R> out[4];
            for (int rI = 0; rI < 4; ++rI) {
                cplx<R> s(0, 0);
                for (int cI = 0; cI < 4; ++cI) s = s + a.mA[4 * rI + cI] * v[4 * ib + cI];
                out[rI] = s;
            }
            for (int rI = 0; rI < 4; ++rI) v[4 * ib + rI] = out[rI];
        }
        // gate B mixes the ib axis for each ia
        for (int ia = 0; ia < 4; ++ia) {
            cplx<R> out[4];
            for (int rI = 0; rI < 4; ++rI) {
                cplx<R> s(0, 0);
                for (int cI = 0; cI < 4; ++cI) s = s + a.mB[4 * rI + cI] * v[4 * cI + ia];
                out[rI] = s;
            }
            for (int rI = 0; rI < 4; ++rI) v[4 * rI + ia] = out[rI];
        }
        for (int b = 0; b < 16; ++b) sv[idx[b]] = v[b];
    }
}

template <typename R>
void launchMtrx2qPair2(cplx<R>* sv, const Gate4x4Pair2Args<R>& a, hipStream_t stream)
{
    hipLaunchKernelGGL(
        (k_mtrx2q_pair2<R>), dim3(gridFor(a.orbits)), dim3(QA_BLOCK), 0, stream, sv, a);
}

// TWO generalized QFT columns per pass for the distributed pager's local
// ladder: targets at ARBITRARY local slots (tPowHi/tPowLo by ROLE, not
// numeric order), ramp bits possibly relocated (RampArgs, which must
// EXCLUDE the low column's target bit — its contribution is the constant
// A = e^{i·sign·π/2}); per-column meta-page scalars phase0Hi/phase0Lo.
// Same algebra as k_qft_col2: f_hi(bLo) = f0·A^bLo·e^{i·phase0Hi},
// f_lo = f0²·e^{i·phase0Lo} with ONE sincos per orbit.
template <typename R, bool PRE>
__global__ void k_qft_col2_gen(cplx<R>* sv, bitCapInt orbits, bitCapInt tPowHi, bitCapInt tPowLo,
    RampArgs a, R phase0Hi, R phase0Lo)
{
    const R s = (R)0.70710678118654752440;
    const R iSign = (a.scale >= 0) ? (R)1 : (R)-1;
    const bitCapInt pLow = (tPowHi < tPowLo) ? tPowHi : tPowLo;
    const bitCapInt pHigh = (tPowHi < tPowLo) ? tPowLo : tPowHi;
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt k = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; k < orbits;
         k += stride) {
        const bitCapInt r = insertZeroBitDev(insertZeroBitDev(k, pLow), pHigh);
        uint64_t frac = (uint64_t)((r >> a.rampStart) & a.inPlaceRelMask);
        for (int t = 0; t < a.nScattered; ++t) {
            if (r & a.sPow[t]) frac += a.sWeight[t];
        }
        R sn, cs;
        devSinCos<R>((R)a.scale * (R)frac, &sn, &cs);
        const cplx<R> f0{ cs, sn };
        const cplx<R> A{ 0, iSign };
        R snh, csh, snl, csl;
        devSinCos<R>(phase0Hi, &snh, &csh);
        devSinCos<R>(phase0Lo, &snl, &csl);
        const cplx<R> fHi0 = f0 * cplx<R>{ csh, snh };
        const cplx<R> fHi1 = fHi0 * A;
        const cplx<R> fLo = f0 * f0 * cplx<R>{ csl, snl };
        cplx<R> a00 = sv[r];
        cplx<R> a01 = sv[r | tPowLo];
        cplx<R> a10 = sv[r | tPowHi];
        cplx<R> a11 = sv[r | tPowHi | tPowLo];
        if (!PRE) {
            cplx<R> b00 = s * (a00 + a10), b01 = s * (a01 + a11);
            cplx<R> b10 = fHi0 * (s * (a00 - a10)), b11 = fHi1 * (s * (a01 - a11));
            sv[r] = s * (b00 + b01);
            sv[r | tPowLo] = fLo * (s * (b00 - b01));
            sv[r | tPowHi] = s * (b10 + b11);
            sv[r | tPowHi | tPowLo] = fLo * (s * (b10 - b11));
        } else {
            a01 = fLo * a01;
            a11 = fLo * a11;
            cplx<R> b00 = s * (a00 + a01), b01 = s * (a00 - a01);
            cplx<R> b10 = s * (a10 + a11), b11 = s * (a10 - a11);
            b10 = fHi0 * b10;
            b11 = fHi1 * b11;
            sv[r] = s * (b00 + b10);
            sv[r | tPowHi] = s * (b00 - b10);
            sv[r | tPowLo] = s * (b01 + b11);
            sv[r | tPowHi | tPowLo] = s * (b01 - b11);
        }
    }
}

template <typename R>
void launchQftColumn2General(cplx<R>* sv, bitCapInt maxQPower, bitCapInt tPowHi, bitCapInt tPowLo,
    const RampArgs& a, double phase0Hi, double phase0Lo, bool pre, hipStream_t stream)
{
    const bitCapInt orbits = maxQPower >> 2u;
    if (pre) {
        hipLaunchKernelGGL((k_qft_col2_gen<R, true>), dim3(gridFor(orbits)), dim3(QA_BLOCK), 0,
            stream, sv, orbits, tPowHi, tPowLo, a, (R)phase0Hi, (R)phase0Lo);
    } else {
        hipLaunchKernelGGL((k_qft_col2_gen<R, false>), dim3(gridFor(orbits)), dim3(QA_BLOCK), 0,
            stream, sv, orbits, tPowHi, tPowLo, a, (R)phase0Hi, (R)phase0Lo);
    }
}

// Ranged top-target fused column with receive-buffer fusion (pipelined
// distributed page exchange; see kernels.hpp). Pair rows r in [itLo, itHi):
//   x (target=0 side) = recvIsLow ? recvSrc[r-itLo] : sv[r]
//   y (target=1 side) = recvIsLow ? sv[r|tPow]      : recvSrc[r-itLo]
// outputs written in place to sv[r], sv[r|tPow]. Because the target is the
// TOP bit, the ramp fraction depends only on r (< tPow) and the pair map is
// the identity — no bit interleave needed.
template <typename R, bool PRE, bool RECV_LOW>
__global__ void k_qft_col_top_range(cplx<R>* sv, bitCapInt tPow, RampArgs a, R phase0,
    bitCapInt itLo, bitCapInt itHi, const cplx<R>* __restrict__ recvSrc)
{
    const R s = (R)0.70710678118654752440;
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt r = itLo + (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; r < itHi;
         r += stride) {
        uint64_t frac = (uint64_t)((r >> a.rampStart) & a.inPlaceRelMask);
        for (int k = 0; k < a.nScattered; ++k) {
            if (r & a.sPow[k]) frac += a.sWeight[k];
        }
        R sn, cs;
        devSinCos<R>((R)a.scale * (R)frac + phase0, &sn, &cs);
        const cplx<R> f{ cs, sn };
        cplx<R> x = RECV_LOW ? recvSrc[r - itLo] : sv[r];
        cplx<R> y = RECV_LOW ? sv[r | tPow] : recvSrc[r - itLo];
        if (PRE) y = f * y;
        cplx<R> o0 = s * (x + y);
        cplx<R> o1 = s * (x - y);
        if (!PRE) o1 = f * o1;
        sv[r] = o0;
        sv[r | tPow] = o1;
    }
}

// float4-vectorized variant: two adjacent pair rows per thread (requires
// even itLo/itHi, which the chunking always produces for qpp >= 2)
template <bool PRE, bool RECV_LOW>
__global__ void k_qft_col_top_range_v(cplx<float>* sv, bitCapInt tPow, RampArgs a, float phase0,
    bitCapInt itLo, bitCapInt itHi, const cplx<float>* __restrict__ recvSrc)
{
    float4* sv4 = reinterpret_cast<float4*>(sv);
    const float4* rc4 = reinterpret_cast<const float4*>(recvSrc);
    const float s = 0.70710678f;
    const bitCapInt lo4 = itLo >> 1u, hi4 = itHi >> 1u;
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt k = lo4 + (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; k < hi4;
         k += stride) {
        const bitCapInt r = 2u * k;
        const bitCapInt t4 = (r | tPow) >> 1u;
        float4 vlo = RECV_LOW ? rc4[k - lo4] : sv4[k];
        float4 vhi = RECV_LOW ? sv4[t4] : rc4[k - lo4];
        uint64_t fr0 = (uint64_t)((r >> a.rampStart) & a.inPlaceRelMask);
        uint64_t fr1 = (uint64_t)(((r + 1u) >> a.rampStart) & a.inPlaceRelMask);
        for (int t = 0; t < a.nScattered; ++t) {
            if (r & a.sPow[t]) fr0 += a.sWeight[t];
            if ((r + 1u) & a.sPow[t]) fr1 += a.sWeight[t];
        }
        float s0, c0, s1, c1;
        __sincosf((float)a.scale * (float)fr0 + phase0, &s0, &c0);
        __sincosf((float)a.scale * (float)fr1 + phase0, &s1, &c1);
        const cplx<float> f0{ c0, s0 }, f1{ c1, s1 };
        cplx<float> x0{ vlo.x, vlo.y }, x1{ vlo.z, vlo.w };
        cplx<float> y0{ vhi.x, vhi.y }, y1{ vhi.z, vhi.w };
        if (PRE) {
            y0 = f0 * y0;
            y1 = f1 * y1;
        }
        cplx<float> a0 = s * (x0 + y0), a1 = s * (x1 + y1);
        cplx<float> b0 = s * (x0 - y0), b1 = s * (x1 - y1);
        if (!PRE) {
            b0 = f0 * b0;
            b1 = f1 * b1;
        }
        sv4[k] = make_float4(a0.re, a0.im, a1.re, a1.im);
        sv4[t4] = make_float4(b0.re, b0.im, b1.re, b1.im);
    }
}

template <typename R>
void launchQftColumnTopRange(cplx<R>* sv, bitCapInt maxQPower, const RampArgs& a, double phase0,
    bool pre, bitCapInt itLo, bitCapInt itHi, const cplx<R>* recvSrc, bool recvIsLow,
    hipStream_t stream)
{
    const bitCapInt tPow = maxQPower >> 1u;
    const bitCapInt n = itHi - itLo;
    if (!n) return;
    if constexpr (std::is_same_v<R, float>) {
        if ((itLo & 1u) == 0u && (itHi & 1u) == 0u) {
            const dim3 g(gridFor(n >> 1u)), b(QA_BLOCK);
            if (pre) {
                if (recvIsLow)
                    hipLaunchKernelGGL((k_qft_col_top_range_v<true, true>), g, b, 0, stream, sv,
                        tPow, a, (float)phase0, itLo, itHi, recvSrc);
                else
                    hipLaunchKernelGGL((k_qft_col_top_range_v<true, false>), g, b, 0, stream, sv,
                        tPow, a, (float)phase0, itLo, itHi, recvSrc);
            } else {
                if (recvIsLow)
                    hipLaunchKernelGGL((k_qft_col_top_range_v<false, true>), g, b, 0, stream, sv,
                        tPow, a, (float)phase0, itLo, itHi, recvSrc);
                else
                    hipLaunchKernelGGL((k_qft_col_top_range_v<false, false>), g, b, 0, stream, sv,
                        tPow, a, (float)phase0, itLo, itHi, recvSrc);
            }
            return;
        }
    }
    const dim3 g(gridFor(n)), b(QA_BLOCK);
    if (pre) {
        if (recvIsLow)
            hipLaunchKernelGGL((k_qft_col_top_range<R, true, true>), g, b, 0, stream, sv, tPow, a,
                (R)phase0, itLo, itHi, recvSrc);
        else
            hipLaunchKernelGGL((k_qft_col_top_range<R, true, false>), g, b, 0, stream, sv, tPow,
                a, (R)phase0, itLo, itHi, recvSrc);
    } else {
        if (recvIsLow)
            hipLaunchKernelGGL((k_qft_col_top_range<R, false, true>), g, b, 0, stream, sv, tPow,
                a, (R)phase0, itLo, itHi, recvSrc);
        else
            hipLaunchKernelGGL((k_qft_col_top_range<R, false, false>), g, b, 0, stream, sv, tPow,
                a, (R)phase0, itLo, itHi, recvSrc);
    }
}

// TWO QFT columns fused in ONE full-state pass (halves the pass count of
// the already-fused per-column ladder): columns (col, col-1) process
// 4-amplitude orbits over bits (tHi, tLo). The cross-column ramp term is a
// CONSTANT factor iF = e^{i·sign·π/2} and the lower column's ramp is the
// square of the upper's, so ONE sincos per orbit drives all three phase
// factors:  f_hi(bLo) = f0·iF^bLo,  f_lo = f0².
// Forward (PRE=false, QFT): H_hi, ramp_hi, H_lo, ramp_lo.
// Inverse (PRE=true, IQFT): ramp_lo, H_lo, ramp_hi, H_hi (exact adjoint).
template <typename R, bool PRE>
__global__ void k_qft_col2(cplx<R>* sv, bitCapInt orbits, bitCapInt tHi, bitCapInt tLo,
    bitLenInt rampStart, bitCapInt lowMask, R scaleHi)
{
    const R s = (R)0.70710678118654752440;
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    // iF = e^{i·sign·π/2}; sign is carried in scaleHi's sign
    const R iSign = (scaleHi >= 0) ? (R)1 : (R)-1;
    for (bitCapInt k = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; k < orbits;
         k += stride) {
        const bitCapInt r = insertZeroBitDev(insertZeroBitDev(k, tLo), tHi);
        const uint64_t lf = (uint64_t)((r >> rampStart) & lowMask);
        R sn, cs;
        devSinCos<R>(scaleHi * (R)lf, &sn, &cs);
        const cplx<R> f0{ cs, sn };
        const cplx<R> iF{ 0, iSign };
        const cplx<R> fHi0 = f0, fHi1 = f0 * iF;
        const cplx<R> fLo = f0 * f0;
        cplx<R> a00 = sv[r];
        cplx<R> a01 = sv[r | tLo];
        cplx<R> a10 = sv[r | tHi];
        cplx<R> a11 = sv[r | tHi | tLo];
        if (!PRE) {
            // H_hi then ramp_hi (phase on bHi=1, depends on bLo)
            cplx<R> b00 = s * (a00 + a10), b01 = s * (a01 + a11);
            cplx<R> b10 = fHi0 * (s * (a00 - a10)), b11 = fHi1 * (s * (a01 - a11));
            // H_lo then ramp_lo (phase on bLo=1)
            sv[r] = s * (b00 + b01);
            sv[r | tLo] = fLo * (s * (b00 - b01));
            sv[r | tHi] = s * (b10 + b11);
            sv[r | tHi | tLo] = fLo * (s * (b10 - b11));
        } else {
            // ramp_lo then H_lo
            a01 = fLo * a01;
            a11 = fLo * a11;
            cplx<R> b00 = s * (a00 + a01), b01 = s * (a00 - a01);
            cplx<R> b10 = s * (a10 + a11), b11 = s * (a10 - a11);
            // ramp_hi then H_hi
            b10 = fHi0 * b10;
            b11 = fHi1 * b11;
            sv[r] = s * (b00 + b10);
            sv[r | tHi] = s * (b00 - b10);
            sv[r | tLo] = s * (b01 + b11);
            sv[r | tHi | tLo] = s * (b01 - b11);
        }
    }
}

// float4-vectorized: each thread drives TWO adjacent orbits (8 amplitudes,
// 4 float4 RMWs). Requires tLo >= 2.
template <bool PRE>
__global__ void k_qft_col2_v(cplx<float>* sv, bitCapInt orbitPairs, bitCapInt tHi, bitCapInt tLo,
    bitLenInt rampStart, bitCapInt lowMask, float scaleHi)
{
    float4* sv4 = reinterpret_cast<float4*>(sv);
    const float s = 0.70710678f;
    const float iSign = (scaleHi >= 0) ? 1.0f : -1.0f;
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt m = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; m < orbitPairs;
         m += stride) {
        const bitCapInt r = insertZeroBitDev(insertZeroBitDev(2u * m, tLo), tHi);
        const bitCapInt i00 = r >> 1u;
        const bitCapInt i01 = (r | tLo) >> 1u;
        const bitCapInt i10 = (r | tHi) >> 1u;
        const bitCapInt i11 = (r | tHi | tLo) >> 1u;
        float4 v00 = sv4[i00], v01 = sv4[i01], v10 = sv4[i10], v11 = sv4[i11];
        const uint64_t lf0 = (uint64_t)((r >> rampStart) & lowMask);
        const uint64_t lf1 = (uint64_t)(((r + 1u) >> rampStart) & lowMask);
        float s0, c0, s1, c1;
        __sincosf(scaleHi * (float)lf0, &s0, &c0);
        __sincosf(scaleHi * (float)lf1, &s1, &c1);
        const cplx<float> f0a{ c0, s0 }, f0b{ c1, s1 };
        const cplx<float> iF{ 0.0f, iSign };
#define QA_COL2_BODY(aA, aB, aC, aD, f0)                                                           \
    {                                                                                              \
        const cplx<float> fHi0 = f0, fHi1 = f0 * iF, fLo = f0 * f0;                                \
        if (!PRE) {                                                                                \
            cplx<float> b00 = s * (aA + aC), b01 = s * (aB + aD);                                  \
            cplx<float> b10 = fHi0 * (s * (aA - aC)), b11 = fHi1 * (s * (aB - aD));                \
            aA = s * (b00 + b01);                                                                  \
            aB = fLo * (s * (b00 - b01));                                                          \
            aC = s * (b10 + b11);                                                                  \
            aD = fLo * (s * (b10 - b11));                                                          \
        } else {                                                                                   \
            cplx<float> t01 = fLo * aB, t11 = fLo * aD;                                            \
            cplx<float> b00 = s * (aA + t01), b01 = s * (aA - t01);                                \
            cplx<float> b10 = s * (aC + t11), b11 = s * (aC - t11);                                \
            b10 = fHi0 * b10;                                                                      \
            b11 = fHi1 * b11;                                                                      \
            aA = s * (b00 + b10);                                                                  \
            aC = s * (b00 - b10);                                                                  \
            aB = s * (b01 + b11);                                                                  \
            aD = s * (b01 - b11);                                                                  \
        }                                                                                          \
    }
        cplx<float> x00{ v00.x, v00.y }, y00{ v00.z, v00.w };
        cplx<float> x01{ v01.x, v01.y }, y01{ v01.z, v01.w };
        cplx<float> x10{ v10.x, v10.y }, y10{ v10.z, v10.w };
        cplx<float> x11{ v11.x, v11.y }, y11{ v11.z, v11.w };
        QA_COL2_BODY(x00, x01, x10, x11, f0a);
        QA_COL2_BODY(y00, y01, y10, y11, f0b);
#undef QA_COL2_BODY
        sv4[i00] = make_float4(x00.re, x00.im, y00.re, y00.im);
        sv4[i01] = make_float4(x01.re, x01.im, y01.re, y01.im);
        sv4[i10] = make_float4(x10.re, x10.im, y10.re, y10.im);
        sv4[i11] = make_float4(x11.re, x11.im, y11.re, y11.im);
    }
}

// THREE QFT columns fused per pass (8-amplitude orbits). Phase structure:
// with f0 = e^{i·s_hi·lf}, A = e^{i·sign·π/2} = ±i, B = e^{i·sign·π/4}:
//   ramp_hi(bMid,bLo) = f0·A^bMid·B^bLo, ramp_mid(bLo) = f0²·A^bLo,
//   ramp_lo = f0⁴ — still ONE sincos per orbit.
template <typename R, bool PRE>
__global__ void k_qft_col3(cplx<R>* sv, bitCapInt orbits, bitCapInt tHi, bitCapInt tMid,
    bitCapInt tLo, bitLenInt rampStart, bitCapInt lowMask, R scaleHi)
{
    const R s = (R)0.70710678118654752440;
    const R c45 = (R)0.70710678118654752440;
    const R iSign = (scaleHi >= 0) ? (R)1 : (R)-1;
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt k = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; k < orbits;
         k += stride) {
        const bitCapInt r =
            insertZeroBitDev(insertZeroBitDev(insertZeroBitDev(k, tLo), tMid), tHi);
        bitCapInt idx[8];
        for (int b = 0; b < 8; ++b) {
            idx[b] = r | ((b & 4) ? tHi : 0u) | ((b & 2) ? tMid : 0u) | ((b & 1) ? tLo : 0u);
        }
        cplx<R> v[8];
        for (int b = 0; b < 8; ++b) v[b] = sv[idx[b]];
        const uint64_t lf = (uint64_t)((r >> rampStart) & lowMask);
        R sn, cs;
        devSinCos<R>(scaleHi * (R)lf, &sn, &cs);
        const cplx<R> f0{ cs, sn };
        const cplx<R> A{ 0, iSign };
        const cplx<R> B{ c45, iSign * c45 };
        const cplx<R> f2 = f0 * f0;
        const cplx<R> f4 = f2 * f2;
        // per-slot hi-ramp factors for bHi=1: f0 * A^bMid * B^bLo
        cplx<R> fh[4];
        fh[0] = f0;
        fh[1] = f0 * B;
        fh[2] = f0 * A;
        fh[3] = f0 * A * B;
        auto hPair = [&](int lo, int hi) {
            const cplx<R> t = s * (v[lo] + v[hi]);
            const cplx<R> u = s * (v[lo] - v[hi]);
            v[lo] = t;
            v[hi] = u;
        };
        if (!PRE) {
            for (int b = 0; b < 4; ++b) hPair(b, b | 4);
            for (int m = 0; m < 4; ++m) v[4 | m] = fh[m] * v[4 | m];
            hPair(0, 2);
            hPair(1, 3);
            hPair(4, 6);
            hPair(5, 7);
            v[2] = f2 * v[2];
            v[3] = f2 * A * v[3];
            v[6] = f2 * v[6];
            v[7] = f2 * A * v[7];
            hPair(0, 1);
            hPair(2, 3);
            hPair(4, 5);
            hPair(6, 7);
            v[1] = f4 * v[1];
            v[3] = f4 * v[3];
            v[5] = f4 * v[5];
            v[7] = f4 * v[7];
        } else {
            // exact adjoint order (angles carry the sign): ramp_lo, H_lo,
            // ramp_mid, H_mid, ramp_hi, H_hi
            v[1] = f4 * v[1];
            v[3] = f4 * v[3];
            v[5] = f4 * v[5];
            v[7] = f4 * v[7];
            hPair(0, 1);
            hPair(2, 3);
            hPair(4, 5);
            hPair(6, 7);
            v[2] = f2 * v[2];
            v[3] = f2 * A * v[3];
            v[6] = f2 * v[6];
            v[7] = f2 * A * v[7];
            hPair(0, 2);
            hPair(1, 3);
            hPair(4, 6);
            hPair(5, 7);
            for (int m = 0; m < 4; ++m) v[4 | m] = fh[m] * v[4 | m];
            for (int b = 0; b < 4; ++b) hPair(b, b | 4);
        }
        for (int b = 0; b < 8; ++b) sv[idx[b]] = v[b];
    }
}

// float4-vectorized 3-column kernel: two adjacent orbits per thread
// (16 amplitudes, 8 float4 RMWs); requires tLo >= 2.
template <bool PRE>
__global__ void k_qft_col3_v(cplx<float>* sv, bitCapInt orbitPairs, bitCapInt tHi, bitCapInt tMid,
    bitCapInt tLo, bitLenInt rampStart, bitCapInt lowMask, float scaleHi)
{
    float4* sv4 = reinterpret_cast<float4*>(sv);
    const float s = 0.70710678f;
    const float iSign = (scaleHi >= 0) ? 1.0f : -1.0f;
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt m = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; m < orbitPairs;
         m += stride) {
        const bitCapInt r =
            insertZeroBitDev(insertZeroBitDev(insertZeroBitDev(2u * m, tLo), tMid), tHi);
        bitCapInt idx[8];
        for (int b = 0; b < 8; ++b) {
            idx[b] =
                (r | ((b & 4) ? tHi : 0u) | ((b & 2) ? tMid : 0u) | ((b & 1) ? tLo : 0u)) >> 1u;
        }
        float4 raw[8];
        for (int b = 0; b < 8; ++b) raw[b] = sv4[idx[b]];
        const uint64_t lf0 = (uint64_t)((r >> rampStart) & lowMask);
        const uint64_t lf1 = (uint64_t)(((r + 1u) >> rampStart) & lowMask);
        float s0, c0, s1, c1;
        __sincosf(scaleHi * (float)lf0, &s0, &c0);
        __sincosf(scaleHi * (float)lf1, &s1, &c1);
        const cplx<float> A{ 0.0f, iSign };
        const cplx<float> B{ 0.70710678f, iSign * 0.70710678f };
        for (int half = 0; half < 2; ++half) {
            const cplx<float> f0 = half ? cplx<float>{ c1, s1 } : cplx<float>{ c0, s0 };
            const cplx<float> f2 = f0 * f0;
            const cplx<float> f4 = f2 * f2;
            cplx<float> fh[4] = { f0, f0 * B, f0 * A, f0 * A * B };
            cplx<float> v[8];
            for (int b = 0; b < 8; ++b) {
                v[b] = half ? cplx<float>{ raw[b].z, raw[b].w } : cplx<float>{ raw[b].x, raw[b].y };
            }
            auto hPair = [&](int lo, int hi) {
                const cplx<float> t = s * (v[lo] + v[hi]);
                const cplx<float> u = s * (v[lo] - v[hi]);
                v[lo] = t;
                v[hi] = u;
            };
            if (!PRE) {
                for (int b = 0; b < 4; ++b) hPair(b, b | 4);
                for (int q = 0; q < 4; ++q) v[4 | q] = fh[q] * v[4 | q];
                hPair(0, 2);
                hPair(1, 3);
                hPair(4, 6);
                hPair(5, 7);
                v[2] = f2 * v[2];
                v[3] = f2 * A * v[3];
                v[6] = f2 * v[6];
                v[7] = f2 * A * v[7];
                hPair(0, 1);
                hPair(2, 3);
                hPair(4, 5);
                hPair(6, 7);
                v[1] = f4 * v[1];
                v[3] = f4 * v[3];
                v[5] = f4 * v[5];
                v[7] = f4 * v[7];
            } else {
                v[1] = f4 * v[1];
                v[3] = f4 * v[3];
                v[5] = f4 * v[5];
                v[7] = f4 * v[7];
                hPair(0, 1);
                hPair(2, 3);
                hPair(4, 5);
                hPair(6, 7);
                v[2] = f2 * v[2];
                v[3] = f2 * A * v[3];
                v[6] = f2 * v[6];
                v[7] = f2 * A * v[7];
                hPair(0, 2);
                hPair(1, 3);
                hPair(4, 6);
                hPair(5, 7);
                for (int q = 0; q < 4; ++q) v[4 | q] = fh[q] * v[4 | q];
                for (int b = 0; b < 4; ++b) hPair(b, b | 4);
            }
            for (int b = 0; b < 8; ++b) {
                if (half) {
                    raw[b].z = v[b].re;
                    raw[b].w = v[b].im;
                } else {
                    raw[b].x = v[b].re;
                    raw[b].y = v[b].im;
                }
            }
        }
        for (int b = 0; b < 8; ++b) sv4[idx[b]] = raw[b];
    }
}

// K-column orbit apply over 16 register-resident amplitudes — the same math
// as k_qft_colK (one f0 plus constant roots of unity drive every ramp), as a
// building block for the LDS-staged ladder kernels. v[b] is the amplitude
// whose group-column bits spell b; U[d] = e^{i·sign·π/2^d}; f0 encodes the
// below-group ramp contribution at the group's HIGHEST column scale.
template <typename R, int K, bool PRE>
__device__ __forceinline__ void qftOrbitColumns(cplx<R>* v, R iSign, cplx<R> f0)
{
    const R s = (R)0.70710678118654752440;
    constexpr int NS = 1 << K;
    // U(d) = e^{i·sign·π/2^d} — compile-time roots of unity (indices below
    // are constants after full unrolling, so these fold into immediates)
    const cplx<R> U[4] = { cplx<R>{ (R)1, (R)0 }, cplx<R>{ (R)0, iSign },
        cplx<R>{ (R)0.70710678118654752440, iSign * (R)0.70710678118654752440 },
        cplx<R>{ (R)0.92387953251128675613, iSign * (R)0.38268343236508977173 } };
    cplx<R> fPow[K];
    fPow[0] = f0;
#pragma unroll
    for (int m = 1; m < K; ++m) fPow[m] = fPow[m - 1] * fPow[m - 1];
    auto column = [&](int c) {
#pragma unroll
        for (int b = 0; b < NS; ++b) {
            if (b & (1 << c)) continue;
            const int hb = b | (1 << c);
            const cplx<R> t = s * (v[b] + v[hb]);
            const cplx<R> u = s * (v[b] - v[hb]);
            v[b] = t;
            v[hb] = u;
        }
#pragma unroll
        for (int b = 0; b < NS; ++b) {
            if (!(b & (1 << c))) continue;
            cplx<R> f = fPow[K - 1 - c];
#pragma unroll
            for (int j = 0; j < K; ++j) {
                if (j < c && (b & (1 << j))) f = f * U[c - j];
            }
            v[b] = f * v[b];
        }
    };
    auto columnInv = [&](int c) {
#pragma unroll
        for (int b = 0; b < NS; ++b) {
            if (!(b & (1 << c))) continue;
            cplx<R> f = fPow[K - 1 - c];
#pragma unroll
            for (int j = 0; j < K; ++j) {
                if (j < c && (b & (1 << j))) f = f * U[c - j];
            }
            v[b] = f * v[b];
        }
#pragma unroll
        for (int b = 0; b < NS; ++b) {
            if (b & (1 << c)) continue;
            const int hb = b | (1 << c);
            const cplx<R> t = s * (v[b] + v[hb]);
            const cplx<R> u = s * (v[b] - v[hb]);
            v[b] = t;
            v[hb] = u;
        }
    };
    if (!PRE) {
#pragma unroll
        for (int c = K - 1; c >= 0; --c) column(c);
    } else {
#pragma unroll
        for (int c = 0; c < K; ++c) columnInv(c);
    }
}

// LDS bank swizzle for the low-ladder kernel: XOR index bits 4..7 into
// bits 0..3 so every access pattern the orbit groups generate (element
// strides 1, 16, 256 within a wave) lands on 16 distinct bank pairs per
// 16 consecutive lanes — the optimum for 8/16-byte LDS elements.
__device__ __forceinline__ int qaSwz(int j) { return j ^ ((j >> 4) & 15); }

// One group of K columns applied over a 2^tb-amplitude tile as 16-amp
// register orbits. K is a template parameter so v[] stays in registers
// (runtime K forces the array to scratch — measured 2.5x slower). Groups
// with gLo >= 6 may read/write HBM directly (lane-contiguous within the
// orbit slot loops); gLo < 6 groups always go through (swizzled) LDS.
template <typename R, int K, bool PRE>
__device__ __forceinline__ void qftLowGroup(cplx<R>* svBase, cplx<R>* lds, int tile, int gLo,
    R scaleHi, R iSign, bool fromGlobal, bool toGlobal)
{
    constexpr int NS = 1 << K;
    const int orbitCount = tile >> K;
    for (int o = threadIdx.x; o < orbitCount; o += blockDim.x) {
        const int below = o & ((1 << gLo) - 1);
        const int r = ((o >> gLo) << (gLo + K)) | below;
        cplx<R> v[NS];
        if (fromGlobal) {
#pragma unroll
            for (int b = 0; b < NS; ++b) v[b] = svBase[r | (b << gLo)];
        } else {
#pragma unroll
            for (int b = 0; b < NS; ++b) v[b] = lds[qaSwz(r | (b << gLo))];
        }
        R sn, cs;
        devSinCos<R>(scaleHi * (R)below, &sn, &cs);
        qftOrbitColumns<R, K, PRE>(v, iSign, cplx<R>{ cs, sn });
        if (toGlobal) {
#pragma unroll
            for (int b = 0; b < NS; ++b) svBase[r | (b << gLo)] = v[b];
        } else {
#pragma unroll
            for (int b = 0; b < NS; ++b) lds[qaSwz(r | (b << gLo))] = v[b];
        }
    }
}



// ALL low-bit QFT columns in ONE pass: when the register starts at bit 0,
// columns colMax..0 act entirely inside a contiguous 2^tb-amplitude tile.
// Columns are processed in groups of (up to) 4 as 16-amplitude register
// orbits — the first group gathers straight from HBM, the last scatters
// straight back, and LDS carries the tile between groups. One global
// read+write, 2 barriers, and 1 sincos per orbit per group replace the
// former per-pair sincos ladder. tb = 12 (fp32) / 11 (fp64).
template <typename R, int K, bool PRE>
__global__ void __launch_bounds__(256, 3)
    k_qft_low_lds(cplx<R>* sv, bitCapInt nTiles, int tb, int colMax, R piSign)
{
    // REQUIRES colMax+1 to be a multiple of K (the engine aligns the ladder
    // length; K = 4 fp32 / 3 fp64) so every group is the single kernel-wide
    // instantiation — keeps the orbit registers allocated without spills.
    extern __shared__ unsigned char qa_lds_raw[];
    cplx<R>* lds = reinterpret_cast<cplx<R>*>(qa_lds_raw);
    const int tile = 1 << tb;
    const int nCols = colMax + 1;
    const int nG = nCols / K;
    const R iSign = (piSign >= 0) ? (R)1 : (R)-1;
    // a group may touch HBM directly only if its orbit slot loops are
    // lane-contiguous, i.e. gLo >= 6; otherwise a contiguous LDS<->HBM
    // copy brackets the ladder on that side
    const int firstG = PRE ? 0 : (nG - 1);
    const int lastG = PRE ? (nG - 1) : 0;
    const bool copyIn = (K * firstG) < 6;
    const bool copyOut = (K * lastG) < 6;
    for (bitCapInt t = blockIdx.x; t < nTiles; t += gridDim.x) {
        cplx<R>* svBase = sv + (t << tb);
        if (copyIn) {
            for (int j = threadIdx.x; j < tile; j += blockDim.x) lds[qaSwz(j)] = svBase[j];
        }
        for (int gi = 0; gi < nG; ++gi) {
            const int g = PRE ? gi : (nG - 1 - gi);
            const int gLo = K * g;
            const bool fromGlobal = (gi == 0) && !copyIn;
            const bool toGlobal = (gi == nG - 1) && !copyOut;
            const R scaleHi = piSign / (R)(1 << (gLo + K - 1));
            if (gi != 0 || copyIn) __syncthreads();
            qftLowGroup<R, K, PRE>(svBase, lds, tile, gLo, scaleHi, iSign, fromGlobal, toGlobal);
        }
        if (copyOut) {
            __syncthreads();
            for (int j = threadIdx.x; j < tile; j += blockDim.x) svBase[j] = lds[qaSwz(j)];
        }
        __syncthreads();
    }
}

template <typename R>
void launchQftLowLds(
    cplx<R>* sv, bitCapInt maxQPower, int tb, int colMax, int sign, bool pre, hipStream_t stream)
{
    constexpr int K = qaLowLadderK<R>();
    const bitCapInt nTiles = maxQPower >> tb;
    const size_t ldsBytes = (size_t(1) << tb) * sizeof(cplx<R>);
    const int grid = (int)std::min<bitCapInt>(nTiles, (bitCapInt)QA_REDUCE_MAX_BLOCKS);
    const R piSign = (R)sign * (R)3.14159265358979323846;
    if (pre) {
        hipLaunchKernelGGL((k_qft_low_lds<R, K, true>), dim3(grid), dim3(QA_BLOCK), ldsBytes,
            stream, sv, nTiles, tb, colMax, piSign);
    } else {
        hipLaunchKernelGGL((k_qft_low_lds<R, K, false>), dim3(grid), dim3(QA_BLOCK), ldsBytes,
            stream, sv, nTiles, tb, colMax, piSign);
    }
}

// One group of K mid columns over a 64 x 2^nCols 2D tile as 16-amp register
// orbits (K templated — see qftLowGroup). Every access, LDS or HBM, keeps
// the 64-value `low` dimension in the lane index, so loads/stores are
// 512 B-contiguous per slot and LDS is bank-conflict-free without swizzle.
template <typename R, int K, bool PRE>
__device__ __forceinline__ void qftMidGroup(cplx<R>* sv, cplx<R>* lds, bitCapInt xBase,
    int colLo, int tileAmps, int gLo, R midFixed, R hbScale, R scaleHi, R iSign,
    bool fromGlobal, bool toGlobal)
{
    constexpr int NS = 1 << K;
    const int orbitCount = tileAmps >> K;
    for (int o = threadIdx.x; o < orbitCount; o += blockDim.x) {
        const int low = o & 63;
        const int cbr = o >> 6;
        const int cbBelow = cbr & ((1 << gLo) - 1);
        const int cb0 = ((cbr >> gLo) << (gLo + K)) | cbBelow;
        cplx<R> v[NS];
        if (fromGlobal) {
#pragma unroll
            for (int b = 0; b < NS; ++b) {
                v[b] = sv[xBase | ((bitCapInt)(cb0 | (b << gLo)) << colLo) | (bitCapInt)low];
            }
        } else {
#pragma unroll
            for (int b = 0; b < NS; ++b) v[b] = lds[((cb0 | (b << gLo)) << 6) | low];
        }
        R sn, cs;
        devSinCos<R>(scaleHi * ((R)low + midFixed + hbScale * (R)cbBelow), &sn, &cs);
        qftOrbitColumns<R, K, PRE>(v, iSign, cplx<R>{ cs, sn });
        if (toGlobal) {
#pragma unroll
            for (int b = 0; b < NS; ++b) {
                sv[xBase | ((bitCapInt)(cb0 | (b << gLo)) << colLo) | (bitCapInt)low] = v[b];
            }
        } else {
#pragma unroll
            for (int b = 0; b < NS; ++b) lds[((cb0 | (b << gLo)) << 6) | low] = v[b];
        }
    }
}

// Up to SIX mid-range QFT columns per pass through a 2D LDS tile: 64
// contiguous low amplitudes (coalesced 512 B runs) x 2^nCols column-bit
// combinations. Column bits are processed in groups of (up to) 4 as
// 16-amplitude register orbits (qftOrbitColumns): the first group gathers
// straight from HBM, the last scatters straight back, LDS carries the tile
// between groups. Ramp for column col: theta = scale_col * (x mod 2^col)
// where x mod 2^col = low6 + midFixed + cbBelow*2^colLo — low6 varies
// in-tile, midFixed (bits 6..colLo-1) is tile-constant, cbBelow is the
// tile's below-group column bits. Requires a start-0 register, colLo >= 6.
template <typename R, int K, bool PRE>
__global__ void __launch_bounds__(256, 3)
    k_qft_mid_lds(cplx<R>* sv, bitCapInt nTiles, int colLo, int nCols, R piSign)
{
    // REQUIRES nCols to be a multiple of K (launcher picks K) so every
    // group is the single kernel-wide instantiation — no register spills.
    extern __shared__ unsigned char qa_lds_raw2[];
    cplx<R>* lds = reinterpret_cast<cplx<R>*>(qa_lds_raw2);
    const int C = 1 << nCols;
    const int tileAmps = 64 * C;
    const int nG = nCols / K;
    const bitCapInt midMask = (ONE_BCI << (colLo - 6)) - 1u;
    const R hbScale = (R)(uint64_t)(ONE_BCI << colLo);
    const R iSign = (piSign >= 0) ? (R)1 : (R)-1;
    for (bitCapInt t = blockIdx.x; t < nTiles; t += gridDim.x) {
        const bitCapInt xBase =
            ((t >> (colLo - 6)) << (colLo + nCols)) | ((t & midMask) << 6);
        const R midFixed = (R)(uint64_t)((t & midMask) << 6);
        for (int gi = 0; gi < nG; ++gi) {
            const int g = PRE ? gi : (nG - 1 - gi);
            const int gLo = K * g;
            const bool fromGlobal = (gi == 0);
            const bool toGlobal = (gi == nG - 1);
            const R scaleHi = piSign / (R)(ONE_BCI << (colLo + gLo + K - 1));
            if (gi != 0) __syncthreads();
            qftMidGroup<R, K, PRE>(sv, lds, xBase, colLo, tileAmps, gLo, midFixed, hbScale,
                scaleHi, iSign, fromGlobal, toGlobal);
        }
        __syncthreads();
    }
}

template <typename R>
void launchQftMidLds(
    cplx<R>* sv, bitCapInt maxQPower, int colLo, int nCols, int sign, bool pre, hipStream_t stream)
{
    const bitCapInt nTiles = maxQPower >> (6 + nCols);
    const size_t ldsBytes = (size_t(64) << nCols) * sizeof(cplx<R>);
    const int grid = (int)std::min<bitCapInt>(nTiles, (bitCapInt)QA_REDUCE_MAX_BLOCKS);
    const R piSign = (R)sign * (R)3.14159265358979323846;
    // uniform group width: 4 | nCols -> K=4, else 3 | nCols -> K=3 (the
    // engine only requests nCols in {4, 6}; other multiples also work)
    if ((nCols & 3) == 0) {
        if (pre) {
            hipLaunchKernelGGL((k_qft_mid_lds<R, 4, true>), dim3(grid), dim3(QA_BLOCK), ldsBytes,
                stream, sv, nTiles, colLo, nCols, piSign);
        } else {
            hipLaunchKernelGGL((k_qft_mid_lds<R, 4, false>), dim3(grid), dim3(QA_BLOCK), ldsBytes,
                stream, sv, nTiles, colLo, nCols, piSign);
        }
    } else if ((nCols % 3) == 0) {
        if (pre) {
            hipLaunchKernelGGL((k_qft_mid_lds<R, 3, true>), dim3(grid), dim3(QA_BLOCK), ldsBytes,
                stream, sv, nTiles, colLo, nCols, piSign);
        } else {
            hipLaunchKernelGGL((k_qft_mid_lds<R, 3, false>), dim3(grid), dim3(QA_BLOCK), ldsBytes,
                stream, sv, nTiles, colLo, nCols, piSign);
        }
    } else {
        if (pre) {
            hipLaunchKernelGGL((k_qft_mid_lds<R, 2, true>), dim3(grid), dim3(QA_BLOCK), ldsBytes,
                stream, sv, nTiles, colLo, nCols, piSign);
        } else {
            hipLaunchKernelGGL((k_qft_mid_lds<R, 2, false>), dim3(grid), dim3(QA_BLOCK), ldsBytes,
                stream, sv, nTiles, colLo, nCols, piSign);
        }
    }
}

// GENERIC K-column fused QFT kernel (2^K-amplitude orbits). The per-column
// ramp factor for the slot with column-c's bit set is
//   f0^(2^(K-1-c)) · Π_{j<c} U(c-j)^(b_j),   U(d) = e^{i·sign·π/2^d}
// — one sincos (f0) plus K-1 constant roots of unity drive every ramp.
template <typename R, int K, bool PRE>
__global__ void k_qft_colK(cplx<R>* sv, bitCapInt orbits, const bitCapInt tPows0,
    const bitCapInt tPows1, const bitCapInt tPows2, const bitCapInt tPows3, const bitCapInt tPows4,
    bitLenInt rampStart, bitCapInt lowMask, R scaleHi)
{
    // tPows: [0]=lowest column bit ... [K-1]=highest
    const bitCapInt tP[5] = { tPows0, tPows1, tPows2, tPows3, tPows4 };
    const R s = (R)0.70710678118654752440;
    const R iSign = (scaleHi >= 0) ? (R)1 : (R)-1;
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    // U(d) = e^{i·sign·π/2^d}, d = 1..K-1
    cplx<R> U[5];
    {
        R ang = iSign * (R)1.57079632679489661923; // π/2
        for (int d = 1; d < K; ++d) {
            R sn, cs;
            devSinCos<R>(ang, &sn, &cs);
            U[d] = cplx<R>{ cs, sn };
            ang = ang * (R)0.5;
        }
    }
    const int NS = 1 << K;
    for (bitCapInt k = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; k < orbits;
         k += stride) {
        bitCapInt r = k;
        for (int b = 0; b < K; ++b) r = insertZeroBitDev(r, tP[b]);
        bitCapInt idx[1 << K];
        for (int b = 0; b < NS; ++b) {
            bitCapInt x = r;
            for (int c = 0; c < K; ++c) {
                if (b & (1 << c)) x |= tP[c];
            }
            idx[b] = x;
        }
        cplx<R> v[1 << K];
        for (int b = 0; b < NS; ++b) v[b] = sv[idx[b]];
        const uint64_t lf = (uint64_t)((r >> rampStart) & lowMask);
        R sn, cs;
        devSinCos<R>(scaleHi * (R)lf, &sn, &cs);
        cplx<R> fPow[5]; // f0^(2^m): fPow[0]=f0, fPow[m]=fPow[m-1]^2
        fPow[0] = cplx<R>{ cs, sn };
        for (int m = 1; m < K; ++m) fPow[m] = fPow[m - 1] * fPow[m - 1];
        // slot bit c corresponds to column index (from hi): c=K-1 is hi
        auto column = [&](int c) {
            // H over axis bit c
            for (int b = 0; b < NS; ++b) {
                if (b & (1 << c)) continue;
                const int hb = b | (1 << c);
                const cplx<R> t = s * (v[b] + v[hb]);
                const cplx<R> u = s * (v[b] - v[hb]);
                v[b] = t;
                v[hb] = u;
            }
            // ramp on slots with bit c set
            for (int b = 0; b < NS; ++b) {
                if (!(b & (1 << c))) continue;
                cplx<R> f = fPow[K - 1 - c];
                for (int j = 0; j < c; ++j) {
                    if (b & (1 << j)) f = f * U[c - j];
                }
                v[b] = f * v[b];
            }
        };
        auto columnInv = [&](int c) {
            for (int b = 0; b < NS; ++b) {
                if (!(b & (1 << c))) continue;
                cplx<R> f = fPow[K - 1 - c];
                for (int j = 0; j < c; ++j) {
                    if (b & (1 << j)) f = f * U[c - j];
                }
                v[b] = f * v[b];
            }
            for (int b = 0; b < NS; ++b) {
                if (b & (1 << c)) continue;
                const int hb = b | (1 << c);
                const cplx<R> t = s * (v[b] + v[hb]);
                const cplx<R> u = s * (v[b] - v[hb]);
                v[b] = t;
                v[hb] = u;
            }
        };
        if (!PRE) {
            for (int c = K - 1; c >= 0; --c) column(c);
        } else {
            for (int c = 0; c < K; ++c) columnInv(c);
        }
        for (int b = 0; b < NS; ++b) sv[idx[b]] = v[b];
    }
}

template <typename R>
void launchQftColumnK(cplx<R>* sv, bitCapInt maxQPower, bitLenInt rampStart, bitLenInt col,
    int kCols, const bitCapInt* tPows, int sign, bool pre, hipStream_t stream)
{
    // tPows[0..kCols-1] ascending (lowest column first); col = highest column
    const bitCapInt lowMask = (ONE_BCI << (col - (kCols - 1))) - 1u;
    const R scaleHi = (R)sign * (R)3.14159265358979323846 / (R)(ONE_BCI << col);
    const bitCapInt orbits = maxQPower >> kCols;
    if (kCols == 5) {
        if (pre) {
            hipLaunchKernelGGL((k_qft_colK<R, 5, true>), dim3(gridFor(orbits)), dim3(QA_BLOCK),
                0, stream, sv, orbits, tPows[0], tPows[1], tPows[2], tPows[3], tPows[4],
                rampStart, lowMask, scaleHi);
        } else {
            hipLaunchKernelGGL((k_qft_colK<R, 5, false>), dim3(gridFor(orbits)), dim3(QA_BLOCK),
                0, stream, sv, orbits, tPows[0], tPows[1], tPows[2], tPows[3], tPows[4],
                rampStart, lowMask, scaleHi);
        }
        return;
    }
    if (kCols != 4) return; // 2/3 have tuned float4 kernels
    if (pre) {
        hipLaunchKernelGGL((k_qft_colK<R, 4, true>), dim3(gridFor(orbits)), dim3(QA_BLOCK), 0,
            stream, sv, orbits, tPows[0], tPows[1], tPows[2], tPows[3], 0u, rampStart, lowMask,
            scaleHi);
    } else {
        hipLaunchKernelGGL((k_qft_colK<R, 4, false>), dim3(gridFor(orbits)), dim3(QA_BLOCK), 0,
            stream, sv, orbits, tPows[0], tPows[1], tPows[2], tPows[3], 0u, rampStart, lowMask,
            scaleHi);
    }
}

template <typename R>
void launchQftColumn3(cplx<R>* sv, bitCapInt maxQPower, bitLenInt rampStart, bitLenInt col,
    bitCapInt tHi, bitCapInt tMid, bitCapInt tLo, int sign, bool pre, hipStream_t stream)
{
    const bitCapInt lowMask = (ONE_BCI << (col - 2u)) - 1u;
    const R scaleHi = (R)sign * (R)3.14159265358979323846 / (R)(ONE_BCI << col);
    const bitCapInt orbits = maxQPower >> 3u;
    if constexpr (std::is_same_v<R, float>) {
        if (tLo >= 2u && (orbits & 1u) == 0u) {
            if (pre) {
                hipLaunchKernelGGL((k_qft_col3_v<true>), dim3(gridFor(orbits >> 1u)),
                    dim3(QA_BLOCK), 0, stream, sv, orbits >> 1u, tHi, tMid, tLo, rampStart,
                    lowMask, (float)scaleHi);
            } else {
                hipLaunchKernelGGL((k_qft_col3_v<false>), dim3(gridFor(orbits >> 1u)),
                    dim3(QA_BLOCK), 0, stream, sv, orbits >> 1u, tHi, tMid, tLo, rampStart,
                    lowMask, (float)scaleHi);
            }
            return;
        }
    }
    if (pre) {
        hipLaunchKernelGGL((k_qft_col3<R, true>), dim3(gridFor(orbits)), dim3(QA_BLOCK), 0,
            stream, sv, orbits, tHi, tMid, tLo, rampStart, lowMask, scaleHi);
    } else {
        hipLaunchKernelGGL((k_qft_col3<R, false>), dim3(gridFor(orbits)), dim3(QA_BLOCK), 0,
            stream, sv, orbits, tHi, tMid, tLo, rampStart, lowMask, scaleHi);
    }
}

template <typename R>
void launchQftColumn2(cplx<R>* sv, bitCapInt maxQPower, bitLenInt rampStart, bitLenInt col,
    bitCapInt tHi, bitCapInt tLo, int sign, bool pre, hipStream_t stream)
{
    // columns (col, col-1): ramp low-bit count = col-1 bits below tLo
    const bitCapInt lowMask = (ONE_BCI << (col - 1u)) - 1u;
    const R scaleHi = (R)sign * (R)3.14159265358979323846 / (R)(ONE_BCI << col);
    const bitCapInt orbits = maxQPower >> 2u;
    if constexpr (std::is_same_v<R, float>) {
        if (tLo >= 2u && (orbits & 1u) == 0u) {
            if (pre) {
                hipLaunchKernelGGL((k_qft_col2_v<true>), dim3(gridFor(orbits >> 1u)),
                    dim3(QA_BLOCK), 0, stream, sv, orbits >> 1u, tHi, tLo, rampStart, lowMask,
                    (float)scaleHi);
            } else {
                hipLaunchKernelGGL((k_qft_col2_v<false>), dim3(gridFor(orbits >> 1u)),
                    dim3(QA_BLOCK), 0, stream, sv, orbits >> 1u, tHi, tLo, rampStart, lowMask,
                    (float)scaleHi);
            }
            return;
        }
    }
    if (pre) {
        hipLaunchKernelGGL((k_qft_col2<R, true>), dim3(gridFor(orbits)), dim3(QA_BLOCK), 0,
            stream, sv, orbits, tHi, tLo, rampStart, lowMask, scaleHi);
    } else {
        hipLaunchKernelGGL((k_qft_col2<R, false>), dim3(gridFor(orbits)), dim3(QA_BLOCK), 0,
            stream, sv, orbits, tHi, tLo, rampStart, lowMask, scaleHi);
    }
}

template <typename R>
void launchQftColumn(cplx<R>* sv, bitCapInt maxQPower, bitLenInt rampStart, bitLenInt col,
    bitCapInt tPow, int sign, bool pre, hipStream_t stream)
{
    const bitCapInt maxI = maxQPower >> 1u;
    const bitCapInt rampMask = (ONE_BCI << col) - 1u;
    const R scale = (R)sign * (R)3.14159265358979323846 / (R)(ONE_BCI << col);
    if constexpr (std::is_same_v<R, float>) {
        if (tPow >= 2u && (maxI & 1u) == 0u) {
            static const bool pipe = []() {
                if (const char* env = std::getenv("QRACK_GPU_QFT_PIPE")) return std::atoi(env) != 0;
                return false;
            }();
            if (pipe) {
                if (pre) {
                    hipLaunchKernelGGL((k_qft_col_v_pipe<true>), dim3(gridFor(maxI >> 1u)),
                        dim3(QA_BLOCK), 0, stream, sv, maxI, tPow, rampStart, rampMask, (float)scale);
                } else {
                    hipLaunchKernelGGL((k_qft_col_v_pipe<false>), dim3(gridFor(maxI >> 1u)),
                        dim3(QA_BLOCK), 0, stream, sv, maxI, tPow, rampStart, rampMask, (float)scale);
                }
                return;
            }
            if (pre) {
                hipLaunchKernelGGL((k_qft_col_v<true>), dim3(gridFor(maxI >> 1u)), dim3(QA_BLOCK),
                    0, stream, sv, maxI, tPow, rampStart, rampMask, (float)scale);
            } else {
                hipLaunchKernelGGL((k_qft_col_v<false>), dim3(gridFor(maxI >> 1u)), dim3(QA_BLOCK),
                    0, stream, sv, maxI, tPow, rampStart, rampMask, (float)scale);
            }
            return;
        }
    }
    if (pre) {
        hipLaunchKernelGGL((k_qft_col<R, true>), dim3(gridFor(maxI)), dim3(QA_BLOCK), 0, stream,
            sv, maxI, tPow, rampStart, rampMask, scale);
    } else {
        hipLaunchKernelGGL((k_qft_col<R, false>), dim3(gridFor(maxI)), dim3(QA_BLOCK), 0, stream,
            sv, maxI, tPow, rampStart, rampMask, scale);
    }
}

// ---- batched independent single-qubit gates ---------------------------------
// k distinct-target 2x2s in ONE pass: each lane owns a 2^k-amplitude orbit in
// registers (k <= 5 fp32 / 4 fp64), so k memory-bound passes collapse to one.

template <typename R, int K>
__global__ void __launch_bounds__(256, K >= 5 ? 2 : (K >= 3 ? 3 : 4))
    k_mtrx_batch(cplx<R>* sv, Batch1qArgs<R> a)
{
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt j = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; j < a.maxI; j += stride) {
        const bitCapInt base = expandBits(j, a.tPow, K);
        cplx<R> v[1 << K];
#pragma unroll
        for (int s = 0; s < (1 << K); ++s) {
            bitCapInt off = 0;
#pragma unroll
            for (int g = 0; g < K; ++g) {
                if (s & (1 << g)) off |= a.tPow[g];
            }
            v[s] = sv[base | off];
        }
#pragma unroll
        for (int g = 0; g < K; ++g) {
            const cplx<R> m0 = a.m[4 * g], m1 = a.m[4 * g + 1], m2 = a.m[4 * g + 2],
                          m3 = a.m[4 * g + 3];
#pragma unroll
            for (int s = 0; s < (1 << K); ++s) {
                if (s & (1 << g)) continue;
                const int t = s | (1 << g);
                const cplx<R> x = v[s], y = v[t];
                v[s] = m0 * x + m1 * y;
                v[t] = m2 * x + m3 * y;
            }
        }
#pragma unroll
        for (int s = 0; s < (1 << K); ++s) {
            bitCapInt off = 0;
#pragma unroll
            for (int g = 0; g < K; ++g) {
                if (s & (1 << g)) off |= a.tPow[g];
            }
            sv[base | off] = v[s];
        }
    }
}

// fp32 float4 variant (requires tPow[0] >= 2): each lane handles TWO adjacent
// orbits via 16 B loads/stores — full-line HBM streams like k_apply2x2_1v.
template <int K>
__global__ void __launch_bounds__(256, K >= 4 ? 2 : (K >= 3 ? 3 : 4))
    k_mtrx_batch_v(cplx<float>* sv, Batch1qArgs<float> a)
{
    float4* sv4 = reinterpret_cast<float4*>(sv);
    const bitCapInt half = a.maxI >> 1u;
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt t = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; t < half; t += stride) {
        const bitCapInt base = expandBits(2u * t, a.tPow, K); // even: bit0 non-target
        float4 v[1 << K];
#pragma unroll
        for (int s = 0; s < (1 << K); ++s) {
            bitCapInt off = 0;
#pragma unroll
            for (int g = 0; g < K; ++g) {
                if (s & (1 << g)) off |= a.tPow[g];
            }
            v[s] = sv4[(base | off) >> 1u];
        }
#pragma unroll
        for (int g = 0; g < K; ++g) {
            const cplx<float> m0 = a.m[4 * g], m1 = a.m[4 * g + 1], m2 = a.m[4 * g + 2],
                              m3 = a.m[4 * g + 3];
#pragma unroll
            for (int s = 0; s < (1 << K); ++s) {
                if (s & (1 << g)) continue;
                const int tt = s | (1 << g);
                const cplx<float> x0{ v[s].x, v[s].y }, x1{ v[s].z, v[s].w };
                const cplx<float> y0{ v[tt].x, v[tt].y }, y1{ v[tt].z, v[tt].w };
                const cplx<float> a0 = m0 * x0 + m1 * y0, a1 = m0 * x1 + m1 * y1;
                const cplx<float> b0 = m2 * x0 + m3 * y0, b1 = m2 * x1 + m3 * y1;
                v[s] = make_float4(a0.re, a0.im, a1.re, a1.im);
                v[tt] = make_float4(b0.re, b0.im, b1.re, b1.im);
            }
        }
#pragma unroll
        for (int s = 0; s < (1 << K); ++s) {
            bitCapInt off = 0;
#pragma unroll
            for (int g = 0; g < K; ++g) {
                if (s & (1 << g)) off |= a.tPow[g];
            }
            sv4[(base | off) >> 1u] = v[s];
        }
    }
}

// ---- MFMA demonstration variant ---------------------------------------------
// The BASELINE north star calls for "MFMA used for the batched 2x2 complex
// mat-vec". A 2x2 complex gate on a pair (x, y) is the 4x4 REAL matrix
// G = [[m0.re,-m0.im,m1.re,-m1.im],[m0.im,m0.re,m1.im,m1.re],
//      [m2.re,-m2.im,m3.re,-m3.im],[m2.im,m2.re,m3.im,m3.re]] acting on
// (x.re, x.im, y.re, y.im). Each wave feeds v_mfma_f32_16x16x4_f32 with
// A[i][k] = G[i%4][k] (the gate replicated down the rows) and
// B[k][j] = component k of pair j, producing 16 pair results per MFMA.
// The workload is HBM-bound at ~0.1 flop/byte, so this is a correctness +
// counter demonstration (QRACK_GPU_MFMA=1), not the default path — the A/B
// numbers are recorded in profiles/KERNELS.md.
typedef float qa_f32x4 __attribute__((ext_vector_type(4)));

__global__ void k_apply2x2_1mfma(cplx<float>* sv, GateArgs<float> a)
{
    const unsigned lane = threadIdx.x & 63u;
    const unsigned j16 = lane & 15u;   // B column / D column
    const unsigned kRow = lane >> 4u;  // A k-index / B k-index
    const bitCapInt p = a.qPowers[0];
    // gate as 4x4 real, row (lane&15)%4, column kRow
    const float G[4][4] = {
        { a.m[0].re, -a.m[0].im, a.m[1].re, -a.m[1].im },
        { a.m[0].im, a.m[0].re, a.m[1].im, a.m[1].re },
        { a.m[2].re, -a.m[2].im, a.m[3].re, -a.m[3].im },
        { a.m[2].im, a.m[2].re, a.m[3].im, a.m[3].re },
    };
    const float aVal = G[j16 & 3u][kRow];
    const bitCapInt wavesPerGrid = ((bitCapInt)gridDim.x * blockDim.x) >> 6u;
    const bitCapInt waveId = (((bitCapInt)blockIdx.x * blockDim.x + threadIdx.x) >> 6u);
    const bitCapInt nGroups = (a.maxI + 15u) >> 4u; // 16 pairs per wave step
    for (bitCapInt g = waveId; g < nGroups; g += wavesPerGrid) {
        const bitCapInt myPair = (g << 4u) | j16;
        float xre = 0, xim = 0, yre = 0, yim = 0;
        const bool live = myPair < a.maxI;
        bitCapInt i = 0;
        if (live) {
            i = ((myPair & ~(p - 1u)) << 1u) | (myPair & (p - 1u));
            const cplx<float> x = sv[i | a.offset1];
            const cplx<float> y = sv[i | a.offset2];
            xre = x.re;
            xim = x.im;
            yre = y.re;
            yim = y.im;
        }
        // B[k][j] = component k of pair j: pull from lane j via wave shuffle
        const int src = (int)j16;
        const float c0 = __shfl(xre, src, 64);
        const float c1 = __shfl(xim, src, 64);
        const float c2 = __shfl(yre, src, 64);
        const float c3 = __shfl(yim, src, 64);
        const float bVal = (kRow == 0u) ? c0 : (kRow == 1u) ? c1 : (kRow == 2u) ? c2 : c3;
        qa_f32x4 acc = { 0.f, 0.f, 0.f, 0.f };
        acc = __builtin_amdgcn_mfma_f32_16x16x4f32(aVal, bVal, acc, 0, 0, 0);
        // D[row][col]: lane holds rows (lane>>4)*4 + t, col = lane&15; every
        // 16-lane group holds the full 4-vector of pair (lane&15) — group 0
        // writes back
        if (live && kRow == 0u) {
            sv[i | a.offset1] = cplx<float>{ acc[0], acc[1] };
            sv[i | a.offset2] = cplx<float>{ acc[2], acc[3] };
        }
    }
}

// ---- LDS-tiled low-bit gate batch -------------------------------------------
// Each workgroup stages a contiguous 2^QA_LDS_TILE_BITS-amplitude tile in
// LDS, applies every gate (targets all inside the tile) at LDS bandwidth,
// then writes the tile back: ONE global RMW pass for up to 12 fused gates,
// with bit-0 targets handled as cheaply as any other.

// Gates are applied in GROUPS of qaLdsBatchK<R>() (4 fp32 / 3 fp64) as
// 2^K-amplitude register orbits: one LDS read + butterfly chain in
// registers + one LDS write per amp per group, instead of a pair-RMW
// through LDS per GATE. The engine pads a.k to a multiple of K with
// identity gates on spare tile bits and sorts each group's targets
// ascending (1q gates on distinct targets commute, so order is free).
template <typename R> __global__ void k_mtrx_batch_lds(cplx<R>* sv, BatchLdsArgs<R> a)
{
    constexpr int TB = qaLdsTileBits<R>();
    constexpr int K = qaLdsBatchK<R>();
    constexpr int NS = 1 << K;
    __shared__ cplx<R> tile[1u << TB];
    constexpr unsigned TILE = 1u << TB;
    const bitCapInt nTiles = a.maxQPower >> TB;
    const int nG = a.k / K;
    for (bitCapInt t = blockIdx.x; t < nTiles; t += gridDim.x) {
        const bitCapInt base = t << TB;
        for (unsigned i = threadIdx.x; i < TILE; i += blockDim.x) {
            tile[qaSwz((int)i)] = sv[base + i];
        }
        __syncthreads();
        for (int gi = 0; gi < nG; ++gi) {
            const int g0 = gi * K;
            unsigned p[K];
#pragma unroll
            for (int g = 0; g < K; ++g) p[g] = (unsigned)a.tPow[g0 + g];
            for (unsigned o = threadIdx.x; o < (TILE >> K); o += blockDim.x) {
                unsigned r = o;
#pragma unroll
                for (int g = 0; g < K; ++g) r = ((r & ~(p[g] - 1u)) << 1u) | (r & (p[g] - 1u));
                cplx<R> v[NS];
#pragma unroll
                for (int s = 0; s < NS; ++s) {
                    unsigned off = r;
#pragma unroll
                    for (int g = 0; g < K; ++g) {
                        if (s & (1 << g)) off |= p[g];
                    }
                    v[s] = tile[qaSwz((int)off)];
                }
#pragma unroll
                for (int g = 0; g < K; ++g) {
                    const cplx<R> m0 = a.m[4 * (g0 + g)], m1 = a.m[4 * (g0 + g) + 1],
                                  m2 = a.m[4 * (g0 + g) + 2], m3 = a.m[4 * (g0 + g) + 3];
#pragma unroll
                    for (int s = 0; s < NS; ++s) {
                        if (s & (1 << g)) continue;
                        const int hb = s | (1 << g);
                        const cplx<R> x = v[s], y = v[hb];
                        v[s] = m0 * x + m1 * y;
                        v[hb] = m2 * x + m3 * y;
                    }
                }
#pragma unroll
                for (int s = 0; s < NS; ++s) {
                    unsigned off = r;
#pragma unroll
                    for (int g = 0; g < K; ++g) {
                        if (s & (1 << g)) off |= p[g];
                    }
                    tile[qaSwz((int)off)] = v[s];
                }
            }
            __syncthreads();
        }
        for (unsigned i = threadIdx.x; i < TILE; i += blockDim.x) {
            sv[base + i] = tile[qaSwz((int)i)];
        }
        __syncthreads();
    }
}

template <typename R>
void launchMtrx1qBatchLds(cplx<R>* sv, const BatchLdsArgs<R>& a, hipStream_t stream)
{
    const bitCapInt nTiles = a.maxQPower >> qaLdsTileBits<R>();
    const int grid = (int)std::min<bitCapInt>(nTiles, (bitCapInt)QA_MAX_BLOCKS);
    hipLaunchKernelGGL((k_mtrx_batch_lds<R>), dim3(grid), dim3(QA_BLOCK), 0, stream, sv, a);
}

// LDS-tiled two-qubit batch: each workgroup stages one tile, then applies
// every 4x4 (pair bits inside the tile) at LDS speed — one global RMW pass
// for a whole disjoint two-qubit layer segment.
template <typename R> __global__ void k_mtrx2q_batch_lds(cplx<R>* sv, Batch2qLdsArgs<R> a)
{
    constexpr int TB = qaLdsTileBits<R>();
    __shared__ cplx<R> tile[1u << TB];
    constexpr unsigned TILE = 1u << TB;
    const bitCapInt nTiles = a.maxQPower >> TB;
    for (bitCapInt t = blockIdx.x; t < nTiles; t += gridDim.x) {
        const bitCapInt base = t << TB;
        for (unsigned i = threadIdx.x; i < TILE; i += blockDim.x) {
            tile[i] = sv[base + i];
        }
        __syncthreads();
        for (int g = 0; g < a.k; ++g) {
            const unsigned lo = (unsigned)a.p1[g];
            const unsigned hi = (unsigned)a.p2[g];
            const cplx<R>* m = &a.m[16 * g];
            for (unsigned j = threadIdx.x; j < (TILE >> 2); j += blockDim.x) {
                // expand j around lo then hi (lo < hi)
                unsigned i0 = ((j & ~(lo - 1u)) << 1u) | (j & (lo - 1u));
                i0 = ((i0 & ~(hi - 1u)) << 1u) | (i0 & (hi - 1u));
                const unsigned i1 = i0 | lo;
                const unsigned i2 = i0 | hi;
                const unsigned i3 = i0 | lo | hi;
                const cplx<R> v0 = tile[i0], v1 = tile[i1], v2 = tile[i2], v3 = tile[i3];
                tile[i0] = m[0] * v0 + m[1] * v1 + m[2] * v2 + m[3] * v3;
                tile[i1] = m[4] * v0 + m[5] * v1 + m[6] * v2 + m[7] * v3;
                tile[i2] = m[8] * v0 + m[9] * v1 + m[10] * v2 + m[11] * v3;
                tile[i3] = m[12] * v0 + m[13] * v1 + m[14] * v2 + m[15] * v3;
            }
            __syncthreads();
        }
        for (unsigned i = threadIdx.x; i < TILE; i += blockDim.x) {
            sv[base + i] = tile[i];
        }
        __syncthreads();
    }
}

template <typename R>
void launchMtrx2qBatchLds(cplx<R>* sv, const Batch2qLdsArgs<R>& a, hipStream_t stream)
{
    const bitCapInt nTiles = a.maxQPower >> qaLdsTileBits<R>();
    const int grid = (int)std::min<bitCapInt>(nTiles, (bitCapInt)QA_MAX_BLOCKS);
    hipLaunchKernelGGL((k_mtrx2q_batch_lds<R>), dim3(grid), dim3(QA_BLOCK), 0, stream, sv, a);
}

template <typename R> __global__ void k_mtrx_2q(cplx<R>* sv, Gate4x4Args<R> a)
{
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt j = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; j < a.maxI; j += stride) {
        bitCapInt i0 = ((j & ~(a.p1 - 1u)) << 1u) | (j & (a.p1 - 1u));
        i0 = ((i0 & ~(a.p2 - 1u)) << 1u) | (i0 & (a.p2 - 1u));
        const bitCapInt i1 = i0 | a.p1;
        const bitCapInt i2 = i0 | a.p2;
        const bitCapInt i3 = i0 | a.p1 | a.p2;
        const cplx<R> v0 = sv[i0], v1 = sv[i1], v2 = sv[i2], v3 = sv[i3];
        sv[i0] = a.m[0] * v0 + a.m[1] * v1 + a.m[2] * v2 + a.m[3] * v3;
        sv[i1] = a.m[4] * v0 + a.m[5] * v1 + a.m[6] * v2 + a.m[7] * v3;
        sv[i2] = a.m[8] * v0 + a.m[9] * v1 + a.m[10] * v2 + a.m[11] * v3;
        sv[i3] = a.m[12] * v0 + a.m[13] * v1 + a.m[14] * v2 + a.m[15] * v3;
    }
}

template <typename R>
void launchMtrx2q(cplx<R>* sv, const Gate4x4Args<R>& a, hipStream_t stream)
{
    hipLaunchKernelGGL((k_mtrx_2q<R>), dim3(gridFor(a.maxI)), dim3(QA_BLOCK), 0, stream, sv, a);
}

// ---- batched disjoint CNOTs: one permutation pass per layer ----------------

template <typename R> __global__ void k_cnot_batch(cplx<R>* sv, CnotBatchArgs a)
{
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt i = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; i < a.maxI; i += stride) {
        bitCapInt xm = 0;
        for (int j = 0; j < a.k; ++j) {
            if (i & a.cPow[j]) xm |= a.tPow[j];
        }
        const bitCapInt p = i ^ xm;
        if (p <= i) continue; // partner handles the swap (or xm == 0)
        const cplx<R> t = sv[i];
        sv[i] = sv[p];
        sv[p] = t;
    }
}

// fp32 float4 variant: adjacent amplitude pairs share xm when neither a
// control nor a target sits at bit 0
__global__ void k_cnot_batch_v(cplx<float>* sv, CnotBatchArgs a)
{
    float4* sv4 = reinterpret_cast<float4*>(sv);
    const bitCapInt half = a.maxI >> 1u;
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt q = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; q < half; q += stride) {
        const bitCapInt i = q << 1u;
        bitCapInt xm = 0;
        for (int j = 0; j < a.k; ++j) {
            if (i & a.cPow[j]) xm |= a.tPow[j];
        }
        const bitCapInt p = i ^ xm;
        if (p <= i) continue;
        const bitCapInt qi = i >> 1u, qp = p >> 1u;
        const float4 t = sv4[qi];
        sv4[qi] = sv4[qp];
        sv4[qp] = t;
    }
}

template <typename R> __global__ void k_cphase_pairs(cplx<R>* sv, CPhasePairsArgs a)
{
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt i = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; i < a.maxI; i += stride) {
        double th = 0;
        for (int j = 0; j < a.k; ++j) {
            if ((i & a.cPow[j]) && (i & a.tPow[j])) th += a.angle[j];
        }
        if (th == 0.0) continue;
        R s, c;
        devSinCos<R>((R)th, &s, &c);
        sv[i] = cplx<R>{ c, s } * sv[i];
    }
}

template <typename R>
void launchCPhasePairs(cplx<R>* sv, const CPhasePairsArgs& a, hipStream_t stream)
{
    hipLaunchKernelGGL(
        (k_cphase_pairs<R>), dim3(gridFor(a.maxI)), dim3(QA_BLOCK), 0, stream, sv, a);
}

template <typename R>
void launchCnotBatch(cplx<R>* sv, const CnotBatchArgs& a, hipStream_t stream)
{
    if constexpr (std::is_same_v<R, float>) {
        bool low = false;
        for (int j = 0; j < a.k; ++j) {
            if (a.cPow[j] < 2u || a.tPow[j] < 2u) low = true;
        }
        if (!low && (a.maxI & 1u) == 0u) {
            hipLaunchKernelGGL((k_cnot_batch_v), dim3(gridFor(a.maxI >> 1u)), dim3(QA_BLOCK), 0,
                stream, sv, a);
            return;
        }
    }
    hipLaunchKernelGGL((k_cnot_batch<R>), dim3(gridFor(a.maxI)), dim3(QA_BLOCK), 0, stream, sv, a);
}

template <typename R>
void launchMtrx1qBatch(cplx<R>* sv, const Batch1qArgs<R>& a, hipStream_t stream)
{
    const int grid = gridFor(a.maxI);
    if constexpr (std::is_same_v<R, float>) {
        if (a.tPow[0] >= 2u && a.maxI >= 2u && (a.maxI & 1u) == 0u && a.k <= 4) {
            const int gridv = gridFor(a.maxI >> 1u);
            switch (a.k) {
            case 2:
                hipLaunchKernelGGL((k_mtrx_batch_v<2>), dim3(gridv), dim3(QA_BLOCK), 0, stream, sv, a);
                return;
            case 3:
                hipLaunchKernelGGL((k_mtrx_batch_v<3>), dim3(gridv), dim3(QA_BLOCK), 0, stream, sv, a);
                return;
            case 4:
                hipLaunchKernelGGL((k_mtrx_batch_v<4>), dim3(gridv), dim3(QA_BLOCK), 0, stream, sv, a);
                return;
            }
        }
    }
    switch (a.k) {
    case 2:
        hipLaunchKernelGGL((k_mtrx_batch<R, 2>), dim3(grid), dim3(QA_BLOCK), 0, stream, sv, a);
        break;
    case 3:
        hipLaunchKernelGGL((k_mtrx_batch<R, 3>), dim3(grid), dim3(QA_BLOCK), 0, stream, sv, a);
        break;
    case 4:
        hipLaunchKernelGGL((k_mtrx_batch<R, 4>), dim3(grid), dim3(QA_BLOCK), 0, stream, sv, a);
        break;
    case 5:
        hipLaunchKernelGGL((k_mtrx_batch<R, 5>), dim3(grid), dim3(QA_BLOCK), 0, stream, sv, a);
        break;
    default:
        throw QrackError("launchMtrx1qBatch: k out of range");
    }
}

template <typename R> __global__ void k_phase_ramp_gen(cplx<R>* sv, bitCapInt maxI, RampArgs a)
{
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt j = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; j < maxI; j += stride) {
        bitCapInt i = j;
        if (a.condPow) {
            i = (((j & ~(a.condPow - 1u)) << 1u) | (j & (a.condPow - 1u))) | a.condPow;
        }
        uint64_t frac = (uint64_t)((i >> a.rampStart) & a.inPlaceRelMask);
        for (int k = 0; k < a.nScattered; ++k) {
            if (i & a.sPow[k]) frac += a.sWeight[k];
        }
        const R theta = (R)a.scale * (R)frac;
        R s, c;
        devSinCos<R>(theta, &s, &c);
        sv[i] = cplx<R>{ c, s } * sv[i];
    }
}

template <typename R>
void launchPhaseRampGeneral(cplx<R>* sv, bitCapInt maxQPower, const RampArgs& a, hipStream_t stream)
{
    const bitCapInt maxI = a.condPow ? (maxQPower >> 1u) : maxQPower;
    hipLaunchKernelGGL(
        (k_phase_ramp_gen<R>), dim3(gridFor(maxI)), dim3(QA_BLOCK), 0, stream, sv, maxI, a);
}

// ---- sampling / inner product / marginals --------------------------------------

template <typename R>
__global__ void k_chunk_sums(const cplx<R>* sv, bitCapInt chunkLen, double* sums)
{
    // one block per contiguous chunk
    const bitCapInt lo = (bitCapInt)blockIdx.x * chunkLen;
    double s = 0;
    for (bitCapInt i = lo + threadIdx.x; i < lo + chunkLen; i += blockDim.x) {
        s += (double)norm(sv[i]);
    }
    s = waveReduceSum(s);
    __shared__ double waveSums[QA_BLOCK / 64];
    const int lane = threadIdx.x & 63;
    const int wid = threadIdx.x >> 6;
    if (lane == 0) waveSums[wid] = s;
    __syncthreads();
    if (wid == 0) {
        double t = (lane < QA_BLOCK / 64) ? waveSums[lane] : 0.0;
        t = waveReduceSum(t);
        if (lane == 0) sums[blockIdx.x] = t;
    }
}

template <typename R>
void launchChunkSums(
    const cplx<R>* sv, bitCapInt nChunks, bitCapInt chunkLen, double* sumsDev, hipStream_t stream)
{
    hipLaunchKernelGGL(
        (k_chunk_sums<R>), dim3((uint32_t)nChunks), dim3(QA_BLOCK), 0, stream, sv, chunkLen, sumsDev);
}

template <typename R>
__global__ void k_inner(
    const cplx<R>* a, const cplx<R>* b, bitCapInt maxI, double* partialsRe, double* partialsIm)
{
    double re = 0, im = 0;
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    for (bitCapInt i = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; i < maxI; i += stride) {
        const cplx<R> x = a[i];
        const cplx<R> y = b[i];
        // conj(b) * a
        re += (double)(y.re * x.re + y.im * x.im);
        im += (double)(y.re * x.im - y.im * x.re);
    }
    re = waveReduceSum(re);
    im = waveReduceSum(im);
    __shared__ double wr[QA_BLOCK / 64];
    __shared__ double wi2[QA_BLOCK / 64];
    const int lane = threadIdx.x & 63;
    const int wid = threadIdx.x >> 6;
    if (lane == 0) {
        wr[wid] = re;
        wi2[wid] = im;
    }
    __syncthreads();
    if (wid == 0) {
        double tr = (lane < QA_BLOCK / 64) ? wr[lane] : 0.0;
        double ti = (lane < QA_BLOCK / 64) ? wi2[lane] : 0.0;
        tr = waveReduceSum(tr);
        ti = waveReduceSum(ti);
        if (lane == 0) {
            partialsRe[blockIdx.x] = tr;
            partialsIm[blockIdx.x] = ti;
        }
    }
}

template <typename R>
int launchInner(const cplx<R>* a, const cplx<R>* b, bitCapInt maxI, double* partialsRe,
    double* partialsIm, hipStream_t stream)
{
    const int grid = gridForReduce(maxI);
    hipLaunchKernelGGL(
        (k_inner<R>), dim3(grid), dim3(QA_BLOCK), 0, stream, a, b, maxI, partialsRe, partialsIm);
    return grid;
}

template <typename R, bool USE_LDS>
__global__ void k_part_probs(
    const cplx<R>* sv, bitCapInt maxQPower, bitLenInt start, bitLenInt length, double* probs)
{
    const bitCapInt lenMask = (ONE_BCI << length) - 1u;
    const bitCapInt stride = (bitCapInt)gridDim.x * blockDim.x;
    if constexpr (USE_LDS) {
        extern __shared__ __attribute__((aligned(16))) char smemRaw[];
        double* hist = reinterpret_cast<double*>(smemRaw);
        const bitCapInt nBins = ONE_BCI << length;
        for (bitCapInt p = threadIdx.x; p < nBins; p += blockDim.x) hist[p] = 0;
        __syncthreads();
        for (bitCapInt i = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; i < maxQPower;
             i += stride) {
            const bitCapInt p = (i >> start) & lenMask;
            atomicAdd(&hist[p], (double)norm(sv[i]));
        }
        __syncthreads();
        for (bitCapInt p = threadIdx.x; p < nBins; p += blockDim.x) {
            if (hist[p] != 0.0) atomicAdd(&probs[p], hist[p]);
        }
    } else {
        for (bitCapInt i = (bitCapInt)blockIdx.x * blockDim.x + threadIdx.x; i < maxQPower;
             i += stride) {
            const bitCapInt p = (i >> start) & lenMask;
            atomicAdd(&probs[p], (double)norm(sv[i]));
        }
    }
}

template <typename R>
void launchPartProbs(const cplx<R>* sv, bitCapInt maxQPower, bitLenInt start, bitLenInt length,
    double* probsDev, hipStream_t stream)
{
    const int grid = gridFor(maxQPower);
    const bitCapInt nBins = ONE_BCI << length;
    if (nBins <= 2048u) {
        hipLaunchKernelGGL((k_part_probs<R, true>), dim3(grid), dim3(QA_BLOCK),
            (uint32_t)(nBins * sizeof(double)), stream, sv, maxQPower, start, length, probsDev);
    } else {
        hipLaunchKernelGGL((k_part_probs<R, false>), dim3(grid), dim3(QA_BLOCK), 0, stream, sv,
            maxQPower, start, length, probsDev);
    }
}

// ---- explicit instantiations ---------------------------------------------------

#define QA_INSTANTIATE(R)                                                                           \
    template void launchApply2x2<R>(cplx<R>*, const GateArgs<R>&, hipStream_t);                     \
    template void launchUniformlyControlled<R>(                                                     \
        cplx<R>*, bitCapInt, bitCapInt, const bitCapInt*, int, const cplx<R>*, hipStream_t);        \
    template void launchXMask<R>(cplx<R>*, bitCapInt, bitCapInt, hipStream_t);                      \
    template void launchParityPhase<R>(cplx<R>*, bitCapInt, bitCapInt, cplx<R>, cplx<R>, hipStream_t); \
    template int launchReduce<R>(const cplx<R>*, const ReduceArgs&, int, double*, hipStream_t);     \
    template int launchArgMax<R>(const cplx<R>*, bitCapInt, double*, bitCapInt*, hipStream_t);      \
    template void launchNormalize<R>(cplx<R>*, bitCapInt, cplx<R>, R, hipStream_t);                 \
    template void launchApplyM<R>(cplx<R>*, bitCapInt, bitCapInt, bitCapInt, cplx<R>, hipStream_t); \
    template void launchApplyParity<R>(cplx<R>*, bitCapInt, bitCapInt, bool, cplx<R>, hipStream_t); \
    template void launchCompose<R>(                                                                 \
        const cplx<R>*, const cplx<R>*, cplx<R>*, bitCapInt, bitLenInt, bitLenInt, hipStream_t);    \
    template void launchDisposeSlice<R>(const cplx<R>*, cplx<R>*, bitCapInt, bitLenInt, bitLenInt,  \
        bitCapInt, cplx<R>, hipStream_t);                                                           \
    template void launchGatherPart<R>(const cplx<R>*, cplx<R>*, bitCapInt, bitLenInt, bitLenInt,    \
        bitCapInt, cplx<R>, hipStream_t);                                                           \
    template void launchAllocateExpand<R>(                                                          \
        const cplx<R>*, cplx<R>*, bitCapInt, bitLenInt, bitLenInt, hipStream_t);                    \
    template void launchShuffleSwap<R>(cplx<R>*, cplx<R>*, bitCapInt, hipStream_t);                 \
    template void launchPermute<R>(const cplx<R>*, cplx<R>*, const PermArgs&, hipStream_t);         \
    template void launchPhaseFlipIfLess<R>(                                                         \
        cplx<R>*, bitCapInt, bitCapInt, bitLenInt, bitCapInt, bitCapInt, hipStream_t);              \
    template void launchChunkSums<R>(const cplx<R>*, bitCapInt, bitCapInt, double*, hipStream_t);   \
    template int launchInner<R>(                                                                    \
        const cplx<R>*, const cplx<R>*, bitCapInt, double*, double*, hipStream_t);                  \
    template void launchPartProbs<R>(                                                               \
        const cplx<R>*, bitCapInt, bitLenInt, bitLenInt, double*, hipStream_t);                     \
    template void launchPhaseRamp<R>(cplx<R>*, bitCapInt, bitLenInt, bitLenInt, bitCapInt, double, hipStream_t);\
    template void launchPhaseRampGeneral<R>(cplx<R>*, bitCapInt, const RampArgs&, hipStream_t);    \
    template void launchQftColumn2<R>(cplx<R>*, bitCapInt, bitLenInt, bitLenInt, bitCapInt,        \
        bitCapInt, int, bool, hipStream_t);                                                         \
    template void launchQftColumn3<R>(cplx<R>*, bitCapInt, bitLenInt, bitLenInt, bitCapInt,        \
        bitCapInt, bitCapInt, int, bool, hipStream_t);                                              \
    template void launchQftColumnK<R>(cplx<R>*, bitCapInt, bitLenInt, bitLenInt, int,               \
        const bitCapInt*, int, bool, hipStream_t);                                                  \
    template void launchQftColumn<R>(                                                               \
        cplx<R>*, bitCapInt, bitLenInt, bitLenInt, bitCapInt, int, bool, hipStream_t);              \
    template void launchMtrx1qBatch<R>(cplx<R>*, const Batch1qArgs<R>&, hipStream_t);                               \
    template void launchCnotBatch<R>(cplx<R>*, const CnotBatchArgs&, hipStream_t);                              \
    template void launchMtrx1qBatchLds<R>(cplx<R>*, const BatchLdsArgs<R>&, hipStream_t);                \
    template void launchMtrx2qBatchLds<R>(cplx<R>*, const Batch2qLdsArgs<R>&, hipStream_t);                \
    template void launchMtrx2q<R>(cplx<R>*, const Gate4x4Args<R>&, hipStream_t);                              \
    template void launchCPhasePairs<R>(cplx<R>*, const CPhasePairsArgs&, hipStream_t);                               \
    template void launchMtrx2qPair2<R>(cplx<R>*, const Gate4x4Pair2Args<R>&, hipStream_t);         \
    template void launchQftLowLds<R>(cplx<R>*, bitCapInt, int, int, int, bool, hipStream_t);        \
    template void launchQftMidLds<R>(cplx<R>*, bitCapInt, int, int, int, bool, hipStream_t);        \
    template void launchQftColumn2General<R>(cplx<R>*, bitCapInt, bitCapInt, bitCapInt,            \
        const RampArgs&, double, double, bool, hipStream_t);                                        \
    template void launchQftColumnTopRange<R>(cplx<R>*, bitCapInt, const RampArgs&, double, bool,    \
        bitCapInt, bitCapInt, const cplx<R>*, bool, hipStream_t);                                   \
    template void launchQftColumnGeneral<R>(                                                        \
        cplx<R>*, bitCapInt, bitCapInt, const RampArgs&, double, bool, hipStream_t);

QA_INSTANTIATE(float)
QA_INSTANTIATE(double)

} // namespace qrack_amd
