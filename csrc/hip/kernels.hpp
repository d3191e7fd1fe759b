// qrack_amd — HIP kernel launch API (host-visible declarations).
//
// Capability parity target: the reference GPU kernel inventory
// (/root/reference/src/common/qengine.cl 74 kernels + qheader_alu.cl;
// SURVEY.md §2.2). Fresh CDNA4 design: wave64 blocks of 256, grid-stride
// loops capped near 8 blocks/CU (256 CUs), float4/double2-vectorized
// amplitude streaming, single-pass two-stage reductions (wave shuffle +
// LDS + per-block partials), and one uniform-opcode permutation kernel for
// the whole ALU family instead of 26 near-identical kernels.
#pragma once

#include "../common/types.hpp"

#include <hip/hip_runtime.h>

namespace qrack_amd {

constexpr int QA_MAX_SKIP_POWERS = 16;
// grid-size cap for gate/reduction launches; partials buffers must hold
// this many (A/B-swept on the 30q QFT: 32768 = 104.4 ms < 16384 = 106.1 <
// 8192 = 108.0 < 2048 = 112.1 — more blocks shrink the grid-stride tails
// and balance the 8 XCDs)
constexpr int QA_REDUCE_MAX_BLOCKS = 32768;

// by-value argument block for gate kernels
template <typename R> struct GateArgs {
    cplx<R> m[4];
    bitCapInt offset1;
    bitCapInt offset2;
    bitCapInt qPowers[QA_MAX_SKIP_POWERS]; // sorted ascending
    int nPowers;
    bitCapInt maxI; // iterations (pairs)
};

enum class PermOp : int {
    INC = 0,
    INCDECC,
    INCS,
    MUL,
    DIV,
    MULMODN,
    IMULMODN,
    POWMODN,
    HASH,
    LDA,
    ADC,
    SBC,
    ROL,
    INCBCD,
};

struct PermArgs {
    int op;
    bitCapInt maxI;                        // iterated indices (sub-space size)
    bitCapInt qPowers[QA_MAX_SKIP_POWERS]; // skip powers for the iteration
    int nPowers;
    bitCapInt controlMask; // OR into every iterated index
    bitLenInt start;       // in/out register start
    bitLenInt length;      // register length
    bitCapInt operand;     // toAdd / toMul / base
    bitCapInt modN;
    bitCapInt carryMask;    // carry / overflow qubit power
    bitLenInt start2;       // out register start (modN ops) / value start (LDA)
    bitLenInt length2;      // out register length
    bitCapInt extra;        // carry-in addend etc.
    const unsigned char* table; // device pointer for HASH/LDA/ADC/SBC
    int tableBytes;             // bytes per table entry
};

enum class ReduceOp : int {
    NORM_ALL = 0,   // sum |amp|^2
    PROB_BITSET,    // sum |amp|^2 where (i & mask) == mask (single-bit use)
    PROB_MASK,      // sum |amp|^2 where (i & mask) == perm
    PROB_PARITY,    // sum |amp|^2 where parity(i & mask)
    EXP_PERM,       // sum value(i) * |amp|^2 (factorized bit values)
    EXP_PERM_SQ,    // sum value(i)^2 * |amp|^2
    NORM_FLOOR,     // sum |amp|^2 with amplitude floor (UpdateRunningNorm)
};

constexpr int QA_REDUCE_MAX_BITS = 32;

struct ReduceArgs {
    bitCapInt maxI;
    bitCapInt mask;
    bitCapInt perm;
    double offset;       // EXP_PERM value offset
    double normThresh;   // NORM_FLOOR
    // EXP_PERM inputs ride IN the kernarg segment (<= 32 listed bits; the
    // engine falls back to the generic host path beyond that) — no device
    // staging buffers, so no stream-ordered-allocator exposure on the
    // expectation/variance query path
    bitLenInt bitsArr[QA_REDUCE_MAX_BITS];
    bitCapInt permsArr[QA_REDUCE_MAX_BITS];
    int nBits;
};

// ---- launches (all asynchronous on `stream`) -------------------------------

template <typename R>
void launchApply2x2(cplx<R>* sv, const GateArgs<R>& a, hipStream_t stream);

template <typename R>
void launchUniformlyControlled(cplx<R>* sv, bitCapInt maxI, bitCapInt targetPower,
    const bitCapInt* ctrlPowersDev, int nCtrls, const cplx<R>* mtrxsDev, hipStream_t stream);

template <typename R>
void launchXMask(cplx<R>* sv, bitCapInt maxQPower, bitCapInt mask, hipStream_t stream);

template <typename R>
void launchParityPhase(
    cplx<R>* sv, bitCapInt maxQPower, bitCapInt mask, cplx<R> even, cplx<R> odd, hipStream_t stream);

// two-stage reduction: stage 1 fills partials[gridSize]; returns grid size used
template <typename R>
int launchReduce(const cplx<R>* sv, const ReduceArgs& a, int op, double* partialsDev,
    hipStream_t stream);

// argmax of |amp|^2: fills (value,index) pairs per block
template <typename R>
int launchArgMax(const cplx<R>* sv, bitCapInt maxI, double* valsDev, bitCapInt* idxDev,
    hipStream_t stream);

template <typename R>
void launchNormalize(
    cplx<R>* sv, bitCapInt maxQPower, cplx<R> factor, R normThresh, hipStream_t stream);

template <typename R>
void launchApplyM(cplx<R>* sv, bitCapInt maxQPower, bitCapInt mask, bitCapInt result, cplx<R> nrm,
    hipStream_t stream);

template <typename R>
void launchApplyParity(cplx<R>* sv, bitCapInt maxQPower, bitCapInt mask, bool odd, cplx<R> nrm,
    hipStream_t stream);

template <typename R>
void launchCompose(const cplx<R>* a, const cplx<R>* b, cplx<R>* out, bitCapInt nMaxQPower,
    bitLenInt start, bitLenInt oQubits, hipStream_t stream);

// out[r] = in[low | (slicePerm << start) | high] * scale  (Dispose/Decompose)
template <typename R>
void launchDisposeSlice(const cplx<R>* in, cplx<R>* out, bitCapInt remPower, bitLenInt start,
    bitLenInt length, bitCapInt slicePerm, cplx<R> scale, hipStream_t stream);

// dest[p] = in[low(rem) | (p << start) | high(rem)] * scale  (Decompose part)
template <typename R>
void launchGatherPart(const cplx<R>* in, cplx<R>* dest, bitCapInt partPower, bitLenInt start,
    bitLenInt length, bitCapInt remIndex, cplx<R> scale, hipStream_t stream);

// out[i] = (mid(i)==0) ? in[collapse(i)] : 0   (Allocate)
template <typename R>
void launchAllocateExpand(const cplx<R>* in, cplx<R>* out, bitCapInt nMaxQPower, bitLenInt start,
    bitLenInt length, hipStream_t stream);

// swap upper half of a with lower half of b
template <typename R>
void launchShuffleSwap(cplx<R>* aHigh, cplx<R>* bLow, bitCapInt half, hipStream_t stream);

template <typename R>
void launchPermute(const cplx<R>* sv, cplx<R>* nsv, const PermArgs& a, hipStream_t stream);

template <typename R>
void launchPhaseFlipIfLess(cplx<R>* sv, bitCapInt maxQPower, bitCapInt greaterPerm, bitLenInt start,
    bitCapInt regMask, bitCapInt flagMask, hipStream_t stream);

int reduceGridSize(bitCapInt n);

// fused diagonal phase ramp: amp *= exp(i*scale*((x>>rampStart) mod
// 2^rampBits)) where (condPower==0) || (x & condPower) — one pass replaces
// rampBits (controlled-)phase gates (the QFT column ladder).
template <typename R>
void launchPhaseRamp(cplx<R>* sv, bitCapInt maxQPower, bitLenInt rampStart, bitLenInt rampBits,
    bitCapInt condPower, double scale, hipStream_t stream);

// generalized diagonal ramp with relocated bits: frac(i) =
// ((i >> rampStart) & inPlaceRelMask) + sum_k (i & sPow[k] ? sWeight[k] : 0);
// amp *= exp(i*scale*frac) where (condPow==0)||(i & condPow). Supports the
// distributed pager's lazily-permuted qubit maps (<= 8 relocated bits).
struct RampArgs {
    bitLenInt rampStart;
    bitCapInt inPlaceRelMask;
    int nScattered;
    bitCapInt sPow[8];
    uint64_t sWeight[8];
    bitCapInt condPow;
    double scale;
};

template <typename R>
void launchPhaseRampGeneral(cplx<R>* sv, bitCapInt maxQPower, const RampArgs& a, hipStream_t stream);

// TWO fused QFT columns in one state pass (see kernels.hip k_qft_col2):
// columns (col, col-1) as 4-amplitude orbits — ONE sincos per orbit drives
// both ramps (cross term is a constant ±i factor, lower ramp = f0^2)
template <typename R>
void launchQftColumn2(cplx<R>* sv, bitCapInt maxQPower, bitLenInt rampStart, bitLenInt col,
    bitCapInt tHi, bitCapInt tLo, int sign, bool pre, hipStream_t stream);

// ladder group width: uniform K=4 fp32 / K=3 fp64 — the engine must align
// low-ladder lengths to a multiple of this
template <typename R> constexpr int qaLowLadderK() { return sizeof(R) == 4 ? 4 : 3; }

// one-pass low-bit QFT ladder: columns colMax..0 applied inside contiguous
// 2^tb-amplitude LDS tiles (start-0 registers only; colMax+1 must be a
// multiple of qaLowLadderK<R>())
template <typename R>
void launchQftLowLds(
    cplx<R>* sv, bitCapInt maxQPower, int tb, int colMax, int sign, bool pre, hipStream_t stream);

// nCols mid-range QFT columns per pass through a 2D LDS tile (64 contiguous
// low amps x 2^nCols column-bit combos; start-0 registers, colLo >= 6)
template <typename R>
void launchQftMidLds(cplx<R>* sv, bitCapInt maxQPower, int colLo, int nCols, int sign, bool pre,
    hipStream_t stream);

// generic K-column fused QFT pass (2^K-amplitude orbits; K=4 instantiated)
template <typename R>
void launchQftColumnK(cplx<R>* sv, bitCapInt maxQPower, bitLenInt rampStart, bitLenInt col,
    int kCols, const bitCapInt* tPows, int sign, bool pre, hipStream_t stream);

// THREE fused QFT columns per pass (8-amplitude orbits; see k_qft_col3)
template <typename R>
void launchQftColumn3(cplx<R>* sv, bitCapInt maxQPower, bitLenInt rampStart, bitLenInt col,
    bitCapInt tHi, bitCapInt tMid, bitCapInt tLo, int sign, bool pre, hipStream_t stream);

// generalized fused QFT column: H on tPow + the (possibly relocated-bit)
// ramp of RampArgs in ONE pass; phase0 is a constant phase folded onto the
// target=1 side (distributed pager meta-page scalar)
template <typename R>
void launchQftColumnGeneral(cplx<R>* sv, bitCapInt maxQPower, bitCapInt tPow, const RampArgs& a,
    double phase0, bool pre, hipStream_t stream);

// TWO generalized columns per pass (arbitrary local target slots,
// relocated ramp bits, per-column meta scalars) — the distributed pager's
// local-ladder pass-halving (see k_qft_col2_gen)
template <typename R>
void launchQftColumn2General(cplx<R>* sv, bitCapInt maxQPower, bitCapInt tPowHi, bitCapInt tPowLo,
    const RampArgs& a, double phase0Hi, double phase0Lo, bool pre, hipStream_t stream);

// Ranged top-target fused QFT column for the distributed pager's PIPELINED
// page exchange: processes pair rows r in [itLo, itHi) of the H on the TOP
// local qubit (tPow = maxQPower/2), reading ONE side of each pair straight
// from the RCCL receive buffer `recvSrc` (indexed r - itLo) instead of the
// state vector — the received chunk is consumed in place, no staging copy.
// recvIsLow: true when the received chunk is the pair's low (target=0) side
// (the HIGH page of an exchange receives the low side). Launched per chunk
// on the caller-supplied stream (torch's current stream) so NCCL chunk c+1
// overlaps with chunk c's compute. Replaces the host-staged ShuffleBuffers
// swap of the reference (opencl.cpp:254-264) with exchange+apply fusion.
template <typename R>
void launchQftColumnTopRange(cplx<R>* sv, bitCapInt maxQPower, const RampArgs& a, double phase0,
    bool pre, bitCapInt itLo, bitCapInt itHi, const cplx<R>* recvSrc, bool recvIsLow,
    hipStream_t stream);

// batched independent single-qubit gates: k distinct-target 2x2s applied in
// ONE full-state pass (2^k-amplitude orbits in registers). The memory-bound
// fusion win: k passes -> 1. fp32 supports k in [2,5], fp64 [2,4].
constexpr int QA_MAX_BATCH_1Q = 5;

template <typename R> struct Batch1qArgs {
    cplx<R> m[4 * QA_MAX_BATCH_1Q];
    bitCapInt tPow[QA_MAX_BATCH_1Q]; // sorted ascending, distinct
    int k;
    bitCapInt maxI; // orbit count = maxQPower >> k
};

template <typename R>
void launchMtrx1qBatch(cplx<R>* sv, const Batch1qArgs<R>& a, hipStream_t stream);

// LDS-tiled low-bit batch: gates whose targets all sit below QA_LDS_TILE_BITS
// apply inside a shared-memory tile — up to 12 gates in ONE global RMW pass
// (the register-orbit kernels degrade on low-bit targets; LDS does not).
// 32 KB LDS tiles both ways: 4096 fp32 amps / 2048 fp64 amps (a 64 KB fp64
// tile would sit exactly on the per-workgroup LDS limit)
template <typename R> constexpr int qaLdsTileBits() { return sizeof(R) == 4 ? 12 : 11; }
// register-orbit group width inside the LDS gate-batch kernel; the engine
// pads batches to a multiple of this with identity gates
template <typename R> constexpr int qaLdsBatchK() { return sizeof(R) == 4 ? 4 : 3; }
constexpr int QA_LDS_TILE_BITS = 12; // fp32 value; use qaLdsTileBits<R>()
constexpr int QA_MAX_BATCH_LDS = 12;

template <typename R> struct BatchLdsArgs {
    cplx<R> m[4 * QA_MAX_BATCH_LDS];
    bitCapInt tPow[QA_MAX_BATCH_LDS]; // sorted ascending, all < 2^QA_LDS_TILE_BITS
    int k;
    bitCapInt maxQPower;
};

template <typename R>
void launchMtrx1qBatchLds(cplx<R>* sv, const BatchLdsArgs<R>& a, hipStream_t stream);

// LDS-tiled batch of disjoint TWO-qubit 4x4 gates (all four qubit bits below
// the tile): a whole fsim/SU(4) layer segment in ONE global pass.
constexpr int QA_MAX_BATCH_2Q = 6;

template <typename R> struct Batch2qLdsArgs {
    cplx<R> m[16 * QA_MAX_BATCH_2Q]; // row-major 4x4 per pair, basis |q2 q1>
    bitCapInt p1[QA_MAX_BATCH_2Q];   // pow2(q1) (low bit of the pair index)
    bitCapInt p2[QA_MAX_BATCH_2Q];   // pow2(q2) (high bit)
    int k;
    bitCapInt maxQPower;
};

template <typename R>
void launchMtrx2qBatchLds(cplx<R>* sv, const Batch2qLdsArgs<R>& a, hipStream_t stream);

// general global two-qubit 4x4 apply (register orbit of 4 amplitudes; any
// qubit positions) — the reference decomposes SU(4) into gate strings, this
// build applies it in ONE pass.
template <typename R> struct Gate4x4Args {
    cplx<R> m[16]; // row-major, basis |q2 q1>
    bitCapInt p1;  // pow2(min qubit)
    bitCapInt p2;  // pow2(max qubit)
    bitCapInt maxI; // maxQPower >> 2
};

template <typename R>
void launchMtrx2q(cplx<R>* sv, const Gate4x4Args<R>& a, hipStream_t stream);

// TWO disjoint 4x4s in ONE full-state pass (16-amplitude orbits — the same
// register-orbit budget the 4-column QFT kernel proved runs at full HBM
// bandwidth). Matrices row-major in |q2 q1> basis with powers pre-sorted
// per gate (pA1<pA2, pB1<pB2); all four bit positions distinct.
template <typename R> struct Gate4x4Pair2Args {
    cplx<R> mA[16];
    cplx<R> mB[16];
    bitCapInt pA1, pA2, pB1, pB2;
    bitCapInt sorted4[4]; // the four powers ascending (orbit insertion)
    bitCapInt orbits;     // maxQPower >> 4
};

template <typename R>
void launchMtrx2qPair2(cplx<R>* sv, const Gate4x4Pair2Args<R>& a, hipStream_t stream);

// batched disjoint CNOTs: a whole layer of k control/target pairs (no qubit
// repeated) applied as ONE in-place permutation pass — amp[i] swaps with
// amp[i ^ xm(i)] where xm(i) XORs tPow[j] for every set control bit. One
// full-state RMW replaces k half-state passes.
constexpr int QA_MAX_BATCH_CNOT = 16;

struct CnotBatchArgs {
    bitCapInt cPow[QA_MAX_BATCH_CNOT];
    bitCapInt tPow[QA_MAX_BATCH_CNOT];
    int k;
    bitCapInt maxI; // full state size
};

template <typename R>
void launchCnotBatch(cplx<R>* sv, const CnotBatchArgs& a, hipStream_t stream);

// batched controlled-phase pairs: amp[i] *= exp(i * sum_j angle_j) over pairs
// with both bits set — one diagonal pass applies a whole CZ/CPhase layer
// (pairs may share qubits; diagonal ops commute).
struct CPhasePairsArgs {
    bitCapInt cPow[QA_MAX_BATCH_CNOT];
    bitCapInt tPow[QA_MAX_BATCH_CNOT];
    double angle[QA_MAX_BATCH_CNOT];
    int k;
    bitCapInt maxI;
};

template <typename R>
void launchCPhasePairs(cplx<R>* sv, const CPhasePairsArgs& a, hipStream_t stream);

// fully fused QFT column (H + the column's phase ramp in one pass);
// pre=false: QFT order (H then ramp), pre=true: IQFT order (ramp then H)
template <typename R>
void launchQftColumn(cplx<R>* sv, bitCapInt maxQPower, bitLenInt rampStart, bitLenInt col,
    bitCapInt tPow, int sign, bool pre, hipStream_t stream);

// contiguous per-chunk |amp|^2 sums (inverse-CDF sampling support):
// sums[c] = sum over [c*chunkLen, (c+1)*chunkLen)
template <typename R>
void launchChunkSums(
    const cplx<R>* sv, bitCapInt nChunks, bitCapInt chunkLen, double* sumsDev, hipStream_t stream);

// complex inner product <b|a>: per-block partial (re, im) pairs
template <typename R>
int launchInner(const cplx<R>* a, const cplx<R>* b, bitCapInt maxI, double* partialsRe,
    double* partialsIm, hipStream_t stream);

// marginal probabilities of the [start, start+length) register:
// probs[p] += |amp|^2 over all amplitudes with register == p  (probsDev zeroed
// by caller). Single pass, LDS histogram when 2^length <= 2048.
template <typename R>
void launchPartProbs(const cplx<R>* sv, bitCapInt maxQPower, bitLenInt start, bitLenInt length,
    double* probsDev, hipStream_t stream);

} // namespace qrack_amd
