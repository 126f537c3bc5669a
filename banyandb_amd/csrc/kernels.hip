// banyandb_amd/csrc/kernels.hip — MI355X (gfx950) decode+fold kernels and
// the C-ABI session around them.
//
// Hot path replaced (apache/skywalking-banyandb):
//   blockCursor.loadData -> block.mustReadFrom (banyand/measure/block.go:818,
//   324) -> BytesToInt64List (pkg/encoding/int_list.go:57) ->
//   BatchAggregation.Consume fold (pkg/query/vectorized/measure/
//   aggregation.go:310-351,464-486; pkg/query/aggregation/function.go).
//
// Design (CDNA4-first, not a translation):
//  * one 64-lane WAVE per (series,block) — 1.2M blocks at the benchmark
//    configs is ~150 blocks per wave slot at full occupancy; no intra-block
//    sync, no LDS;
//  * the varint stream (zigzag, 7-bit groups LSB-first, terminator = high
//    bit clear — pkg/encoding/int.go:81-148) decodes in 256-byte quad
//    windows: four bytes per lane from one u32 load, __ballot terminator
//    masks, up to four 1-2-byte values decoded per lane with one 32-bit
//    DPP scan per window (shfl compiles to ds_bpermute — an LDS op — so
//    the scans use DPP row_shr + v_readlane instead); wider varints fall
//    back to a 64-byte ballot window;
//  * SUM/COUNT need no prefix reconstruction at all: for delta streams
//    sum(v_i, i=r0..r1) = nsel*first + sum_j w_j * d_j with closed-form
//    weights, so the kernel does a weighted fold of raw deltas (exact mod
//    2^64 — Go int64 addition wraps identically);
//  * MIN/MAX (and float64 restore) reconstruct values with a wave-wide
//    inclusive scan (__shfl_up, 6 steps) — the reference's serial carry
//    chain (delta.go:45-117) becomes a parallel prefix sum;
//  * float64 columns stay in the decimal-int domain in-kernel (monotone
//    restore, float.go:69-102 applied at finalize); the mantissa sum is
//    carried per block into a double;
//  * per-block partials land in dense per-group accumulators with one
//    device-scope atomic set per wave per block (Guideline 12).
#include "../../include/bydb_gpu.h"

#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <string>
#include <vector>

#define WAVE 64

// exact powers of ten for per-block decimal-exponent rescale
__device__ __constant__ int64_t c_pow10i[19] = {
    1LL, 10LL, 100LL, 1000LL, 10000LL, 100000LL, 1000000LL, 10000000LL,
    100000000LL, 1000000000LL, 10000000000LL, 100000000000LL,
    1000000000000LL, 10000000000000LL, 100000000000000LL,
    1000000000000000LL, 10000000000000000LL, 100000000000000000LL,
    1000000000000000000LL};

// Plain (non-dictionary) tag columns — the >256-distinct-values fallback
// (column.go:266-278, dictionary.go:58) — are zstd-compressed on disk; the
// host normalizes them at part registration (encode.cpp) into a sidecar
// arena, flagged in the stored descriptor by the top offset bit.
#define TAG_SIDECAR_BIT (1ull << 63)
#define PLAIN_BM_WORDS 128  // bitmap words per block (8192 rows max)
bool bydb_normalize_plain_tag(const uint8_t *src, uint64_t src_len,
                              uint64_t nrows, std::vector<uint8_t> &out);
bool bydb_normalize_dict_tag(const uint8_t *src, uint64_t src_len,
                             bool *needed, std::vector<uint8_t> &out);
bool bydb_normalize_plain_field(const uint8_t *src, uint64_t src_len,
                                uint64_t nrows, std::vector<uint8_t> &out);

// ---------------- device helpers ----------------

__device__ __forceinline__ int64_t zz_dec(uint64_t u) {
    return (int64_t)(u >> 1) ^ -(int64_t)(u & 1);
}

__device__ __forceinline__ uint64_t lanemask_lt(int lane) {
    return (lane == 0) ? 0ull : (~0ull >> (64 - lane));
}

// DPP row-shift: pure-VALU lane shift within 16-lane rows (bound_ctrl
// zero-fills lanes whose source is outside the row).  __shfl_up compiles
// to ds_bpermute (an LDS op, ~40-cycle latency); a 6-step shfl scan is a
// dependent LDS chain that dominates the scan kernels' wait time, so the
// scans below use 4 DPP row_shr steps + v_readlane cross-row fixup —
// zero LDS traffic.
template <int CTRL>
__device__ __forceinline__ uint32_t dpp_mov32(uint32_t v) {
    return (uint32_t)__builtin_amdgcn_update_dpp(0, (int)v, CTRL, 0xF, 0xF,
                                                 true);
}

__device__ __forceinline__ uint64_t readlane64(uint64_t v, int l) {
    uint32_t lo = (uint32_t)__builtin_amdgcn_readlane((int)(uint32_t)v, l);
    uint32_t hi = (uint32_t)__builtin_amdgcn_readlane((int)(uint32_t)(v >> 32),
                                                      l);
    return ((uint64_t)hi << 32) | lo;
}

// wave-wide inclusive scan of a u64 (wrapping adds)
__device__ __forceinline__ uint64_t wave_incl_scan(uint64_t v, int lane) {
#define BYDB_SCAN_STEP64(C)                                                  \
    {                                                                        \
        uint64_t t = ((uint64_t)dpp_mov32<C>((uint32_t)(v >> 32)) << 32) |   \
                     dpp_mov32<C>((uint32_t)v);                              \
        v += t;                                                              \
    }
    BYDB_SCAN_STEP64(0x111)   // row_shr:1
    BYDB_SCAN_STEP64(0x112)   // row_shr:2
    BYDB_SCAN_STEP64(0x114)   // row_shr:4
    BYDB_SCAN_STEP64(0x118)   // row_shr:8
#undef BYDB_SCAN_STEP64
    uint64_t s15 = readlane64(v, 15);
    uint64_t s31 = readlane64(v, 31);
    uint64_t s47 = readlane64(v, 47);
    // branchless: nested ternaries here compile to exec-mask branch
    // cascades; mask-ANDs keep it pure VALU
    int r = lane >> 4;
    uint64_t m1 = (uint64_t)-(int64_t)(r >= 1);
    uint64_t m2 = (uint64_t)-(int64_t)(r >= 2);
    uint64_t m3 = (uint64_t)-(int64_t)(r >= 3);
    return v + (s15 & m1) + (s31 & m2) + (s47 & m3);
}

// reductions, DPP-style like the scans: 4 row_shr steps leave each
// 16-lane row's total at its last lane; v_readlane combines the rows.
// All lanes return the wave total (as the shfl_xor butterfly did).
__device__ __forceinline__ uint64_t wave_reduce_add(uint64_t v) {
#define BYDB_RSTEP64(C)                                                      \
    {                                                                        \
        uint64_t t = ((uint64_t)dpp_mov32<C>((uint32_t)(v >> 32)) << 32) |   \
                     dpp_mov32<C>((uint32_t)v);                              \
        v += t;                                                              \
    }
    BYDB_RSTEP64(0x111) BYDB_RSTEP64(0x112)
    BYDB_RSTEP64(0x114) BYDB_RSTEP64(0x118)
#undef BYDB_RSTEP64
    return readlane64(v, 15) + readlane64(v, 31) + readlane64(v, 47) +
           readlane64(v, 63);
}

__device__ __forceinline__ double wave_reduce_addf(double v) {
    uint64_t u;
    __builtin_memcpy(&u, &v, 8);
#define BYDB_RSTEPF(C)                                                       \
    {                                                                        \
        uint64_t tb = ((uint64_t)dpp_mov32<C>((uint32_t)(u >> 32)) << 32) |  \
                      dpp_mov32<C>((uint32_t)u);                             \
        double t;                                                            \
        __builtin_memcpy(&t, &tb, 8);                                        \
        v += t;                                                              \
        __builtin_memcpy(&u, &v, 8);                                         \
    }
    BYDB_RSTEPF(0x111) BYDB_RSTEPF(0x112)
    BYDB_RSTEPF(0x114) BYDB_RSTEPF(0x118)
#undef BYDB_RSTEPF
    double a, b, c, d;
    uint64_t x;
    x = readlane64(u, 15); __builtin_memcpy(&a, &x, 8);
    x = readlane64(u, 31); __builtin_memcpy(&b, &x, 8);
    x = readlane64(u, 47); __builtin_memcpy(&c, &x, 8);
    x = readlane64(u, 63); __builtin_memcpy(&d, &x, 8);
    return a + b + c + d;
}

__device__ __forceinline__ int64_t wave_reduce_min(int64_t v) {
    // row_shr moves are zero-filled out of row; feed the identity by
    // selecting against the shifted VALIDITY instead: use bound_ctrl 0's
    // zero fill and compare via a mask-free trick — shift lanes carry a
    // zero, so compare only where the source lane exists (lane%16 >= off)
    const int lane15 = (int)(threadIdx.x & 15);
#define BYDB_RMIN(C, OFF)                                                    \
    {                                                                        \
        int64_t t = (int64_t)(((uint64_t)dpp_mov32<C>((uint32_t)((uint64_t)v >> 32)) << 32) | \
                              dpp_mov32<C>((uint32_t)(uint64_t)v));          \
        if (lane15 >= OFF && t < v) v = t;                                   \
    }
    BYDB_RMIN(0x111, 1) BYDB_RMIN(0x112, 2)
    BYDB_RMIN(0x114, 4) BYDB_RMIN(0x118, 8)
#undef BYDB_RMIN
    int64_t a = (int64_t)readlane64((uint64_t)v, 15);
    int64_t b = (int64_t)readlane64((uint64_t)v, 31);
    int64_t c = (int64_t)readlane64((uint64_t)v, 47);
    int64_t d = (int64_t)readlane64((uint64_t)v, 63);
    int64_t m = a < b ? a : b;
    m = c < m ? c : m;
    return d < m ? d : m;
}

__device__ __forceinline__ int64_t wave_reduce_max(int64_t v) {
    const int lane15 = (int)(threadIdx.x & 15);
#define BYDB_RMAX(C, OFF)                                                    \
    {                                                                        \
        int64_t t = (int64_t)(((uint64_t)dpp_mov32<C>((uint32_t)((uint64_t)v >> 32)) << 32) | \
                              dpp_mov32<C>((uint32_t)(uint64_t)v));          \
        if (lane15 >= OFF && t > v) v = t;                                   \
    }
    BYDB_RMAX(0x111, 1) BYDB_RMAX(0x112, 2)
    BYDB_RMAX(0x114, 4) BYDB_RMAX(0x118, 8)
#undef BYDB_RMAX
    int64_t a = (int64_t)readlane64((uint64_t)v, 15);
    int64_t b = (int64_t)readlane64((uint64_t)v, 31);
    int64_t c = (int64_t)readlane64((uint64_t)v, 47);
    int64_t d = (int64_t)readlane64((uint64_t)v, 63);
    int64_t m = a > b ? a : b;
    m = c > m ? c : m;
    return d > m ? d : m;
}

// Decode one varint serially starting at p (used for wave-uniform headers:
// the DeltaConst stride and the delta-of-delta d1 — every lane runs the
// same loop on the same bytes, so the result is wave-uniform).
__device__ __forceinline__ int64_t decode_one_varint(const uint8_t *p, int *len) {
    uint8_t c = p[0];
    if (c < 0x80) { *len = 1; return zz_dec(c); }
    uint64_t u = (uint64_t)(c & 0x7f);
    int i = 1;
    unsigned shift = 0;
    while (c >= 0x80 && i <= 10) {
        c = p[i++];
        shift += 7;
        u |= (uint64_t)(c & 0x7f) << shift;
    }
    *len = i;
    return zz_dec(u);
}

// Triangular number T(m) = m(m+1)/2 for m <= 8192 (fits u32 domain widely).
__device__ __forceinline__ uint64_t tri(uint64_t m) { return m * (m + 1) / 2; }

struct DevErr {
    unsigned code;      // first error code seen (atomicCAS)
    unsigned block_lo;  // low 32 bits of the offending block index
};

enum {
    DERR_NONE = 0,
    DERR_BAD_STREAM = 1,    // no terminator in a 64-byte window
    DERR_BAD_ENC = 2,       // unsupported encode type in kernel
    DERR_DESC_TS = 3,       // descending timestamps (reference never writes them)
    DERR_GROUP_RANGE = 4,   // block group_code >= configured n_groups
    DERR_F64_SUM_OVF = 6,   // float64 block mantissa sum would exceed int64
    DERR_EXP_MISMATCH = 5,  // float64 block's decimal exponent differs from
                            // the configured session exponent (per-block
                            // rescale is a later row; silent mixing would
                            // corrupt sums and make min/max incomparable)
};

__device__ __forceinline__ void dev_set_err(DevErr *e, unsigned code, uint64_t bi) {
    if (atomicCAS(&e->code, DERR_NONE, code) == DERR_NONE)
        e->block_lo = (unsigned)bi;
}

// ---------------- per-block decode primitives ----------------

// ---- dense all-1-byte delta fold ----
// When the stream length equals the delta count, every varint is one byte
// (int.go:84-88), so the stream IS the zigzag byte array and the weighted
// fold becomes dense byte math: per dword, SWAR zigzag
// (d = (b>>1) ^ (-(b&1) as 0x00/0xff)) turns 4 bytes into 4 signed i8
// lanes, and v_dot4_i32_i8 folds them against {1,1,1,1} / {0,1,2,3}
// weights.  Per-lane accumulators stay in int32 for a whole block
// (<= 128 bytes/lane * 63 * 8191 < 2^31).
__device__ __forceinline__ int32_t dot4_i8(uint32_t a, uint32_t b, int32_t c) {
#if defined(__gfx950__) || defined(__AMDGCN__)
    return __builtin_amdgcn_sdot4((int)a, (int)b, c, false);
#else
    int32_t r = c;
    for (int k = 0; k < 4; k++)
        r += (int32_t)(int8_t)(a >> (8 * k)) * (int32_t)(int8_t)(b >> (8 * k));
    return r;
#endif
}

__device__ __forceinline__ uint32_t swar_zigzag(uint32_t w) {
    uint32_t mag = (w >> 1) & 0x7f7f7f7fu;
    uint32_t sgn = w & 0x01010101u;
    return mag ^ (sgn * 0xffu);  // per-byte signed i8 delta
}

// Fold Sum(d_j) and Sum(bidx*d_j) over stream bytes bidx in [lo, hi)
// (delta j = bidx+1).  Interior dwords go through the dense dot4 loop;
// the <=6 boundary bytes are folded by lane 0.
__device__ void dense_region(const uint8_t *stream, int64_t lo, int64_t hi,
                             int lane, int64_t *out_sum_d, int64_t *out_sum_jd) {
    *out_sum_d = 0;
    *out_sum_jd = 0;
    if (hi <= lo) return;
    // aligned interior: first dword-aligned byte >= lo, last < hi
    uintptr_t s0 = (uintptr_t)stream;
    int64_t alo = (int64_t)(((s0 + (uint64_t)lo + 3) & ~(uintptr_t)3) - s0);
    int64_t ahi = (int64_t)(((s0 + (uint64_t)hi) & ~(uintptr_t)3) - s0);
    int32_t acc_d = 0;
    int64_t acc_jd = 0;   // 64-bit: boundary folds may exceed int32 ranges? no — keep 64 for safety on jd only at fold
    int32_t acc_jd32 = 0;
    if (ahi > alo) {
        const uint32_t *p = (const uint32_t *)(s0 + (uint64_t)alo);
        int64_t ndw = (ahi - alo) / 4;
        int64_t q = lane;
        // software-pipelined 4-deep batches: fold batch N while batch
        // N+1's four loads are in flight (the plain 4-deep loop leaves
        // the wave parked on its own loads ~79% of cycles)
        if (q + 192 < ndw) {
            uint32_t w0 = p[q], w1 = p[q + 64], w2 = p[q + 128],
                     w3 = p[q + 192];
            for (;;) {
                int64_t qn = q + 256;
                bool more = qn + 192 < ndw;
                uint32_t n0 = 0, n1 = 0, n2 = 0, n3 = 0;
                if (more) {
                    n0 = p[qn];
                    n1 = p[qn + 64];
                    n2 = p[qn + 128];
                    n3 = p[qn + 192];
                }
                int32_t b0 = (int32_t)(alo + 4 * q);
                uint32_t s_0 = swar_zigzag(w0), s_1 = swar_zigzag(w1),
                         s_2 = swar_zigzag(w2), s_3 = swar_zigzag(w3);
                int32_t t0 = dot4_i8(s_0, 0x01010101u, 0);
                int32_t t1 = dot4_i8(s_1, 0x01010101u, 0);
                int32_t t2 = dot4_i8(s_2, 0x01010101u, 0);
                int32_t t3 = dot4_i8(s_3, 0x01010101u, 0);
                acc_d += t0 + t1 + t2 + t3;
                acc_jd32 += dot4_i8(s_0, 0x03020100u, 0) +
                            dot4_i8(s_1, 0x03020100u, 0) +
                            dot4_i8(s_2, 0x03020100u, 0) +
                            dot4_i8(s_3, 0x03020100u, 0);
                acc_jd32 += b0 * t0 + (b0 + 256) * t1 + (b0 + 512) * t2 +
                            (b0 + 768) * t3;
                q = qn;
                if (!more) break;
                w0 = n0;
                w1 = n1;
                w2 = n2;
                w3 = n3;
            }
        }
        for (; q < ndw; q += 64) {
            uint32_t w0 = p[q];
            uint32_t s_0 = swar_zigzag(w0);
            int32_t t0 = dot4_i8(s_0, 0x01010101u, 0);
            int32_t k0 = dot4_i8(s_0, 0x03020100u, 0);
            acc_d += t0;
            acc_jd32 += k0 + (int32_t)(alo + 4 * q) * t0;
        }
    }
    acc_jd += acc_jd32;
    // boundary bytes on lane 0: with an interior, the head [lo, alo) and
    // tail [ahi, hi); without one, the whole [lo, hi)
    if (lane == 0) {
        int64_t h_lo = lo, h_hi, t_lo, t_hi;
        if (ahi > alo) { h_hi = alo; t_lo = ahi; t_hi = hi; }
        else { h_hi = hi; t_lo = 0; t_hi = 0; }
        for (int64_t b = h_lo; b < h_hi; b++) {
            uint32_t c = stream[b];
            int32_t d = (int32_t)(c >> 1) ^ -(int32_t)(c & 1);
            acc_d += d;
            acc_jd += (int64_t)b * d;
        }
        for (int64_t b = t_lo; b < t_hi; b++) {
            uint32_t c = stream[b];
            int32_t d = (int32_t)(c >> 1) ^ -(int32_t)(c & 1);
            acc_d += d;
            acc_jd += (int64_t)b * d;
        }
    }
    *out_sum_d = (int64_t)acc_d;
    *out_sum_jd = acc_jd;
}

// wave-wide inclusive scan of an int32 (per-window lane sums)
// row-masked DPP: rows outside ROWMASK receive `old` = 0, so `v += ...`
// is a no-op there — the canonical GCN cross-row scan combine.
template <int CTRL, int ROWMASK>
__device__ __forceinline__ uint32_t dpp_mov32_rm(uint32_t v) {
    return (uint32_t)__builtin_amdgcn_update_dpp(0, (int)v, CTRL, ROWMASK,
                                                 0xF, true);
}

__device__ __forceinline__ int32_t wave_incl_scan32(int32_t v, int lane) {
    v += (int32_t)dpp_mov32<0x111>((uint32_t)v);
    v += (int32_t)dpp_mov32<0x112>((uint32_t)v);
    v += (int32_t)dpp_mov32<0x114>((uint32_t)v);
    v += (int32_t)dpp_mov32<0x118>((uint32_t)v);
    // cross-row combine via row broadcasts instead of three readlanes +
    // masked adds: row_bcast15 adds lane 15 into row 1 and lane 47 into
    // row 3; row_bcast31 then adds lane 31 (= rows 0+1 total) into rows
    // 2 and 3 — two DPP ops, no SALU round trips
    v += (int32_t)dpp_mov32_rm<0x142, 0xa>((uint32_t)v);  // ROW_BCAST15
    v += (int32_t)dpp_mov32_rm<0x143, 0xc>((uint32_t)v);  // ROW_BCAST31
    (void)lane;
    return v;
}

// Dense min/max over an all-1-byte delta stream: fold min/max of the
// values v_j for delta bytes b in [b_lo, b_hi) (v after applying byte b),
// with base_val = v at delta index b_lo (carry-in).  256-byte windows,
// SWAR zigzag, per-window lane-sum scan — replaces the latency-chained
// 64-byte ballot scan for the dominant stream shape.
__device__ void dense_minmax(const uint8_t *stream, int64_t b_lo, int64_t b_hi,
                             int64_t base_val, int lane, int64_t *out_mn,
                             int64_t *out_mx) {
    int64_t lmn = INT64_MAX, lmx = INT64_MIN;
    if (b_hi > b_lo) {
        const uint32_t *wp =
            (const uint32_t *)((uintptr_t)(stream + b_lo) & ~(uintptr_t)3);
        int64_t wbase = (int64_t)((uintptr_t)wp - (uintptr_t)stream);
        uint64_t carry = (uint64_t)base_val;  // wrap-safe (Go int64 wraps)
        uint32_t w = wp[lane];
        while (wbase < b_hi) {
            uint32_t w_nxt = wp[64 + lane];   // next window, one ahead
            uint32_t sb = swar_zigzag(w);
            // mask bytes outside [b_lo, b_hi): zero their deltas and skip fold
            int64_t g0 = wbase + 4 * lane;
            uint32_t keep = 0;
#pragma unroll
            for (int k = 0; k < 4; k++) {
                int64_t g = g0 + k;
                if (g >= b_lo && g < b_hi) keep |= 0xffu << (8 * k);
            }
            sb &= keep;
            int32_t t = dot4_i8(sb, 0x01010101u, 0);
            int32_t incl = wave_incl_scan32(t, lane);
            uint64_t lane_base = carry + (uint64_t)(int64_t)(incl - t);
            int32_t cum = 0;
#pragma unroll
            for (int k = 0; k < 4; k++) {
                int32_t d = (int32_t)(int8_t)(sb >> (8 * k));
                cum += d;
                int64_t g = g0 + k;
                if (g >= b_lo && g < b_hi) {
                    int64_t v = (int64_t)(lane_base + (uint64_t)(int64_t)cum);
                    lmn = v < lmn ? v : lmn;
                    lmx = v > lmx ? v : lmx;
                }
            }
            carry += (uint64_t)(int64_t)__builtin_amdgcn_readlane(incl, 63);
            wp += 64;
            wbase += 256;
            w = w_nxt;
        }
    }
    *out_mn = wave_reduce_min(lmn);
    *out_mx = wave_reduce_max(lmx);
}

// Weighted fold over an all-1-byte delta stream:
//   sum_{j=1..jend} d_j * w(j),  w(j) = nsel (j<=r0) | r1-j+1 (else)
// via two dense regions (delta j lives at byte j-1):
//   region1 bytes [0, min(r0,jend)):          contributes nsel * S1d
//   region2 bytes [min(r0,jend), jend):       contributes (r1+1)*S2d - S2jd'
// where S2jd' uses delta index j = bidx+1.
__device__ uint64_t dense_delta_weighted(const uint8_t *stream,
                                         int64_t n_deltas, int64_t r0,
                                         int64_t r1, int lane) {
    int64_t jend = r1 < n_deltas ? r1 : n_deltas;
    if (jend < 1) return 0;
    uint64_t nsel = (uint64_t)(r1 - r0 + 1);
    int64_t b1 = r0 < jend ? r0 : jend;  // deltas 1..b1 have weight nsel
    int64_t s1d, s1jd, s2d, s2jd;
    dense_region(stream, 0, b1, lane, &s1d, &s1jd);
    dense_region(stream, b1, jend, lane, &s2d, &s2jd);
    // j = bidx + 1  =>  sum j*d = sum bidx*d + sum d
    uint64_t sum_jd2 = (uint64_t)s2jd + (uint64_t)s2d;
    uint64_t acc = nsel * (uint64_t)s1d;
    acc += (uint64_t)(r1 + 1) * (uint64_t)s2d - sum_jd2;
    return acc;
}

// Fast weighted delta fold: 256-byte windows (one dword per lane) with an
// all-1-byte fast case — the dominant shape for delta-encoded telemetry
// (zigzag deltas < 64 encode to one byte, int.go:84-88).  For a window
// fully inside the linear-weight region (j > r0, tail within [.., jend]),
//   sum d_j * (r1 - j + 1) = (r1+1) * S1 - S2,  S1 = sum d_j, S2 = sum j*d_j
// so the per-value work is 32-bit byte math folded into two lane
// accumulators.  Windows with multi-byte varints or boundary weights fall
// back to one 64-byte ballot step (exact same semantics), then resume.
// The next window's dwords are loaded before the current one is processed
// (the fast path's advance is a compile-time +256), hiding HBM latency.
__device__ uint64_t fold_delta_weighted_fast(const uint8_t *stream,
                                             int64_t n_deltas, int64_t r0,
                                             int64_t r1, int lane, DevErr *derr,
                                             uint64_t bi) {
    uint64_t gen_acc = 0;   // generic-weight accumulator (boundary/slow)
    uint64_t s1 = 0;        // sum of d_j  (linear region)
    uint64_t s2 = 0;        // sum of j*d_j (linear region)
    uint64_t pos = 0;
    int64_t j = 1;
    int64_t jend = r1 < n_deltas ? r1 : n_deltas;
    uint64_t nsel = (uint64_t)(r1 - r0 + 1);
    // 256-byte windows at 4-byte-aligned absolute addresses; `off` is the
    // unconsumed-prefix length (bytes before the next value start).
    const uint32_t *wp =
        (const uint32_t *)((uintptr_t)(stream + pos) & ~(uintptr_t)3);
    unsigned off = (unsigned)((uintptr_t)(stream + pos) - (uintptr_t)wp);
    uint32_t w = wp[lane];
    while (j <= jend) {
        uint32_t w_next = wp[64 + lane];               // prefetch next window
        uint32_t valid = (lane == 0 && off) ? (0xFFFFFFFFu << (8 * off))
                                            : 0xFFFFFFFFu;
        uint32_t want = 0x80808080u & valid;
        bool all_term = ((~w) & want) == want;
        int64_t nvals_window = 256 - (int64_t)off;
        bool fast = __all(all_term) && (jend - j + 1) >= nvals_window;
        if (fast) {
            bool linear = j > r0;   // whole window past the r0 boundary
            int32_t sum_d = 0, sum_kd = 0;
#pragma unroll
            for (int k = 0; k < 4; k++) {
                int32_t bidx = 4 * lane + k;
                uint32_t b = (w >> (8 * k)) & 0xffu;
                int32_t d = (int32_t)(b >> 1) ^ -(int32_t)(b & 1);
                if ((unsigned)bidx < off) d = 0;       // lane-0 masked prefix
                if (linear) {
                    sum_d += d;
                    sum_kd += (bidx - (int32_t)off) * d;
                } else {
                    int64_t myj = j + (bidx - (int64_t)off);
                    uint64_t wt = myj <= r0 ? nsel : (uint64_t)(r1 - myj + 1);
                    if ((unsigned)bidx >= off)
                        gen_acc += (uint64_t)(int64_t)d * wt;
                }
            }
            if (linear) {
                s1 += (uint64_t)(int64_t)sum_d;
                s2 += (uint64_t)j * (uint64_t)(int64_t)sum_d +
                      (uint64_t)(int64_t)sum_kd;
            }
            pos += (uint64_t)nvals_window;
            j += nvals_window;
            wp += 64;
            off = 0;
            w = w_next;
            continue;
        }
        // slow step: one 64-byte ballot window from pos (multi-byte
        // varints and/or the stream tail) — identical semantics to v1
        {
            uint8_t b = stream[pos + (uint64_t)lane];
            uint64_t emask = __ballot(b < 0x80);
            if (emask == 0) { dev_set_err(derr, DERR_BAD_STREAM, bi); break; }
            int rank = __popcll(emask & lanemask_lt(lane));
            int64_t myj = j + rank;
            bool is_term = (b < 0x80) && (myj <= jend);
            // <=2-byte window fast case: decode via the left neighbour,
            // no backward byte loop, no 64-bit bookkeeping
            uint64_t cont2 = ~emask;
            if ((cont2 & (cont2 << 1)) == 0 &&
                (int64_t)__popcll(emask) <= jend - j + 1) {
                uint32_t prevb = (uint32_t)__shfl_up((int)b, 1);
                bool is2b = lane > 0 && (prevb & 0x80u);
                uint32_t u2 = is2b ? ((prevb & 0x7fu) | ((uint32_t)b << 7))
                                   : (uint32_t)b;
                int32_t d32 = (int32_t)(u2 >> 1) ^ -(int32_t)(u2 & 1);
                if (b < 0x80) {
                    uint64_t wt = myj <= r0 ? nsel : (uint64_t)(r1 - myj + 1);
                    gen_acc += (uint64_t)(int64_t)d32 * wt;
                }
                int ll2 = 63 - __clzll(emask);
                int nterm2 = __popcll(emask);
                if (j + nterm2 > jend) { j = jend + 1; break; }
                j += nterm2;
                pos += (uint64_t)(ll2 + 1);
                wp = (const uint32_t *)((uintptr_t)(stream + pos) & ~(uintptr_t)3);
                off = (unsigned)((uintptr_t)(stream + pos) - (uintptr_t)wp);
                w = wp[lane];
                continue;
            }
            int64_t d;
            if (emask == ~0ull) {
                d = zz_dec(b);
            } else if (is_term) {
                uint64_t below = emask & lanemask_lt(lane);
                int start = below ? (64 - __clzll(below)) : 0;
                uint64_t u = 0;
                unsigned sh = 0;
                for (int i = start; i < lane; ++i) {
                    u |= (uint64_t)(stream[pos + (uint64_t)i] & 0x7f) << sh;
                    sh += 7;
                }
                u |= (uint64_t)b << sh;
                d = zz_dec(u);
            } else {
                d = 0;
            }
            if (is_term) {
                uint64_t wt = myj <= r0 ? nsel : (uint64_t)(r1 - myj + 1);
                gen_acc += (uint64_t)d * wt;
            }
            int nterm = __popcll(emask);
            if (j + nterm > jend) { j = jend + 1; break; }
            j += nterm;
            pos += (uint64_t)(64 - __clzll(emask));
            wp = (const uint32_t *)((uintptr_t)(stream + pos) & ~(uintptr_t)3);
            off = (unsigned)((uintptr_t)(stream + pos) - (uintptr_t)wp);
            w = wp[lane];
        }
    }
    // fold the linear region: sum d_j*(r1-j+1) = (r1+1)*S1 - S2 (mod 2^64)
    return gen_acc + (uint64_t)(r1 + 1) * s1 - s2;
}

// Weighted delta-of-delta fold: sum_{i=r0..r1} v_i with
//   v_i = first + i*d1 + sum_{k=2..i} d2_k * (i-k+1)
// => contribution of d2_k (k in [2, r1]):  W_k = T(r1-k+1) - T(max(k,r0)-k).
__device__ uint64_t fold_dod_weighted(const uint8_t *stream, int64_t n_deltas,
                                      int64_t r0, int64_t r1, int lane,
                                      DevErr *derr, uint64_t bi) {
    uint64_t acc = 0;
    uint64_t pos = 0;
    int64_t k = 2;                       // d2 indices run 2..n-1
    int64_t kend = r1 < n_deltas ? r1 : n_deltas;
    while (k <= kend) {
        uint8_t b = stream[pos + (uint64_t)lane];
        uint64_t emask = __ballot(b < 0x80);
        if (emask == 0) { dev_set_err(derr, DERR_BAD_STREAM, bi); return acc; }
        int rank = __popcll(emask & lanemask_lt(lane));
        int64_t myk = k + rank;
        bool is_term = (b < 0x80) && (myk <= kend);
        int64_t d;
        if (emask == ~0ull) {
            d = zz_dec(b);
        } else if (is_term) {
            uint64_t below = emask & lanemask_lt(lane);
            int start = below ? (64 - __clzll(below)) : 0;
            uint64_t u = 0;
            unsigned sh = 0;
            for (int i = start; i < lane; ++i) {
                u |= (uint64_t)(stream[pos + (uint64_t)i] & 0x7f) << sh;
                sh += 7;
            }
            u |= (uint64_t)b << sh;
            d = zz_dec(u);
        } else {
            d = 0;
        }
        if (is_term) {
            int64_t a = myk > r0 ? myk : r0;
            uint64_t w = tri((uint64_t)(r1 - myk + 1)) - tri((uint64_t)(a - myk));
            acc += (uint64_t)d * w;
        }
        int nterm = __popcll(emask);
        if (k + nterm > kend) break;
        k += nterm;
        pos += (uint64_t)(64 - __clzll(emask));
    }
    return acc;
}

// ---------------- tag predicate (dictionary codes) ----------------
// The reference evaluates tag-equality predicates on decoded tag values
// per row (vec scan/filter; tags are dictionary-encoded per block,
// column.go:266-278, dictionary.go:79-115).  Here a resolve prepass maps
// the predicate value to a per-block CODE mask by comparing it against the
// block's dictionary values, and the scan walks the block's bit-packed RLE
// code runs (dictionary.go:158-260) to test rows — codes only, the value
// bytes are never expanded per row.

struct PredBlock {
    uint64_t mask[4];      // bit c: dict code c equals the predicate value
    uint64_t rle_bit_off;  // absolute bit offset of packed RLE entries
    uint32_t nentries;     // packed entry count (2 per run: value, count)
    uint8_t width;         // bits per packed entry
    uint8_t active;        // block has a tag column on this slot
    uint8_t err;           // unparseable on device -> error
    uint8_t uniform;       // single run covers the whole block
    uint8_t plain;         // plain (non-dictionary) column: rle_bit_off is
                           // the per-row match-bitmap WORD offset instead
    uint8_t disabled;      // slot carries no predicate (empty value)
};

// MSB-first bit read at an arbitrary bit offset (reader.go:39-79 order);
// n <= 32, so 8 gathered bytes always cover shift+n.
__device__ __forceinline__ uint64_t rd_bits_be(const uint8_t *p,
                                               uint64_t bitpos, uint32_t n) {
    uint64_t byte = bitpos >> 3;
    uint32_t sh = (uint32_t)(bitpos & 7);
    uint64_t acc = 0;
#pragma unroll
    for (int i = 0; i < 8; i++) acc = (acc << 8) | p[byte + (uint64_t)i];
    return (acc >> (64 - sh - n)) & ((1ull << n) - 1);
}

// compress_block section header inside a dictionary stream: [0][u8 len]
// plain (<128 B, bytes.go:291-295) or [2][u32le len] raw — the framing the
// host normalization writes after decompressing a zstd'd section.  Returns
// the section start, or nullptr on an unparseable marker (zstd reaching
// the device unnormalized).
__device__ __forceinline__ const uint8_t *dict_section(const uint8_t *p,
                                                       const uint8_t *end,
                                                       uint64_t *len_out) {
    if (p >= end) return nullptr;
    if (*p == 0) {
        if (p + 2 > end) return nullptr;
        *len_out = p[1];
        return p + 2;
    }
    if (*p == 2) {
        if (p + 5 > end) return nullptr;
        *len_out = (uint64_t)p[1] | ((uint64_t)p[2] << 8) |
                   ((uint64_t)p[3] << 16) | ((uint64_t)p[4] << 24);
        return p + 5;
    }
    return nullptr;
}

struct PredWalk {
    const uint8_t *payload;
    const uint64_t *bm;        // plain columns: per-row match bitmap (O(1))
    uint64_t bit0;
    uint32_t nentries, width;
    uint64_t mask[4];
    uint32_t entry;            // next entry index (2 per run)
    int64_t run_lo, run_hi;    // rows [run_lo, run_hi) of the current run
    bool run_match;
};

__device__ __forceinline__ void pred_init(PredWalk *pw, const uint8_t *payload,
                                          const uint8_t *sidecar,
                                          const PredBlock *pb,
                                          const uint64_t *bm_arena) {
    // bit 63 of rle_bit_off: the dictionary stream was host-normalized
    // into the sidecar arena (zstd'd sections); offsets are sidecar-based
    pw->payload = (pb->rle_bit_off & TAG_SIDECAR_BIT) ? sidecar : payload;
    pw->bm = pb->plain ? bm_arena + pb->rle_bit_off : nullptr;
    pw->bit0 = pb->rle_bit_off & ~TAG_SIDECAR_BIT;
    pw->nentries = pb->nentries;
    pw->width = pb->width;
#pragma unroll
    for (int i = 0; i < 4; i++) pw->mask[i] = pb->mask[i];
    pw->entry = 0;
    pw->run_lo = 0;
    pw->run_hi = 0;
    pw->run_match = false;
}

__device__ __forceinline__ void pred_advance(PredWalk *pw) {
    if (pw->entry + 1 >= pw->nentries) {   // RLE exhausted: no-match tail
        pw->run_lo = pw->run_hi;
        pw->run_hi = INT64_MAX;
        pw->run_match = false;
        pw->entry = pw->nentries + 2;
        return;
    }
    uint64_t code = rd_bits_be(pw->payload, pw->bit0 + (uint64_t)pw->entry * pw->width,
                               pw->width);
    uint64_t cnt = rd_bits_be(pw->payload,
                              pw->bit0 + (uint64_t)(pw->entry + 1) * pw->width,
                              pw->width);
    pw->entry += 2;
    pw->run_lo = pw->run_hi;
    pw->run_hi += (int64_t)cnt;
    pw->run_match = (pw->mask[(code >> 6) & 3] >> (code & 63)) & 1;
}

// Assign `match` for each lane's row (rows ascend across calls).  `need`
// lanes get the run covering their row; the walker only moves forward.
__device__ __forceinline__ bool pred_match_rows(PredWalk *pw, int64_t row,
                                                bool need) {
    if (pw->bm) {   // plain column: stateless bitmap lookup, no wave sync
        if (!need) return false;
        return (pw->bm[row >> 6] >> (row & 63)) & 1;
    }
    bool match = false;
    while (true) {
        bool mine = need && row >= pw->run_lo && row < pw->run_hi;
        if (mine) { match = pw->run_match; need = false; }
        if (__all(!need)) break;
        pred_advance(pw);
    }
    return match;
}

// ---- per-row group-by on a dictionary tag ----
// The reference groups rows by the encoded tag-value key, materialising
// groups in first-seen order (computeKey/appendKeyComponent,
// vectorized/measure/aggregation.go:523, groupby.go:287-364).  Here the
// host supplies the group DOMAIN (value -> dense group id); a resolve
// prepass maps each block's dictionary codes to group ids — per-block
// uniform tags (entity tags, the common case) collapse to a single id and
// keep the fast fold paths; row-varying tags fold per RLE run.
struct GroupDomain {
    const uint8_t *blob;     // concatenated domain value bytes
    const uint64_t *offs;    // n+1 offsets into blob
    const uint64_t *hashes;  // open-addressing table: slot -> hash
    const uint32_t *gids;    // slot -> gid (0xFFFFFFFF empty)
    uint32_t table_size;     // power of two
    uint32_t n;
};

struct GroupBlock {
    uint32_t uniform_gid;   // gid when the whole block is one run;
                            // 0xFFFFFFFE = row-varying, 0xFFFFFFFF = no tag
    uint32_t map_off;       // offset into the code->gid arena (row-varying)
    uint64_t rle_bit_off;
    uint32_t nentries;
    uint8_t width;
    uint8_t err;
    uint8_t _p[2];
};

#define GID_NONE 0xFFFFFFFFu
#define GID_VARYING 0xFFFFFFFEu

// FNV-1a 64 over value bytes (hash choice is internal; equality is by
// bytes, mirroring the reference's "equality is on key bytes" note)
__device__ __host__ inline uint64_t fnv1a(const uint8_t *p, uint64_t n) {
    uint64_t h = 1469598103934665603ULL;
    for (uint64_t i = 0; i < n; i++) {
        h ^= p[i];
        h *= 1099511628211ULL;
    }
    return h;
}

__device__ inline uint32_t domain_lookup(const GroupDomain *d, const uint8_t *v,
                                         uint64_t len) {
    uint64_t h = fnv1a(v, len);
    uint32_t mask = d->table_size - 1;
    uint32_t slot = (uint32_t)h & mask;
    for (uint32_t probe = 0; probe <= mask; probe++) {
        uint32_t g = d->gids[slot];
        if (g == GID_NONE) return GID_NONE;
        if (d->hashes[slot] == h) {
            uint64_t lo = d->offs[g], hi = d->offs[g + 1];
            if (hi - lo == len) {
                bool eq = true;
                for (uint64_t k = 0; k < len; k++)
                    if (d->blob[lo + k] != v[k]) { eq = false; break; }
                if (eq) return g;
            }
        }
        slot = (slot + 1) & mask;
    }
    return GID_NONE;
}

// one thread per block: parse the tag dictionary, map codes to gids
__global__ void k_resolve_groups(const uint8_t *__restrict__ payload,
                                 const uint8_t *__restrict__ sidecar,
                                 const bydb_block_desc *__restrict__ blocks,
                                 int64_t n_blocks, int slot, GroupDomain dom,
                                 GroupBlock *__restrict__ out,
                                 uint16_t *__restrict__ map_arena,
                                 uint32_t map_base) {
    int64_t bi = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (bi >= n_blocks) return;
    const bydb_block_desc *bd = &blocks[bi];
    uint64_t toff = slot == 0 ? bd->tag_off : slot == 1 ? bd->tag2_off : bd->tag3_off;
    uint64_t tlen = slot == 0 ? bd->tag_len : slot == 1 ? bd->tag2_len : bd->tag3_len;
    GroupBlock gb;
    gb.uniform_gid = GID_NONE;
    gb.map_off = map_base + (uint32_t)(bi * 256);
    gb.rle_bit_off = 0; gb.nentries = 0; gb.width = 0; gb.err = 0;
    gb._p[0] = gb._p[1] = 0;
    if (tlen == 0) { out[bi] = gb; return; }
    const bool in_sidecar = (toff & TAG_SIDECAR_BIT) != 0;
    const uint8_t *base = in_sidecar ? sidecar : payload;
    const uint8_t *p = base + (toff & ~TAG_SIDECAR_BIT);
    const uint8_t *end = p + tlen;
    if (*p != BYDB_ENC_DICTIONARY) {
        if (*p == BYDB_ENC_PLAIN) {
            // plain (>256-distinct) column: per-row gids are resolved by
            // the follow-up wave kernel (k_resolve_plain_groups) into
            // (gid,count) run pairs; width 0xFF + nentries 0 marks
            // "pending" and the scan errors loudly if it stays that way
            gb.width = 0xFF;
            gb.uniform_gid = GID_VARYING;
            gb.nentries = 0;
            out[bi] = gb;
            return;
        }
        gb.err = 1;
        out[bi] = gb;
        return;
    }
    p++;
    uint64_t count = 0;
    unsigned sh = 0;
    while (p < end) {
        uint8_t c = *p++;
        count |= (uint64_t)(c & 0x7f) << sh;
        if (c < 0x80) break;
        sh += 7;
    }
    uint64_t ll = 0;
    const uint8_t *lens_blk = dict_section(p, end, &ll);
    if (!lens_blk) { gb.err = 1; out[bi] = gb; return; }
    p = lens_blk + ll;
    uint64_t vl = 0;
    const uint8_t *vals = dict_section(p, end, &vl);
    if (!vals) { gb.err = 1; out[bi] = gb; return; }
    p = vals + vl;
    uint8_t wt = lens_blk[0];
    uint32_t wbytes = wt == 0 ? 1 : wt == 1 ? 2 : wt == 2 ? 4 : 8;
    const uint8_t *lp = lens_blk + 1;
    uint64_t voff = 0;
    uint16_t *map = map_arena + gb.map_off;
    for (uint64_t v = 0; v < count && v < 256; v++) {
        uint64_t alen = 0;
        for (uint32_t b = 0; b < wbytes; b++) alen = (alen << 8) | lp[v * wbytes + b];
        uint32_t g = GID_NONE;
        if (alen > 0) {
            uint64_t vlen = alen - 1;
            g = domain_lookup(&dom, vals + voff, vlen);
            voff += vlen;
        }
        map[v] = g == GID_NONE ? 0xFFFFu : (uint16_t)g;
    }
    uint64_t bit0 = ((uint64_t)(p - base)) * 8;
    uint32_t nentries = (uint32_t)rd_bits_be(base, bit0, 32);
    uint32_t width = nentries ? (uint32_t)rd_bits_be(base, bit0 + 32, 8) : 0;
    gb.nentries = nentries;
    gb.width = (uint8_t)width;
    gb.rle_bit_off = (bit0 + 40) | (in_sidecar ? TAG_SIDECAR_BIT : 0);
    if (nentries == 2) {
        uint64_t code = rd_bits_be(base, bit0 + 40, width);
        uint64_t cnt = rd_bits_be(base, bit0 + 40 + width, width);
        if (cnt >= bd->count) {
            uint16_t m = map[code < 256 ? code : 0];
            gb.uniform_gid = m == 0xFFFFu ? GID_NONE : m;
            out[bi] = gb;
            return;
        }
    }
    gb.uniform_gid = GID_VARYING;
    out[bi] = gb;
}

// Per-row group-by on a PLAIN (>256-distinct) tag column — computeKey
// groups on any key column (groupby.go:287-364); the reference's plain
// fallback (column.go:266-278) stores a bytes block, host-normalized into
// the sidecar at part registration.  One wave per block: each row's value
// maps to its domain gid (equality by bytes, like appendKeyComponent);
// consecutive equal gids compress into (gid,count) u16 run pairs in the
// run arena, so the group merge walks them with the same cursor shape as
// dictionary RLE.  Single-run blocks collapse to a uniform gid and keep
// every fast fold path.  Nil and out-of-domain rows become gid 0xFFFF
// (dropped rows).
#define GRUN_ENTRIES_PER_STREAM (2 * (8192 + 1))
__global__ __launch_bounds__(WAVE) void k_resolve_plain_groups(
    const uint8_t *__restrict__ payload, const uint8_t *__restrict__ sidecar,
    const bydb_block_desc *__restrict__ blocks, int64_t n_blocks, int slot,
    GroupDomain dom, GroupBlock *__restrict__ out,
    uint16_t *__restrict__ gruns, uint64_t grun_cap_streams,
    uint32_t *__restrict__ ctr) {
    const int lane = threadIdx.x;
    for (int64_t bi = blockIdx.x; bi < n_blocks; bi += gridDim.x) {
        GroupBlock gb = out[bi];
        if (gb.width != 0xFF || gb.err) continue;
        const bydb_block_desc *bd = &blocks[bi];
        uint64_t toff = slot == 0 ? bd->tag_off
                                  : slot == 1 ? bd->tag2_off : bd->tag3_off;
        const uint8_t *p = (toff & TAG_SIDECAR_BIT)
                               ? sidecar + (toff & ~TAG_SIDECAR_BIT)
                               : payload + toff;
        p++;  // ENC_PLAIN
        uint64_t n = (uint64_t)p[0] | ((uint64_t)p[1] << 8) |
                     ((uint64_t)p[2] << 16) | ((uint64_t)p[3] << 24);
        p += 4;
        uint8_t wt = *p++;
        uint32_t wbytes = wt == 0 ? 1 : wt == 1 ? 2 : wt == 2 ? 4 : 8;
        const uint8_t *lens = p;
        const uint8_t *vals = lens + n * wbytes;
        if (n != (uint64_t)bd->count) {
            if (lane == 0) out[bi].err = 1;
            continue;
        }
        uint32_t slot_id = 0;
        if (lane == 0) slot_id = atomicAdd(ctr, 1u);
        slot_id = (uint32_t)__shfl((int)slot_id, 0);
        if (slot_id >= grun_cap_streams) {
            if (lane == 0) out[bi].err = 1;   // arena exhausted: loud
            continue;
        }
        uint16_t *rp = gruns + (uint64_t)slot_id * GRUN_ENTRIES_PER_STREAM;
        uint64_t carry_bytes = 0;
        uint32_t n_pairs = 0;
        int32_t cur_gid = -2;   // no open run yet
        uint32_t cur_cnt = 0;
        for (uint64_t base = 0; base < n; base += WAVE) {
            uint64_t row = base + (uint64_t)lane;
            uint64_t lp1 = 0;
            if (row < n)
                for (uint32_t b = 0; b < wbytes; b++)
                    lp1 = (lp1 << 8) | lens[row * wbytes + b];
            uint64_t vlen = lp1 ? lp1 - 1 : 0;
            uint64_t incl = wave_incl_scan(vlen, lane);
            uint64_t myoff = carry_bytes + incl - vlen;
            int32_t g = -1;     // nil or out-of-domain: dropped row
            if (row < n && lp1) {
                uint32_t gg = domain_lookup(&dom, vals + myoff, vlen);
                g = gg == GID_NONE ? -1 : (int32_t)gg;
            }
            int32_t prev = __shfl_up(g, 1);
            if (lane == 0) prev = cur_gid;
            bool valid = row < n;
            bool is_start = valid && g != prev;
            uint64_t starts = __ballot(is_start);
            uint32_t chunk_rows =
                (uint32_t)(n - base < WAVE ? n - base : (uint64_t)WAVE);
            if (starts == 0) {
                cur_cnt += chunk_rows;
            } else {
                int F = (int)__builtin_ctzll(starts);
                bool head = (cur_cnt + (uint32_t)F) > 0;
                if (lane == 0 && head) {
                    rp[2 * n_pairs] =
                        cur_gid < 0 ? 0xFFFFu : (uint16_t)cur_gid;
                    rp[2 * n_pairs + 1] = (uint16_t)(cur_cnt + (uint32_t)F);
                }
                int last = 63 - __clzll(starts);
                int nst = __popcll(starts);
                if (is_start && lane != last) {
                    int rank = __popcll(starts & lanemask_lt(lane));
                    uint64_t hi = starts >> (lane + 1);   // lane < 63 here
                    int nxt = lane + 1 + (int)__builtin_ctzll(hi);
                    uint32_t idx = n_pairs + (head ? 1u : 0u) + (uint32_t)rank;
                    rp[2 * idx] = g < 0 ? 0xFFFFu : (uint16_t)g;
                    rp[2 * idx + 1] = (uint16_t)(nxt - lane);
                }
                n_pairs += (head ? 1u : 0u) + (uint32_t)(nst - 1);
                cur_gid = __shfl(g, last);
                cur_cnt = chunk_rows - (uint32_t)last;
            }
            carry_bytes += readlane64(incl, WAVE - 1);
        }
        if (cur_cnt > 0) {
            if (lane == 0) {
                rp[2 * n_pairs] = cur_gid < 0 ? 0xFFFFu : (uint16_t)cur_gid;
                rp[2 * n_pairs + 1] = (uint16_t)cur_cnt;
            }
            n_pairs++;
        }
        if (lane == 0) {
            GroupBlock *o = &out[bi];
            if (n_pairs == 1) {
                // whole block one run: uniform fast path, no runs walked
                uint16_t g16 = rp[0];
                o->uniform_gid = g16 == 0xFFFFu ? GID_NONE : (uint32_t)g16;
                o->width = 0;
                o->nentries = 0;
            } else {
                o->uniform_gid = GID_VARYING;
                o->rle_bit_off =
                    (uint64_t)slot_id * GRUN_ENTRIES_PER_STREAM;
                o->nentries = 2 * n_pairs;
            }
        }
    }
}

// Resolve prepass: one thread per block parses the dictionary header and
// builds the code mask.  Only plain (<128 B) compress_block sections are
// parseable on device (bytes.go:291-303); zstd-compressed dictionaries are
// flagged and surfaced as a device error if a predicate needs them.
#define PF_CLEAR 0
#define PF_SKIP 1
#define PF_WALK 2
#define PF_ERR 3

// Resolve one predicate slot for one block; returns the slot verdict
// (PF_CLEAR uniform-match / PF_SKIP miss-or-nil / PF_WALK row-varying /
// PF_ERR unparseable) and fills *out for the walker path.
__device__ int resolve_one_pred(const uint8_t *__restrict__ payload,
                                const uint8_t *__restrict__ sidecar,
                                const bydb_block_desc *__restrict__ bd,
                                int slot, const uint8_t *__restrict__ pred,
                                uint64_t pred_len, PredBlock *out) {
    uint64_t toff = slot == 0 ? bd->tag_off : slot == 1 ? bd->tag2_off : bd->tag3_off;
    uint64_t tlen = slot == 0 ? bd->tag_len : slot == 1 ? bd->tag2_len : bd->tag3_len;
    PredBlock pb;
    for (int i = 0; i < 4; i++) pb.mask[i] = 0;
    pb.rle_bit_off = 0; pb.nentries = 0; pb.width = 0;
    pb.active = 0; pb.err = 0; pb.uniform = 0; pb.plain = 0;
    pb.disabled = 0;
    if (pred_len == 0) {     // slot disabled: no constraint
        pb.disabled = 1;
        *out = pb;
        return PF_CLEAR;
    }
    if (tlen == 0) { *out = pb; return PF_SKIP; }   // nil tag: never equal
    const bool in_sidecar = (toff & TAG_SIDECAR_BIT) != 0;
    const uint8_t *base = in_sidecar ? sidecar : payload;
    const uint8_t *p = base + (toff & ~TAG_SIDECAR_BIT);
    const uint8_t *end = p + tlen;
    if (*p == BYDB_ENC_PLAIN) {
        // host-normalized plain column: a second pass (k_resolve_plain)
        // fills this block's match bitmap.  A plain stream that did NOT
        // go through host normalization (no sidecar bit) cannot be
        // parsed here — flag it loudly.
        if (!in_sidecar) { pb.err = 1; *out = pb; return PF_ERR; }
        pb.active = 1;
        pb.plain = 1;
        *out = pb;
        return PF_WALK;
    }
    if (*p != BYDB_ENC_DICTIONARY) { pb.err = 1; *out = pb; return PF_ERR; }
    p++;
    pb.active = 1;
    // varuint count (int.go:152-199)
    uint64_t count = 0;
    unsigned sh = 0;
    while (p < end) {
        uint8_t c = *p++;
        count |= (uint64_t)(c & 0x7f) << sh;
        if (c < 0x80) break;
        sh += 7;
    }
    // lengths block: compress_block(u64list) — plain or host-normalized
    uint64_t ll = 0;
    const uint8_t *lens_blk = dict_section(p, end, &ll);
    if (!lens_blk) { pb.err = 1; *out = pb; return PF_ERR; }
    p = lens_blk + ll;
    // values payload block
    uint64_t vl = 0;
    const uint8_t *vals = dict_section(p, end, &vl);
    if (!vals) { pb.err = 1; *out = pb; return PF_ERR; }
    p = vals + vl;
    // parse width-typed lengths (bytes.go:205-235)
    uint8_t wt = lens_blk[0];
    uint32_t wbytes = wt == 0 ? 1 : wt == 1 ? 2 : wt == 2 ? 4 : 8;
    const uint8_t *lp = lens_blk + 1;
    uint64_t voff = 0;
    for (uint64_t v = 0; v < count && v < 256; v++) {
        uint64_t alen = 0;
        for (uint32_t b = 0; b < wbytes; b++) alen = (alen << 8) | lp[v * wbytes + b];
        if (alen > 0) {
            uint64_t vlen = alen - 1;
            if (vlen == pred_len) {
                bool eq = true;
                for (uint64_t k = 0; k < vlen; k++)
                    if (vals[voff + k] != pred[k]) { eq = false; break; }
                if (eq) pb.mask[(v >> 6) & 3] |= 1ull << (v & 63);
            }
            voff += vlen;
        }
    }
    // bit-packed RLE: [32b entry count][8b width][entries...] MSB-first
    uint64_t bit0 = ((uint64_t)(p - base)) * 8;
    uint32_t nentries = (uint32_t)rd_bits_be(base, bit0, 32);
    uint32_t width = nentries ? (uint32_t)rd_bits_be(base, bit0 + 32, 8) : 0;
    pb.nentries = nentries;
    pb.width = (uint8_t)width;
    pb.rle_bit_off = (bit0 + 40) | (in_sidecar ? TAG_SIDECAR_BIT : 0);
    int verdict = PF_WALK;
    if (nentries == 2) {
        uint64_t code = rd_bits_be(base, bit0 + 40, width);
        uint64_t cnt = rd_bits_be(base, bit0 + 40 + width, width);
        if (cnt >= bd->count) {
            pb.uniform = 1;
            bool match = (pb.mask[(code >> 6) & 3] >> (code & 63)) & 1;
            verdict = match ? PF_CLEAR : PF_SKIP;
        }
    }
    *out = pb;
    return verdict;
}

// One fused launch resolves every predicate slot for a block (the
// descriptor is read once) and writes the combined flag byte the scan's
// hot loop prices a non-matching block at:
//   0 = every predicated slot is uniform-match -> fold unpredicated
//   1 = some slot misses (uniform non-match or nil tag) -> skip block
//   2 = some slot is row-varying -> full walker path
//   3 = unparseable tag stream -> device error
__global__ void k_resolve_preds(
    const uint8_t *__restrict__ payload, const uint8_t *__restrict__ sidecar,
    const bydb_block_desc *__restrict__ blocks, int64_t n_blocks,
    const uint8_t *__restrict__ pred_bytes, uint64_t o0, uint64_t l0,
    uint64_t o1, uint64_t l1, uint64_t o2, uint64_t l2, int n_preds,
    PredBlock *__restrict__ out, uint8_t *__restrict__ flags,
    uint32_t *__restrict__ walk_count) {
    int64_t bi = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (bi >= n_blocks) return;
    const bydb_block_desc *bd = &blocks[bi];
    const uint64_t offs[3] = {o0, o1, o2};
    const uint64_t lens[3] = {l0, l1, l2};
    uint8_t f = PF_CLEAR;
    for (int sl = 0; sl < n_preds; sl++) {
        PredBlock pb;
        int v = resolve_one_pred(payload, sidecar, bd, sl,
                                 pred_bytes + offs[sl], lens[sl], &pb);
        out[(int64_t)sl * n_blocks + bi] = pb;
        if (v == PF_ERR) { f = PF_ERR; break; }
        if (v == PF_SKIP) { f = PF_SKIP; break; }
        if (v == PF_WALK) f = PF_WALK;
    }
    flags[bi] = f;
    // walk-block census: lets the heavy scan pass exit immediately when
    // every predicate verdict is uniform (the dominant entity-tag case)
    uint64_t wmask = __ballot(f == PF_WALK);
    if ((threadIdx.x & 63) == 0 && wmask)
        atomicAdd(walk_count, (uint32_t)__popcll(wmask));
}

// Second resolve pass for plain (non-dictionary) tag columns: one wave per
// block evaluates the equality predicate on every row of the normalized
// stream ([ENC_PLAIN][u32le n][wt][lens BE][payload]) and writes a per-row
// match bitmap.  The residual-predicate pushdown the reference performs
// row-by-row on decoded [][]byte (aggregation.go:310 over column.go:410
// Plain columns) becomes 64 rows per step with a wave prefix-scan over the
// value lengths for the payload offsets.
__global__ __launch_bounds__(WAVE) void k_resolve_plain(
    const uint8_t *__restrict__ payload, const uint8_t *__restrict__ sidecar,
    const bydb_block_desc *__restrict__ blocks, int64_t n_blocks,
    const uint8_t *__restrict__ pred, uint64_t pred_len, int slot,
    PredBlock *__restrict__ out, uint64_t *__restrict__ bm,
    uint32_t *__restrict__ ctr) {
    const int lane = threadIdx.x;
    for (int64_t bi = blockIdx.x; bi < n_blocks; bi += gridDim.x) {
        PredBlock pb = out[bi];
        if (!pb.plain || pb.err) continue;
        const bydb_block_desc *bd = &blocks[bi];
        uint64_t toff = slot == 0 ? bd->tag_off
                                  : slot == 1 ? bd->tag2_off : bd->tag3_off;
        const uint8_t *p = (toff & TAG_SIDECAR_BIT)
                               ? sidecar + (toff & ~TAG_SIDECAR_BIT)
                               : payload + toff;
        p++;  // ENC_PLAIN
        uint64_t n = (uint64_t)p[0] | ((uint64_t)p[1] << 8) |
                     ((uint64_t)p[2] << 16) | ((uint64_t)p[3] << 24);
        p += 4;
        uint8_t wt = *p++;
        uint32_t wbytes = wt == 0 ? 1 : wt == 1 ? 2 : wt == 2 ? 4 : 8;
        const uint8_t *lens = p;
        const uint8_t *vals = lens + n * wbytes;
        if (n != (uint64_t)bd->count || n > (uint64_t)PLAIN_BM_WORDS * 64) {
            if (lane == 0) out[bi].err = 1;
            continue;
        }
        uint32_t slot_id = 0;
        if (lane == 0) slot_id = atomicAdd(ctr, 1u);
        slot_id = (uint32_t)__shfl((int)slot_id, 0);
        uint64_t w0 = (uint64_t)slot_id * PLAIN_BM_WORDS;
        uint64_t carry = 0;  // payload bytes consumed by prior chunks
        for (uint64_t base = 0; base < n; base += WAVE) {
            uint64_t row = base + (uint64_t)lane;
            uint64_t lp1 = 0;
            if (row < n)
                for (uint32_t b = 0; b < wbytes; b++)
                    lp1 = (lp1 << 8) | lens[row * wbytes + b];
            uint64_t vlen = lp1 ? lp1 - 1 : 0;
            uint64_t incl = wave_incl_scan(vlen, lane);
            uint64_t myoff = carry + incl - vlen;
            bool match = false;
            if (row < n && lp1 && vlen == pred_len) {
                match = true;
                for (uint64_t k = 0; k < vlen; k++)
                    if (vals[myoff + k] != pred[k]) { match = false; break; }
            }
            uint64_t word = __ballot(match);
            if (lane == 0) bm[w0 + (base >> 6)] = word;
            carry += readlane64(incl, WAVE - 1);
        }
        if (lane == 0) out[bi].rle_bit_off = w0;
    }
}

// Fold arithmetic-progression blocks (Const dd=0 / DeltaConst) under a
// per-row predicate: walk rows 64 at a time through the RLE runs.
__device__ void fold_arith_pred(int64_t first, int64_t dd, int64_t r0,
                                int64_t r1, int lane, PredWalk *pw0,
                                PredWalk *pw1, PredWalk *pw2,
                                uint64_t *out_sum, uint64_t *out_cnt,
                                int64_t *out_mn, int64_t *out_mx) {
    uint64_t lsum = 0, lcnt = 0;
    int64_t lmn = INT64_MAX, lmx = INT64_MIN;
    for (int64_t base = r0; base <= r1; base += 64) {
        int64_t row = base + lane;
        bool need = row <= r1;
        bool match = true;
        if (pw0) match = pred_match_rows(pw0, row, need && match) && match;
        if (pw1) match = pred_match_rows(pw1, row, need && match) && match;
        if (pw2) match = pred_match_rows(pw2, row, need && match) && match;
        if (need && match) {
            int64_t v = (int64_t)((uint64_t)first + (uint64_t)row * (uint64_t)dd);
            lsum += (uint64_t)v;
            lcnt++;
            lmn = v < lmn ? v : lmn;
            lmx = v > lmx ? v : lmx;
        }
    }
    *out_sum = wave_reduce_add(lsum);
    *out_cnt = wave_reduce_add(lcnt);
    *out_mn = wave_reduce_min(lmn);
    *out_mx = wave_reduce_max(lmx);
}

// Full value reconstruction over a delta or delta-of-delta stream with a
// per-value callback encoded as flags (fold min/max/sum/sumf, or count
// ts-range bounds).  dod=false: values are rows j=1..n-1 (row 0 = first,
// handled by the caller).  dod=true: stream holds d2 for rows 2..n-1 and
// d1_init has been parsed; rows 0,1 handled by the caller.
struct ScanFold {
    uint64_t sum;       // wrapping sum of selected values
    int64_t mn, mx;     // min/max of selected values
    uint64_t nsel;      // selected row count
};

// full: compile-time promise that [r0, r1] covers every streamed row
// (the clamp-free instantiations pass the template constant !EN_CLAMP),
// letting the per-value selection branch dead-code out of the quad fold
// — less register pressure for the 7-wave build.
__device__ void scan_stream(const uint8_t *stream, int64_t n_deltas, bool dod,
                            int64_t first_plus /* v at row (dod?1:0) */,
                            int64_t d1_init, int64_t r0, int64_t r1,
                            int lane, ScanFold *f, DevErr *derr, uint64_t bi,
                            PredWalk *pw0, PredWalk *pw1, PredWalk *pw2,
                            bool full = false) {
    uint64_t pos = 0;
    int64_t j = dod ? 2 : 1;
    int64_t jmax = n_deltas;            // always scan the whole stream
    uint64_t v_carry = (uint64_t)first_plus;
    uint64_t d1_carry = (uint64_t)d1_init;
    uint64_t l_sum = 0, l_nsel = 0;
    int64_t l_mn = INT64_MAX, l_mx = INT64_MIN;
    // Fixed-size window advance with a cross-window varint carry
    // (carry_u holds the carried low groups, carry_n its byte count), so
    // window addresses are induction variables.  The primary window is
    // 128 B with TWO bytes per lane (one u16 load): when every varint is
    // 1-2 bytes — the dominant telemetry shape — each lane decodes up to
    // two values and one 32-bit scan covers 128 B, halving per-byte
    // instruction and load counts vs the 64-B window.  Windows that fail
    // the pattern (longer varints, walkers, the stream tail, a multi-byte
    // carry) fall back to the 64-B machinery below.  Terminators past
    // jmax (bytes of the following stream / the part's 1-KiB slack) fold
    // as zeros, so the wave-wide scans read their totals at lane 63.
    uint64_t carry_u = 0;
    uint32_t carry_n = 0;
    const bool no_walk = pw0 == nullptr && pw1 == nullptr && pw2 == nullptr;
#define QLEAD 4
    uint32_t wq[QLEAD];  // wq[q] = this lane's u32 of the window at pos+256q
    bool done = false;
    while (j <= jmax) {
        // The quad path is a SELF-CONTAINED inner loop: with the 64-B
        // fallback in the same loop body, the compiler hoists the
        // fallback's byte load above the quad branch and fences it with
        // s_waitcnt vmcnt(0) — draining the whole QLEAD window pipeline
        // every iteration (measured: the scan ran at full memory latency
        // per 256-B window).  Inside this loop the only vector loads are
        // the window dwords, so the waits stay at vmcnt(QLEAD-1).
        if (!dod && no_walk && carry_n <= 1) {
            // pipeline fill: QLEAD windows in flight, then exactly ONE
            // load per steady-state iteration (at the bottom, QLEAD-1
            // windows ahead of its use)
#pragma unroll
            for (int q = 0; q < QLEAD; q++)
                __builtin_memcpy(&wq[q],
                                 stream + pos + 256 * (uint64_t)q +
                                     4 * (uint64_t)lane, 4);
            // Ring by PHASE, not by shifting: moving wq[q+1] into wq[q]
            // would make every iteration wait on the NEWEST outstanding
            // load (vmcnt(0) to read the register being shifted),
            // re-serialising the pipeline.  The unrolled phase loop keeps
            // indices static, so reading wq[ph] waits only for the load
            // issued QLEAD-1 iterations ago.
            bool quad_exit = false;
            while (!quad_exit) {
#pragma unroll
            for (int ph = 0; ph < QLEAD; ph++) {
            // 256-B quad window: FOUR bytes per lane (one u32 load), each
            // lane decoding up to four 1-2-byte varints; one 32-bit scan
            // covers 256 B.
            const uint32_t w_cur = wq[ph];
            uint32_t b0 = w_cur & 0xffu, b1 = (w_cur >> 8) & 0xffu,
                     b2 = (w_cur >> 16) & 0xffu, b3 = w_cur >> 24;
            uint64_t e0 = __ballot(b0 < 0x80);
            uint64_t e1 = __ballot(b1 < 0x80);
            uint64_t e2 = __ballot(b2 < 0x80);
            uint64_t e3 = __ballot(b3 < 0x80);
            uint64_t c0 = ~e0, c1 = ~e1, c2 = ~e2, c3 = ~e3;
            int nt = __popcll(e0) + __popcll(e1) + __popcll(e2) +
                     __popcll(e3);
            if (((c0 & c1) | (c1 & c2) | (c2 & c3)) == 0 &&
                (c3 & (c0 >> 1)) == 0 &&
                (int64_t)nt <= jmax - j + 1 &&
                (carry_n == 0 || (e0 & 1))) {
                // previous lane's b3 heads a value ending at my b0
                uint32_t pb3 = dpp_mov32<0x111>(b3);
                uint32_t q15 = (uint32_t)__builtin_amdgcn_readlane((int)b3, 15);
                uint32_t q31 = (uint32_t)__builtin_amdgcn_readlane((int)b3, 31);
                uint32_t q47 = (uint32_t)__builtin_amdgcn_readlane((int)b3, 47);
                if ((lane & 15) == 0 && lane)
                    pb3 = lane == 16 ? q15 : lane == 32 ? q31 : q47;
                if (lane == 0)
                    pb3 = carry_n ? ((uint32_t)carry_u | 0x80u) : 0;
                bool t0 = b0 < 0x80, t1 = b1 < 0x80, t2 = b2 < 0x80,
                     t3 = b3 < 0x80;
                uint32_t u0 = (pb3 & 0x80u) ? ((pb3 & 0x7fu) | (b0 << 7)) : b0;
                uint32_t u1 = (b0 & 0x80u) ? ((b0 & 0x7fu) | (b1 << 7)) : b1;
                uint32_t u2 = (b1 & 0x80u) ? ((b1 & 0x7fu) | (b2 << 7)) : b2;
                uint32_t u3 = (b2 & 0x80u) ? ((b2 & 0x7fu) | (b3 << 7)) : b3;
                int32_t dA = t0 ? ((int32_t)(u0 >> 1) ^ -(int32_t)(u0 & 1)) : 0;
                int32_t dB = t1 ? ((int32_t)(u1 >> 1) ^ -(int32_t)(u1 & 1)) : 0;
                int32_t dC = t2 ? ((int32_t)(u2 >> 1) ^ -(int32_t)(u2 & 1)) : 0;
                int32_t dD = t3 ? ((int32_t)(u3 >> 1) ^ -(int32_t)(u3 & 1)) : 0;
                int32_t sl = dA + dB + dC + dD;
                int32_t S = wave_incl_scan32(sl, lane);
                int32_t cum = S - sl;
                if (full || (j >= r0 && j + nt - 1 <= r1)) {
                    // whole window selected (the dominant shape once the
                    // block-level clamp resolved): no per-value index
                    // bookkeeping — every terminator folds
#define BYDB_QVF(tk, dk)                                                     \
                    if (tk) {                                                \
                        cum += dk;                                           \
                        int64_t sv =                                         \
                            (int64_t)(v_carry + (uint64_t)(int64_t)cum);     \
                        l_sum += (uint64_t)sv;                               \
                        l_mn = sv < l_mn ? sv : l_mn;                        \
                        l_mx = sv > l_mx ? sv : l_mx;                        \
                    }
                    BYDB_QVF(t0, dA)
                    BYDB_QVF(t1, dB)
                    BYDB_QVF(t2, dC)
                    BYDB_QVF(t3, dD)
#undef BYDB_QVF
                    l_nsel += (uint64_t)((int)t0 + (int)t1 + (int)t2 +
                                         (int)t3);
                } else {
                    int rb = __popcll(e0 & lanemask_lt(lane)) +
                             __popcll(e1 & lanemask_lt(lane)) +
                             __popcll(e2 & lanemask_lt(lane)) +
                             __popcll(e3 & lanemask_lt(lane));
                    int64_t idx = j + rb;
#define BYDB_QVAL(tk, dk)                                                    \
                    if (tk) {                                                \
                        cum += dk;                                           \
                        int64_t sv =                                         \
                            (int64_t)(v_carry + (uint64_t)(int64_t)cum);     \
                        if (idx >= r0 && idx <= r1) {                        \
                            l_sum += (uint64_t)sv;                           \
                            l_nsel++;                                        \
                            l_mn = sv < l_mn ? sv : l_mn;                    \
                            l_mx = sv > l_mx ? sv : l_mx;                    \
                        }                                                    \
                        idx++;                                               \
                    }
                    BYDB_QVAL(t0, dA)
                    BYDB_QVAL(t1, dB)
                    BYDB_QVAL(t2, dC)
                    BYDB_QVAL(t3, dD)
#undef BYDB_QVAL
                }
                v_carry +=
                    (uint64_t)(int64_t)__builtin_amdgcn_readlane(S, 63);
                if (e3 >> 63) {
                    carry_n = 0;
                    carry_u = 0;
                } else {
                    carry_n = 1;
                    carry_u = (uint64_t)(
                        (uint32_t)__builtin_amdgcn_readlane((int)b3, 63) &
                        0x7f);
                }
                j += nt;
                pos += 256;
                if (j > jmax) { done = true; quad_exit = true; break; }
                // refill this phase's slot with the window QLEAD-1 ahead
                __builtin_memcpy(&wq[ph],
                                 stream + pos + 256 * (uint64_t)(QLEAD - 1) +
                                     4 * (uint64_t)lane, 4);
                continue;
            }
            quad_exit = true;  // pattern failed: 64-B fallback below
            break;
            }        // phase loop
            }        // inner quad loop
            if (done) break;
        }
        uint8_t b = stream[pos + (uint64_t)lane];
        uint64_t emask = __ballot(b < 0x80);
        if (emask == 0) { dev_set_err(derr, DERR_BAD_STREAM, bi); break; }
        int rank = __popcll(emask & lanemask_lt(lane));
        int64_t myj = j + rank;
        bool is_term = (b < 0x80) && (myj <= jmax);
        int nterm = __popcll(emask);
        // fast window: every varint is 1 or 2 bytes (no two consecutive
        // continuation bits), not the tail window, and any carried byte
        // completes at lane 0 — decode via the left neighbour's byte
        // (shfl) and scan in 32-bit (window delta sums fit: <= 64 * 2^21)
        uint64_t cont = ~emask;
        if (pw0 == nullptr && pw1 == nullptr && pw2 == nullptr &&
            (cont & (cont << 1)) == 0 &&
            (int64_t)nterm <= jmax - j + 1 &&
            (carry_n == 0 || (carry_n == 1 && (emask & 1)))) {
            uint32_t prev = (uint32_t)__shfl_up((int)b, 1);
            bool is2 = lane > 0 && (prev & 0x80u);
            uint32_t u = is2 ? ((prev & 0x7fu) | ((uint32_t)b << 7)) : b;
            if (lane == 0 && carry_n)
                u = (uint32_t)carry_u | ((uint32_t)b << 7);
            int32_t d32 = (int32_t)(u >> 1) ^ -(int32_t)(u & 1);
            if (!(b < 0x80)) d32 = 0;          // continuation lanes carry 0
            int32_t s32 = wave_incl_scan32(d32, lane);
            uint64_t sv_u;
            if (dod) {
                // delta-of-delta in two 32-bit scans regardless of the
                // 64-bit d1 carry:  v_j = v_carry + cnt_j*d1_carry + t_j
                // with t_j = scan over terminators of (scan of d2)
                int32_t t32 = wave_incl_scan32((b < 0x80) ? s32 : 0, lane);
                sv_u = v_carry + (uint64_t)(int64_t)(rank + 1) * d1_carry +
                       (uint64_t)(int64_t)t32;
                int32_t t_tot = __builtin_amdgcn_readlane(t32, 63);
                int32_t s_tot = __builtin_amdgcn_readlane(s32, 63);
                v_carry += (uint64_t)(int64_t)nterm * d1_carry +
                           (uint64_t)(int64_t)t_tot;
                d1_carry += (uint64_t)(int64_t)s_tot;
            } else {
                sv_u = v_carry + (uint64_t)(int64_t)s32;
                v_carry += (uint64_t)(int64_t)__builtin_amdgcn_readlane(s32, 63);
            }
            if (b < 0x80) {
                int64_t sv = (int64_t)sv_u;
                if (full || (myj >= r0 && myj <= r1)) {
                    l_sum += (uint64_t)sv;
                    l_nsel++;
                    l_mn = sv < l_mn ? sv : l_mn;
                    l_mx = sv > l_mx ? sv : l_mx;
                }
            }
            if (emask >> 63) {
                carry_n = 0;
                carry_u = 0;
            } else {
                carry_n = 1;
                carry_u = (uint64_t)((uint32_t)__builtin_amdgcn_readlane((int)b, 63) & 0x7f);
            }
            j += nterm;
            pos += 64;
            continue;
        }
        uint64_t d = 0;
        if (emask == ~0ull && carry_n == 0) {
            d = (uint64_t)zz_dec(b);
        } else {
            // assemble each terminator's varint from the bytes already in
            // registers: byte k back arrives by one __shfl_up at a
            // wave-uniform distance (trip count = longest in-window varint
            // from the widest terminator gap); the first terminator also
            // prepends the carried bytes.  LSB-first groups
            // (int.go:81-103): the byte k back holds bits (len-1-k)*7.
            uint64_t below = emask & lanemask_lt(lane);
            int start = below ? (64 - __clzll(below)) : 0;
            int mylen = lane - start + 1;        // in-window bytes
            if (mylen > 10) mylen = 10;  // valid varints are <=10 B; longer
                                         // gaps only occur on corrupt input
            // wave-uniform max varint length over terminator lanes
            int maxlen = (b < 0x80) ? mylen : 1;
            for (int off2 = 32; off2 > 0; off2 >>= 1) {
                int t = __shfl_xor(maxlen, off2);
                maxlen = t > maxlen ? t : maxlen;
            }
            uint64_t u = (uint64_t)(b & 0x7f) << (7 * (mylen - 1));
            for (int k = 1; k < maxlen; ++k) {
                uint32_t bk = (uint32_t)__shfl_up((int)b, k);
                if (k < mylen)
                    u |= (uint64_t)(bk & 0x7f) << (7 * (mylen - 1 - k));
            }
            if (below == 0 && carry_n)
                u = carry_u | (u << (7 * carry_n));
            if (is_term) d = (uint64_t)zz_dec(u);
        }
        if (!is_term) d = 0;
        uint64_t val;
        if (dod) {
            // first scan: d1_j = d1_carry + incl_scan(d2)
            uint64_t s1 = wave_incl_scan(d, lane);
            uint64_t d1j = is_term ? (d1_carry + s1) : 0;
            // second scan: v_j = v_carry + incl_scan(d1_j over values)
            uint64_t s2 = wave_incl_scan(d1j, lane);
            val = v_carry + s2;
            v_carry += readlane64(s2, 63);
            d1_carry += readlane64(s1, 63);
        } else {
            uint64_t sscan = wave_incl_scan(d, lane);
            val = v_carry + sscan;
            v_carry += readlane64(sscan, 63);
        }
        bool in_sel = is_term && (full || (myj >= r0 && myj <= r1));
        if (pw0) in_sel = pred_match_rows(pw0, myj, in_sel) && in_sel;
        if (pw1) in_sel = pred_match_rows(pw1, myj, in_sel) && in_sel;
        if (pw2) in_sel = pred_match_rows(pw2, myj, in_sel) && in_sel;
        if (in_sel) {
            int64_t sv = (int64_t)val;
            l_sum += val;
            l_nsel++;
            l_mn = sv < l_mn ? sv : l_mn;
            l_mx = sv > l_mx ? sv : l_mx;
        }
        if (j + nterm > jmax) break;
        // carry out: bytes after the last terminator head the next value
        int last_lane = 63 - __clzll(emask);
        int n_tail = 63 - last_lane;
        if (n_tail) {
            uint64_t tail = 0;
            if (lane > last_lane) {
                int sh2 = lane - last_lane - 1;
                if (sh2 > 9) sh2 = 9;        // corrupt-input guard
                tail = (uint64_t)(b & 0x7f) << (7 * sh2);
            }
            for (int off2 = 32; off2 > 0; off2 >>= 1)
                tail |= (uint64_t)__shfl_xor((long long)tail, off2);
            carry_u = tail;
            carry_n = n_tail > 9 ? 9 : (uint32_t)n_tail;
        } else {
            carry_u = 0;
            carry_n = 0;
        }
        j += nterm;
        pos += 64;
    }
    f->sum = l_sum;
    f->mn = l_mn;
    f->mx = l_mx;
    f->nsel = l_nsel;
}

// Count rows of a delta / delta-of-delta stream whose value falls below
// lo_bound (n_lo) or above hi_bound (n_hi) — all the FindRange clamp
// needs (range.go:143-170; ascending blocks).  This is the ONLY consumer
// of value bounds, kept OUT of scan_stream so the fold kernels (incl.
// the closed-form-only headline instantiation, which needs the clamp but
// not value folds) carry none of the scan's fast-window register state.
// Correctness path: it runs only on boundary blocks whose ts range
// partially overlaps the query; generic 64-B ballot windows.  Totals are
// wave-reduced; all lanes return them.
__device__ void clamp_count_stream(
    const uint8_t *stream, int64_t n_deltas,
                                   bool dod, int64_t first_plus,
                                   int64_t d1_init, int64_t lo_bound,
                                   int64_t hi_bound, int lane,
                                   uint64_t *out_nlo, uint64_t *out_nhi,
                                   DevErr *derr, uint64_t bi) {
    uint64_t pos = 0;
    int64_t j = dod ? 2 : 1;
    int64_t jmax = n_deltas;
    uint64_t v_carry = (uint64_t)first_plus;
    uint64_t d1_carry = (uint64_t)d1_init;
    uint64_t l_nlo = 0, l_nhi = 0;
    uint64_t carry_u = 0;
    uint32_t carry_n = 0;
    while (j <= jmax) {
        uint8_t b = stream[pos + (uint64_t)lane];
        uint64_t emask = __ballot(b < 0x80);
        if (emask == 0) { dev_set_err(derr, DERR_BAD_STREAM, bi); break; }
        int rank = __popcll(emask & lanemask_lt(lane));
        int64_t myj = j + rank;
        bool is_term = (b < 0x80) && (myj <= jmax);
        int nterm = __popcll(emask);
        uint64_t d = 0;
        if (emask == ~0ull && carry_n == 0) {
            d = (uint64_t)zz_dec(b);
        } else {
            uint64_t below = emask & lanemask_lt(lane);
            int start = below ? (64 - __clzll(below)) : 0;
            int mylen = lane - start + 1;
            if (mylen > 10) mylen = 10;
            int maxlen = (b < 0x80) ? mylen : 1;
            for (int off2 = 32; off2 > 0; off2 >>= 1) {
                int t = __shfl_xor(maxlen, off2);
                maxlen = t > maxlen ? t : maxlen;
            }
            uint64_t u = (uint64_t)(b & 0x7f) << (7 * (mylen - 1));
            for (int k = 1; k < maxlen; ++k) {
                uint32_t bk = (uint32_t)__shfl_up((int)b, k);
                if (k < mylen)
                    u |= (uint64_t)(bk & 0x7f) << (7 * (mylen - 1 - k));
            }
            if (below == 0 && carry_n)
                u = carry_u | (u << (7 * carry_n));
            if (is_term) d = (uint64_t)zz_dec(u);
        }
        if (!is_term) d = 0;
        uint64_t val;
        if (dod) {
            uint64_t s1 = wave_incl_scan(d, lane);
            uint64_t d1j = is_term ? (d1_carry + s1) : 0;
            uint64_t s2 = wave_incl_scan(d1j, lane);
            val = v_carry + s2;
            v_carry += readlane64(s2, 63);
            d1_carry += readlane64(s1, 63);
        } else {
            uint64_t sscan = wave_incl_scan(d, lane);
            val = v_carry + sscan;
            v_carry += readlane64(sscan, 63);
        }
        if (is_term) {
            int64_t sv = (int64_t)val;
            if (sv < lo_bound) l_nlo++;
            if (sv > hi_bound) l_nhi++;
        }
        if (j + nterm > jmax) break;
        int last_lane = 63 - __clzll(emask);
        int n_tail = 63 - last_lane;
        if (n_tail) {
            uint64_t tail = 0;
            if (lane > last_lane) {
                int sh2 = lane - last_lane - 1;
                if (sh2 > 9) sh2 = 9;
                tail = (uint64_t)(b & 0x7f) << (7 * sh2);
            }
            for (int off2 = 32; off2 > 0; off2 >>= 1)
                tail |= (uint64_t)__shfl_xor((long long)tail, off2);
            carry_u = tail;
            carry_n = n_tail > 9 ? 9 : (uint32_t)n_tail;
        } else {
            carry_u = 0;
            carry_n = 0;
        }
        j += nterm;
        pos += 64;
    }
    *out_nlo = wave_reduce_add(l_nlo);
    *out_nhi = wave_reduce_add(l_nhi);
}

// ---------------- varint segment index ----------------
// The value-reconstruction path over a mixed-width varint stream is a
// serial chain of ballot windows (the next window's address depends on the
// current ballot).  For a RESIDENT part that chain can be broken once:
// a build pass walks each eligible stream and records, every SEG_ROWS
// values, the byte offset and running value at the segment start.  The
// scan then processes (block, segment) pairs independently — 8x more
// parallel units and no cross-segment dependency.  The index is built
// once per part upload + agg shape (like the reference's block metadata,
// it is derived state over immutable part bytes).
#define SEG_ROWS 8192
#define MAX_SEGS 1
#define SEG_INELIGIBLE 0xFFFFFFFFu

struct SegEntry {
    uint32_t byte_off;   // offset of the segment's first delta in the stream
    uint32_t _pad;
    int64_t v_start;     // value at row seg*SEG_ROWS (before the first delta)
    int64_t d1_start;    // delta-of-delta streams: d1 at row seg*SEG_ROWS
};

__global__ __launch_bounds__(256) void k_build_seg_index(
    const uint8_t *__restrict__ payload, const bydb_block_desc *__restrict__ blocks,
    int64_t n_blocks, SegEntry *__restrict__ segs, DevErr *derr) {
    const int lane = threadIdx.x & 63;
    int64_t wave_id = (int64_t)blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
    int64_t n_waves = (int64_t)gridDim.x * (blockDim.x >> 6);
    for (int64_t bi = wave_id; bi < n_blocks; bi += n_waves) {
        const bydb_block_desc *bd = &blocks[bi];
        SegEntry *bseg = segs + bi * MAX_SEGS;
        const int64_t n = (int64_t)bd->count;
        const bool dod = bd->field_enc == BYDB_ENC_DELTA_OF_DELTA;
        bool eligible =
            n > SEG_ROWS &&
            ((bd->field_enc == BYDB_ENC_DELTA &&
              bd->field_len != (uint64_t)(n - 1)) ||
             dod);
        if (!eligible) {
            if (lane == 0) bseg[0].byte_off = SEG_INELIGIBLE;
            continue;
        }
        const uint8_t *stream = payload + bd->field_off;
        uint64_t d2_off = 0;
        uint64_t d1_carry = 0;
        if (dod) {
            int vl;
            d1_carry = (uint64_t)decode_one_varint(stream, &vl);
            d2_off = (uint64_t)vl;
        }
        if (lane == 0) {
            bseg[0].byte_off = (uint32_t)d2_off;
            bseg[0].v_start =
                dod ? (int64_t)((uint64_t)bd->field_first + d1_carry)
                    : bd->field_first;
            bseg[0].d1_start = (int64_t)d1_carry;
        }
        const uint8_t *d2s = stream + d2_off;
        const int64_t n_deltas = n - 1 - (dod ? 1 : 0);
        // j here indexes the per-stream varint: absolute row = j (delta)
        // or j + 1 (delta-of-delta)
        uint64_t pos = 0;
        int64_t j = 1;
        uint64_t v_carry = (uint64_t)bd->field_first + (dod ? d1_carry : 0);
        // delta streams walk 128-B paired-lane windows (the scan_stream
        // fast shape) and record boundaries per value; non-qualifying
        // windows rewind any carried byte and use the 64-B machinery
        uint64_t carry_u2 = 0;
        uint32_t carry_n2 = 0;
        while (j <= n_deltas) {
            if (!dod && carry_n2 <= 1) {
                uint16_t pr;
                __builtin_memcpy(&pr, d2s + pos + 2 * (uint64_t)lane, 2);
                uint32_t b0 = pr & 0xffu, b1 = (uint32_t)pr >> 8;
                uint64_t e0 = __ballot(b0 < 0x80);
                uint64_t e1 = __ballot(b1 < 0x80);
                uint64_t c0 = ~e0, c1 = ~e1;
                int nt = __popcll(e0) + __popcll(e1);
                if ((c0 & c1) == 0 && (c1 & (c0 >> 1)) == 0 &&
                    (int64_t)nt <= n_deltas - j + 1 &&
                    (carry_n2 == 0 || (e0 & 1))) {
                    uint32_t pb1 = dpp_mov32<0x111>(b1);
                    uint32_t q15 =
                        (uint32_t)__builtin_amdgcn_readlane((int)b1, 15);
                    uint32_t q31 =
                        (uint32_t)__builtin_amdgcn_readlane((int)b1, 31);
                    uint32_t q47 =
                        (uint32_t)__builtin_amdgcn_readlane((int)b1, 47);
                    if ((lane & 15) == 0 && lane)
                        pb1 = lane == 16 ? q15 : lane == 32 ? q31 : q47;
                    if (lane == 0)
                        pb1 = carry_n2 ? ((uint32_t)carry_u2 | 0x80u) : 0;
                    bool t0 = b0 < 0x80, t1 = b1 < 0x80;
                    uint32_t u0 =
                        (pb1 & 0x80u) ? ((pb1 & 0x7fu) | (b0 << 7)) : b0;
                    uint32_t u1 =
                        (b0 & 0x80u) ? ((b0 & 0x7fu) | (b1 << 7)) : b1;
                    int32_t dA =
                        t0 ? ((int32_t)(u0 >> 1) ^ -(int32_t)(u0 & 1)) : 0;
                    int32_t dB =
                        t1 ? ((int32_t)(u1 >> 1) ^ -(int32_t)(u1 & 1)) : 0;
                    int32_t sl = dA + dB;
                    int32_t S = wave_incl_scan32(sl, lane);
                    int32_t pre = S - sl;
                    int rb = __popcll(e0 & lanemask_lt(lane)) +
                             __popcll(e1 & lanemask_lt(lane));
                    if (t0) {
                        int64_t row = j + rb;
                        if ((row % SEG_ROWS) == 0 &&
                            row / SEG_ROWS < MAX_SEGS) {
                            SegEntry E;
                            E.byte_off =
                                (uint32_t)(d2_off + pos + 2 * (uint64_t)lane +
                                           1);
                            E._pad = 0;
                            E.v_start = (int64_t)(v_carry +
                                                  (uint64_t)(int64_t)(pre + dA));
                            E.d1_start = 0;
                            bseg[row / SEG_ROWS] = E;
                        }
                    }
                    if (t1) {
                        int64_t row = j + rb + (t0 ? 1 : 0);
                        if ((row % SEG_ROWS) == 0 &&
                            row / SEG_ROWS < MAX_SEGS) {
                            SegEntry E;
                            E.byte_off =
                                (uint32_t)(d2_off + pos + 2 * (uint64_t)lane +
                                           2);
                            E._pad = 0;
                            E.v_start =
                                (int64_t)(v_carry +
                                          (uint64_t)(int64_t)(pre + dA + dB));
                            E.d1_start = 0;
                            bseg[row / SEG_ROWS] = E;
                        }
                    }
                    v_carry +=
                        (uint64_t)(int64_t)__builtin_amdgcn_readlane(S, 63);
                    if (e1 >> 63) {
                        carry_n2 = 0;
                        carry_u2 = 0;
                    } else {
                        carry_n2 = 1;
                        carry_u2 = (uint64_t)(
                            (uint32_t)__builtin_amdgcn_readlane((int)b1, 63) &
                            0x7f);
                    }
                    j += nt;
                    pos += 128;
                    continue;
                }
            }
            if (carry_n2) {  // rewind the carried byte for the 64-B walk
                pos -= carry_n2;
                carry_n2 = 0;
                carry_u2 = 0;
            }
            uint8_t b = d2s[pos + (uint64_t)lane];
            uint64_t emask = __ballot(b < 0x80);
            if (emask == 0) { dev_set_err(derr, DERR_BAD_STREAM, (uint64_t)bi); break; }
            int rank = __popcll(emask & lanemask_lt(lane));
            int64_t myj = j + rank;
            bool is_term = (b < 0x80) && (myj <= n_deltas);
            uint64_t d = 0;
            if (emask == ~0ull) {
                d = (uint64_t)zz_dec(b);
            } else if (is_term) {
                uint64_t below = emask & lanemask_lt(lane);
                int start = below ? (64 - __clzll(below)) : 0;
                uint64_t u = 0;
                unsigned sh = 0;
                for (int i = start; i < lane; ++i) {
                    u |= (uint64_t)(d2s[pos + (uint64_t)i] & 0x7f) << sh;
                    sh += 7;
                }
                u |= (uint64_t)b << sh;
                d = (uint64_t)zz_dec(u);
            }
            if (!is_term) d = 0;
            uint64_t s1 = wave_incl_scan(d, lane);
            uint64_t vinc;
            uint64_t d1j = 0;
            if (dod) {
                // v advances by d1_j = d1_carry + scan(d2); scan the d1_j
                d1j = is_term ? (d1_carry + s1) : 0;
                vinc = wave_incl_scan(d1j, lane);
            } else {
                vinc = s1;
            }
            // a lane whose absolute row is a SEG_ROWS multiple records the
            // next segment's start (byte after my terminator)
            int64_t myrow = myj + (dod ? 1 : 0);
            if (is_term && (myrow % SEG_ROWS) == 0 &&
                myrow / SEG_ROWS < MAX_SEGS) {
                SegEntry e;
                e.byte_off = (uint32_t)(d2_off + pos + (uint64_t)lane + 1);
                e._pad = 0;
                e.v_start = (int64_t)(v_carry + vinc);
                e.d1_start = (int64_t)(d1_carry + s1);
                bseg[myrow / SEG_ROWS] = e;
            }
            int nterm_all = __popcll(emask);
            int64_t nvals = (n_deltas - j + 1) < (int64_t)nterm_all
                                ? (n_deltas - j + 1) : (int64_t)nterm_all;
            int last_lane;
            if (nvals == (int64_t)nterm_all) {
                last_lane = 63 - __clzll(emask);
            } else {
                uint64_t mm = emask;
                last_lane = 0;
                for (int t = 0; t < nvals; t++) {
                    last_lane = __ffsll((unsigned long long)mm) - 1;
                    mm &= mm - 1;
                }
            }
            v_carry += (uint64_t)__shfl((long long)vinc, last_lane);
            if (dod) d1_carry += (uint64_t)__shfl((long long)s1, last_lane);
            if (j + nterm_all > n_deltas) break;
            j += nterm_all;
            pos += (uint64_t)(64 - __clzll(emask));
        }
    }
}

// ---------------- the scan+aggregate kernel ----------------

enum {
    KF_NEED_VALUES = 1,  // min/max requested -> reconstruct values
    KF_FLOAT = 2,        // float64 field: also accumulate mantissa sum as double
                         // (bits 16-31 of flags carry the configured
                         // decimal exponent for the per-block guard)
    KF_WALK_ONLY = 4,    // process ONLY PF_WALK blocks (the heavy pass of
                         // the light/heavy predicate split below)
};

// wfirst: the (item << 13 | row) key where this wave first ENTERED the
// group (rows that pass predicates and map to the group, before the
// null-field check — matching where the reference's computeKey
// materialises a group, aggregation.go:523).  The global atomicMin over
// these keys is the group's first appearance in storage order, which is
// the reference's first-seen materialisation order (blocks iterate in
// (sid, minTs) order, rows ascend).  ~0 = no entry this flush.
__device__ __forceinline__ void flush_partial(bydb_partial *partials,
                                              int64_t group, uint64_t wsum,
                                              uint64_t wcnt, int64_t wmin,
                                              int64_t wmax, double wsumf,
                                              int lane,
                                              uint64_t *first_seen,
                                              uint64_t wfirst) {
    if (group < 0 || lane != 0) return;
    bydb_partial *p = &partials[group];
    if (first_seen != nullptr && wfirst != ~0ull)
        atomicMin((unsigned long long *)&first_seen[group],
                  (unsigned long long)wfirst);
    if (wsum) atomicAdd((unsigned long long *)&p->sum_i, (unsigned long long)wsum);
    if (wcnt) atomicAdd((unsigned long long *)&p->count, (unsigned long long)wcnt);
    if (wmin != INT64_MAX) atomicMin((long long *)&p->min_i, (long long)wmin);
    if (wmax != INT64_MIN) atomicMax((long long *)&p->max_i, (long long)wmax);
    if (wsumf != 0.0) atomicAdd(&p->sum_f, wsumf);
}

// Fold rows [a, b] of a block's field column: returns the wrapping sum,
// selected count and optional min/max.  Shared by the per-run grouped
// fold; mirrors the main kernel's dispatch.  bw0..bw2: optional
// BITMAP-mode predicate walkers (plain >256-card tag columns,
// k_resolve_plain) — stateless O(1) per-row lookups, so they filter any
// sub-range; RLE walkers never reach here (the run merge intersects
// their runs into [a, b] instead).
__device__ void fold_range(const uint8_t *fstream, uint8_t fenc, int64_t first,
                           uint64_t field_len, int64_t n, int64_t a, int64_t b,
                           bool need_values, int lane, uint64_t *out_sum,
                           int64_t *out_mn, int64_t *out_mx, bool *out_have_mm,
                           uint64_t *out_cnt, const uint8_t *plain_norm,
                           bool raw_f64, DevErr *derr, uint64_t bi,
                           PredWalk *bw0 = nullptr, PredWalk *bw1 = nullptr,
                           PredWalk *bw2 = nullptr) {
    uint64_t nsel = (uint64_t)(b - a + 1);
    uint64_t bsum = 0;
    int64_t bmn = INT64_MAX, bmx = INT64_MIN;
    bool have = false;
    *out_cnt = nsel;
    if (fenc == BYDB_ENC_PLAIN) {
        // null-bearing cell block (host-normalized; see the main Plain
        // fold branch): count excludes nulls
        if (!plain_norm) {
            dev_set_err(derr, DERR_BAD_ENC, bi);
            *out_sum = 0; *out_mn = bmn; *out_mx = bmx; *out_have_mm = false;
            *out_cnt = 0;
            return;
        }
        uint32_t nn = (uint32_t)plain_norm[0] | ((uint32_t)plain_norm[1] << 8) |
                      ((uint32_t)plain_norm[2] << 16) |
                      ((uint32_t)plain_norm[3] << 24);
        const uint64_t *vbm = (const uint64_t *)(plain_norm + 8);
        const uint64_t *cells = vbm + ((uint64_t)nn + 63) / 64;
        uint64_t lsum = 0, lcnt = 0;
        double lsumf = 0.0;
        int64_t lmn = INT64_MAX, lmx = INT64_MIN;
        for (int64_t base = a; base <= b; base += WAVE) {
            int64_t row = base + lane;
            bool ok = row <= b && ((vbm[row >> 6] >> (row & 63)) & 1);
            if (bw0) ok = pred_match_rows(bw0, row, ok) && ok;
            if (bw1) ok = pred_match_rows(bw1, row, ok) && ok;
            if (bw2) ok = pred_match_rows(bw2, row, ok) && ok;
            if (ok) {
                uint64_t u = __builtin_bswap64(cells[row]);
                if (raw_f64) {
                    double dv;
                    __builtin_memcpy(&dv, &u, 8);
                    lsumf += dv;
                    int64_t kv = (u >> 63)
                                     ? (int64_t)(~u ^ 0x8000000000000000ull)
                                     : (int64_t)u;
                    lmn = kv < lmn ? kv : lmn;
                    lmx = kv > lmx ? kv : lmx;
                } else {
                    int64_t v = (u >> 63) ? (int64_t)(u & ~(1ull << 63))
                                          : -(int64_t)((1ull << 63) - u);
                    lsum += (uint64_t)v;
                    lmn = v < lmn ? v : lmn;
                    lmx = v > lmx ? v : lmx;
                }
                lcnt++;
            }
        }
        if (raw_f64) {
            double ds = wave_reduce_addf(lsumf);
            uint64_t db;
            __builtin_memcpy(&db, &ds, 8);
            bsum = db;
        } else {
            bsum = wave_reduce_add(lsum);
        }
        *out_cnt = wave_reduce_add(lcnt);
        bmn = wave_reduce_min(lmn);
        bmx = wave_reduce_max(lmx);
        have = *out_cnt > 0;
        *out_sum = bsum;
        *out_mn = bmn;
        *out_mx = bmx;
        *out_have_mm = have;
        return;
    }
    if (bw0 || bw1 || bw2) {
        // bitmap-filtered fold over the non-Plain encodings: arithmetic
        // progressions via the predicated row walk, varint streams via
        // the walker-aware scan (the closed forms cannot filter rows)
        if (fenc == BYDB_ENC_CONST || fenc == BYDB_ENC_DELTA_CONST) {
            int64_t dd = 0;
            if (fenc == BYDB_ENC_DELTA_CONST) {
                int vl;
                dd = decode_one_varint(fstream, &vl);
            }
            uint64_t psum, pcnt;
            fold_arith_pred(first, dd, a, b, lane, bw0, bw1, bw2, &psum,
                            &pcnt, &bmn, &bmx);
            *out_sum = psum;   // wave_reduce_add already wave-uniform
            *out_cnt = pcnt;
            *out_mn = bmn;
            *out_mx = bmx;
            *out_have_mm = pcnt > 0;
            return;
        }
        if (fenc == BYDB_ENC_DELTA || fenc == BYDB_ENC_DELTA_OF_DELTA) {
            bool dod = fenc == BYDB_ENC_DELTA_OF_DELTA;
            const uint8_t *st = fstream;
            int64_t d1 = 0;
            if (dod) {
                int vl;
                d1 = decode_one_varint(st, &vl);
                st += vl;
            }
            int64_t row1 = (int64_t)((uint64_t)first + (uint64_t)d1);
            uint64_t lsum = 0, lcnt = 0;
            int64_t lmn = INT64_MAX, lmx = INT64_MIN;
            bool need0 = lane == 0 && a <= 0 && 0 <= b;
            bool m0 = true;
            if (bw0) m0 = pred_match_rows(bw0, 0, need0 && m0) && m0;
            if (bw1) m0 = pred_match_rows(bw1, 0, need0 && m0) && m0;
            if (bw2) m0 = pred_match_rows(bw2, 0, need0 && m0) && m0;
            if (need0 && m0) {
                lsum += (uint64_t)first;
                lcnt++;
                lmn = first;
                lmx = first;
            }
            if (dod) {
                bool need1 = lane == 0 && a <= 1 && 1 <= b;
                bool m1 = true;
                if (bw0) m1 = pred_match_rows(bw0, 1, need1 && m1) && m1;
                if (bw1) m1 = pred_match_rows(bw1, 1, need1 && m1) && m1;
                if (bw2) m1 = pred_match_rows(bw2, 1, need1 && m1) && m1;
                if (need1 && m1) {
                    lsum += (uint64_t)row1;
                    lcnt++;
                    lmn = row1 < lmn ? row1 : lmn;
                    lmx = row1 > lmx ? row1 : lmx;
                }
            }
            ScanFold ff;
            scan_stream(st, n - 1, dod, dod ? row1 : first, d1, a, b,
                        lane, &ff, derr, bi, bw0, bw1, bw2);
            lsum += ff.sum;
            lcnt += ff.nsel;
            lmn = ff.mn < lmn ? ff.mn : lmn;
            lmx = ff.mx > lmx ? ff.mx : lmx;
            uint64_t wsum2 = wave_reduce_add(lsum);
            *out_cnt = wave_reduce_add(lcnt);
            *out_mn = wave_reduce_min(lmn);
            *out_mx = wave_reduce_max(lmx);
            *out_have_mm = *out_cnt > 0;
            *out_sum = wsum2;
            return;
        }
        dev_set_err(derr, DERR_BAD_ENC, bi);
        *out_sum = 0;
        *out_mn = bmn;
        *out_mx = bmx;
        *out_have_mm = false;
        *out_cnt = 0;
        return;
    }
    if (fenc == BYDB_ENC_CONST) {
        if (lane == 0) bsum = (uint64_t)first * nsel;
        bmn = bmx = first;
        have = true;
    } else if (fenc == BYDB_ENC_DELTA_CONST) {
        int vl;
        int64_t dd = decode_one_varint(fstream, &vl);
        if (lane == 0) {
            uint64_t si = (uint64_t)(a + b) * nsel / 2;
            bsum = (uint64_t)first * nsel + (uint64_t)dd * si;
        }
        if (need_values) {
            int64_t lmn = INT64_MAX, lmx = INT64_MIN;
            for (int64_t base = a; base <= b; base += 64) {
                int64_t i = base + lane;
                if (i <= b) {
                    int64_t v = (int64_t)((uint64_t)first + (uint64_t)i * (uint64_t)dd);
                    lmn = v < lmn ? v : lmn;
                    lmx = v > lmx ? v : lmx;
                }
            }
            bmn = wave_reduce_min(lmn);
            bmx = wave_reduce_max(lmx);
            have = true;
        }
    } else if (fenc == BYDB_ENC_DELTA) {
        uint64_t acc;
        bool dense = field_len == (uint64_t)(n - 1);
        if (dense) acc = dense_delta_weighted(fstream, n - 1, a, b, lane);
        else acc = fold_delta_weighted_fast(fstream, n - 1, a, b, lane, derr, bi);
        acc = wave_reduce_add(acc);
        if (lane == 0) bsum = (uint64_t)first * nsel + acc;
        if (need_values) {
            if (dense) {
                int64_t psd, psjd;
                dense_region(fstream, 0, a, lane, &psd, &psjd);
                int64_t vr0 = (int64_t)((uint64_t)first +
                                        wave_reduce_add((uint64_t)psd));
                dense_minmax(fstream, a, b, vr0, lane, &bmn, &bmx);
                bmn = vr0 < bmn ? vr0 : bmn;
                bmx = vr0 > bmx ? vr0 : bmx;
            } else {
                ScanFold ff;
                scan_stream(fstream, n - 1, false, first, 0, a, b,
                            lane, &ff, derr, bi, nullptr, nullptr,
                            nullptr);
                int64_t lmn = ff.mn, lmx = ff.mx;
                if (lane == 0 && a <= 0 && 0 <= b) {
                    lmn = first < lmn ? first : lmn;
                    lmx = first > lmx ? first : lmx;
                }
                bmn = wave_reduce_min(lmn);
                bmx = wave_reduce_max(lmx);
            }
            have = true;
        }
    } else if (fenc == BYDB_ENC_DELTA_OF_DELTA) {
        int vl;
        int64_t d1 = decode_one_varint(fstream, &vl);
        uint64_t acc = fold_dod_weighted(fstream + vl, n - 1, a, b, lane, derr, bi);
        acc = wave_reduce_add(acc);
        if (lane == 0) {
            uint64_t si = (uint64_t)(a + b) * nsel / 2;
            bsum = (uint64_t)first * nsel + (uint64_t)d1 * si + acc;
        }
        if (need_values) {
            int64_t row1 = (int64_t)((uint64_t)first + (uint64_t)d1);
            ScanFold ff;
            scan_stream(fstream + vl, n - 1, true, row1, d1, a, b,
                        lane, &ff, derr, bi, nullptr, nullptr,
                        nullptr);
            int64_t lmn = ff.mn, lmx = ff.mx;
            if (lane == 0) {
                if (a <= 0 && 0 <= b) {
                    lmn = first < lmn ? first : lmn;
                    lmx = first > lmx ? first : lmx;
                }
                if (a <= 1 && 1 <= b) {
                    lmn = row1 < lmn ? row1 : lmn;
                    lmx = row1 > lmx ? row1 : lmx;
                }
            }
            bmn = wave_reduce_min(lmn);
            bmx = wave_reduce_max(lmx);
            have = true;
        }
    } else {
        dev_set_err(derr, DERR_BAD_ENC, bi);
    }
    *out_sum = (uint64_t)__shfl((long long)bsum, 0);
    *out_mn = bmn;
    *out_mx = bmx;
    *out_have_mm = have;
}

// Compile-time specialization: the FAST instantiation (no value
// reconstruction, no predicates) sheds the scan/walker machinery and its
// register pressure; the full one keeps everything.  (EN_* are constant
// guards — dead branches are eliminated per instantiation.)
//
// EN_WALK — the light/heavy predicate split.  Row-varying predicate
// walkers (PredWalk x3, ~300 B of per-lane state) force scratch spills
// into the value-scan hot loop even when every block resolves to a
// uniform verdict (entity tags — the dominant case).  With predicates
// the host launches TWO passes on the same stream: a light kernel
// (EN_WALK=false) that folds every PF_CLEAR/PF_SKIP block and skips
// PF_WALK ones, carrying no walker state at all, then a heavy kernel
// (EN_WALK=true, flags & KF_WALK_ONLY) that folds only the PF_WALK
// blocks.  When no block needs a walker the heavy pass is a ~flag-byte
// sweep.
// EN_CLAMP — compile the FindRange row clamp only when the query's
// [min_ts, max_ts] does NOT already cover the whole resident part (the
// host tracks the part's ts extent and selects the instantiation).  The
// clamp's scan machinery otherwise inflates register pressure enough to
// spill scratch reloads into the dense fold loop (~7% on the headline).
template <bool EN_VALUES, bool EN_PREDS, bool EN_GROUPS, bool EN_WALK,
          bool EN_CLAMP>
// Occupancy: 4 waves/SIMD for the value-scan instantiations — requesting
// 6 forces scratch spills into the window loop and measured ~1.6x SLOWER;
// the closed-form-only instantiation fits 6 without spills.
// Occupancy: the clamp-free value-scan instantiations (the analytic
// full-range f64/min-max path) run at EIGHT waves/SIMD — swept 4..8 on
// hardware twice: 7 won before the compile-time full-range flag
// (6.40 -> 5.67 ms on configs[2]), 8 wins after it shed the selection
// branch's registers (-> ~5.25 ms); the extra waves hide the decode's
// dependency bubbles and the VGPR-spill cost stays below the win.  The
// other instantiations keep 4 (more state: clamp / groups), the
// closed-form-only ones 6 (8 measured WORSE there: 2.84 -> 3.25 ms).
__global__ __launch_bounds__(256, (EN_VALUES && !EN_CLAMP && !EN_GROUPS) ? 8 : (EN_VALUES || EN_PREDS || EN_GROUPS) ? 4 : 6) void k_scan_agg_t(
    const uint8_t *__restrict__ payload, const uint8_t *__restrict__ sidecar,
    const bydb_block_desc *__restrict__ blocks,
    int64_t n_blocks, int64_t min_ts, int64_t max_ts, int flags,
    const PredBlock *__restrict__ preds_in, int n_preds,
    const uint64_t *__restrict__ pred_bm,
    const uint8_t *__restrict__ pred_flags,
    const SegEntry *__restrict__ segs_in,
    const GroupBlock *__restrict__ groups_in,
    const uint16_t *__restrict__ gmap_in,
    const uint16_t *__restrict__ gruns_in, int n_gslots, int64_t gm0,
    int64_t gm1, int64_t gm2, int64_t n_groups,
    bydb_partial *__restrict__ partials, DevErr *derr,
    const uint32_t *__restrict__ walk_count,
    uint64_t *__restrict__ first_seen) {
    // heavy pass with no PF_WALK blocks anywhere: one scalar load, done
    if ((flags & KF_WALK_ONLY) && walk_count != nullptr &&
        *walk_count == 0)
        return;
    const int64_t gmul[3] = {gm0, gm1, gm2};
    const PredBlock *preds = EN_PREDS ? preds_in : nullptr;
    const SegEntry *segs = EN_VALUES ? segs_in : nullptr;
    const GroupBlock *groups = EN_GROUPS ? groups_in : nullptr;
    const int lane = threadIdx.x & 63;
    const int wave_in_block = threadIdx.x >> 6;
    int64_t wave_id = (int64_t)blockIdx.x * (blockDim.x >> 6) + wave_in_block;
    int64_t n_waves = (int64_t)gridDim.x * (blockDim.x >> 6);

    // per-wave partial (flushed on group change / at the end)
    int64_t cur_group = -1;
    uint64_t wsum = 0, wcnt = 0;
    uint64_t wfirst = ~0ull;   // first (item,row) entry key of cur_group
    int64_t wmin = INT64_MAX, wmax = INT64_MIN;
    double wsumf = 0.0;

    // with a segment index, the work item is a (block, segment) pair —
    // 8x the parallel units on the value-scan path, no serial chain.
    // Strided assignment (not contiguous ranges): predicates skip blocks
    // in long runs, and contiguous ranges turn those runs into idle waves.
    const int64_t n_items = segs ? n_blocks * MAX_SEGS : n_blocks;
    for (int64_t wi = wave_id; wi < n_items; wi += n_waves) {
        const int64_t bi = segs ? wi / MAX_SEGS : wi;
        const int seg = segs ? (int)(wi % MAX_SEGS) : 0;
        // per-item cold-start: issue the independent scalar loads
        // (descriptor, predicate flag, segment entries) back-to-back so
        // the item pays ONE memory latency, not a chain of them
        const bydb_block_desc *bd = &blocks[bi];
        const uint8_t pf_pre =
            (EN_PREDS && preds != nullptr) ? pred_flags[bi] : PF_CLEAR;
        uint32_t seg0_off = SEG_INELIGIBLE;
        SegEntry e_pre;
        e_pre.byte_off = 0; e_pre._pad = 0; e_pre.v_start = 0;
        e_pre.d1_start = 0;
        if (segs) {
            seg0_off = segs[bi * MAX_SEGS].byte_off;
            e_pre = segs[bi * MAX_SEGS + seg];
        }
        const bool seg_eligible = segs && seg0_off != SEG_INELIGIBLE;
        if (seg != 0 && !seg_eligible) continue;
        // light/heavy split routing (see the template comment): the heavy
        // pass owns exactly the PF_WALK blocks, the light pass everything
        // else — checked before any clamp work so the off-pass blocks
        // cost one flag byte
        if (EN_PREDS && preds != nullptr) {
            if (flags & KF_WALK_ONLY) {
                if (pf_pre != PF_WALK) continue;
            } else if (!EN_WALK && pf_pre == PF_WALK) {
                continue;
            }
        }
        const int64_t n = (int64_t)bd->count;
        if (seg != 0 && (int64_t)seg * SEG_ROWS + 1 > n - 1) continue;
        const int64_t ts_min = bd->ts_min, ts_max = bd->ts_max;
        // float64 blocks fold in the decimal-int domain; a block whose
        // exponent EXCEEDS the session's rescales its partials by
        // 10^(block_exp - cfg_exp) at the merge below (exact for the
        // monotone min/max mantissas; the sum factor is a double).  The
        // session exponent must be the part's minimum exponent — a block
        // BELOW it cannot be represented and errors loudly.
        int fdiff = 0;
        bool raw_f64 = false;
        if (flags & KF_FLOAT) {
            int16_t cfge = (int16_t)(uint16_t)((uint32_t)flags >> 16);
            if (cfge == INT16_MIN) {
                // raw-float session (nullable f64, IEEE-754 cells):
                // every block must be a Plain cell block
                raw_f64 = true;
                if (bd->field_enc != BYDB_ENC_PLAIN) {
                    dev_set_err(derr, DERR_EXP_MISMATCH, (uint64_t)bi);
                    continue;
                }
            } else if (bd->field_enc == BYDB_ENC_PLAIN) {
                // raw cells inside a decimal session: domains clash
                dev_set_err(derr, DERR_EXP_MISMATCH, (uint64_t)bi);
                continue;
            } else {
                fdiff = (int)bd->exp - (int)cfge;
                if (fdiff < 0 || fdiff > 18) {
                    dev_set_err(derr, DERR_EXP_MISMATCH, (uint64_t)bi);
                    continue;
                }
                // mantissa-sum overflow guard: the engine
                // carries each block's decimal-int mantissa sum in int64
                // (exact, then one double convert); the reference's
                // per-row float64 adds never wrap.  A block whose
                // |firstValue| alone implies the sum exceeds int64
                // (|mean| > 2^63/8192 ~ 1.1e15 — the realistic overflow
                // mode is a large baseline) errs loudly instead of
                // returning a silently wrong sum.  Delta-driven escapes
                // from a small firstValue remain a documented limit.
                uint64_t afirst = bd->field_first < 0
                                      ? (uint64_t)(-(uint64_t)bd->field_first)
                                      : (uint64_t)bd->field_first;
                if (afirst > (1ull << 63) / (uint64_t)(bd->count + 1u)) {
                    dev_set_err(derr, DERR_F64_SUM_OVF, (uint64_t)bi);
                    continue;
                }
            }
        }

        // ---- row clamp (timestamp.FindRange, range.go:143-170) ----
        int64_t r0 = 0, r1 = n - 1;
        if (ts_min > ts_max) { dev_set_err(derr, DERR_DESC_TS, (uint64_t)bi); continue; }
        if (EN_CLAMP && !(min_ts <= ts_min && max_ts >= ts_max)) {
            if (ts_min > max_ts || ts_max < min_ts) continue;  // no overlap
            uint8_t tenc = bd->ts_enc_with_version;
            // common type (encoding.GetCommonType)
            uint8_t tc = tenc >= BYDB_ENC_CONST_WV && tenc <= BYDB_ENC_DELTA_OF_DELTA_WV
                             ? tenc - 4 : tenc;
            const uint8_t *tstream = payload + bd->ts_off;
            if (tc == BYDB_ENC_CONST) {
                // all == ts_min; overlap already implies in-range
            } else if (tc == BYDB_ENC_DELTA_CONST) {
                int vl;
                int64_t dd = decode_one_varint(tstream, &vl);
                if (dd <= 0) { dev_set_err(derr, DERR_DESC_TS, (uint64_t)bi); continue; }
                // first i with ts_min + i*dd >= min_ts
                if (min_ts > ts_min)
                    r0 = (min_ts - ts_min + dd - 1) / dd;
                if (max_ts < ts_max)
                    r1 = (max_ts - ts_min) / dd;
                if (r1 > n - 1) r1 = n - 1;
            } else if (tc == BYDB_ENC_DELTA || tc == BYDB_ENC_DELTA_OF_DELTA) {
                bool dod = tc == BYDB_ENC_DELTA_OF_DELTA;
                const uint8_t *s = tstream;
                int64_t d1 = 0;
                int64_t vfirst = ts_min;
                if (dod) {
                    int vl;
                    d1 = decode_one_varint(s, &vl);
                    s += vl;
                }
                int64_t row1 = (int64_t)((uint64_t)ts_min + (uint64_t)d1);
                uint64_t nlo, nhi;
                clamp_count_stream(s, n - 1, dod, dod ? row1 : vfirst, d1,
                                   min_ts, max_ts, lane, &nlo, &nhi, derr,
                                   (uint64_t)bi);
                // rows 0 (and 1 for dod) were not in the stream
                if (ts_min < min_ts) nlo++;
                if (ts_min > max_ts) nhi++;
                if (dod) {
                    if (row1 < min_ts) nlo++;
                    if (row1 > max_ts) nhi++;
                }
                r0 = (int64_t)nlo;
                r1 = n - 1 - (int64_t)nhi;
            }
            if (r0 > r1) continue;
        }
        const uint64_t nsel = (uint64_t)(r1 - r0 + 1);

        // ---- per-row tag predicates (conjunctive, dictionary codes) ----
        // One combined flag byte decides the common cases; only PF_WALK
        // blocks pay for walker setup.
        PredWalk pw0, pw1, pw2;
        PredWalk *wp0 = nullptr, *wp1 = nullptr, *wp2 = nullptr;
        if (preds != nullptr) {
            const uint8_t pf = pf_pre;
            if (pf == PF_SKIP) continue;
            if (pf == PF_ERR) {
                dev_set_err(derr, DERR_BAD_ENC, (uint64_t)bi);
                continue;
            }
            if (EN_WALK && pf == PF_WALK) {
                bool skip_block = false;
#pragma unroll
                for (int sl = 0; sl < 3; sl++) {
                    if (sl >= n_preds || skip_block) continue;
                    PredBlock pb = preds[(int64_t)sl * n_blocks + bi];
                    if (pb.disabled) continue;   // empty slot: no constraint
                    if (!pb.active) { skip_block = true; continue; }
                    PredWalk *w = sl == 0 ? &pw0 : sl == 1 ? &pw1 : &pw2;
                    pred_init(w, payload, sidecar, &pb, pred_bm);
                    if (pb.uniform) {
                        // block-uniform slot inside a varying conjunction
                        pred_advance(w);
                        if (!w->run_match) skip_block = true;
                        // uniform + match: walker dropped, slot is a no-op
                    } else {
                        if (sl == 0) wp0 = w;
                        else if (sl == 1) wp1 = w;
                        else wp2 = w;
                    }
                }
                if (skip_block) continue;
            }
            // PF_CLEAR: every slot uniform-match -> fold unpredicated
        }
        bool pred_on = wp0 || wp1 || wp2;

        // ---- per-row group-by on dictionary tags (EN_GROUPS) ----
        // Composite keys over up to 3 tag slots: gid = g0 + n0*g1 +
        // n0*n1*g2 (computeKey's multi-component key, aggregation.go:523).
        // All-uniform blocks (entity tags) take the fast fold paths; any
        // row-varying slot folds per intersection of the RLE runs.
        int64_t block_group = (int64_t)bd->group_code;
        if (!EN_GROUPS && block_group >= n_groups) {
            dev_set_err(derr, DERR_GROUP_RANGE, (uint64_t)bi);
            continue;
        }
        if (EN_GROUPS && groups) {
            GroupBlock gb[3];
            bool varying = false, skip_blk = false, errd = false;
            for (int sl = 0; sl < n_gslots; sl++) {
                gb[sl] = groups[(int64_t)sl * n_blocks + bi];
                if (gb[sl].err) errd = true;
                else if (gb[sl].uniform_gid == GID_NONE) skip_blk = true;
                else if (gb[sl].uniform_gid == GID_VARYING) varying = true;
            }
            if (errd) { dev_set_err(derr, DERR_BAD_ENC, (uint64_t)bi); continue; }
            if (skip_blk) continue;   // nil/unmapped on a grouped slot
            if (!varying) {
                block_group = 0;
                for (int sl = 0; sl < n_gslots; sl++)
                    block_group += (int64_t)gb[sl].uniform_gid * gmul[sl];
            } else {
                // row-varying predicates join the run merge below:
                // RLE walkers as additional run cursors; BITMAP-mode
                // (plain-tag) walkers have no runs, so they filter
                // per-row inside fold_range instead (stateless O(1)
                // lookups — heavy-pass (EN_WALK) instantiations only)
                PredWalk *bw0 = nullptr, *bw1 = nullptr, *bw2 = nullptr;
                if (EN_WALK) {
                    if (wp0 && wp0->bm) { bw0 = wp0; wp0 = nullptr; }
                    if (wp1 && wp1->bm) { bw1 = wp1; wp1 = nullptr; }
                    if (wp2 && wp2->bm) { bw2 = wp2; wp2 = nullptr; }
                }
                if (seg != 0) continue;  // whole block folded at seg 0
                const uint8_t *gstream = payload + bd->field_off;
                const uint8_t *plain_norm =
                    (bd->field_enc == BYDB_ENC_PLAIN &&
                     (bd->field_off & TAG_SIDECAR_BIT))
                        ? sidecar + (bd->field_off & ~TAG_SIDECAR_BIT)
                        : nullptr;
                // per-slot run cursors (uniform slots are one infinite
                // run); plain slots (width 0xFF) walk (gid,count) u16
                // pairs from the run arena instead of bit-packed RLE
                const uint8_t *gsrc[3];
                uint64_t bit[3];
                uint32_t ent[3];
                int64_t run_hi_s[3];
                int64_t gid_s[3];
                bool bad_plain = false;
                for (int sl = 0; sl < n_gslots; sl++) {
                    gsrc[sl] = (gb[sl].rle_bit_off & TAG_SIDECAR_BIT)
                                   ? sidecar : payload;
                    if (gb[sl].uniform_gid != GID_VARYING) {
                        gid_s[sl] = (int64_t)gb[sl].uniform_gid;
                        run_hi_s[sl] = n;
                        ent[sl] = gb[sl].nentries;  // exhausted
                        bit[sl] = 0;
                    } else {
                        if (gb[sl].width == 0xFF && gb[sl].nentries == 0)
                            bad_plain = true;  // resolve pass never ran
                        bit[sl] = gb[sl].rle_bit_off & ~TAG_SIDECAR_BIT;
                        ent[sl] = 0;
                        run_hi_s[sl] = 0;
                        gid_s[sl] = -1;
                    }
                }
                if (bad_plain) {
                    dev_set_err(derr, DERR_BAD_ENC, (uint64_t)bi);
                    continue;
                }
                auto advance = [&](int sl) {
                    if (ent[sl] + 1 >= gb[sl].nentries) {
                        run_hi_s[sl] = n;
                        gid_s[sl] = -1;  // RLE exhausted: no group
                        return;
                    }
                    if (gb[sl].width == 0xFF) {
                        const uint16_t *rp = gruns_in + gb[sl].rle_bit_off;
                        uint16_t g16 = rp[ent[sl]];
                        uint16_t cnt16 = rp[ent[sl] + 1];
                        ent[sl] += 2;
                        gid_s[sl] = g16 == 0xFFFFu ? -1 : (int64_t)g16;
                        run_hi_s[sl] += (int64_t)cnt16;
                        return;
                    }
                    uint64_t code = rd_bits_be(gsrc[sl], bit[sl], gb[sl].width);
                    bit[sl] += gb[sl].width;
                    uint64_t cnt = rd_bits_be(gsrc[sl], bit[sl], gb[sl].width);
                    bit[sl] += gb[sl].width;
                    ent[sl] += 2;
                    uint16_t m = (gmap_in + gb[sl].map_off)[code < 256 ? code : 0];
                    gid_s[sl] = m == 0xFFFFu ? -1 : (int64_t)m;
                    run_hi_s[sl] += (int64_t)cnt;
                };
                for (int sl = 0; sl < n_gslots; sl++)
                    if (gb[sl].uniform_gid == GID_VARYING) advance(sl);
                int64_t lo = 0;
                while (lo < n) {
                    // predicate cursors advance lazily to cover `lo`
                    if (wp0) while (wp0->run_hi <= lo) pred_advance(wp0);
                    if (wp1) while (wp1->run_hi <= lo) pred_advance(wp1);
                    if (wp2) while (wp2->run_hi <= lo) pred_advance(wp2);
                    int64_t hi = n;
                    for (int sl = 0; sl < n_gslots; sl++)
                        hi = run_hi_s[sl] < hi ? run_hi_s[sl] : hi;
                    if (wp0 && wp0->run_hi < hi) hi = wp0->run_hi;
                    if (wp1 && wp1->run_hi < hi) hi = wp1->run_hi;
                    if (wp2 && wp2->run_hi < hi) hi = wp2->run_hi;
                    int64_t aa = lo > r0 ? lo : r0;
                    int64_t bb2 = (hi - 1) < r1 ? (hi - 1) : r1;
                    int64_t comp = 0;
                    bool ok = true;
                    for (int sl = 0; sl < n_gslots; sl++) {
                        if (gid_s[sl] < 0) ok = false;
                        else comp += gid_s[sl] * gmul[sl];
                    }
                    if (wp0) ok = ok && wp0->run_match;
                    if (wp1) ok = ok && wp1->run_match;
                    if (wp2) ok = ok && wp2->run_match;
                    if (ok && aa <= bb2) {
                        uint64_t rsum, rcnt;
                        int64_t rmn, rmx;
                        bool rhave;
                        fold_range(gstream, bd->field_enc, bd->field_first,
                                   bd->field_len, n, aa, bb2,
                                   EN_VALUES && (flags & KF_NEED_VALUES), lane,
                                   &rsum, &rmn, &rmx, &rhave, &rcnt,
                                   plain_norm, raw_f64, derr, (uint64_t)bi,
                                   bw0, bw1, bw2);
                        double rfs = 1.0;
                        if ((flags & KF_FLOAT) && fdiff) {
                            int64_t mfac = c_pow10i[fdiff];
                            rfs = (double)mfac;
                            if (rhave) {
                                int64_t smin, smax;
                                if (__builtin_mul_overflow(rmn, mfac, &smin) ||
                                    __builtin_mul_overflow(rmx, mfac, &smax)) {
                                    dev_set_err(derr, DERR_EXP_MISMATCH,
                                                (uint64_t)bi);
                                    rhave = false;
                                } else {
                                    rmn = smin;
                                    rmx = smax;
                                }
                            }
                        }
                        if (comp != cur_group) {
                            flush_partial(partials, cur_group, wsum, wcnt,
                                          wmin, wmax, wsumf, lane,
                                          first_seen, wfirst);
                            cur_group = comp;
                            wsum = 0; wcnt = 0; wmin = INT64_MAX;
                            wmax = INT64_MIN; wsumf = 0;
                            wfirst = ((uint64_t)wi << 13) | (uint64_t)aa;
                        } else if (wfirst == ~0ull) {
                            wfirst = ((uint64_t)wi << 13) | (uint64_t)aa;
                        }
                        wsum += rsum;
                        wcnt += rcnt;
                        if (rhave) {
                            wmin = rmn < wmin ? rmn : wmin;
                            wmax = rmx > wmax ? rmx : wmax;
                        }
                        if (flags & KF_FLOAT) {
                            if (raw_f64) {
                                double db;
                                uint64_t ub = rsum;
                                __builtin_memcpy(&db, &ub, 8);
                                wsumf += db;
                            } else {
                                wsumf += (double)(int64_t)rsum * rfs;
                            }
                        }
                    }
                    for (int sl = 0; sl < n_gslots; sl++)
                        if (run_hi_s[sl] == hi && gb[sl].uniform_gid == GID_VARYING)
                            advance(sl);
                    if (hi <= lo) break;  // safety against stuck cursors
                    lo = hi;
                }
                continue;
            }
        }
        const bool scan_path =
            EN_VALUES && (flags & KF_NEED_VALUES) &&
            ((bd->field_enc == BYDB_ENC_DELTA &&
              bd->field_len != (uint64_t)(n - 1)) ||
             bd->field_enc == BYDB_ENC_DELTA_OF_DELTA);
        const bool use_seg = seg_eligible && scan_path && !pred_on;
        if (!use_seg && seg != 0) continue;

        // ---- field fold ----
        const uint8_t fenc = bd->field_enc;
        const int64_t first = bd->field_first;
        const uint8_t *fstream = payload + bd->field_off;
        uint64_t bsum = 0;          // block sum (wrapping)
        int64_t bmin = INT64_MAX, bmax = INT64_MIN;
        bool have_minmax = false;

        uint64_t nsel_eff = nsel;
        if (pred_on && (fenc == BYDB_ENC_CONST || fenc == BYDB_ENC_DELTA_CONST)) {
            int64_t dd = 0;
            if (fenc == BYDB_ENC_DELTA_CONST) {
                int vl;
                dd = decode_one_varint(fstream, &vl);
            }
            uint64_t psum, pcnt;
            fold_arith_pred(first, dd, r0, r1, lane, wp0, wp1, wp2, &psum,
                            &pcnt, &bmin, &bmax);
            bsum = lane == 0 ? psum : 0;
            nsel_eff = pcnt;
            have_minmax = (EN_VALUES && (flags & KF_NEED_VALUES)) && pcnt > 0;
        } else if (pred_on &&
                   (fenc == BYDB_ENC_DELTA || fenc == BYDB_ENC_DELTA_OF_DELTA)) {
            bool dod = fenc == BYDB_ENC_DELTA_OF_DELTA;
            const uint8_t *s = fstream;
            int64_t d1 = 0;
            if (dod) {
                int vl;
                d1 = decode_one_varint(s, &vl);
                s += vl;
            }
            int64_t row1 = (int64_t)((uint64_t)first + (uint64_t)d1);
            // rows outside the stream first (walker rows must ascend)
            uint64_t lsum = 0, lcnt = 0;
            int64_t lmn = INT64_MAX, lmx = INT64_MIN;
            bool need0 = lane == 0 && r0 <= 0 && 0 <= r1;
            bool m0 = true;
            if (wp0) m0 = pred_match_rows(wp0, 0, need0 && m0) && m0;
            if (wp1) m0 = pred_match_rows(wp1, 0, need0 && m0) && m0;
            if (wp2) m0 = pred_match_rows(wp2, 0, need0 && m0) && m0;
            if (need0 && m0) { lsum += (uint64_t)first; lcnt++;
                lmn = first; lmx = first; }
            if (dod) {
                bool need1 = lane == 0 && r0 <= 1 && 1 <= r1;
                bool m1 = true;
                if (wp0) m1 = pred_match_rows(wp0, 1, need1 && m1) && m1;
                if (wp1) m1 = pred_match_rows(wp1, 1, need1 && m1) && m1;
                if (wp2) m1 = pred_match_rows(wp2, 1, need1 && m1) && m1;
                if (need1 && m1) { lsum += (uint64_t)row1; lcnt++;
                    lmn = row1 < lmn ? row1 : lmn; lmx = row1 > lmx ? row1 : lmx; }
            }
            ScanFold ff;
            scan_stream(s, n - 1, dod, dod ? row1 : first, d1, r0, r1,
                        lane, &ff, derr, (uint64_t)bi, wp0, wp1, wp2,
                        !EN_CLAMP);
            lsum += ff.sum;
            lcnt += ff.nsel;
            lmn = ff.mn < lmn ? ff.mn : lmn;
            lmx = ff.mx > lmx ? ff.mx : lmx;
            bsum = wave_reduce_add(lsum);
            nsel_eff = wave_reduce_add(lcnt);
            bmin = wave_reduce_min(lmn);
            bmax = wave_reduce_max(lmx);
            have_minmax = nsel_eff > 0;
            bsum = lane == 0 ? bsum : 0;
        } else if (fenc == BYDB_ENC_CONST) {
            if (lane == 0) bsum = (uint64_t)first * nsel;
            bmin = bmax = first;
            have_minmax = true;
        } else if (fenc == BYDB_ENC_DELTA_CONST) {
            int vl;
            int64_t dd = decode_one_varint(fstream, &vl);
            if (lane == 0) {
                // sum_{i=r0..r1}(first + i*dd) = nsel*first + dd * sum i
                uint64_t si = (uint64_t)(r0 + r1) * nsel / 2;
                bsum = (uint64_t)first * nsel + (uint64_t)dd * si;
            }
            if (EN_VALUES && (flags & KF_NEED_VALUES)) {
                // reconstruct min/max exactly (wrap-safe): walk rows by lanes
                int64_t lmn = INT64_MAX, lmx = INT64_MIN;
                for (int64_t base = r0; base <= r1; base += WAVE) {
                    int64_t i = base + lane;
                    if (i <= r1) {
                        int64_t v = (int64_t)((uint64_t)first + (uint64_t)i * (uint64_t)dd);
                        lmn = v < lmn ? v : lmn;
                        lmx = v > lmx ? v : lmx;
                    }
                }
                bmin = wave_reduce_min(lmn);
                bmax = wave_reduce_max(lmx);
                have_minmax = true;
            }
        } else if (fenc == BYDB_ENC_DELTA || fenc == BYDB_ENC_DELTA_OF_DELTA) {
            bool dod = fenc == BYDB_ENC_DELTA_OF_DELTA;
            if (!(EN_VALUES && (flags & KF_NEED_VALUES))) {
                uint64_t acc;
                if (!dod) {
                    if (bd->field_len == (uint64_t)(n - 1)) {
                        // stream length == delta count -> every varint is
                        // one byte: dense dot4 fold
                        acc = dense_delta_weighted(fstream, n - 1, r0, r1,
                                                   lane);
                    } else {
                        acc = fold_delta_weighted_fast(fstream, n - 1, r0, r1,
                                                       lane, derr,
                                                       (uint64_t)bi);
                    }
                    acc = wave_reduce_add(acc);
                    if (lane == 0) {
                        bsum = (uint64_t)first * nsel + acc;
                    }
                } else {
                    int vl;
                    int64_t d1 = decode_one_varint(fstream, &vl);
                    acc = fold_dod_weighted(fstream + vl, n - 1, r0, r1, lane,
                                            derr, (uint64_t)bi);
                    acc = wave_reduce_add(acc);
                    if (lane == 0) {
                        // nsel*first + d1 * sum_{i=r0..r1} i + weighted d2
                        uint64_t si = (uint64_t)(r0 + r1) * nsel / 2;
                        bsum = (uint64_t)first * nsel + (uint64_t)d1 * si + acc;
                    }
                }
            } else if (use_seg) {
                // (SegEntry preloaded at item start)
                // segment-parallel value scan: this wave folds only rows
                // [seg*SEG_ROWS+1 .. min((seg+1)*SEG_ROWS, n-1)] (+ rows 0
                // and, for delta-of-delta, 1 on segment 0), starting from
                // the indexed byte offset and carries — no cross-segment
                // dependency.  For delta-of-delta, segment k >= 1 relabels
                // absolute row (k*SEG_ROWS - 1 + j'), so j' = 2 lands on
                // the segment's first row and e.v_start/e.d1_start are the
                // carries at row k*SEG_ROWS.
                const SegEntry e = e_pre;
                int64_t sj_lo = (int64_t)seg * SEG_ROWS + 1;
                int64_t sj_hi = ((int64_t)seg + 1) * SEG_ROWS;
                if (sj_hi > n - 1) sj_hi = n - 1;
                int64_t a = sj_lo > r0 ? sj_lo : r0;
                int64_t b = sj_hi < r1 ? sj_hi : r1;
                bool handle0 = seg == 0 && r0 <= 0 && 0 <= r1;
                bool handle1 = dod && seg == 0 && r0 <= 1 && 1 <= r1;
                if (dod && seg == 0 && a < 2) a = 2;
                if (a > b && !handle0 && !handle1) continue;
                uint64_t lsum = 0, lcnt = 0;
                int64_t lmn = INT64_MAX, lmx = INT64_MIN;
                if (a <= b) {
                    ScanFold ff;
                    int64_t rel0 = dod ? ((int64_t)seg * SEG_ROWS -
                                          (seg ? 1 : 0))
                                       : (int64_t)seg * SEG_ROWS;
                    scan_stream(fstream + e.byte_off, b - rel0, dod,
                                e.v_start, e.d1_start, a - rel0, b - rel0,
                                lane, &ff, derr, (uint64_t)bi, nullptr,
                                nullptr, nullptr, !EN_CLAMP);
                    lsum = ff.sum;
                    lcnt = ff.nsel;
                    lmn = ff.mn;
                    lmx = ff.mx;
                }
                if (lane == 0 && handle0) {
                    lsum += (uint64_t)first;
                    lcnt++;
                    lmn = first < lmn ? first : lmn;
                    lmx = first > lmx ? first : lmx;
                }
                if (lane == 0 && handle1) {
                    int64_t row1 = e.v_start;  // seg 0: v at row 1
                    lsum += (uint64_t)row1;
                    lcnt++;
                    lmn = row1 < lmn ? row1 : lmn;
                    lmx = row1 > lmx ? row1 : lmx;
                }
                bsum = wave_reduce_add(lsum);
                nsel_eff = wave_reduce_add(lcnt);
                bmin = wave_reduce_min(lmn);
                bmax = wave_reduce_max(lmx);
                have_minmax = nsel_eff > 0;
                bsum = lane == 0 ? bsum : 0;
            } else if (!dod && bd->field_len == (uint64_t)(n - 1)) {
                // all-1-byte delta stream: dense weighted sum + dense
                // min/max (no serial window chain)
                uint64_t acc = dense_delta_weighted(fstream, n - 1, r0, r1,
                                                    lane);
                acc = wave_reduce_add(acc);
                bsum = (uint64_t)first * nsel + acc;
                bsum = lane == 0 ? bsum : 0;
                // v at delta index r0 (v_{r0}) = first + sum of bytes [0,r0)
                int64_t psd, psjd;
                dense_region(fstream, 0, r0, lane, &psd, &psjd);
                int64_t vr0 = (int64_t)((uint64_t)first +
                                        wave_reduce_add((uint64_t)psd));
                dense_minmax(fstream, r0, r1, vr0, lane, &bmin, &bmax);
                bmin = vr0 < bmin ? vr0 : bmin;  // row r0 itself
                bmax = vr0 > bmax ? vr0 : bmax;
                have_minmax = true;
            } else {
                const uint8_t *s = fstream;
                int64_t d1 = 0;
                if (dod) {
                    int vl;
                    d1 = decode_one_varint(s, &vl);
                    s += vl;
                }
                int64_t row1 = (int64_t)((uint64_t)first + (uint64_t)d1);
                ScanFold ff;
                scan_stream(s, n - 1, dod, dod ? row1 : first, d1, r0, r1,
                            lane, &ff, derr, (uint64_t)bi, nullptr, nullptr,
                            nullptr, !EN_CLAMP);
                uint64_t lsum = ff.sum;
                int64_t lmn = ff.mn, lmx = ff.mx;
                // rows outside the stream: row 0 (value=first) and, for
                // dod, row 1 (value=row1)
                if (lane == 0) {
                    if (r0 <= 0 && 0 <= r1) {
                        lsum += (uint64_t)first;
                        lmn = first < lmn ? first : lmn;
                        lmx = first > lmx ? first : lmx;
                    }
                    if (dod && r0 <= 1 && 1 <= r1) {
                        lsum += (uint64_t)row1;
                        lmn = row1 < lmn ? row1 : lmn;
                        lmx = row1 > lmx ? row1 : lmx;
                    }
                }
                bsum = wave_reduce_add(lsum);
                bmin = wave_reduce_min(lmn);
                bmax = wave_reduce_max(lmx);
                have_minmax = true;
            }
        } else if (fenc == BYDB_ENC_PLAIN) {
            // null-bearing Plain column, host-normalized into the sidecar
            // ([u32 n][pad][validity bitmap][8-B sign-flip cells]): fold
            // valid selected rows, decoding cells in-lane
            // (convert/number.go:95-108); nulls drop from count and every
            // aggregate (aggregation.go:310).
            if (!(bd->field_off & TAG_SIDECAR_BIT)) {
                dev_set_err(derr, DERR_BAD_ENC, (uint64_t)bi);
                continue;
            }
            const uint8_t *fp = sidecar + (bd->field_off & ~TAG_SIDECAR_BIT);
            uint32_t nn = (uint32_t)fp[0] | ((uint32_t)fp[1] << 8) |
                          ((uint32_t)fp[2] << 16) | ((uint32_t)fp[3] << 24);
            if (nn != (uint32_t)n) {
                dev_set_err(derr, DERR_BAD_ENC, (uint64_t)bi);
                continue;
            }
            const uint64_t *vbm = (const uint64_t *)(fp + 8);
            const uint64_t *cells = vbm + ((uint64_t)nn + 63) / 64;
            uint64_t lsum = 0, lcnt = 0;
            double lsumf = 0.0;
            int64_t lmn = INT64_MAX, lmx = INT64_MIN;
            for (int64_t base = r0; base <= r1; base += WAVE) {
                int64_t row = base + lane;
                bool ok = row <= r1 &&
                          ((vbm[row >> 6] >> (row & 63)) & 1);
                if (wp0) ok = pred_match_rows(wp0, row, ok) && ok;
                if (wp1) ok = pred_match_rows(wp1, row, ok) && ok;
                if (wp2) ok = pred_match_rows(wp2, row, ok) && ok;
                if (ok) {
                    uint64_t c = cells[row];
                    // cells are big-endian on the wire
                    uint64_t u = __builtin_bswap64(c);
                    if (raw_f64) {
                        // IEEE-754 bits: double sum + ordered-bits
                        // min/max (Float64ToOrderedBytes map,
                        // convert/number.go:148-157)
                        double dv;
                        __builtin_memcpy(&dv, &u, 8);
                        lsumf += dv;
                        // order-preserving map INTO THE SIGNED domain
                        // (min/max compare as int64): positives keep raw
                        // bits, negatives flip magnitude and sign
                        int64_t kv =
                            (u >> 63)
                                ? (int64_t)(~u ^ 0x8000000000000000ull)
                                : (int64_t)u;
                        lmn = kv < lmn ? kv : lmn;
                        lmx = kv > lmx ? kv : lmx;
                    } else {
                        int64_t v = (u >> 63)
                                        ? (int64_t)(u & ~(1ull << 63))
                                        : -(int64_t)((1ull << 63) - u);
                        lsum += (uint64_t)v;
                        lmn = v < lmn ? v : lmn;
                        lmx = v > lmx ? v : lmx;
                    }
                    lcnt++;
                }
            }
            if (raw_f64) {
                double ds = wave_reduce_addf(lsumf);
                uint64_t db;
                __builtin_memcpy(&db, &ds, 8);
                bsum = lane == 0 ? db : 0;
            } else {
                bsum = wave_reduce_add(lsum);
                bsum = lane == 0 ? bsum : 0;
            }
            nsel_eff = wave_reduce_add(lcnt);
            bmin = wave_reduce_min(lmn);
            bmax = wave_reduce_max(lmx);
            have_minmax = nsel_eff > 0;
        } else {
            dev_set_err(derr, DERR_BAD_ENC, (uint64_t)bi);
            continue;
        }

        // rows 0 (+1 for dod) fast path bookkeeping: the weighted folds
        // above already include first (and d1) in their closed forms.

        // ---- accumulate into the per-wave register partial ----
        // One atomic set per BLOCK on a single group serialises on one
        // cacheline (~88 atomics/us on one word) and dominates the whole
        // scan; instead fold block totals into per-wave registers and
        // flush only when the group changes or the wave is done.
        bsum = (uint64_t)__shfl((long long)bsum, 0);  // lane0 holds closed forms
        double fscale = 1.0;
        if ((flags & KF_FLOAT) && fdiff) {
            int64_t mfac = c_pow10i[fdiff];
            fscale = (double)mfac;
            if (have_minmax) {
                int64_t smin, smax;
                if (__builtin_mul_overflow(bmin, mfac, &smin) ||
                    __builtin_mul_overflow(bmax, mfac, &smax)) {
                    dev_set_err(derr, DERR_EXP_MISMATCH, (uint64_t)bi);
                    continue;
                }
                bmin = smin;
                bmax = smax;
            }
        }
        if (block_group != cur_group) {
            flush_partial(partials, cur_group, wsum, wcnt, wmin, wmax, wsumf,
                          lane, first_seen, wfirst);
            cur_group = block_group;
            wsum = 0; wcnt = 0; wmin = INT64_MAX; wmax = INT64_MIN; wsumf = 0;
            wfirst = ((uint64_t)wi << 13) | (uint64_t)r0;
        } else if (wfirst == ~0ull) {
            wfirst = ((uint64_t)wi << 13) | (uint64_t)r0;
        }
        wsum += bsum;
        wcnt += nsel_eff;
        if (have_minmax) {
            wmin = bmin < wmin ? bmin : wmin;
            wmax = bmax > wmax ? bmax : wmax;
        }
        if (flags & KF_FLOAT) {
            if (raw_f64) {
                double db;
                uint64_t ub = bsum;
                __builtin_memcpy(&db, &ub, 8);
                wsumf += db;
            } else {
                wsumf += (double)(int64_t)bsum * fscale;
            }
        }
    }
    flush_partial(partials, cur_group, wsum, wcnt, wmin, wmax, wsumf, lane,
                  first_seen, wfirst);
}

// init kernel: set partials to the fold identity (Map.Reset, function.go)
__global__ void k_reset_partials(bydb_partial *p, int64_t n) {
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i < n) {
        p[i].sum_i = 0;
        p[i].count = 0;
        p[i].min_i = INT64_MAX;
        p[i].max_i = INT64_MIN;
        p[i].sum_f = 0.0;
        p[i]._pad = 0.0;
    }
}

// ===================== host session =====================

struct bydb_session {
    int device = 0;
    hipStream_t stream = nullptr;
    hipEvent_t ev_start = nullptr, ev_stop = nullptr;
    std::string err;
    // part residency
    uint8_t *d_payload = nullptr;
    uint64_t payload_cap = 0, payload_len = 0;
    int64_t part_ts_min = INT64_MAX, part_ts_max = INT64_MIN;
    bydb_block_desc *d_blocks = nullptr;
    int64_t blocks_cap = 0, n_blocks = 0;
    // aggregation config
    int field_vtype = BYDB_VT_INT64;
    uint32_t func_mask = 0;
    uint32_t n_groups = 0;
    int mode = BYDB_MODE_ALL;
    bydb_partial *d_partials = nullptr;   // owned buffer
    bydb_partial *d_acc = nullptr;        // active accumulation target
    uint64_t partials_cap = 0;
    DevErr *d_err = nullptr;
    PredBlock *d_preds = nullptr;
    int64_t preds_cap = 0;
    SegEntry *d_segs = nullptr;
    int64_t segs_cap = 0;
    bool segs_built = false;
    // per-row group-by on dictionary tags (composite keys over <=3 slots)
    int n_gslots = 0;                  // 0 = group by block group_code
    int gslots[3] = {-1, -1, -1};
    int64_t gmul[3] = {1, 1, 1};
    GroupBlock *d_groups = nullptr;    // [slot][block]
    uint16_t *d_gmap = nullptr;        // [slot][block][256]
    int64_t groups_cap = 0;
    bool groups_built = false;
    // plain-tag group-by: (gid,count) run pairs per normalized plain
    // stream on a grouped slot (k_resolve_plain_groups)
    uint16_t *d_gruns = nullptr;
    uint64_t gruns_cap_streams = 0;
    uint32_t *d_grun_ctr = nullptr;
    uint8_t *d_dom_blob[3] = {};
    uint64_t *d_dom_offs[3] = {};
    uint64_t *d_dom_hashes[3] = {};
    uint32_t *d_dom_gids[3] = {};
    uint32_t dom_table_size[3] = {};
    uint32_t dom_n[3] = {};
    uint8_t *d_pred_bytes = nullptr;
    uint64_t pred_bytes_cap = 0;
    // plain (non-dictionary) tag columns: host-normalized sidecar arena +
    // per-row match bitmaps filled by k_resolve_plain
    uint8_t *d_sidecar = nullptr;
    uint64_t sidecar_len = 0, sidecar_cap = 0;
    uint64_t n_plain_host = 0;  // (block,slot) plain streams normalized
    uint32_t *d_plain_ctr = nullptr;
    uint64_t *d_pred_bm = nullptr;
    uint64_t pred_bm_cap = 0;  // words
    uint8_t *d_pred_flags = nullptr;
    int64_t pred_flags_cap = 0;
    uint32_t *d_walk_count = nullptr;
    uint64_t *d_first_seen = nullptr;
    uint64_t first_seen_cap = 0;
    float last_ms = 0.0f;
    bool consumed = false;
    int16_t float_exp = 0;  // shared decimal exponent for float64 restore
};

#define HIP_TRY(s, call)                                                      \
    do {                                                                      \
        hipError_t _e = (call);                                               \
        if (_e != hipSuccess) {                                               \
            (s)->err = std::string(#call) + ": " + hipGetErrorString(_e);     \
            return BYDB_ERR_HIP;                                              \
        }                                                                     \
    } while (0)

extern "C" bydb_session *bydb_session_create(int device) {
    int n = 0;
    hipError_t e = hipGetDeviceCount(&n);
    if (e != hipSuccess || n <= device) {
        fprintf(stderr, "bydb_session_create: hipGetDeviceCount=%s n=%d dev=%d\n",
                hipGetErrorString(e), n, device);
        return nullptr;
    }
    if ((e = hipSetDevice(device)) != hipSuccess) {
        fprintf(stderr, "bydb_session_create: hipSetDevice=%s\n", hipGetErrorString(e));
        return nullptr;
    }
    bydb_session *s = new bydb_session();
    s->device = device;
    if ((e = hipStreamCreate(&s->stream)) != hipSuccess ||
        (e = hipEventCreate(&s->ev_start)) != hipSuccess ||
        (e = hipEventCreate(&s->ev_stop)) != hipSuccess ||
        (e = hipMalloc(&s->d_err, sizeof(DevErr))) != hipSuccess ||
        (e = hipMalloc(&s->d_plain_ctr, sizeof(uint32_t))) != hipSuccess) {
        fprintf(stderr, "bydb_session_create: init failed: %s\n", hipGetErrorString(e));
        delete s;
        return nullptr;
    }
    (void)hipMemset(s->d_err, 0, sizeof(DevErr));
    (void)hipMemset(s->d_plain_ctr, 0, sizeof(uint32_t));
    return s;
}

extern "C" void bydb_session_destroy(bydb_session *s) {
    if (!s) return;
    (void)hipSetDevice(s->device);
    if (s->d_payload) (void)hipFree(s->d_payload);
    if (s->d_blocks) (void)hipFree(s->d_blocks);
    if (s->d_partials) (void)hipFree(s->d_partials);
    if (s->d_preds) (void)hipFree(s->d_preds);
    if (s->d_pred_bytes) (void)hipFree(s->d_pred_bytes);
    if (s->d_segs) (void)hipFree(s->d_segs);
    if (s->d_groups) (void)hipFree(s->d_groups);
    if (s->d_gmap) (void)hipFree(s->d_gmap);
    if (s->d_gruns) (void)hipFree(s->d_gruns);
    if (s->d_grun_ctr) (void)hipFree(s->d_grun_ctr);
    for (int sl = 0; sl < 3; sl++) {
        if (s->d_dom_blob[sl]) (void)hipFree(s->d_dom_blob[sl]);
        if (s->d_dom_offs[sl]) (void)hipFree(s->d_dom_offs[sl]);
        if (s->d_dom_hashes[sl]) (void)hipFree(s->d_dom_hashes[sl]);
        if (s->d_dom_gids[sl]) (void)hipFree(s->d_dom_gids[sl]);
    }
    if (s->d_err) (void)hipFree(s->d_err);
    if (s->d_sidecar) (void)hipFree(s->d_sidecar);
    if (s->d_plain_ctr) (void)hipFree(s->d_plain_ctr);
    if (s->d_pred_bm) (void)hipFree(s->d_pred_bm);
    if (s->d_pred_flags) (void)hipFree(s->d_pred_flags);
    if (s->d_walk_count) (void)hipFree(s->d_walk_count);
    if (s->d_first_seen) (void)hipFree(s->d_first_seen);
    if (s->ev_start) (void)hipEventDestroy(s->ev_start);
    if (s->ev_stop) (void)hipEventDestroy(s->ev_stop);
    if (s->stream) (void)hipStreamDestroy(s->stream);
    delete s;
}

extern "C" const char *bydb_last_error(bydb_session *s) {
    return s ? s->err.c_str() : "null session";
}

extern "C" int bydb_part_reserve(bydb_session *s, uint64_t payload_bytes,
                                 int64_t n_blocks) {
    HIP_TRY(s, hipSetDevice(s->device));
    if (s->d_payload) { (void)hipFree(s->d_payload); s->d_payload = nullptr; }
    if (s->d_blocks) { (void)hipFree(s->d_blocks); s->d_blocks = nullptr; }
    // +4KiB slack: the 256-byte quad-window loop runs an 8-window-deep
    // load pipeline (2 KiB of lead) and may read past the last stream's
    // end
    HIP_TRY(s, hipMalloc(&s->d_payload, payload_bytes + 4096));
    HIP_TRY(s, hipMemset(s->d_payload + payload_bytes, 0, 4096));
    HIP_TRY(s, hipMalloc(&s->d_blocks, sizeof(bydb_block_desc) * (size_t)n_blocks));
    s->payload_cap = payload_bytes;
    s->blocks_cap = n_blocks;
    s->payload_len = 0;
    s->n_blocks = 0;
    s->part_ts_min = INT64_MAX;
    s->part_ts_max = INT64_MIN;
    s->sidecar_len = 0;
    s->n_plain_host = 0;
    s->segs_built = false;
    s->groups_built = false;
    return BYDB_OK;
}

extern "C" int bydb_part_append(bydb_session *s, const uint8_t *payload,
                                uint64_t len, const bydb_block_desc *blocks,
                                int64_t n_blocks) {
    HIP_TRY(s, hipSetDevice(s->device));
    s->segs_built = false;
    s->groups_built = false;
    if (s->payload_len + len > s->payload_cap ||
        s->n_blocks + n_blocks > s->blocks_cap) {
        s->err = "part_append exceeds reservation";
        return BYDB_ERR_BAD_ARG;
    }
    // Plain (non-dictionary) tag columns are zstd-compressed in the part;
    // normalize them here — where the reference decompresses at part open
    // (zstd.go:49) — into the sidecar arena, and repoint the stored
    // descriptor (TAG_SIDECAR_BIT).  The payload arena layout is untouched
    // so later chunks' absolute offsets stay valid.
    std::vector<bydb_block_desc> fixed;
    std::vector<uint8_t> extra;
    const uint64_t part_end = s->payload_len + len;
    for (int64_t i = 0; i < n_blocks; i++) {
        const bydb_block_desc *bd = &blocks[i];
        if (bd->ts_min < s->part_ts_min) s->part_ts_min = bd->ts_min;
        if (bd->ts_max > s->part_ts_max) s->part_ts_max = bd->ts_max;
        // loud host-side validation: every stream the descriptor names
        // must lie inside the part appended so far (catches builders whose
        // offset base drifted from this session's arena)
        if (bd->ts_off + bd->ts_len > part_end ||
            bd->field_off + bd->field_len > part_end ||
            (bd->tag_len && bd->tag_off + bd->tag_len > part_end) ||
            (bd->tag2_len && bd->tag2_off + bd->tag2_len > part_end) ||
            (bd->tag3_len && bd->tag3_off + bd->tag3_len > part_end)) {
            s->err = "descriptor stream offsets exceed appended payload";
            return BYDB_ERR_BAD_ARG;
        }
        for (int sl = 0; sl < 3; sl++) {
            uint64_t toff = sl == 0 ? bd->tag_off
                                    : sl == 1 ? bd->tag2_off : bd->tag3_off;
            uint64_t tlen = sl == 0 ? bd->tag_len
                                    : sl == 1 ? bd->tag2_len : bd->tag3_len;
            if (tlen == 0 || (toff & TAG_SIDECAR_BIT)) continue;
            if (toff < s->payload_len || toff + tlen > s->payload_len + len)
                continue;  // stream not in this chunk: leave as-is
            const uint8_t *src = payload + (toff - s->payload_len);
            std::vector<uint8_t> norm;
            if (src[0] == BYDB_ENC_PLAIN) {
                if (!bydb_normalize_plain_tag(src, tlen, bd->count, norm)) {
                    s->err = "plain tag column normalization failed";
                    return BYDB_ERR_BAD_DATA;
                }
                s->n_plain_host++;
            } else if (src[0] == BYDB_ENC_DICTIONARY) {
                // dictionaries whose compress_block sections are zstd'd
                // (>=128 B) get the same host decompress; small plain
                // dictionaries parse on device untouched
                bool needed = false;
                if (!bydb_normalize_dict_tag(src, tlen, &needed, norm)) {
                    if (needed) {
                        s->err = "dictionary tag normalization failed";
                        return BYDB_ERR_BAD_DATA;
                    }
                    continue;
                }
                if (!needed) continue;
            } else {
                continue;
            }
            if (fixed.empty()) fixed.assign(blocks, blocks + n_blocks);
            uint64_t noff = TAG_SIDECAR_BIT | (s->sidecar_len + extra.size());
            bydb_block_desc &fd = fixed[(size_t)i];
            if (sl == 0) { fd.tag_off = noff; fd.tag_len = norm.size(); }
            else if (sl == 1) { fd.tag2_off = noff; fd.tag2_len = norm.size(); }
            else { fd.tag3_off = noff; fd.tag3_len = norm.size(); }
            extra.insert(extra.end(), norm.begin(), norm.end());
        }
        // null-bearing Plain FIELD columns (8-B sign-flip cell blocks):
        // normalize into the sidecar as bitmap + fixed cells so the fold
        // can skip nulls per row (aggregation.go:310 null check)
        if (bd->field_enc == BYDB_ENC_PLAIN &&
            !(bd->field_off & TAG_SIDECAR_BIT)) {
            const uint8_t *fsrc = payload + (bd->field_off - s->payload_len);
            std::vector<uint8_t> norm;
            if (!bydb_normalize_plain_field(fsrc, bd->field_len, bd->count,
                                            norm)) {
                s->err = "plain field column normalization failed";
                return BYDB_ERR_BAD_DATA;
            }
            if (fixed.empty()) fixed.assign(blocks, blocks + n_blocks);
            while ((s->sidecar_len + extra.size()) & 7) extra.push_back(0);
            bydb_block_desc &fd = fixed[(size_t)i];
            fd.field_off = TAG_SIDECAR_BIT | (s->sidecar_len + extra.size());
            fd.field_len = norm.size();
            extra.insert(extra.end(), norm.begin(), norm.end());
        }
    }
    if (!extra.empty()) {
        uint64_t need = s->sidecar_len + extra.size() + 64;
        if (need > s->sidecar_cap) {
            uint64_t ncap = s->sidecar_cap ? s->sidecar_cap : 4096;
            while (ncap < need) ncap *= 2;
            uint8_t *nb = nullptr;
            HIP_TRY(s, hipMalloc(&nb, ncap));
            if (s->sidecar_len)
                HIP_TRY(s, hipMemcpy(nb, s->d_sidecar, s->sidecar_len,
                                     hipMemcpyDeviceToDevice));
            if (s->d_sidecar) (void)hipFree(s->d_sidecar);
            s->d_sidecar = nb;
            s->sidecar_cap = ncap;
        }
        HIP_TRY(s, hipMemcpy(s->d_sidecar + s->sidecar_len, extra.data(),
                             extra.size(), hipMemcpyHostToDevice));
        s->sidecar_len += extra.size();
    }
    HIP_TRY(s, hipMemcpy(s->d_payload + s->payload_len, payload, len,
                         hipMemcpyHostToDevice));
    HIP_TRY(s, hipMemcpy(s->d_blocks + s->n_blocks,
                         fixed.empty() ? blocks : fixed.data(),
                         sizeof(bydb_block_desc) * (size_t)n_blocks,
                         hipMemcpyHostToDevice));
    s->payload_len += len;
    s->n_blocks += n_blocks;
    return BYDB_OK;
}

extern "C" int bydb_part_clear(bydb_session *s) {
    s->payload_len = 0;
    s->n_blocks = 0;
    s->part_ts_min = INT64_MAX;
    s->part_ts_max = INT64_MIN;
    s->sidecar_len = 0;
    s->n_plain_host = 0;
    return BYDB_OK;
}

extern "C" int bydb_agg_configure(bydb_session *s, int field_vtype,
                                  uint32_t func_mask, uint32_t n_groups,
                                  int mode) {
    HIP_TRY(s, hipSetDevice(s->device));
    if (n_groups < 1) { s->err = "n_groups < 1"; return BYDB_ERR_BAD_ARG; }
    s->field_vtype = field_vtype;
    s->func_mask = func_mask;
    s->n_groups = n_groups;
    s->mode = mode;
    if (s->partials_cap < n_groups) {
        if (s->d_partials) (void)hipFree(s->d_partials);
        HIP_TRY(s, hipMalloc(&s->d_partials, sizeof(bydb_partial) * n_groups));
        s->partials_cap = n_groups;
    }
    if (s->first_seen_cap < n_groups) {
        if (s->d_first_seen) (void)hipFree(s->d_first_seen);
        HIP_TRY(s, hipMalloc(&s->d_first_seen, sizeof(uint64_t) * n_groups));
        s->first_seen_cap = n_groups;
    }
    s->d_acc = s->d_partials;
    s->n_gslots = 0;
    return bydb_reset(s);
}

// Load one slot's group domain onto the device (open-addressing table,
// equality by bytes).
static int load_domain(bydb_session *s, int pos, const uint8_t *dom_blob,
                       const uint64_t *dom_offs, uint32_t n_values) {
    uint32_t tsize = 4;
    while (tsize < n_values * 4) tsize <<= 1;
    std::vector<uint64_t> hashes(tsize, 0);
    std::vector<uint32_t> gids(tsize, GID_NONE);
    for (uint32_t g = 0; g < n_values; g++) {
        const uint8_t *v = dom_blob + dom_offs[g];
        uint64_t len = dom_offs[g + 1] - dom_offs[g];
        uint64_t h = fnv1a(v, len);
        uint32_t sl = (uint32_t)h & (tsize - 1);
        while (gids[sl] != GID_NONE) sl = (sl + 1) & (tsize - 1);
        hashes[sl] = h;
        gids[sl] = g;
    }
    uint64_t blob_len = dom_offs[n_values];
    if (s->d_dom_blob[pos]) { (void)hipFree(s->d_dom_blob[pos]); s->d_dom_blob[pos] = nullptr; }
    if (s->d_dom_offs[pos]) { (void)hipFree(s->d_dom_offs[pos]); s->d_dom_offs[pos] = nullptr; }
    if (s->d_dom_hashes[pos]) { (void)hipFree(s->d_dom_hashes[pos]); s->d_dom_hashes[pos] = nullptr; }
    if (s->d_dom_gids[pos]) { (void)hipFree(s->d_dom_gids[pos]); s->d_dom_gids[pos] = nullptr; }
    HIP_TRY(s, hipMalloc(&s->d_dom_blob[pos], blob_len ? blob_len : 1));
    HIP_TRY(s, hipMalloc(&s->d_dom_offs[pos], sizeof(uint64_t) * (n_values + 1)));
    HIP_TRY(s, hipMalloc(&s->d_dom_hashes[pos], sizeof(uint64_t) * tsize));
    HIP_TRY(s, hipMalloc(&s->d_dom_gids[pos], sizeof(uint32_t) * tsize));
    HIP_TRY(s, hipMemcpy(s->d_dom_blob[pos], dom_blob, blob_len, hipMemcpyHostToDevice));
    HIP_TRY(s, hipMemcpy(s->d_dom_offs[pos], dom_offs,
                         sizeof(uint64_t) * (n_values + 1), hipMemcpyHostToDevice));
    HIP_TRY(s, hipMemcpy(s->d_dom_hashes[pos], hashes.data(),
                         sizeof(uint64_t) * tsize, hipMemcpyHostToDevice));
    HIP_TRY(s, hipMemcpy(s->d_dom_gids[pos], gids.data(),
                         sizeof(uint32_t) * tsize, hipMemcpyHostToDevice));
    s->dom_table_size[pos] = tsize;
    s->dom_n[pos] = n_values;
    return BYDB_OK;
}

// Group rows by up to 3 dictionary tags (composite key, computeKey
// semantics): gid = g0 + n0*g1 + n0*n1*g2; per-slot domains concatenated
// in blobs/offs (offs holds, per slot, n_i+1 cumulative offsets).
extern "C" int bydb_agg_configure_by_tags(bydb_session *s, int field_vtype,
                                          uint32_t func_mask, const int *slots,
                                          int n_slots,
                                          const uint8_t *const *dom_blobs,
                                          const uint64_t *const *dom_offs,
                                          const uint32_t *n_values, int mode) {
    if (n_slots < 1 || n_slots > 3) { s->err = "n_slots 1..3"; return BYDB_ERR_BAD_ARG; }
    uint64_t total = 1;
    for (int i = 0; i < n_slots; i++) {
        if (slots[i] < 0 || slots[i] > 2 || n_values[i] < 1) {
            s->err = "bad slot/domain";
            return BYDB_ERR_BAD_ARG;
        }
        total *= n_values[i];
    }
    if (total > 65000) { s->err = "composite group space too large"; return BYDB_ERR_BAD_ARG; }
    int rc = bydb_agg_configure(s, field_vtype, func_mask, (uint32_t)total, mode);
    if (rc != BYDB_OK) return rc;
    s->n_gslots = n_slots;
    s->groups_built = false;
    int64_t mul = 1;
    for (int i = 0; i < n_slots; i++) {
        s->gslots[i] = slots[i];
        s->gmul[i] = mul;
        mul *= n_values[i];
        rc = load_domain(s, i, dom_blobs[i], dom_offs[i], n_values[i]);
        if (rc != BYDB_OK) return rc;
    }
    return BYDB_OK;
}

extern "C" int bydb_agg_configure_by_tag(bydb_session *s, int field_vtype,
                                         uint32_t func_mask, int tag_slot,
                                         const uint8_t *dom_blob,
                                         const uint64_t *dom_offs,
                                         uint32_t n_values, int mode) {
    const int slots[1] = {tag_slot};
    const uint8_t *blobs[1] = {dom_blob};
    const uint64_t *offs[1] = {dom_offs};
    const uint32_t nv[1] = {n_values};
    return bydb_agg_configure_by_tags(s, field_vtype, func_mask, slots, 1,
                                      blobs, offs, nv, mode);
}

extern "C" int bydb_set_partials_buffer(bydb_session *s, void *dev_ptr,
                                        uint64_t len) {
    if (dev_ptr == nullptr) {
        s->d_acc = s->d_partials;
        return BYDB_OK;
    }
    if (len < sizeof(bydb_partial) * s->n_groups) {
        s->err = "partials buffer too small";
        return BYDB_ERR_BAD_ARG;
    }
    s->d_acc = (bydb_partial *)dev_ptr;
    return bydb_reset(s);
}

extern "C" int bydb_reset(bydb_session *s) {
    HIP_TRY(s, hipSetDevice(s->device));
    if (!s->d_acc) { s->err = "configure first"; return BYDB_ERR_STATE; }
    int threads = 256;
    int blocks = (int)((s->n_groups + threads - 1) / threads);
    hipLaunchKernelGGL(k_reset_partials, dim3(blocks), dim3(threads), 0,
                       s->stream, s->d_acc, (int64_t)s->n_groups);
    HIP_TRY(s, hipGetLastError());
    if (s->d_first_seen)
        HIP_TRY(s, hipMemsetAsync(s->d_first_seen, 0xFF,
                                  sizeof(uint64_t) * s->n_groups, s->stream));
    HIP_TRY(s, hipMemsetAsync(s->d_err, 0, sizeof(DevErr), s->stream));
    s->consumed = false;
    return BYDB_OK;
}

extern "C" int bydb_consume_multi(bydb_session *s, int64_t min_ts,
                                  int64_t max_ts, const uint8_t *const *preds_in,
                                  const uint64_t *pred_lens, int n_preds) {
    HIP_TRY(s, hipSetDevice(s->device));
    if (!s->d_acc) { s->err = "configure first"; return BYDB_ERR_STATE; }
    if (s->n_blocks == 0) { s->err = "no part resident"; return BYDB_ERR_STATE; }
    if (n_preds < 0 || n_preds > 3) { s->err = "n_preds must be 0..3"; return BYDB_ERR_BAD_ARG; }
    // drop trailing disabled slots; interior empty slots mean "no
    // constraint on that tag"
    while (n_preds > 0 && pred_lens[n_preds - 1] == 0) n_preds--;
    PredBlock *preds = nullptr;
    if (n_preds > 0) {
        if (s->preds_cap < s->n_blocks * 3) {
            if (s->d_preds) (void)hipFree(s->d_preds);
            HIP_TRY(s, hipMalloc(&s->d_preds,
                                 sizeof(PredBlock) * (size_t)s->n_blocks * 3));
            s->preds_cap = s->n_blocks * 3;
        }
        if (s->pred_flags_cap < s->n_blocks) {
            if (s->d_pred_flags) (void)hipFree(s->d_pred_flags);
            HIP_TRY(s, hipMalloc(&s->d_pred_flags, (size_t)s->n_blocks));
            s->pred_flags_cap = s->n_blocks;
        }
        if (!s->d_walk_count)
            HIP_TRY(s, hipMalloc(&s->d_walk_count, sizeof(uint32_t)));
        HIP_TRY(s, hipMemsetAsync(s->d_walk_count, 0, sizeof(uint32_t),
                                  s->stream));
        uint64_t total = 0;
        for (int i = 0; i < n_preds; i++) total += pred_lens[i];
        if (s->pred_bytes_cap < total) {
            if (s->d_pred_bytes) (void)hipFree(s->d_pred_bytes);
            HIP_TRY(s, hipMalloc(&s->d_pred_bytes, total));
            s->pred_bytes_cap = total;
        }
        uint64_t off = 0;
        uint64_t offs[3] = {0, 0, 0};
        uint64_t lens3[3] = {0, 0, 0};
        int rthreads = 256;
        int rblocks = (int)((s->n_blocks + rthreads - 1) / rthreads);
        for (int i = 0; i < n_preds; i++) {
            if (pred_lens[i])   // disabled slots may pass a NULL pointer
                HIP_TRY(s, hipMemcpyAsync(s->d_pred_bytes + off, preds_in[i],
                                          pred_lens[i], hipMemcpyHostToDevice,
                                          s->stream));
            offs[i] = off;
            lens3[i] = pred_lens[i];
            off += pred_lens[i];
        }
        // one fused launch resolves all slots + the combined flag byte
        hipLaunchKernelGGL(k_resolve_preds, dim3(rblocks), dim3(rthreads), 0,
                           s->stream, s->d_payload, s->d_sidecar, s->d_blocks,
                           s->n_blocks, s->d_pred_bytes, offs[0], lens3[0],
                           offs[1], lens3[1], offs[2], lens3[2], n_preds,
                           s->d_preds, s->d_pred_flags, s->d_walk_count);
        HIP_TRY(s, hipGetLastError());
        // plain (non-dictionary) columns present: the host counted every
        // normalized (block,slot) stream at part_append, so the bitmap
        // arena is sized without a device round-trip
        if (s->n_plain_host > 0) {
            uint64_t words = s->n_plain_host * PLAIN_BM_WORDS;
            if (words > s->pred_bm_cap) {
                if (s->d_pred_bm) (void)hipFree(s->d_pred_bm);
                HIP_TRY(s, hipMalloc(&s->d_pred_bm, words * sizeof(uint64_t)));
                s->pred_bm_cap = words;
            }
            HIP_TRY(s, hipMemsetAsync(s->d_plain_ctr, 0, sizeof(uint32_t),
                                      s->stream));
            int pgrid = (int)(s->n_blocks < 65535 ? s->n_blocks : 65535);
            off = 0;
            for (int i = 0; i < n_preds; i++) {
                if (pred_lens[i] == 0) continue;
                hipLaunchKernelGGL(k_resolve_plain, dim3(pgrid), dim3(WAVE), 0,
                                   s->stream, s->d_payload, s->d_sidecar,
                                   s->d_blocks, s->n_blocks,
                                   s->d_pred_bytes + off, pred_lens[i], i,
                                   s->d_preds + (int64_t)i * s->n_blocks,
                                   s->d_pred_bm, s->d_plain_ctr);
                HIP_TRY(s, hipGetLastError());
                off += pred_lens[i];
            }
        }
        preds = s->d_preds;
    }
    int flags = 0;
    if (s->func_mask & ((1u << BYDB_AGG_MIN) | (1u << BYDB_AGG_MAX)))
        flags |= KF_NEED_VALUES;
    if (s->field_vtype == BYDB_VT_FLOAT64)
        flags |= KF_FLOAT | ((int)(uint16_t)s->float_exp << 16);
    const int threads = 256;                       // 4 waves per workgroup
    SegEntry *segs = nullptr;
    if (flags & KF_NEED_VALUES) {
        // build (once per part) and use the varint segment index
        if (s->segs_cap < s->n_blocks * MAX_SEGS) {
            if (s->d_segs) (void)hipFree(s->d_segs);
            HIP_TRY(s, hipMalloc(&s->d_segs, sizeof(SegEntry) *
                                                 (size_t)s->n_blocks * MAX_SEGS));
            s->segs_cap = s->n_blocks * MAX_SEGS;
            s->segs_built = false;
        }
        if (!s->segs_built) {
            int64_t wgs_b = (s->n_blocks + 3) / 4;
            int grid_b = (int)(wgs_b < 8192 ? wgs_b : 8192);
            hipLaunchKernelGGL(k_build_seg_index, dim3(grid_b), dim3(threads),
                               0, s->stream, s->d_payload, s->d_blocks,
                               s->n_blocks, s->d_segs, s->d_err);
            HIP_TRY(s, hipGetLastError());
            s->segs_built = true;
        }
        segs = s->d_segs;
    }
    GroupBlock *groups = nullptr;
    if (s->n_gslots > 0) {
        if (s->groups_cap < s->n_blocks) {
            if (s->d_groups) (void)hipFree(s->d_groups);
            if (s->d_gmap) (void)hipFree(s->d_gmap);
            HIP_TRY(s, hipMalloc(&s->d_groups,
                                 sizeof(GroupBlock) * 3 * (size_t)s->n_blocks));
            HIP_TRY(s, hipMalloc(&s->d_gmap,
                                 sizeof(uint16_t) * 256 * 3 * (size_t)s->n_blocks));
            s->groups_cap = s->n_blocks;
            s->groups_built = false;
        }
        if (!s->groups_built) {
            int rthreads = 256;
            int rblocks = (int)((s->n_blocks + rthreads - 1) / rthreads);
            // plain (>256-distinct) tag columns on grouped slots resolve
            // into (gid,count) run pairs; size the arena by the count of
            // normalized plain streams (an upper bound)
            if (s->n_plain_host > 0) {
                uint64_t want = s->n_plain_host;
                uint64_t bytes =
                    want * GRUN_ENTRIES_PER_STREAM * sizeof(uint16_t);
                if (bytes > (4ull << 30)) {
                    s->err = "plain-tag group-by run arena exceeds 4 GiB";
                    return BYDB_ERR_OOM;
                }
                if (want > s->gruns_cap_streams) {
                    if (s->d_gruns) (void)hipFree(s->d_gruns);
                    HIP_TRY(s, hipMalloc(&s->d_gruns, bytes));
                    s->gruns_cap_streams = want;
                }
                if (!s->d_grun_ctr)
                    HIP_TRY(s, hipMalloc(&s->d_grun_ctr, sizeof(uint32_t)));
                HIP_TRY(s, hipMemsetAsync(s->d_grun_ctr, 0, sizeof(uint32_t),
                                          s->stream));
            }
            for (int i = 0; i < s->n_gslots; i++) {
                GroupDomain dom;
                dom.blob = s->d_dom_blob[i];
                dom.offs = s->d_dom_offs[i];
                dom.hashes = s->d_dom_hashes[i];
                dom.gids = s->d_dom_gids[i];
                dom.table_size = s->dom_table_size[i];
                dom.n = s->dom_n[i];
                hipLaunchKernelGGL(
                    k_resolve_groups, dim3(rblocks), dim3(rthreads), 0,
                    s->stream, s->d_payload, s->d_sidecar, s->d_blocks,
                    s->n_blocks, s->gslots[i], dom,
                    s->d_groups + (int64_t)i * s->n_blocks,
                    s->d_gmap, (uint32_t)((int64_t)i * 256 * s->n_blocks));
                HIP_TRY(s, hipGetLastError());
                if (s->n_plain_host > 0) {
                    int pgrid =
                        (int)(s->n_blocks < 65535 ? s->n_blocks : 65535);
                    hipLaunchKernelGGL(
                        k_resolve_plain_groups, dim3(pgrid), dim3(WAVE), 0,
                        s->stream, s->d_payload, s->d_sidecar, s->d_blocks,
                        s->n_blocks, s->gslots[i], dom,
                        s->d_groups + (int64_t)i * s->n_blocks, s->d_gruns,
                        s->gruns_cap_streams, s->d_grun_ctr);
                    HIP_TRY(s, hipGetLastError());
                }
            }
            s->groups_built = true;
        }
        groups = s->d_groups;
    }
    int64_t waves_needed = segs ? s->n_blocks * MAX_SEGS : s->n_blocks;
    int64_t wgs = (waves_needed + 3) / 4;
    int grid = (int)(wgs < 8192 ? wgs : 8192);     // grid-stride beyond
    if (grid < 1) grid = 1;
    HIP_TRY(s, hipEventRecord(s->ev_start, s->stream));
    const bool en_values = (flags & KF_NEED_VALUES) != 0;
    const bool en_preds = n_preds > 0;
    const bool en_groups = groups != nullptr;
    void (*kfn)(const uint8_t *, const uint8_t *, const bydb_block_desc *,
                int64_t, int64_t, int64_t, int, const PredBlock *, int,
                const uint64_t *, const uint8_t *, const SegEntry *,
                const GroupBlock *, const uint16_t *, const uint16_t *, int,
                int64_t, int64_t, int64_t, int64_t, bydb_partial *,
                DevErr *, const uint32_t *, uint64_t *);
    void (*kfn_walk)(const uint8_t *, const uint8_t *,
                     const bydb_block_desc *, int64_t, int64_t, int64_t, int,
                     const PredBlock *, int, const uint64_t *,
                     const uint8_t *, const SegEntry *, const GroupBlock *,
                     const uint16_t *, const uint16_t *, int, int64_t,
                     int64_t, int64_t, int64_t, bydb_partial *, DevErr *,
                     const uint32_t *, uint64_t *) = nullptr;
    // the clamp-free instantiations run when the query range covers the
    // whole resident part (the common analytic scan; per-block FindRange
    // is then an identity)
    const bool en_clamp = !(s->n_blocks > 0 && min_ts <= s->part_ts_min &&
                            max_ts >= s->part_ts_max);
#define BYDB_KSEL(C)                                                         \
    if (en_values) {                                                         \
        if (en_preds) {                                                      \
            kfn = en_groups ? k_scan_agg_t<true, true, true, false, C>       \
                            : k_scan_agg_t<true, true, false, false, C>;     \
            kfn_walk = en_groups ? k_scan_agg_t<true, true, true, true, C>   \
                                 : k_scan_agg_t<true, true, false, true, C>; \
        } else {                                                             \
            kfn = en_groups ? k_scan_agg_t<true, false, true, false, C>      \
                            : k_scan_agg_t<true, false, false, false, C>;    \
        }                                                                    \
    } else {                                                                 \
        if (en_preds) {                                                      \
            kfn = en_groups ? k_scan_agg_t<false, true, true, false, C>      \
                            : k_scan_agg_t<false, true, false, false, C>;    \
            kfn_walk = en_groups ? k_scan_agg_t<false, true, true, true, C>  \
                                 : k_scan_agg_t<false, true, false, true, C>;\
        } else {                                                             \
            kfn = en_groups ? k_scan_agg_t<false, false, true, false, C>     \
                            : k_scan_agg_t<false, false, false, false, C>;   \
        }                                                                    \
    }
    if (en_clamp) {
        BYDB_KSEL(true)
    } else {
        BYDB_KSEL(false)
    }
#undef BYDB_KSEL
    // light pass: every block whose predicate verdict is uniform (no
    // walker state, no scratch spills in the scan loop)
    hipLaunchKernelGGL(kfn, dim3(grid), dim3(threads), 0, s->stream,
                       s->d_payload, s->d_sidecar, s->d_blocks, s->n_blocks,
                       min_ts, max_ts, flags, preds, n_preds, s->d_pred_bm,
                       s->d_pred_flags, segs, groups, s->d_gmap, s->d_gruns,
                       s->n_gslots, s->gmul[0], s->gmul[1], s->gmul[2],
                       (int64_t)s->n_groups, s->d_acc, s->d_err, nullptr,
                       s->d_first_seen);
    HIP_TRY(s, hipGetLastError());
    // heavy pass: only the PF_WALK blocks (row-varying predicates).  When
    // none exist this is a flag-byte sweep (~tens of us)
    if (kfn_walk != nullptr) {
        hipLaunchKernelGGL(kfn_walk, dim3(grid), dim3(threads), 0, s->stream,
                           s->d_payload, s->d_sidecar, s->d_blocks,
                           s->n_blocks, min_ts, max_ts, flags | KF_WALK_ONLY,
                           preds, n_preds, s->d_pred_bm, s->d_pred_flags,
                           segs, groups, s->d_gmap, s->d_gruns, s->n_gslots,
                           s->gmul[0], s->gmul[1], s->gmul[2],
                           (int64_t)s->n_groups, s->d_acc, s->d_err,
                           s->d_walk_count, s->d_first_seen);
        HIP_TRY(s, hipGetLastError());
    }
    HIP_TRY(s, hipEventRecord(s->ev_stop, s->stream));
    s->consumed = true;
    return BYDB_OK;
}

extern "C" int bydb_consume(bydb_session *s, int64_t min_ts, int64_t max_ts,
                            const uint8_t *pred, uint64_t pred_len) {
    if (pred_len == 0) {
        return bydb_consume_multi(s, min_ts, max_ts, nullptr, nullptr, 0);
    }
    const uint8_t *preds[1] = {pred};
    uint64_t lens[1] = {pred_len};
    return bydb_consume_multi(s, min_ts, max_ts, preds, lens, 1);
}

extern "C" double bydb_last_consume_ms(bydb_session *s) { return s->last_ms; }

static int finalize_common(bydb_session *s, bydb_partial *host_parts,
                           int64_t n_groups) {
    HIP_TRY(s, hipSetDevice(s->device));
    if ((uint32_t)n_groups != s->n_groups) {
        s->err = "n_groups mismatch";
        return BYDB_ERR_BAD_ARG;
    }
    HIP_TRY(s, hipStreamSynchronize(s->stream));
    if (s->consumed) {
        float ms = 0;
        if (hipEventElapsedTime(&ms, s->ev_start, s->ev_stop) == hipSuccess)
            s->last_ms = ms;
    }
    DevErr de;
    HIP_TRY(s, hipMemcpy(&de, s->d_err, sizeof de, hipMemcpyDeviceToHost));
    if (de.code != DERR_NONE) {
        char buf[128];
        snprintf(buf, sizeof buf, "device decode error %u at block %u", de.code,
                 de.block_lo);
        s->err = buf;
        return BYDB_ERR_BAD_DATA;
    }
    HIP_TRY(s, hipMemcpy(host_parts, s->d_acc,
                         sizeof(bydb_partial) * (size_t)n_groups,
                         hipMemcpyDeviceToHost));
    return BYDB_OK;
}

// Go math.Pow10 restated (decode-side float restore must reproduce its
// table-multiply rounding — float.go:77-102).
static const double h_pow10tab[32] = {
    1e0, 1e1, 1e2, 1e3, 1e4, 1e5, 1e6, 1e7, 1e8, 1e9, 1e10, 1e11, 1e12,
    1e13, 1e14, 1e15, 1e16, 1e17, 1e18, 1e19, 1e20, 1e21, 1e22, 1e23,
    1e24, 1e25, 1e26, 1e27, 1e28, 1e29, 1e30, 1e31};
static const double h_pow10postab32[10] = {
    1e0, 1e32, 1e64, 1e96, 1e128, 1e160, 1e192, 1e224, 1e256, 1e288};
static double go_pow10(int nn) {
    if (0 <= nn && nn <= 308)
        return h_pow10postab32[(unsigned)nn / 32] * h_pow10tab[(unsigned)nn % 32];
    return 0.0;  // callers only use 0..308
}

static double restore_f64(int64_t v, int16_t exp) {
    if (exp >= 0) return (double)v * go_pow10(exp);
    double r = (double)v;
    int neg = -(int)exp;
    while (neg > 0) {
        int step = neg < 308 ? neg : 308;
        r /= go_pow10(step);
        neg -= step;
    }
    return r;
}

static double restore_f64_from_double(double v, int16_t exp) {
    if (exp >= 0) return v * go_pow10(exp);
    double r = v;
    int neg = -(int)exp;
    while (neg > 0) {
        int step = neg < 308 ? neg : 308;
        r /= go_pow10(step);
        neg -= step;
    }
    return r;
}

// The float64 restore at finalize needs the column's decimal exponent
// (column payload header, column.go:253-263).  One query aggregates one
// field column, whose blocks share the exponent at the benchmark configs;
// the host sets it per session before finalize.
extern "C" int bydb_set_float_exp(bydb_session *s, int16_t exp) {
    s->float_exp = exp;
    return BYDB_OK;
}

extern "C" int bydb_finalize_partials(bydb_session *s, bydb_partial *out,
                                      int64_t n_groups) {
    return finalize_common(s, out, n_groups);
}

/* Per-group first-seen keys ((item << 13) | row in storage order; ~0 =
 * group never entered).  Sorting group ids by these keys reproduces the
 * reference's first-seen group materialisation order (computeKey
 * creates groups in row-iteration order, aggregation.go:523), so a Go
 * twin can emit NextBatch rows in reference order from the host-domain
 * buffers.  Caveats (documented in DESIGN.md): a group whose only rows
 * fail a BITMAP-mode (plain-tag) predicate inside a row-varying merge
 * may still record the run-entry key; within one (item,row) tie the
 * order is exact. */
extern "C" int bydb_group_first_seen(bydb_session *s, uint64_t *out,
                                     int64_t n_groups) {
    HIP_TRY(s, hipSetDevice(s->device));
    if ((uint32_t)n_groups != s->n_groups || !s->d_first_seen) {
        s->err = "n_groups mismatch or not configured";
        return BYDB_ERR_BAD_ARG;
    }
    HIP_TRY(s, hipStreamSynchronize(s->stream));
    HIP_TRY(s, hipMemcpy(out, s->d_first_seen,
                         sizeof(uint64_t) * (size_t)n_groups,
                         hipMemcpyDeviceToHost));
    return BYDB_OK;
}

static void partial_to_result(const bydb_partial *p, int field_vtype,
                              int16_t float_exp, bydb_result *r) {
    r->sum_i = p->sum_i;
    r->count = p->count;
    r->min_i = p->min_i;
    r->max_i = p->max_i;
    if (field_vtype == BYDB_VT_FLOAT64 && float_exp == INT16_MIN) {
        // raw IEEE-754 domain (nullable float columns): min/max carried
        // as order-preserving bit keys (convert/number.go:148-157), sum
        // carried directly as a double
        auto key_to_f64 = [](int64_t k) {
            uint64_t bits = k >= 0 ? (uint64_t)k
                                   : ~((uint64_t)k ^ 0x8000000000000000ull);
            double d;
            memcpy(&d, &bits, 8);
            return d;
        };
        r->min_f = p->count ? key_to_f64(p->min_i) : 1.7976931348623157e308;
        r->max_f = p->count ? key_to_f64(p->max_i) : -1.7976931348623157e308;
        r->sum_f = p->sum_f;
        double c = (double)p->count;
        r->mean_f = c == 0 ? 0 : (r->sum_f / c < 1 ? 1 : r->sum_f / c);
        r->mean_i = 0;
    } else if (field_vtype == BYDB_VT_FLOAT64) {
        // decimal-int domain -> float64 (monotone restore; float.go:69-102)
        r->min_f = p->count ? restore_f64(p->min_i, float_exp) : 1.7976931348623157e308;
        r->max_f = p->count ? restore_f64(p->max_i, float_exp) : -1.7976931348623157e308;
        r->sum_f = restore_f64_from_double(p->sum_f, float_exp);
        double c = (double)p->count;
        r->mean_f = c == 0 ? 0 : (r->sum_f / c < 1 ? 1 : r->sum_f / c);
        r->mean_i = 0;
    } else {
        r->sum_f = 0;
        r->min_f = 0;
        r->max_f = 0;
        r->mean_f = 0;
        // meanFunc.Val (function.go:30-45)
        r->mean_i = p->count == 0 ? 0
                    : (p->sum_i / p->count < 1 ? 1 : p->sum_i / p->count);
    }
}

extern "C" int bydb_finalize(bydb_session *s, bydb_result *out,
                             int64_t n_groups) {
    std::vector<bydb_partial> parts((size_t)n_groups);
    int rc = finalize_common(s, parts.data(), n_groups);
    if (rc != BYDB_OK) return rc;
    for (int64_t g = 0; g < n_groups; g++)
        partial_to_result(&parts[(size_t)g], s->field_vtype, s->float_exp, &out[(size_t)g]);
    return BYDB_OK;
}

extern "C" int bydb_reduce_partials2(const bydb_partial *parts,
                                     int64_t n_parts_per_group, int64_t n_groups,
                                     int field_vtype, int16_t float_exp,
                                     bydb_result *out) {
    for (int64_t g = 0; g < n_groups; g++) {
        bydb_partial acc;
        acc.sum_i = 0;
        acc.count = 0;
        acc.min_i = INT64_MAX;
        acc.max_i = INT64_MIN;
        acc.sum_f = 0;
        // Combine — aggregation_reduce.go:120-138 / function.go Reduce
        for (int64_t k = 0; k < n_parts_per_group; k++) {
            const bydb_partial *p = &parts[k * n_groups + g];
            acc.sum_i = (int64_t)((uint64_t)acc.sum_i + (uint64_t)p->sum_i);
            acc.count += p->count;
            if (p->min_i < acc.min_i) acc.min_i = p->min_i;
            if (p->max_i > acc.max_i) acc.max_i = p->max_i;
            acc.sum_f += p->sum_f;
        }
        partial_to_result(&acc, field_vtype, float_exp, &out[g]);
    }
    return BYDB_OK;
}
