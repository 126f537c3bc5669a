/* oracle/bydb_oracle.c — CPU oracle (checker only; see header).
 *
 * Plain-C restatement of the reference Go hot-path algorithms.  Every
 * function cites the reference file:line it follows.  Arithmetic that wraps
 * in Go (int64 +,-,*) is done here in uint64_t so C's signed-overflow UB is
 * never triggered; results are bit-identical to Go's wrapping semantics.
 */
#include "bydb_oracle.h"

#include <errno.h>
#include <math.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <dlfcn.h>

/* ===================== varint / zigzag =====================
 * VarInt64ListToBytes — pkg/encoding/int.go:81-103
 * BytesToVarInt64List — pkg/encoding/int.go:111-148
 * Format: zigzag, then 7-bit groups LSB-first; continuation bytes have the
 * high bit SET, the terminating byte has it CLEAR. */

size_t bo_varint64_list_encode(uint8_t *dst, const int64_t *vs, int64_t n) {
    size_t o = 0;
    for (int64_t i = 0; i < n; i++) {
        int64_t v = vs[i];
        if (v < 0x40 && v > -0x40) {
            /* int.go:84-88: single-byte fast path, zigzag in int8 */
            int8_t c = (int8_t)v;
            uint8_t z = (uint8_t)((c << 1) ^ (c >> 7));
            dst[o++] = z;
            continue;
        }
        uint64_t u = ((uint64_t)v << 1) ^ (uint64_t)(v >> 63);
        while (u > 0x7f) {
            dst[o++] = (uint8_t)(0x80u | (u & 0xff));
            u >>= 7;
        }
        dst[o++] = (uint8_t)u;
    }
    return o;
}

int bo_varint64_list_decode(int64_t *dst, int64_t n, const uint8_t *src,
                            size_t src_len, size_t *consumed) {
    size_t idx = 0;
    for (int64_t i = 0; i < n; i++) {
        if (idx >= src_len) return BO_ERR_TRUNCATED;
        uint8_t c = src[idx++];
        if (c < 0x80) {
            /* int.go:122: v := int8(c>>1) ^ (int8(c<<7) >> 7) */
            int8_t v = (int8_t)(c >> 1) ^ (int8_t)((int8_t)(c << 7) >> 7);
            dst[i] = (int64_t)v;
            continue;
        }
        uint64_t u = (uint64_t)(c & 0x7f);
        size_t start_idx = idx - 1;
        unsigned shift = 0;
        while (c >= 0x80) {
            if (idx >= src_len) return BO_ERR_TRUNCATED;
            if (idx - start_idx > 9) return BO_ERR_TOO_LONG_VARINT;
            c = src[idx++];
            shift += 7;
            u |= (uint64_t)(c & 0x7f) << shift;
        }
        dst[i] = (int64_t)(u >> 1) ^ -(int64_t)(u & 1);
    }
    if (consumed) *consumed = idx;
    return BO_OK;
}

/* VarUint64ToBytes — int.go:152-199 */
size_t bo_varuint64_encode(uint8_t *dst, uint64_t u) {
    size_t o = 0;
    while (u > 0x7f) {
        dst[o++] = (uint8_t)(0x80u | (u & 0xff));
        u >>= 7;
    }
    dst[o++] = (uint8_t)u;
    return o;
}

int bo_varuint64_decode(const uint8_t *src, size_t src_len, uint64_t *out,
                        size_t *consumed) {
    size_t idx = 0;
    if (idx >= src_len) return BO_ERR_TRUNCATED;
    uint8_t c = src[idx++];
    if (c < 0x80) {
        *out = c;
        if (consumed) *consumed = idx;
        return BO_OK;
    }
    uint64_t u = (uint64_t)(c & 0x7f);
    size_t start_idx = idx - 1;
    unsigned shift = 0;
    while (c >= 0x80) {
        if (idx >= src_len) return BO_ERR_TRUNCATED;
        if (idx - start_idx > 9) return BO_ERR_TOO_LONG_VARINT;
        c = src[idx++];
        shift += 7;
        u |= (uint64_t)(c & 0x7f) << shift;
    }
    *out = u;
    if (consumed) *consumed = idx;
    return BO_OK;
}

/* ===================== encode-type selection =====================
 * isConst / isDelta / isIncremental — pkg/encoding/int_list.go:112-179 */

static int is_const(const int64_t *a, int64_t n) {
    if (n == 0) return 0;
    for (int64_t i = 1; i < n; i++)
        if (a[i] != a[0]) return 0;
    return 1;
}

static int64_t sign_bit(int64_t v) { return (int64_t)(((uint64_t)v >> 63) & 1); }

static void is_delta(const int64_t *a, int64_t n, int *is_d, int *is_dc) {
    *is_d = 0;
    *is_dc = 0;
    if (n < 2) return;
    int ct = 1;
    int64_t d1 = (int64_t)((uint64_t)a[1] - (uint64_t)a[0]);
    int64_t asc = sign_bit(d1);
    int64_t prev = a[1];
    for (int64_t i = 2; i < n; i++) {
        int64_t d = (int64_t)((uint64_t)a[i] - (uint64_t)prev);
        if ((sign_bit(d) ^ asc) == 1) return;
        if (ct && d != d1) ct = 0;
        prev = a[i];
    }
    *is_d = 1;
    *is_dc = ct;
}

static int is_incremental(const int64_t *a, int64_t n) {
    if (n < 2) return 0;
    int64_t resets = 0;
    int64_t v_prev = a[0];
    if (v_prev < 0) return 1;
    for (int64_t i = 1; i < n; i++) {
        int64_t v = a[i];
        if (v < v_prev) {
            if (v < 0) return 0;
            if (v > (v_prev >> 3)) return 0;
            resets++;
        }
        v_prev = v;
    }
    if (resets <= 2) return 1;
    return resets < (n >> 3);
}

/* int64ListDeltaToBytes — pkg/encoding/delta.go:26-44 */
static size_t delta_encode(uint8_t *dst, const int64_t *src, int64_t n,
                           int64_t *first) {
    *first = src[0];
    size_t o = 0;
    int64_t v = src[0];
    for (int64_t i = 1; i < n; i++) {
        int64_t d = (int64_t)((uint64_t)src[i] - (uint64_t)v);
        v = (int64_t)((uint64_t)v + (uint64_t)d);
        o += bo_varint64_list_encode(dst + o, &d, 1);
    }
    return o;
}

/* int64sDeltaOfDeltaToBytes — delta.go:72-90 */
static size_t dod_encode(uint8_t *dst, const int64_t *src, int64_t n,
                         int64_t *first) {
    *first = src[0];
    int64_t d1 = (int64_t)((uint64_t)src[1] - (uint64_t)src[0]);
    size_t o = bo_varint64_list_encode(dst, &d1, 1);
    int64_t v = src[1];
    for (int64_t i = 2; i < n; i++) {
        int64_t d2 = (int64_t)((uint64_t)src[i] - (uint64_t)v - (uint64_t)d1);
        d1 = (int64_t)((uint64_t)d1 + (uint64_t)d2);
        v = (int64_t)((uint64_t)v + (uint64_t)d1);
        o += bo_varint64_list_encode(dst + o, &d2, 1);
    }
    return o;
}

/* Int64ListToBytes — pkg/encoding/int_list.go:27-54 */
int bo_int64_list_encode(uint8_t *dst, size_t cap, const int64_t *a, int64_t n,
                         size_t *out_len, uint8_t *out_type, int64_t *out_first) {
    if (n < 1) return BO_ERR_EMPTY;
    (void)cap; /* callers size dst at >= 10*n bytes */
    if (is_const(a, n)) {
        *out_first = a[0];
        *out_type = BO_ENC_CONST;
        *out_len = 0;
        return BO_OK;
    }
    int is_d, is_dc;
    is_delta(a, n, &is_d, &is_dc);
    if (is_dc) {
        *out_first = a[0];
        int64_t d = (int64_t)((uint64_t)a[1] - (uint64_t)a[0]);
        *out_len = bo_varint64_list_encode(dst, &d, 1);
        *out_type = BO_ENC_DELTA_CONST;
        return BO_OK;
    }
    if (is_d || is_incremental(a, n)) {
        *out_len = dod_encode(dst, a, n, out_first);
        *out_type = BO_ENC_DELTA_OF_DELTA;
        return BO_OK;
    }
    *out_len = delta_encode(dst, a, n, out_first);
    *out_type = BO_ENC_DELTA;
    return BO_OK;
}

/* bytesDeltaToInt64List / bytesDeltaOfDeltaToInt64s / BytesToInt64List —
 * delta.go:45-118, int_list.go:57-101 */
int bo_int64_list_decode(int64_t *dst, const uint8_t *src, size_t src_len,
                         uint8_t mt, int64_t first_value, int64_t items_count) {
    switch (mt) {
    case BO_ENC_DELTA: {
        if (items_count < 1) return BO_ERR_EMPTY;
        int64_t nd = items_count - 1;
        int64_t *is = (int64_t *)malloc(sizeof(int64_t) * (size_t)(nd > 0 ? nd : 1));
        size_t consumed = 0;
        int rc = bo_varint64_list_decode(is, nd, src, src_len, &consumed);
        if (rc == BO_OK && consumed != src_len) rc = BO_ERR_TAIL;
        if (rc != BO_OK) { free(is); return rc; }
        uint64_t v = (uint64_t)first_value;
        dst[0] = (int64_t)v;
        for (int64_t i = 0; i < nd; i++) {
            v += (uint64_t)is[i];
            dst[i + 1] = (int64_t)v;
        }
        free(is);
        return BO_OK;
    }
    case BO_ENC_DELTA_OF_DELTA: {
        if (items_count < 2) return BO_ERR_EMPTY;
        int64_t nd = items_count - 1;
        int64_t *is = (int64_t *)malloc(sizeof(int64_t) * (size_t)nd);
        size_t consumed = 0;
        int rc = bo_varint64_list_decode(is, nd, src, src_len, &consumed);
        if (rc == BO_OK && consumed != src_len) rc = BO_ERR_TAIL;
        if (rc != BO_OK) { free(is); return rc; }
        uint64_t v = (uint64_t)first_value;
        uint64_t d1 = (uint64_t)is[0];
        dst[0] = (int64_t)v;
        v += d1;
        dst[1] = (int64_t)v;
        for (int64_t i = 1; i < nd; i++) {
            d1 += (uint64_t)is[i];
            v += d1;
            dst[i + 1] = (int64_t)v;
        }
        free(is);
        return BO_OK;
    }
    case BO_ENC_CONST: {
        if (src_len > 0) return BO_ERR_TAIL;
        for (int64_t i = 0; i < items_count; i++) dst[i] = first_value;
        return BO_OK;
    }
    case BO_ENC_DELTA_CONST: {
        int64_t d;
        size_t consumed = 0;
        int rc = bo_varint64_list_decode(&d, 1, src, src_len, &consumed);
        if (rc != BO_OK) return rc;
        if (consumed != src_len) return BO_ERR_TAIL;
        uint64_t v = (uint64_t)first_value;
        for (int64_t i = 0; i < items_count; i++) {
            dst[i] = (int64_t)v;
            v += (uint64_t)d;
        }
        return BO_OK;
    }
    default:
        return BO_ERR_BAD_TYPE;
    }
}

/* ===================== Go math.Pow10 =====================
 * Exact restatement of Go's math.Pow10 (go/src/math/pow10.go) — the decode
 * path's float scaling MUST reproduce its table-multiply rounding. */
static const double go_pow10tab[32] = {
    1e0, 1e1, 1e2, 1e3, 1e4, 1e5, 1e6, 1e7, 1e8, 1e9, 1e10, 1e11, 1e12,
    1e13, 1e14, 1e15, 1e16, 1e17, 1e18, 1e19, 1e20, 1e21, 1e22, 1e23,
    1e24, 1e25, 1e26, 1e27, 1e28, 1e29, 1e30, 1e31,
};
static const double go_pow10postab32[10] = {
    1e0, 1e32, 1e64, 1e96, 1e128, 1e160, 1e192, 1e224, 1e256, 1e288,
};
static const double go_pow10negtab32[11] = {
    1e-0, 1e-32, 1e-64, 1e-96, 1e-128, 1e-160, 1e-192, 1e-224, 1e-256,
    1e-288, 1e-320,
};

double bo_go_pow10(int n) {
    if (0 <= n && n <= 308)
        return go_pow10postab32[(unsigned)n / 32] * go_pow10tab[(unsigned)n % 32];
    if (-323 <= n && n <= 0)
        return go_pow10negtab32[(unsigned)(-n) / 32] / go_pow10tab[(unsigned)(-n) % 32];
    if (n > 308) return HUGE_VAL;
    return 0.0;
}

/* ===================== decimal float codec =====================
 * floatToDecimal — pkg/encoding/float.go:105-124 (fast int path) and
 * floatToDecimalSlow :126-180 (shortest round-trip via strconv; here the
 * unique shortest correctly-rounded decimal is found by the smallest %.*e
 * precision that round-trips — the same digits Go's AppendFloat(-1) emits). */
static const int64_t bo_pow10_i64[19] = {
    1LL, 10LL, 100LL, 1000LL, 10000LL, 100000LL, 1000000LL, 10000000LL,
    100000000LL, 1000000000LL, 10000000000LL, 100000000000LL, 1000000000000LL,
    10000000000000LL, 100000000000000LL, 1000000000000000LL,
    10000000000000000LL, 100000000000000000LL, 1000000000000000000LL,
};

/* mulPow10Fast / mulPow10Large — float.go:189-226 */
static int mul_pow10(int64_t v, int n, int64_t *out) {
    if (n < 0) return 0;
    while (n >= 19) {
        if (v > INT64_MAX / bo_pow10_i64[18] || v < INT64_MIN / bo_pow10_i64[18])
            return 0;
        v *= bo_pow10_i64[18];
        n -= 18;
    }
    if (n > 0) {
        if (v > INT64_MAX / bo_pow10_i64[n] || v < INT64_MIN / bo_pow10_i64[n])
            return 0;
        v *= bo_pow10_i64[n];
    }
    *out = v;
    return 1;
}

static int float_to_decimal(double f, int64_t *mant, int16_t *exp) {
    if (isnan(f) || isinf(f)) return 0;
    if (f == 0) { *mant = 0; *exp = 0; return 1; }
    if (f >= -9.2233720368547758e18 && f <= 9.2233720368547758e18) {
        /* float.go:113: if u := int64(f); float64(u) == f */
        int64_t u = (int64_t)f;
        if ((double)u == f) {
            int16_t e = 0;
            while (u != 0 && u % 10 == 0) { u /= 10; e++; }
            *mant = u; *exp = e; return 1;
        }
    }
    /* slow path: shortest round-trip 'e' representation */
    char buf[64];
    int p;
    for (p = 0; p <= 17; p++) {
        snprintf(buf, sizeof buf, "%.*e", p, f);
        double back = strtod(buf, NULL);
        if (back == f) break;
    }
    if (p > 17) return 0;
    /* parse d[.ddd]e±XX exactly as float.go:131-179 */
    char *e = strchr(buf, 'e');
    if (!e) return 0;
    long sci_exp = strtol(e + 1, NULL, 10);
    char digits[32];
    int nd = 0;
    int frac_digits = 0;
    int negative = buf[0] == '-';
    const char *s = buf + (negative ? 1 : 0);
    int seen_dot = 0;
    for (; s < e; s++) {
        if (*s == '.') { seen_dot = 1; continue; }
        digits[nd++] = *s;
        if (seen_dot) frac_digits++;
    }
    /* strip trailing zeros (float.go:160-163), keep at least 1 digit */
    while (nd > 1 && digits[nd - 1] == '0') { nd--; frac_digits--; }
    digits[nd] = 0;
    if (nd > 19) return 0;
    errno = 0;
    long long m = strtoll(digits, NULL, 10);
    if (sci_exp > 32767 || sci_exp < -32768) return 0;
    int32_t ex = (int32_t)sci_exp - frac_digits;
    if (negative) m = -m;
    *mant = (int64_t)m;
    *exp = (int16_t)ex;
    return 1;
}

/* Float64ListToDecimalIntList — float.go:30-66 */
int bo_float_list_to_decimal(const double *src, int64_t n, int64_t *out_ints,
                             int16_t *out_exp) {
    if (n == 0) { *out_exp = 0; return BO_OK; }
    int16_t *exps = (int16_t *)malloc(sizeof(int16_t) * (size_t)n);
    int16_t min_exp = INT16_MAX;
    for (int64_t i = 0; i < n; i++) {
        int64_t d; int16_t e;
        if (!float_to_decimal(src[i], &d, &e)) { free(exps); return BO_ERR_LOSSY_FLOAT; }
        out_ints[i] = d;
        exps[i] = e;
        if (e < min_exp) min_exp = e;
    }
    for (int64_t i = 0; i < n; i++) {
        int diff = exps[i] - min_exp;
        if (diff == 0) continue;
        int64_t scaled;
        if (!mul_pow10(out_ints[i], diff, &scaled)) { free(exps); return BO_ERR_LOSSY_FLOAT; }
        out_ints[i] = scaled;
    }
    free(exps);
    *out_exp = min_exp;
    return BO_OK;
}

/* DecimalIntListToFloat64List — float.go:69-102 */
int bo_decimal_to_float_list(double *dst, const int64_t *vs, int16_t exponent,
                             int64_t n) {
    if (n == 0) return BO_OK;
    if (exponent >= 0) {
        double scale = bo_go_pow10(exponent);
        for (int64_t i = 0; i < n; i++) dst[i] = (double)vs[i] * scale;
    } else {
        double divisors[4];
        int nd = 0;
        int neg_exp = -(int)exponent;
        while (neg_exp > 0) {
            int step = neg_exp < 308 ? neg_exp : 308;
            divisors[nd++] = bo_go_pow10(step);
            neg_exp -= step;
        }
        for (int64_t i = 0; i < n; i++) {
            double r = (double)vs[i];
            for (int k = 0; k < nd; k++) r /= divisors[k];
            dst[i] = r;
        }
    }
    return BO_OK;
}

/* ===================== int64 cell codec =====================
 * convert.Int64ToBytes — pkg/convert/number.go:33-46:
 *   i >= 0: u = |i| | 1<<63 ; i < 0: u = 1<<63 - |i| ; big-endian bytes. */
void bo_cell_i64_to_bytes(uint8_t out[8], int64_t v) {
    uint64_t u;
    if (v >= 0) u = (uint64_t)v | (1ULL << 63);
    else u = (1ULL << 63) - (uint64_t)(-(uint64_t)v);
    for (int i = 0; i < 8; i++) out[i] = (uint8_t)(u >> (56 - 8 * i));
}

/* convert.BytesToInt64 — number.go:95-108 */
int64_t bo_cell_bytes_to_i64(const uint8_t in[8]) {
    uint64_t u = 0;
    for (int i = 0; i < 8; i++) u = (u << 8) | in[i];
    if (in[0] >= 128) {
        u ^= 1ULL << 63;
        return (int64_t)u;
    }
    u = (1ULL << 63) - u;
    return -(int64_t)u;
}

/* ===================== column payload codec =====================
 * encodeInt64Column — banyand/measure/column.go:183-214
 * decodeInt64Column — column.go:331-363 (minus the [][]byte
 * re-materialisation, which is presentation-layer overhead)
 * encode/decodeFloat64Column — column.go:216-263, 365-408 */
int bo_column_i64_encode(uint8_t *dst, size_t cap, const int64_t *vals,
                         int64_t n, size_t *out_len) {
    uint8_t type; int64_t first; size_t stream_len;
    int rc = bo_int64_list_encode(dst + 9, cap - 9, vals, n, &stream_len, &type, &first);
    if (rc != BO_OK) return rc;
    dst[0] = type;
    bo_cell_i64_to_bytes(dst + 1, first);
    *out_len = 9 + stream_len;
    return BO_OK;
}

int bo_column_i64_decode(int64_t *dst, const uint8_t *payload, size_t len,
                         int64_t n) {
    if (len < 1) return BO_ERR_TRUNCATED;
    uint8_t type = payload[0];
    if (type == BO_ENC_PLAIN) return BO_ERR_BAD_TYPE; /* nulls: out of scope */
    if (len < 9) return BO_ERR_TRUNCATED;
    int64_t first = bo_cell_bytes_to_i64(payload + 1);
    return bo_int64_list_decode(dst, payload + 9, len - 9, type, first, n);
}

int bo_column_f64_encode(uint8_t *dst, size_t cap, const double *vals,
                         int64_t n, size_t *out_len) {
    int64_t *ints = (int64_t *)malloc(sizeof(int64_t) * (size_t)n);
    int16_t exp;
    int rc = bo_float_list_to_decimal(vals, n, ints, &exp);
    if (rc != BO_OK) { free(ints); return rc; }
    uint8_t type; int64_t first; size_t stream_len;
    rc = bo_int64_list_encode(dst + 11, cap - 11, ints, n, &stream_len, &type, &first);
    free(ints);
    if (rc != BO_OK) return rc;
    dst[0] = type;
    dst[1] = (uint8_t)((uint16_t)exp >> 8); /* convert.Int16ToBytes: BE */
    dst[2] = (uint8_t)((uint16_t)exp & 0xff);
    bo_cell_i64_to_bytes(dst + 3, first);
    *out_len = 11 + stream_len;
    (void)cap;
    return BO_OK;
}

int bo_column_f64_decode(double *dst, const uint8_t *payload, size_t len,
                         int64_t n) {
    if (len < 1) return BO_ERR_TRUNCATED;
    uint8_t type = payload[0];
    if (type == BO_ENC_PLAIN) return BO_ERR_BAD_TYPE;
    if (len < 11) return BO_ERR_TRUNCATED;
    int16_t exp = (int16_t)(((uint16_t)payload[1] << 8) | payload[2]);
    int64_t first = bo_cell_bytes_to_i64(payload + 3);
    int64_t *ints = (int64_t *)malloc(sizeof(int64_t) * (size_t)n);
    int rc = bo_int64_list_decode(ints, payload + 11, len - 11, type, first, n);
    if (rc == BO_OK) rc = bo_decimal_to_float_list(dst, ints, exp, n);
    free(ints);
    return rc;
}

/* ===================== timestamps payload =====================
 * mustWriteTimestampsTo — block.go:386-404 (ts stream stored under the
 * WithVersion encode-type variant, versions stream appended);
 * mustDecodeTimestampsWithVersions — block.go:425-443. */
static uint8_t to_version_type(uint8_t et) {
    switch (et) {                      /* encoding.GetVersionType, encoding.go:100-114 */
    case BO_ENC_CONST: return BO_ENC_CONST_WITH_VERSION;
    case BO_ENC_DELTA_CONST: return BO_ENC_DELTA_CONST_WITH_VERSION;
    case BO_ENC_DELTA: return BO_ENC_DELTA_WITH_VERSION;
    case BO_ENC_DELTA_OF_DELTA: return BO_ENC_DELTA_OF_DELTA_WITH_VERSION;
    default: return BO_ENC_UNKNOWN;
    }
}
static uint8_t to_common_type(uint8_t et) {
    switch (et) {                      /* encoding.GetCommonType, encoding.go:116-130 */
    case BO_ENC_CONST_WITH_VERSION: return BO_ENC_CONST;
    case BO_ENC_DELTA_CONST_WITH_VERSION: return BO_ENC_DELTA_CONST;
    case BO_ENC_DELTA_WITH_VERSION: return BO_ENC_DELTA;
    case BO_ENC_DELTA_OF_DELTA_WITH_VERSION: return BO_ENC_DELTA_OF_DELTA;
    default: return BO_ENC_UNKNOWN;
    }
}

int bo_timestamps_encode(uint8_t *dst, size_t cap, const int64_t *ts,
                         const int64_t *versions, int64_t n, size_t *out_len,
                         uint8_t *ts_enc_with_version, int64_t *ts_min,
                         int64_t *ts_max, uint64_t *version_offset,
                         uint8_t *version_enc, int64_t *version_first) {
    uint8_t tenc; int64_t tfirst; size_t tlen;
    int rc = bo_int64_list_encode(dst, cap, ts, n, &tlen, &tenc, &tfirst);
    if (rc != BO_OK) return rc;
    uint8_t wv = to_version_type(tenc);
    if (wv == BO_ENC_UNKNOWN) return BO_ERR_BAD_TYPE;
    *ts_enc_with_version = wv;
    *ts_min = tfirst;
    *ts_max = ts[n - 1];
    *version_offset = tlen;
    uint8_t venc; int64_t vfirst; size_t vlen;
    rc = bo_int64_list_encode(dst + tlen, cap - tlen, versions, n, &vlen, &venc, &vfirst);
    if (rc != BO_OK) return rc;
    *version_enc = venc;
    *version_first = vfirst;
    *out_len = tlen + vlen;
    return BO_OK;
}

int bo_timestamps_decode(int64_t *ts, int64_t *versions, const uint8_t *payload,
                         size_t len, uint8_t ts_enc_with_version,
                         int64_t ts_min, uint64_t version_offset,
                         uint8_t version_enc, int64_t version_first, int64_t n) {
    uint8_t t = to_common_type(ts_enc_with_version);
    if (t == BO_ENC_UNKNOWN) return BO_ERR_BAD_TYPE;
    if (len < version_offset) return BO_ERR_TRUNCATED;
    int rc = bo_int64_list_decode(ts, payload, version_offset, t, ts_min, n);
    if (rc != BO_OK) return rc;
    if (versions)
        rc = bo_int64_list_decode(versions, payload + version_offset,
                                  len - version_offset, version_enc,
                                  version_first, n);
    return rc;
}

/* ===================== FindRange =====================
 * pkg/timestamp/range.go:143-170 — linear scans from both ends,
 * ascending- or descending-aware, inclusive [min,max]. */
int bo_find_range(const int64_t *ts, int64_t n, int64_t min_val,
                  int64_t max_val, int64_t *start, int64_t *end) {
    if (n == 0) { *start = -1; *end = -1; return 0; }
    int is_asc = ts[0] <= ts[n - 1];
    if (is_asc && (ts[0] > max_val || ts[n - 1] < min_val)) { *start = -1; *end = -1; return 0; }
    if (!is_asc && (ts[0] < min_val || ts[n - 1] > max_val)) { *start = -1; *end = -1; return 0; }
    int64_t s = -1, e = n;
    while (s < n - 1) {
        s++;
        if ((is_asc && ts[s] >= min_val) || (!is_asc && ts[s] <= max_val)) break;
    }
    while (e > 0) {
        e--;
        if ((is_asc && ts[e] <= max_val) || (!is_asc && ts[e] >= min_val)) break;
    }
    *start = s;
    *end = e;
    return s <= e;
}

/* ===================== zstd (dlopen; format-compatible per RFC 8878) ===== */
typedef size_t (*zstd_compress_fn)(void *, size_t, const void *, size_t, int);
typedef size_t (*zstd_decompress_fn)(void *, size_t, const void *, size_t);
typedef unsigned (*zstd_iserr_fn)(size_t);
static zstd_compress_fn p_zstd_compress;
static zstd_decompress_fn p_zstd_decompress;
static zstd_iserr_fn p_zstd_iserr;
static int zstd_loaded = -1;

static int load_zstd(void) {
    if (zstd_loaded >= 0) return zstd_loaded;
    void *h = dlopen("libzstd.so.1", RTLD_NOW | RTLD_GLOBAL);
    if (!h) h = dlopen("libzstd.so", RTLD_NOW | RTLD_GLOBAL);
    if (h) {
        p_zstd_compress = (zstd_compress_fn)dlsym(h, "ZSTD_compress");
        p_zstd_decompress = (zstd_decompress_fn)dlsym(h, "ZSTD_decompress");
        p_zstd_iserr = (zstd_iserr_fn)dlsym(h, "ZSTD_isError");
    }
    zstd_loaded = (p_zstd_compress && p_zstd_decompress && p_zstd_iserr) ? 1 : 0;
    return zstd_loaded;
}

/* compressBlock — pkg/encoding/bytes.go:291-305: <128 B -> [0, len] plain,
 * else [1, varuint(len), zstd level 1]. */
static int compress_block(uint8_t *dst, size_t *o, const uint8_t *src, size_t n) {
    if (n < 128) {
        dst[(*o)++] = 0;
        dst[(*o)++] = (uint8_t)n;
        memcpy(dst + *o, src, n);
        *o += n;
        return BO_OK;
    }
    if (!load_zstd()) return BO_ERR_ZSTD_UNAVAILABLE;
    dst[(*o)++] = 1;
    uint8_t tmp_head[10];
    size_t bound = n + n / 2 + 256;
    uint8_t *tmp = (uint8_t *)malloc(bound);
    size_t clen = p_zstd_compress(tmp, bound, src, n, 1);
    if (p_zstd_iserr(clen)) { free(tmp); return BO_ERR_BAD_DATA; }
    size_t hl = bo_varuint64_encode(tmp_head, clen);
    memcpy(dst + *o, tmp_head, hl); *o += hl;
    memcpy(dst + *o, tmp, clen); *o += clen;
    free(tmp);
    return BO_OK;
}

/* decompressBlock — bytes.go:306-345 */
static int decompress_block(uint8_t *dst, size_t dst_cap, size_t *dst_len,
                            const uint8_t *src, size_t src_len, size_t *consumed) {
    if (src_len < 1) return BO_ERR_TRUNCATED;
    size_t o = 0;
    uint8_t bt = src[o++];
    if (bt == 0) {
        if (src_len < 2) return BO_ERR_TRUNCATED;
        size_t bl = src[o++];
        if (src_len - o < bl) return BO_ERR_TRUNCATED;
        if (bl > dst_cap) return BO_ERR_CAPACITY;
        memcpy(dst, src + o, bl);
        o += bl;
        *dst_len = bl;
        *consumed = o;
        return BO_OK;
    }
    if (bt == 1) {
        uint64_t bl; size_t c;
        int rc = bo_varuint64_decode(src + o, src_len - o, &bl, &c);
        if (rc != BO_OK) return rc;
        o += c;
        if (src_len - o < bl) return BO_ERR_TRUNCATED;
        if (!load_zstd()) return BO_ERR_ZSTD_UNAVAILABLE;
        size_t dl = p_zstd_decompress(dst, dst_cap, src + o, bl);
        if (p_zstd_iserr(dl)) return BO_ERR_BAD_DATA;
        o += bl;
        *dst_len = dl;
        *consumed = o;
        return BO_OK;
    }
    return BO_ERR_BAD_TYPE;
}

/* encodeUint64List — bytes.go:205-235 (width-typed: 0=u8,1=u16 BE,2=u32 BE,3=u64 BE) */
static size_t encode_u64_list(uint8_t *dst, const uint64_t *a, int64_t n) {
    uint64_t nmax = 0;
    for (int64_t i = 0; i < n; i++) if (a[i] > nmax) nmax = a[i];
    size_t o = 0;
    if (nmax < (1ULL << 8)) {
        dst[o++] = 0;
        for (int64_t i = 0; i < n; i++) dst[o++] = (uint8_t)a[i];
    } else if (nmax < (1ULL << 16)) {
        dst[o++] = 1;
        for (int64_t i = 0; i < n; i++) { dst[o++] = (uint8_t)(a[i] >> 8); dst[o++] = (uint8_t)a[i]; }
    } else if (nmax < (1ULL << 32)) {
        dst[o++] = 2;
        for (int64_t i = 0; i < n; i++)
            for (int b = 3; b >= 0; b--) dst[o++] = (uint8_t)(a[i] >> (8 * b));
    } else {
        dst[o++] = 3;
        for (int64_t i = 0; i < n; i++)
            for (int b = 7; b >= 0; b--) dst[o++] = (uint8_t)(a[i] >> (8 * b));
    }
    return o;
}

static int decode_u64_list(uint64_t *dst, const uint8_t *src, size_t src_len,
                           int64_t n) {
    if (src_len < 1) return BO_ERR_TRUNCATED;
    uint8_t bt = src[0];
    const uint8_t *p = src + 1;
    size_t width = bt == 0 ? 1 : bt == 1 ? 2 : bt == 2 ? 4 : bt == 3 ? 8 : 0;
    if (!width) return BO_ERR_BAD_TYPE;
    if (src_len - 1 != width * (size_t)n) return BO_ERR_BAD_DATA;
    for (int64_t i = 0; i < n; i++) {
        uint64_t v = 0;
        for (size_t b = 0; b < width; b++) v = (v << 8) | p[i * width + b];
        dst[i] = v;
    }
    return BO_OK;
}

/* EncodeUint64Block / DecodeUint64Block — bytes.go:168-196 */
static int encode_u64_block(uint8_t *dst, size_t *o, const uint64_t *a, int64_t n) {
    uint8_t *tmp = (uint8_t *)malloc((size_t)n * 8 + 1);
    size_t tl = encode_u64_list(tmp, a, n);
    int rc = compress_block(dst, o, tmp, tl);
    free(tmp);
    return rc;
}

static int decode_u64_block(uint64_t *dst, const uint8_t *src, size_t src_len,
                            int64_t n, size_t *consumed) {
    size_t cap = (size_t)n * 8 + 1;
    uint8_t *tmp = (uint8_t *)malloc(cap);
    size_t tl = 0, c = 0;
    int rc = decompress_block(tmp, cap, &tl, src, src_len, &c);
    if (rc == BO_OK) rc = decode_u64_list(dst, tmp, tl, n);
    free(tmp);
    if (rc == BO_OK && consumed) *consumed = c;
    return rc;
}

/* EncodeBytesBlock — bytes.go:45-70: width-typed uint64 lengths block
 * (nil -> 0, value -> len+1), then compressed concatenated payload.
 * lens[i] < 0 denotes nil. */
int bo_bytes_block_encode(uint8_t *dst, size_t cap, const uint8_t *data,
                          const int64_t *lens, int64_t n, size_t *out_len) {
    (void)cap;
    uint64_t *alens = (uint64_t *)malloc(sizeof(uint64_t) * (size_t)n);
    size_t total = 0;
    for (int64_t i = 0; i < n; i++) {
        if (lens[i] < 0) alens[i] = 0;
        else { alens[i] = (uint64_t)lens[i] + 1; total += (size_t)lens[i]; }
    }
    size_t o = 0;
    int rc = encode_u64_block(dst, &o, alens, n);
    free(alens);
    if (rc != BO_OK) return rc;
    /* data is already the concatenation of non-nil values in row order */
    rc = compress_block(dst, &o, data, total);
    if (rc != BO_OK) return rc;
    *out_len = o;
    return BO_OK;
}

/* Plain (null-bearing) int64 column: a bytes block of 8-B sign-flip
 * cells (convert/number.go:33-46), nil rows zero-length
 * (column.go:214-233 write fallback).  Fills dst and valid; the fold
 * must skip invalid rows (aggregation.go:310 null check). */
static int column_i64_decode_plain(int64_t *dst, uint8_t *valid,
                                   const uint8_t *src, size_t len,
                                   int64_t n) {
    if (len < 1 || src[0] != BO_ENC_PLAIN) return BO_ERR_BAD_TYPE;
    size_t cap = (size_t)n * 8 + 16;
    uint8_t *data = (uint8_t *)malloc(cap);
    int64_t *lens = (int64_t *)malloc(sizeof(int64_t) * (size_t)n);
    size_t dl = 0;
    int rc = bo_bytes_block_decode(data, cap, lens, src + 1, len - 1, n, &dl);
    if (rc == BO_OK) {
        size_t off = 0;
        for (int64_t i = 0; i < n; i++) {
            if (lens[i] < 0) { valid[i] = 0; dst[i] = 0; continue; }
            if (lens[i] != 8) { rc = BO_ERR_BAD_DATA; break; }
            valid[i] = 1;
            dst[i] = bo_cell_bytes_to_i64(data + off);
            off += 8;
        }
    }
    free(data); free(lens);
    return rc;
}

/* Plain (null-bearing) float64 column: 8-B big-endian IEEE-754 cells
 * (convert/number.go:128-132), nil rows zero-length. */
static int column_f64_decode_plain(double *dst, uint8_t *valid,
                                   const uint8_t *src, size_t len,
                                   int64_t n) {
    if (len < 1 || src[0] != BO_ENC_PLAIN) return BO_ERR_BAD_TYPE;
    size_t cap = (size_t)n * 8 + 16;
    uint8_t *data = (uint8_t *)malloc(cap);
    int64_t *lens = (int64_t *)malloc(sizeof(int64_t) * (size_t)n);
    size_t dl = 0;
    int rc = bo_bytes_block_decode(data, cap, lens, src + 1, len - 1, n, &dl);
    if (rc == BO_OK) {
        size_t off = 0;
        for (int64_t i = 0; i < n; i++) {
            if (lens[i] < 0) { valid[i] = 0; dst[i] = 0; continue; }
            if (lens[i] != 8) { rc = BO_ERR_BAD_DATA; break; }
            valid[i] = 1;
            uint64_t bits = 0;
            for (int k = 0; k < 8; k++) bits = (bits << 8) | data[off + k];
            memcpy(&dst[i], &bits, 8);
            off += 8;
        }
    }
    free(data); free(lens);
    return rc;
}

/* BytesBlockDecoder.Decode — bytes.go:84-130 */
int bo_bytes_block_decode(uint8_t *data_out, size_t data_cap, int64_t *lens_out,
                          const uint8_t *src, size_t src_len, int64_t n,
                          size_t *data_len_out) {
    uint64_t *alens = (uint64_t *)malloc(sizeof(uint64_t) * (size_t)n);
    size_t c = 0;
    int rc = decode_u64_block(alens, src, src_len, n, &c);
    if (rc != BO_OK) { free(alens); return rc; }
    size_t dl = 0, c2 = 0;
    rc = decompress_block(data_out, data_cap, &dl, src + c, src_len - c, &c2);
    if (rc != BO_OK) { free(alens); return rc; }
    if (c + c2 != src_len) { free(alens); return BO_ERR_TAIL; }
    size_t need = 0;
    for (int64_t i = 0; i < n; i++) {
        if (alens[i] == 0) { lens_out[i] = -1; continue; }
        uint64_t al = alens[i] - 1;
        lens_out[i] = (int64_t)al;
        need += al;
    }
    if (need != dl) { free(alens); return BO_ERR_BAD_DATA; }
    free(alens);
    *data_len_out = dl;
    return BO_OK;
}

/* ---- MSB-first bit writer/reader — pkg/encoding/writer.go:24-93,
 * reader.go:24-100 ---- */
typedef struct { uint8_t *buf; size_t len; uint8_t cache; uint8_t avail; } bitw;
static void bw_init(bitw *w, uint8_t *buf) { w->buf = buf; w->len = 0; w->cache = 0; w->avail = 8; }
static void bw_byte(bitw *w, uint8_t b) {
    w->buf[w->len++] = (uint8_t)(w->cache | (b >> (8 - w->avail)));
    w->cache = (uint8_t)(w->avail == 8 ? 0 : b << w->avail);
}
static void bw_bits(bitw *w, uint64_t u, int num) {
    u <<= (64 - (unsigned)num);
    for (; num >= 8; num -= 8) { bw_byte(w, (uint8_t)(u >> 56)); u <<= 8; }
    uint8_t rem = (uint8_t)(u >> 56);
    for (; num > 0; num--) {
        if (rem & 0x80) w->cache |= (uint8_t)(1u << (w->avail - 1));
        w->avail--;
        if (w->avail == 0) { w->buf[w->len++] = w->cache; w->cache = 0; w->avail = 8; }
        rem <<= 1;
    }
}
static void bw_flush(bitw *w) {
    if (w->avail != 8) { w->buf[w->len++] = w->cache; }
    w->cache = 0; w->avail = 8;
}

typedef struct { const uint8_t *buf; size_t len, pos; uint8_t cache; uint8_t nbits; } bitr;
static void br_init(bitr *r, const uint8_t *buf, size_t len) { r->buf = buf; r->len = len; r->pos = 0; r->cache = 0; r->nbits = 0; }
static int br_bool(bitr *r, int *out) {
    if (r->nbits == 0) {
        if (r->pos >= r->len) return BO_ERR_TRUNCATED;
        r->cache = r->buf[r->pos++];
        r->nbits = 8;
    }
    r->nbits--;
    *out = (r->cache & 0x80) != 0;
    r->cache <<= 1;
    return BO_OK;
}
static int br_byte(bitr *r, uint8_t *out) {
    if (r->pos >= r->len) return BO_ERR_TRUNCATED;
    uint8_t b = r->buf[r->pos++];
    if (r->nbits == 0) { *out = b; r->cache = b; return BO_OK; }
    *out = (uint8_t)(r->cache | (b >> r->nbits));
    r->cache = (uint8_t)(b << (8 - r->nbits));
    return BO_OK;
}
static int br_bits(bitr *r, int num, uint64_t *out) {
    uint64_t result = 0;
    for (; num >= 8; num -= 8) {
        uint8_t b;
        int rc = br_byte(r, &b);
        if (rc != BO_OK) return rc;
        result = (result << 8) | b;
    }
    for (; num > 0; num--) {
        int bit;
        int rc = br_bool(r, &bit);
        if (rc != BO_OK) return rc;
        result = (result << 1) | (uint64_t)bit;
    }
    *out = result;
    return BO_OK;
}

/* encodeRLE/decodeRLE + bit-packing — dictionary.go:158-260 */
static int64_t rle_encode(uint32_t *dst, const uint32_t *src, int64_t n) {
    if (n == 0) return 0;
    int64_t o = 0;
    uint32_t cur = src[0], count = 1;
    for (int64_t i = 1; i < n; i++) {
        if (src[i] == cur) count++;
        else { dst[o++] = cur; dst[o++] = count; cur = src[i]; count = 1; }
    }
    dst[o++] = cur; dst[o++] = count;
    return o;
}

static size_t bitpack_encode(uint8_t *dst, const uint32_t *src, int64_t n) {
    bitw w; bw_init(&w, dst);
    if (n == 0) { bw_bits(&w, 0, 32); bw_flush(&w); return w.len; }
    bw_bits(&w, (uint64_t)n, 32);
    uint32_t maxv = 0;
    for (int64_t i = 0; i < n; i++) if (src[i] > maxv) maxv = src[i];
    int width = 1;
    if (maxv > 0) { width = 0; uint32_t m = maxv; while (m) { width++; m >>= 1; } }
    bw_bits(&w, (uint64_t)width, 8);
    for (int64_t i = 0; i < n; i++) bw_bits(&w, src[i], width);
    bw_flush(&w);
    return w.len;
}

static int bitpack_decode(uint32_t *dst, int64_t cap, const uint8_t *src,
                          size_t src_len, int64_t *out_n) {
    bitr r; br_init(&r, src, src_len);
    uint64_t length;
    int rc = br_bits(&r, 32, &length);
    if (rc != BO_OK) return rc;
    if (length == 0) { *out_n = 0; return BO_OK; }
    uint64_t width;
    rc = br_bits(&r, 8, &width);
    if (rc != BO_OK) return rc;
    if ((int64_t)length > cap) return BO_ERR_CAPACITY;
    for (uint64_t i = 0; i < length; i++) {
        uint64_t v;
        rc = br_bits(&r, (int)width, &v);
        if (rc != BO_OK) return rc;
        dst[i] = (uint32_t)v;
    }
    *out_n = (int64_t)length;
    return BO_OK;
}

/* Dictionary.Encode — dictionary.go:79-87:
 * varuint(count) ++ EncodeBytesBlock(values) ++ bitpack(RLE(indices)) */
int bo_dictionary_encode(uint8_t *dst, size_t cap, const uint8_t *data,
                         const int64_t *lens, int64_t n, size_t *out_len) {
    (void)cap;
    /* Dictionary.Add — dictionary.go:51-66 (nil-aware equality :69-77) */
    const uint8_t *vals[256];
    int64_t vlens[256];
    int64_t nvals = 0;
    uint32_t *indices = (uint32_t *)malloc(sizeof(uint32_t) * (size_t)n);
    const uint8_t *p = data;
    for (int64_t i = 0; i < n; i++) {
        const uint8_t *v = lens[i] < 0 ? NULL : p;
        int64_t vl = lens[i];
        if (vl > 0) p += vl;
        int64_t found = -1;
        for (int64_t j = 0; j < nvals; j++) {
            int eq;
            if (vlens[j] < 0 && vl < 0) eq = 1;
            else if (vlens[j] < 0 || vl < 0) eq = 0;
            else eq = (vlens[j] == vl) && (memcmp(vals[j], v, (size_t)vl) == 0);
            if (eq) { found = j; break; }
        }
        if (found < 0) {
            if (nvals == 256) { free(indices); return BO_ERR_CAPACITY; }
            vals[nvals] = v; vlens[nvals] = vl;
            found = nvals++;
        }
        indices[i] = (uint32_t)found;
    }
    size_t o = bo_varuint64_encode(dst, (uint64_t)nvals);
    /* concatenated dict values + lens for EncodeBytesBlock */
    size_t vtotal = 0;
    for (int64_t j = 0; j < nvals; j++) if (vlens[j] > 0) vtotal += (size_t)vlens[j];
    uint8_t *vdata = (uint8_t *)malloc(vtotal ? vtotal : 1);
    size_t vo = 0;
    for (int64_t j = 0; j < nvals; j++)
        if (vlens[j] > 0) { memcpy(vdata + vo, vals[j], (size_t)vlens[j]); vo += (size_t)vlens[j]; }
    size_t bl = 0;
    int rc = bo_bytes_block_encode(dst + o, cap - o, vdata, vlens, nvals, &bl);
    free(vdata);
    if (rc != BO_OK) { free(indices); return rc; }
    o += bl;
    uint32_t *rle = (uint32_t *)malloc(sizeof(uint32_t) * 2 * (size_t)n);
    int64_t rn = rle_encode(rle, indices, n);
    o += bitpack_encode(dst + o, rle, rn);
    free(rle);
    free(indices);
    *out_len = o;
    return BO_OK;
}

/* Dictionary decode to per-row codes — dictionary.go:90-115 minus value
 * materialisation (kernel parity target: codes only). */
int bo_dictionary_decode_codes(uint32_t *codes_out, const uint8_t *src,
                               size_t src_len, int64_t n) {
    uint64_t count; size_t c;
    int rc = bo_varuint64_decode(src, src_len, &count, &c);
    if (rc != BO_OK) return rc;
    if (count == 0) return n == 0 ? BO_OK : BO_ERR_BAD_DATA;
    src += c; src_len -= c;
    /* skip the values bytes block: lengths block + payload block */
    uint64_t *alens = (uint64_t *)malloc(sizeof(uint64_t) * count);
    size_t c1 = 0;
    rc = decode_u64_block(alens, src, src_len, (int64_t)count, &c1);
    free(alens);
    if (rc != BO_OK) return rc;
    src += c1; src_len -= c1;
    /* skip compressed payload */
    size_t big = 1 << 22;
    uint8_t *tmp = (uint8_t *)malloc(big);
    size_t dl = 0, c2 = 0;
    rc = decompress_block(tmp, big, &dl, src, src_len, &c2);
    free(tmp);
    if (rc != BO_OK) return rc;
    src += c2; src_len -= c2;
    /* bit-unpack RLE pairs then expand */
    int64_t rcap = 2 * n + 2;
    uint32_t *rle = (uint32_t *)malloc(sizeof(uint32_t) * (size_t)rcap);
    int64_t rn = 0;
    rc = bitpack_decode(rle, rcap, src, src_len, &rn);
    if (rc != BO_OK) { free(rle); return rc; }
    int64_t o = 0;
    for (int64_t i = 0; i + 1 < rn; i += 2) {
        uint32_t v = rle[i], cnt = rle[i + 1];
        for (uint32_t k = 0; k < cnt; k++) {
            if (o >= n) { free(rle); return BO_ERR_BAD_DATA; }
            codes_out[o++] = v;
        }
    }
    free(rle);
    return o == n ? BO_OK : BO_ERR_BAD_DATA;
}

/* Dictionary.Decode — dictionary.go:90-115 (full values) */
int bo_dictionary_decode(uint8_t *data_out, size_t data_cap, int64_t *lens_out,
                         const uint8_t *src, size_t src_len, int64_t n,
                         size_t *data_len_out) {
    uint64_t count; size_t c;
    int rc = bo_varuint64_decode(src, src_len, &count, &c);
    if (rc != BO_OK) return rc;
    if (count == 0) { *data_len_out = 0; return n == 0 ? BO_OK : BO_ERR_BAD_DATA; }
    src += c; src_len -= c;
    uint64_t *alens = (uint64_t *)malloc(sizeof(uint64_t) * count);
    size_t c1 = 0;
    rc = decode_u64_block(alens, src, src_len, (int64_t)count, &c1);
    if (rc != BO_OK) { free(alens); return rc; }
    src += c1; src_len -= c1;
    size_t big = data_cap + 16;
    uint8_t *vdata = (uint8_t *)malloc(big);
    size_t dl = 0, c2 = 0;
    rc = decompress_block(vdata, big, &dl, src, src_len, &c2);
    if (rc != BO_OK) { free(alens); free(vdata); return rc; }
    src += c2; src_len -= c2;
    /* dict value offsets */
    uint64_t voff[257];
    voff[0] = 0;
    for (uint64_t j = 0; j < count; j++) {
        uint64_t al = alens[j] == 0 ? 0 : alens[j] - 1;
        voff[j + 1] = voff[j] + al;
    }
    if (voff[count] != dl) { free(alens); free(vdata); return BO_ERR_BAD_DATA; }
    uint32_t *codes = (uint32_t *)malloc(sizeof(uint32_t) * (size_t)n);
    /* re-decode codes from the remaining bitpack */
    int64_t rcap = 2 * n + 2;
    uint32_t *rle = (uint32_t *)malloc(sizeof(uint32_t) * (size_t)rcap);
    int64_t rn = 0;
    rc = bitpack_decode(rle, rcap, src, src_len, &rn);
    int64_t o = 0;
    if (rc == BO_OK) {
        for (int64_t i = 0; i + 1 < rn && rc == BO_OK; i += 2) {
            uint32_t v = rle[i], cnt = rle[i + 1];
            for (uint32_t k = 0; k < cnt; k++) {
                if (o >= n || v >= count) { rc = BO_ERR_BAD_DATA; break; }
                codes[o++] = v;
            }
        }
        if (rc == BO_OK && o != n) rc = BO_ERR_BAD_DATA;
    }
    size_t po = 0;
    if (rc == BO_OK) {
        for (int64_t i = 0; i < n; i++) {
            uint32_t v = codes[i];
            if (alens[v] == 0) { lens_out[i] = -1; continue; }
            uint64_t al = alens[v] - 1;
            if (po + al > data_cap) { rc = BO_ERR_CAPACITY; break; }
            memcpy(data_out + po, vdata + voff[v], al);
            lens_out[i] = (int64_t)al;
            po += al;
        }
    }
    free(rle); free(codes); free(alens); free(vdata);
    if (rc == BO_OK) *data_len_out = po;
    return rc;
}

/* ===================== MEAN finalisation =====================
 * meanFunc.Val — pkg/query/aggregation/function.go:30-45: sum/count,
 * clamped to >= 1 (Go int64 division truncates toward zero, same as C). */
int64_t bo_mean_val_i64(int64_t sum, int64_t count) {
    if (count == 0) return 0;
    int64_t v = sum / count;
    return v < 1 ? 1 : v;
}

double bo_mean_val_f64(double sum, double count) {
    if (count == 0) return 0;
    double v = sum / count;
    return v < 1 ? 1 : v;
}

/* ===================== scan + aggregate ===================== */
static void agg_reset(bo_agg_result *r) {
    r->sum_i = 0; r->sum_f = 0; r->count = 0;
    r->min_i = INT64_MAX; r->max_i = INT64_MIN;       /* function.go:Reset */
    r->min_f = 1.7976931348623157e308; r->max_f = -1.7976931348623157e308;
}

static int scan_block(const uint8_t *payload, const bo_block_desc *b,
                      int field_vtype, int64_t min_ts, int64_t max_ts,
                      const uint8_t *preds_concat, const int64_t *pred_lens,
                      bo_agg_result *r,
                      int64_t *ts_buf, int64_t *i64_buf, double *f64_buf,
                      uint8_t *tagdata_buf, int64_t *taglen_buf,
                      uint8_t *rowmatch_buf) {
    int64_t n = (int64_t)b->count;
    int rc = bo_timestamps_decode(ts_buf, NULL, payload + b->ts_off,
                                  (size_t)(b->ts_len + b->ver_len),
                                  b->ts_enc_with_version, b->ts_min, b->ts_len,
                                  b->version_enc, b->version_first, n);
    if (rc != BO_OK) return rc;
    int64_t r0, r1;
    if (!bo_find_range(ts_buf, n, min_ts, max_ts, &r0, &r1)) return BO_OK;
    /* conjunctive tag-equality predicates (up to 3, one per tag slot).
     * A block lacking a predicated tag column has nil tags: never equal. */
    int have_pred = 0;
    for (int64_t i = 0; i < n; i++) rowmatch_buf[i] = 1;
    int64_t pred_off = 0;
    for (int sl = 0; sl < 3; sl++) {
        int64_t plen = pred_lens ? pred_lens[sl] : 0;
        if (plen == 0) continue;
        const uint8_t *pred = preds_concat + pred_off;
        pred_off += plen;
        have_pred = 1;
        uint64_t toff = sl == 0 ? b->tag_off : sl == 1 ? b->tag2_off : b->tag3_off;
        uint64_t tlen = sl == 0 ? b->tag_len : sl == 1 ? b->tag2_len : b->tag3_len;
        if (tlen == 0) return BO_OK;  /* nil tags: exclude whole block */
        uint8_t tag_type = payload[toff];
        size_t tagdata_len = 0;
        if (tag_type == BO_ENC_DICTIONARY) {
            rc = bo_dictionary_decode(tagdata_buf, (size_t)1 << 24, taglen_buf,
                                      payload + toff + 1, tlen - 1, n,
                                      &tagdata_len);
            if (rc != BO_OK) return rc;
        } else if (tag_type == BO_ENC_PLAIN) {
            /* residual pushdown on a plain bytes column — the >256-distinct
             * dictionary bail (column.go:266-278, dictionary.go:58):
             * decode the bytes block and compare per row */
            rc = bo_bytes_block_decode(tagdata_buf, (size_t)1 << 24,
                                       taglen_buf, payload + toff + 1,
                                       tlen - 1, n, &tagdata_len);
            if (rc != BO_OK) return rc;
        } else {
            return BO_ERR_BAD_TYPE;
        }
        size_t tago = 0;
        for (int64_t i = 0; i < n; i++) {
            int64_t tl = taglen_buf[i];
            size_t my = tago;
            if (tl > 0) tago += (size_t)tl;
            if (tl != plen ||
                (tl > 0 && memcmp(tagdata_buf + my, pred, (size_t)tl) != 0))
                rowmatch_buf[i] = 0;
        }
    }
    (void)have_pred;
    if (field_vtype == BO_VT_INT64) {
        if (payload[b->col_off] == BO_ENC_PLAIN) {
            uint8_t valid[8192];
            rc = column_i64_decode_plain(i64_buf, valid,
                                         payload + b->col_off, b->col_len, n);
            if (rc != BO_OK) return rc;
            for (int64_t i = 0; i < n; i++)
                if (!valid[i]) rowmatch_buf[i] = 0;
        } else {
            rc = bo_column_i64_decode(i64_buf, payload + b->col_off,
                                      b->col_len, n);
            if (rc != BO_OK) return rc;
        }
        for (int64_t i = r0; i <= r1; i++) {
            if (!rowmatch_buf[i]) continue;
            int64_t v = i64_buf[i];
            r->sum_i = (int64_t)((uint64_t)r->sum_i + (uint64_t)v);
            r->count++;
            if (v < r->min_i) r->min_i = v;
            if (v > r->max_i) r->max_i = v;
        }
    } else if (field_vtype == BO_VT_FLOAT64) {
        if (payload[b->col_off] == BO_ENC_PLAIN) {
            uint8_t validf[8192];
            rc = column_f64_decode_plain(f64_buf, validf,
                                         payload + b->col_off, b->col_len, n);
            if (rc != BO_OK) return rc;
            for (int64_t i = 0; i < n; i++)
                if (!validf[i]) rowmatch_buf[i] = 0;
        } else {
            rc = bo_column_f64_decode(f64_buf, payload + b->col_off,
                                      b->col_len, n);
            if (rc != BO_OK) return rc;
        }
        for (int64_t i = r0; i <= r1; i++) {
            if (!rowmatch_buf[i]) continue;
            double v = f64_buf[i];
            r->sum_f += v;
            r->count++;
            if (v < r->min_f) r->min_f = v;
            if (v > r->max_f) r->max_f = v;
        }
    } else {
        return BO_ERR_BAD_TYPE;
    }
    return BO_OK;
}

#define MAX_BLOCK_ROWS 8192 /* banyand/measure/measure.go:41-46 */

int bo_scan_agg(const uint8_t *payload, const bo_block_desc *blocks,
                int64_t n_blocks, int field_vtype, int64_t min_ts,
                int64_t max_ts, const uint8_t *pred, int64_t pred_len,
                bo_agg_result *out) {
    return bo_scan_agg_grouped(payload, blocks, n_blocks, field_vtype, min_ts,
                               max_ts, pred, pred_len, out, 1);
}

int bo_scan_agg_grouped(const uint8_t *payload, const bo_block_desc *blocks,
                        int64_t n_blocks, int field_vtype, int64_t min_ts,
                        int64_t max_ts, const uint8_t *pred, int64_t pred_len,
                        bo_agg_result *out, int64_t n_groups) {
    int64_t lens[3] = {pred_len, 0, 0};
    return bo_scan_agg_multi(payload, blocks, n_blocks, field_vtype, min_ts,
                             max_ts, pred, lens, out, n_groups);
}

int bo_scan_agg_multi(const uint8_t *payload, const bo_block_desc *blocks,
                      int64_t n_blocks, int field_vtype, int64_t min_ts,
                      int64_t max_ts, const uint8_t *preds_concat,
                      const int64_t pred_lens[3], bo_agg_result *out,
                      int64_t n_groups) {
    for (int64_t g = 0; g < n_groups; g++) agg_reset(&out[g]);
    int64_t *ts_buf = (int64_t *)malloc(sizeof(int64_t) * MAX_BLOCK_ROWS);
    int64_t *i64_buf = (int64_t *)malloc(sizeof(int64_t) * MAX_BLOCK_ROWS);
    double *f64_buf = (double *)malloc(sizeof(double) * MAX_BLOCK_ROWS);
    uint8_t *tagdata_buf = (uint8_t *)malloc((size_t)1 << 24);
    int64_t *taglen_buf = (int64_t *)malloc(sizeof(int64_t) * MAX_BLOCK_ROWS);
    uint8_t *rowmatch_buf = (uint8_t *)malloc(MAX_BLOCK_ROWS);
    int rc = BO_OK;
    for (int64_t i = 0; i < n_blocks && rc == BO_OK; i++) {
        uint32_t g = blocks[i].group_code;
        if ((int64_t)g >= n_groups) { rc = BO_ERR_BAD_DATA; break; }
        rc = scan_block(payload, &blocks[i], field_vtype, min_ts, max_ts,
                        preds_concat, pred_lens, &out[g], ts_buf, i64_buf,
                        f64_buf, tagdata_buf, taglen_buf, rowmatch_buf);
    }
    free(ts_buf); free(i64_buf); free(f64_buf); free(tagdata_buf);
    free(taglen_buf); free(rowmatch_buf);
    return rc;
}

int bo_scan_agg_bytags(const uint8_t *payload, const bo_block_desc *blocks,
                       int64_t n_blocks, int field_vtype, int64_t min_ts,
                       int64_t max_ts, const int *slots, int n_slots,
                       const uint8_t *const *dom_blobs,
                       const int64_t *const *dom_lens, const int64_t *n_doms,
                       const uint8_t *preds_concat, const int64_t *pred_lens,
                       bo_agg_result *out) {
    int64_t total = 1;
    for (int i = 0; i < n_slots; i++) total *= n_doms[i];
    for (int64_t g = 0; g < total; g++) agg_reset(&out[g]);
    int64_t *ts_buf = (int64_t *)malloc(sizeof(int64_t) * MAX_BLOCK_ROWS);
    int64_t *i64_buf = (int64_t *)malloc(sizeof(int64_t) * MAX_BLOCK_ROWS);
    double *f64_buf = (double *)malloc(sizeof(double) * MAX_BLOCK_ROWS);
    uint8_t *tagdata = (uint8_t *)malloc((size_t)1 << 24);
    int64_t *taglen = (int64_t *)malloc(sizeof(int64_t) * MAX_BLOCK_ROWS);
    int64_t *rowgid = (int64_t *)malloc(sizeof(int64_t) * MAX_BLOCK_ROWS);
    int rc = BO_OK;
    for (int64_t i = 0; i < n_blocks && rc == BO_OK; i++) {
        const bo_block_desc *b = &blocks[i];
        int64_t n = (int64_t)b->count;
        rc = bo_timestamps_decode(ts_buf, NULL, payload + b->ts_off,
                                  (size_t)(b->ts_len + b->ver_len),
                                  b->ts_enc_with_version, b->ts_min, b->ts_len,
                                  b->version_enc, b->version_first, n);
        if (rc != BO_OK) break;
        int64_t r0, r1;
        if (!bo_find_range(ts_buf, n, min_ts, max_ts, &r0, &r1)) continue;
        for (int64_t r = 0; r < n; r++) rowgid[r] = 0;
        /* conjunctive tag-equality predicates (same contract as the
         * scalar fold: nil-tag block excluded; dictionary or plain) */
        int pdropped = 0;
        if (pred_lens) {
            int64_t pred_off = 0;
            for (int sl = 0; sl < 3 && rc == BO_OK && !pdropped; sl++) {
                int64_t plen = pred_lens[sl];
                if (plen == 0) continue;
                const uint8_t *pred = preds_concat + pred_off;
                pred_off += plen;
                uint64_t toff2 = sl == 0 ? b->tag_off
                                 : sl == 1 ? b->tag2_off : b->tag3_off;
                uint64_t tlen2 = sl == 0 ? b->tag_len
                                 : sl == 1 ? b->tag2_len : b->tag3_len;
                if (tlen2 == 0) { pdropped = 1; break; }
                uint8_t tt = payload[toff2];
                size_t tdl = 0;
                if (tt == BO_ENC_DICTIONARY)
                    rc = bo_dictionary_decode(tagdata, (size_t)1 << 24,
                                              taglen, payload + toff2 + 1,
                                              tlen2 - 1, n, &tdl);
                else if (tt == BO_ENC_PLAIN)
                    rc = bo_bytes_block_decode(tagdata, (size_t)1 << 24,
                                               taglen, payload + toff2 + 1,
                                               tlen2 - 1, n, &tdl);
                else
                    rc = BO_ERR_BAD_TYPE;
                if (rc != BO_OK) break;
                size_t tago = 0;
                for (int64_t r = 0; r < n; r++) {
                    int64_t tl = taglen[r];
                    size_t my = tago;
                    if (tl > 0) tago += (size_t)tl;
                    if (tl != plen ||
                        (tl > 0 &&
                         memcmp(tagdata + my, pred, (size_t)tl) != 0))
                        rowgid[r] = -1;
                }
            }
        }
        if (rc != BO_OK) break;
        if (pdropped) continue;
        int64_t mul = 1;
        int dropped = 0;
        for (int si = 0; si < n_slots && rc == BO_OK && !dropped; si++) {
            int slot = slots[si];
            uint64_t toff = slot == 0 ? b->tag_off : slot == 1 ? b->tag2_off : b->tag3_off;
            uint64_t tlen = slot == 0 ? b->tag_len : slot == 1 ? b->tag2_len : b->tag3_len;
            if (tlen == 0) { dropped = 1; break; }
            /* computeKey groups on any key column (groupby.go:287-364):
             * dictionary columns and the plain (>256-distinct) fallback
             * (column.go:266-278) both decode to per-row values here */
            size_t tdl = 0;
            if (payload[toff] == BO_ENC_DICTIONARY)
                rc = bo_dictionary_decode(tagdata, (size_t)1 << 24, taglen,
                                          payload + toff + 1, tlen - 1, n,
                                          &tdl);
            else if (payload[toff] == BO_ENC_PLAIN)
                rc = bo_bytes_block_decode(tagdata, (size_t)1 << 24, taglen,
                                           payload + toff + 1, tlen - 1, n,
                                           &tdl);
            else
                rc = BO_ERR_BAD_TYPE;
            if (rc != BO_OK) break;
            const uint8_t *dv = dom_blobs[si];
            const int64_t *dl = dom_lens[si];
            int64_t nd = n_doms[si];
            size_t tago = 0;
            for (int64_t r = 0; r < n; r++) {
                int64_t tl = taglen[r];
                size_t my = tago;
                if (tl > 0) tago += (size_t)tl;
                if (rowgid[r] < 0) continue;
                int64_t gid = -1;
                if (tl >= 0) {
                    const uint8_t *p = dv;
                    for (int64_t g = 0; g < nd; g++) {
                        int64_t gl = dl[g] < 0 ? -1 : dl[g];
                        if (gl == tl &&
                            (tl == 0 || memcmp(p, tagdata + my, (size_t)tl) == 0)) {
                            gid = g;
                            break;
                        }
                        if (gl > 0) p += gl;
                    }
                }
                if (gid < 0) rowgid[r] = -1;
                else rowgid[r] += gid * mul;
            }
            mul *= nd;
        }
        if (rc != BO_OK) break;
        if (dropped) continue;
        if (field_vtype == BO_VT_INT64) {
            if (payload[b->col_off] == BO_ENC_PLAIN) {
                uint8_t validp[8192];
                rc = column_i64_decode_plain(i64_buf, validp,
                                             payload + b->col_off,
                                             b->col_len, n);
                if (rc == BO_OK)
                    for (int64_t q = 0; q < n; q++)
                        if (!validp[q]) rowgid[q] = -1;
            } else {
                rc = bo_column_i64_decode(i64_buf, payload + b->col_off,
                                          b->col_len, n);
            }
        } else if (field_vtype == BO_VT_FLOAT64) {
            if (payload[b->col_off] == BO_ENC_PLAIN) {
                uint8_t validf[8192];
                rc = column_f64_decode_plain(f64_buf, validf,
                                             payload + b->col_off,
                                             b->col_len, n);
                if (rc == BO_OK)
                    for (int64_t q = 0; q < n; q++)
                        if (!validf[q]) rowgid[q] = -1;
            } else {
                rc = bo_column_f64_decode(f64_buf, payload + b->col_off,
                                          b->col_len, n);
            }
        } else rc = BO_ERR_BAD_TYPE;
        if (rc != BO_OK) break;
        for (int64_t r = r0; r <= r1; r++) {
            if (rowgid[r] < 0) continue;
            bo_agg_result *o = &out[rowgid[r]];
            if (field_vtype == BO_VT_INT64) {
                int64_t v = i64_buf[r];
                o->sum_i = (int64_t)((uint64_t)o->sum_i + (uint64_t)v);
                o->count++;
                if (v < o->min_i) o->min_i = v;
                if (v > o->max_i) o->max_i = v;
            } else {
                double v = f64_buf[r];
                o->sum_f += v;
                o->count++;
                if (v < o->min_f) o->min_f = v;
                if (v > o->max_f) o->max_f = v;
            }
        }
    }
    free(ts_buf); free(i64_buf); free(f64_buf); free(tagdata); free(taglen);
    free(rowgid);
    return rc;
}

int bo_scan_agg_bytag(const uint8_t *payload, const bo_block_desc *blocks,
                      int64_t n_blocks, int field_vtype, int64_t min_ts,
                      int64_t max_ts, int slot, const uint8_t *dom_blob,
                      const int64_t *dom_lens, int64_t n_dom,
                      bo_agg_result *out) {
    const int slots[1] = {slot};
    const uint8_t *blobs[1] = {dom_blob};
    const int64_t *lens[1] = {dom_lens};
    const int64_t nd[1] = {n_dom};
    return bo_scan_agg_bytags(payload, blocks, n_blocks, field_vtype, min_ts,
                              max_ts, slots, 1, blobs, lens, nd, NULL, NULL,
                              out);
}

/* ===================== xxhash64 =====================
 * Canonical XXH64 (seed 0) — pins cespare/xxhash v2.3.0 used for
 * Entity -> SeriesID (pkg/convert/hash.go:23). */
static const uint64_t P1 = 11400714785074694791ULL;
static const uint64_t P2 = 14029467366897019727ULL;
static const uint64_t P3 = 1609587929392839161ULL;
static const uint64_t P4 = 9650029242287828579ULL;
static const uint64_t P5 = 2870177450012600261ULL;

static inline uint64_t rotl64(uint64_t x, int r) { return (x << r) | (x >> (64 - r)); }
static inline uint64_t rd64(const uint8_t *p) {
    uint64_t v;
    memcpy(&v, p, 8);
    return v; /* little-endian host */
}
static inline uint32_t rd32(const uint8_t *p) {
    uint32_t v;
    memcpy(&v, p, 4);
    return v;
}
static inline uint64_t xxh_round(uint64_t acc, uint64_t input) {
    acc += input * P2;
    acc = rotl64(acc, 31);
    acc *= P1;
    return acc;
}
static inline uint64_t xxh_merge(uint64_t acc, uint64_t val) {
    val = xxh_round(0, val);
    acc ^= val;
    acc = acc * P1 + P4;
    return acc;
}

uint64_t bo_xxhash64(const uint8_t *p, size_t len) {
    const uint8_t *end = p + len;
    uint64_t h;
    if (len >= 32) {
        uint64_t v1 = P1 + P2, v2 = P2, v3 = 0, v4 = (uint64_t)0 - P1;
        const uint8_t *limit = end - 32;
        do {
            v1 = xxh_round(v1, rd64(p)); p += 8;
            v2 = xxh_round(v2, rd64(p)); p += 8;
            v3 = xxh_round(v3, rd64(p)); p += 8;
            v4 = xxh_round(v4, rd64(p)); p += 8;
        } while (p <= limit);
        h = rotl64(v1, 1) + rotl64(v2, 7) + rotl64(v3, 12) + rotl64(v4, 18);
        h = xxh_merge(h, v1);
        h = xxh_merge(h, v2);
        h = xxh_merge(h, v3);
        h = xxh_merge(h, v4);
    } else {
        h = P5;
    }
    h += (uint64_t)len;
    while (p + 8 <= end) {
        h ^= xxh_round(0, rd64(p));
        h = rotl64(h, 27) * P1 + P4;
        p += 8;
    }
    if (p + 4 <= end) {
        h ^= (uint64_t)rd32(p) * P1;
        h = rotl64(h, 23) * P2 + P3;
        p += 4;
    }
    while (p < end) {
        h ^= (uint64_t)(*p) * P5;
        h = rotl64(h, 11) * P1;
        p++;
    }
    h ^= h >> 33;
    h *= P2;
    h ^= h >> 29;
    h *= P3;
    h ^= h >> 32;
    return h;
}
