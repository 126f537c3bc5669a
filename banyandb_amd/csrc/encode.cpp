// banyandb_amd/csrc/encode.cpp — host-side part builder (fixture writer).
//
// Mirrors the reference measure block write path so the encoded streams the
// GPU kernels consume are byte-identical to what the reference's own writer
// produces:
//   - varint/zigzag list:      pkg/encoding/int.go:81-103
//   - encode-type selection:   pkg/encoding/int_list.go:27-54,112-179
//   - delta / delta-of-delta:  pkg/encoding/delta.go:26-44,72-90
//   - decimal float -> ints:   pkg/encoding/float.go:30-66,105-180
//   - timestamps+versions:     banyand/measure/block.go:386-404
//   - int64/float64 columns:   banyand/measure/column.go:183-263
//   - dictionary tag columns:  pkg/encoding/dictionary.go:51-87 +
//                              bytes.go:45-70,205-305 (plain blocks <128B)
// Pure host code; no GPU required.  Validated byte-for-byte against the CPU
// oracle in tests/test_product_encoder.py.
#include "../../include/bydb_gpu.h"

#include <algorithm>
#include <cmath>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <string>
#include <vector>

namespace {

// ---------- varint / zigzag (int.go:81-103) ----------
inline void varint_append(std::vector<uint8_t> &dst, int64_t v) {
    if (v < 0x40 && v > -0x40) {
        int8_t c = (int8_t)v;
        dst.push_back((uint8_t)((c << 1) ^ (c >> 7)));
        return;
    }
    uint64_t u = ((uint64_t)v << 1) ^ (uint64_t)(v >> 63);
    while (u > 0x7f) {
        dst.push_back((uint8_t)(0x80u | (u & 0xff)));
        u >>= 7;
    }
    dst.push_back((uint8_t)u);
}

inline void varuint_append(std::vector<uint8_t> &dst, uint64_t u) {
    while (u > 0x7f) {
        dst.push_back((uint8_t)(0x80u | (u & 0xff)));
        u >>= 7;
    }
    dst.push_back((uint8_t)u);
}

// ---------- selection predicates (int_list.go:112-179) ----------
inline int64_t sign_bit(int64_t v) { return (int64_t)(((uint64_t)v >> 63) & 1); }

bool is_const_list(const int64_t *a, int64_t n) {
    for (int64_t i = 1; i < n; i++)
        if (a[i] != a[0]) return false;
    return n > 0;
}

void is_delta_list(const int64_t *a, int64_t n, bool *is_d, bool *is_dc) {
    *is_d = false;
    *is_dc = false;
    if (n < 2) return;
    bool ct = true;
    int64_t d1 = (int64_t)((uint64_t)a[1] - (uint64_t)a[0]);
    int64_t asc = sign_bit(d1);
    int64_t prev = a[1];
    for (int64_t i = 2; i < n; i++) {
        int64_t d = (int64_t)((uint64_t)a[i] - (uint64_t)prev);
        if ((sign_bit(d) ^ asc) == 1) return;
        if (ct && d != d1) ct = false;
        prev = a[i];
    }
    *is_d = true;
    *is_dc = ct;
}

bool is_incremental_list(const int64_t *a, int64_t n) {
    if (n < 2) return false;
    int64_t resets = 0;
    int64_t v_prev = a[0];
    if (v_prev < 0) return true;
    for (int64_t i = 1; i < n; i++) {
        int64_t v = a[i];
        if (v < v_prev) {
            if (v < 0) return false;
            if (v > (v_prev >> 3)) return false;
            resets++;
        }
        v_prev = v;
    }
    if (resets <= 2) return true;
    return resets < (n >> 3);
}

// Int64ListToBytes (int_list.go:27-54); appends the stream to dst.
uint8_t int64_list_append(std::vector<uint8_t> &dst, const int64_t *a,
                          int64_t n, int64_t *first) {
    if (is_const_list(a, n)) {
        *first = a[0];
        return BYDB_ENC_CONST;
    }
    bool is_d, is_dc;
    is_delta_list(a, n, &is_d, &is_dc);
    if (is_dc) {
        *first = a[0];
        varint_append(dst, (int64_t)((uint64_t)a[1] - (uint64_t)a[0]));
        return BYDB_ENC_DELTA_CONST;
    }
    if (is_d || is_incremental_list(a, n)) {
        // delta.go:72-90
        *first = a[0];
        int64_t d1 = (int64_t)((uint64_t)a[1] - (uint64_t)a[0]);
        varint_append(dst, d1);
        int64_t v = a[1];
        for (int64_t i = 2; i < n; i++) {
            int64_t d2 = (int64_t)((uint64_t)a[i] - (uint64_t)v - (uint64_t)d1);
            d1 = (int64_t)((uint64_t)d1 + (uint64_t)d2);
            v = (int64_t)((uint64_t)v + (uint64_t)d1);
            varint_append(dst, d2);
        }
        return BYDB_ENC_DELTA_OF_DELTA;
    }
    // delta.go:26-44
    *first = a[0];
    int64_t v = a[0];
    for (int64_t i = 1; i < n; i++) {
        int64_t d = (int64_t)((uint64_t)a[i] - (uint64_t)v);
        v = (int64_t)((uint64_t)v + (uint64_t)d);
        varint_append(dst, d);
    }
    return BYDB_ENC_DELTA;
}

// ---------- decimal float encode (float.go:30-66,105-180) ----------
const int64_t kPow10[19] = {
    1LL, 10LL, 100LL, 1000LL, 10000LL, 100000LL, 1000000LL, 10000000LL,
    100000000LL, 1000000000LL, 10000000000LL, 100000000000LL, 1000000000000LL,
    10000000000000LL, 100000000000000LL, 1000000000000000LL,
    10000000000000000LL, 100000000000000000LL, 1000000000000000000LL};

bool mul_pow10(int64_t v, int n, int64_t *out) {
    if (n < 0) return false;
    while (n >= 19) {
        if (v > INT64_MAX / kPow10[18] || v < INT64_MIN / kPow10[18]) return false;
        v *= kPow10[18];
        n -= 18;
    }
    if (n > 0) {
        if (v > INT64_MAX / kPow10[n] || v < INT64_MIN / kPow10[n]) return false;
        v *= kPow10[n];
    }
    *out = v;
    return true;
}

bool float_to_decimal(double f, int64_t *mant, int16_t *exp) {
    if (std::isnan(f) || std::isinf(f)) return false;
    if (f == 0) { *mant = 0; *exp = 0; return true; }
    if (f >= -9.2233720368547758e18 && f <= 9.2233720368547758e18) {
        int64_t u = (int64_t)f;
        if ((double)u == f) {
            int16_t e = 0;
            while (u != 0 && u % 10 == 0) { u /= 10; e++; }
            *mant = u; *exp = e; return true;
        }
    }
    char buf[64];
    int p;
    for (p = 0; p <= 17; p++) {
        snprintf(buf, sizeof buf, "%.*e", p, f);
        if (strtod(buf, nullptr) == f) break;
    }
    if (p > 17) return false;
    char *e = strchr(buf, 'e');
    if (!e) return false;
    long sci_exp = strtol(e + 1, nullptr, 10);
    char digits[32];
    int nd = 0, frac_digits = 0;
    bool negative = buf[0] == '-';
    bool seen_dot = false;
    for (const char *s = buf + (negative ? 1 : 0); s < e; s++) {
        if (*s == '.') { seen_dot = true; continue; }
        digits[nd++] = *s;
        if (seen_dot) frac_digits++;
    }
    while (nd > 1 && digits[nd - 1] == '0') { nd--; frac_digits--; }
    digits[nd] = 0;
    if (nd > 19) return false;
    long long m = strtoll(digits, nullptr, 10);
    int32_t ex = (int32_t)sci_exp - frac_digits;
    if (negative) m = -m;
    *mant = (int64_t)m;
    *exp = (int16_t)ex;
    return true;
}

bool floats_to_decimal_list(const double *src, int64_t n, int64_t *out,
                            int16_t *out_exp) {
    int16_t min_exp = INT16_MAX;
    std::vector<int16_t> exps((size_t)n);
    for (int64_t i = 0; i < n; i++) {
        int64_t d; int16_t e;
        if (!float_to_decimal(src[i], &d, &e)) return false;
        out[i] = d;
        exps[(size_t)i] = e;
        if (e < min_exp) min_exp = e;
    }
    for (int64_t i = 0; i < n; i++) {
        int diff = exps[(size_t)i] - min_exp;
        if (diff == 0) continue;
        int64_t scaled;
        if (!mul_pow10(out[i], diff, &scaled)) return false;
        out[i] = scaled;
    }
    *out_exp = min_exp;
    return true;
}

// ---------- int64 cell codec (convert/number.go:33-46) ----------
inline void cell_append(std::vector<uint8_t> &dst, int64_t v) {
    uint64_t u;
    if (v >= 0) u = (uint64_t)v | (1ULL << 63);
    else u = (1ULL << 63) - (uint64_t)(-(uint64_t)v);
    for (int i = 7; i >= 0; i--) dst.push_back((uint8_t)(u >> (8 * i)));
}

// ---------- bytes block (bytes.go:45-70,205-305) ----------
// compressBlock plain path only (<128 B); zstd framing via libzstd at need.
extern "C" {
typedef size_t (*zstd_compress_fn)(void *, size_t, const void *, size_t, int);
typedef size_t (*zstd_decompress_fn)(void *, size_t, const void *, size_t);
typedef unsigned (*zstd_iserr_fn)(size_t);
}
#include <dlfcn.h>
zstd_compress_fn p_zstd_compress = nullptr;
zstd_decompress_fn p_zstd_decompress = nullptr;
zstd_iserr_fn p_zstd_iserr = nullptr;
bool zstd_load() {
    static int loaded = -1;
    if (loaded >= 0) return loaded;
    void *h = dlopen("libzstd.so.1", RTLD_NOW | RTLD_GLOBAL);
    if (!h) h = dlopen("libzstd.so", RTLD_NOW | RTLD_GLOBAL);
    if (h) {
        p_zstd_compress = (zstd_compress_fn)dlsym(h, "ZSTD_compress");
        p_zstd_decompress = (zstd_decompress_fn)dlsym(h, "ZSTD_decompress");
        p_zstd_iserr = (zstd_iserr_fn)dlsym(h, "ZSTD_isError");
    }
    loaded = (p_zstd_compress && p_zstd_decompress && p_zstd_iserr) ? 1 : 0;
    return loaded;
}

// decompressBlock (bytes.go:306-345): [0][u8 len][bytes] plain (<128 B) or
// [1][varuint clen][zstd frame].  Returns bytes consumed in *consumed.
bool decompress_block_host(std::vector<uint8_t> &out, const uint8_t *src,
                           uint64_t src_len, uint64_t dst_hint,
                           uint64_t *consumed) {
    if (src_len < 2) return false;
    if (src[0] == 0) {
        uint64_t n = src[1];
        if (2 + n > src_len) return false;
        out.assign(src + 2, src + 2 + n);
        *consumed = 2 + n;
        return true;
    }
    if (src[0] != 1 || !zstd_load()) return false;
    uint64_t clen = 0, o = 1;
    unsigned sh = 0;
    while (o < src_len) {
        uint8_t c = src[o++];
        clen |= (uint64_t)(c & 0x7f) << sh;
        if (c < 0x80) break;
        sh += 7;
    }
    if (o + clen > src_len) return false;
    out.resize(dst_hint);
    size_t dl = p_zstd_decompress(out.data(), out.size(), src + o, clen);
    if (p_zstd_iserr(dl)) return false;
    out.resize(dl);
    *consumed = o + clen;
    return true;
}

bool compress_append(std::vector<uint8_t> &dst, const uint8_t *src, size_t n) {
    if (n < 128) {
        dst.push_back(0);
        dst.push_back((uint8_t)n);
        dst.insert(dst.end(), src, src + n);
        return true;
    }
    if (!zstd_load()) return false;
    dst.push_back(1);
    std::vector<uint8_t> tmp(n + n / 2 + 256);
    size_t clen = p_zstd_compress(tmp.data(), tmp.size(), src, n, 1);
    if (p_zstd_iserr(clen)) return false;
    varuint_append(dst, clen);
    dst.insert(dst.end(), tmp.data(), tmp.data() + clen);
    return true;
}

void u64list_append(std::vector<uint8_t> &dst, const uint64_t *a, int64_t n) {
    uint64_t nmax = 0;
    for (int64_t i = 0; i < n; i++) nmax = a[i] > nmax ? a[i] : nmax;
    if (nmax < (1ULL << 8)) {
        dst.push_back(0);
        for (int64_t i = 0; i < n; i++) dst.push_back((uint8_t)a[i]);
    } else if (nmax < (1ULL << 16)) {
        dst.push_back(1);
        for (int64_t i = 0; i < n; i++) {
            dst.push_back((uint8_t)(a[i] >> 8));
            dst.push_back((uint8_t)a[i]);
        }
    } else if (nmax < (1ULL << 32)) {
        dst.push_back(2);
        for (int64_t i = 0; i < n; i++)
            for (int b = 3; b >= 0; b--) dst.push_back((uint8_t)(a[i] >> (8 * b)));
    } else {
        dst.push_back(3);
        for (int64_t i = 0; i < n; i++)
            for (int b = 7; b >= 0; b--) dst.push_back((uint8_t)(a[i] >> (8 * b)));
    }
}

bool u64block_append(std::vector<uint8_t> &dst, const uint64_t *a, int64_t n) {
    std::vector<uint8_t> tmp;
    u64list_append(tmp, a, n);
    return compress_append(dst, tmp.data(), tmp.size());
}

// EncodeBytesBlock (bytes.go:45-70); lens[i] < 0 means nil.
bool bytes_block_append(std::vector<uint8_t> &dst, const uint8_t *data,
                        const int64_t *lens, int64_t n) {
    std::vector<uint64_t> alens((size_t)n);
    size_t total = 0;
    for (int64_t i = 0; i < n; i++) {
        if (lens[i] < 0) alens[(size_t)i] = 0;
        else { alens[(size_t)i] = (uint64_t)lens[i] + 1; total += (size_t)lens[i]; }
    }
    if (!u64block_append(dst, alens.data(), n)) return false;
    return compress_append(dst, data, total);
}

// ---------- MSB-first bit writer (writer.go:24-93) ----------
struct BitW {
    std::vector<uint8_t> &buf;
    uint8_t cache = 0;
    uint8_t avail = 8;
    explicit BitW(std::vector<uint8_t> &b) : buf(b) {}
    void byte(uint8_t b) {
        buf.push_back((uint8_t)(cache | (b >> (8 - avail))));
        cache = (uint8_t)(avail == 8 ? 0 : b << avail);
    }
    void bits(uint64_t u, int num) {
        u <<= (64 - (unsigned)num);
        for (; num >= 8; num -= 8) { byte((uint8_t)(u >> 56)); u <<= 8; }
        uint8_t rem = (uint8_t)(u >> 56);
        for (; num > 0; num--) {
            if (rem & 0x80) cache |= (uint8_t)(1u << (avail - 1));
            avail--;
            if (avail == 0) { buf.push_back(cache); cache = 0; avail = 8; }
            rem <<= 1;
        }
    }
    void flush() {
        if (avail != 8) buf.push_back(cache);
        cache = 0;
        avail = 8;
    }
};

// Dictionary.Encode (dictionary.go:51-87,158-230)
bool dictionary_append(std::vector<uint8_t> &dst, const uint8_t *data,
                       const int64_t *lens, int64_t n) {
    const uint8_t *vals[256];
    int64_t vlens[256];
    int64_t nvals = 0;
    std::vector<uint32_t> indices((size_t)n);
    const uint8_t *p = data;
    for (int64_t i = 0; i < n; i++) {
        const uint8_t *v = lens[i] < 0 ? nullptr : p;
        int64_t vl = lens[i];
        if (vl > 0) p += vl;
        int64_t found = -1;
        for (int64_t j = 0; j < nvals; j++) {
            bool eq;
            if (vlens[j] < 0 && vl < 0) eq = true;
            else if (vlens[j] < 0 || vl < 0) eq = false;
            else eq = vlens[j] == vl && memcmp(vals[j], v, (size_t)vl) == 0;
            if (eq) { found = j; break; }
        }
        if (found < 0) {
            if (nvals == 256) return false;  // dictionary.go:58 bail -> Plain
            vals[nvals] = v;
            vlens[nvals] = vl;
            found = nvals++;
        }
        indices[(size_t)i] = (uint32_t)found;
    }
    varuint_append(dst, (uint64_t)nvals);
    std::vector<uint8_t> vdata;
    for (int64_t j = 0; j < nvals; j++)
        if (vlens[j] > 0) vdata.insert(vdata.end(), vals[j], vals[j] + vlens[j]);
    if (!bytes_block_append(dst, vdata.data(), vlens, nvals)) return false;
    // RLE (dictionary.go:158-176)
    std::vector<uint32_t> rle;
    uint32_t cur = indices[0], count = 1;
    for (int64_t i = 1; i < n; i++) {
        if (indices[(size_t)i] == cur) count++;
        else { rle.push_back(cur); rle.push_back(count); cur = indices[(size_t)i]; count = 1; }
    }
    rle.push_back(cur);
    rle.push_back(count);
    // bit-packing ([32b count][8b width] MSB-first, dictionary.go:196-230)
    BitW w(dst);
    w.bits((uint64_t)rle.size(), 32);
    uint32_t maxv = 0;
    for (uint32_t v : rle) maxv = v > maxv ? v : maxv;
    int width = 1;
    if (maxv > 0) { width = 0; uint32_t m = maxv; while (m) { width++; m >>= 1; } }
    w.bits((uint64_t)width, 8);
    for (uint32_t v : rle) w.bits(v, width);
    w.flush();
    return true;
}

}  // namespace

// Host-side normalization of a reference-form Plain tag column
// (column.go:266-278 fallback: [EncodeTypePlain][bytes block], written when
// the dictionary bails at >256 distinct values, dictionary.go:58) into the
// device-parseable form [ENC_PLAIN][u32le nrows][wt][lens BE][payload] —
// the same decompress-at-part-open the reference performs (zstd.go:49)
// before row access; the GPU then evaluates the per-row predicate itself.
bool bydb_normalize_plain_tag(const uint8_t *src, uint64_t src_len,
                              uint64_t nrows, std::vector<uint8_t> &out) {
    if (src_len < 1 || src[0] != BYDB_ENC_PLAIN) return false;
    const uint8_t *p = src + 1;
    uint64_t rem = src_len - 1, used = 0;
    std::vector<uint8_t> lens;
    if (!decompress_block_host(lens, p, rem, 1 + nrows * 8, &used)) return false;
    p += used;
    rem -= used;
    if (lens.empty()) return false;
    uint8_t wt = lens[0];
    uint32_t wbytes = wt == 0 ? 1 : wt == 1 ? 2 : wt == 2 ? 4 : 8;
    if (lens.size() != 1 + (size_t)nrows * wbytes) return false;
    uint64_t total = 0;
    for (uint64_t i = 0; i < nrows; i++) {
        uint64_t ap1 = 0;
        for (uint32_t b = 0; b < wbytes; b++)
            ap1 = (ap1 << 8) | lens[1 + i * wbytes + b];
        if (ap1) total += ap1 - 1;
    }
    std::vector<uint8_t> vals;
    if (!decompress_block_host(vals, p, rem, total, &used)) return false;
    if (used != rem || vals.size() != total) return false;
    out.clear();
    out.reserve(6 + (lens.size() - 1) + vals.size());
    out.push_back(BYDB_ENC_PLAIN);
    for (int b = 0; b < 4; b++) out.push_back((uint8_t)(nrows >> (8 * b)));
    out.push_back(wt);
    out.insert(out.end(), lens.begin() + 1, lens.end());
    out.insert(out.end(), vals.begin(), vals.end());
    return true;
}

// Same normalization for dictionary tag columns whose compress_block
// sections are zstd'd (>=128 B, bytes.go:291-303): decompress the lengths
// and values sections and rewrite them with the raw framing
// [2][u32le len][bytes] the device parsers accept; the bit-packed RLE
// section is raw bits and is copied verbatim.  Returns false when the
// stream needs no normalization (both sections already plain).
bool bydb_normalize_dict_tag(const uint8_t *src, uint64_t src_len,
                             bool *needed, std::vector<uint8_t> &out) {
    *needed = false;
    if (src_len < 2 || src[0] != BYDB_ENC_DICTIONARY) return false;
    const uint8_t *p = src + 1;
    const uint8_t *end = src + src_len;
    // varuint count (int.go:152-199)
    uint64_t count = 0;
    unsigned sh = 0;
    while (p < end) {
        uint8_t c = *p++;
        count |= (uint64_t)(c & 0x7f) << sh;
        if (c < 0x80) break;
        sh += 7;
    }
    if (p >= end) return false;
    if (p[0] == 0) {
        // lengths plain: values section decides
        const uint8_t *v = p + 2 + p[1];
        if (v >= end || v[0] == 0) return true;  // fully plain: no work
    }
    *needed = true;
    std::vector<uint8_t> lens, vals;
    uint64_t used = 0;
    if (!decompress_block_host(lens, p, (uint64_t)(end - p),
                               1 + count * 8, &used))
        return false;
    p += used;
    // values payload upper bound: decompressed lens give the exact total
    if (lens.empty()) return false;
    uint8_t wt = lens[0];
    uint32_t wbytes = wt == 0 ? 1 : wt == 1 ? 2 : wt == 2 ? 4 : 8;
    if (lens.size() != 1 + (size_t)count * wbytes) return false;
    uint64_t total = 0;
    for (uint64_t i = 0; i < count; i++) {
        uint64_t ap1 = 0;
        for (uint32_t b = 0; b < wbytes; b++)
            ap1 = (ap1 << 8) | lens[1 + i * wbytes + b];
        if (ap1) total += ap1 - 1;
    }
    if (!decompress_block_host(vals, p, (uint64_t)(end - p), total, &used))
        return false;
    p += used;
    if (vals.size() != total) return false;
    out.clear();
    out.push_back(BYDB_ENC_DICTIONARY);
    {
        uint64_t u = count;
        while (u >= 0x80) { out.push_back((uint8_t)(u | 0x80)); u >>= 7; }
        out.push_back((uint8_t)u);
    }
    auto raw_section = [&out](const std::vector<uint8_t> &sec) {
        out.push_back(2);
        uint32_t l = (uint32_t)sec.size();
        for (int b = 0; b < 4; b++) out.push_back((uint8_t)(l >> (8 * b)));
        out.insert(out.end(), sec.begin(), sec.end());
    };
    raw_section(lens);
    raw_section(vals);
    out.insert(out.end(), p, end);  // bit-packed RLE verbatim
    return true;
}

// Normalize a Plain (null-bearing) FIELD column — the 8-B sign-flip cell
// block written above — into the device fold form:
// [u32le n][u32le 0][validity bitmap ceil(n/64)*8 B, bit=1 valid]
// [n x 8-B cells big-endian, null rows zeroed].  8-B aligned throughout
// so the kernel loads cells as u64.
bool bydb_normalize_plain_field(const uint8_t *src, uint64_t src_len,
                                uint64_t nrows, std::vector<uint8_t> &out) {
    if (src_len < 1 || src[0] != BYDB_ENC_PLAIN) return false;
    const uint8_t *p = src + 1;
    uint64_t rem = src_len - 1, used = 0;
    std::vector<uint8_t> lens;
    if (!decompress_block_host(lens, p, rem, 1 + nrows * 8, &used)) return false;
    p += used;
    rem -= used;
    if (lens.empty()) return false;
    uint8_t wt = lens[0];
    uint32_t wbytes = wt == 0 ? 1 : wt == 1 ? 2 : wt == 2 ? 4 : 8;
    if (lens.size() != 1 + (size_t)nrows * wbytes) return false;
    uint64_t total = 0;
    for (uint64_t i = 0; i < nrows; i++) {
        uint64_t ap1 = 0;
        for (uint32_t b = 0; b < wbytes; b++)
            ap1 = (ap1 << 8) | lens[1 + i * wbytes + b];
        if (ap1 == 0) continue;
        if (ap1 != 9) return false;  // cells are exactly 8 bytes
        total += 8;
    }
    std::vector<uint8_t> vals;
    if (!decompress_block_host(vals, p, rem, total, &used)) return false;
    if (used != rem || vals.size() != total) return false;
    uint64_t nw = (nrows + 63) / 64;
    out.assign(8 + nw * 8 + nrows * 8, 0);
    out[0] = (uint8_t)nrows;
    out[1] = (uint8_t)(nrows >> 8);
    out[2] = (uint8_t)(nrows >> 16);
    out[3] = (uint8_t)(nrows >> 24);
    uint8_t *bm = out.data() + 8;
    uint8_t *cells = out.data() + 8 + nw * 8;
    uint64_t voff = 0;
    for (uint64_t i = 0; i < nrows; i++) {
        uint64_t ap1 = 0;
        for (uint32_t b = 0; b < wbytes; b++)
            ap1 = (ap1 << 8) | lens[1 + i * wbytes + b];
        if (ap1 == 0) continue;
        bm[(i >> 3)] |= (uint8_t)(1u << (i & 7));
        memcpy(cells + i * 8, vals.data() + voff, 8);
        voff += 8;
    }
    return true;
}

// ===================== part builder =====================
struct bydb_part_builder {
    std::vector<uint8_t> payload;
    std::vector<bydb_block_desc> blocks;
    uint64_t base_off = 0;  // absolute offset of payload[0] within the part
    std::string err;
    // per-slot tag tables: the generators attach tag_table[slot][series %
    // table_size] as a constant (entity) tag column to every block
    std::vector<std::vector<uint8_t>> tag_table[3];
    // scratch
    std::vector<int64_t> scratch_i64;
    std::vector<double> scratch_f64;
    std::vector<int64_t> scratch_ts, scratch_ver;
    // duplicate-(sid,ts) fold bookkeeping for the LAST added block
    // (part.go:192-198): when rows were dropped, last_keep holds the
    // surviving source indices so set_block_tag can fold its per-row tag
    // values the same way (the reference's dps.skip drops the
    // tagFamilies row with the datapoint)
    std::vector<int64_t> last_keep;
    int64_t last_orig_n = 0;
};

extern "C" bydb_part_builder *bydb_part_builder_create(void) {
    return new bydb_part_builder();
}

extern "C" void bydb_part_builder_destroy(bydb_part_builder *b) { delete b; }

extern "C" const char *bydb_part_builder_error(bydb_part_builder *b) {
    return b->err.c_str();
}

static int add_block_common(bydb_part_builder *b, uint64_t series_id,
                            const int64_t *ts, const int64_t *versions,
                            const int64_t *field_ints, int16_t exp,
                            uint8_t vtype, int64_t n, uint32_t group_code,
                            const uint8_t *valid = nullptr) {
    if (n < 1 || n > 8192) {  // measure.go:41-46
        b->err = "block row count out of range";
        return BYDB_ERR_BAD_ARG;
    }
    bydb_block_desc d;
    memset(&d, 0, sizeof d);
    d.series_id = series_id;
    d.count = (uint32_t)n;
    d.group_code = group_code;
    d.field_vtype = vtype;
    d.exp = exp;
    // timestamps payload (block.go:386-404): ts stream ++ version stream
    int64_t first;
    size_t start = b->payload.size();
    uint8_t tenc = int64_list_append(b->payload, ts, n, &first);
    uint8_t wv = 0;
    switch (tenc) {  // encoding.GetVersionType (encoding.go:100-114)
    case BYDB_ENC_CONST: wv = BYDB_ENC_CONST_WV; break;
    case BYDB_ENC_DELTA_CONST: wv = BYDB_ENC_DELTA_CONST_WV; break;
    case BYDB_ENC_DELTA: wv = BYDB_ENC_DELTA_WV; break;
    case BYDB_ENC_DELTA_OF_DELTA: wv = BYDB_ENC_DELTA_OF_DELTA_WV; break;
    default:
        b->err = "unexpected timestamp encode type";
        return BYDB_ERR_BAD_DATA;
    }
    d.ts_enc_with_version = wv;
    d.ts_min = first;
    d.ts_max = ts[n - 1];
    d.ts_off = b->base_off + start;
    d.ts_len = b->payload.size() - start;
    size_t vstart = b->payload.size();
    int64_t vfirst;
    d.version_enc = int64_list_append(b->payload, versions, n, &vfirst);
    d.version_first = vfirst;
    (void)vstart;
    // field column stream (header fields parsed into the desc; the payload
    // keeps only the varint stream, header-free — kernels read streams)
    size_t fstart = b->payload.size();
    if (valid) {
        // null-bearing column: the int-list encodings cannot represent
        // nulls, so the writer stores Plain — a bytes block of 8-B
        // sign-flip cells (convert/number.go:33-46), nil rows zero-length
        // (column.go:214-233 + :266-278 fallback).  The fold skips nulls
        // (vectorized/measure/aggregation.go:310 null check).
        b->payload.push_back(BYDB_ENC_PLAIN);
        std::vector<uint8_t> cells;
        std::vector<int64_t> clens((size_t)n);
        cells.reserve((size_t)n * 8);
        for (int64_t i = 0; i < n; i++) {
            if (valid[i]) {
                cell_append(cells, field_ints[i]);
                clens[(size_t)i] = 8;
            } else {
                clens[(size_t)i] = -1;
            }
        }
        if (!bytes_block_append(b->payload, cells.data(), clens.data(), n)) {
            b->err = "cell block encode failed (zstd unavailable?)";
            b->payload.resize(start);
            return BYDB_ERR_BAD_DATA;
        }
        d.field_enc = BYDB_ENC_PLAIN;
        d.field_first = 0;
    } else {
        int64_t ffirst;
        uint8_t fenc = int64_list_append(b->payload, field_ints, n, &ffirst);
        d.field_enc = fenc;
        d.field_first = ffirst;
    }
    d.field_off = b->base_off + fstart;
    d.field_len = b->payload.size() - fstart;
    b->blocks.push_back(d);
    b->last_keep.clear();
    b->last_orig_n = n;
    return BYDB_OK;
}

// Duplicate-(sid,ts) fold at part build, exactly as the reference's
// mustInitFromDataPoints does (part.go:178 sort + :192-198 skip loop with
// datapoints.go:189-197 Less): rows order by (ts asc, version desc) and
// only the FIRST row of each timestamp — the highest version — survives.
// Returns false when the input is already strictly-ascending unique (the
// common case; no work).  keep holds the surviving source indices in
// write order.  (Not reproduced: the reference's tsPrev==0 initialization
// would also drop a first row whose timestamp is exactly 0 — a
// 1970-epoch corner with no real inputs.)
static bool dedup_block_rows(const int64_t *ts, const int64_t *versions,
                             int64_t n, std::vector<int64_t> &keep) {
    bool sorted_unique = true;
    for (int64_t i = 1; i < n; i++)
        if (ts[i] <= ts[i - 1]) { sorted_unique = false; break; }
    if (sorted_unique) return false;
    keep.resize((size_t)n);
    for (int64_t i = 0; i < n; i++) keep[(size_t)i] = i;
    std::stable_sort(keep.begin(), keep.end(), [&](int64_t a, int64_t c) {
        if (ts[a] != ts[c]) return ts[a] < ts[c];
        return versions[a] > versions[c];
    });
    size_t w = 0;
    int64_t prev_ts = 0;
    for (size_t i = 0; i < keep.size(); i++) {
        int64_t src = keep[i];
        if (w == 0 || ts[src] != prev_ts) {
            keep[w++] = src;
            prev_ts = ts[src];
        }
    }
    keep.resize(w);
    return true;
}

// Null-bearing float64 column: 8-B big-endian IEEE-754 cells
// (convert/number.go:128-132 Float64ToBytes), nil rows zero-length.
extern "C" int bydb_part_builder_add_block_f64_nullable(
    bydb_part_builder *b, uint64_t series_id, const int64_t *ts,
    const int64_t *versions, const double *vals, const uint8_t *valid,
    int64_t n, uint32_t group_code) {
    if (!valid) {
        b->err = "valid mask required";
        return BYDB_ERR_BAD_ARG;
    }
    // reuse the common path with a pre-built cell list: encode the cells
    // here and hand add_block_common a sentinel that writes them
    if (n < 1 || n > 8192) {
        b->err = "block row count out of range";
        return BYDB_ERR_BAD_ARG;
    }
    {
        std::vector<int64_t> keep;
        if (dedup_block_rows(ts, versions, n, keep)) {
            std::vector<int64_t> ts2, ver2;
            std::vector<double> vv2;
            std::vector<uint8_t> va2;
            for (int64_t src : keep) {
                ts2.push_back(ts[src]);
                ver2.push_back(versions[src]);
                vv2.push_back(vals[src]);
                va2.push_back(valid[src]);
            }
            int rc = bydb_part_builder_add_block_f64_nullable(
                b, series_id, ts2.data(), ver2.data(), vv2.data(),
                va2.data(), (int64_t)keep.size(), group_code);
            if (rc == BYDB_OK) {
                b->last_keep = keep;
                b->last_orig_n = n;
            }
            return rc;
        }
    }
    // ts/version/desc handling matches add_block_common; the field stream
    // is [ENC_PLAIN][bytes block of float cells]
    b->scratch_i64.resize((size_t)n);
    for (int64_t i = 0; i < n; i++) b->scratch_i64[(size_t)i] = 0;
    int rc = add_block_common(b, series_id, ts, versions,
                              b->scratch_i64.data(), 0, BYDB_VT_FLOAT64, n,
                              group_code, valid);
    if (rc != BYDB_OK) return rc;
    // rewrite the just-written Plain cell payload with float bits: the
    // common path wrote i64 sign-flip cells of zeros; redo the field
    // stream properly
    bydb_block_desc &d = b->blocks.back();
    size_t fstart = (size_t)(d.field_off - b->base_off);
    b->payload.resize(fstart);
    b->payload.push_back(BYDB_ENC_PLAIN);
    std::vector<uint8_t> cells;
    std::vector<int64_t> clens((size_t)n);
    cells.reserve((size_t)n * 8);
    for (int64_t i = 0; i < n; i++) {
        if (valid[i]) {
            uint64_t bits;
            memcpy(&bits, &vals[i], 8);
            for (int k = 7; k >= 0; k--)
                cells.push_back((uint8_t)(bits >> (8 * k)));
            clens[(size_t)i] = 8;
        } else {
            clens[(size_t)i] = -1;
        }
    }
    if (!bytes_block_append(b->payload, cells.data(), clens.data(), n)) {
        b->err = "cell block encode failed (zstd unavailable?)";
        b->blocks.pop_back();
        b->payload.resize(fstart);
        return BYDB_ERR_BAD_DATA;
    }
    d.field_len = (b->base_off + b->payload.size()) - d.field_off;
    d.exp = 0;
    return BYDB_OK;
}

// Null-bearing int64 column: valid[i] == 0 marks row i null.
extern "C" int bydb_part_builder_add_block_i64_nullable(
    bydb_part_builder *b, uint64_t series_id, const int64_t *ts,
    const int64_t *versions, const int64_t *vals, const uint8_t *valid,
    int64_t n, uint32_t group_code) {
    if (!valid) {
        b->err = "valid mask required";
        return BYDB_ERR_BAD_ARG;
    }
    {
        std::vector<int64_t> keep;
        if (dedup_block_rows(ts, versions, n, keep)) {
            std::vector<int64_t> ts2, ver2, vv2;
            std::vector<uint8_t> va2;
            for (int64_t src : keep) {
                ts2.push_back(ts[src]);
                ver2.push_back(versions[src]);
                vv2.push_back(vals[src]);
                va2.push_back(valid[src]);
            }
            int rc = bydb_part_builder_add_block_i64_nullable(
                b, series_id, ts2.data(), ver2.data(), vv2.data(),
                va2.data(), (int64_t)keep.size(), group_code);
            if (rc == BYDB_OK) {
                b->last_keep = keep;
                b->last_orig_n = n;
            }
            return rc;
        }
    }
    return add_block_common(b, series_id, ts, versions, vals, 0,
                            BYDB_VT_INT64, n, group_code, valid);
}

extern "C" int bydb_part_builder_add_block_i64(bydb_part_builder *b,
                                               uint64_t series_id,
                                               const int64_t *ts,
                                               const int64_t *versions,
                                               const int64_t *vals, int64_t n,
                                               uint32_t group_code) {
    {
        std::vector<int64_t> keep;
        if (dedup_block_rows(ts, versions, n, keep)) {
            std::vector<int64_t> ts2, ver2, vv2;
            for (int64_t src : keep) {
                ts2.push_back(ts[src]);
                ver2.push_back(versions[src]);
                vv2.push_back(vals[src]);
            }
            int rc = bydb_part_builder_add_block_i64(
                b, series_id, ts2.data(), ver2.data(), vv2.data(),
                (int64_t)keep.size(), group_code);
            if (rc == BYDB_OK) {
                b->last_keep = keep;
                b->last_orig_n = n;
            }
            return rc;
        }
    }
    return add_block_common(b, series_id, ts, versions, vals, 0, BYDB_VT_INT64,
                            n, group_code);
}

extern "C" int bydb_part_builder_add_block_f64(bydb_part_builder *b,
                                               uint64_t series_id,
                                               const int64_t *ts,
                                               const int64_t *versions,
                                               const double *vals, int64_t n,
                                               uint32_t group_code) {
    {
        std::vector<int64_t> keep;
        if (dedup_block_rows(ts, versions, n, keep)) {
            std::vector<int64_t> ts2, ver2;
            std::vector<double> vv2;
            for (int64_t src : keep) {
                ts2.push_back(ts[src]);
                ver2.push_back(versions[src]);
                vv2.push_back(vals[src]);
            }
            int rc = bydb_part_builder_add_block_f64(
                b, series_id, ts2.data(), ver2.data(), vv2.data(),
                (int64_t)keep.size(), group_code);
            if (rc == BYDB_OK) {
                b->last_keep = keep;
                b->last_orig_n = n;
            }
            return rc;
        }
    }
    b->scratch_i64.resize((size_t)n);
    int16_t exp;
    if (!floats_to_decimal_list(vals, n, b->scratch_i64.data(), &exp)) {
        b->err = "cannot encode float64 losslessly as decimal int";
        return BYDB_ERR_BAD_DATA;
    }
    return add_block_common(b, series_id, ts, versions, b->scratch_i64.data(),
                            exp, BYDB_VT_FLOAT64, n, group_code);
}

extern "C" int bydb_part_builder_set_block_tag(bydb_part_builder *b,
                                               const uint8_t *data,
                                               const int64_t *lens, int64_t n) {
    if (b->blocks.empty()) {
        b->err = "no block to attach tag to";
        return BYDB_ERR_STATE;
    }
    bydb_block_desc &d = b->blocks.back();
    if ((int64_t)d.count != n) {
        // If the last block folded duplicate-(sid,ts) rows (part.go:
        // 192-198), fold this per-row tag list the same way: the
        // reference's dps.skip drops the tagFamilies row with the
        // datapoint, so the caller may pass the ORIGINAL row list.
        if (n == b->last_orig_n &&
            (int64_t)b->last_keep.size() == (int64_t)d.count) {
            std::vector<int64_t> offs((size_t)n);
            int64_t o = 0;
            for (int64_t i = 0; i < n; i++) {
                offs[(size_t)i] = o;
                if (lens[i] > 0) o += lens[i];
            }
            std::vector<uint8_t> data2;
            std::vector<int64_t> lens2;
            for (int64_t src : b->last_keep) {
                lens2.push_back(lens[src]);
                if (lens[src] > 0)
                    data2.insert(data2.end(), data + offs[(size_t)src],
                                 data + offs[(size_t)src] + lens[src]);
            }
            static const uint8_t kEmpty = 0;
            return bydb_part_builder_set_block_tag(
                b, data2.empty() ? &kEmpty : data2.data(), lens2.data(),
                (int64_t)lens2.size());
        }
        b->err = "tag row count mismatch";
        return BYDB_ERR_BAD_ARG;
    }
    int slot = d.tag_len == 0 ? 0 : d.tag2_len == 0 ? 1 : d.tag3_len == 0 ? 2 : 3;
    if (slot == 3) {
        b->err = "at most 3 tag columns per block";
        return BYDB_ERR_BAD_ARG;
    }
    size_t start = b->payload.size();
    // column.encodeDefault (column.go:266-278): try dictionary, else Plain
    b->payload.push_back(BYDB_ENC_DICTIONARY);
    if (!dictionary_append(b->payload, data, lens, n)) {
        b->payload.resize(start);
        b->payload.push_back(BYDB_ENC_PLAIN);
        if (!bytes_block_append(b->payload, data, lens, n)) {
            b->err = "bytes block encode failed (zstd unavailable?)";
            b->payload.resize(start);
            return BYDB_ERR_BAD_DATA;
        }
    }
    uint64_t off = b->base_off + start;
    uint64_t len = b->payload.size() - start;
    if (slot == 0) { d.tag_off = off; d.tag_len = len; }
    else if (slot == 1) { d.tag2_off = off; d.tag2_len = len; }
    else { d.tag3_off = off; d.tag3_len = len; }
    return BYDB_OK;
}

extern "C" uint64_t bydb_part_builder_payload_len(bydb_part_builder *b) {
    return b->payload.size();
}
extern "C" const uint8_t *bydb_part_builder_payload(bydb_part_builder *b) {
    return b->payload.data();
}
extern "C" int64_t bydb_part_builder_n_blocks(bydb_part_builder *b) {
    return (int64_t)b->blocks.size();
}
extern "C" const bydb_block_desc *bydb_part_builder_blocks(bydb_part_builder *b) {
    return b->blocks.data();
}
// raw helpers for the part reader (part_io.cpp): append payload bytes /
// a pre-built descriptor with offsets relative to the whole part
extern "C" int bydb_part_builder_append_raw(bydb_part_builder *b,
                                            const uint8_t *data, uint64_t len) {
    b->payload.insert(b->payload.end(), data, data + len);
    return BYDB_OK;
}

extern "C" int bydb_part_builder_append_desc(bydb_part_builder *b,
                                             const bydb_block_desc *d) {
    b->blocks.push_back(*d);
    return BYDB_OK;
}

extern "C" int bydb_part_builder_drain(bydb_part_builder *b) {
    b->base_off += b->payload.size();
    b->payload.clear();
    b->blocks.clear();
    return BYDB_OK;
}

extern "C" int bydb_part_builder_set_tag_table(bydb_part_builder *b, int slot,
                                               const uint8_t *data,
                                               const int64_t *lens,
                                               int64_t n_values) {
    if (slot < 0 || slot > 2) return BYDB_ERR_BAD_ARG;
    b->tag_table[slot].clear();
    const uint8_t *p = data;
    for (int64_t i = 0; i < n_values; i++) {
        if (lens[i] < 0) { b->tag_table[slot].emplace_back(); continue; }
        b->tag_table[slot].emplace_back(p, p + lens[i]);
        p += lens[i];
    }
    return BYDB_OK;
}

// attach each configured tag table's entry for this series to the block
// just added (entity tags: constant per series)
static int attach_auto_tags(bydb_part_builder *b, uint64_t series_index,
                            int64_t n) {
    for (int slot = 0; slot < 3; slot++) {
        if (b->tag_table[slot].empty()) continue;
        const auto &v = b->tag_table[slot][series_index % b->tag_table[slot].size()];
        std::vector<uint8_t> data;
        std::vector<int64_t> lens((size_t)n, (int64_t)v.size());
        data.reserve(v.size() * (size_t)n);
        for (int64_t i = 0; i < n; i++) data.insert(data.end(), v.begin(), v.end());
        int rc = bydb_part_builder_set_block_tag(b, data.data(), lens.data(), n);
        if (rc != BYDB_OK) return rc;
    }
    return BYDB_OK;
}

// ===================== synthetic generator =====================
// splitmix64 — deterministic synthetic noise stream (seed spec: DESIGN.md).
static inline uint64_t splitmix64(uint64_t *state) {
    uint64_t z = (*state += 0x9E3779B97F4A7C15ULL);
    z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ULL;
    z = (z ^ (z >> 27)) * 0x94D049BB133111EBULL;
    return z ^ (z >> 31);
}

extern "C" int bydb_gen_series_i64(bydb_part_builder *b, uint64_t series_index,
                                   int64_t n_dp, int64_t t0, int64_t stride_ns,
                                   int64_t base, int64_t ramp, uint64_t seed,
                                   uint32_t group_code) {
    uint64_t st = seed ^ (series_index * 0x9E3779B97F4A7C15ULL + 1);
    uint64_t sid = series_index + 1;  // series ids are opaque; stable mapping
    int64_t done = 0;
    b->scratch_i64.reserve(8192);
    b->scratch_ts.reserve(8192);
    b->scratch_ver.reserve(8192);
    while (done < n_dp) {
        int64_t n = n_dp - done > 8192 ? 8192 : n_dp - done;  // part.go:201-202
        b->scratch_i64.resize((size_t)n);
        b->scratch_ts.resize((size_t)n);
        b->scratch_ver.resize((size_t)n);
        for (int64_t i = 0; i < n; i++) {
            int64_t gi = done + i;
            int64_t noise = (int64_t)(splitmix64(&st) % 7) - 3;
            b->scratch_i64[(size_t)i] = base + gi * ramp + noise;
            b->scratch_ts[(size_t)i] = t0 + gi * stride_ns;
            b->scratch_ver[(size_t)i] = 1;
        }
        int rc = bydb_part_builder_add_block_i64(
            b, sid, b->scratch_ts.data(), b->scratch_ver.data(),
            b->scratch_i64.data(), n, group_code);
        if (rc == BYDB_OK) rc = attach_auto_tags(b, series_index, n);
        if (rc != BYDB_OK) return rc;
        done += n;
    }
    return BYDB_OK;
}

extern "C" int bydb_gen_series_f64(bydb_part_builder *b, uint64_t series_index,
                                   int64_t n_dp, int64_t t0, int64_t stride_ns,
                                   double base, double ramp, uint64_t seed,
                                   uint32_t group_code) {
    uint64_t st = seed ^ (series_index * 0x9E3779B97F4A7C15ULL + 1);
    uint64_t sid = series_index + 1;
    int64_t base_cents = (int64_t)llround(base * 100.0);
    int64_t ramp_cents = (int64_t)llround(ramp * 100.0);
    int64_t done = 0;
    while (done < n_dp) {
        int64_t n = n_dp - done > 8192 ? 8192 : n_dp - done;
        b->scratch_f64.resize((size_t)n);
        b->scratch_ts.resize((size_t)n);
        b->scratch_ver.resize((size_t)n);
        b->scratch_i64.resize((size_t)n);
        bool fast_ok = true;
        int16_t min_exp = INT16_MAX;
        std::vector<int16_t> exps((size_t)n);
        for (int64_t i = 0; i < n; i++) {
            int64_t gi = done + i;
            int64_t noise = (int64_t)(splitmix64(&st) % 601) - 300;  // cents
            int64_t cents = base_cents + gi * ramp_cents + noise;
            b->scratch_ts[(size_t)i] = t0 + gi * stride_ns;
            b->scratch_ver[(size_t)i] = 1;
            b->scratch_f64[(size_t)i] = (double)cents / 100.0;
            if (cents > 1000000000000LL || cents < -1000000000000LL)
                fast_ok = false;
            // floatToDecimal(cents/100.0) == (cents stripped of trailing
            // zeros, -2 + strips) for |cents| <= 1e12 (shortest-repr
            // uniqueness at <= 13 significant digits)
            int64_t d = cents;
            int16_t e = -2;
            if (d == 0) e = 0;
            else while (d % 10 == 0) { d /= 10; e++; }
            b->scratch_i64[(size_t)i] = d;
            exps[(size_t)i] = e;
            if (e < min_exp) min_exp = e;
        }
        int rc;
        if (fast_ok) {
            for (int64_t i = 0; i < n; i++) {
                int diff = exps[(size_t)i] - min_exp;
                while (diff-- > 0) b->scratch_i64[(size_t)i] *= 10;
            }
            rc = add_block_common(b, sid, b->scratch_ts.data(),
                                  b->scratch_ver.data(), b->scratch_i64.data(),
                                  min_exp, BYDB_VT_FLOAT64, n, group_code);
        } else {
            rc = bydb_part_builder_add_block_f64(
                b, sid, b->scratch_ts.data(), b->scratch_ver.data(),
                b->scratch_f64.data(), n, group_code);
        }
        if (rc == BYDB_OK) rc = attach_auto_tags(b, series_index, n);
        if (rc != BYDB_OK) return rc;
        done += n;
    }
    return BYDB_OK;
}

// ===================== threaded bulk generator =====================
// Generates [first_index, first_index+n_series) series in parallel with
// per-thread private builders, then splices them into b preserving series
// order.  base(series) = series_index * base_step; group_code =
// series_index % group_mod (group_mod 0 -> all group 0).
#include <thread>

static void splice_builder(bydb_part_builder *dst, bydb_part_builder *src) {
    uint64_t shift = dst->base_off + dst->payload.size();
    dst->payload.insert(dst->payload.end(), src->payload.begin(),
                        src->payload.end());
    for (bydb_block_desc d : src->blocks) {
        d.ts_off += shift;
        d.field_off += shift;
        if (d.tag_len) d.tag_off += shift;
        if (d.tag2_len) d.tag2_off += shift;
        if (d.tag3_len) d.tag3_off += shift;
        dst->blocks.push_back(d);
    }
}

template <typename GenFn>
static int bulk_gen(bydb_part_builder *b, uint64_t first_index,
                    int64_t n_series, int n_threads, GenFn gen) {
    if (n_threads < 1) n_threads = 1;
    if (n_threads > 64) n_threads = 64;
    if ((int64_t)n_threads > n_series) n_threads = (int)n_series;
    std::vector<bydb_part_builder> locals((size_t)n_threads);
    for (auto &lb : locals)
        for (int sl = 0; sl < 3; sl++) lb.tag_table[sl] = b->tag_table[sl];
    std::vector<int> rcs((size_t)n_threads, BYDB_OK);
    std::vector<std::thread> threads;
    int64_t per = (n_series + n_threads - 1) / n_threads;
    for (int t = 0; t < n_threads; t++) {
        threads.emplace_back([&, t]() {
            int64_t lo = t * per, hi = lo + per < n_series ? lo + per : n_series;
            for (int64_t si = lo; si < hi; si++) {
                int rc = gen(&locals[(size_t)t], first_index + (uint64_t)si);
                if (rc != BYDB_OK) { rcs[(size_t)t] = rc; return; }
            }
        });
    }
    for (auto &th : threads) th.join();
    for (int t = 0; t < n_threads; t++) {
        if (rcs[(size_t)t] != BYDB_OK) {
            b->err = locals[(size_t)t].err;
            return rcs[(size_t)t];
        }
        splice_builder(b, &locals[(size_t)t]);
    }
    return BYDB_OK;
}

extern "C" int bydb_gen_series_bulk_i64(bydb_part_builder *b,
                                        uint64_t first_index, int64_t n_series,
                                        int64_t n_dp, int64_t t0,
                                        int64_t stride_ns, int64_t base_step,
                                        int64_t ramp, uint64_t seed,
                                        uint32_t group_mod, int n_threads) {
    return bulk_gen(b, first_index, n_series, n_threads,
                    [&](bydb_part_builder *lb, uint64_t si) {
                        uint32_t gc = group_mod ? (uint32_t)(si % group_mod) : 0;
                        return bydb_gen_series_i64(lb, si, n_dp, t0, stride_ns,
                                                   (int64_t)si * base_step, ramp,
                                                   seed, gc);
                    });
}

extern "C" int bydb_gen_series_bulk_f64(bydb_part_builder *b,
                                        uint64_t first_index, int64_t n_series,
                                        int64_t n_dp, int64_t t0,
                                        int64_t stride_ns, double base_step,
                                        double ramp, uint64_t seed,
                                        uint32_t group_mod, int n_threads) {
    return bulk_gen(b, first_index, n_series, n_threads,
                    [&](bydb_part_builder *lb, uint64_t si) {
                        uint32_t gc = group_mod ? (uint32_t)(si % group_mod) : 0;
                        return bydb_gen_series_f64(lb, si, n_dp, t0, stride_ns,
                                                   (double)si * base_step, ramp,
                                                   seed, gc);
                    });
}
