// banyandb_amd/csrc/frame.cpp — raw columnar wire-frame codec (egress).
//
// Restates the reference's shared vec frame layout byte-for-byte
// (pkg/query/vectorized/frame/encode.go:42-170, decode.go; measure wire
// numbering from frame/golden_test.go:38-121):
//   magic {0x00,'V','F','R'} + version 3 + uvarint(nrows) + uvarint(ncols),
//   then per column: [role byte][type byte][uvarint len + name]
//   [uvarint len + family][validity bitmap ceil(N/8) B, LE bit-packed,
//   1 = null][data: int64/float64 = N x 8 B little-endian; string/bytes =
//   uvarint(len)+bytes per row (null -> len 0); TagValue/FieldValue =
//   uvarint(len)+proto bytes per cell].
// Wire roles: ts=1 version=2 series=3 shard=4 tag=5 field=6.
// Wire types: i64=1 f64=2 str=3 bytes=4 TagValue=5 FieldValue=6.
// This is the Map->Reduce partial transport / egress slot (SURVEY section
// 2 "Columnar wire frame"; raw_emit.go) — GPU results are emitted into it
// without host re-materialisation of cells.
#include "../../include/bydb_gpu.h"

#include <cstring>
#include <string>
#include <vector>

namespace {

const uint8_t kMagic[4] = {0x00, 'V', 'F', 'R'};
const uint8_t kWireVersion = 3;

void uvarint_append(std::vector<uint8_t> &b, uint64_t u) {
    while (u >= 0x80) {
        b.push_back((uint8_t)(u | 0x80));
        u >>= 7;
    }
    b.push_back((uint8_t)u);
}

void le64_append(std::vector<uint8_t> &b, uint64_t v) {
    for (int i = 0; i < 8; i++) b.push_back((uint8_t)(v >> (8 * i)));
}

struct Col {
    uint8_t role, type;
    std::string name, family;
    std::vector<int64_t> i64;      // also float64 bit patterns
    std::vector<uint8_t> var_data; // concatenated var-width cells
    std::vector<int64_t> var_lens; // -1 = null
    std::vector<uint8_t> nulls;    // per-row 0/1 for fixed-width
};

}  // namespace

struct bydb_frame_builder {
    uint64_t nrows;
    std::vector<Col> cols;
    std::vector<uint8_t> out;
    std::string err;
};

extern "C" bydb_frame_builder *bydb_frame_builder_create(uint64_t nrows) {
    auto *b = new bydb_frame_builder();
    b->nrows = nrows;
    return b;
}

extern "C" void bydb_frame_builder_destroy(bydb_frame_builder *b) { delete b; }
extern "C" const char *bydb_frame_builder_error(bydb_frame_builder *b) {
    return b->err.c_str();
}

static Col *add_col(bydb_frame_builder *b, uint8_t role, uint8_t type,
                    const char *name, const char *family) {
    b->cols.emplace_back();
    Col *c = &b->cols.back();
    c->role = role;
    c->type = type;
    c->name = name ? name : "";
    c->family = family ? family : "";
    return c;
}

extern "C" int bydb_frame_add_i64(bydb_frame_builder *b, uint8_t role,
                                  const char *name, const char *family,
                                  const int64_t *vals, const uint8_t *nulls) {
    Col *c = add_col(b, role, 1, name, family);
    c->i64.assign(vals, vals + b->nrows);
    if (nulls) c->nulls.assign(nulls, nulls + b->nrows);
    else c->nulls.assign(b->nrows, 0);
    return BYDB_OK;
}

extern "C" int bydb_frame_add_f64(bydb_frame_builder *b, uint8_t role,
                                  const char *name, const char *family,
                                  const double *vals, const uint8_t *nulls) {
    Col *c = add_col(b, role, 2, name, family);
    c->i64.resize(b->nrows);
    memcpy(c->i64.data(), vals, b->nrows * 8);
    c->nulls.assign(b->nrows, 0);
    if (nulls) c->nulls.assign(nulls, nulls + b->nrows);
    return BYDB_OK;
}

// var-width column: data = concatenated non-null cells, lens[i] (-1 = null)
static int add_var(bydb_frame_builder *b, uint8_t role, uint8_t type,
                   const char *name, const char *family, const uint8_t *data,
                   const int64_t *lens) {
    Col *c = add_col(b, role, type, name, family);
    c->var_lens.assign(lens, lens + b->nrows);
    size_t total = 0;
    for (uint64_t i = 0; i < b->nrows; i++)
        if (lens[i] > 0) total += (size_t)lens[i];
    c->var_data.assign(data, data + total);
    return BYDB_OK;
}

extern "C" int bydb_frame_add_str(bydb_frame_builder *b, uint8_t role,
                                  const char *name, const char *family,
                                  const uint8_t *data, const int64_t *lens) {
    return add_var(b, role, 3, name, family, data, lens);
}

extern "C" int bydb_frame_add_bytes(bydb_frame_builder *b, uint8_t role,
                                    const char *name, const char *family,
                                    const uint8_t *data, const int64_t *lens) {
    return add_var(b, role, 4, name, family, data, lens);
}

// TagValue / FieldValue cells: minimal proto emit of the scalar variants
// (model/v1/common.proto: TagValue oneof str=2, int=4, binary_data=6;
// FieldValue oneof str=2, int=3, binary_data=4, float=5).
static void proto_varint(std::vector<uint8_t> &b, uint64_t u) { uvarint_append(b, u); }

static void tagvalue_str_cell(std::vector<uint8_t> &out, const uint8_t *s,
                              size_t n, bool field_value) {
    // inner Str { string value = 1; }
    std::vector<uint8_t> inner;
    inner.push_back(0x0A);  // field 1, LEN
    proto_varint(inner, n);
    inner.insert(inner.end(), s, s + n);
    out.push_back(field_value ? 0x12 : 0x12);  // str = field 2 both messages
    proto_varint(out, inner.size());
    out.insert(out.end(), inner.begin(), inner.end());
}

static void tagvalue_int_cell(std::vector<uint8_t> &out, int64_t v,
                              bool field_value) {
    // inner Int { int64 value = 1; } — proto int64 = two's-complement varint
    std::vector<uint8_t> inner;
    inner.push_back(0x08);  // field 1, VARINT
    proto_varint(inner, (uint64_t)v);
    out.push_back(field_value ? 0x1A : 0x22);  // FieldValue int=3, TagValue int=4
    proto_varint(out, inner.size());
    out.insert(out.end(), inner.begin(), inner.end());
}

// str cells for TagValue (type 5) / FieldValue (type 6) columns.
extern "C" int bydb_frame_add_tagvalue_str(bydb_frame_builder *b, uint8_t role,
                                           const char *name, const char *family,
                                           const uint8_t *data,
                                           const int64_t *lens, int field_value) {
    Col *c = add_col(b, role, field_value ? 6 : 5, name, family);
    c->var_lens.resize(b->nrows);
    const uint8_t *p = data;
    for (uint64_t i = 0; i < b->nrows; i++) {
        if (lens[i] < 0) {
            c->var_lens[i] = -1;
            continue;
        }
        size_t before = c->var_data.size();
        tagvalue_str_cell(c->var_data, p, (size_t)lens[i], field_value);
        p += lens[i];
        c->var_lens[i] = (int64_t)(c->var_data.size() - before);
    }
    return BYDB_OK;
}

extern "C" int bydb_frame_add_tagvalue_int(bydb_frame_builder *b, uint8_t role,
                                           const char *name, const char *family,
                                           const int64_t *vals,
                                           const uint8_t *nulls, int field_value) {
    Col *c = add_col(b, role, field_value ? 6 : 5, name, family);
    c->var_lens.resize(b->nrows);
    for (uint64_t i = 0; i < b->nrows; i++) {
        if (nulls && nulls[i]) {
            c->var_lens[i] = -1;
            continue;
        }
        size_t before = c->var_data.size();
        tagvalue_int_cell(c->var_data, vals[i], field_value);
        c->var_lens[i] = (int64_t)(c->var_data.size() - before);
    }
    return BYDB_OK;
}

extern "C" int bydb_frame_finish(bydb_frame_builder *b) {
    b->out.clear();
    auto &o = b->out;
    o.insert(o.end(), kMagic, kMagic + 4);
    o.push_back(kWireVersion);
    uvarint_append(o, b->nrows);
    uvarint_append(o, b->cols.size());
    for (const Col &c : b->cols) {
        o.push_back(c.role);
        o.push_back(c.type);
        uvarint_append(o, c.name.size());
        o.insert(o.end(), c.name.begin(), c.name.end());
        uvarint_append(o, c.family.size());
        o.insert(o.end(), c.family.begin(), c.family.end());
        // validity bitmap: bit j set <=> row j null (LE bit packing)
        uint64_t n = b->nrows;
        if (n > 0) {
            size_t nbytes = (size_t)((n + 7) / 8);
            size_t start = o.size();
            o.resize(start + nbytes, 0);
            for (uint64_t j = 0; j < n; j++) {
                bool isnull = c.type == 1 || c.type == 2
                                  ? (c.nulls.size() > j && c.nulls[j] != 0)
                                  : c.var_lens[j] < 0;
                if (isnull) o[start + j / 8] |= (uint8_t)(1u << (j % 8));
            }
        }
        if (c.type == 1 || c.type == 2) {
            for (uint64_t j = 0; j < n; j++) le64_append(o, (uint64_t)c.i64[j]);
        } else {
            const uint8_t *p = c.var_data.data();
            for (uint64_t j = 0; j < n; j++) {
                int64_t l = c.var_lens[j] < 0 ? 0 : c.var_lens[j];
                uvarint_append(o, (uint64_t)l);
                o.insert(o.end(), p, p + l);
                p += l;
            }
        }
    }
    return BYDB_OK;
}

extern "C" uint64_t bydb_frame_len(bydb_frame_builder *b) { return b->out.size(); }
extern "C" const uint8_t *bydb_frame_data(bydb_frame_builder *b) {
    return b->out.data();
}

// ---- minimal decode (round-trip checking / reduce-side ingest) ----
// Walks the frame and surfaces each column's raw sections; validation per
// frame/validate.go: magic, version, lengths within the buffer.
struct bydb_frame_reader {
    std::vector<uint8_t> buf;
    uint64_t nrows = 0, ncols = 0;
    struct RCol {
        uint8_t role, type;
        std::string name, family;
        size_t bitmap_off, data_off, data_len;
    };
    std::vector<RCol> cols;
    std::string err;
};

static bool rd_uvarint(const uint8_t *p, size_t len, size_t *pos, uint64_t *out) {
    uint64_t u = 0;
    unsigned sh = 0;
    while (*pos < len) {
        uint8_t c = p[(*pos)++];
        u |= (uint64_t)(c & 0x7f) << sh;
        if (c < 0x80) {
            *out = u;
            return true;
        }
        sh += 7;
        if (sh > 63) return false;
    }
    return false;
}

extern "C" bydb_frame_reader *bydb_frame_open(const uint8_t *data, uint64_t len) {
    auto *r = new bydb_frame_reader();
    r->buf.assign(data, data + len);
    const uint8_t *p = r->buf.data();
    if (len < 7 || memcmp(p, kMagic, 4) != 0 || p[4] != kWireVersion) {
        r->err = len < 7 ? "truncated" : memcmp(p, kMagic, 4) ? "bad magic" : "bad version";
        return r;
    }
    size_t pos = 5;
    if (!rd_uvarint(p, len, &pos, &r->nrows) ||
        !rd_uvarint(p, len, &pos, &r->ncols)) {
        r->err = "truncated header";
        return r;
    }
    for (uint64_t ci = 0; ci < r->ncols; ci++) {
        bydb_frame_reader::RCol c;
        if (pos + 2 > len) { r->err = "truncated column header"; return r; }
        c.role = p[pos++];
        c.type = p[pos++];
        uint64_t nl, fl;
        if (!rd_uvarint(p, len, &pos, &nl) || pos + nl > len) { r->err = "bad name"; return r; }
        c.name.assign((const char *)p + pos, nl);
        pos += nl;
        if (!rd_uvarint(p, len, &pos, &fl) || pos + fl > len) { r->err = "bad family"; return r; }
        c.family.assign((const char *)p + pos, fl);
        pos += fl;
        uint64_t n = r->nrows;
        c.bitmap_off = pos;
        if (n > 0) pos += (size_t)((n + 7) / 8);
        c.data_off = pos;
        if (c.type == 1 || c.type == 2) {
            pos += (size_t)n * 8;
        } else {
            for (uint64_t j = 0; j < n; j++) {
                uint64_t l;
                if (!rd_uvarint(p, len, &pos, &l) || pos + l > len) {
                    r->err = "bad var cell";
                    return r;
                }
                pos += l;
            }
        }
        if (pos > len) { r->err = "truncated column"; return r; }
        c.data_len = pos - c.data_off;
        r->cols.push_back(c);
    }
    return r;
}

extern "C" void bydb_frame_close(bydb_frame_reader *r) { delete r; }
extern "C" const char *bydb_frame_reader_error(bydb_frame_reader *r) {
    return r->err.c_str();
}
extern "C" uint64_t bydb_frame_nrows(bydb_frame_reader *r) { return r->nrows; }
extern "C" uint64_t bydb_frame_ncols(bydb_frame_reader *r) { return r->ncols; }
extern "C" int bydb_frame_col_info(bydb_frame_reader *r, uint64_t ci,
                                   uint8_t *role, uint8_t *type, char *name,
                                   uint64_t name_cap, char *family,
                                   uint64_t family_cap) {
    if (ci >= r->cols.size()) return BYDB_ERR_BAD_ARG;
    auto &c = r->cols[ci];
    *role = c.role;
    *type = c.type;
    snprintf(name, name_cap, "%s", c.name.c_str());
    snprintf(family, family_cap, "%s", c.family.c_str());
    return BYDB_OK;
}
#include <cstdio>
extern "C" int bydb_frame_col_null(bydb_frame_reader *r, uint64_t ci, uint64_t row) {
    auto &c = r->cols[ci];
    return (r->buf[c.bitmap_off + row / 8] >> (row % 8)) & 1;
}
extern "C" int bydb_frame_col_i64(bydb_frame_reader *r, uint64_t ci,
                                  int64_t *out) {
    auto &c = r->cols[ci];
    if (c.type != 1 && c.type != 2) return BYDB_ERR_BAD_ARG;
    memcpy(out, r->buf.data() + c.data_off, (size_t)r->nrows * 8);
    return BYDB_OK;
}
extern "C" int bydb_frame_col_var(bydb_frame_reader *r, uint64_t ci,
                                  uint8_t *data_out, uint64_t data_cap,
                                  int64_t *lens_out, uint64_t *data_len) {
    auto &c = r->cols[ci];
    if (c.type == 1 || c.type == 2) return BYDB_ERR_BAD_ARG;
    const uint8_t *p = r->buf.data();
    size_t pos = c.data_off;
    uint64_t o = 0;
    for (uint64_t j = 0; j < r->nrows; j++) {
        uint64_t l;
        if (!rd_uvarint(p, r->buf.size(), &pos, &l)) return BYDB_ERR_BAD_DATA;
        int isnull = bydb_frame_col_null(r, ci, j);
        lens_out[j] = isnull ? -1 : (int64_t)l;
        if (o + l > data_cap) return BYDB_ERR_OOM;
        memcpy(data_out + o, p + pos, l);
        o += l;
        pos += l;
    }
    *data_len = o;
    return BYDB_OK;
}

// ---- top-N over group results ----
// Restates BatchTop's ordering contract (pkg/query/vectorized/measure/
// top.go:31-121 + ApplyTopToReduce, reduce.go:290): asc keeps the lowest N
// values (output smallest first), desc the highest N (largest first);
// ties surface in insertion order (earlier group wins, top.go:78-81);
// empty groups (no rows folded) are nulls and sort lowest (top.go:34,95).
// value_sel: 0 sum_i, 1 count, 2 min_i, 3 max_i, 4 mean_i,
//            5 sum_f, 6 min_f, 7 max_f, 8 mean_f.
#include <algorithm>

extern "C" int bydb_top_groups(const bydb_result *results, int64_t n_groups,
                               int value_sel, int64_t k, int asc,
                               int64_t *out_idx, int64_t *out_n) {
    if (k < 0 || value_sel < 0 || value_sel > 8) return BYDB_ERR_BAD_ARG;
    struct Row { int64_t idx; int64_t iv; double fv; bool is_f; bool null; };
    std::vector<Row> rows((size_t)n_groups);
    bool is_f = value_sel >= 5;
    for (int64_t g = 0; g < n_groups; g++) {
        const bydb_result *r = &results[g];
        Row &w = rows[(size_t)g];
        w.idx = g;
        w.is_f = is_f;
        w.null = r->count == 0;
        switch (value_sel) {
        case 0: w.iv = r->sum_i; break;
        case 1: w.iv = r->count; break;
        case 2: w.iv = r->min_i; break;
        case 3: w.iv = r->max_i; break;
        case 4: w.iv = r->mean_i; break;
        case 5: w.fv = r->sum_f; break;
        case 6: w.fv = r->min_f; break;
        case 7: w.fv = r->max_f; break;
        default: w.fv = r->mean_f; break;
        }
    }
    auto cmp = [&](const Row &a, const Row &b) {
        // cmpTopVal: nulls lowest (top.go:95-121)
        int c;
        if (a.null && b.null) c = 0;
        else if (a.null) c = -1;
        else if (b.null) c = 1;
        else if (is_f) c = a.fv < b.fv ? -1 : a.fv > b.fv ? 1 : 0;
        else c = a.iv < b.iv ? -1 : a.iv > b.iv ? 1 : 0;
        if (c != 0) return asc ? c < 0 : c > 0;
        return a.idx < b.idx;  // ties: insertion order (earlier wins)
    };
    std::stable_sort(rows.begin(), rows.end(), cmp);
    int64_t n = k < n_groups ? k : n_groups;
    for (int64_t i = 0; i < n; i++) out_idx[i] = rows[(size_t)i].idx;
    *out_n = n;
    return BYDB_OK;
}
