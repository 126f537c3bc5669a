/* oracle/bydb_oracle.h
 *
 * TEST INFRASTRUCTURE ONLY — CPU oracle for the BanyanDB measure
 * scan+aggregate hot path.  This library is a plain-C restatement of the
 * reference Go algorithms (apache/skywalking-banyandb, snapshot at
 * /root/reference) used exclusively as the parity checker and the reported
 * CPU baseline.  Nothing in the product path may link, import or call it;
 * only tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg do.
 *
 * Parity pinning: validated against the reference's own table-driven test
 * vectors transcribed into tests/golden/ (pkg/encoding/delta_test.go,
 * int_list_test.go, float_test.go, int_test.go, dictionary_test.go).
 */
#ifndef BYDB_ORACLE_H
#define BYDB_ORACLE_H

#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

/* EncodeType — pkg/encoding/encoding.go:86-98 */
enum {
    BO_ENC_UNKNOWN = 0,
    BO_ENC_CONST = 1,
    BO_ENC_DELTA_CONST = 2,
    BO_ENC_DELTA = 3,
    BO_ENC_DELTA_OF_DELTA = 4,
    BO_ENC_CONST_WITH_VERSION = 5,
    BO_ENC_DELTA_CONST_WITH_VERSION = 6,
    BO_ENC_DELTA_WITH_VERSION = 7,
    BO_ENC_DELTA_OF_DELTA_WITH_VERSION = 8,
    BO_ENC_PLAIN = 9,
    BO_ENC_DICTIONARY = 10,
};

/* ValueType — pkg/pb/v1/valuetype */
enum {
    BO_VT_UNKNOWN = 0,
    BO_VT_STR = 1,
    BO_VT_INT64 = 2,
    BO_VT_FLOAT64 = 3,
    BO_VT_BINARY = 4,
};

/* Error codes (0 = ok) */
enum {
    BO_OK = 0,
    BO_ERR_EMPTY = -1,
    BO_ERR_TRUNCATED = -2,
    BO_ERR_TOO_LONG_VARINT = -3,
    BO_ERR_TAIL = -4,
    BO_ERR_BAD_TYPE = -5,
    BO_ERR_CAPACITY = -6,
    BO_ERR_LOSSY_FLOAT = -7,
    BO_ERR_ZSTD_UNAVAILABLE = -8,
    BO_ERR_BAD_DATA = -9,
};

/* ---- varint / zigzag (pkg/encoding/int.go:81-148) ---- */
size_t bo_varint64_list_encode(uint8_t *dst, const int64_t *vs, int64_t n);
int bo_varint64_list_decode(int64_t *dst, int64_t n, const uint8_t *src,
                            size_t src_len, size_t *consumed);
size_t bo_varuint64_encode(uint8_t *dst, uint64_t u);
int bo_varuint64_decode(const uint8_t *src, size_t src_len, uint64_t *out,
                        size_t *consumed);

/* ---- int64 list codec with encode-type selection
 *      (pkg/encoding/int_list.go:27-101, delta.go:26-118) ---- */
int bo_int64_list_encode(uint8_t *dst, size_t cap, const int64_t *a, int64_t n,
                         size_t *out_len, uint8_t *out_type, int64_t *out_first);
int bo_int64_list_decode(int64_t *dst, const uint8_t *src, size_t src_len,
                         uint8_t mt, int64_t first_value, int64_t items_count);

/* ---- decimal float codec (pkg/encoding/float.go:30-226) ---- */
int bo_float_list_to_decimal(const double *src, int64_t n, int64_t *out_ints,
                             int16_t *out_exp);
int bo_decimal_to_float_list(double *dst, const int64_t *vs, int16_t exponent,
                             int64_t n);
double bo_go_pow10(int n); /* exact restatement of Go math.Pow10 */

/* ---- order-preserving int64 cell codec (pkg/convert/number.go:33-60,95-110) ---- */
void bo_cell_i64_to_bytes(uint8_t out[8], int64_t v);
int64_t bo_cell_bytes_to_i64(const uint8_t in[8]);

/* ---- column payload codec (banyand/measure/column.go:157-278, 331-423)
 * int64 payload  = [type 1B][firstValue 8B cell] ++ stream
 * float64 payload= [type 1B][exp 2B BE][firstValue 8B cell] ++ stream     */
int bo_column_i64_encode(uint8_t *dst, size_t cap, const int64_t *vals,
                         int64_t n, size_t *out_len);
int bo_column_i64_decode(int64_t *dst, const uint8_t *payload, size_t len,
                         int64_t n);
int bo_column_f64_encode(uint8_t *dst, size_t cap, const double *vals,
                         int64_t n, size_t *out_len);
int bo_column_f64_decode(double *dst, const uint8_t *payload, size_t len,
                         int64_t n);

/* ---- timestamps+versions payload (banyand/measure/block.go:386-443)
 * payload = ts stream ++ versions stream; metadata returned separately.   */
int bo_timestamps_encode(uint8_t *dst, size_t cap, const int64_t *ts,
                         const int64_t *versions, int64_t n, size_t *out_len,
                         uint8_t *ts_enc_with_version, int64_t *ts_min,
                         int64_t *ts_max, uint64_t *version_offset,
                         uint8_t *version_enc, int64_t *version_first);
int bo_timestamps_decode(int64_t *ts, int64_t *versions, const uint8_t *payload,
                         size_t len, uint8_t ts_enc_with_version,
                         int64_t ts_min, uint64_t version_offset,
                         uint8_t version_enc, int64_t version_first, int64_t n);

/* ---- FindRange row clamp (pkg/timestamp/range.go:143-170) ---- */
int bo_find_range(const int64_t *ts, int64_t n, int64_t min_val,
                  int64_t max_val, int64_t *start, int64_t *end);

/* ---- bytes block + dictionary (pkg/encoding/bytes.go:45-330,
 *      dictionary.go:27-300, reader.go/writer.go MSB-first bitstream) ----
 * Values are passed as a concatenated buffer + lengths; length -1 = nil.  */
int bo_bytes_block_encode(uint8_t *dst, size_t cap, const uint8_t *data,
                          const int64_t *lens, int64_t n, size_t *out_len);
int bo_bytes_block_decode(uint8_t *data_out, size_t data_cap, int64_t *lens_out,
                          const uint8_t *src, size_t src_len, int64_t n,
                          size_t *data_len_out);
int bo_dictionary_encode(uint8_t *dst, size_t cap, const uint8_t *data,
                         const int64_t *lens, int64_t n, size_t *out_len);
int bo_dictionary_decode(uint8_t *data_out, size_t data_cap, int64_t *lens_out,
                         const uint8_t *src, size_t src_len, int64_t n,
                         size_t *data_len_out);
/* decode only the per-row dictionary code indices (GPU-parity helper) */
int bo_dictionary_decode_codes(uint32_t *codes_out, const uint8_t *src,
                               size_t src_len, int64_t n);

/* ---- scan + aggregate over a block directory ----
 * Restates blockCursor.loadData -> block.mustReadFrom -> fold
 * (banyand/measure/block.go:324-443,818-860; column.go:331-423;
 *  pkg/query/vectorized/measure/aggregation.go:310-351,464-486;
 *  pkg/query/aggregation/function.go).  Blocks are fed in directory order
 * (sid asc, min_ts asc), rows in ascending ts order — the same order the
 * reference's merged batch path produces.                                  */
typedef struct {
    uint64_t series_id;
    uint32_t count;
    uint8_t ts_enc_with_version;
    uint8_t version_enc;
    uint8_t _pad[2];
    int64_t ts_min;  /* timestampsMetadata.min == first ts value */
    int64_t ts_max;
    int64_t version_first;
    uint64_t ts_off;   /* offset of ts payload in the payload buffer */
    uint64_t ts_len;   /* bytes of ts stream (== versionOffset) */
    uint64_t ver_len;  /* bytes of versions stream (follows ts stream) */
    uint64_t col_off;  /* offset of full field column payload (incl. header) */
    uint64_t col_len;
    uint64_t tag_off;  /* offset of tag column payload (incl. header), or 0 */
    uint64_t tag_len;  /* slot 0; slots 1-2 below */
    uint64_t tag2_off;
    uint64_t tag2_len;
    uint64_t tag3_off;
    uint64_t tag3_len;
    uint32_t group_code; /* dense group index for group-by parity; else 0 */
    uint32_t _pad2;
} bo_block_desc;

typedef struct {
    int64_t sum_i;      /* int64 sum (wrapping) */
    double sum_f;       /* float64 sum, reference fold order */
    int64_t count;      /* rows folded (nulls skipped) */
    int64_t min_i, max_i;
    double min_f, max_f;
} bo_agg_result;

/* field_vtype: BO_VT_INT64 or BO_VT_FLOAT64.
 * min_ts/max_ts: inclusive clamp (INT64_MIN/INT64_MAX for none).
 * predicate: if pred_len > 0, keep only rows whose tag value equals
 * pred[0..pred_len) (tag column must be dictionary-encoded).             */
int bo_scan_agg(const uint8_t *payload, const bo_block_desc *blocks,
                int64_t n_blocks, int field_vtype, int64_t min_ts,
                int64_t max_ts, const uint8_t *pred, int64_t pred_len,
                bo_agg_result *out);

/* Grouped variant: group = blocks[i].group_code in [0, n_groups);
 * out is an array of n_groups results (callers pre-zero? no — the function
 * initialises all slots to the reference's Reset() sentinels).            */
int bo_scan_agg_grouped(const uint8_t *payload, const bo_block_desc *blocks,
                        int64_t n_blocks, int field_vtype, int64_t min_ts,
                        int64_t max_ts, const uint8_t *pred, int64_t pred_len,
                        bo_agg_result *out, int64_t n_groups);

/* Conjunctive multi-tag variant: preds = concatenated predicate bytes,
 * pred_lens[3] (0 disables a slot); preds[i] applies to tag slot i. */
int bo_scan_agg_multi(const uint8_t *payload, const bo_block_desc *blocks,
                      int64_t n_blocks, int field_vtype, int64_t min_ts,
                      int64_t max_ts, const uint8_t *preds_concat,
                      const int64_t pred_lens[3], bo_agg_result *out,
                      int64_t n_groups);

/* Per-row group-by on a dictionary tag: group id = index of the row's
 * tag value in the domain list (host-controlled order = the reference's
 * first-seen materialisation, computeKey aggregation.go:523).  Rows with
 * nil or out-of-domain tags are dropped.  slot: 0..2. */
int bo_scan_agg_bytags(const uint8_t *payload, const bo_block_desc *blocks,
                       int64_t n_blocks, int field_vtype, int64_t min_ts,
                       int64_t max_ts, const int *slots, int n_slots,
                       const uint8_t *const *dom_blobs,
                       const int64_t *const *dom_lens, const int64_t *n_doms,
                       const uint8_t *preds_concat, const int64_t *pred_lens,
                       bo_agg_result *out);
int bo_scan_agg_bytag(const uint8_t *payload, const bo_block_desc *blocks,
                      int64_t n_blocks, int field_vtype, int64_t min_ts,
                      int64_t max_ts, int slot, const uint8_t *dom_blob,
                      const int64_t *dom_lens, int64_t n_dom,
                      bo_agg_result *out);

/* MEAN finalisation — pkg/query/aggregation/function.go:30-45 (clamps to >=1) */
int64_t bo_mean_val_i64(int64_t sum, int64_t count);
double bo_mean_val_f64(double sum, double count);

/* xxhash64 (cespare/xxhash v2.3.0 == canonical XXH64, seed 0) for
 * Entity -> SeriesID derivation (pkg/convert/hash.go:23).                 */
uint64_t bo_xxhash64(const uint8_t *data, size_t len);

#ifdef __cplusplus
}
#endif
#endif /* BYDB_ORACLE_H */
