/* include/bydb_gpu.h — C-ABI of the MI355X-native measure scan+aggregate
 * engine (libbydb_gpu.so).
 *
 * This is the drop-in boundary a Go host binds over cgo.  Each entry point
 * names the reference interface it replaces (apache/skywalking-banyandb):
 *
 *  - session/configure/consume/finalize mirror the lifecycle of
 *    vectorized.BreakerOperator (pkg/query/vectorized/operator.go:65-70:
 *    Init / Consume / Finalize / NextBatch / Close) as built for
 *    BatchAggregation by BuildOperators
 *    (pkg/query/vectorized/measure/plan.go:61-131) — one session per
 *    pipeline, sticky errors, Close releases everything.
 *  - part upload replaces the storage seam
 *    model.MeasureBatchResult.PullBatch (pkg/query/model/batch.go:45-55) +
 *    blockCursor.loadData/block.mustReadFrom (banyand/measure/block.go:818,
 *    324): instead of decoding on the host, the encoded block streams are
 *    made resident in HBM and decode+fold happens in HIP kernels.
 *  - the partial layout mirrors aggregation.Partial[N]
 *    (pkg/query/aggregation/aggregation.go:36-55) with MEAN's count sidecar
 *    (vectorized/measure/aggregation.go meanCountSuffix) and AggModeMap /
 *    AggModeReduce semantics (aggregation.go:57-64).
 *
 * Plain pointers and sizes only; no torch types.  See INTEGRATION.md for
 * the cgo binding a BanyanDB maintainer would add.
 */
#ifndef BYDB_GPU_H
#define BYDB_GPU_H

#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

/* EncodeType — pkg/encoding/encoding.go:86-98 */
enum {
    BYDB_ENC_UNKNOWN = 0,
    BYDB_ENC_CONST = 1,
    BYDB_ENC_DELTA_CONST = 2,
    BYDB_ENC_DELTA = 3,
    BYDB_ENC_DELTA_OF_DELTA = 4,
    BYDB_ENC_CONST_WV = 5,
    BYDB_ENC_DELTA_CONST_WV = 6,
    BYDB_ENC_DELTA_WV = 7,
    BYDB_ENC_DELTA_OF_DELTA_WV = 8,
    BYDB_ENC_PLAIN = 9,
    BYDB_ENC_DICTIONARY = 10,
};

/* ValueType — pkg/pb/v1/valuetype */
enum { BYDB_VT_INT64 = 2, BYDB_VT_FLOAT64 = 3 };

/* AggFunc — pkg/query/vectorized/measure/aggregation.go:95-107 */
enum {
    BYDB_AGG_SUM = 0,
    BYDB_AGG_COUNT = 1,
    BYDB_AGG_MIN = 2,
    BYDB_AGG_MAX = 3,
    BYDB_AGG_MEAN = 4,
};

/* AggMode — vectorized/measure/aggregation.go:57-64 */
enum { BYDB_MODE_ALL = 0, BYDB_MODE_MAP = 1, BYDB_MODE_REDUCE = 2 };

/* Status codes (0 = ok).  Sticky per session, mirroring the operator's
 * sticky Go error contract. */
enum {
    BYDB_OK = 0,
    BYDB_ERR = -1,
    BYDB_ERR_HIP = -2,
    BYDB_ERR_BAD_ARG = -3,
    BYDB_ERR_BAD_DATA = -4,
    BYDB_ERR_STATE = -5,
    BYDB_ERR_NO_GPU = -6,
    BYDB_ERR_OOM = -7,
};

/* One (series, block) entry of the resident part directory.  Field
 * semantics follow blockMetadata + timestampsMetadata + columnMetadata
 * (banyand/measure/block_metadata.go:254-311, column_metadata.go:47-52):
 * the column payload headers ([type][exp][firstValue], column.go:183-263)
 * are parsed at load time on the host, so kernels see pure streams. */
typedef struct {
    uint64_t series_id;
    uint32_t count;               /* rows in block, <= 8192 (measure.go:41-46) */
    uint8_t ts_enc_with_version;  /* timestampsMetadata.encodeType */
    uint8_t version_enc;
    uint8_t field_enc;            /* common EncodeType of the field column */
    uint8_t field_vtype;          /* BYDB_VT_* */
    int64_t ts_min;               /* == first timestamp (block ascending) */
    int64_t ts_max;
    int64_t version_first;
    int64_t field_first;          /* decoded firstValue of the field column */
    int16_t exp;                  /* float64 decimal exponent (float.go:69) */
    uint8_t _pad[6];
    uint64_t ts_off;              /* ts varint stream (header-free) */
    uint64_t ts_len;
    uint64_t field_off;           /* field varint stream (header-free) */
    uint64_t field_len;
    uint64_t tag_off;             /* tag column payload incl. type byte, or 0 */
    uint64_t tag_len;             /* slot 0; slots 1-2 below (conjunctive) */
    uint64_t tag2_off;
    uint64_t tag2_len;
    uint64_t tag3_off;
    uint64_t tag3_len;
    uint32_t group_code;          /* dense group index (group-by), else 0 */
    uint32_t _pad2;
} bydb_block_desc;

/* Dense per-group partial accumulator — one slot per group.
 * sum_i/count are exact (wrapping) int64; min/max are on the int64 domain
 * (for float64 fields these are the decimal-int min/max, restored to
 * float64 at finalize — restore is monotone, float.go:69-102); sum_f is
 * the float64 mantissa sum (double).                                      */
typedef struct {
    int64_t sum_i;
    int64_t count;
    int64_t min_i;
    int64_t max_i;
    double sum_f;
    double _pad;
} bydb_partial;

/* Finalised per-group result (AggModeAll shape). */
typedef struct {
    int64_t sum_i;
    double sum_f;
    int64_t count;
    int64_t min_i;
    int64_t max_i;
    double min_f;
    double max_f;
    int64_t mean_i;   /* meanFunc.Val with the >=1 clamp (function.go:30-45) */
    double mean_f;
} bydb_result;

typedef struct bydb_session bydb_session;

/* ---- session lifecycle (BreakerOperator Init/Close) ---- */
bydb_session *bydb_session_create(int device);
void bydb_session_destroy(bydb_session *s);
const char *bydb_last_error(bydb_session *s);

/* ---- part residency (storage seam) ----
 * reserve once, then append payload chunks + their block descs; offsets in
 * descs are absolute within the whole part payload. */
int bydb_part_reserve(bydb_session *s, uint64_t payload_bytes, int64_t n_blocks);
int bydb_part_append(bydb_session *s, const uint8_t *payload, uint64_t len,
                     const bydb_block_desc *blocks, int64_t n_blocks);
int bydb_part_clear(bydb_session *s);

/* ---- aggregation configure (BuildOperators / AggSpec) ----
 * funcs: bitmask of (1<<BYDB_AGG_*); n_groups >= 1 (1 = scalar aggregate,
 * matching aggAllIterator).  field_vtype selects int64/float64 semantics. */
int bydb_agg_configure(bydb_session *s, int field_vtype, uint32_t func_mask,
                       uint32_t n_groups, int mode);

/* Optional: accumulate partials into an external device buffer of
 * n_groups * sizeof(bydb_partial) bytes (e.g. a torch CUDA tensor for an
 * RCCL merge).  Pass NULL to use the session's own buffer. */
int bydb_set_partials_buffer(bydb_session *s, void *dev_ptr, uint64_t len);

/* ---- consume (BreakerOperator.Consume over the resident part) ----
 * min_ts/max_ts: inclusive row clamp (timestamp.FindRange,
 * pkg/timestamp/range.go:143-170).  pred: optional tag-equality predicate
 * value bytes (dictionary tags only), pred_len 0 = none.  Asynchronous:
 * returns after launch; bydb_finalize syncs. */
int bydb_consume(bydb_session *s, int64_t min_ts, int64_t max_ts,
                 const uint8_t *pred, uint64_t pred_len);

/* Conjunctive multi-tag predicate (<=3 values; preds[i] applies to the
 * block's tag slot i).  pred_lens[i] == 0 disables slot i. */
int bydb_consume_multi(bydb_session *s, int64_t min_ts, int64_t max_ts,
                       const uint8_t *const *preds, const uint64_t *pred_lens,
                       int n_preds);

/* ---- finalize + emit (Finalize/NextBatch) ----
 * Syncs the stream, downloads partials and finalises per-group results
 * (MEAN = sum/count with the >=1 clamp; float64 restore per float.go).
 * out must hold n_groups entries.  Resets partials for the next epoch. */
int bydb_finalize(bydb_session *s, bydb_result *out, int64_t n_groups);

/* Finalize partials only (AggModeMap: emit Partial state for an external
 * reduce — e.g. RCCL — without applying Val()).  Leaves device partials
 * in the accumulation buffer; host copy written to out. */
int bydb_finalize_partials(bydb_session *s, bydb_partial *out, int64_t n_groups);

/* Reset accumulators to the fold identity (Map.Reset, function.go). */
int bydb_reset(bydb_session *s);

/* Per-group first-seen keys ((item << 13) | row in storage order; ~0 =
 * group never entered).  Sorting group ids by these keys reproduces the
 * reference's first-seen group materialisation order (computeKey creates
 * groups in row-iteration order, vectorized/measure/aggregation.go:523),
 * so the Go twin can emit NextBatch rows in reference order from the
 * host-domain result buffers.  out holds n_groups entries. */
int bydb_group_first_seen(bydb_session *s, uint64_t *out, int64_t n_groups);

/* Combine external partials (AggModeReduce Combine semantics,
 * aggregation_reduce.go:120-138) into results on the host.  parts is
 * [n_parts_per_group][n_groups]; float_exp selects the float64 restore
 * domain (BYDB_FLOAT_RAW_EXP = raw IEEE-754 cells). */
int bydb_reduce_partials2(const bydb_partial *parts, int64_t n_parts_per_group,
                          int64_t n_groups, int field_vtype, int16_t float_exp,
                          bydb_result *out);

/* Timing of the last consume's kernels in milliseconds (HIP events on the
 * session stream) — feeds bench.py's roofline.achieved. */
double bydb_last_consume_ms(bydb_session *s);

/* ---- host-side part builder (fixture writer) ----
 * Mirrors the reference block write path: mustInitFromDataPoints ->
 * block.mustWriteTo (block.go:60,139), mustWriteTimestampsTo
 * (block.go:386-404), column.mustWriteTo (column.go:157-278).  Pure host
 * code; no GPU needed.  Produces payload bytes + bydb_block_desc entries
 * whose encoded streams are byte-identical to the reference's (validated
 * against the oracle in tests/).                                          */
typedef struct bydb_part_builder bydb_part_builder;
bydb_part_builder *bydb_part_builder_create(void);
void bydb_part_builder_destroy(bydb_part_builder *b);
const char *bydb_part_builder_error(bydb_part_builder *b);

int bydb_part_builder_add_block_i64(bydb_part_builder *b, uint64_t series_id,
                                    const int64_t *ts, const int64_t *versions,
                                    const int64_t *vals, int64_t n,
                                    uint32_t group_code);
int bydb_part_builder_add_block_f64(bydb_part_builder *b, uint64_t series_id,
                                    const int64_t *ts, const int64_t *versions,
                                    const double *vals, int64_t n,
                                    uint32_t group_code);
/* Raw-float sentinel for bydb_set_float_exp / bydb_reduce_partials2:
 * sessions over null-bearing FLOAT64 columns (Plain IEEE-754 cell blocks,
 * convert/number.go:128-132) fold in the raw-bits domain — ordered-bit
 * min/max keys, direct double sums — selected by this exponent value.
 * Mixing raw and decimal float blocks in one session is a loud error. */
#define BYDB_FLOAT_RAW_EXP ((int16_t)-32768)

int bydb_part_builder_add_block_f64_nullable(
    bydb_part_builder *b, uint64_t series_id, const int64_t *ts,
    const int64_t *versions, const double *vals, const uint8_t *valid,
    int64_t n, uint32_t group_code);
/* null-bearing int64 column (valid[i]==0 -> row i null): stored as the
 * reference's Plain fallback — a bytes block of 8-B sign-flip cells with
 * nil rows zero-length (column.go:214-233, convert/number.go:33-46);
 * the fold skips nulls (vectorized/measure/aggregation.go:310) */
int bydb_part_builder_add_block_i64_nullable(
    bydb_part_builder *b, uint64_t series_id, const int64_t *ts,
    const int64_t *versions, const int64_t *vals, const uint8_t *valid,
    int64_t n, uint32_t group_code);
/* attach a dictionary-encoded tag column to the block just added:
 * tag value of row i = tag_values[codes[i]] (code order = first-seen) */
int bydb_part_builder_set_block_tag(bydb_part_builder *b, const uint8_t *data,
                                    const int64_t *lens, int64_t n);

uint64_t bydb_part_builder_payload_len(bydb_part_builder *b);
const uint8_t *bydb_part_builder_payload(bydb_part_builder *b);
int64_t bydb_part_builder_n_blocks(bydb_part_builder *b);
const bydb_block_desc *bydb_part_builder_blocks(bydb_part_builder *b);
/* drop payload bytes + descs already consumed, keep absolute offsets
 * running (chunked upload) */
int bydb_part_builder_drain(bydb_part_builder *b);

/* ---- synthetic workload generator (BASELINE.json section 8d) ----
 * Generates one series' datapoints and appends its blocks (<=8192 rows
 * each).  values = base + i*ramp + noise, noise in [-3,3] from
 * splitmix64(seed ^ series_index).  threads: host threads for encoding. */
int bydb_gen_series_i64(bydb_part_builder *b, uint64_t series_index,
                        int64_t n_dp, int64_t t0, int64_t stride_ns,
                        int64_t base, int64_t ramp, uint64_t seed,
                        uint32_t group_code);
int bydb_gen_series_f64(bydb_part_builder *b, uint64_t series_index,
                        int64_t n_dp, int64_t t0, int64_t stride_ns,
                        double base, double ramp, uint64_t seed,
                        uint32_t group_code);
/* bulk generators: n_series consecutive series (first_index ...) encoded
 * on `threads` host threads; base = series_index * base_step.
 * group_mod > 0 assigns group_code = series_index %% group_mod. */
int bydb_gen_series_bulk_i64(bydb_part_builder *b, uint64_t first_index,
                             int64_t n_series, int64_t n_dp, int64_t t0,
                             int64_t stride_ns, int64_t base_step, int64_t ramp,
                             uint64_t seed, uint32_t group_mod, int threads);
int bydb_gen_series_bulk_f64(bydb_part_builder *b, uint64_t first_index,
                             int64_t n_series, int64_t n_dp, int64_t t0,
                             int64_t stride_ns, double base_step, double ramp,
                             uint64_t seed, uint32_t group_mod, int threads);
/* entity-tag table for the generators: every generated block gets
 * values[series_index %% n_values] as a constant tag column in `slot`.
 * data = concatenated values; lens[i] < 0 marks a nil value. */
int bydb_part_builder_set_tag_table(bydb_part_builder *b, int slot,
                                    const uint8_t *data, const int64_t *lens,
                                    int64_t n_values);
/* raw helpers used by the part reader */
int bydb_part_builder_append_raw(bydb_part_builder *b, const uint8_t *data,
                                 uint64_t len);
int bydb_part_builder_append_desc(bydb_part_builder *b,
                                  const bydb_block_desc *d);

/* ---- per-row group-by on dictionary tag columns ----
 * computeKey/appendKeyComponent semantics (vectorized/measure/
 * aggregation.go:523 + groupby.go:287-364) with a host-supplied group
 * DOMAIN: dense group id = index into the domain value list (the host
 * layer controls materialisation order; the reference's is first-seen).
 * Nil or out-of-domain rows drop.  dom_blob = concatenated values,
 * dom_offs[n_values+1] prefix offsets. */
int bydb_agg_configure_by_tag(bydb_session *s, int field_vtype,
                              uint32_t func_mask, int tag_slot,
                              const uint8_t *dom_blob,
                              const uint64_t *dom_offs, uint32_t n_values,
                              int mode);
/* composite key over up to 3 tag slots: gid = g0 + n0*g1 + n0*n1*g2 */
int bydb_agg_configure_by_tags(bydb_session *s, int field_vtype,
                               uint32_t func_mask, const int *slots,
                               int n_slots, const uint8_t *const *dom_blobs,
                               const uint64_t *const *dom_offs,
                               const uint32_t *n_values, int mode);

/* ---- on-disk part directory (Appendix A; banyand/measure/part.go:37-56)
 * write_dir emits metadata.json / meta.bin / primary.bin / timestamps.bin
 * / fv.bin / <family>.tfm/.tf in the reference's exact layout; read_dir
 * loads a part directory (ours or the reference's) into a builder. */
int bydb_part_write_dir(bydb_part_builder *b, const char *path,
                        const char *field_name, const char *tag_family,
                        const char *const *tag_names, int n_tags);
int bydb_part_read_dir(bydb_part_builder *b, const char *path);

/* ---- columnar wire frame codec (egress / Map->Reduce transport) ----
 * Byte-stable layout per pkg/query/vectorized/frame (frame.go:26-95,
 * encode.go:42-170): magic "\0VFR", version, uvarint nrows/ncols, per
 * column [role, type, name, family, validity bitmap (1=null, LE
 * bit-packed), data (numeric = N x 8B LE; var = uvarint len + bytes)]. */
typedef struct bydb_frame_builder bydb_frame_builder;
typedef struct bydb_frame_reader bydb_frame_reader;
bydb_frame_builder *bydb_frame_builder_create(uint64_t nrows);
void bydb_frame_builder_destroy(bydb_frame_builder *b);
const char *bydb_frame_builder_error(bydb_frame_builder *b);
int bydb_frame_add_i64(bydb_frame_builder *b, uint8_t role, const char *name,
                       const char *family, const int64_t *vals,
                       const uint8_t *nulls);
int bydb_frame_add_f64(bydb_frame_builder *b, uint8_t role, const char *name,
                       const char *family, const double *vals,
                       const uint8_t *nulls);
int bydb_frame_add_str(bydb_frame_builder *b, uint8_t role, const char *name,
                       const char *family, const uint8_t *data,
                       const int64_t *lens);
int bydb_frame_add_bytes(bydb_frame_builder *b, uint8_t role, const char *name,
                         const char *family, const uint8_t *data,
                         const int64_t *lens);
/* minimal TagValue/FieldValue proto cells (raw_emit.go egress shape);
 * field_value selects FieldValue (type 6) over TagValue (type 5) */
int bydb_frame_add_tagvalue_str(bydb_frame_builder *b, uint8_t role,
                                const char *name, const char *family,
                                const uint8_t *data, const int64_t *lens,
                                int field_value);
int bydb_frame_add_tagvalue_int(bydb_frame_builder *b, uint8_t role,
                                const char *name, const char *family,
                                const int64_t *vals, const uint8_t *nulls,
                                int field_value);
int bydb_frame_finish(bydb_frame_builder *b);
uint64_t bydb_frame_len(bydb_frame_builder *b);
const uint8_t *bydb_frame_data(bydb_frame_builder *b);
bydb_frame_reader *bydb_frame_open(const uint8_t *data, uint64_t len);
void bydb_frame_close(bydb_frame_reader *r);
const char *bydb_frame_reader_error(bydb_frame_reader *r);
uint64_t bydb_frame_nrows(bydb_frame_reader *r);
uint64_t bydb_frame_ncols(bydb_frame_reader *r);
int bydb_frame_col_info(bydb_frame_reader *r, uint64_t ci, uint8_t *role,
                        uint8_t *type, char *name, uint64_t name_cap,
                        char *family, uint64_t family_cap);
int bydb_frame_col_null(bydb_frame_reader *r, uint64_t ci, uint64_t row);
int bydb_frame_col_i64(bydb_frame_reader *r, uint64_t ci, int64_t *out);
int bydb_frame_col_var(bydb_frame_reader *r, uint64_t ci, uint8_t *data_out,
                       uint64_t data_cap, int64_t *lens_out,
                       uint64_t *data_len);

/* ---- BatchTop ordering over group results ----
 * top.go:31-121 + ApplyTopToReduce (reduce.go:290): asc keeps the lowest
 * k (smallest first), desc the highest k; ties keep insertion order;
 * empty groups are nulls and sort lowest.  value_sel: 0 sum_i, 1 count,
 * 2 min_i, 3 max_i, 4 mean_i, 5 sum_f, 6 min_f, 7 max_f, 8 mean_f. */
int bydb_top_groups(const bydb_result *results, int64_t n_groups,
                    int value_sel, int64_t k, int asc, int64_t *out_idx,
                    int64_t *out_n);

/* ---- AggModeReduce over raw map frames, with replica dedup ----
 * Restates the liaison reduce (aggregation_reduce.go:83-138 +
 * ReduceRawFrames, reduce.go:78): rows keyed by the key columns
 * (appendKeyComponent semantics); the FIRST row per (shard_id, group_key)
 * pair Combines, replica duplicates drop; MEAN count sidecars are located
 * by the "__agg_count" name walk; Val() applies meanReduce's >=1 clamp.
 * shard_col < 0 dedups on the group key alone (the reference's
 * shardIDIdx == -1 fallback).  Outputs in first-seen group order:
 * out_i64/out_f64 are [n_groups][n_specs] finals (int64 or float64
 * meaningful per the value column's frame type); key_buf/key_offs
 * (optional) receive the packed group keys.  Returns BYDB_OK and
 * *out_n_groups, or BYDB_ERR_OOM when out_cap/key_buf_cap are small. */
typedef struct {
    int32_t input_col;   /* frame column index of the partial value */
    int32_t func;        /* BYDB_AGG_* */
} bydb_reduce_spec;
int bydb_reduce_frames(const uint8_t *const *frames,
                       const uint64_t *frame_lens, int n_frames,
                       int shard_col, const int32_t *key_cols, int n_key_cols,
                       const bydb_reduce_spec *specs, int n_specs,
                       int64_t out_cap, int64_t *out_i64, double *out_f64,
                       uint8_t *key_buf, uint64_t key_buf_cap,
                       uint64_t *key_offs, int64_t *out_n_groups);

#ifdef __cplusplus
}
#endif
#endif /* BYDB_GPU_H */
