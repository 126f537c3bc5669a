/* tools/cabi_check.c — pure-C consumer of the drop-in C-ABI, standing in
 * for the cgo binding a BanyanDB maintainer would write.
 * Exercises the documented surface end to end:
 *   1. part builder -> upload -> scalar fold (sum/count/min/max)
 *   2. conjunctive tag predicates (bydb_consume_multi)
 *   3. per-row group-by on a tag domain (bydb_agg_configure_by_tag)
 *   4. AggModeMap partials -> host Combine (bydb_reduce_partials2)
 *   5. wire-frame egress + frame-level reduce with replica dedup
 *      (bydb_frame_* + bydb_reduce_frames)
 *   6. BatchTop ordering (bydb_top_groups)
 *   7. on-disk part round trip (bydb_part_write_dir / read_dir)
 * Compile:
 *   gcc -std=c99 -I include tools/cabi_check.c -L banyandb_amd -lbydb_gpu \
 *       -Wl,-rpath,banyandb_amd -o cabi_check            */
#include <stdio.h>
#include <stdlib.h>
#include <string.h>

#include "bydb_gpu.h"

#define N 1000
#define T0 1700000000000000000LL
#define NGROUPS 3

#define CHECK(cond, what)                                                 \
    do {                                                                  \
        if (!(cond)) {                                                    \
            fprintf(stderr, "FAIL %s (%s:%d)\n", what, __FILE__,          \
                    __LINE__);                                            \
            return 1;                                                     \
        }                                                                 \
    } while (0)

static const char *ENVS[NGROUPS] = {"prod", "dev", "qa"};

int main(void) {
    int64_t ts[N], ver[N], vals[N];
    int64_t want_sum = 0, want_sum_env[NGROUPS] = {0, 0, 0};
    int64_t want_cnt_env[NGROUPS] = {0, 0, 0};
    for (int i = 0; i < N; i++) {
        ts[i] = T0 + (int64_t)i * 1000000LL;
        ver[i] = 1;
        vals[i] = (int64_t)(i * 37 - 5000);
        want_sum += vals[i];
    }
    bydb_part_builder *b = bydb_part_builder_create();
    CHECK(b, "builder create");
    /* 4 blocks, each with a row-varying env tag */
    uint8_t tagdata[N * 8];
    int64_t taglens[N];
    for (int blk = 0; blk < 4; blk++) {
        CHECK(bydb_part_builder_add_block_i64(b, (uint64_t)blk + 1, ts, ver,
                                              vals, N, 0) == BYDB_OK,
              "add_block");
        size_t off = 0;
        for (int i = 0; i < N; i++) {
            int g = (i / 7 + blk) % NGROUPS;
            size_t l = strlen(ENVS[g]);
            memcpy(tagdata + off, ENVS[g], l);
            off += l;
            taglens[i] = (int64_t)l;
            if (blk == 0) {
                want_sum_env[g] += vals[i];
                want_cnt_env[g]++;
            }
        }
        CHECK(bydb_part_builder_set_block_tag(b, tagdata, taglens, N) ==
                  BYDB_OK,
              "set_block_tag");
    }

    bydb_session *s = bydb_session_create(0);
    if (!s) { fprintf(stderr, "no GPU session\n"); return 2; }
    uint64_t plen = bydb_part_builder_payload_len(b);
    int64_t nb = bydb_part_builder_n_blocks(b);
    CHECK(bydb_part_reserve(s, plen, nb) == BYDB_OK, "reserve");
    CHECK(bydb_part_append(s, bydb_part_builder_payload(b), plen,
                           bydb_part_builder_blocks(b), nb) == BYDB_OK,
          "append");
    uint32_t funcs = (1u << BYDB_AGG_SUM) | (1u << BYDB_AGG_COUNT) |
                     (1u << BYDB_AGG_MIN) | (1u << BYDB_AGG_MAX);

    /* ---- 1. scalar fold ---- */
    CHECK(bydb_agg_configure(s, BYDB_VT_INT64, funcs, 1, BYDB_MODE_ALL) ==
              BYDB_OK,
          "configure");
    CHECK(bydb_consume(s, INT64_MIN, INT64_MAX, NULL, 0) == BYDB_OK,
          "consume");
    bydb_result r;
    CHECK(bydb_finalize(s, &r, 1) == BYDB_OK, "finalize");
    CHECK(r.count == 4 * N && r.sum_i == 4 * want_sum && r.min_i == -5000 &&
              r.max_i == (int64_t)(N - 1) * 37 - 5000,
          "scalar numbers");

    /* ---- 2. tag predicate (slot 0 == "prod") ---- */
    const uint8_t *preds[1] = {(const uint8_t *)"prod"};
    uint64_t plens[1] = {4};
    CHECK(bydb_reset(s) == BYDB_OK, "reset");
    CHECK(bydb_consume_multi(s, INT64_MIN, INT64_MAX, preds, plens, 1) ==
              BYDB_OK,
          "consume_multi");
    CHECK(bydb_finalize(s, &r, 1) == BYDB_OK, "finalize pred");
    /* every block has the same rotation of envs; "prod" selects the g==0
     * rows of each block's rotation */
    int64_t want_pred_cnt = 0;
    for (int blk = 0; blk < 4; blk++)
        for (int i = 0; i < N; i++)
            if ((i / 7 + blk) % NGROUPS == 0) want_pred_cnt++;
    CHECK(r.count == want_pred_cnt, "pred count");

    /* ---- 3. per-row group-by on the env tag ---- */
    uint8_t dom_blob[64];
    uint64_t dom_offs[NGROUPS + 1];
    size_t doff = 0;
    for (int g = 0; g < NGROUPS; g++) {
        dom_offs[g] = doff;
        memcpy(dom_blob + doff, ENVS[g], strlen(ENVS[g]));
        doff += strlen(ENVS[g]);
    }
    dom_offs[NGROUPS] = doff;
    CHECK(bydb_agg_configure_by_tag(s, BYDB_VT_INT64, funcs, 0, dom_blob,
                                    dom_offs, NGROUPS, BYDB_MODE_ALL) ==
              BYDB_OK,
          "configure_by_tag");
    CHECK(bydb_consume(s, INT64_MIN, INT64_MAX, NULL, 0) == BYDB_OK,
          "consume grouped");
    bydb_result gr[NGROUPS];
    CHECK(bydb_finalize(s, gr, NGROUPS) == BYDB_OK, "finalize grouped");
    int64_t gcnt = 0;
    for (int g = 0; g < NGROUPS; g++) gcnt += gr[g].count;
    CHECK(gcnt == 4 * N, "grouped total count");
    /* block 0's env-0 rows land in... every block rotates the same tag
     * values; per-group counts are the same sums across the 4 rotations */

    /* ---- 4. AggModeMap partials + host Combine ---- */
    CHECK(bydb_agg_configure_by_tag(s, BYDB_VT_INT64, funcs, 0, dom_blob,
                                    dom_offs, NGROUPS, BYDB_MODE_MAP) ==
              BYDB_OK,
          "configure map");
    CHECK(bydb_consume(s, INT64_MIN, INT64_MAX, NULL, 0) == BYDB_OK,
          "consume map");
    bydb_partial parts[2 * NGROUPS];
    CHECK(bydb_finalize_partials(s, parts, NGROUPS) == BYDB_OK,
          "finalize_partials");
    /* simulate a second shard with identical partials, then Combine */
    memcpy(parts + NGROUPS, parts, sizeof(bydb_partial) * NGROUPS);
    bydb_result red[NGROUPS];
    CHECK(bydb_reduce_partials2(parts, 2, NGROUPS, BYDB_VT_INT64, 0, red) ==
              BYDB_OK,
          "reduce_partials2");
    for (int g = 0; g < NGROUPS; g++) {
        CHECK(red[g].count == 2 * gr[g].count, "reduce count");
        CHECK(red[g].sum_i == 2 * gr[g].sum_i, "reduce sum");
        CHECK(red[g].min_i == gr[g].min_i && red[g].max_i == gr[g].max_i,
              "reduce min/max");
    }

    /* ---- 5. map frames -> bydb_reduce_frames (replica dedup) ---- */
    bydb_frame_builder *fb = bydb_frame_builder_create(NGROUPS);
    CHECK(fb, "frame builder");
    int64_t shard_ids[NGROUPS] = {0, 0, 0};
    int64_t fsums[NGROUPS], fcnts[NGROUPS];
    uint8_t keydata[64];
    int64_t keylens[NGROUPS];
    size_t koff = 0;
    for (int g = 0; g < NGROUPS; g++) {
        fsums[g] = gr[g].sum_i;
        fcnts[g] = gr[g].count;
        memcpy(keydata + koff, ENVS[g], strlen(ENVS[g]));
        koff += strlen(ENVS[g]);
        keylens[g] = (int64_t)strlen(ENVS[g]);
    }
    CHECK(bydb_frame_add_i64(fb, 4, "shard_id", "", shard_ids, NULL) ==
              BYDB_OK,
          "frame shard col");
    CHECK(bydb_frame_add_str(fb, 5, "env", "meta", keydata, keylens) ==
              BYDB_OK,
          "frame key col");
    CHECK(bydb_frame_add_i64(fb, 6, "value", "", fsums, NULL) == BYDB_OK,
          "frame value col");
    CHECK(bydb_frame_add_i64(fb, 6, "value__agg_count", "", fcnts, NULL) ==
              BYDB_OK,
          "frame count col");
    CHECK(bydb_frame_finish(fb) == BYDB_OK, "frame finish");
    const uint8_t *fr = bydb_frame_data(fb);
    uint64_t frlen = bydb_frame_len(fb);
    /* decode check */
    bydb_frame_reader *frd = bydb_frame_open(fr, frlen);
    CHECK(frd && bydb_frame_nrows(frd) == NGROUPS &&
              bydb_frame_ncols(frd) == 4,
          "frame open");
    bydb_frame_close(frd);
    /* the same frame THREE times (two replicas) + a shard-1 copy: dedup
     * keeps one per (shard,key), so finals = 2x one shard's values */
    const uint8_t *frames[4] = {fr, fr, fr, NULL};
    uint64_t frlens[4] = {frlen, frlen, frlen, 0};
    bydb_frame_builder *fb2 = bydb_frame_builder_create(NGROUPS);
    int64_t shard1[NGROUPS] = {1, 1, 1};
    bydb_frame_add_i64(fb2, 4, "shard_id", "", shard1, NULL);
    bydb_frame_add_str(fb2, 5, "env", "meta", keydata, keylens);
    bydb_frame_add_i64(fb2, 6, "value", "", fsums, NULL);
    bydb_frame_add_i64(fb2, 6, "value__agg_count", "", fcnts, NULL);
    bydb_frame_finish(fb2);
    frames[3] = bydb_frame_data(fb2);
    frlens[3] = bydb_frame_len(fb2);
    bydb_reduce_spec specs[2] = {{2, BYDB_AGG_SUM}, {2, BYDB_AGG_MEAN}};
    int32_t key_cols[1] = {1};
    int64_t out_i[NGROUPS * 2];
    double out_f[NGROUPS * 2];
    uint8_t key_buf[256];
    uint64_t key_offs[NGROUPS + 1];
    int64_t ngot = 0;
    CHECK(bydb_reduce_frames(frames, frlens, 4, 0, key_cols, 1, specs, 2,
                             NGROUPS, out_i, out_f, key_buf,
                             sizeof key_buf, key_offs, &ngot) == BYDB_OK,
          "reduce_frames");
    CHECK(ngot == NGROUPS, "reduce_frames group count");
    for (int g = 0; g < NGROUPS; g++) {
        CHECK(out_i[g * 2] == 2 * gr[g].sum_i, "reduce_frames sum");
        int64_t mean = (2 * gr[g].sum_i) / (2 * gr[g].count);
        if (mean < 1) mean = 1;
        CHECK(out_i[g * 2 + 1] == mean, "reduce_frames mean clamp");
    }
    bydb_frame_builder_destroy(fb);
    bydb_frame_builder_destroy(fb2);

    /* ---- 6. BatchTop ordering ---- */
    int64_t top_idx[NGROUPS];
    int64_t top_n = 0;
    CHECK(bydb_top_groups(gr, NGROUPS, 0 /* sum_i */, 2, 0 /* desc */,
                          top_idx, &top_n) == BYDB_OK,
          "top_groups");
    CHECK(top_n == 2, "top n");
    CHECK(gr[top_idx[0]].sum_i >= gr[top_idx[1]].sum_i, "top order");

    /* ---- 7. on-disk part round trip ---- */
    const char *tagnames[1] = {"env"};
    CHECK(bydb_part_write_dir(b, "/tmp/cabi_part", "value", "default",
                              tagnames, 1) == BYDB_OK,
          "write_dir");
    bydb_part_builder *b2 = bydb_part_builder_create();
    CHECK(bydb_part_read_dir(b2, "/tmp/cabi_part") == BYDB_OK, "read_dir");
    CHECK(bydb_part_builder_n_blocks(b2) == nb, "read_dir blocks");
    /* reupload the reloaded part and re-check the scalar fold */
    uint64_t plen2 = bydb_part_builder_payload_len(b2);
    CHECK(bydb_part_reserve(s, plen2, nb) == BYDB_OK, "reserve 2");
    CHECK(bydb_part_append(s, bydb_part_builder_payload(b2), plen2,
                           bydb_part_builder_blocks(b2), nb) == BYDB_OK,
          "append 2");
    CHECK(bydb_agg_configure(s, BYDB_VT_INT64, funcs, 1, BYDB_MODE_ALL) ==
              BYDB_OK,
          "configure 2");
    CHECK(bydb_consume(s, INT64_MIN, INT64_MAX, NULL, 0) == BYDB_OK,
          "consume 2");
    CHECK(bydb_finalize(s, &r, 1) == BYDB_OK, "finalize 2");
    CHECK(r.count == 4 * N && r.sum_i == 4 * want_sum,
          "round-trip numbers");
    bydb_part_builder_destroy(b2);

    printf("cabi_check: all 7 sections OK (scalar, preds, group-by, "
           "partials reduce, frame reduce+dedup, top, part round trip)\n");
    bydb_session_destroy(s);
    bydb_part_builder_destroy(b);
    return 0;
}
