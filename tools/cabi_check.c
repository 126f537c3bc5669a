/* tools/cabi_check.c — pure-C consumer of the drop-in C-ABI, standing in
 * for the cgo binding a BanyanDB maintainer would write.
 * Builds a tiny part with the builder API, uploads it, folds sum/count/
 * min/max on the GPU and checks the numbers.  Compile:
 *   gcc -std=c99 -I include tools/cabi_check.c -L banyandb_amd -lbydb_gpu \
 *       -Wl,-rpath,banyandb_amd -o cabi_check            */
#include <stdio.h>
#include <stdlib.h>

#include "bydb_gpu.h"

#define N 1000
#define T0 1700000000000000000LL

int main(void) {
    int64_t ts[N], ver[N], vals[N];
    int64_t want_sum = 0;
    for (int i = 0; i < N; i++) {
        ts[i] = T0 + (int64_t)i * 1000000LL;
        ver[i] = 1;
        vals[i] = (int64_t)(i * 37 - 5000);
        want_sum += vals[i];
    }
    bydb_part_builder *b = bydb_part_builder_create();
    if (!b) { fprintf(stderr, "builder create failed\n"); return 1; }
    if (bydb_part_builder_add_block_i64(b, 1, ts, ver, vals, N, 0) != BYDB_OK) {
        fprintf(stderr, "add_block: %s\n", bydb_part_builder_error(b));
        return 1;
    }
    bydb_session *s = bydb_session_create(0);
    if (!s) { fprintf(stderr, "no GPU session\n"); return 2; }
    uint64_t plen = bydb_part_builder_payload_len(b);
    int64_t nb = bydb_part_builder_n_blocks(b);
    if (bydb_part_reserve(s, plen, nb) != BYDB_OK ||
        bydb_part_append(s, bydb_part_builder_payload(b), plen,
                         bydb_part_builder_blocks(b), nb) != BYDB_OK) {
        fprintf(stderr, "upload: %s\n", bydb_last_error(s));
        return 1;
    }
    uint32_t funcs = (1u << BYDB_AGG_SUM) | (1u << BYDB_AGG_COUNT) |
                     (1u << BYDB_AGG_MIN) | (1u << BYDB_AGG_MAX);
    if (bydb_agg_configure(s, BYDB_VT_INT64, funcs, 1, BYDB_MODE_ALL) != BYDB_OK ||
        bydb_consume(s, INT64_MIN, INT64_MAX, NULL, 0) != BYDB_OK) {
        fprintf(stderr, "consume: %s\n", bydb_last_error(s));
        return 1;
    }
    bydb_result r;
    if (bydb_finalize(s, &r, 1) != BYDB_OK) {
        fprintf(stderr, "finalize: %s\n", bydb_last_error(s));
        return 1;
    }
    int ok = r.count == N && r.sum_i == want_sum && r.min_i == -5000 &&
             r.max_i == (int64_t)(N - 1) * 37 - 5000;
    printf("cabi_check: count=%lld sum=%lld min=%lld max=%lld -> %s\n",
           (long long)r.count, (long long)r.sum_i, (long long)r.min_i,
           (long long)r.max_i, ok ? "OK" : "MISMATCH");
    bydb_session_destroy(s);
    bydb_part_builder_destroy(b);
    return ok ? 0 : 1;
}
