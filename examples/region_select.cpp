// region_select.cpp — a standalone C++ embedding of the engine, re-creating
// exactly the driver loop a baikalStore region runs for one SELECT
// (Region::select_normal, src/store/region.cpp:3166-3216):
//
//   build plan -> ExecNode::create_tree -> open -> while(!eos) get_next -> close
//
// Usage: region_select <file.parquet>
//   runs  SELECT g, COUNT(*), SUM(v), MIN(w) FROM t WHERE v < 80 GROUP BY g
//   and   SELECT g, v FROM t ORDER BY v, g LIMIT 5
// over the ingested file and prints the result rows. No Python anywhere:
// this is the product path an embedder links (libbkgpu.so + headers).
#include <cstdio>
#include <cstring>
#include <string>
#include <vector>

#include "../include/bk_exec.h"
#include "../include/bk_arrow.h"

static int run_agg(BkgTable* table) {
    BkPlanNodeDesc plan[3];
    memset(plan, 0, sizeof plan);
    /* pre-order: AGG -> FILTER -> SCAN (plan.proto:495-510 flattening) */
    plan[0].node_type = BK_AGG_NODE;
    plan[0].num_children = 1;
    plan[0].limit = -1;
    plan[0].n_group = 1;
    plan[0].group_cols[0] = 0;
    plan[0].n_aggs = 3;
    plan[0].aggs[0] = {BK_AGG_COUNT_STAR, -1};
    plan[0].aggs[1] = {BK_AGG_SUM, 2};
    plan[0].aggs[2] = {BK_AGG_MIN, 1};
    plan[0].expected_groups = 1 << 8;
    plan[1].node_type = BK_WHERE_FILTER_NODE;
    plan[1].num_children = 1;
    plan[1].limit = -1;
    plan[1].n_conjuncts = 1;
    plan[1].conjuncts[0].col = 2;
    plan[1].conjuncts[0].op = BK_OP_LT;
    plan[1].conjuncts[0].cmp_type = BK_INT64;
    plan[1].conjuncts[0].lit_i = 80;
    plan[2].node_type = BK_SCAN_NODE;
    plan[2].limit = -1;
    plan[2].table = table;

    BkExecTree* t = bkexec_create_tree(plan, 3);
    if (!t) { fprintf(stderr, "create_tree: %s\n", bkgpu_last_error()); return 1; }
    if (bkexec_open(t) < 0) { fprintf(stderr, "open failed\n"); return 1; }
    int ns = bkexec_n_slots(t);
    std::vector<int32_t> tag(16 * ns);
    std::vector<int64_t> vi(16 * ns);
    std::vector<double> vd(16 * ns);
    std::vector<uint8_t> nul(16 * ns);
    int eos = 0;
    int64_t total = 0;
    char word[64];
    while (!eos) {
        int64_t n = bkexec_get_next(t, 16, tag.data(), vi.data(), vd.data(),
                                    nul.data(), &eos);
        if (n < 0) { fprintf(stderr, "get_next failed\n"); return 1; }
        for (int64_t r = 0; r < n; r++, total++) {
            printf("row %lld:", (long long)total);
            for (int s = 0; s < ns; s++) {
                size_t i = (size_t)r * ns + s;
                if (nul[i]) { printf(" NULL"); continue; }
                if (tag[i] == BK_DOUBLE) printf(" %.6g", vd[i]);
                else if (tag[i] == BK_STRING) {
                    if (bkgpu_table_dict_word(table, s == 0 ? 0 : 1,
                                              vi[i], word, sizeof word) >= 0)
                        printf(" %s", word);
                    else
                        printf(" code:%lld", (long long)vi[i]);
                } else printf(" %lld", (long long)vi[i]);
            }
            printf("\n");
        }
    }
    printf("scan_rows=%lld filtered=%lld returned=%lld\n",
           (long long)bkexec_num_scan_rows(t),
           (long long)bkexec_num_filter_rows(t),
           (long long)bkexec_num_rows_returned(t));
    bkexec_close(t);
    return 0;
}

static int run_sort(BkgTable* table) {
    BkPlanNodeDesc plan[2];
    memset(plan, 0, sizeof plan);
    plan[0].node_type = BK_SORT_NODE;
    plan[0].num_children = 1;
    plan[0].limit = 5;
    plan[0].n_order = 2;
    plan[0].order[0] = {2, 1, 1, 0};   /* v asc */
    plan[0].order[1] = {0, 1, 1, 0};   /* g asc */
    plan[0].n_out_cols = 2;
    plan[0].out_cols[0] = 0;
    plan[0].out_cols[1] = 2;
    plan[1].node_type = BK_SCAN_NODE;
    plan[1].limit = -1;
    plan[1].table = table;

    BkExecTree* t = bkexec_create_tree(plan, 2);
    if (!t) { fprintf(stderr, "create_tree: %s\n", bkgpu_last_error()); return 1; }
    if (bkexec_open(t) < 0) { fprintf(stderr, "sort open failed\n"); return 1; }
    int ns = bkexec_n_slots(t);
    std::vector<int32_t> tag(8 * ns);
    std::vector<int64_t> vi(8 * ns);
    std::vector<double> vd(8 * ns);
    std::vector<uint8_t> nul(8 * ns);
    int eos = 0;
    char word[64];
    printf("top-5 by (v, g):\n");
    while (!eos) {
        int64_t n = bkexec_get_next(t, 8, tag.data(), vi.data(), vd.data(),
                                    nul.data(), &eos);
        if (n < 0) { fprintf(stderr, "sort get_next failed\n"); return 1; }
        for (int64_t r = 0; r < n; r++) {
            size_t i0 = (size_t)r * ns;
            if (bkgpu_table_dict_word(table, 0, vi[i0], word, sizeof word) < 0)
                snprintf(word, sizeof word, "code:%lld", (long long)vi[i0]);
            printf("  g=%s v=%lld\n", word, (long long)vi[i0 + 1]);
        }
    }
    bkexec_close(t);
    return 0;
}

static int run_window(BkgTable* table) {
    /* SELECT g, v, ROW_NUMBER() OVER w, SUM(v) OVER w, RANK() OVER w
     * FROM t WINDOW w AS (PARTITION BY g ORDER BY v)  -- non-frame mode */
    BkPlanNodeDesc plan[2];
    memset(plan, 0, sizeof plan);
    plan[0].node_type = BK_WINDOW_NODE;
    plan[0].num_children = 1;
    plan[0].limit = 6;                  /* print the first rows only */
    plan[0].part_col = 0;
    plan[0].n_order = 1;
    plan[0].order[0] = {2, 1, 1, 0};
    plan[0].n_winfns = 3;
    plan[0].winfns[0] = {BK_WIN_ROW_NUMBER, -1, 0, 0, 0, 0, 0.0};
    plan[0].winfns[1] = {BK_WIN_SUM, 2, 0, 0, 0, 0, 0.0};
    plan[0].winfns[2] = {BK_WIN_RANK, -1, 0, 0, 0, 0, 0.0};
    plan[0].n_out_cols = 2;
    plan[0].out_cols[0] = 0;
    plan[0].out_cols[1] = 2;
    plan[1].node_type = BK_SCAN_NODE;
    plan[1].limit = -1;
    plan[1].table = table;

    BkExecTree* t = bkexec_create_tree(plan, 2);
    if (!t) { fprintf(stderr, "create_tree: %s\n", bkgpu_last_error()); return 1; }
    if (bkexec_open(t) < 0) { fprintf(stderr, "window open failed\n"); return 1; }
    int ns = bkexec_n_slots(t);
    std::vector<int32_t> tag(8 * ns);
    std::vector<int64_t> vi(8 * ns);
    std::vector<double> vd(8 * ns);
    std::vector<uint8_t> nul(8 * ns);
    int eos = 0;
    char word[64];
    printf("window (PARTITION BY g ORDER BY v): first rows\n");
    while (!eos) {
        int64_t n = bkexec_get_next(t, 8, tag.data(), vi.data(), vd.data(),
                                    nul.data(), &eos);
        if (n < 0) { fprintf(stderr, "window get_next failed\n"); return 1; }
        for (int64_t r = 0; r < n; r++) {
            size_t i0 = (size_t)r * ns;
            if (bkgpu_table_dict_word(table, 0, vi[i0], word, sizeof word) < 0)
                snprintf(word, sizeof word, "code:%lld", (long long)vi[i0]);
            printf("  g=%s v=%lld row_number=%lld sum_over_g=%lld rank=%lld\n",
                   word, (long long)vi[i0 + 1], (long long)vi[i0 + 2],
                   (long long)vi[i0 + 3], (long long)vi[i0 + 4]);
        }
    }
    bkexec_close(t);
    return 0;
}

static int run_distinct(BkgTable* table) {
    /* SELECT g, COUNT(*), COUNT(DISTINCT v) FROM t GROUP BY g — the AggNode
     * applies the reference's multi-distinct rewrite internally
     * (agg_node.cpp:247-258); a high expected_groups routes level 1 through
     * the sort-dedup path (bkdedup.inc) with automatic fallback. */
    BkPlanNodeDesc plan[2];
    memset(plan, 0, sizeof plan);
    plan[0].node_type = BK_AGG_NODE;
    plan[0].num_children = 1;
    plan[0].limit = 5;
    plan[0].n_group = 1;
    plan[0].group_cols[0] = 0;
    plan[0].n_aggs = 2;
    plan[0].aggs[0] = {BK_AGG_COUNT_STAR, -1};
    plan[0].aggs[1] = {BK_AGG_COUNT_DISTINCT, 2};
    plan[0].expected_groups = 1 << 18;   /* high: sort-dedup level 1 */
    plan[1].node_type = BK_SCAN_NODE;
    plan[1].limit = -1;
    plan[1].table = table;

    BkExecTree* t = bkexec_create_tree(plan, 2);
    if (!t) { fprintf(stderr, "create_tree: %s\n", bkgpu_last_error()); return 1; }
    if (bkexec_open(t) < 0) { fprintf(stderr, "distinct open failed\n"); return 1; }
    int ns = bkexec_n_slots(t);
    std::vector<int32_t> tag(8 * ns);
    std::vector<int64_t> vi(8 * ns);
    std::vector<double> vd(8 * ns);
    std::vector<uint8_t> nul(8 * ns);
    int eos = 0;
    char word[64];
    printf("count-distinct (GROUP BY g): first rows\n");
    while (!eos) {
        int64_t n = bkexec_get_next(t, 8, tag.data(), vi.data(), vd.data(),
                                    nul.data(), &eos);
        if (n < 0) { fprintf(stderr, "distinct get_next failed\n"); return 1; }
        for (int64_t r = 0; r < n; r++) {
            size_t i0 = (size_t)r * ns;
            if (bkgpu_table_dict_word(table, 0, vi[i0], word, sizeof word) < 0)
                snprintf(word, sizeof word, "code:%lld", (long long)vi[i0]);
            printf("  g=%s count=%lld count_distinct_v=%lld\n",
                   word, (long long)vi[i0 + 1], (long long)vi[i0 + 2]);
        }
    }
    bkexec_close(t);
    return 0;
}

/* run the same GROUP BY tree and serialize the result batch as an Arrow
 * IPC stream — the vectorized-result bytes Region::select returns to the
 * frontend (region.cpp:2905-2918), written here by the from-scratch
 * serializer (include/bk_arrow.h); pyarrow round-trips the file in
 * tests/test_parquet.py. */
static int run_agg_arrow(BkgTable* table, const char* out_path) {
    BkPlanNodeDesc plan[3];
    memset(plan, 0, sizeof plan);
    plan[0].node_type = BK_AGG_NODE;
    plan[0].num_children = 1;
    plan[0].limit = -1;
    plan[0].n_group = 1;
    plan[0].group_cols[0] = 0;
    plan[0].n_aggs = 3;
    plan[0].aggs[0] = {BK_AGG_COUNT_STAR, -1};
    plan[0].aggs[1] = {BK_AGG_SUM, 2};
    plan[0].aggs[2] = {BK_AGG_MIN, 1};
    plan[0].expected_groups = 1 << 8;
    plan[1].node_type = BK_WHERE_FILTER_NODE;
    plan[1].num_children = 1;
    plan[1].limit = -1;
    plan[1].n_conjuncts = 1;
    plan[1].conjuncts[0].col = 2;
    plan[1].conjuncts[0].op = BK_OP_LT;
    plan[1].conjuncts[0].cmp_type = BK_INT64;
    plan[1].conjuncts[0].lit_i = 80;
    plan[2].node_type = BK_SCAN_NODE;
    plan[2].limit = -1;
    plan[2].table = table;

    BkExecTree* t = bkexec_create_tree(plan, 3);
    if (!t) { fprintf(stderr, "create_tree: %s\n", bkgpu_last_error()); return 1; }
    if (bkexec_open(t) < 0) { fprintf(stderr, "open failed\n"); return 1; }
    int ns = bkexec_n_slots(t);
    std::vector<int32_t> tag(64 * ns);
    std::vector<int64_t> vi(64 * ns);
    std::vector<double> vd(64 * ns);
    std::vector<uint8_t> nul(64 * ns);
    /* collect the whole result columnar (the batch the store would hand to
     * SerializeRecordBatch) */
    std::vector<std::vector<int64_t>> ci(ns);
    std::vector<std::vector<double>> cd(ns);
    std::vector<std::vector<uint8_t>> cv(ns);
    std::vector<int32_t> ctype(ns, BK_INT64);
    int eos = 0;
    while (!eos) {
        int64_t n = bkexec_get_next(t, 64, tag.data(), vi.data(), vd.data(),
                                    nul.data(), &eos);
        if (n < 0) { fprintf(stderr, "get_next failed\n"); return 1; }
        for (int64_t r = 0; r < n; r++)
            for (int s = 0; s < ns; s++) {
                size_t i = (size_t)r * ns + s;
                if (!nul[i]) ctype[s] = tag[i];
                ci[s].push_back(vi[i]);
                cd[s].push_back(vd[i]);
                cv[s].push_back(nul[i] ? 0 : 1);
            }
    }
    int64_t nrows = ci.empty() ? 0 : (int64_t)ci[0].size();
    /* dictionary words for string slots (codes -> words) */
    std::vector<std::vector<std::string>> wstore(ns);
    std::vector<std::vector<const char*>> wptr(ns);
    std::vector<BkArrowCol> cols(ns);
    std::vector<std::string> names(ns);
    std::vector<const char*> nameptr(ns);
    std::vector<std::vector<int32_t>> codes32(ns);
    for (int s = 0; s < ns; s++) {
        names[s] = "c" + std::to_string(s);
        nameptr[s] = names[s].c_str();
        cols[s].col_type = ctype[s];
        cols[s].valid = cv[s].data();
        cols[s].words = nullptr;
        cols[s].nwords = 0;
        if (ctype[s] == BK_DOUBLE) {
            cols[s].data = cd[s].data();
        } else if (ctype[s] == BK_STRING) {
            int64_t maxc = -1;
            for (int64_t r = 0; r < nrows; r++)
                if (cv[s][r] && ci[s][r] > maxc) maxc = ci[s][r];
            char word[256];
            /* slot 0 = the group key (table col 0); other string slots in
             * this plan come from MIN(w) on table col 1 (same mapping the
             * row printer uses) */
            int src = s == 0 ? 0 : 1;
            for (int64_t c = 0; c <= maxc; c++) {
                if (bkgpu_table_dict_word(table, src, c, word,
                                          sizeof word) >= 0)
                    wstore[s].push_back(word);
                else
                    wstore[s].push_back("");
            }
            for (auto& w : wstore[s]) wptr[s].push_back(w.c_str());
            codes32[s].resize(nrows);
            for (int64_t r = 0; r < nrows; r++)
                codes32[s][r] = (int32_t)ci[s][r];
            cols[s].data = codes32[s].data();
            cols[s].words = wptr[s].data();
            cols[s].nwords = (int64_t)wstore[s].size();
        } else {
            cols[s].data = ci[s].data();
        }
    }
    void* buf = nullptr;
    int64_t blen = 0;
    if (bk_arrow_ipc_stream(ns, cols.data(), nrows, nameptr.data(), &buf,
                            &blen) != 0) {
        fprintf(stderr, "arrow serialize failed\n");
        return 1;
    }
    FILE* f = fopen(out_path, "wb");
    if (!f) { fprintf(stderr, "cannot write %s\n", out_path); return 1; }
    fwrite(buf, 1, (size_t)blen, f);
    fclose(f);
    bk_arrow_free(buf);
    printf("arrow ipc: %lld rows, %lld bytes -> %s\n", (long long)nrows,
           (long long)blen, out_path);
    bkexec_close(t);
    return 0;
}

int main(int argc, char** argv) {
    if (argc != 2 && argc != 3) {
        fprintf(stderr, "usage: %s <file.parquet> [out.arrow]\n", argv[0]);
        return 2;
    }
    if (bkgpu_device_count() < 1) {
        fprintf(stderr, "no HIP device (this example needs a GPU)\n");
        return 3;
    }
    BkgTable* table = (BkgTable*)bkgpu_table_from_parquet(argv[1]);
    if (!table) {
        fprintf(stderr, "ingest failed: %s\n", bkparquet_last_error());
        return 1;
    }
    printf("ingested %lld rows x %d cols\n",
           (long long)bkgpu_table_nrows(table), bkgpu_table_ncols(table));
    int rc = run_agg(table);
    if (rc == 0) rc = run_sort(table);
    if (rc == 0) rc = run_window(table);
    if (rc == 0) rc = run_distinct(table);
    if (rc == 0 && argc == 3) rc = run_agg_arrow(table, argv[2]);
    bkgpu_table_free(table);
    return rc;
}
