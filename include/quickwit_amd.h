/* quickwit_amd — C-ABI for the MI355X-native Quickwit leaf-search hot path.
 *
 * This is the drop-in boundary of SURVEY.md §8b: it replaces the work the
 * reference ships to its rayon pool inside `leaf_search_single_split`
 * (quickwit/quickwit-search/src/leaf.rs:899-959), speaking the protobuf
 * messages of quickwit-proto/protos/quickwit/search.proto verbatim:
 *   - qw_leaf_search:  LeafSearchRequest (search.proto:362) in,
 *                      LeafSearchResponse (search.proto:618) out.
 *   - qw_ctx mirrors SearcherContext (quickwit-search/src/service.rs:405-427)
 *     for the fields that affect results/limits.
 * A Rust host binds these with a plain extern "C" block (INTEGRATION.md shows
 * the binding that would sit under leaf.rs:900).
 *
 * Ownership: request/split buffers are caller-owned and immutable for the
 * call; response buffers are callee-allocated and freed with qw_buf_free.
 * Errors: negative status + UTF-8 message via qw_last_error, mapped onto the
 * reference's SearchError variants; per-split failures are reported INSIDE
 * the response (LeafSearchResponse.failed_splits, search.proto:626), exactly
 * as the reference does (leaf.rs:2143-2147).
 */
#ifndef QUICKWIT_AMD_H
#define QUICKWIT_AMD_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef struct qw_ctx qw_ctx;

typedef struct qw_buf {
    uint8_t* data;
    size_t len;
} qw_buf;

/* Status codes (negative = error), mapped onto quickwit SearchError
 * (quickwit-search/src/error.rs). */
enum qw_status {
    QW_OK = 0,
    QW_ERR_INVALID_ARGUMENT = -1, /* SearchError::InvalidArgument */
    QW_ERR_INVALID_QUERY = -2,    /* SearchError::InvalidQuery */
    QW_ERR_NOT_FOUND = -3,        /* unknown split id */
    QW_ERR_INTERNAL = -4,         /* SearchError::Internal */
    QW_ERR_NO_GPU = -5,           /* no HIP device: the product path never
                                     falls back to CPU */
    QW_ERR_BAD_SPLIT = -6,        /* malformed QWA1 container */
    QW_ERR_OVER_MEMORY_BUDGET = -7, /* HBM budget refusal: the permit-provider
                                     * memory gate (search_permit_provider.rs:
                                     * 43-110) mapped to a synchronous C-ABI —
                                     * refuse instead of queueing */
};

/* config_json mirrors the result-affecting subset of SearcherConfig
 * (quickwit-config/src/node_config/mod.rs:408-461) plus device placement:
 *   {"device": 0,                       // HIP device ordinal
 *    "aggregation_memory_limit": 500000000,
 *    "aggregation_bucket_limit": 65000}
 * NULL/empty means defaults (device 0). Creation succeeds without a GPU;
 * device init is lazy (first add_split/search touching the device fails with
 * QW_ERR_NO_GPU instead). */
qw_ctx* qw_ctx_create(const char* config_json);
void qw_ctx_free(qw_ctx* ctx);

/* Register a split (QWA1 container bytes) and upload its sections to device
 * HBM — the analog of the reference's warmup having populated the
 * ByteRangeCache (leaf.rs:302-377): after this call the split is "warm" and
 * qw_leaf_search times pure search, like cpu_search_microsecs
 * (leaf.rs:905-946). The data buffer may be released by the caller after
 * return. */
int32_t qw_ctx_add_split(qw_ctx* ctx, const char* split_id,
                         const uint8_t* data, size_t len);
int32_t qw_ctx_remove_split(qw_ctx* ctx, const char* split_id);

/* LeafSearchRequest protobuf in, LeafSearchResponse protobuf out.
 * Splits referenced by LeafRequestRef.split_offsets must have been added.
 * Synchronous; internally one HIP stream per split. */
int32_t qw_leaf_search(qw_ctx* ctx, const uint8_t* leaf_search_request_pb,
                       size_t len, qw_buf* response_pb_out);

void qw_buf_free(qw_buf* buf);

/* fetch_docs phase 2 (quickwit-search/src/fetch_docs.rs; root.rs:903):
 * FetchDocsRequest protobuf in (search.proto:674), FetchDocsResponse out
 * (:695). Fetches the stored documents for the given PartialHits from the
 * splits' row store (zlib blocks + doc->block index);
 * leaf_json is the ingested document's canonical JSON. Snippets
 * (snippet_request) are not produced in round 1. Hits are returned grouped
 * by (split_id, doc_id) — the root matches them by partial_hit. */
int32_t qw_fetch_docs(qw_ctx* ctx, const uint8_t* fetch_docs_request_pb,
                      size_t len, qw_buf* response_pb_out);

/* leaf_list_terms (quickwit-search/src/list_terms.rs:211-322):
 * LeafListTermsRequest protobuf in (search.proto:732), LeafListTermsResponse
 * out (search.proto:745). Per split: sorted term-dictionary range scan over
 * [start_key, end_key) limited to max_hits, then k-merge + dedup across
 * splits. Term bytes are the RAW term value bytes (declared deviation: the
 * reference prefixes tantivy's serialized-Term header, an engine-internal
 * encoding with no analog here — DESIGN.md §7). Host-only (term dictionaries
 * stay on CPU); splits must have been added. */
int32_t qw_leaf_list_terms(qw_ctx* ctx, const uint8_t* leaf_list_terms_request_pb,
                           size_t len, qw_buf* response_pb_out);

/* Root-side helpers replacing the reference's tantivy-side merge
 * (collector.rs:832-861 merge_fruits / aggregation merge): merge N
 * LeafSearchResponse protobufs (top-K with reference tie-breaks + QAGG1
 * aggregation blobs + counters) into one, honoring the SearchRequest's
 * sort/max_hits. */
int32_t qw_merge_leaf_responses(const uint8_t* search_request_pb,
                                size_t search_request_len,
                                const uint8_t** response_pbs,
                                const size_t* response_lens, size_t n,
                                qw_buf* merged_out);

/* Finalize an intermediate aggregation blob (QAGG1) into the ES-shaped JSON
 * the reference's root returns for the same aggregation request. */
int32_t qw_finalize_agg_to_json(const uint8_t* blob, size_t len,
                                const char* agg_request_json,
                                qw_buf* json_out);

/* Last error message for this ctx (UTF-8, valid until next call on ctx).
 * For ctx-less entry points (merge/finalize), pass NULL to read the
 * thread-local message. */
const char* qw_last_error(const qw_ctx* ctx);

/* ---- instrumentation (bench/roofline; not part of the reference surface) */

/* Accumulated device time (HIP events on the launch stream) and launch count
 * for a named kernel since the last reset. Returns QW_ERR_NOT_FOUND for an
 * unknown name. Known names: "union_bm25", "range_filter", "column_agg",
 * "topk_select". */
int32_t qw_kernel_stats(qw_ctx* ctx, const char* kernel_name,
                        double* total_ms, uint64_t* launches);
void qw_kernel_stats_reset(qw_ctx* ctx);
int32_t qw_ctx_device_sync(qw_ctx* ctx);

/* HBM footprint accounting (SearchPermitProvider memory-budget analog):
 * used = split images + scratch/result buffers + cached hitset bitmaps;
 * budget = configured "hbm_memory_budget" or 97% of device free memory at
 * first use; splits_bytes = resident split images alone. */
int32_t qw_ctx_memory_stats(qw_ctx* ctx, uint64_t* used_bytes,
                            uint64_t* budget_bytes, uint64_t* splits_bytes);

/* Negative/absence term cache counters (leaf.rs:761-827): probes that
 * short-circuited a split (hits), probes over required terms that found no
 * cached absence (misses), and resident keys. Replaces the reference's
 * prometheus-side SplitSearchOutcomeCounters for this cache. */
int32_t qw_absence_cache_stats(qw_ctx* ctx, uint64_t* hits, uint64_t* misses,
                               uint64_t* entries);

/* Library version + build arch, e.g. "quickwit_amd 0.1 gfx950". */
const char* qw_version(void);

#ifdef __cplusplus
} /* extern "C" */
#endif

#endif /* QUICKWIT_AMD_H */
