/*
 * uidalgo — C-ABI boundary of the MI355X-native posting-list set-algebra
 * engine (libuidalgo.so).
 *
 * This is the drop-in surface for dgraph's `algo` package (reference
 * interfaces each export replaces are cited below; see INTEGRATION.md for the
 * cgo binding a dgraph maintainer would add).  All memory is caller-owned,
 * errors are int status codes, a ua_ctx wraps one GPU + one HIP stream and is
 * callable from any thread (calls on one ctx are serialized by its stream).
 *
 * Two levels:
 *  - host-pointer convenience calls that mirror algo's signatures 1:1
 *    (upload + compute + download; PCIe-inclusive), and
 *  - device-resident batched calls (`*_dev`), the batched engine that a
 *    worker/task.go:834-971 fan-out would feed: one grid per
 *    SubGraph's thousands of per-key list ops.
 *
 * No torch types, no C++ types: plain pointers and sizes only.
 */
#ifndef UIDALGO_H
#define UIDALGO_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- status codes ---- */
enum {
    UA_OK = 0,
    UA_ERR_HIP = 1,      /* HIP runtime error (see ua_last_hip_error) */
    UA_ERR_NOMEM = 2,    /* allocation failed */
    UA_ERR_INVALID = 3,  /* bad argument (e.g. block with num_uids > 256) */
    UA_ERR_NO_GPU = 4,   /* no HIP device visible */
};

const char *ua_strerror(int code);
int ua_version(void);

/* ---- context ---- */
typedef struct ua_ctx ua_ctx;
int ua_ctx_create(ua_ctx **out, int device);
void ua_ctx_destroy(ua_ctx *ctx);

/* ---- device memory (for callers without their own allocator) ---- */
int ua_dev_alloc(ua_ctx *, uint64_t bytes, void **dptr);
int ua_dev_free(ua_ctx *, void *dptr);
int ua_h2d(ua_ctx *, void *dst_dev, const void *src_host, uint64_t bytes);
int ua_d2h(ua_ctx *, void *dst_host, const void *src_dev, uint64_t bytes);
int ua_sync(ua_ctx *);

/* ---- engine stats (HIP-event timing of the dominant kernels, for the
 * roofline leg of bench.py: kernel_ms is accumulated on the engine's own
 * stream across launches since the last reset) ---- */
int ua_stats_reset(ua_ctx *);
int ua_stats_get(ua_ctx *, uint64_t *n_launches, double *kernel_ms,
                 uint64_t *bytes_algorithmic);

/* ---- pb.UidPack mirror (pb.proto:379-400) ---- */
typedef struct {
    uint64_t base;          /* pb.UidBlock.base */
    uint32_t num_uids;      /* pb.UidBlock.num_uids (includes base) */
    uint32_t deltas_len;
    const uint8_t *deltas;  /* group-varint deltas */
} ua_block;

typedef struct {
    uint32_t block_size;    /* pb.UidPack.block_size */
    uint64_t n_blocks;
    const ua_block *blocks;
} ua_pack;

/* Flattened pack in device memory — the engine's native layout. */
typedef struct {
    uint32_t block_size;
    uint64_t n_blocks;
    const uint64_t *bases;      /* device [n_blocks] */
    const uint32_t *num_uids;   /* device [n_blocks] */
    const uint64_t *delta_offs; /* device [n_blocks+1], offsets into deltas */
    const uint8_t *deltas;      /* device blob */
    uint64_t total_uids;        /* codec.ExactLen */
} ua_dpack;

/* ---- host-side codec (replaces codec.Encode, codec.go:393) ---- */
typedef struct ua_owned_pack ua_owned_pack;
int ua_encode(const uint64_t *uids, uint64_t n, uint32_t block_size,
              ua_owned_pack **out);
const ua_pack *ua_owned_pack_view(ua_owned_pack *);
void ua_owned_pack_free(ua_owned_pack *);

uint64_t ua_pack_exact_len(const ua_pack *);  /* codec.ExactLen  codec.go:427 */
uint64_t ua_pack_approx_len(const ua_pack *); /* codec.ApproxLen codec.go:418 */

/* Flatten a host pack into the engine layout (arrays caller-allocated:
 * bases[n_blocks], num_uids[n_blocks], delta_offs[n_blocks+1],
 * deltas_blob[total deltas bytes]).  Query sizes first. */
int ua_pack_flat_sizes(const ua_pack *, uint64_t *n_blocks,
                       uint64_t *deltas_bytes, uint64_t *total_uids);
int ua_pack_flatten(const ua_pack *, uint64_t *bases, uint32_t *num_uids,
                    uint64_t *delta_offs, uint8_t *deltas_blob);

/* Decode a pack on the GPU (replaces codec.Decode, codec.go:444).
 * out: device buffer, capacity total_uids. */
int ua_decode_dev(ua_ctx *, const ua_dpack *, uint64_t seek_uid,
                  uint64_t *out, uint64_t *out_n);

/* GPU codec.Encode (codec.go:393): uids (device, sorted) -> flattened pack
 * arrays (device, caller-allocated worst case: bases/num_uids/delta_offs for
 * max_blocks = n blocks (+1 for delta_offs), deltas capacity >= 6*n + 24
 * bytes).  Byte-identical to the host encoder / reference format.
 * block_size <= 256 (the engine's block bound); 0 packs per-uid blocks like
 * the reference. */
int ua_encode_dev(ua_ctx *, const uint64_t *uids, uint64_t n,
                  uint32_t block_size, uint64_t *bases, uint32_t *num_uids,
                  uint64_t *delta_offs, uint8_t *deltas,
                  uint64_t *n_blocks_out, uint64_t *deltas_bytes_out);

/* ---- batched device-resident set algebra ----
 * All pointers in ua_dpair are DEVICE pointers; the descriptor array itself
 * and out_lens live on the host.  Inputs are sorted uint64 lists; results are
 * bit-exact vs the reference on duplicate-free inputs (the parity domain
 * pinned by uidlist_test.go:394,536-542 — see DESIGN.md §semantics).
 * On inputs that violate the contract (duplicates / unsorted), results are
 * unspecified — like the reference's bin variants — but all writes stay
 * within each pair's out capacity (the engine clamps; the reference is
 * memory-safe on such inputs and so is this ABI). */
typedef struct {
    const uint64_t *u; /* device */
    uint64_t n;
    const uint64_t *v; /* device */
    uint64_t m;
    uint64_t *out;     /* device; capacity: intersect >= min(n,m),
                        *                   merge >= n+m, difference >= n */
} ua_dpair;

/* batched algo.IntersectWith (uidlist.go:142) */
int ua_intersect_batch_dev(ua_ctx *, const ua_dpair *pairs, int n_pairs,
                           uint64_t *out_lens);

/* Prepared batch (the repeated-query path): descriptor upload and the
 * merge-path tile partition are done ONCE at create; each run is launches
 * only, and the FIRST run additionally caches the per-thread merge-path
 * splits (4 B/tile-thread, device) that every later run loads instead of
 * re-searching.  The pairs' device contents must not change between
 * create and the runs (both caches depend on them; violating this gives
 * unspecified results but stays memory-safe); out capacities must fit the
 * op (intersect >= min(n,m), merge >= n+m, difference >= n). */
typedef struct ua_batch ua_batch;
enum { UA_OP_INTERSECT = 0, UA_OP_MERGE = 1, UA_OP_DIFFERENCE = 2 };
int ua_batch_create(ua_ctx *, const ua_dpair *pairs, int n_pairs, ua_batch **out);
int ua_batch_run(ua_ctx *, ua_batch *, int op, uint64_t *out_lens);
/* n_runs passes enqueued back-to-back with ONE sync (the repeated-query
 * serving shape); out_lens are the final pass's lengths (all passes over an
 * immutable batch produce identical results). */
int ua_batch_run_n(ua_ctx *, ua_batch *, int op, int n_runs,
                   uint64_t *out_lens);
void ua_batch_destroy(ua_ctx *, ua_batch *);
/* batched pairwise algo.MergeSorted semantics (dedup union, uidlist.go:448) */
int ua_merge_batch_dev(ua_ctx *, const ua_dpair *pairs, int n_pairs,
                       uint64_t *out_lens);
/* batched algo.Difference (uidlist.go:332) */
int ua_difference_batch_dev(ua_ctx *, const ua_dpair *pairs, int n_pairs,
                            uint64_t *out_lens);

/* batched duplicate-KEEPING merge of sorted runs (no dedup, unlike
 * MergeSorted): out capacity exactly n+m; out_lens[i] = n+m.  The building
 * block of the segmented sort. */
int ua_merge_all_batch_dev(ua_ctx *, const ua_dpair *pairs, int n_pairs,
                           uint64_t *out_lens);

/* Batched segmented sort, u64 ascending, duplicates kept, in place.
 * tmp: device scratch with capacity n per segment.  The engine primitive
 * behind the sort path's UidMatrix shapes (worker/sort.go:48,139,189 —
 * SURVEY.md §8f row 3) and for unsorted ingest ahead of Encode. */
typedef struct {
    uint64_t *data; /* device, n elements; sorted ascending on return */
    uint64_t n;
    uint64_t *tmp;  /* device scratch, capacity n */
} ua_dseg;
int ua_sort_segments_dev(ua_ctx *, const ua_dseg *segs, int n_segs);

/* k-way fold algo.IntersectSorted (uidlist.go:297): lists/lens host arrays of
 * device pointers; out device, capacity min(lens). */
int ua_intersect_k_dev(ua_ctx *, const uint64_t *const *lists,
                       const uint64_t *lens, int k, uint64_t *out,
                       uint64_t *out_n);
/* k-way algo.MergeSorted (uidlist.go:448): pairwise tree on device.
 * out device, capacity sum(lens). */
int ua_merge_k_dev(ua_ctx *, const uint64_t *const *lists,
                   const uint64_t *lens, int k, uint64_t *out,
                   uint64_t *out_n);

/* batched algo.IndexOf (uidlist.go:546): queries/out device. */
int ua_index_of_batch_dev(ua_ctx *, const uint64_t *u, uint64_t n,
                          const uint64_t *queries, uint64_t nq, int64_t *out);

/* batched algo.ApplyFilter (uidlist.go:21; callers worker/task.go:1403,
 * query/query.go:1431): order-preserving mask compaction.  The reference
 * takes a Go closure f(uid, i); across the C-ABI the filter arrives as a
 * precomputed per-element byte mask (callers evaluate per-uid predicates
 * upstream).  out may equal u (in-place, the reference's shape). */
typedef struct {
    const uint64_t *u;    /* device */
    uint64_t n;
    const uint8_t *mask;  /* device, n bytes; nonzero = keep */
    uint64_t *out;        /* device, capacity >= n; may alias u */
} ua_dfilter;
int ua_apply_filter_batch_dev(ua_ctx *, const ua_dfilter *tasks, int n_tasks,
                              uint64_t *out_lens);

/* fused decode+intersect: algo.IntersectCompressedWith (uidlist.go:33) over a
 * device pack; v device; out device, capacity min(total_uids, m). */
int ua_intersect_packed_dev(ua_ctx *, const ua_dpack *, uint64_t after_uid,
                            const uint64_t *v, uint64_t m, uint64_t *out,
                            uint64_t *out_n);

/* Batched fused decode+intersect — ONE grid over a whole per-key fan-out
 * (worker/task.go:834-971 handleUidPostings: thousands of keys' packs, each
 * intersected with its filter list, often one shared q.UidList).  All packs
 * are concatenated into one flat block arena (bases/num_uids/delta_offs/
 * deltas, like ua_dpack); pack_block_base[n_packs+1] delimits each pack's
 * blocks.  Per-pack v/out/after are device pointers in the tasks array. */
typedef struct {
    const uint64_t *v; /* device */
    uint64_t m;
    uint64_t *out;     /* device, capacity >= min(pack uids, m) */
    uint64_t after_uid;
} ua_ptask;

int ua_intersect_packed_batch_dev(ua_ctx *, const uint64_t *bases,
                                  const uint32_t *num_uids,
                                  const uint64_t *delta_offs,
                                  const uint8_t *deltas,
                                  const uint64_t *pack_block_base /* host [n_packs+1] */,
                                  int n_packs, const ua_ptask *tasks /* host */,
                                  uint64_t *out_lens);

/* Prepared pack fan-out (a standing query plan): task/boundary upload and
 * workspace sizing once at create; each run is launches + one lens read.
 * The flat pack arena and the tasks' v/out buffers are caller-owned device
 * memory that must outlive the handle; their contents may change between
 * runs (the plan caches no data-dependent state, unlike ua_batch). */
typedef struct ua_pbatch ua_pbatch;
int ua_pbatch_create(ua_ctx *, const uint64_t *bases, const uint32_t *num_uids,
                     const uint64_t *delta_offs, const uint8_t *deltas,
                     const uint64_t *pack_block_base /* host [n_packs+1] */,
                     int n_packs, const ua_ptask *tasks /* host */,
                     ua_pbatch **out);
int ua_pbatch_run(ua_ctx *, ua_pbatch *, uint64_t *out_lens);
void ua_pbatch_destroy(ua_ctx *, ua_pbatch *);

/* ---- host-pointer convenience (upload+compute+download; mirrors the algo
 * package signatures 1:1 for the cgo shim) ---- */
int ua_intersect(ua_ctx *, const uint64_t *u, uint64_t n, const uint64_t *v,
                 uint64_t m, uint64_t *out, uint64_t *out_n);  /* IntersectWith */
int ua_intersect_k(ua_ctx *, const uint64_t *const *lists, const uint64_t *lens,
                   int k, uint64_t *out, uint64_t *out_n);     /* IntersectSorted */
int ua_merge_k(ua_ctx *, const uint64_t *const *lists, const uint64_t *lens,
               int k, uint64_t *out, uint64_t *out_n);         /* MergeSorted */
int ua_difference(ua_ctx *, const uint64_t *u, uint64_t n, const uint64_t *v,
                  uint64_t m, uint64_t *out, uint64_t *out_n); /* Difference */
int64_t ua_index_of(const uint64_t *u, uint64_t n, uint64_t uid); /* IndexOf:
                  * one log(n) probe — host binary search like the reference;
                  * the batched GPU form is ua_index_of_batch_dev */
int ua_apply_filter(ua_ctx *, uint64_t *u /* compacted in place */, uint64_t n,
                    const uint8_t *mask, uint64_t *out_n); /* ApplyFilter */
int ua_intersect_packed(ua_ctx *, const ua_pack *, uint64_t after_uid,
                        const uint64_t *v, uint64_t m, uint64_t *out,
                        uint64_t *out_n);            /* IntersectCompressedWith */

#ifdef __cplusplus
}
#endif
#endif /* UIDALGO_H */
