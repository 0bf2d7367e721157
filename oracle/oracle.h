/*
 * ORACLE — TEST INFRASTRUCTURE ONLY.
 *
 * Bit-exact CPU restatement of the dgraph-io/dgraph posting-list set-algebra
 * hot path:
 *   - /root/reference/algo/uidlist.go   (whole file: IntersectWith{,Lin,Jump,Bin},
 *     binIntersect, IntersectSorted, MergeSorted/internalMergeSort, Difference,
 *     IndexOf, ApplyFilter, IntersectCompressedWith{,LinJump,Bin})
 *   - /root/reference/algo/heap.go      (uint64Heap used by the merge)
 *   - /root/reference/codec/codec.go    (Encoder/packBlock/Add/Done, Decoder
 *     UnpackBlock/Seek/SeekToBlock/LinearSeek/PeekNextBase/Next, ApproxLen,
 *     ExactLen, Encode, Decode, match32MSB)
 *   - struct layouts from /root/reference/protos/pb.proto:22,379-400
 *
 * Third-party arithmetic boundary: github.com/dgryski/go-groupvarint
 * v0.0.0-20230630160417-2bfb7969fb3c (reference go.mod:20) — the 4x-uint32
 * group-varint codec called at codec/codec.go:87,189-190.  That library is
 * NOT present in this container; orc_gv_encode4/orc_gv_decode4 below restate
 * the published Group Varint format (1 control byte, bit-pairs [1:0],[3:2],
 * [5:4],[7:6] = byte-length-1 of v0..v3, values little-endian).  The
 * reference pins results across this boundary only as encode->decode
 * roundtrips (codec_test.go:37-58,74-111,190-217,306-334), i.e. decoded-u64
 * parity is pinned, byte-level pack parity is unpinned.
 *
 * ONLY tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg may
 * call, link or execute this code.  The product path (dgraph_amd/ +
 * libuidalgo.so) must never route through it.
 */
#ifndef UIDALGO_ORACLE_H
#define UIDALGO_ORACLE_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- pb.UidBlock / pb.UidPack mirrors (pb.proto:379-400) ---- */
typedef struct {
    uint64_t base;       /* pb.UidBlock.base */
    uint32_t num_uids;   /* pb.UidBlock.num_uids (includes base) */
    uint32_t deltas_len; /* len(pb.UidBlock.deltas) */
    uint8_t *deltas;     /* group-varint encoded deltas */
} orc_block;

typedef struct {
    uint32_t block_size; /* pb.UidPack.block_size */
    size_t n_blocks;
    orc_block *blocks;
} orc_pack;

/* ---- group-varint (go-groupvarint Encode4/Decode4/BytesUsed) ---- */
extern const uint8_t orc_gv_bytes_used[256];
size_t orc_gv_encode4(uint8_t *buf, const uint32_t v[4]); /* returns bytes written (1+sum) */
void orc_gv_decode4(uint32_t out[4], const uint8_t *p);

/* ---- codec (codec/codec.go) ---- */
/* codec.Encode(uids, blockSize)  codec.go:393.  Caller frees with orc_pack_free. */
orc_pack *orc_encode(const uint64_t *uids, size_t n, int block_size);
void orc_pack_free(orc_pack *p);

/* codec.ApproxLen / ExactLen  codec.go:418,427 */
size_t orc_pack_approx_len(const orc_pack *p);
size_t orc_pack_exact_len(const orc_pack *p);

/* Decoder (codec.go:139).  uids/n_uids expose the Go decoder's d.uids view. */
typedef struct {
    const orc_pack *pack;
    int block_idx;     /* d.blockIdx */
    uint64_t *uids;    /* current view (owned buffer + trim offset applied) */
    size_t n_uids;
    /* internals */
    uint64_t *buf;
    size_t buf_cap;
} orc_dec;

enum { ORC_SEEK_START = 0, ORC_SEEK_CURRENT = 1 }; /* codec.go:24-29 */

void orc_dec_init(orc_dec *d, const orc_pack *pack);
void orc_dec_free(orc_dec *d);
/* Each returns the new d.uids view (also stored in d). */
void orc_dec_unpack_block(orc_dec *d);         /* codec.go:154 */
void orc_dec_seek(orc_dec *d, uint64_t uid, int whence);          /* codec.go:279 */
void orc_dec_seek_to_block(orc_dec *d, uint64_t uid, int whence); /* codec.go:219 */
void orc_dec_linear_seek(orc_dec *d, uint64_t seek);              /* codec.go:349 */
uint64_t orc_dec_peek_next_base(const orc_dec *d);                /* codec.go:362 */
void orc_dec_next(orc_dec *d);                                    /* codec.go:376 */
int orc_dec_valid(const orc_dec *d);                              /* codec.go:371 */

/* codec.Decode(pack, seek)  codec.go:444 — caller buffer of ApproxLen capacity. */
size_t orc_decode(const orc_pack *p, uint64_t seek, uint64_t *out);

/* ---- algo (algo/uidlist.go) ----
 * Output buffers are caller-owned: capacity >= min(n,m) for intersect,
 * >= n for difference, >= sum(lens) for merge. Returns written count. */

/* IntersectWithLin/Jump/Bin keep the reference's (i,k)/return contracts. */
void orc_intersect_with_lin(const uint64_t *u, size_t n, const uint64_t *v, size_t m,
                            uint64_t *o, size_t *o_n, size_t *i_out, size_t *k_out); /* :170 */
void orc_intersect_with_jump(const uint64_t *u, size_t n, const uint64_t *v, size_t m,
                             uint64_t *o, size_t *o_n, size_t *i_out, size_t *k_out); /* :195 */
size_t orc_intersect_with_bin(const uint64_t *d, size_t ld, const uint64_t *q, size_t lq,
                              uint64_t *o, size_t *o_n); /* :226, returns maxq */

/* IntersectWith dispatch (ratio 100/500)  :142 */
size_t orc_intersect_with(const uint64_t *u, size_t n, const uint64_t *v, size_t m,
                          uint64_t *out);

/* IntersectSorted  :297 (stable sort by length; fold smallest-first) */
size_t orc_intersect_sorted(const uint64_t *const *lists, const size_t *lens, size_t k,
                            uint64_t *out);

/* MergeSorted  :448 → heap merge :392-433 (dedup'd k-way union) */
size_t orc_merge_sorted(const uint64_t *const *lists, const size_t *lens, size_t k,
                        uint64_t *out);

/* Difference  :332 */
size_t orc_difference(const uint64_t *u, size_t n, const uint64_t *v, size_t m,
                      uint64_t *out);

/* IndexOf  :546 */
int64_t orc_index_of(const uint64_t *u, size_t n, uint64_t uid);
/* ApplyFilter  :21 (in-place; mask = precomputed f(uid,i)) */
size_t orc_apply_filter(uint64_t *u, size_t n, const uint8_t *mask);

/* IntersectCompressedWith dispatch (linVsBinRatio=10)  :33 */
size_t orc_intersect_compressed_with(const orc_pack *pack, uint64_t after_uid,
                                     const uint64_t *v, size_t m, uint64_t *out);
/* The two variants, taking a positioned decoder like the reference. */
void orc_intersect_compressed_with_lin_jump(orc_dec *dec, const uint64_t *v, size_t m,
                                            uint64_t *o, size_t *o_n); /* :63 */
void orc_intersect_compressed_with_bin(orc_dec *dec, const uint64_t *q, size_t lq,
                                       uint64_t *o, size_t *o_n); /* :87 */

/* ---- batched CPU baseline (OpenMP across pairs; bench.py cpu_baseline leg) ----
 * threads_used (nullable): distinct threads that ran >=1 pair. */
void orc_intersect_batch_cpu(int n_pairs,
                             const uint64_t *const *us, const size_t *ns,
                             const uint64_t *const *vs, const size_t *ms,
                             uint64_t *const *outs, size_t *out_ns,
                             int n_threads, int *threads_used);
int orc_omp_max_threads(void);

#ifdef __cplusplus
}
#endif
#endif /* UIDALGO_ORACLE_H */
