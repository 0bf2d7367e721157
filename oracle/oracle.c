/*
 * ORACLE — TEST INFRASTRUCTURE ONLY (see oracle.h header comment).
 *
 * Bit-exact C restatement of dgraph-io/dgraph algo/uidlist.go +
 * codec/codec.go.  Every function cites the reference file:line it follows.
 * Only tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg may
 * use this library; the product path must never route through it.
 */
#include "oracle.h"

#include <stdlib.h>
#include <string.h>

#ifdef _OPENMP
#include <omp.h>
#endif

/* ===================== group-varint (go-groupvarint) =====================
 * Standard Group Varint: control byte bit-pairs [1:0],[3:2],[5:4],[7:6] hold
 * (byte-length - 1) of v0..v3; values little-endian. */

const uint8_t orc_gv_bytes_used[256] = {
#define B(t) (uint8_t)(1 + ((t) & 3) + 1 + (((t) >> 2) & 3) + 1 + (((t) >> 4) & 3) + 1 + (((t) >> 6) & 3) + 1)
#define R4(t) B(t), B(t + 1), B(t + 2), B(t + 3)
#define R16(t) R4(t), R4(t + 4), R4(t + 8), R4(t + 12)
#define R64(t) R16(t), R16(t + 16), R16(t + 32), R16(t + 48)
    R64(0), R64(64), R64(128), R64(192)
#undef R64
#undef R16
#undef R4
#undef B
};

size_t orc_gv_encode4(uint8_t *buf, const uint32_t v[4]) {
    uint8_t *p = buf + 1;
    uint8_t tag = 0;
    for (int i = 0; i < 4; i++) {
        uint32_t x = v[i];
        int len = 1 + (x > 0xffu) + (x > 0xffffu) + (x > 0xffffffu);
        tag = (uint8_t)(tag | ((len - 1) << (2 * i)));
        for (int b = 0; b < len; b++) {
            *p++ = (uint8_t)(x & 0xff);
            x >>= 8;
        }
    }
    buf[0] = tag;
    return (size_t)(p - buf);
}

void orc_gv_decode4(uint32_t out[4], const uint8_t *p) {
    uint8_t tag = p[0];
    const uint8_t *q = p + 1;
    for (int i = 0; i < 4; i++) {
        int len = ((tag >> (2 * i)) & 3) + 1;
        uint32_t x = 0;
        for (int b = 0; b < len; b++) x |= ((uint32_t)q[b]) << (8 * b);
        q += len;
        out[i] = x;
    }
}

/* ===================== small helpers ===================== */

/* sort.Search(n, f) with f = a[i] >= key  (first index where a[i] >= key) */
static size_t lower_bound_u64(const uint64_t *a, size_t n, uint64_t key) {
    size_t lo = 0, hi = n;
    while (lo < hi) {
        size_t mid = lo + (hi - lo) / 2;
        if (a[mid] >= key) hi = mid;
        else lo = mid + 1;
    }
    return lo;
}

/* first index where a[i] > key */
static size_t upper_bound_u64(const uint64_t *a, size_t n, uint64_t key) {
    size_t lo = 0, hi = n;
    while (lo < hi) {
        size_t mid = lo + (hi - lo) / 2;
        if (a[mid] > key) hi = mid;
        else lo = mid + 1;
    }
    return lo;
}

/* ===================== codec: Encoder (codec.go:36-136,393) ===================== */

/* codec.go:469 match32MSB, bitMask codec.go:32 */
static int match32msb(uint64_t a, uint64_t b) {
    const uint64_t mask = 0xffffffff00000000ull;
    return (a & mask) == (b & mask);
}

typedef struct {
    int block_size;
    orc_pack *pack;
    uint64_t *uids;
    size_t n_uids, cap_uids;
    uint8_t *buf; /* e.buf */
    size_t buf_len, buf_cap;
    size_t cap_blocks;
} orc_enc;

static void enc_buf_append(orc_enc *e, const uint8_t *p, size_t n) {
    if (e->buf_len + n > e->buf_cap) {
        e->buf_cap = (e->buf_cap ? e->buf_cap * 2 : 64);
        if (e->buf_cap < e->buf_len + n) e->buf_cap = e->buf_len + n;
        e->buf = (uint8_t *)realloc(e->buf, e->buf_cap);
    }
    memcpy(e->buf + e->buf_len, p, n);
    e->buf_len += n;
}

/* codec.go:57 packBlock */
static void enc_pack_block(orc_enc *e) {
    if (e->n_uids == 0) return;

    if (e->pack->n_blocks >= e->cap_blocks) {
        e->cap_blocks = e->cap_blocks ? e->cap_blocks * 2 : 8;
        e->pack->blocks = (orc_block *)realloc(e->pack->blocks, e->cap_blocks * sizeof(orc_block));
    }
    orc_block *blk = &e->pack->blocks[e->pack->n_blocks];
    blk->base = e->uids[0];
    blk->num_uids = (uint32_t)e->n_uids;

    uint64_t last = e->uids[0];
    size_t off = 1, rem = e->n_uids - 1; /* e.uids = e.uids[1:] (codec.go:71) */

    e->buf_len = 0; /* e.buf.Reset() */
    uint8_t gbuf[17];
    uint32_t tmp[4];
    for (;;) {
        for (int i = 0; i < 4; i++) {
            if ((size_t)i >= rem) {
                tmp[i] = 0; /* zero padding, codec.go:80 */
            } else {
                tmp[i] = (uint32_t)(e->uids[off + (size_t)i] - last);
                last = e->uids[off + (size_t)i];
            }
        }
        size_t sz = orc_gv_encode4(gbuf, tmp);
        enc_buf_append(e, gbuf, sz);
        if (rem <= 4) break; /* codec.go:91 (always encodes >= 1 group, incl. pad-only) */
        off += 4;
        rem -= 4;
    }

    blk->deltas_len = (uint32_t)e->buf_len;
    blk->deltas = (uint8_t *)malloc(e->buf_len ? e->buf_len : 1);
    memcpy(blk->deltas, e->buf, e->buf_len);
    e->pack->n_blocks++;
    e->n_uids = 0;
}

/* codec.go:107 Add */
static void enc_add(orc_enc *e, uint64_t uid) {
    if (!e->pack) {
        e->pack = (orc_pack *)calloc(1, sizeof(orc_pack));
        e->pack->block_size = (uint32_t)e->block_size;
    }
    if (e->n_uids > 0 && !match32msb(e->uids[e->n_uids - 1], uid)) {
        enc_pack_block(e);
    }
    if (e->n_uids >= e->cap_uids) {
        e->cap_uids = e->cap_uids ? e->cap_uids * 2 : 16;
        e->uids = (uint64_t *)realloc(e->uids, e->cap_uids * sizeof(uint64_t));
    }
    e->uids[e->n_uids++] = uid;
    if ((long long)e->n_uids >= (long long)e->block_size) {
        enc_pack_block(e);
    }
}

/* codec.go:393 Encode (+ Done :130).  n==0 returns an empty pack (Go returns a
 * nil *pb.UidPack; every consumer treats both as the empty list). */
orc_pack *orc_encode(const uint64_t *uids, size_t n, int block_size) {
    orc_enc e;
    memset(&e, 0, sizeof(e));
    e.block_size = block_size;
    for (size_t i = 0; i < n; i++) enc_add(&e, uids[i]);
    enc_pack_block(&e); /* Done */
    free(e.uids);
    free(e.buf);
    if (!e.pack) {
        e.pack = (orc_pack *)calloc(1, sizeof(orc_pack));
        e.pack->block_size = (uint32_t)block_size;
    }
    return e.pack;
}

void orc_pack_free(orc_pack *p) {
    if (!p) return;
    for (size_t i = 0; i < p->n_blocks; i++) free(p->blocks[i].deltas);
    free(p->blocks);
    free(p);
}

/* codec.go:418 ApproxLen */
size_t orc_pack_approx_len(const orc_pack *p) {
    if (!p) return 0;
    return p->n_blocks * (size_t)p->block_size;
}

/* codec.go:427 ExactLen (NumUids includes the base UID) */
size_t orc_pack_exact_len(const orc_pack *p) {
    if (!p) return 0;
    size_t num = 0;
    for (size_t i = 0; i < p->n_blocks; i++) num += p->blocks[i].num_uids;
    return num;
}

/* ===================== codec: Decoder (codec.go:139-384) ===================== */

void orc_dec_init(orc_dec *d, const orc_pack *pack) {
    memset(d, 0, sizeof(*d));
    d->pack = pack; /* Decoder{Pack: pack}: blockIdx=0, uids empty */
}

void orc_dec_free(orc_dec *d) {
    free(d->buf);
    memset(d, 0, sizeof(*d));
}

/* codec.go:154 UnpackBlock */
void orc_dec_unpack_block(orc_dec *d) {
    d->n_uids = 0;
    d->uids = d->buf;
    if (!d->pack || d->block_idx < 0 || (size_t)d->block_idx >= d->pack->n_blocks) return;
    const orc_block *b = &d->pack->blocks[d->block_idx];

    size_t need = (size_t)b->num_uids + 4; /* groups of 4 may overshoot before truncation */
    if (d->buf_cap < need) {
        d->buf_cap = need;
        d->buf = (uint64_t *)realloc(d->buf, d->buf_cap * sizeof(uint64_t));
        d->uids = d->buf;
    }

    uint64_t last = b->base;
    d->buf[d->n_uids++] = last;

    const uint8_t *enc = b->deltas;
    size_t enc_len = b->deltas_len;
    uint8_t padded[17];
    uint32_t t4[4];
    while (d->n_uids < b->num_uids) {
        const uint8_t *p = enc;
        if (enc_len < 17) { /* 17-byte pad workaround, codec.go:178-187 */
            memset(padded, 0, sizeof(padded));
            memcpy(padded, enc, enc_len);
            p = padded;
        }
        orc_gv_decode4(t4, p);
        size_t used = orc_gv_bytes_used[p[0]];
        if (used > enc_len) used = enc_len; /* valid packs never hit this */
        enc += used;
        enc_len -= used;
        for (int i = 0; i < 4; i++) {
            last = last + (uint64_t)t4[i];
            d->buf[d->n_uids++] = last;
        }
    }
    d->n_uids = b->num_uids; /* d.uids = d.uids[:block.NumUids] (codec.go:198) */
}

/* codec.go:362 PeekNextBase */
uint64_t orc_dec_peek_next_base(const orc_dec *d) {
    size_t bidx = (size_t)(d->block_idx + 1);
    if (d->pack && bidx < d->pack->n_blocks) return d->pack->blocks[bidx].base;
    return UINT64_MAX;
}

/* codec.go:371 Valid */
int orc_dec_valid(const orc_dec *d) {
    return d->pack && (size_t)d->block_idx < d->pack->n_blocks;
}

/* codec.go:376 Next */
void orc_dec_next(orc_dec *d) {
    d->block_idx++;
    orc_dec_unpack_block(d);
}

/* codec.go:279 Seek */
void orc_dec_seek(orc_dec *d, uint64_t uid, int whence) {
    if (!d->pack) {
        d->n_uids = 0;
        return;
    }
    d->block_idx = 0;
    if (uid == 0) {
        orc_dec_unpack_block(d);
        return;
    }
    const orc_pack *pack = d->pack;
    size_t nb = pack->n_blocks;

    size_t lo = 0, hi = nb; /* sort.Search over blocks (codec.go:300) */
    while (lo < hi) {
        size_t mid = lo + (hi - lo) / 2;
        uint64_t base = pack->blocks[mid].base;
        int ge = (whence == ORC_SEEK_START) ? (base >= uid) : (base > uid);
        if (ge) hi = mid;
        else lo = mid + 1;
    }
    size_t idx = lo;

    if (idx == 0) {
        orc_dec_unpack_block(d);
        return;
    }
    if (idx < nb && pack->blocks[idx].base == uid) {
        d->block_idx = (int)idx;
        orc_dec_unpack_block(d);
        return;
    }
    d->block_idx = (int)idx - 1;
    orc_dec_unpack_block(d);

    /* in-block search (codec.go:317-329) */
    size_t uidx = (whence == ORC_SEEK_START) ? lower_bound_u64(d->uids, d->n_uids, uid)
                                             : upper_bound_u64(d->uids, d->n_uids, uid);
    if (uidx < d->n_uids) {
        d->uids += uidx; /* d.uids = d.uids[uidx:] */
        d->n_uids -= uidx;
        return;
    }
    orc_dec_next(d);
}

/* codec.go:219 SeekToBlock */
void orc_dec_seek_to_block(orc_dec *d, uint64_t uid, int whence) {
    if (!d->pack) {
        d->n_uids = 0;
        return;
    }
    int prev = d->block_idx;
    d->block_idx = 0;
    if (uid == 0) {
        orc_dec_unpack_block(d);
        return;
    }
    const orc_pack *pack = d->pack;
    size_t nb = pack->n_blocks;
    if ((size_t)prev >= nb) {
        /* Decoder exhausted (blockIdx == len(Blocks)).  The reference PANICS
         * here — index out of range at codec.go:230 Blocks[prevBlockIdx] —
         * reachable via IntersectCompressedWithBin when afterUID > every
         * pack uid.  Result is undefined upstream; both this oracle and the
         * GPU engine define it as the mathematically consistent empty set
         * ({x in pack : x >= after} ∩ v = ∅). */
        d->n_uids = 0;
        return;
    }
    if (prev > 0 && uid < pack->blocks[prev].base) prev = 0;

    /* sort.Search over Blocks[prev:] (codec.go:245) */
    size_t cnt = nb - (size_t)prev;
    size_t lo = 0, hi = cnt;
    while (lo < hi) {
        size_t mid = lo + (hi - lo) / 2;
        uint64_t base = pack->blocks[mid + (size_t)prev].base;
        int ge = (whence == ORC_SEEK_START) ? (base >= uid) : (base > uid);
        if (ge) hi = mid;
        else lo = mid + 1;
    }
    size_t idx = lo + (size_t)prev;

    if (idx == 0) {
        orc_dec_unpack_block(d);
        return;
    }
    if (idx < nb && pack->blocks[idx].base == uid) {
        d->block_idx = (int)idx;
        orc_dec_unpack_block(d);
        return;
    }
    d->block_idx = (int)idx - 1;
    if (d->block_idx != prev) {
        orc_dec_unpack_block(d); /* codec.go:260-262 */
    }
    if (d->n_uids > 0 && uid <= d->uids[d->n_uids - 1]) { /* codec.go:264 */
        return;
    }
    orc_dec_next(d);
}

/* codec.go:349 LinearSeek */
void orc_dec_linear_seek(orc_dec *d, uint64_t seek) {
    for (;;) {
        uint64_t v = orc_dec_peek_next_base(d);
        if (seek < v) break;
        d->block_idx++;
    }
    orc_dec_unpack_block(d);
}

/* codec.go:444 Decode */
size_t orc_decode(const orc_pack *p, uint64_t seek, uint64_t *out) {
    size_t n = 0;
    orc_dec d;
    orc_dec_init(&d, p);
    orc_dec_seek(&d, seek, ORC_SEEK_START);
    while (d.n_uids > 0) {
        memcpy(out + n, d.uids, d.n_uids * sizeof(uint64_t));
        n += d.n_uids;
        orc_dec_next(&d);
    }
    orc_dec_free(&d);
    return n;
}

/* ===================== algo (algo/uidlist.go) ===================== */

#define ORC_JUMP 32          /* uidlist.go:17 */
#define ORC_LIN_VS_BIN 10    /* uidlist.go:18 */

/* uidlist.go:170 IntersectWithLin */
void orc_intersect_with_lin(const uint64_t *u, size_t n, const uint64_t *v, size_t m,
                            uint64_t *o, size_t *o_n, size_t *i_out, size_t *k_out) {
    size_t i = 0, k = 0;
    while (i < n && k < m) {
        uint64_t uid = u[i], vid = v[k];
        if (uid > vid) {
            for (k = k + 1; k < m && v[k] < uid; k++) {}
        } else if (uid == vid) {
            o[(*o_n)++] = uid;
            k++;
            i++;
        } else {
            for (i = i + 1; i < n && u[i] < vid; i++) {}
        }
    }
    *i_out = i;
    *k_out = k;
}

/* uidlist.go:195 IntersectWithJump */
void orc_intersect_with_jump(const uint64_t *u, size_t n, const uint64_t *v, size_t m,
                             uint64_t *o, size_t *o_n, size_t *i_out, size_t *k_out) {
    size_t i = 0, k = 0;
    while (i < n && k < m) {
        uint64_t uid = u[i], vid = v[k];
        if (uid == vid) {
            o[(*o_n)++] = uid;
            k++;
            i++;
        } else if (k + ORC_JUMP < m && uid > v[k + ORC_JUMP]) {
            k += ORC_JUMP;
        } else if (i + ORC_JUMP < n && vid > u[i + ORC_JUMP]) {
            i += ORC_JUMP;
        } else if (uid > vid) {
            for (k = k + 1; k < m && v[k] < uid; k++) {}
        } else {
            for (i = i + 1; i < n && u[i] < vid; i++) {}
        }
    }
    *i_out = i;
    *k_out = k;
}

/* uidlist.go:254 binIntersect (signed indices: Go's midd-- may go to -1) */
static void bin_intersect(const uint64_t *d, int64_t ld, const uint64_t *q, int64_t lq,
                          uint64_t *f, size_t *fn) {
    if (ld == 0 || lq == 0) return;
    int64_t midq = lq / 2;
    uint64_t qval = q[midq];
    int64_t midd = (int64_t)lower_bound_u64(d, (size_t)ld, qval);

    if (midd > midq) {
        bin_intersect(d, midd, q, midq, f, fn);
    } else {
        bin_intersect(q, midq, d, midd, f, fn);
    }

    if (midd >= ld) return;
    if (d[midd] == qval) {
        f[(*fn)++] = qval;
    } else {
        midd--;
    }

    const uint64_t *dd = d + midd + 1;
    int64_t ldd = ld - (midd + 1);
    const uint64_t *qq = q + midq + 1;
    int64_t lqq = lq - (midq + 1);
    if (ldd > lqq) {
        bin_intersect(dd, ldd, qq, lqq, f, fn);
    } else {
        bin_intersect(qq, lqq, dd, ldd, f, fn);
    }
}

/* uidlist.go:226 IntersectWithBin — returns maxq */
size_t orc_intersect_with_bin(const uint64_t *d, size_t ld, const uint64_t *q, size_t lq,
                              uint64_t *o, size_t *o_n) {
    if (ld < lq) {
        const uint64_t *t = d;
        d = q;
        q = t;
        size_t ts = ld;
        ld = lq;
        lq = ts;
    }
    if (ld == 0 || lq == 0 || d[ld - 1] < q[0] || q[lq - 1] < d[0]) return 0;

    uint64_t val = d[0];
    size_t minq = lower_bound_u64(q, lq, val);
    val = d[ld - 1];
    size_t maxq = upper_bound_u64(q, lq, val);

    bin_intersect(d, (int64_t)ld, q + minq, (int64_t)(maxq - minq), o, o_n);
    return maxq;
}

/* uidlist.go:142 IntersectWith (ratio dispatch 100/500) */
size_t orc_intersect_with(const uint64_t *u, size_t n, const uint64_t *v, size_t m,
                          uint64_t *out) {
    size_t nn = n, mm = m;
    if (nn > mm) {
        size_t t = nn;
        nn = mm;
        mm = t;
    }
    size_t o_n = 0;
    if (nn == 0) nn = 1;
    double ratio = (double)mm / (double)nn;
    size_t i, k;
    if (ratio < 100) {
        orc_intersect_with_lin(u, n, v, m, out, &o_n, &i, &k);
    } else if (ratio < 500) {
        orc_intersect_with_jump(u, n, v, m, out, &o_n, &i, &k);
    } else {
        orc_intersect_with_bin(u, n, v, m, out, &o_n);
    }
    return o_n;
}

/* uidlist.go:297 IntersectSorted (stable sort by length, fold smallest-first) */
size_t orc_intersect_sorted(const uint64_t *const *lists, const size_t *lens, size_t k,
                            uint64_t *out) {
    if (k == 0) return 0;
    /* stable insertion sort of indices by length (Go sort.Slice is unstable,
     * but on duplicate-free inputs the fold result is order-independent) */
    size_t *ord = (size_t *)malloc(k * sizeof(size_t));
    for (size_t i = 0; i < k; i++) ord[i] = i;
    for (size_t i = 1; i < k; i++) {
        size_t key = ord[i];
        size_t j = i;
        while (j > 0 && lens[ord[j - 1]] > lens[key]) {
            ord[j] = ord[j - 1];
            j--;
        }
        ord[j] = key;
    }
    if (k == 1) {
        memcpy(out, lists[ord[0]], lens[ord[0]] * sizeof(uint64_t));
        size_t r = lens[ord[0]];
        free(ord);
        return r;
    }
    size_t cap = lens[ord[0]];
    uint64_t *tmp = (uint64_t *)malloc((cap ? cap : 1) * sizeof(uint64_t));
    size_t cur = orc_intersect_with(lists[ord[0]], lens[ord[0]],
                                    lists[ord[1]], lens[ord[1]], out);
    for (size_t j = 2; j < k; j++) {
        cur = orc_intersect_with(out, cur, lists[ord[j]], lens[ord[j]], tmp);
        memcpy(out, tmp, cur * sizeof(uint64_t));
        if (cur == 0) break; /* uidlist.go:324 early exit */
    }
    free(tmp);
    free(ord);
    return cur;
}

/* uidlist.go:392-433 internalMergeSort(WithBuffer) via uint64Heap (heap.go).
 * MergeSorted (:448/:465) produces the same dedup'd union. */
typedef struct {
    uint64_t val;
    size_t list_idx;
} orc_helem;

static void heap_sift_down(orc_helem *h, size_t n, size_t i) {
    for (;;) {
        size_t l = 2 * i + 1, r = 2 * i + 2, s = i;
        if (l < n && h[l].val < h[s].val) s = l;
        if (r < n && h[r].val < h[s].val) s = r;
        if (s == i) break;
        orc_helem t = h[i];
        h[i] = h[s];
        h[s] = t;
        i = s;
    }
}

static void heap_sift_up(orc_helem *h, size_t i) {
    while (i > 0) {
        size_t p = (i - 1) / 2;
        if (h[p].val <= h[i].val) break;
        orc_helem t = h[i];
        h[i] = h[p];
        h[p] = t;
        i = p;
    }
}

size_t orc_merge_sorted(const uint64_t *const *lists, const size_t *lens, size_t k,
                        uint64_t *out) {
    if (k == 0) return 0;
    orc_helem *h = (orc_helem *)malloc((k ? k : 1) * sizeof(orc_helem));
    size_t *idx = (size_t *)calloc(k ? k : 1, sizeof(size_t));
    size_t hn = 0;
    for (size_t i = 0; i < k; i++) {
        if (lists[i] == NULL || lens[i] == 0) continue;
        h[hn].val = lists[i][0];
        h[hn].list_idx = i;
        heap_sift_up(h, hn);
        hn++;
    }
    size_t o = 0;
    uint64_t last = 0;
    while (hn > 0) {
        orc_helem me = h[0];
        if (o == 0 || me.val != last) { /* dedup, uidlist.go:417 */
            out[o++] = me.val;
            last = me.val;
        }
        size_t li = me.list_idx;
        if (idx[li] >= lens[li] - 1) {
            h[0] = h[hn - 1]; /* heap.Pop */
            hn--;
            heap_sift_down(h, hn, 0);
        } else {
            idx[li]++;
            h[0].val = lists[li][idx[li]];
            heap_sift_down(h, hn, 0); /* heap.Fix(h, 0) */
        }
    }
    free(h);
    free(idx);
    return o;
}

/* uidlist.go:332 Difference (keeps u's duplicates) */
size_t orc_difference(const uint64_t *u, size_t n, const uint64_t *v, size_t m,
                      uint64_t *out) {
    if (u == NULL || v == NULL) return 0; /* Go nil check :333 */
    size_t i = 0, k = 0, o = 0;
    while (i < n && k < m) {
        uint64_t uid = u[i], vid = v[k];
        if (uid < vid) {
            while (i < n && u[i] < vid) {
                out[o++] = u[i];
                i++;
            }
        } else if (uid == vid) {
            i++;
            k++;
        } else {
            for (k = k + 1; k < m && v[k] < uid; k++) {}
        }
    }
    while (i < n && k >= m) {
        out[o++] = u[i];
        i++;
    }
    return o;
}

/* uidlist.go:21 ApplyFilter: in-place keep-where-f compaction; the Go
 * closure f(uid, i) is a precomputed mask here (the C-ABI's contract). */
size_t orc_apply_filter(uint64_t *u, size_t n, const uint8_t *mask) {
    size_t o = 0;
    for (size_t i = 0; i < n; i++) {
        if (mask[i]) u[o++] = u[i];
    }
    return o;
}

/* uidlist.go:546 IndexOf */
int64_t orc_index_of(const uint64_t *u, size_t n, uint64_t uid) {
    size_t i = lower_bound_u64(u, n, uid);
    if (i < n && u[i] == uid) return (int64_t)i;
    return -1;
}

/* uidlist.go:63 IntersectCompressedWithLinJump */
void orc_intersect_compressed_with_lin_jump(orc_dec *dec, const uint64_t *v, size_t m,
                                            uint64_t *o, size_t *o_n) {
    size_t k = 0, i_, off;
    orc_intersect_with_lin(dec->uids, dec->n_uids, v + k, m - k, o, o_n, &i_, &off);
    k += off;
    while (k < m) {
        orc_dec_linear_seek(dec, v[k]);
        if (dec->n_uids == 0) break;
        orc_intersect_with_lin(dec->uids, dec->n_uids, v + k, m - k, o, o_n, &i_, &off);
        if (off == 0) off = 1; /* if v[k] isn't in u, move forward (:76) */
        k += off;
    }
}

/* uidlist.go:87 IntersectCompressedWithBin */
void orc_intersect_compressed_with_bin(orc_dec *dec, const uint64_t *q, size_t lq,
                                       uint64_t *o, size_t *o_n) {
    size_t ld = orc_pack_exact_len(dec->pack);
    if (lq == 0) return;

    if (ld <= lq) { /* iterate blocks, jump-intersect each (:99-113) */
        const uint64_t *qq = q;
        size_t qn = lq;
        for (;;) {
            if (dec->n_uids == 0) break;
            size_t i_, off;
            orc_intersect_with_jump(dec->uids, dec->n_uids, qq, qn, o, o_n, &i_, &off);
            qq += off;
            qn -= off;
            if (qn == 0) return;
            orc_dec_next(dec);
        }
        return;
    }

    size_t qidx = 0; /* per-q-element seek path (:115-137) */
    for (;;) {
        if (qidx >= lq) return;
        uint64_t u = q[qidx];
        if (dec->n_uids == 0 || u > dec->uids[dec->n_uids - 1]) {
            if (lq * ORC_LIN_VS_BIN < ld) {
                orc_dec_linear_seek(dec, u);
            } else {
                orc_dec_seek_to_block(dec, u, ORC_SEEK_CURRENT);
            }
            if (dec->n_uids == 0) return;
        }
        size_t i_, off;
        orc_intersect_with_jump(dec->uids, dec->n_uids, q + qidx, lq - qidx, o, o_n, &i_, &off);
        if (off == 0) off = 1;
        qidx += off;
    }
}

/* uidlist.go:33 IntersectCompressedWith (linVsBinRatio=10 dispatch) */
size_t orc_intersect_compressed_with(const orc_pack *pack, uint64_t after_uid,
                                     const uint64_t *v, size_t m, uint64_t *out) {
    if (!pack) return 0;
    orc_dec dec;
    orc_dec_init(&dec, pack);
    orc_dec_seek(&dec, after_uid, ORC_SEEK_START);
    /* dec.ApproxLen() (codec.go:203): BlockSize * (len(Blocks) - blockIdx) */
    long long nblk = (long long)pack->n_blocks - (long long)dec.block_idx;
    if (nblk < 0) nblk = 0;
    size_t n = (size_t)pack->block_size * (size_t)nblk;
    size_t mm = m;
    if (n > mm) {
        size_t t = n;
        n = mm;
        mm = t;
    }
    size_t o_n = 0;
    if (n == 0) n = 1;
    double ratio = (double)mm / (double)n;
    if (ratio < ORC_LIN_VS_BIN) {
        orc_intersect_compressed_with_lin_jump(&dec, v, m, out, &o_n);
    } else {
        orc_intersect_compressed_with_bin(&dec, v, m, out, &o_n);
    }
    orc_dec_free(&dec);
    return o_n;
}

/* ===================== batched CPU baseline ===================== */

int orc_omp_max_threads(void) {
#ifdef _OPENMP
    return omp_get_max_threads();
#else
    return 1;
#endif
}

/* threads_used (optional): how many distinct OpenMP threads actually ran
 * >=1 pair — evidence that the baseline really used the cores it claims
 * (bench.py reports it as cpu_baseline.cores). */
void orc_intersect_batch_cpu(int n_pairs,
                             const uint64_t *const *us, const size_t *ns,
                             const uint64_t *const *vs, const size_t *ms,
                             uint64_t *const *outs, size_t *out_ns,
                             int n_threads, int *threads_used) {
    static unsigned char used[4096];
    memset(used, 0, sizeof(used));
#ifdef _OPENMP
    if (n_threads > 0) omp_set_num_threads(n_threads);
#pragma omp parallel for schedule(dynamic)
#endif
    for (int p = 0; p < n_pairs; p++) {
#ifdef _OPENMP
        int tid = omp_get_thread_num();
#else
        int tid = 0;
#endif
        if (tid >= 0 && tid < (int)sizeof(used)) used[tid] = 1;
        out_ns[p] = orc_intersect_with(us[p], ns[p], vs[p], ms[p], outs[p]);
    }
    if (threads_used) {
        int tu = 0;
        for (size_t i = 0; i < sizeof(used); i++) tu += used[i];
        *threads_used = tu;
    }
}
