/*
 * uidalgo — MI355X-native (gfx950 / CDNA4) posting-list set-algebra engine.
 *
 * Implements the C-ABI in include/uidalgo.h: a from-scratch GPU engine for
 * the dgraph algo/uidlist.go + codec/ hot path (reference interfaces cited
 * per function).  Design:
 *
 *  - batched merge-path set algebra: every (u,v) list pair is cut into
 *    TILE-element merge-path tiles; one 256-thread workgroup intersects /
 *    unions / differences one tile out of LDS; per-tile match counts are
 *    scanned (flat hierarchical scan) and compacted so outputs stay sorted.
 *    All pairs of a batch go through ONE grid (the reference fans out one
 *    goroutine per key, worker/task.go:834-971; here the whole fan-out is
 *    one launch).
 *  - fused decode+intersect: one 64-lane wavefront decodes one group-varint
 *    block (codec.go:154 UnpackBlock format) into LDS — control-byte walk,
 *    per-lane 4-delta extract, wave prefix-sum — and intersects it with its
 *    v-range without a global round trip.
 *
 * Everything is integer/HBM-bound: no MFMA.  Wavefront = 64; ballot masks
 * are 64-bit.
 *
 * Semantics contract (parity with the Go reference is bit-exact on
 * duplicate-free sorted inputs — the domain the reference itself pins:
 * uidlist_test.go:394 "behaviour of bin intersect is not defined when
 * duplicates are present"; posting lists are sorted sets, paper/dgraph.tex:269).
 */
#include <hip/hip_runtime.h>

#include <algorithm>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <mutex>
#include <unordered_map>
#include <vector>

#include "../../include/uidalgo.h"

typedef uint64_t u64;
typedef uint32_t u32;
typedef uint16_t u16;
typedef uint8_t u8;

#ifndef UA_ABLATE
#define UA_ABLATE 0 /* perf-bisection builds only (tools/ubench.py): 1 = fill-only,
                     * 2 = walk-on-garbage (no fill), 3 = no scan/write-back */
#endif

#ifndef UA_TILE
#define UA_TILE 2048
#endif      /* merge-path elements per tile */
#define UA_BLOCK 256      /* threads per workgroup (aux kernels) */
#ifndef UA_TBLOCK
#define UA_TBLOCK 256     /* threads per TILE-kernel workgroup: smaller WGs on
                           * smaller tiles raise independent tiles per CU (the
                           * 16-blocks/CU cap) — the chip's own latency hiding,
                           * which measured better than every software pipeline */
#endif
#define UA_WPT (UA_TILE / UA_TBLOCK) /* path elements per thread */
#ifndef UA_PAD32
#define UA_PAD32 0 /* 1 = skewed LDS layout (+1 u64 pad per 32): kills the
                    * 4-way ds_read_b64 bank conflicts of the walk/search
                    * gathers at power-of-2 lane strides (SQ: conflicts were
                    * 51% of LDS cycles); fill switches to 4B-lane LDS-DMA
                    * on 32-element chunks */
#endif
#if UA_PAD32
#define UA_PX(x) ((x) + ((x) >> 5)) /* flat logical u64 index -> physical */
#define UA_SMEMN (UA_TILE + UA_TILE / 32 + 8)
#else
#define UA_PX(x) (x)
#define UA_SMEMN (UA_TILE + 4)
#endif
#define UA_SCAN_CHUNK 2048
#define UA_STAGE_P 64 /* dense primary staging slots (u64/tile): tiles whose
                       * total emission fits go here (512 B apart -> DRAM
                       * row-local compaction reads) instead of the sparse
                       * stage_stride region (8-16 KB apart), which stays as
                       * the overflow path for dense-overlap tiles */
#define UA_PKW 4          /* packed-decode waves (blocks) per workgroup */
#define UA_MAX_BLOCK_UIDS 256
#define UA_MAX_DELTAS 1092 /* 64 groups x 17 B, padded */

static_assert(UA_TILE % UA_TBLOCK == 0, "tile must divide evenly over threads");
static_assert(UA_TILE / UA_TBLOCK >= 1 && UA_TILE / UA_TBLOCK <= 32,
              "emission flags are one u32 bit per path step per thread");
static_assert(UA_TBLOCK % 64 == 0 && UA_TBLOCK >= 64 && UA_TBLOCK <= 1024,
              "tile kernel is built from whole wavefronts");
static_assert(UA_BLOCK == 256, "wave geometry (4 waves/WG) is hard-coded in "
                               "scan/compact/packed kernels");

/* ==================== device helpers ==================== */

__device__ __forceinline__ u64 d_lower_bound(const u64 *__restrict__ a, u64 n, u64 key) {
    u64 lo = 0, hi = n;
    while (lo < hi) {
        u64 mid = (lo + hi) >> 1;
        if (a[mid] >= key) hi = mid;
        else lo = mid + 1;
    }
    return lo;
}

/* merge-path split: #A consumed after `diag` merge steps, A-priority on ties */
__device__ __forceinline__ u64 d_merge_path(const u64 *__restrict__ A, u64 n,
                                            const u64 *__restrict__ B, u64 m, u64 diag) {
    u64 lo = (diag > m) ? diag - m : 0;
    u64 hi = diag < n ? diag : n;
    while (lo < hi) {
        u64 mid = (lo + hi) >> 1;
        if (A[mid] <= B[diag - 1 - mid]) lo = mid + 1;
        else hi = mid;
    }
    return lo;
}

__device__ __forceinline__ int d_merge_path_lds(const u64 *A, int n, const u64 *B, int m,
                                                int diag) {
    int lo = diag > m ? diag - m : 0;
    int hi = diag < n ? diag : n;
    while (lo < hi) {
        int mid = (lo + hi) >> 1;
        if (A[mid] <= B[diag - 1 - mid]) lo = mid + 1;
        else hi = mid;
    }
    return lo;
}

#ifndef UA_WALK3
#define UA_WALK3 1 /* 1 = maskless A-value walk for intersect/diff (no em[] bookkeeping) */
#endif

#ifndef UA_NARY
#define UA_NARY 0 /* MEASURED REJECT (0.735 vs 0.710 ms on cfg2): 3 probe
                   * pairs per round (one lgkm wait) halve the search rounds,
                   * but the extra probe issue outweighs the latency saved —
                   * the phase is issue-bound after walk3 */
#endif

/* merge-path probe through the (possibly padded) tile layout */
__device__ __forceinline__ int d_merge_path_px(const u64 *smem, int aoff, int n,
                                               int boff, int m, int diag) {
    int lo = diag > m ? diag - m : 0;
    int hi = diag < n ? diag : n;
#if UA_NARY
    while (hi - lo > 3) {
        int q = (hi - lo) >> 2;
        int m1 = lo + q, m2 = lo + 2 * q, m3 = lo + 3 * q;
        /* all six gathers issue before one wait */
        u64 a1 = smem[UA_PX(aoff + m1)], b1 = smem[UA_PX(boff + diag - 1 - m1)];
        u64 a2 = smem[UA_PX(aoff + m2)], b2 = smem[UA_PX(boff + diag - 1 - m2)];
        u64 a3 = smem[UA_PX(aoff + m3)], b3 = smem[UA_PX(boff + diag - 1 - m3)];
        bool p1 = a1 <= b1, p2 = a2 <= b2, p3 = a3 <= b3;
        lo = p3 ? m3 + 1 : (p2 ? m2 + 1 : (p1 ? m1 + 1 : lo));
        hi = p1 ? (p2 ? (p3 ? hi : m3) : m2) : m1;
    }
#endif
    while (lo < hi) {
        int mid = (lo + hi) >> 1;
        if (smem[UA_PX(aoff + mid)] <= smem[UA_PX(boff + diag - 1 - mid)]) lo = mid + 1;
        else hi = mid;
    }
    return lo;
}

__device__ __forceinline__ int d_lower_bound_i(const u64 *a, int n, u64 key) {
    int lo = 0, hi = n;
    while (lo < hi) {
        int mid = (lo + hi) >> 1;
        if (a[mid] >= key) hi = mid;
        else lo = mid + 1;
    }
    return lo;
}

/* No scan-3 pass: the flat exclusive scan stays split as (chunk-local offs,
 * chunk partials); every consumer computes the final offset with one extra
 * L2 load — saves a full read+write sweep of offs plus a launch. */
__device__ __forceinline__ u64 d_off(const u64 *__restrict__ offs,
                                     const u64 *__restrict__ partials, u64 x) {
    return offs[x] + partials[x / UA_SCAN_CHUNK];
}

/* ==================== batch descriptors ==================== */

struct alignas(16) UaDesc {
    const u64 *u;
    u64 n;
    const u64 *v;
    u64 m;
    u64 *out;
    u64 tile_base; /* index of this pair's first tile in the global tile array */
};

/* ==================== kernel: tile partition ==================== */

/* Two-level tile partition.  Phase 0: full-range diagonal search for every
 * 8th tile of each pair (and records tile->pair for all tiles).  Phase 1:
 * the other 7/8 of tiles search only inside the bracket their two coarse
 * neighbours pin (merge path is monotone), a <=16k-element window that lands
 * in L2 — ~8x less random HBM fetch than a flat per-tile search. */
#define UA_COARSE 8
__global__ __launch_bounds__(UA_BLOCK) void k_partition(
    const UaDesc *__restrict__ descs, const u64 *__restrict__ tb, int n_pairs,
    u64 total_tiles, u32 *__restrict__ tile_pair, u32 *__restrict__ tile_a0, int phase) {
    u64 t = (u64)blockIdx.x * UA_BLOCK + threadIdx.x;
    if (t >= total_tiles) return;
    /* find pair p with tb[p] <= t < tb[p+1] */
    int lo = 0, hi = n_pairs - 1;
    while (lo < hi) {
        int mid = (lo + hi + 1) >> 1;
        if (tb[mid] <= t) lo = mid;
        else hi = mid - 1;
    }
    int p = lo;
    u64 lt = t - tb[p];
    int is_coarse = ((lt & (UA_COARSE - 1)) == 0);
    if (phase == 0) {
        tile_pair[t] = (u32)p;
        if (!is_coarse) return;
    } else {
        if (is_coarse) return;
    }
    UaDesc d = descs[p];
    u64 path = d.n + d.m;
    u64 diag = lt * UA_TILE;
    if (diag > path) diag = path;
    u64 slo = (diag > d.m) ? diag - d.m : 0;
    u64 shi = diag < d.n ? diag : d.n;
    if (phase == 1) {
        u64 t_down = t - (lt & (UA_COARSE - 1));
        u64 bl = tile_a0[t_down];
        u64 t_up = t_down + UA_COARSE;
        u64 bh = (t_up < tb[p + 1]) ? (u64)tile_a0[t_up] : d.n;
        if (bl > slo) slo = bl;
        if (bh < shi) shi = bh;
    }
    while (slo < shi) {
        u64 mid = (slo + shi) >> 1;
        if (d.u[mid] <= d.v[diag - 1 - mid]) slo = mid + 1;
        else shi = mid;
    }
    tile_a0[t] = (u32)slo;
}

/* ==================== kernel: tile set-algebra ==================== */

enum { OP_INTERSECT = 0, OP_UNION = 1, OP_DIFF = 2, OP_MERGE_ALL = 3 };
enum { MODE_STAGE = 0, MODE_COUNT = 1, MODE_WRITE = 2, MODE_DIRECT = 3,
       MODE_LOOKBACK = 4 };

/* ---- decoupled-lookback tile prefix (MODE_LOOKBACK) ----
 * Single-pass pipeline: each tile walks once, publishes its emission count,
 * resolves its pair-local exclusive prefix by looking back over predecessor
 * tiles (CUB-style: 64 lanes inspect 64 predecessors per round), and writes
 * its emissions at final positions — no staging buffer, no separate scan /
 * compact / pair_out launches, and union needs ONE walk instead of
 * count+write.  Safe on CDNA4: workgroups dispatch first->last
 * (MI355X_MICROARCH.md §dispatch; the same assumption rocPRIM's device scan
 * ships on), and the 8-byte flag word is an untorn agent-scope (sc1)
 * load/store granule, so no fences are needed — the word itself carries the
 * value.
 *
 * Flag word: [63:48] generation (stale-run guard, wraps via memset),
 * [47:46] status (0 invalid / 1 aggregate / 2 pair-local inclusive prefix),
 * [45:0] value. */
#define UA_LB_GEN_MAX 0xFFFFull
__device__ __forceinline__ u64 d_lb_word(u64 gen, u64 status, u64 val) {
    return (gen << 48) | (status << 46) | val;
}

/* wave0 (tid<64) resolves the pair-local exclusive prefix of tile t */
__device__ __forceinline__ u64 d_lb_resolve(u64 *lbf, u64 gen, u64 t, u64 base_t,
                                            int lane) {
    u64 run = 0;
    u64 hi = t;
    for (;;) {
        long idx = (long)hi - 1 - lane;
        int st = 2;
        u64 val = 0;
        if (idx >= (long)base_t) {
            u64 w = __hip_atomic_load(&lbf[idx], __ATOMIC_RELAXED,
                                      __HIP_MEMORY_SCOPE_AGENT);
            while ((w >> 48) != gen || !((w >> 46) & 3)) {
                __builtin_amdgcn_s_sleep(1);
                w = __hip_atomic_load(&lbf[idx], __ATOMIC_RELAXED,
                                      __HIP_MEMORY_SCOPE_AGENT);
            }
            st = (int)((w >> 46) & 3);
            val = w & ((1ull << 46) - 1);
        }
        u64 pm = __ballot(st == 2);
        /* nearest PREFIX lane (smallest back distance); lanes past it add
         * nothing.  pm==0: whole window is aggregates — sum and slide. */
        u64 contrib;
        if (pm) {
            int k = __ffsll((unsigned long long)pm) - 1;
            contrib = (lane <= k) ? val : 0;
        } else {
            contrib = val;
        }
#pragma unroll
        for (int o = 32; o >= 1; o >>= 1) contrib += __shfl_xor(contrib, o);
        run += contrib;
        if (pm) return run;
        hi -= 64;
    }
}

/* One pass over the thread's merge-path segment.  Emissions are recorded in
 * a statically-indexed register array (em[s], flag bit s) so the walk runs
 * ONCE: count + payload together, no re-walk, no scratch spill (guide §5.4
 * rule 20: runtime-indexed local arrays go to scratch; per-step static
 * indices do not). */
/* UNION dedup: the reference's MergeSorted emits the merged stream with ALL
 * duplicates collapsed (uidlist.go:417: output==last skips), including
 * duplicates WITHIN one list (TestMergeSorted6/9/10).  Rule: emit iff value
 * != previous emitted value == previous merged-stream value.  A thread's
 * initial "previous" is max(last A before it, last B before it), using the
 * staged tile-boundary elements when its segment starts at a range head. */
__device__ __forceinline__ u64 d_prev_stream(const u64 *As, int i0, u64 a_before,
                                             bool has_ab, const u64 *Bs, int j0,
                                             u64 b_before, bool has_bb, bool &has_prev) {
    u64 pa = (i0 > 0) ? As[i0 - 1] : a_before;
    bool hpa = (i0 > 0) || has_ab;
    u64 pb = (j0 > 0) ? Bs[j0 - 1] : b_before;
    bool hpb = (j0 > 0) || has_bb;
    has_prev = hpa || hpb;
    if (!hpa) return pb;
    if (!hpb) return pa;
    return pa > pb ? pa : pb;
}

template <int OP>
__device__ __forceinline__ int tile_walk(const u64 *As, int alen, const u64 *Bs, int blen,
                                         u64 a_before, bool has_ab, u64 b_before,
                                         bool has_bb, bool has_bn,
                                         int s0, int s1, int i0,
                                         u64 (&em)[UA_WPT], u32 &flags) {
    int i = i0, j = s0 - i0;
    int cnt = 0;
    flags = 0;
    int steps = s1 - s0;
    int blen_ext = blen + (has_bn ? 1 : 0); /* Bs[blen] is the lookahead slot */
    /* frontier values live in registers: ONE masked LDS reload per step
     * instead of re-gathering As[i]/Bs[j] every comparison */
    u64 a = (i < alen) ? As[i] : 0;
    u64 b = (j < blen_ext) ? Bs[j] : 0;
    bool has_prev = false;
    u64 prev_out = 0;
    if (OP == OP_UNION)
        prev_out = d_prev_stream(As, i, a_before, has_ab, Bs, j, b_before, has_bb,
                                 has_prev);
#pragma unroll
    for (int s = 0; s < UA_WPT; s++) {
        if (s >= steps || (i >= alen && j >= blen)) break;
        bool takeA = (i < alen) && (j >= blen || a <= b);
        if (takeA) {
            if (OP == OP_INTERSECT) {
                bool match = (j < blen_ext) && (a == b);
                if (match) {
                    em[s] = a;
                    flags |= 1u << s;
                    cnt++;
                }
            } else if (OP == OP_DIFF) {
                bool match = (j < blen_ext) && (a == b);
                if (!match) {
                    em[s] = a;
                    flags |= 1u << s;
                    cnt++;
                }
            } else { /* UNION: emit iff != previous merged-stream value */
                if (!has_prev || a != prev_out) {
                    em[s] = a;
                    flags |= 1u << s;
                    cnt++;
                }
                prev_out = a;
                has_prev = true;
            }
            i++;
            if (i < alen) a = As[i];
        } else {
            if (OP == OP_UNION) {
                if (!has_prev || b != prev_out) {
                    em[s] = b;
                    flags |= 1u << s;
                    cnt++;
                }
                prev_out = b;
                has_prev = true;
            }
            j++;
            if (j < blen_ext) b = Bs[j];
        }
    }
    return cnt;
}

/* block-wide exclusive scan of per-thread counts: wave __shfl scan + one
 * cross-wave combine — 1 barrier instead of Hillis-Steele's 17 */
template <int NB>
__device__ __forceinline__ void d_block_scan(int tid, u32 cnt, u32 *wsum,
                                             u32 &excl, u32 &total) {
    int lane = tid & 63, wv = tid >> 6;
    u32 incl = cnt;
#pragma unroll
    for (int o = 1; o < 64; o <<= 1) {
        u32 x = __shfl_up(incl, o);
        if (lane >= o) incl += x;
    }
    if (lane == 63) wsum[wv] = incl;
    __syncthreads();
    u32 wbase = 0;
    u32 tot = 0;
#pragma unroll
    for (int w = 0; w < NB / 64; w++) {
        u32 s = wsum[w];
        if (w < wv) wbase += s;
        tot += s;
    }
    total = tot;
    excl = wbase + incl - cnt;
}

/* Advance j to the first index in [j, bext) with Bs[idx] >= a: gallop then
 * binary — O(log gap) LDS probes, no serial per-element dependence. */
__device__ __forceinline__ int d_advance(const u64 *Bs, int bext, int j, u64 a) {
    if (j >= bext || Bs[j] >= a) return j;
    int st = 1;
    while (j + st < bext && Bs[j + st] < a) st <<= 1;
    int lo = j + (st >> 1) + 1; /* Bs[j + st/2] < a */
    int hi = j + st;
    if (hi > bext) hi = bext;
    while (lo < hi) {
        int mid = (lo + hi) >> 1;
        if (Bs[mid] >= a) hi = mid;
        else lo = mid + 1;
    }
    return lo;
}

/* A-indexed tile pass for INTERSECT/DIFF: threads take contiguous A-index
 * chunks (output order = A order since only A elements emit), and find each
 * A element in the tile's B range by monotone galloping search.  ~2x fewer
 * instructions than the merge walk (the kernel is issue-bound), and no
 * per-thread diagonal search. */
template <int OP>
__device__ __forceinline__ int tile_search(const u64 *As, int alen, const u64 *Bs,
                                           int blen, bool has_bn, int tid,
                                           u64 (&em)[UA_WPT], u32 &flags) {
    int lo = (int)(((long)tid * alen) / UA_TBLOCK);
    int hi = (int)(((long)(tid + 1) * alen) / UA_TBLOCK);
    int bext = blen + (has_bn ? 1 : 0);
    int cnt = 0;
    flags = 0;
    if (lo >= hi) return 0;
    int j = d_lower_bound_i(Bs, bext, As[lo]);
#pragma unroll
    for (int s = 0; s < UA_WPT; s++) {
        if (lo + s >= hi) break;
        u64 a = As[lo + s];
        j = d_advance(Bs, bext, j, a);
        bool found = (j < bext) && (Bs[j] == a);
        bool keep = (OP == OP_INTERSECT) ? found : !found;
        if (keep) {
            em[s] = a;
            flags |= 1u << s;
            cnt++;
        }
    }
    return cnt;
}

/* Branchless walk: the asm of the if/else walk is ~25% exec-mask control
 * (s_and_saveexec/s_or_b64/s_cbranch per step).  This variant keeps a
 * 2-deep frontier per side in registers and does ONE unconditional LDS
 * gather per step at a selected clamped address — no divergent branches in
 * the loop body.  As and Bs must live in the SAME LDS array (smem) so the
 * refill address can select between them.  aoff/boff are u64-element
 * offsets of As/Bs within smem. */
template <int OP, int W>
__device__ __forceinline__ int tile_walk2(const u64 *smembase, int aoff, int alen,
                                          int boff, int blen, u64 a_before, bool has_ab,
                                          u64 b_before, bool has_bb,
                                          bool has_bn, int s0, int s1, int i0,
                                          u64 (&em)[W], u32 &flags) {
    int i = i0, j = s0 - i0;
    int cnt = 0;
    flags = 0;
    int steps = s1 - s0;
    int blen_ext = blen + (has_bn ? 1 : 0);
    int amax = alen > 0 ? alen - 1 : 0;
    int bmax = blen_ext > 0 ? blen_ext - 1 : 0;
    /* frontier: cur + next per side, clamped loads (garbage guarded by
     * i/j bound checks in the predicates); all LDS gathers go through
     * UA_PX (identity unless UA_PAD32) */
    u64 a = smembase[UA_PX(aoff + (i < alen ? i : amax))];
    u64 an = smembase[UA_PX(aoff + ((i + 1) < alen ? (i + 1) : amax))];
    u64 b = smembase[UA_PX(boff + (j < blen_ext ? j : bmax))];
    u64 bn = smembase[UA_PX(boff + ((j + 1) < blen_ext ? (j + 1) : bmax))];
    bool has_prev = false;
    u64 prev_out = 0;
    if (OP == OP_UNION) {
        u64 pa = (i > 0) ? smembase[UA_PX(aoff + i - 1)] : a_before;
        bool hpa = (i > 0) || has_ab;
        u64 pb = (j > 0) ? smembase[UA_PX(boff + j - 1)] : b_before;
        bool hpb = (j > 0) || has_bb;
        has_prev = hpa || hpb;
        prev_out = !hpa ? pb : (!hpb ? pa : (pa > pb ? pa : pb));
    }
#pragma unroll
    for (int s = 0; s < W; s++) {
        if (s >= steps) break;
        bool inA = i < alen, inB = j < blen;
        if (!inA && !inB) break;
        bool takeA = inA && (!inB || a <= b);
        bool eq = (a == b) && (j < blen_ext);
        bool emit;
        if (OP == OP_INTERSECT) {
            emit = takeA && eq;
            em[s] = a;
        } else if (OP == OP_DIFF) {
            emit = takeA && !eq;
            em[s] = a;
        } else if (OP == OP_MERGE_ALL) { /* duplicate-keeping merge (sort runs) */
            emit = true;
            em[s] = takeA ? a : b;
        } else { /* UNION: emit iff != previous merged-stream value */
            u64 val = takeA ? a : b;
            emit = !has_prev || val != prev_out;
            em[s] = val;
            prev_out = val;
            has_prev = true;
        }
        flags |= ((u32)emit) << s;
        cnt += emit;
        int ni = i + (takeA ? 1 : 0);
        int nj = j + (takeA ? 0 : 1);
        /* one refill: next lookahead of the consumed side */
        int ra = (ni + 1) < alen ? (ni + 1) : amax;
        int rb = (nj + 1) < blen_ext ? (nj + 1) : bmax;
        int raddr = takeA ? (aoff + ra) : (boff + rb);
        u64 r = smembase[UA_PX(raddr)];
        a = takeA ? an : a;
        b = takeA ? b : bn;
        an = takeA ? r : an;
        bn = takeA ? bn : r;
        i = ni;
        j = nj;
    }
    return cnt;
}

/* COUNT-only walk: tile_walk2 minus the em[] value writes — MODE_COUNT
 * needs only the per-thread emission count (union's first pass; walk3
 * already covers intersect/diff counts).  Dedup state is kept so union
 * counts match tile_walk2 exactly. */
template <int OP, int W>
__device__ __forceinline__ int tile_walk2c(const u64 *smembase, int aoff, int alen,
                                           int boff, int blen, u64 a_before,
                                           bool has_ab, u64 b_before, bool has_bb,
                                           bool has_bn, int s0, int s1, int i0) {
    int i = i0, j = s0 - i0;
    int cnt = 0;
    int steps = s1 - s0;
    int blen_ext = blen + (has_bn ? 1 : 0);
    int amax = alen > 0 ? alen - 1 : 0;
    int bmax = blen_ext > 0 ? blen_ext - 1 : 0;
    u64 a = smembase[UA_PX(aoff + (i < alen ? i : amax))];
    u64 an = smembase[UA_PX(aoff + ((i + 1) < alen ? (i + 1) : amax))];
    u64 b = smembase[UA_PX(boff + (j < blen_ext ? j : bmax))];
    u64 bn = smembase[UA_PX(boff + ((j + 1) < blen_ext ? (j + 1) : bmax))];
    bool has_prev = false;
    u64 prev_out = 0;
    if (OP == OP_UNION) {
        u64 pa = (i > 0) ? smembase[UA_PX(aoff + i - 1)] : a_before;
        bool hpa = (i > 0) || has_ab;
        u64 pb = (j > 0) ? smembase[UA_PX(boff + j - 1)] : b_before;
        bool hpb = (j > 0) || has_bb;
        has_prev = hpa || hpb;
        prev_out = !hpa ? pb : (!hpb ? pa : (pa > pb ? pa : pb));
    }
#pragma unroll
    for (int s = 0; s < W; s++) {
        if (s >= steps) break;
        bool inA = i < alen, inB = j < blen;
        if (!inA && !inB) break;
        bool takeA = inA && (!inB || a <= b);
        bool eq = (a == b) && (j < blen_ext);
        bool emit;
        if (OP == OP_INTERSECT) emit = takeA && eq;
        else if (OP == OP_DIFF) emit = takeA && !eq;
        else if (OP == OP_MERGE_ALL) emit = true;
        else {
            u64 val = takeA ? a : b;
            emit = !has_prev || val != prev_out;
            prev_out = val;
            has_prev = true;
        }
        cnt += emit;
        int ni = i + (takeA ? 1 : 0);
        int nj = j + (takeA ? 0 : 1);
        int ra = (ni + 1) < alen ? (ni + 1) : amax;
        int rb = (nj + 1) < blen_ext ? (nj + 1) : bmax;
        int raddr = takeA ? (aoff + ra) : (boff + rb);
        u64 r = smembase[UA_PX(raddr)];
        a = takeA ? an : a;
        b = takeA ? b : bn;
        an = takeA ? r : an;
        bn = takeA ? bn : r;
        i = ni;
        j = nj;
    }
    return cnt;
}

/* INTERSECT/DIFF-only walk with NO value array: emissions are always
 * A-values, so instead of maintaining em[W] (whose per-step conditional
 * writes and register shuffling dominate the issue-bound walk loop), track
 * one takeA bitmask and reconstruct the few matched values from LDS after
 * the scan: A-index at step s = i0 + popcount(amask below s).  ~0.04
 * matches/thread on the 1%-overlap headline shape. */
template <int OP, int W>
__device__ __forceinline__ int tile_walk3(const u64 *smembase, int aoff, int alen,
                                          int boff, int blen, bool has_bn,
                                          int s0, int s1, int i0,
                                          u32 &flags, u32 &amask) {
    static_assert(W <= 32, "masks are u32");
    int i = i0, j = s0 - i0;
    int cnt = 0;
    flags = 0;
    amask = 0;
    int steps = s1 - s0;
    int blen_ext = blen + (has_bn ? 1 : 0);
    int amax = alen > 0 ? alen - 1 : 0;
    int bmax = blen_ext > 0 ? blen_ext - 1 : 0;
    u64 a = smembase[UA_PX(aoff + (i < alen ? i : amax))];
    u64 an = smembase[UA_PX(aoff + ((i + 1) < alen ? (i + 1) : amax))];
    u64 b = smembase[UA_PX(boff + (j < blen_ext ? j : bmax))];
    u64 bn = smembase[UA_PX(boff + ((j + 1) < blen_ext ? (j + 1) : bmax))];
#pragma unroll
    for (int s = 0; s < W; s++) {
        if (s >= steps) break;
        bool inA = i < alen, inB = j < blen;
        if (!inA && !inB) break;
        bool takeA = inA && (!inB || a <= b);
        bool eq = (a == b) && (j < blen_ext);
        bool emit = (OP == OP_INTERSECT) ? (takeA && eq) : (takeA && !eq);
        flags |= ((u32)emit) << s;
        amask |= ((u32)takeA) << s;
        cnt += emit;
        int ni = i + (takeA ? 1 : 0);
        int nj = j + (takeA ? 0 : 1);
        int ra = (ni + 1) < alen ? (ni + 1) : amax;
        int rb = (nj + 1) < blen_ext ? (nj + 1) : bmax;
        int raddr = takeA ? (aoff + ra) : (boff + rb);
        u64 r = smembase[UA_PX(raddr)];
        a = takeA ? an : a;
        b = takeA ? b : bn;
        an = takeA ? r : an;
        bn = takeA ? bn : r;
        i = ni;
        j = nj;
    }
    return cnt;
}

/* Dual-chain maskless walk (intersect/diff): two independent 4-step
 * gather chains per thread (halved serial LDS latency), walk3's
 * no-em[] bookkeeping.  Viable because the prepared batch caches BOTH
 * splits (i0 at s0, i0b at smid) — round 1 rejected the dual-chain
 * em[] walk precisely because its second diagonal search cost more
 * than the added MLP recovered. */
template <int OP, int W>
__device__ __forceinline__ int tile_walk3x(const u64 *smembase, int aoff, int alen,
                                           int boff, int blen, bool has_bn,
                                           int s0, int smid, int s1, int i0a,
                                           int i0b, u32 &flags, u32 &amA, u32 &amB) {
    static_assert(W <= 32, "masks are u32");
    constexpr int H = W / 2;
    int blen_ext = blen + (has_bn ? 1 : 0);
    int amax = alen > 0 ? alen - 1 : 0;
    int bmax = blen_ext > 0 ? blen_ext - 1 : 0;
    int ia = i0a, ja = s0 - i0a;
    int ib = i0b, jb = smid - i0b;
    int stepsA = smid - s0, stepsB = s1 - smid;
    u64 aA = smembase[UA_PX(aoff + (ia < alen ? ia : amax))];
    u64 bA = smembase[UA_PX(boff + (ja < blen_ext ? ja : bmax))];
    u64 aB = smembase[UA_PX(aoff + (ib < alen ? ib : amax))];
    u64 bB = smembase[UA_PX(boff + (jb < blen_ext ? jb : bmax))];
    int cnt = 0;
    flags = 0;
    amA = 0;
    amB = 0;
#pragma unroll
    for (int q = 0; q < H; q++) {
        if (q < stepsA && (ia < alen || ja < blen)) {
            bool inA = ia < alen, inB = ja < blen;
            bool takeA = inA && (!inB || aA <= bA);
            bool eq = (aA == bA) && (ja < blen_ext);
            bool emit = (OP == OP_INTERSECT) ? (takeA && eq) : (takeA && !eq);
            flags |= ((u32)emit) << q;
            amA |= ((u32)takeA) << q;
            cnt += emit;
            int ni = ia + (takeA ? 1 : 0), nj = ja + (takeA ? 0 : 1);
            int raddr = takeA ? (aoff + (ni < alen ? ni : amax))
                              : (boff + (nj < blen_ext ? nj : bmax));
            u64 r = smembase[UA_PX(raddr)];
            aA = takeA ? r : aA;
            bA = takeA ? bA : r;
            ia = ni;
            ja = nj;
        }
        if (q < stepsB && (ib < alen || jb < blen)) {
            bool inA = ib < alen, inB = jb < blen;
            bool takeA = inA && (!inB || aB <= bB);
            bool eq = (aB == bB) && (jb < blen_ext);
            bool emit = (OP == OP_INTERSECT) ? (takeA && eq) : (takeA && !eq);
            flags |= ((u32)emit) << (H + q);
            amB |= ((u32)takeA) << q;
            cnt += emit;
            int ni = ib + (takeA ? 1 : 0), nj = jb + (takeA ? 0 : 1);
            int raddr = takeA ? (aoff + (ni < alen ? ni : amax))
                              : (boff + (nj < blen_ext ? nj : bmax));
            u64 r = smembase[UA_PX(raddr)];
            aB = takeA ? r : aB;
            bB = takeA ? bB : r;
            ib = ni;
            jb = nj;
        }
    }
    return cnt;
}

/* Dual-chain walk (UA_WALK2X): each thread walks TWO independent
 * half-segments (diagonals s0 and s0+W/2) with interleaved steps — two
 * independent LDS-gather dependency chains per thread to close the
 * latency/MLP gap the SQ counters show (waves park 60% while HBM sits at
 * ~51%). */
template <int OP>
__device__ __forceinline__ int tile_walk2x(const u64 *smembase, int aoff, int alen,
                                           int boff, int blen, u64 a_before, bool has_ab,
                                           u64 b_before, bool has_bb, bool has_bn,
                                           int s0, int s1, int i0a, int i0b, int smid,
                                           u64 (&em)[UA_WPT], u32 &flags) {
    const u64 *As = smembase + aoff;
    const u64 *Bs = smembase + boff;
    int blen_ext = blen + (has_bn ? 1 : 0);
    int amax = alen > 0 ? alen - 1 : 0;
    int bmax = blen_ext > 0 ? blen_ext - 1 : 0;
    constexpr int H = UA_WPT / 2;

    int ia = i0a, ja = s0 - i0a;
    int ib = i0b, jb = smid - i0b;
    int stepsA = smid - s0, stepsB = s1 - smid;
    u64 aA = As[ia < alen ? ia : amax], bA = Bs[ja < blen_ext ? ja : bmax];
    u64 aB = As[ib < alen ? ib : amax], bB = Bs[jb < blen_ext ? jb : bmax];
    bool hpA = false, hpB = false;
    u64 poA = 0, poB = 0;
    if (OP == OP_UNION) {
        poA = d_prev_stream(As, ia, a_before, has_ab, Bs, ja, b_before, has_bb, hpA);
        poB = d_prev_stream(As, ib, a_before, has_ab, Bs, jb, b_before, has_bb, hpB);
    }
    int cnt = 0;
    flags = 0;
#pragma unroll
    for (int q = 0; q < H; q++) {
        /* chain A step */
        if (q < stepsA && (ia < alen || ja < blen)) {
            bool inA = ia < alen, inB = ja < blen;
            bool takeA = inA && (!inB || aA <= bA);
            bool eq = (aA == bA) && (ja < blen_ext);
            bool emit;
            u64 val = takeA ? aA : bA;
            if (OP == OP_INTERSECT) emit = takeA && eq;
            else if (OP == OP_DIFF) emit = takeA && !eq;
            else if (OP == OP_MERGE_ALL) emit = true;
            else {
                emit = !hpA || val != poA;
                poA = val;
                hpA = true;
            }
            em[q] = (OP == OP_INTERSECT || OP == OP_DIFF) ? aA : val;
            flags |= ((u32)emit) << q;
            cnt += emit;
            int ni = ia + (takeA ? 1 : 0), nj = ja + (takeA ? 0 : 1);
            int raddr = takeA ? (aoff + (ni < alen ? ni : amax))
                              : (boff + (nj < blen_ext ? nj : bmax));
            u64 r = smembase[raddr];
            aA = takeA ? r : aA;
            bA = takeA ? bA : r;
            ia = ni;
            ja = nj;
        }
        /* chain B step (independent registers: interleaves with A's gather) */
        if (q < stepsB && (ib < alen || jb < blen)) {
            bool inA = ib < alen, inB = jb < blen;
            bool takeA = inA && (!inB || aB <= bB);
            bool eq = (aB == bB) && (jb < blen_ext);
            bool emit;
            u64 val = takeA ? aB : bB;
            if (OP == OP_INTERSECT) emit = takeA && eq;
            else if (OP == OP_DIFF) emit = takeA && !eq;
            else if (OP == OP_MERGE_ALL) emit = true;
            else {
                emit = !hpB || val != poB;
                poB = val;
                hpB = true;
            }
            em[H + q] = (OP == OP_INTERSECT || OP == OP_DIFF) ? aB : val;
            flags |= ((u32)emit) << (H + q);
            cnt += emit;
            int ni = ib + (takeA ? 1 : 0), nj = jb + (takeA ? 0 : 1);
            int raddr = takeA ? (aoff + (ni < alen ? ni : amax))
                              : (boff + (nj < blen_ext ? nj : bmax));
            u64 r = smembase[raddr];
            aB = takeA ? r : aB;
            bB = takeA ? bB : r;
            ib = ni;
            jb = nj;
        }
    }
    return cnt;
}

/* cooperative global->LDS fill, 16-B vectorized on the aligned body
 * (8-B/lane loads cap ~60% of the dwordx4 HBM rate — guide §2/G13) */
__device__ __forceinline__ void d_fill_lds(u64 *dst, const u64 *__restrict__ src,
                                           int len, int tid) {
    int head = (int)(((uintptr_t)src >> 3) & 1); /* 1 if src ≡ 8 (mod 16) */
    if (head > len) head = len;
    if (tid == 0 && head) dst[0] = src[0];
    int nvec = (len - head) >> 1;
    const ulonglong2 *vs = (const ulonglong2 *)(src + head);
    if ((((uintptr_t)(dst + head)) & 15) == 0) {
        ulonglong2 *vd = (ulonglong2 *)(dst + head);
        for (int i = tid; i < nvec; i += UA_TBLOCK) vd[i] = vs[i];
    } else {
        for (int i = tid; i < nvec; i += UA_TBLOCK) {
            ulonglong2 x = vs[i];
            dst[head + 2 * i] = x.x;
            dst[head + 2 * i + 1] = x.y;
        }
    }
    for (int i = head + 2 * nvec + tid; i < len; i += UA_TBLOCK) dst[i] = src[i];
}

template <int OP, int MODE>
__global__ __launch_bounds__(UA_TBLOCK) void k_tiles(
    const UaDesc *__restrict__ descs, const u32 *__restrict__ tile_pair,
    const u32 *__restrict__ tile_a0, u64 total_tiles,
    u64 *__restrict__ staging, u64 stage_stride, u32 *__restrict__ tile_cnt,
    const u64 *__restrict__ offs, const u64 *__restrict__ partials /* MODE_WRITE only */,
    const u32 *__restrict__ isplit_in /* cached per-thread merge-path splits,
        PACKED lo16 = i0 at diagonal s0, hi16 = i0b at smid (prepared batch:
        inputs immutable, so splits are run-invariant like the tile
        partition); null = compute */,
    u32 *__restrict__ isplit_out /* store packed splits on the first run */) {
    __shared__ __align__(16) u64 smem[UA_SMEMN];
    __shared__ u32 scan[UA_TBLOCK / 64]; /* per-wave totals for d_block_scan */
    __shared__ u64 s_abefore;
    __shared__ u64 s_bbefore;
    __shared__ u64 s_run; /* MODE_LOOKBACK: pair-local exclusive prefix */

    u64 t = blockIdx.x;
    int tid = threadIdx.x;
    u32 p = tile_pair[t];
    UaDesc d = descs[p];
    u64 lt = t - d.tile_base;
    u64 path = d.n + d.m;
    u64 d0 = lt * UA_TILE;
    if (d0 > path) d0 = path; /* capacity-tiled batches can overshoot the path */
    u64 d1 = d0 + UA_TILE;
    if (d1 > path) d1 = path;
    u32 a0 = tile_a0[t];
    u32 a1 = (t + 1 < total_tiles && tile_pair[t + 1] == p) ? tile_a0[t + 1] : (u32)d.n;
    u32 b0 = (u32)(d0 - a0), b1 = (u32)(d1 - a1);
    int alen = (int)(a1 - a0), blen = (int)(b1 - b0);
    /* cached splits: issued HERE so the 4B load's latency hides under the
     * fill drain; consumed only after the barrier */
    int i0_cached = -1, i0b_cached = -1;
    if (isplit_in) {
        u32 w = isplit_in[t * UA_TBLOCK + tid];
        i0_cached = (int)(w & 0xffffu);
        i0b_cached = (int)(w >> 16);
    }

#ifndef UA_GLDS
#define UA_GLDS 1 /* global_load_lds (LDS-DMA) fill: +13% vs the 16B reg-staged fill (4.03 vs 3.58 TB/s on cfg2); 0 = reg-staged */
#endif
#if UA_PAD32
#if !UA_GLDS || UA_SEARCH || UA_WALK2X || (defined(UA_WALK2) && UA_WALK2 == 0)
#error "UA_PAD32 supports only the default glds + walk2 configuration"
#endif
    /* padded layout: 4B-lane DMA has no 16B parity constraint */
    int aoff = 0;
    int boff = alen;
#else
#if UA_GLDS
    /* LDS bases chosen so each side's 16B-aligned glds body lines up with
     * its source POINTER parity (glds writes wave-uniform base + lane*16;
     * list base pointers are not always 16B aligned — merge-tree runs and
     * sliced outputs sit at odd u64 offsets) */
    int ashift = (int)((((uintptr_t)(d.u + a0)) >> 3) & 1);
    int bshift = (int)((((uintptr_t)(d.v + b0)) >> 3) & 1);
    u64 *As = smem + ashift;
    int boff_base = ((ashift + alen + 1) & ~1) + bshift;
    u64 *Bs = smem + boff_base;
#else
    u64 *As = smem;                          /* 16-B aligned */
    u64 *Bs = smem + ((alen + 1) & ~1);      /* rounded up to even: 16-B aligned */
#endif
    int aoff = (int)(As - smem);
    int boff = (int)(Bs - smem);
#endif

    bool has_ab = (a0 > 0);
    bool has_bb = (b0 > 0);
    bool has_bn = ((u64)b1 < d.m);
#if UA_ABLATE == 5 /* meta + dispatch floor: no fill, no walk, no scan */
    if (tid == 0) tile_cnt[t] = (u32)((a0 ^ (u32)d.m) & 1u);
    (void)has_ab; (void)has_bb; (void)has_bn;
    return;
#endif
#if UA_ABLATE != 2
#if UA_PAD32
    {
        /* 4B-lane LDS-DMA on 32-element flat-aligned chunks: each chunk's
         * physical 256B LDS run is contiguous (pads fall between chunks) */
        int wv4 = tid >> 6, lane = tid & 63;
        constexpr int NW = UA_TBLOCK / 64;
        int c1 = alen & ~31; /* aoff == 0 is chunk-aligned */
        for (int f = wv4 * 32; f < c1; f += NW * 32) {
            const u32 *srcp = (const u32 *)(d.u + a0 + f) + lane;
            __builtin_amdgcn_global_load_lds(srcp, (u32 *)&smem[UA_PX(f)], 4, 0, 0);
        }
        for (int i = c1 + tid; i < alen; i += UA_TBLOCK)
            smem[UA_PX(i)] = d.u[a0 + i];

        int bfirst = ((boff + 31) & ~31);          /* first aligned flat chunk */
        int b_lead = bfirst - boff;
        if (b_lead > blen) b_lead = blen;
        int bend = (boff + blen) & ~31;            /* flat end of aligned body */
        for (int f = bfirst + wv4 * 32; f < bend; f += NW * 32) {
            const u32 *srcp = (const u32 *)(d.v + b0 + (f - boff)) + lane;
            __builtin_amdgcn_global_load_lds(srcp, (u32 *)&smem[UA_PX(f)], 4, 0, 0);
        }
        for (int i = tid; i < b_lead; i += UA_TBLOCK)
            smem[UA_PX(boff + i)] = d.v[b0 + i];
        int btail = bend - boff;
        if (btail < b_lead) btail = b_lead;
        for (int i = btail + tid; i < blen; i += UA_TBLOCK)
            smem[UA_PX(boff + i)] = d.v[b0 + i];
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
#elif UA_GLDS
    {
        /* per-wave LDS-DMA fill: 128 u64 per wave-call (64 lanes x 16 B) */
        int wv4 = tid >> 6, lane = tid & 63;
        int ahead = ashift; /* elements before the aligned body (0 or 1) */
        if (ahead > alen) ahead = alen;
        int abody = (alen - ahead) & ~127;
        for (int e = wv4 * 128; e < abody; e += (UA_TBLOCK / 64) * 128) {
            const u64 *g = d.u + a0 + ahead + e + lane * 2;
            __builtin_amdgcn_global_load_lds((const u32 *)g,
                                             (u32 *)&smem[ashift + ahead + e], 16, 0, 0);
        }
        for (int i = ahead + abody + tid; i < alen; i += UA_TBLOCK)
            As[i] = d.u[a0 + i];
        if (tid < ahead) As[tid] = d.u[a0 + tid];

        int bhead = bshift;
        if (bhead > blen) bhead = blen;
        int bbody = (blen - bhead) & ~127;
        for (int e = wv4 * 128; e < bbody; e += (UA_TBLOCK / 64) * 128) {
            const u64 *g = d.v + b0 + bhead + e + lane * 2;
            __builtin_amdgcn_global_load_lds((const u32 *)g,
                                             (u32 *)&smem[boff_base + bhead + e], 16, 0, 0);
        }
        for (int i = bhead + bbody + tid; i < blen; i += UA_TBLOCK)
            Bs[i] = d.v[b0 + i];
        if (tid < bhead) Bs[tid] = d.v[b0 + tid];
        /* drain the LDS-DMA before the barrier: the glds writes count on
         * vmcnt, and the compiler's barrier wait cannot be relied on here */
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
#else
    d_fill_lds(As, d.u + a0, alen, tid);
    d_fill_lds(Bs, d.v + b0, blen, tid);
#endif
    if (tid == 0) {
        s_abefore = has_ab ? d.u[a0 - 1] : 0;
        if (OP == OP_UNION) s_bbefore = has_bb ? d.v[b0 - 1] : 0;
        smem[UA_PX(boff + blen)] = has_bn ? d.v[b1] : 0;
    }
#endif
    __syncthreads();

#ifndef UA_SEARCH
#define UA_SEARCH 0 /* 1 = A-indexed gallop search for intersect/diff; measured 3.2 vs 3.45 TB/s for the merge walk on cfg2 - kept for skewed-ratio experiments */
#endif
#if UA_SEARCH && UA_WALK3
#error "UA_SEARCH fills em[]; build it with -DUA_WALK3=0"
#endif
#if UA_ABLATE == 1 /* fill-only: keep the loads live, skip search+walk */
    u64 ablate_x = smem[UA_PX(aoff + tid)] + smem[UA_PX(boff + (tid & 127))];
    asm volatile("" ::"v"(ablate_x));
    u64 em[UA_WPT];
    u32 flags = 0;
    u32 amA = 0, amB = 0;
    int w_i0 = 0, w_i0b = 0;
    int cnt = 0;
    (void)s_abefore;
#else
    u64 em[UA_WPT];
    u32 flags;
    u32 amA = 0, amB = 0; /* walk3x: per-chain takeA bits for reconstruction */
    int w_i0 = 0, w_i0b = 0;
    int cnt;
#ifndef UA_WALK2
#define UA_WALK2 1 /* 0 = the branchy register-frontier walk */
#endif
#if !UA_PAD32
    if (UA_SEARCH && OP != OP_UNION && OP != OP_MERGE_ALL) {
        cnt = tile_search<OP>(As, alen, Bs, blen, has_bn, tid, em, flags);
    } else
#endif
    {
        int tilelen = alen + blen;
        int s0 = tid * UA_WPT;
        int s1 = s0 + UA_WPT;
        if (s0 > tilelen) s0 = tilelen;
        if (s1 > tilelen) s1 = tilelen;
        int smid = s0 + UA_WPT / 2;
        if (smid > tilelen) smid = tilelen;
        if (smid < s0) smid = s0;
        if (smid > s1) smid = s1;
        int i0 = (i0_cached >= 0) ? i0_cached
                                  : d_merge_path_px(smem, aoff, alen, boff, blen, s0);
        int i0b = (i0b_cached >= 0)
                      ? i0b_cached
                      : d_merge_path_px(smem, aoff, alen, boff, blen, smid);
        if (isplit_out)
            isplit_out[t * UA_TBLOCK + tid] = (u32)i0 | ((u32)i0b << 16);
        u64 a_before = s_abefore;
        u64 b_before = (OP == OP_UNION) ? s_bbefore : 0;
#ifndef UA_WALK2X
#define UA_WALK2X 0 /* 1 = dual-chain walk (2 independent gather chains/thread) */
#endif
#if !UA_PAD32
        if (UA_WALK2X) {
            int smid = s0 + UA_WPT / 2;
            if (smid > tilelen) smid = tilelen;
            if (smid < s0) smid = s0;
            if (smid > s1) smid = s1;
            int i0b = d_merge_path_lds(As, alen, Bs, blen, smid);
            cnt = tile_walk2x<OP>(smem, (int)(As - smem), alen, (int)(Bs - smem), blen,
                                  a_before, has_ab, b_before, has_bb, has_bn, s0, s1,
                                  i0, i0b, smid, em, flags);
        } else if (!UA_WALK2) {
            cnt = tile_walk<OP>(As, alen, Bs, blen, a_before, has_ab, b_before,
                                has_bb, has_bn, s0, s1, i0, em, flags);
        } else
#endif
        if (UA_WALK3 && (OP == OP_INTERSECT || OP == OP_DIFF)) {
#ifndef UA_DUALWALK
#define UA_DUALWALK 1 /* 0 = single-chain walk3 (A/B toggle) */
#endif
#if UA_DUALWALK
            cnt = tile_walk3x<OP, UA_WPT>(smem, aoff, alen, boff, blen, has_bn,
                                          s0, smid, s1, i0, i0b, flags, amA, amB);
            w_i0 = i0;
            w_i0b = i0b;
#else
            u32 amall;
            cnt = tile_walk3<OP, UA_WPT>(smem, aoff, alen, boff, blen, has_bn,
                                         s0, s1, i0, flags, amall);
            /* express the single mask in the dual-reconstruction form:
             * amB = high half, w_i0b = i0 + takeA count of the low half */
            constexpr u32 HM = (1u << (UA_WPT / 2)) - 1;
            amA = amall & HM;
            amB = amall >> (UA_WPT / 2);
            w_i0 = i0;
            w_i0b = i0 + __popc(amA);
#endif
        } else if (MODE == MODE_COUNT) {
            /* count pass needs no values — skip the em[] bookkeeping */
            cnt = tile_walk2c<OP, UA_WPT>(smem, aoff, alen, boff, blen, a_before,
                                          has_ab, b_before, has_bb, has_bn, s0, s1,
                                          i0);
            flags = 0;
        } else {
            cnt = tile_walk2<OP>(smem, aoff, alen, boff, blen,
                                 a_before, has_ab, b_before, has_bb, has_bn, s0, s1,
                                 i0, em, flags);
        }
        if (MODE == MODE_DIRECT) {
            /* OP_MERGE_ALL: every path element emits, so the output position
             * IS the path position — no scan, no staging, no compaction */
            u64 *dst = d.out + d0 + (u64)s0;
            int steps = s1 - s0;
#pragma unroll
            for (int s = 0; s < UA_WPT; s++) {
                if (s < steps) dst[s] = em[s];
            }
            return;
        }
    }
#endif
#if UA_ABLATE == 3 /* fill+walk, skip scan/write-back */
    asm volatile("" ::"v"(cnt), "v"(flags));
    cnt = 0;
    flags = 0;
#endif

    u32 excl, total;
    d_block_scan<UA_TBLOCK>(tid, (u32)cnt, scan, excl, total);

    if (MODE == MODE_COUNT) {
        if (tid == 0) tile_cnt[t] = total;
        return;
    }
    if (MODE == MODE_LOOKBACK) {
        /* single-pass: staging = flag array, stage_stride = generation,
         * offs = per-pair output lengths (d_pout).  See d_lb_resolve. */
        u64 *lbf = staging;
        u64 gen = stage_stride;
        u64 *pout = (u64 *)offs;
        if (tid == 0) {
            /* first tile's aggregate IS its inclusive prefix — publish
             * PREFIX directly so successor windows terminate fast */
            u64 st = (lt == 0) ? 2ull : 1ull;
            __hip_atomic_store(&lbf[t], d_lb_word(gen, st, total),
                               __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
            if (lt == 0) s_run = 0;
        }
        if (lt != 0 && tid < 64) {
            u64 run = d_lb_resolve(lbf, gen, t, d.tile_base, tid);
            if (tid == 0) {
                s_run = run;
                __hip_atomic_store(&lbf[t], d_lb_word(gen, 2ull, run + total),
                                   __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
            }
        }
        __syncthreads();
        u64 run = s_run;
        /* out-capacity clamp: on the contract's duplicate-free sorted inputs
         * emissions never exceed capacity; on invalid inputs results are
         * unspecified (like the reference's bin variants) but writes stay
         * in bounds (the reference is memory-safe there, so are we). */
        u64 cap = (OP == OP_INTERSECT) ? (d.n < d.m ? d.n : d.m)
                  : (OP == OP_DIFF) ? d.n
                                    : (d.n + d.m);
        u64 gbase = run + (u64)excl;
        if (cnt > 0 && gbase < cap) {
            u64 *dst = d.out + gbase;
            u64 room = cap - gbase;
            int lim = (int)((room < (u64)cnt) ? room : (u64)cnt);
            if (UA_WALK3 && (OP == OP_INTERSECT || OP == OP_DIFF)) {
                /* reconstruct emitted A-values from the per-chain masks */
                constexpr int H = UA_WPT / 2;
                u32 f = flags;
                int k = 0;
                while (f && k < lim) {
                    int s = __builtin_ctz(f);
                    f &= f - 1;
                    int idx = (s < H)
                                  ? w_i0 + __popc(amA & ((1u << s) - 1))
                                  : w_i0b + __popc(amB & ((1u << (s - H)) - 1));
                    dst[k++] = smem[UA_PX(aoff + idx)];
                }
            } else {
                int k = 0;
#pragma unroll
                for (int s = 0; s < UA_WPT; s++) {
                    if (flags & (1u << s)) {
                        if (k < lim) dst[k] = em[s];
                        k++;
                    }
                }
            }
        }
        if (tid == 0) {
            bool lastt = (t + 1 == total_tiles) || (tile_pair[t + 1] != p);
            if (lastt) {
                u64 inc = run + total;
                pout[p] = inc < cap ? inc : cap;
            }
        }
        return;
    }
    u64 *dst;
    u32 lim = (u32)cnt;
    if (MODE == MODE_STAGE) {
        /* clamp to the staging stride: invalid (duplicate/unsorted) inputs
         * can emit more than stride entries per tile; results there are
         * unspecified, OOB writes are not.  Small totals go to
         * the dense primary region (offs = its base, see batch_tiles_seq);
         * stage_stride stays the sparse overflow. */
        bool prim = (offs != nullptr) && (total <= UA_STAGE_P);
        u32 cap_t = prim ? (u32)UA_STAGE_P : (u32)stage_stride;
        u32 c0 = excl < cap_t ? cap_t - excl : 0;
        if (lim > c0) lim = c0;
        dst = prim ? ((u64 *)offs + t * UA_STAGE_P)
                   : (staging + t * stage_stride + excl);
        if (prim) dst += excl;
        if (tid == 0)
            tile_cnt[t] = total < (u32)stage_stride ? total : (u32)stage_stride;
    } else { /* MODE_WRITE: (offs, partials) is the split flat scan */
        dst = d.out +
              (d_off(offs, partials, t) - d_off(offs, partials, d.tile_base)) + excl;
    }
    if (cnt > 0) {
        if (UA_WALK3 && (OP == OP_INTERSECT || OP == OP_DIFF)) {
            /* reconstruct emitted A-values from the per-chain masks */
            constexpr int H = UA_WPT / 2;
            u32 f = flags;
            u32 k = 0;
            while (f && k < lim) {
                int s = __builtin_ctz(f);
                f &= f - 1;
                int idx = (s < H)
                              ? w_i0 + __popc(amA & ((1u << s) - 1))
                              : w_i0b + __popc(amB & ((1u << (s - H)) - 1));
                dst[k++] = smem[UA_PX(aoff + idx)];
            }
        } else {
            u32 k = 0;
#pragma unroll
            for (int s = 0; s < UA_WPT; s++) {
                if (flags & (1u << s)) {
                    if (k < lim) dst[k] = em[s];
                    k++;
                }
            }
        }
    }
}

/* ---- 2-tile software-pipelined variant: one workgroup processes tiles
 * (2w, 2w+1); the global loads for tile 2w+1 are issued into registers
 * before the walk of tile 2w, so HBM latency hides under compute.  Double
 * LDS buffer (2x16.4KB -> 4 WGs/CU).  Toggle: UA_PIPE. ---- */

struct TileMeta {
    u32 p;
    u32 a0, b0;
    int alen, blen;
    bool valid, has_ab, has_bn;
    UaDesc d;
};

__device__ __forceinline__ TileMeta d_tile_meta(const UaDesc *__restrict__ descs,
                                                const u32 *__restrict__ tile_pair,
                                                const u32 *__restrict__ tile_a0,
                                                u64 total_tiles, u64 t) {
    TileMeta m;
    m.valid = (t < total_tiles);
    if (!m.valid) {
        m.alen = m.blen = 0;
        m.a0 = m.b0 = 0;
        m.has_ab = m.has_bn = false;
        return m;
    }
    m.p = tile_pair[t];
    m.d = descs[m.p];
    u64 lt = t - m.d.tile_base;
    u64 path = m.d.n + m.d.m;
    u64 d0 = lt * UA_TILE;
    if (d0 > path) d0 = path;
    u64 d1 = d0 + UA_TILE;
    if (d1 > path) d1 = path;
    u32 a1 = (t + 1 < total_tiles && tile_pair[t + 1] == m.p) ? tile_a0[t + 1]
                                                              : (u32)m.d.n;
    m.a0 = tile_a0[t];
    m.b0 = (u32)(d0 - m.a0);
    u32 b1 = (u32)(d1 - a1);
    m.alen = (int)(a1 - m.a0);
    m.blen = (int)(b1 - m.b0);
    m.has_ab = (m.a0 > 0);
    m.has_bn = ((u64)b1 < m.d.m);
    return m;
}

/* unified A|B element load into registers (one code path, coalesced) */
#define UA_PIPE_REGS (UA_WPT)
__device__ __forceinline__ void d_reg_load(const TileMeta &m, int tid,
                                           u64 (&val)[UA_PIPE_REGS], u64 &abefore,
                                           u64 &bbefore, u64 &bnext) {
#pragma unroll
    for (int k = 0; k < UA_PIPE_REGS; k++) {
        int e = tid + k * UA_BLOCK;
        u64 x = 0;
        if (e < m.alen) x = m.d.u[m.a0 + e];
        else if (e - m.alen < m.blen) x = m.d.v[m.b0 + (e - m.alen)];
        val[k] = x;
    }
    if (tid == 0) {
        abefore = m.has_ab ? m.d.u[m.a0 - 1] : 0;
        bbefore = (m.b0 > 0) ? m.d.v[m.b0 - 1] : 0;
        bnext = m.has_bn ? m.d.v[m.b0 + m.blen] : 0;
    }
}

__device__ __forceinline__ void d_reg_commit(const TileMeta &m, int tid, u64 *smem,
                                             const u64 (&val)[UA_PIPE_REGS],
                                             u64 abefore, u64 bbefore, u64 bnext,
                                             u64 *s_abefore, u64 *s_bbefore) {
    int boff = (m.alen + 1) & ~1;
#pragma unroll
    for (int k = 0; k < UA_PIPE_REGS; k++) {
        int e = tid + k * UA_BLOCK;
        if (e < m.alen) smem[e] = val[k];
        else if (e - m.alen < m.blen) smem[boff + (e - m.alen)] = val[k];
    }
    if (tid == 0) {
        *s_abefore = abefore;
        *s_bbefore = bbefore;
        smem[boff + m.blen] = bnext; /* lookahead slot (garbage if !has_bn) */
    }
}

template <int OP, int MODE>
__device__ __forceinline__ void d_tile_body(const TileMeta &m, u64 t, int tid,
                                            const u64 *smem, u64 a_before, u64 b_before,
                                            u64 *__restrict__ staging, u64 stage_stride,
                                            u32 *__restrict__ tile_cnt,
                                            const u64 *__restrict__ offs,
                                            const u64 *__restrict__ partials,
                                            u32 *scanbuf) {
    int alen = m.alen, blen = m.blen;
    int boff = (alen + 1) & ~1;
    int tilelen = alen + blen;
    int s0 = tid * UA_WPT;
    int s1 = s0 + UA_WPT;
    if (s0 > tilelen) s0 = tilelen;
    if (s1 > tilelen) s1 = tilelen;
    u64 em[UA_WPT];
    u32 flags = 0;
    int cnt = 0;
    if (m.valid) {
        int i0 = d_merge_path_lds(smem, alen, smem + boff, blen, s0);
        cnt = tile_walk2<OP>(smem, 0, alen, boff, blen, a_before, m.has_ab, b_before,
                             m.b0 > 0, m.has_bn, s0, s1, i0, em, flags);
    }
    u32 excl, total;
    d_block_scan<UA_BLOCK>(tid, (u32)cnt, scanbuf, excl, total);
    if (!m.valid) return;
    if (MODE == MODE_COUNT) {
        if (tid == 0) tile_cnt[t] = total;
        return;
    }
    u64 *dst;
    if (MODE == MODE_STAGE) {
        dst = staging + t * stage_stride + excl;
        if (tid == 0) tile_cnt[t] = total;
    } else {
        dst = m.d.out + (d_off(offs, partials, t) - d_off(offs, partials, m.d.tile_base)) +
              excl;
    }
    if (cnt > 0) {
        int k = 0;
#pragma unroll
        for (int s = 0; s < UA_WPT; s++) {
            if (flags & (1u << s)) dst[k++] = em[s];
        }
    }
}

template <int OP, int MODE>
__global__ __launch_bounds__(UA_BLOCK) void k_tiles_pipe(
    const UaDesc *__restrict__ descs, const u32 *__restrict__ tile_pair,
    const u32 *__restrict__ tile_a0, u64 total_tiles,
    u64 *__restrict__ staging, u64 stage_stride, u32 *__restrict__ tile_cnt,
    const u64 *__restrict__ offs, const u64 *__restrict__ partials) {
    __shared__ __align__(16) u64 smem[2][UA_TILE + 4];
    __shared__ u32 scanbuf[2][UA_BLOCK / 64];
    __shared__ u64 s_abefore[2];
    __shared__ u64 s_bbefore[2];

    int tid = threadIdx.x;
    u64 t0 = (u64)blockIdx.x * 2;
    u64 t1 = t0 + 1;
    TileMeta m0 = d_tile_meta(descs, tile_pair, tile_a0, total_tiles, t0);
    TileMeta m1 = d_tile_meta(descs, tile_pair, tile_a0, total_tiles, t1);

    u64 v0[UA_PIPE_REGS], ab0 = 0, bb0 = 0, bn0 = 0;
    d_reg_load(m0, tid, v0, ab0, bb0, bn0);
    d_reg_commit(m0, tid, smem[0], v0, ab0, bb0, bn0, &s_abefore[0], &s_bbefore[0]);
    /* issue tile-1 loads NOW: they stay in flight across the barrier and
     * the tile-0 walk */
    u64 v1[UA_PIPE_REGS], ab1 = 0, bb1 = 0, bn1 = 0;
    if (m1.valid) d_reg_load(m1, tid, v1, ab1, bb1, bn1);
    __syncthreads();

    d_tile_body<OP, MODE>(m0, t0, tid, smem[0], s_abefore[0], s_bbefore[0], staging,
                          stage_stride, tile_cnt, offs, partials, scanbuf[0]);
    if (!m1.valid) return;
    d_reg_commit(m1, tid, smem[1], v1, ab1, bb1, bn1, &s_abefore[1], &s_bbefore[1]);
    __syncthreads();
    d_tile_body<OP, MODE>(m1, t1, tid, smem[1], s_abefore[1], s_bbefore[1], staging,
                          stage_stride, tile_cnt, offs, partials, scanbuf[1]);
}


/* ---- 2-buffer glds pipeline (UA_PIPE2): persistent workgroups grid-stride
 * over tiles; the NEXT tile's LDS-DMA fill is issued before walking the
 * current one, and drained only at the iteration-end raw barrier — the
 * guide's canonical glds 2-buffer overlap.  Scalar head/tail/boundary loads
 * are ordered BEFORE the glds issue (an ordinary load-result use after a
 * glds makes hipcc drain vmcnt(0) early — guide §5 trap 4b).  LDS 2x16.4KB
 * -> 4 WGs/CU. ---- */

struct P2Fill {
    int aoff, boff;
};

__device__ __forceinline__ P2Fill d_p2_issue(const TileMeta &m, u64 *buf, int tid,
                                             u64 *s_ab, u64 *s_bb, int for_union) {
    P2Fill f{0, 0};
    if (!m.valid) return f;
    int ashift = (int)((((uintptr_t)(m.d.u + m.a0)) >> 3) & 1);
    int bshift = (int)((((uintptr_t)(m.d.v + m.b0)) >> 3) & 1);
    f.aoff = ashift;
    f.boff = ((ashift + m.alen + 1) & ~1) + bshift;
    int wv4 = tid >> 6, lane = tid & 63;
    int ahead = ashift < m.alen ? ashift : m.alen;
    int abody = (m.alen - ahead) & ~127;
    int bhead = bshift < m.blen ? bshift : m.blen;
    int bbody = (m.blen - bhead) & ~127;
    /* scalar pieces FIRST (ordinary loads complete before any glds issues) */
    for (int i = ahead + abody + tid; i < m.alen; i += UA_BLOCK)
        buf[f.aoff + i] = m.d.u[m.a0 + i];
    if (tid < ahead) buf[f.aoff + tid] = m.d.u[m.a0 + tid];
    for (int i = bhead + bbody + tid; i < m.blen; i += UA_BLOCK)
        buf[f.boff + i] = m.d.v[m.b0 + i];
    if (tid < bhead) buf[f.boff + tid] = m.d.v[m.b0 + tid];
    if (tid == 0) {
        *s_ab = m.has_ab ? m.d.u[m.a0 - 1] : 0;
        if (for_union) *s_bb = (m.b0 > 0) ? m.d.v[m.b0 - 1] : 0;
        buf[f.boff + m.blen] = m.has_bn ? m.d.v[m.b0 + m.blen] : 0;
    }
    /* LDS-DMA body LAST */
    for (int e = wv4 * 128; e < abody; e += 4 * 128)
        __builtin_amdgcn_global_load_lds((const u32 *)(m.d.u + m.a0 + ahead + e + lane * 2),
                                         (u32 *)&buf[f.aoff + ahead + e], 16, 0, 0);
    for (int e = wv4 * 128; e < bbody; e += 4 * 128)
        __builtin_amdgcn_global_load_lds((const u32 *)(m.d.v + m.b0 + bhead + e + lane * 2),
                                         (u32 *)&buf[f.boff + bhead + e], 16, 0, 0);
    return f;
}

/* block scan with a RAW barrier (no vmcnt drain: a prefetch glds may be in
 * flight during the scan) */
__device__ __forceinline__ void d_block_scan_raw(int tid, u32 cnt, u32 *wsum,
                                                 u32 &excl, u32 &total) {
    int lane = tid & 63, wv = tid >> 6;
    u32 incl = cnt;
#pragma unroll
    for (int o = 1; o < 64; o <<= 1) {
        u32 x = __shfl_up(incl, o);
        if (lane >= o) incl += x;
    }
    if (lane == 63) wsum[wv] = incl;
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    u32 wbase = 0;
#pragma unroll
    for (int w = 0; w < UA_BLOCK / 64; w++) {
        u32 sv = wsum[w];
        if (w < wv) wbase += sv;
    }
    total = wsum[0] + wsum[1] + wsum[2] + wsum[3];
    excl = wbase + incl - cnt;
}

template <int OP, int MODE>
__global__ __launch_bounds__(UA_BLOCK) void k_tiles_p2(
    const UaDesc *__restrict__ descs, const u32 *__restrict__ tile_pair,
    const u32 *__restrict__ tile_a0, u64 total_tiles,
    u64 *__restrict__ staging, u64 stage_stride, u32 *__restrict__ tile_cnt,
    const u64 *__restrict__ offs, const u64 *__restrict__ partials) {
    __shared__ __align__(16) u64 smem[2][UA_TILE + 4];
    __shared__ u32 wsum[UA_BLOCK / 64];
    __shared__ u64 s_ab[2], s_bb[2];

    int tid = threadIdx.x;
    u64 G = gridDim.x;
    u64 t = blockIdx.x;
    if (t >= total_tiles) return;

    TileMeta m = d_tile_meta(descs, tile_pair, tile_a0, total_tiles, t);
    P2Fill f = d_p2_issue(m, smem[0], tid, &s_ab[0], &s_bb[0], OP == OP_UNION);
    asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();

    int cur = 0;
    for (; t < total_tiles; t += G, cur ^= 1) {
        /* issue the NEXT tile's fill into the other buffer; it stays in
         * flight across this tile's walk */
        u64 tn = t + G;
        TileMeta mn;
        P2Fill fn{0, 0};
        mn.valid = false;
        if (tn < total_tiles) {
            mn = d_tile_meta(descs, tile_pair, tile_a0, total_tiles, tn);
            fn = d_p2_issue(mn, smem[cur ^ 1], tid, &s_ab[cur ^ 1], &s_bb[cur ^ 1],
                            OP == OP_UNION);
        }

        u64 *buf = smem[cur];
        int alen = m.alen, blen = m.blen;
        int tilelen = alen + blen;
        int s0 = tid * UA_WPT;
        int s1 = s0 + UA_WPT;
        if (s0 > tilelen) s0 = tilelen;
        if (s1 > tilelen) s1 = tilelen;
        int i0 = d_merge_path_lds(buf + f.aoff, alen, buf + f.boff, blen, s0);
        u64 em[UA_WPT];
        u32 flags;
        int cnt = tile_walk2<OP>(buf, f.aoff, alen, f.boff, blen, s_ab[cur], m.has_ab,
                                 s_bb[cur], m.b0 > 0, m.has_bn, s0, s1, i0, em, flags);

        if (MODE == MODE_DIRECT) {
            u64 lt = t - m.d.tile_base;
            u64 d0 = lt * UA_TILE;
            u64 *dst = m.d.out + d0 + (u64)s0;
            int steps = s1 - s0;
#pragma unroll
            for (int q = 0; q < UA_WPT; q++)
                if (q < steps) dst[q] = em[q];
        } else {
            u32 excl, total;
            d_block_scan_raw(tid, (u32)cnt, wsum, excl, total);
            if (MODE == MODE_COUNT) {
                if (tid == 0) tile_cnt[t] = total;
            } else {
                u64 *dst;
                if (MODE == MODE_STAGE) {
                    dst = staging + t * stage_stride + excl;
                    if (tid == 0) tile_cnt[t] = total;
                } else {
                    dst = m.d.out +
                          (d_off(offs, partials, t) - d_off(offs, partials, m.d.tile_base)) +
                          excl;
                }
                if (cnt > 0) {
                    int k = 0;
#pragma unroll
                    for (int q = 0; q < UA_WPT; q++)
                        if (flags & (1u << q)) dst[k++] = em[q];
                }
            }
        }
        /* drain the prefetch + publish this iteration's LDS writes */
        asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
        m = mn;
        f = fn;
    }
}

/* ---- register-staged single-buffer pipeline (UA_RPIPE, runtime env) ----
 *
 * The two round-1 pipeline rejects both HALVED residency (double LDS buffer
 * -> 4 WGs/CU); this variant keeps ONE tile-sized LDS buffer and stages the
 * NEXT tile in registers instead: 512-thread persistent workgroups
 * grid-stride over tiles; each iteration ds_writes the staged registers
 * into LDS, immediately issues the 16-B global loads for tile t+G, and only
 * then walks — so the next tile's HBM latency hides under this tile's
 * search+walk, with LDS residency unchanged.  All barriers are RAW
 * (s_waitcnt lgkmcnt only): __syncthreads fences global too and would drain
 * vmcnt, killing the in-flight prefetch (the presumed reason UA_PIPE saw no
 * overlap).  STAGE/COUNT/WRITE/DIRECT only — LOOKBACK spins on global
 * atomics whose vmcnt waits would serialize the prefetch, and its
 * inter-block dependence needs grid <= residency; it keeps k_tiles. */

#define RP_BLOCK 512
#define RP_WPT (UA_TILE / RP_BLOCK)

/* global-address-space loads: the staged pointers come out of UaDesc (a
 * value loaded from memory), so hipcc emits FLAT loads for them — and on
 * gfx9-family flat ops count on lgkmcnt too, which would make every raw
 * lgkmcnt barrier wait for the in-flight prefetch.  The addrspacecast pins
 * them to global_load_* (vmcnt only). */
#define UA_AS_GLOBAL __attribute__((address_space(1)))
typedef u64 u64x2 __attribute__((ext_vector_type(2)));
__device__ __forceinline__ u64x2 d_gload16(const u64 *p) {
    return *(const UA_AS_GLOBAL u64x2 *)(uintptr_t)p;
}
__device__ __forceinline__ u64 d_gload8(const u64 *p) {
    return *(const UA_AS_GLOBAL u64 *)(uintptr_t)p;
}

/* Issue the global loads for tile m into registers (two 16-B body units +
 * one lane-designated single: 0=a_before 1=b_before 2=b_next 3=a_head
 * 4=b_head 5=a_tail 6=b_tail).  Bodies are the 16-B aligned middles (same
 * parity layout as the glds fill: As = smem+ashift, Bs = smem+boff so LDS
 * commits are 16-B aligned too).  Bare locals, not a struct: hipcc spills a
 * loop-carried struct of vectors to scratch.  The single is ONE
 * address-selected load, not a switch of loads: exec-masked case loads
 * into one register WAW-serialize on vmcnt(0) drains. */
__device__ __forceinline__ void d_rp_load(const TileMeta &m, int tid, u64x2 &st0,
                                          u64x2 &st1, u64 &sng, const u64 *safe) {
    if (!m.valid) return;
    int ash = (int)((((uintptr_t)(m.d.u + m.a0)) >> 3) & 1);
    int bsh = (int)((((uintptr_t)(m.d.v + m.b0)) >> 3) & 1);
    int ah = ash < m.alen ? ash : m.alen;
    int bh = bsh < m.blen ? bsh : m.blen;
    int na2 = (m.alen - ah) >> 1, atail = (m.alen - ah) & 1;
    int nb2 = (m.blen - bh) >> 1, btail = (m.blen - bh) & 1;
    /* branchless single load per unit: exec-masked per-branch loads into one
     * register WAW-serialize on vmcnt drains; out-of-range lanes read `safe`
     * (a broadcast dummy) instead */
    {
        int u = tid;
        const u64 *p = (u < na2)         ? m.d.u + m.a0 + ah + 2 * u
                       : (u - na2 < nb2) ? m.d.v + m.b0 + bh + 2 * (u - na2)
                                         : safe;
        st0 = d_gload16(p);
    }
    {
        int u = tid + RP_BLOCK;
        const u64 *p = (u < na2)         ? m.d.u + m.a0 + ah + 2 * u
                       : (u - na2 < nb2) ? m.d.v + m.b0 + bh + 2 * (u - na2)
                                         : safe;
        st1 = d_gload16(p);
    }
    const u64 *sp = safe;
    switch (tid) {
    case 0: if (m.has_ab) sp = m.d.u + m.a0 - 1; break;
    case 1: if (m.b0 > 0) sp = m.d.v + m.b0 - 1; break;
    case 2: if (m.has_bn) sp = m.d.v + m.b0 + m.blen; break;
    case 3: if (ah) sp = m.d.u + m.a0; break;
    case 4: if (bh) sp = m.d.v + m.b0; break;
    case 5: if (atail) sp = m.d.u + m.a0 + m.alen - 1; break;
    case 6: if (btail) sp = m.d.v + m.b0 + m.blen - 1; break;
    default: break;
    }
    /* opaque barrier on sp: keep hipcc from sinking one load per case */
    asm volatile("" : "+v"(sp));
    /* raw value; the not-applicable zeroing happens at COMMIT time so no
     * instruction here consumes the load (a use would emit a vmcnt drain
     * in front of the walk and kill the whole prefetch) */
    sng = d_gload8(sp);
}

/* ds_write the staged tile into LDS (consumes the staged regs: the
 * compiler's counted vmcnt waits land here, not at a barrier). */
__device__ __forceinline__ void d_rp_commit(const TileMeta &m, int tid, u64 *smem,
                                            u64x2 st0, u64x2 st1, u64 sng,
                                            int &aoff, int &boff, u64 *s_ab,
                                            u64 *s_bb) {
    int ash = (int)((((uintptr_t)(m.d.u + m.a0)) >> 3) & 1);
    int bsh = (int)((((uintptr_t)(m.d.v + m.b0)) >> 3) & 1);
    aoff = ash;
    boff = ((ash + m.alen + 1) & ~1) + bsh;
    if (!m.valid) return;
    int ah = ash < m.alen ? ash : m.alen;
    int bh = bsh < m.blen ? bsh : m.blen;
    int na2 = (m.alen - ah) >> 1, atail = (m.alen - ah) & 1;
    int nb2 = (m.blen - bh) >> 1, btail = (m.blen - bh) & 1;
    {
        int u = tid;
        if (u < na2)
            *(u64x2 *)&smem[aoff + ah + 2 * u] = st0;
        else if (u - na2 < nb2)
            *(u64x2 *)&smem[boff + bh + 2 * (u - na2)] = st0;
    }
    {
        int u = tid + RP_BLOCK;
        if (u < na2)
            *(u64x2 *)&smem[aoff + ah + 2 * u] = st1;
        else if (u - na2 < nb2)
            *(u64x2 *)&smem[boff + bh + 2 * (u - na2)] = st1;
    }
    /* singles: zero the not-applicable cases here (deferred from load) */
    switch (tid) {
    case 0: *s_ab = m.has_ab ? sng : 0; break;
    case 1: *s_bb = (m.b0 > 0) ? sng : 0; break;
    case 2: smem[boff + m.blen] = m.has_bn ? sng : 0; break; /* lookahead */
    case 3: if (ah) smem[aoff] = sng; break;
    case 4: if (bh) smem[boff] = sng; break;
    case 5: if (atail) smem[aoff + m.alen - 1] = sng; break;
    case 6: if (btail) smem[boff + m.blen - 1] = sng; break;
    default: break;
    }
}

/* block scan for NB threads with a RAW barrier (no vmcnt drain) */
template <int NB>
__device__ __forceinline__ void d_block_scan_rawN(int tid, u32 cnt, u32 *wsum,
                                                  u32 &excl, u32 &total) {
    int lane = tid & 63, wv = tid >> 6;
    u32 incl = cnt;
#pragma unroll
    for (int o = 1; o < 64; o <<= 1) {
        u32 x = __shfl_up(incl, o);
        if (lane >= o) incl += x;
    }
    if (lane == 63) wsum[wv] = incl;
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    u32 wbase = 0, tot = 0;
#pragma unroll
    for (int w = 0; w < NB / 64; w++) {
        u32 sv = wsum[w];
        if (w < wv) wbase += sv;
        tot += sv;
    }
    total = tot;
    excl = wbase + incl - cnt;
}

/* d_tile_meta with the pair descriptor cached from the previous
 * (consecutive) tile: descs[p] is a 3-line struct load that would otherwise
 * sit as a serial L2 round trip at every loop head. */
__device__ __forceinline__ TileMeta d_tile_meta_seq(const UaDesc *__restrict__ descs,
                                                    const u32 *__restrict__ tile_pair,
                                                    const u32 *__restrict__ tile_a0,
                                                    u64 total_tiles, u64 t, u64 tend,
                                                    const TileMeta &prev) {
    TileMeta m;
    m.valid = (t < tend);
    if (!m.valid) {
        m.alen = m.blen = 0;
        m.a0 = m.b0 = 0;
        m.has_ab = m.has_bn = false;
        return m;
    }
    m.p = tile_pair[t];
    if (prev.valid && prev.p == m.p) m.d = prev.d;
    else m.d = descs[m.p];
    u64 lt = t - m.d.tile_base;
    u64 path = m.d.n + m.d.m;
    u64 d0 = lt * UA_TILE;
    if (d0 > path) d0 = path;
    u64 d1 = d0 + UA_TILE;
    if (d1 > path) d1 = path;
    u32 a1 = (t + 1 < total_tiles && tile_pair[t + 1] == m.p) ? tile_a0[t + 1]
                                                              : (u32)m.d.n;
    m.a0 = tile_a0[t];
    m.b0 = (u32)(d0 - m.a0);
    u32 b1 = (u32)(d1 - a1);
    m.alen = (int)(a1 - m.a0);
    m.blen = (int)(b1 - m.b0);
    m.has_ab = (m.a0 > 0);
    m.has_bn = ((u64)b1 < m.d.m);
    return m;
}

template <int OP, int MODE>
__global__ __launch_bounds__(RP_BLOCK, 8) void k_tiles_rp(
    const UaDesc *__restrict__ descs, const u32 *__restrict__ tile_pair,
    const u32 *__restrict__ tile_a0, u64 total_tiles,
    u64 *__restrict__ staging, u64 stage_stride, u32 *__restrict__ tile_cnt,
    const u64 *__restrict__ offs, const u64 *__restrict__ partials) {
    __shared__ __align__(16) u64 smem[UA_TILE + 4];
    __shared__ u32 wsum[RP_BLOCK / 64];
    __shared__ u64 s_ab, s_bb;

    int tid = threadIdx.x;
    /* contiguous chunk per workgroup: consecutive tiles keep the meta words
     * (tile_pair/tile_a0, u32) L1-hot and the pair descriptor reg-cached */
    u64 nb = gridDim.x;
    u64 len = (total_tiles + nb - 1) / nb;
    u64 t = (u64)blockIdx.x * len;
    u64 tend = t + len;
    if (tend > total_tiles) tend = total_tiles;
    if (t >= tend) return;

    TileMeta mz;
    mz.valid = false;
    TileMeta m = d_tile_meta_seq(descs, tile_pair, tile_a0, total_tiles, t, tend, mz);
    u64x2 st0 = {}, st1 = {};
    u64 sng = 0;
    const u64 *safe = (const u64 *)descs;
    d_rp_load(m, tid, st0, st1, sng, safe);

    for (; t < tend; t++) {
        TileMeta mn =
            d_tile_meta_seq(descs, tile_pair, tile_a0, total_tiles, t + 1, tend, m);
        int aoff, boff;
        d_rp_commit(m, tid, smem, st0, st1, sng, aoff, boff, &s_ab, &s_bb);
        /* pin the commit's stores BEFORE the next tile's load issue: left
         * free, the scheduler moves the singles ds_writes below the next
         * sng load sharing their register, and the resulting vmcnt(0)
         * drains stall wave 0 (and through the barrier, the whole WG) on
         * the prefetch */
        __builtin_amdgcn_sched_barrier(0);
        u64x2 nst0 = {}, nst1 = {};
        u64 nsng = 0;
        if (mn.valid) d_rp_load(mn, tid, nst0, nst1, nsng, safe); /* in flight across the walk */
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();

        int alen = m.alen, blen = m.blen;
        int tilelen = alen + blen;
        int s0 = tid * RP_WPT;
        int s1 = s0 + RP_WPT;
        if (s0 > tilelen) s0 = tilelen;
        if (s1 > tilelen) s1 = tilelen;
        u64 em[RP_WPT];
        u32 flags = 0;
        int cnt = 0;
        {
            const u64 *As = smem + aoff, *Bs = smem + boff;
            int i0 = d_merge_path_lds(As, alen, Bs, blen, s0);
            cnt = tile_walk2<OP>(smem, aoff, alen, boff, blen, s_ab, m.has_ab, s_bb,
                                 m.b0 > 0, m.has_bn, s0, s1, i0, em, flags);
        }
        if (MODE == MODE_DIRECT) {
            u64 lt = t - m.d.tile_base;
            u64 *dst = m.d.out + lt * UA_TILE + (u64)s0;
            int steps = s1 - s0;
#pragma unroll
            for (int q = 0; q < RP_WPT; q++)
                if (q < steps) dst[q] = em[q];
        } else {
            u32 excl, total;
            d_block_scan_rawN<RP_BLOCK>(tid, (u32)cnt, wsum, excl, total);
            if (MODE == MODE_COUNT) {
                if (tid == 0) tile_cnt[t] = total;
            } else {
                u64 *dst;
                u32 lim = (u32)cnt;
                if (MODE == MODE_STAGE) {
                    /* stride clamp: invalid (duplicate/unsorted) inputs may
                     * over-emit; keep writes in bounds */
                    u32 c0 = excl < (u32)stage_stride ? (u32)stage_stride - excl : 0;
                    if (lim > c0) lim = c0;
                    dst = staging + t * stage_stride + excl;
                    if (tid == 0)
                        tile_cnt[t] = total < (u32)stage_stride ? total
                                                                : (u32)stage_stride;
                } else { /* MODE_WRITE: consumed only after the walk, so the
                            d_off loads draining the prefetch costs little */
                    dst = m.d.out +
                          (d_off(offs, partials, t) - d_off(offs, partials, m.d.tile_base)) +
                          excl;
                }
                if (cnt > 0) {
                    u32 k = 0;
#pragma unroll
                    for (int q = 0; q < RP_WPT; q++) {
                        if (flags & (1u << q)) {
                            if (k < lim) dst[k] = em[q];
                            k++;
                        }
                    }
                }
            }
        }
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier(); /* LDS reads done before the next commit */
        /* keep the register shift BELOW the walk: its copies consume the
         * prefetched values, and hoisted above the barrier they would drain
         * vmcnt before the walk ever starts */
        __builtin_amdgcn_sched_barrier(0);
        m = mn;
        st0 = nst0;
        st1 = nst1;
        sng = nsng;
    }
}


/* ---- double-buffered glds pipeline at FULL occupancy (UA_PP, runtime env;
 * build with -DUA_TILE=1024 so two buffers fit the round-1 LDS budget) ----
 *
 * PIPE2's structure (issue the next tile's LDS-DMA before walking the
 * current, one vmcnt(0) at the swap) was rejected in round 1 at 4-wave WGs
 * = 16 waves/CU.  At UA_TILE=1024 both buffers fit in 16.4 KB, so 8 WGs x
 * 4 waves keep the full 32 waves/CU AND every WG always has a fill in
 * flight.  This attacks the measured convoy loss: fill-only 0.474 ms,
 * walk-only 0.476, but fill+walk 0.652 — the equal-share HBM service
 * synchronizes fill completions across WGs, so whole CUs alternate
 * all-fill / all-walk phases and HBM idles ~27% of the time.
 * STAGE/COUNT/WRITE/DIRECT only (LOOKBACK keeps k_tiles: its resolve spins
 * on global atomics whose waits would drain the in-flight DMA). */
template <int OP, int MODE>
__global__ __launch_bounds__(UA_TBLOCK, 8) void k_tiles_pp(
    const UaDesc *__restrict__ descs, const u32 *__restrict__ tile_pair,
    const u32 *__restrict__ tile_a0, u64 total_tiles,
    u64 *__restrict__ staging, u64 stage_stride, u32 *__restrict__ tile_cnt,
    const u64 *__restrict__ offs, const u64 *__restrict__ partials,
    const u32 *__restrict__ isplit_in, u32 *__restrict__ isplit_out) {
    static_assert(UA_TBLOCK == UA_BLOCK, "d_p2_issue strides are 4-wave");
    __shared__ __align__(16) u64 smem[2][UA_TILE + 4];
    __shared__ u32 scanb[UA_TBLOCK / 64];
    __shared__ u64 s_ab[2], s_bb[2];

    int tid = threadIdx.x;
    u64 nb = gridDim.x;
    u64 len = (total_tiles + nb - 1) / nb;
    u64 t = (u64)blockIdx.x * len;
    u64 tend = t + len;
    if (tend > total_tiles) tend = total_tiles;
    if (t >= tend) return;

    TileMeta mz;
    mz.valid = false;
    TileMeta m = d_tile_meta_seq(descs, tile_pair, tile_a0, total_tiles, t, tend, mz);
    P2Fill f = d_p2_issue(m, smem[0], tid, &s_ab[0], &s_bb[0], OP == OP_UNION);
    int cur = 0;
    asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();

    for (; t < tend; t++, cur ^= 1) {
        TileMeta mn =
            d_tile_meta_seq(descs, tile_pair, tile_a0, total_tiles, t + 1, tend, m);
        P2Fill fn{0, 0};
        if (mn.valid)
            fn = d_p2_issue(mn, smem[cur ^ 1], tid, &s_ab[cur ^ 1], &s_bb[cur ^ 1],
                            OP == OP_UNION); /* DMA in flight across the walk */

        const u64 *buf = smem[cur];
        int alen = m.alen, blen = m.blen;
        int tilelen = alen + blen;
        int s0 = tid * UA_WPT;
        int s1 = s0 + UA_WPT;
        if (s0 > tilelen) s0 = tilelen;
        if (s1 > tilelen) s1 = tilelen;
        u64 em[UA_WPT];
        u32 flags = 0, amask = 0;
        int cnt = 0, w_i0 = 0;
        int i0 = isplit_in ? (int)(isplit_in[t * UA_TBLOCK + tid] & 0xffffu)
                           : d_merge_path_px(buf, f.aoff, alen, f.boff, blen, s0);
        (void)isplit_out; /* pp never participates in the cache */
        if (UA_WALK3 && (OP == OP_INTERSECT || OP == OP_DIFF)) {
            cnt = tile_walk3<OP, UA_WPT>(buf, f.aoff, alen, f.boff, blen, m.has_bn,
                                         s0, s1, i0, flags, amask);
            w_i0 = i0;
        } else {
            cnt = tile_walk2<OP>(buf, f.aoff, alen, f.boff, blen, s_ab[cur], m.has_ab,
                                 s_bb[cur], m.b0 > 0, m.has_bn, s0, s1, i0, em, flags);
        }
        if (MODE == MODE_DIRECT) {
            u64 lt = t - m.d.tile_base;
            u64 *dst = m.d.out + lt * UA_TILE + (u64)s0;
            int steps = s1 - s0;
#pragma unroll
            for (int q = 0; q < UA_WPT; q++)
                if (q < steps) dst[q] = em[q];
        } else {
            u32 excl, total;
            d_block_scan_rawN<UA_TBLOCK>(tid, (u32)cnt, scanb, excl, total);
            if (MODE == MODE_COUNT) {
                if (tid == 0) tile_cnt[t] = total;
            } else {
                u64 *dst;
                u32 lim = (u32)cnt;
                if (MODE == MODE_STAGE) {
                    /* stride clamp as in k_tiles */
                    u32 c0 = excl < (u32)stage_stride ? (u32)stage_stride - excl : 0;
                    if (lim > c0) lim = c0;
                    dst = staging + t * stage_stride + excl;
                    if (tid == 0)
                        tile_cnt[t] = total < (u32)stage_stride ? total
                                                                : (u32)stage_stride;
                } else { /* MODE_WRITE: post-walk d_off loads may drain the
                            DMA, but it has already had the walk to land */
                    dst = m.d.out +
                          (d_off(offs, partials, t) - d_off(offs, partials, m.d.tile_base)) +
                          excl;
                }
                if (cnt > 0) {
                    if (UA_WALK3 && (OP == OP_INTERSECT || OP == OP_DIFF)) {
                        u32 fl = flags;
                        u32 k = 0;
                        while (fl && k < lim) {
                            int s = __builtin_ctz(fl);
                            fl &= fl - 1;
                            int idx = w_i0 + __popc(amask & ((1u << s) - 1));
                            dst[k++] = buf[UA_PX(f.aoff + idx)];
                        }
                    } else {
                        u32 k = 0;
#pragma unroll
                        for (int s = 0; s < UA_WPT; s++) {
                            if (flags & (1u << s)) {
                                if (k < lim) dst[k] = em[s];
                                k++;
                            }
                        }
                    }
                }
            }
        }
        /* drain the next tile's DMA (walk covered its latency) + make sure
         * every wave is done reading this buffer before it is refilled */
        asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
        m = mn;
        f = fn;
    }
}


/* ==================== wave-register intersect (UA_AISECT) ====================
 *
 * Intersect/difference WITHOUT the merge-path walk: A is cut into 512-
 * element wave-tiles (A-indexed, so output order is A order and no
 * per-thread diagonal search exists at all); each WAVE streams 64-wide
 * windows of A and B through REGISTERS (coalesced global loads, no LDS,
 * no barriers) and tests membership with a 6-round shuffle binary search
 * over the B window held one element per lane.  Window advance: A consumes
 * elements <= the window's B max (their membership is fully decided), B
 * consumes elements <= the window's A max (they can never match later A,
 * which is strictly larger).  The phase ablations that motivated this:
 * fill-only 0.474 ms, walk-only 0.476, per-block floor 0.045 — the
 * merge-path kernel's compute phase cost as much as the HBM transfer, and
 * nearly all of it was the per-thread LDS search+walk this design deletes.
 * Aux tail (scan/compact/pair_out) is reused as-is over the A-tile layout
 * (stride UA_AT). */

#define UA_AT 512 /* A elements per wave-tile */

/* per A-tile: owning pair + the lower_bound of its first A value in B
 * (computed once at batch create, like the merge-path partition) */
__global__ __launch_bounds__(UA_BLOCK) void k_apartition(
    const UaDesc *__restrict__ descs, const u64 *__restrict__ tba, int n_pairs,
    u64 total_atiles, u32 *__restrict__ tpair_a, u32 *__restrict__ bstart_a) {
    u64 t = (u64)blockIdx.x * UA_BLOCK + threadIdx.x;
    if (t >= total_atiles) return;
    int lo = 0, hi = n_pairs - 1;
    while (lo < hi) {
        int mid = (lo + hi + 1) >> 1;
        if (tba[mid] <= t) lo = mid;
        else hi = mid - 1;
    }
    int p = lo;
    tpair_a[t] = (u32)p;
    UaDesc d = descs[p];
    u64 a_base = (t - tba[p]) * UA_AT;
    u64 key = (a_base < d.n) ? d.u[a_base] : ~0ull;
    bstart_a[t] = (u32)d_lower_bound(d.v, d.m, key);
}

/* lower_bound of a within the b values held one per lane (lanes [0, nb)):
 * 6 fixed rounds of varying-lane shuffles (ds_bpermute). */
__device__ __forceinline__ int d_wave_lb(u64 b_l, int nb, u64 a) {
    int pos = 0;
#pragma unroll
    for (int step = 32; step >= 1; step >>= 1) {
        int cand = pos + step;
        u64 bm = __shfl(b_l, cand - 1);
        if (cand <= nb && bm < a) pos = cand;
    }
    return pos; /* = count of b < a */
}

template <int OP>
__global__ __launch_bounds__(UA_BLOCK) void k_aisect(
    const UaDesc *__restrict__ descs, const u32 *__restrict__ tpair_a,
    const u64 *__restrict__ tba, const u32 *__restrict__ bstart_a,
    u64 total_atiles, u64 *__restrict__ staging, u32 *__restrict__ tile_cnt) {
    static_assert(OP == OP_INTERSECT || OP == OP_DIFF, "A-indexed ops only");
    u64 t = (u64)blockIdx.x * (UA_BLOCK / 64) + (threadIdx.x >> 6);
    int lane = threadIdx.x & 63;
    if (t >= total_atiles) return;
    u32 p = tpair_a[t];
    UaDesc d = descs[p];
    u64 a_base = (t - tba[p]) * UA_AT;
    if (a_base >= d.n) { /* empty-A pair tail */
        if (lane == 0) tile_cnt[t] = 0;
        return;
    }
    u64 ia = a_base;
    u64 ia_end = a_base + UA_AT;
    if (ia_end > d.n) ia_end = d.n;
    u64 ib = bstart_a[t];
    u64 m = d.m;
    u64 *dst = staging + t * UA_AT;
    u32 running = 0;

    /* 3-deep register FIFO per side: regs xk[lane] = X[clamp(base+64k+lane)]
     * for k=0..2; the current 64-wide window [w, w+64) satisfies
     * w - base < 64, so a window value is a 2-shuffle select over x0/x1 and
     * x2 is PURE PREFETCH (its load latency hides behind ~a window of
     * compute before a shift consumes it).  Reloading per window put the
     * dependent load latency on the serial chain — measured 0.92 ms; the
     * FIFO takes the loads off the chain. */
    u64 baseA = ia, baseB = ib;
    u64 a0r = d.u[(baseA + (u64)lane < ia_end) ? baseA + (u64)lane : ia_end - 1];
    u64 a1r = d.u[(baseA + 64 + (u64)lane < ia_end) ? baseA + 64 + (u64)lane
                                                    : ia_end - 1];
    u64 a2r = d.u[(baseA + 128 + (u64)lane < ia_end) ? baseA + 128 + (u64)lane
                                                     : ia_end - 1];
    u64 b0r = 0, b1r = 0, b2r = 0;
    if (m > 0) {
        b0r = d.v[(baseB + (u64)lane < m) ? baseB + (u64)lane : m - 1];
        b1r = d.v[(baseB + 64 + (u64)lane < m) ? baseB + 64 + (u64)lane : m - 1];
        b2r = d.v[(baseB + 128 + (u64)lane < m) ? baseB + 128 + (u64)lane : m - 1];
    }

    while (ia < ia_end) {
        int na = (int)((ia_end - ia < 64) ? (ia_end - ia) : 64);
        int nb = (int)((m - ib < 64) ? (m - ib) : 64);
        bool valid_a = lane < na;
        /* materialize the windows: off in [0, 127), never reaches x2 */
        int offA = (int)(ia - baseA) + lane;
        u64 aw0 = __shfl(a0r, offA & 63);
        u64 aw1 = __shfl(a1r, offA & 63);
        u64 a = (offA < 64) ? aw0 : aw1;
        bool b_exh = (ib + (u64)nb >= m);
        bool eq = false;
        u64 b_hi = 0;
        int cB = 0;
        if (nb > 0) {
            int offB = (int)(ib - baseB) + lane;
            u64 bw0 = __shfl(b0r, offB & 63);
            u64 bw1 = __shfl(b1r, offB & 63);
            u64 b = (offB < 64) ? bw0 : bw1;
            int pos = d_wave_lb(b, nb, a);
            /* UNCONDITIONAL shuffle: under a && short-circuit, lanes with
             * pos >= nb would go inactive, and a bpermute SOURCE lane that
             * is inactive in the instruction contributes garbage. */
            int posc = pos < nb ? pos : 0;
            u64 bv = __shfl(b, posc);
            eq = (pos < nb) && (bv == a);
            b_hi = __shfl(b, nb - 1);
            /* B consumes elements <= this window's A max: later A values
             * are strictly larger (duplicate-free contract), so those b
             * can never match again */
            u64 a_hi = __shfl(a, na - 1);
            u64 bm = __ballot((lane < nb) && (b <= a_hi));
            cB = __popcll(bm);
        }
        bool consume = valid_a && (b_exh || a <= b_hi);
        bool emit = (OP == OP_INTERSECT) ? (consume && eq) : (consume && !eq);
        u64 em = __ballot(emit);
        if (emit) {
            int rank = __popcll(em & ((1ull << lane) - 1));
            dst[running + rank] = a;
        }
        running += (u32)__popcll(em);
        u64 cm = __ballot(consume);
        int cA = __popcll(cm);
        if (nb == 0 && cA == 0) break; /* no progress possible */
        ia += (u64)cA;
        ib += (u64)cB;
        if (ia - baseA >= 64) {
            baseA += 64;
            a0r = a1r;
            a1r = a2r;
            a2r = d.u[(baseA + 128 + (u64)lane < ia_end) ? baseA + 128 + (u64)lane
                                                         : ia_end - 1];
        }
        if (m > 0 && ib - baseB >= 64) {
            baseB += 64;
            b0r = b1r;
            b1r = b2r;
            b2r = d.v[(baseB + 128 + (u64)lane < m) ? baseB + 128 + (u64)lane
                                                    : m - 1];
        }
    }
    if (lane == 0) tile_cnt[t] = running;
}

/* ==================== kernel: bitonic chunk sort (segmented sort stage 1) ====================
 * One workgroup sorts one <=2048-element chunk in LDS (u64 ascending,
 * duplicates kept; padded with UINT64_MAX).  Stage 2 is the merge-path
 * OP_MERGE_ALL tree. */

#define UA_SORT_N 2048 /* bitonic chunk length: power of two, independent of UA_TILE */

struct UaChunk {
    const u64 *src;
    u64 *dst;
    u32 len;
    u32 pad_;
};

__global__ __launch_bounds__(UA_BLOCK) void k_sort_chunks(
    const UaChunk *__restrict__ chunks, u64 n_chunks) {
    __shared__ u64 s[UA_SORT_N];
    u64 cidx = blockIdx.x;
    if (cidx >= n_chunks) return;
    UaChunk ch = chunks[cidx];
    int tid = threadIdx.x;
    for (int i = tid; i < UA_SORT_N; i += UA_BLOCK)
        s[i] = (i < (int)ch.len) ? ch.src[i] : UINT64_MAX;
    __syncthreads();
    for (int k = 2; k <= UA_SORT_N; k <<= 1) {
        for (int j = k >> 1; j > 0; j >>= 1) {
            for (int i = tid; i < UA_SORT_N; i += UA_BLOCK) {
                int p = i ^ j;
                if (p > i) {
                    bool up = ((i & k) == 0);
                    u64 x = s[i], y = s[p];
                    if ((x > y) == up) {
                        s[i] = y;
                        s[p] = x;
                    }
                }
            }
            __syncthreads();
        }
    }
    for (int i = tid; i < (int)ch.len; i += UA_BLOCK) ch.dst[i] = s[i];
}

/* ==================== kernels: flat hierarchical scan (u32 -> u64) ==================== */

__global__ __launch_bounds__(UA_BLOCK) void k_scan1(const u32 *__restrict__ cnt, u64 n,
                                                    u64 *__restrict__ offs,
                                                    u64 *__restrict__ partials) {
    __shared__ u64 sblk[UA_BLOCK];
    u64 base = (u64)blockIdx.x * UA_SCAN_CHUNK;
    int tid = threadIdx.x;
    u64 loc[UA_SCAN_CHUNK / UA_BLOCK];
    u64 sum = 0;
    for (int j = 0; j < UA_SCAN_CHUNK / UA_BLOCK; j++) {
        u64 i = base + (u64)tid * (UA_SCAN_CHUNK / UA_BLOCK) + j;
        u32 c = (i < n) ? cnt[i] : 0;
        loc[j] = sum;
        sum += c;
    }
    sblk[tid] = sum;
    __syncthreads();
    for (int off = 1; off < UA_BLOCK; off <<= 1) {
        u64 x = (tid >= off) ? sblk[tid - off] : 0;
        __syncthreads();
        sblk[tid] += x;
        __syncthreads();
    }
    u64 excl = sblk[tid] - sum;
    for (int j = 0; j < UA_SCAN_CHUNK / UA_BLOCK; j++) {
        u64 i = base + (u64)tid * (UA_SCAN_CHUNK / UA_BLOCK) + j;
        if (i < n) offs[i] = excl + loc[j];
    }
    if (tid == UA_BLOCK - 1) partials[blockIdx.x] = sblk[UA_BLOCK - 1];
}

__global__ __launch_bounds__(UA_BLOCK) void k_scan2(u64 *__restrict__ partials, u64 nchunks) {
    __shared__ u64 sblk[UA_BLOCK];
    int tid = threadIdx.x;
    u64 carry = 0;
    for (u64 base = 0; base < nchunks; base += UA_BLOCK) {
        u64 i = base + tid;
        u64 x = (i < nchunks) ? partials[i] : 0;
        sblk[tid] = x;
        __syncthreads();
        for (int off = 1; off < UA_BLOCK; off <<= 1) {
            u64 y = (tid >= off) ? sblk[tid - off] : 0;
            __syncthreads();
            sblk[tid] += y;
            __syncthreads();
        }
        if (i < nchunks) partials[i] = carry + sblk[tid] - x;
        u64 tot = sblk[UA_BLOCK - 1];
        __syncthreads();
        carry += tot;
    }
}

__global__ __launch_bounds__(UA_BLOCK) void k_pair_out(const u64 *__restrict__ offs,
                                                       const u64 *__restrict__ partials,
                                                       const u64 *__restrict__ tb, int n_pairs,
                                                       u64 *__restrict__ pout) {
    int p = blockIdx.x * UA_BLOCK + threadIdx.x;
    if (p < n_pairs) pout[p] = d_off(offs, partials, tb[p + 1]) - d_off(offs, partials, tb[p]);
}

/* ==================== kernel: compaction ==================== */

/* one wavefront per tile, 4 tiles per wave, 16 per workgroup: few blocks to
 * dispatch, and tiles with cnt==0 cost one load */
#define UA_COMPACT_SMALL 32 /* cnt <= SMALL tiles go to the thread-per-tile
                              * compactor; the wave compactor keeps the rest
                              * (headline tiles emit ~10, so one thread per
                              * tile amortizes the 3-load dependent chain
                              * over 187k parallel threads instead of 47k
                              * waves) */

/* thread-per-tile compactor for small counts (1..UA_COMPACT_SMALL) */
__global__ __launch_bounds__(UA_BLOCK) void k_compact_small(
    const UaDesc *__restrict__ descs, const u32 *__restrict__ tile_pair,
    const u64 *__restrict__ tb, const u32 *__restrict__ tile_cnt,
    const u64 *__restrict__ offs, const u64 *__restrict__ partials,
    const u64 *__restrict__ staging, u64 stage_stride,
    const u64 *__restrict__ prim, u64 total_tiles, int op) {
    u64 t = (u64)blockIdx.x * UA_BLOCK + threadIdx.x;
    if (t >= total_tiles) return;
    u32 cnt = tile_cnt[t];
    if (cnt == 0 || cnt > UA_COMPACT_SMALL) return;
    u32 p = tile_pair[t];
    UaDesc d = descs[p];
    u64 goff = d_off(offs, partials, t) - d_off(offs, partials, tb[p]);
    u64 cap = (op == OP_INTERSECT) ? (d.n < d.m ? d.n : d.m)
              : (op == OP_DIFF) ? d.n
                                : (d.n + d.m);
    if (goff >= cap) return;
    u64 room = cap - goff;
    if ((u64)cnt > room) cnt = (u32)room;
    u64 *dst = d.out + goff;
    const u64 *src = prim ? prim + t * UA_STAGE_P : staging + t * stage_stride;
    for (u32 i = 0; i < cnt; i++) dst[i] = src[i];
}

__global__ __launch_bounds__(UA_BLOCK) void k_compact(
    const UaDesc *__restrict__ descs, const u32 *__restrict__ tile_pair,
    const u64 *__restrict__ tb /* per-pair first tile in the ACTIVE layout:
                                  == UaDesc.tile_base for merge tiles, tba
                                  for the A-indexed (k_aisect) layout */,
    const u32 *__restrict__ tile_cnt, const u64 *__restrict__ offs,
    const u64 *__restrict__ partials, const u64 *__restrict__ staging,
    u64 stage_stride, const u64 *__restrict__ prim, u64 total_tiles, int op) {
    u64 base = ((u64)blockIdx.x * 4 + (threadIdx.x >> 6)) * 4;
    int lane = threadIdx.x & 63;
    if (base >= total_tiles) return;
    /* batch the whole dependent chain per 4-tile group: cnts, pairs, offs
     * issue together (independent), then descs (one dependent hop), so the
     * wave pays ~2 scattered-load round trips instead of 4x3 */
    u32 cnts4[4], pair4[4];
    u64 off4[4];
#pragma unroll
    for (int q = 0; q < 4; q++) {
        bool v = base + q < total_tiles;
        cnts4[q] = v ? tile_cnt[base + q] : 0;
        pair4[q] = v ? tile_pair[base + q] : 0;
        off4[q] = v ? d_off(offs, partials, base + q) : 0;
    }
#pragma unroll
    for (int q = 0; q < 4; q++) {
        u64 t = base + q;
        u32 cnt = cnts4[q];
        if (cnt <= UA_COMPACT_SMALL) continue; /* k_compact_small's tiles */
        UaDesc d = descs[pair4[q]];
        u64 goff = off4[q] - d_off(offs, partials, tb[pair4[q]]);
        /* pair out-capacity clamp (invalid duplicate/unsorted inputs must
         * stay memory-safe, like the reference) */
        u64 cap = (op == OP_INTERSECT) ? (d.n < d.m ? d.n : d.m)
                  : (op == OP_DIFF) ? d.n
                                    : (d.n + d.m);
        if (goff >= cap) continue;
        u64 room = cap - goff;
        if ((u64)cnt > room) cnt = (u32)room;
        u64 *dst = d.out + goff;
        const u64 *src = (prim && cnt <= (u32)UA_STAGE_P) ? prim + t * UA_STAGE_P
                                                          : staging + t * stage_stride;
        for (u32 i = lane; i < cnt; i += 64) dst[i] = src[i];
    }
}

__global__ __launch_bounds__(UA_BLOCK) void k_compact_flat(
    u64 *__restrict__ out, const u32 *__restrict__ cnts, const u64 *__restrict__ offs,
    const u64 *__restrict__ partials, const u64 *__restrict__ staging,
    u64 stage_stride, u64 n_blocks) {
    u64 base = ((u64)blockIdx.x * 4 + (threadIdx.x >> 6)) * 4;
    int lane = threadIdx.x & 63;
    if (base >= n_blocks) return;
    u32 cnts4[4];
#pragma unroll
    for (int q = 0; q < 4; q++) cnts4[q] = (base + q < n_blocks) ? cnts[base + q] : 0;
#pragma unroll
    for (int q = 0; q < 4; q++) {
        u64 b = base + q;
        u32 cnt = cnts4[q];
        if (cnt == 0) continue;
        u64 *dst = out + d_off(offs, partials, b);
        const u64 *src = staging + b * stage_stride;
        for (u32 i = lane; i < cnt; i += 64) dst[i] = src[i];
    }
}

/* ==================== kernel: batched ApplyFilter ==================== */

/* algo.ApplyFilter (uidlist.go:21): in-place mask compaction.  The Go
 * closure f(uid, i) becomes a precomputed per-element byte mask across the
 * C-ABI (callers evaluate predicates upstream, worker/task.go:1403).  One
 * workgroup per 2048-element chunk; per-task output offsets by the same
 * decoupled-lookback protocol as k_tiles (MODE_LOOKBACK).  out == u
 * (in-place, the reference's shape) is safe: every chunk's loads land
 * (vmcnt) before its count is published, and successors only write after
 * all predecessors published. */
struct alignas(16) UaFDesc {
    const u64 *u;
    u64 n;
    const u8 *mask;
    u64 *out;
    u64 tile_base;
};

__global__ __launch_bounds__(UA_BLOCK) void k_filter(
    const UaFDesc *__restrict__ descs, const u64 *__restrict__ tb, int n_tasks,
    u64 total_tiles, u64 *__restrict__ lbf, u64 gen, u64 *__restrict__ pout) {
    __shared__ u32 scan[UA_BLOCK / 64];
    __shared__ u64 s_run;
    u64 t = blockIdx.x;
    int tid = threadIdx.x;
    /* task p with tb[p] <= t < tb[p+1] */
    int lo = 0, hi = n_tasks - 1;
    while (lo < hi) {
        int mid = (lo + hi + 1) >> 1;
        if (tb[mid] <= t) lo = mid;
        else hi = mid - 1;
    }
    int p = lo;
    UaFDesc d = descs[p];
    u64 lt = t - d.tile_base;
    u64 d0 = lt * UA_TILE;
    u64 d1 = d0 + UA_TILE;
    if (d1 > d.n) d1 = d.n;
    /* per-thread contiguous 8-element run (order-preserving compaction;
     * the 64-B-strided lane pattern coalesces through L1 over the 8 steps) */
    u64 base = d0 + (u64)tid * UA_WPT;
    u64 vals[UA_WPT];
    u32 flags = 0;
    int cnt = 0;
#pragma unroll
    for (int s = 0; s < UA_WPT; s++) {
        u64 i = base + s;
        if (i < d1) {
            vals[s] = d.u[i];
            u32 keep = d.mask[i] ? 1u : 0u;
            flags |= keep << s;
            cnt += (int)keep;
        }
    }
    /* in-place safety: loads must LAND chip-wide before this chunk's count
     * is published (a successor may overwrite these addresses) */
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    u32 excl, total;
    d_block_scan<UA_BLOCK>(tid, (u32)cnt, scan, excl, total);
    if (tid == 0) {
        u64 st = (lt == 0) ? 2ull : 1ull;
        __hip_atomic_store(&lbf[t], d_lb_word(gen, st, total), __ATOMIC_RELAXED,
                           __HIP_MEMORY_SCOPE_AGENT);
        if (lt == 0) s_run = 0;
    }
    if (lt != 0 && tid < 64) {
        u64 run = d_lb_resolve(lbf, gen, t, d.tile_base, tid);
        if (tid == 0) {
            s_run = run;
            __hip_atomic_store(&lbf[t], d_lb_word(gen, 2ull, run + total),
                               __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
        }
    }
    __syncthreads();
    u64 run = s_run;
    if (cnt > 0) {
        u64 *dst = d.out + run + excl;
        int k = 0;
#pragma unroll
        for (int s = 0; s < UA_WPT; s++) {
            if (flags & (1u << s)) dst[k++] = vals[s];
        }
    }
    if (tid == 0) {
        bool lastt = (lt + 1) * UA_TILE >= d.n;
        if (lastt) pout[p] = run + total;
    }
}

/* ==================== kernel: batched IndexOf ==================== */

__global__ __launch_bounds__(UA_BLOCK) void k_index_of(const u64 *__restrict__ u, u64 n,
                                                       const u64 *__restrict__ q, u64 nq,
                                                       int64_t *__restrict__ out) {
    u64 i = (u64)blockIdx.x * UA_BLOCK + threadIdx.x;
    if (i >= nq) return;
    u64 key = q[i];
    u64 pos = d_lower_bound(u, n, key);
    out[i] = (pos < n && u[pos] == key) ? (int64_t)pos : -1;
}

/* ==================== kernel: fused group-varint decode (+intersect) ====================
 * One 64-lane wave per pb.UidBlock: load deltas to LDS, lane-0 control-byte
 * walk -> group offsets, per-lane 4-delta extract, wave prefix-sum (+base),
 * then either emit uids >= after (decode) or match a v-range against the
 * decoded block (intersect).  Format: codec.go:57-101 packBlock /
 * go-groupvarint Encode4 (see oracle.h header). */

__device__ __forceinline__ u32 d_load_le(const u8 *p, int len) {
    u32 x = 0;
    for (int b = 0; b < len; b++) x |= ((u32)p[b]) << (8 * b);
    return x;
}

/* per-block pack lookup for the batched form */
__device__ __forceinline__ int d_pack_of(const u64 *__restrict__ pbb, int n_packs, u64 b) {
    int lo = 0, hi = n_packs - 1;
    while (lo < hi) {
        int mid = (lo + hi + 1) >> 1;
        if (pbb[mid] <= b) lo = mid;
        else hi = mid - 1;
    }
    return lo;
}

template <int DECODE_ONLY>
__global__ __launch_bounds__(UA_BLOCK) void k_packed(
    const u64 *__restrict__ bases, const u32 *__restrict__ nums,
    const u64 *__restrict__ doffs, const u8 *__restrict__ deltas, u64 n_blocks,
    u64 after, const u64 *v, u64 m, u64 *__restrict__ staging,
    u32 *__restrict__ blk_cnt,
    const u64 *__restrict__ pbb /* batch: [n_packs+1] or null */, int n_packs,
    const ua_ptask *__restrict__ tasks /* batch: per-pack v/m/out/after */) {
    __shared__ u8 sdel[UA_PKW][UA_MAX_DELTAS];
    __shared__ u64 sdec[UA_PKW][UA_MAX_BLOCK_UIDS + 4];
    __shared__ u16 sgoff[UA_PKW][64];

    int wv = threadIdx.x >> 6;
    int lane = threadIdx.x & 63;
    u64 b = (u64)blockIdx.x * UA_PKW + wv;
    /* no early return: __syncthreads() below must be reached by every thread */
    bool active = (b < n_blocks);
    u32 num = 0;
    u64 base = 0, off0 = 0;
    u32 dlen = 0;
    if (active) {
        num = nums[b];
        base = bases[b];
        off0 = doffs[b];
        dlen = (u32)(doffs[b + 1] - off0);
        if (num == 0 || num > UA_MAX_BLOCK_UIDS || dlen > UA_MAX_DELTAS) {
            /* host validates; guard anyway */
            if (lane == 0) blk_cnt[b] = 0;
            active = false;
        }
    }
    if (active && tasks != nullptr) {
        ua_ptask tk = tasks[d_pack_of(pbb, n_packs, b)];
        after = tk.after_uid;
        v = tk.v;
        m = tk.m;
    }
    if (active) {
        for (u32 i = lane; i < dlen; i += 64) sdel[wv][i] = deltas[off0 + i];
    }
    __syncthreads(); /* sdel visible to lane 0's control-byte walk */

    u32 ng = (num > 1) ? ((num + 2) >> 2) : 0; /* ceil((num-1)/4) */
    if (active && lane == 0) {
        u32 o = 0;
        for (u32 g = 0; g < ng; g++) {
            sgoff[wv][g] = (u16)o;
            u8 tag = sdel[wv][o];
            o += 5 + (tag & 3) + ((tag >> 2) & 3) + ((tag >> 4) & 3) + ((tag >> 6) & 3);
        }
    }
    __syncthreads(); /* sgoff visible to all lanes */

    u32 d0 = 0, d1 = 0, d2 = 0, d3 = 0;
    u64 s_local = 0;
    if (active && lane < (int)ng) {
        u32 o = sgoff[wv][lane];
        u8 tag = sdel[wv][o];
        const u8 *p = &sdel[wv][o + 1];
        int l0 = (tag & 3) + 1, l1 = ((tag >> 2) & 3) + 1, l2 = ((tag >> 4) & 3) + 1,
            l3 = ((tag >> 6) & 3) + 1;
        d0 = d_load_le(p, l0);
        p += l0;
        d1 = d_load_le(p, l1);
        p += l1;
        d2 = d_load_le(p, l2);
        p += l2;
        d3 = d_load_le(p, l3);
        s_local = (u64)d0 + d1 + d2 + d3;
    }
    u64 incl = s_local;
    for (int o = 1; o < 64; o <<= 1) {
        u64 x = __shfl_up(incl, o);
        if (lane >= o) incl += x;
    }
    u64 excl = incl - s_local;
    if (active && lane == 0) sdec[wv][0] = base;
    if (active && lane < (int)ng) {
        u64 acc = base + excl;
        acc += d0; sdec[wv][lane * 4 + 1] = acc;
        acc += d1; sdec[wv][lane * 4 + 2] = acc;
        acc += d2; sdec[wv][lane * 4 + 3] = acc;
        acc += d3; sdec[wv][lane * 4 + 4] = acc;
    }
    __syncthreads(); /* decoded block visible to every lane */
    if (!active) return;
    /* writes past num land in the +4 pad and are logically truncated (codec.go:198) */
    u64 first = base;
    u64 last = sdec[wv][num - 1];
    u64 out_base = b * (u64)UA_MAX_BLOCK_UIDS;
    u32 cnt = 0;

    if (DECODE_ONLY) {
        for (u32 s0 = 0; s0 < num; s0 += 64) {
            u32 idx = s0 + lane;
            bool ok = false;
            u64 val = 0;
            if (idx < num) {
                val = sdec[wv][idx];
                ok = (val >= after);
            }
            u64 mask = __ballot(ok);
            if (ok) {
                u32 r = (u32)__popcll(mask & ((1ull << lane) - 1));
                staging[out_base + cnt + r] = val;
            }
            cnt += (u32)__popcll(mask);
        }
    } else {
        u64 lo = d_lower_bound(v, m, first > after ? first : after);
        u64 hi = (last == UINT64_MAX) ? m : d_lower_bound(v, m, last + 1);
        for (u64 s0 = lo; s0 < hi; s0 += 64) {
            u64 idx = s0 + lane;
            bool ok = false;
            u64 val = 0;
            if (idx < hi) {
                val = v[idx];
                u64 pos = d_lower_bound(sdec[wv], num, val);
                ok = (pos < num && sdec[wv][pos] == val);
            }
            u64 mask = __ballot(ok);
            if (ok) {
                u32 r = (u32)__popcll(mask & ((1ull << lane) - 1));
                staging[out_base + cnt + r] = val;
            }
            cnt += (u32)__popcll(mask);
        }
    }
    if (lane == 0) blk_cnt[b] = cnt;
}

/* compact for the multi-pack batch: per-pack out destinations */
__global__ __launch_bounds__(UA_BLOCK) void k_compact_pack(
    const ua_ptask *__restrict__ tasks, const u64 *__restrict__ pbb, int n_packs,
    const u32 *__restrict__ cnts, const u64 *__restrict__ offs,
    const u64 *__restrict__ partials, const u64 *__restrict__ staging, u64 n_blocks) {
    u64 basei = ((u64)blockIdx.x * 4 + (threadIdx.x >> 6)) * 4;
    int lane = threadIdx.x & 63;
    if (basei >= n_blocks) return;
    u32 cnts4[4];
#pragma unroll
    for (int q = 0; q < 4; q++) cnts4[q] = (basei + q < n_blocks) ? cnts[basei + q] : 0;
#pragma unroll
    for (int q = 0; q < 4; q++) {
        u64 b = basei + q;
        u32 cnt = cnts4[q];
        if (cnt == 0) continue;
        int p = d_pack_of(pbb, n_packs, b);
        u64 *dst = tasks[p].out +
                   (d_off(offs, partials, b) - d_off(offs, partials, pbb[p]));
        const u64 *src = staging + b * (u64)UA_MAX_BLOCK_UIDS;
        for (u32 i = lane; i < cnt; i += 64) dst[i] = src[i];
    }
}

/* ==================== kernels: GPU codec.Encode ====================
 * Parallel restatement of codec.go:57-136 (packBlock/Add): block boundaries
 * are 32-MSB changes (match32MSB :469) plus every block_size-th element
 * within a 32-MSB run; each block's group-varint bytes are produced by one
 * wavefront (lane g encodes group g; a wave shfl-scan of group byte-lengths
 * places them).  Byte-identical to the reference format. */

#define UA_ENC_STRIDE 1104 /* per-block staging bytes (64 groups x 17, padded) */

__global__ __launch_bounds__(UA_BLOCK) void k_enc_msb_flags(
    const u64 *__restrict__ uids, u64 n, u32 *__restrict__ flags) {
    u64 i = (u64)blockIdx.x * UA_BLOCK + threadIdx.x;
    if (i >= n) return;
    flags[i] = (i == 0) || (((uids[i] ^ uids[i - 1]) >> 32) != 0);
}

/* run/block starts: out[rank(i)] = i for flagged i; thread n writes the
 * total sentinel out[count] = n */
__global__ __launch_bounds__(UA_BLOCK) void k_enc_mark_start(
    const u32 *__restrict__ flags, const u64 *__restrict__ offs,
    const u64 *__restrict__ partials, u64 n, u64 *__restrict__ out) {
    u64 i = (u64)blockIdx.x * UA_BLOCK + threadIdx.x;
    if (i > n) return;
    if (i == n) {
        out[d_off(offs, partials, n)] = n;
        return;
    }
    if (flags[i]) out[d_off(offs, partials, i)] = i;
}

/* msb flags -> block flags (in place): also split every block_size-th
 * element of each 32-MSB run (codec.go:117-126; blockSize 0 -> 1-uid blocks) */
__global__ __launch_bounds__(UA_BLOCK) void k_enc_block_flags(
    u32 *__restrict__ flags, const u64 *__restrict__ offs,
    const u64 *__restrict__ partials, const u64 *__restrict__ run_start, u64 n,
    u32 bs) {
    u64 i = (u64)blockIdx.x * UA_BLOCK + threadIdx.x;
    if (i >= n) return;
    if (flags[i]) return; /* already a boundary */
    u64 rid = d_off(offs, partials, i) + flags[i] - 1; /* inclusive rank - 1 */
    u64 rs = run_start[rid];
    u32 eff = bs ? bs : 1;
    flags[i] = (((i - rs) % eff) == 0);
}

__global__ __launch_bounds__(UA_BLOCK) void k_encode_blocks(
    const u64 *__restrict__ uids, const u64 *__restrict__ bstart, u64 nb,
    u64 *__restrict__ bases, u32 *__restrict__ nums, u32 *__restrict__ blk_bytes,
    u8 *__restrict__ stage) {
    __shared__ u8 sbytes[UA_PKW][UA_ENC_STRIDE];
    int wv = threadIdx.x >> 6;
    int lane = threadIdx.x & 63;
    u64 b = (u64)blockIdx.x * UA_PKW + wv;
    bool active = (b < nb);
    u64 s = 0, e = 0;
    u32 num = 0, ng = 0;
    if (active) {
        s = bstart[b];
        e = bstart[b + 1];
        num = (u32)(e - s);
        ng = (num > 1) ? ((num + 2) >> 2) : 1; /* >=1 group incl. pad-only (codec.go:76-96) */
        if (lane == 0) {
            bases[b] = uids[s];
            nums[b] = num;
        }
    }
    /* lane g: encode group g (deltas for uids[s+1+4g .. s+4+4g], zero-padded) */
    u32 glen = 0;
    u32 dl[4];
    u8 ln[4];
    if (active && lane < (int)ng) {
        u64 prev = uids[s + (u64)(4 * lane)];
        u32 tot = 1;
        for (int k = 0; k < 4; k++) {
            u64 idx = s + (u64)(4 * lane) + 1 + (u64)k;
            u32 d = (idx < e) ? (u32)(uids[idx] - prev) : 0;
            if (idx < e) prev = uids[idx];
            dl[k] = d;
            ln[k] = (u8)(1 + (d > 0xffu) + (d > 0xffffu) + (d > 0xffffffu));
            tot += ln[k];
        }
        glen = tot;
    }
    /* wave exclusive scan of group byte-lengths */
    u32 incl = glen;
#pragma unroll
    for (int o = 1; o < 64; o <<= 1) {
        u32 x = __shfl_up(incl, o);
        if (lane >= o) incl += x;
    }
    u32 goff = incl - glen;
    u32 total = (u32)__shfl(incl, 63);
    if (active && lane < (int)ng) {
        u8 *p = &sbytes[wv][goff];
        u8 tag = (u8)((ln[0] - 1) | ((ln[1] - 1) << 2) | ((ln[2] - 1) << 4) |
                      ((ln[3] - 1) << 6));
        *p++ = tag;
        for (int k = 0; k < 4; k++) {
            u32 d = dl[k];
            for (int bcount = 0; bcount < ln[k]; bcount++) {
                *p++ = (u8)(d & 0xff);
                d >>= 8;
            }
        }
    }
    __syncthreads(); /* cross-lane LDS visibility before the copy-out */
    if (!active) return;
    u8 *gdst = stage + b * UA_ENC_STRIDE;
    for (u32 i = lane; i < total; i += 64) gdst[i] = sbytes[wv][i];
    if (lane == 0) blk_bytes[b] = total;
}

__global__ __launch_bounds__(UA_BLOCK) void k_enc_finalize(
    const u8 *__restrict__ stage, const u32 *__restrict__ blk_bytes,
    const u64 *__restrict__ offs, const u64 *__restrict__ partials, u64 nb,
    u8 *__restrict__ deltas, u64 *__restrict__ delta_offs) {
    u64 b = (u64)blockIdx.x * 4 + (threadIdx.x >> 6);
    int lane = threadIdx.x & 63;
    if (b > nb) return;
    u64 off = d_off(offs, partials, b);
    if (lane == 0) delta_offs[b] = off;
    if (b == nb) return;
    u32 bytes = blk_bytes[b];
    const u8 *src = stage + b * UA_ENC_STRIDE;
    u8 *dst = deltas + off;
    for (u32 i = lane; i < bytes; i += 64) dst[i] = src[i];
}


/* ---- device-chained merge tree helpers ---- */

/* fill a round's descriptor lengths from the previous round's device lens */
__global__ __launch_bounds__(UA_BLOCK) void k_make_descs(
    UaDesc *__restrict__ descs, const u64 *__restrict__ lens_prev, int nk_prev,
    int npair) {
    int j = blockIdx.x * UA_BLOCK + threadIdx.x;
    if (j >= npair) return;
    descs[j].n = lens_prev[2 * j];
    descs[j].m = (2 * j + 1 < nk_prev) ? lens_prev[2 * j + 1] : 0;
}

/* copy src[0..*len_ptr) to out without a host round-trip for the length */
__global__ __launch_bounds__(UA_BLOCK) void k_copy_len(
    u64 *__restrict__ out, const u64 *__restrict__ src, const u64 *__restrict__ len_ptr) {
    u64 n = *len_ptr;
    for (u64 i = (u64)blockIdx.x * UA_BLOCK + threadIdx.x; i < n;
         i += (u64)gridDim.x * UA_BLOCK)
        out[i] = src[i];
}

#ifndef UA_PIPE
#define UA_PIPE 0 /* 1 = 2-tile software-pipelined tile kernel (A/B toggle) */
#endif
#ifndef UA_PIPE2
#define UA_PIPE2 0 /* 1 = persistent 2-buffer glds pipeline (A/B toggle) */
#endif

template <int OP, int MODE>
static void launch_tiles(ua_ctx *c, const UaDesc *descs, const u32 *tpair,
                         const u32 *ta0, u64 T, u64 *stage, u64 stride, u32 *tcnt,
                         const u64 *offs, const u64 *part,
                         const u32 *isin = nullptr, u32 *isout = nullptr);

/* ==================== host shim ==================== */

static thread_local hipError_t g_last_hip = hipSuccess;

#define HIP_TRY(x)                         \
    do {                                   \
        hipError_t _e = (x);               \
        if (_e != hipSuccess) {            \
            g_last_hip = _e;               \
            return UA_ERR_HIP;             \
        }                                  \
    } while (0)

enum {
    WS_DESC = 0, WS_TB, WS_TPAIR, WS_TA0, WS_TCNT, WS_TOFF, WS_PARTIAL,
    WS_STAGE, WS_STAGEP, WS_POUT, WS_HU, WS_HV, WS_HOUT, WS_PACK, WS_SCRATCH_A,
    WS_SCRATCH_B, WS_LBF, WS_COUNT
};

struct ua_ctx {
    int device = 0;
    hipStream_t stream = nullptr;
    void *ws[WS_COUNT] = {};
    size_t ws_cap[WS_COUNT] = {};
    hipEvent_t ev[4] = {};
    hipStream_t stream_tail = nullptr; /* aux-tail overlap (run_n) */
    hipEvent_t oev[4] = {}; /* [0..1]=tile_done per set, [2..3]=tail_done */
    /* decoupled-lookback flag array (WS_LBF) bookkeeping */
    uint32_t lbf_gen = 0;
    size_t lbf_cleared = 0; /* bytes of the CURRENT WS_LBF allocation zeroed */
    uint32_t *d_viol = nullptr; /* pack-limit validation flag (1 u32) */
    /* stats */
    uint64_t n_launches = 0;
    double kernel_ms = 0.0;
    uint64_t bytes_algo = 0;
    std::recursive_mutex mu; /* compound ops hold it across internal stages */
};

extern "C" const char *ua_strerror(int code) {
    switch (code) {
        case UA_OK: return "ok";
        case UA_ERR_HIP: return hipGetErrorString(g_last_hip);
        case UA_ERR_NOMEM: return "allocation failed";
        case UA_ERR_INVALID: return "invalid argument";
        case UA_ERR_NO_GPU: return "no HIP device visible";
        default: return "unknown error";
    }
}

extern "C" int ua_version(void) { return 11; }

extern "C" int ua_ctx_create(ua_ctx **out, int device) {
    int ndev = 0;
    hipError_t e = hipGetDeviceCount(&ndev);
    if (e != hipSuccess || ndev == 0) return UA_ERR_NO_GPU;
    if (device < 0 || device >= ndev) return UA_ERR_INVALID;
    ua_ctx *c = new ua_ctx();
    c->device = device;
    HIP_TRY(hipSetDevice(device));
    HIP_TRY(hipStreamCreate(&c->stream));
    for (int i = 0; i < 4; i++) {
        HIP_TRY(hipEventCreate(&c->ev[i]));
        /* record once eagerly: an event that has never been recorded cannot
         * be captured into a graph on this ROCm (invalid resource handle) */
        HIP_TRY(hipEventRecord(c->ev[i], c->stream));
    }
    HIP_TRY(hipStreamCreate(&c->stream_tail));
    for (int i = 0; i < 4; i++)
        HIP_TRY(hipEventCreateWithFlags(&c->oev[i], hipEventDisableTiming));
    HIP_TRY(hipStreamSynchronize(c->stream));
    *out = c;
    return UA_OK;
}

extern "C" void ua_ctx_destroy(ua_ctx *c) {
    if (!c) return;
    (void)hipSetDevice(c->device);
    for (int i = 0; i < WS_COUNT; i++)
        if (c->ws[i]) (void)hipFree(c->ws[i]);
    for (int i = 0; i < 4; i++)
        if (c->ev[i]) (void)hipEventDestroy(c->ev[i]);
    for (int i = 0; i < 4; i++)
        if (c->oev[i]) (void)hipEventDestroy(c->oev[i]);
    if (c->stream_tail) (void)hipStreamDestroy(c->stream_tail);
    if (c->stream) (void)hipStreamDestroy(c->stream);
    delete c;
}

static int ws_reserve(ua_ctx *c, int slot, size_t bytes) {
    if (bytes <= c->ws_cap[slot]) return UA_OK;
    if (c->ws[slot]) (void)hipFree(c->ws[slot]);
    c->ws[slot] = nullptr;
    c->ws_cap[slot] = 0;
    size_t cap = bytes + bytes / 4; /* 25% headroom limits realloc churn */
    hipError_t e = hipMalloc(&c->ws[slot], cap);
    if (e != hipSuccess) {
        g_last_hip = e;
        /* retry exact */
        e = hipMalloc(&c->ws[slot], bytes);
        if (e != hipSuccess) return UA_ERR_NOMEM;
        cap = bytes;
    }
    c->ws_cap[slot] = cap;
    return UA_OK;
}

/* Which ops run the single-pass decoupled-lookback pipeline.
 * MEASURED (r02, cfg2 192x1Mx1M): lookback doubles the intersect kernel
 * (0.76 -> 1.47 ms) — a tile cannot write until every same-pair predecessor
 * has finished its walk, so each tile's lifetime spans ~2 walk phases and
 * 8 WGs/CU cannot hide the parked resolve.  For intersect/difference the
 * staged scan+compact pipeline (outputs ~1% of inputs -> staging is free)
 * is strictly better; for UNION the lookback removes a FULL second data
 * pass (count+write -> one walk), which outweighs the same wait.  */
#ifndef UA_LOOKBACK
#define UA_LOOKBACK 0 /* intersect/diff: 1 = lookback, 0 = staged (default) */
#endif
#ifndef UA_LOOKBACK_UNION
#define UA_LOOKBACK_UNION 0 /* union/merge-tree single-pass lookback: MEASURED
                             * WORSE on cfg3 (125 vs 145 G elems/s) — the
                             * same-pair predecessor wait costs more than the
                             * saved second read on long tile chains */
#endif

/* Acquire the ctx-level lookback flag array for T tiles with a fresh
 * generation.  The array must read as "stale" for the new generation:
 * guaranteed by zeroing on (re)allocation and re-zeroing when the 16-bit
 * generation wraps. */
[[maybe_unused]] static int lb_acquire_ws(ua_ctx *c, u64 T, u64 **flags, u64 *gen) {
    int rc = ws_reserve(c, WS_LBF, (size_t)T * sizeof(u64));
    if (rc) return rc;
    u64 *f = (u64 *)c->ws[WS_LBF];
    if (c->lbf_cleared < c->ws_cap[WS_LBF] || c->lbf_gen >= UA_LB_GEN_MAX) {
        HIP_TRY(hipMemsetAsync(f, 0, c->ws_cap[WS_LBF], c->stream));
        c->lbf_cleared = c->ws_cap[WS_LBF];
        c->lbf_gen = 0;
    }
    c->lbf_gen += 1;
    *flags = f;
    *gen = c->lbf_gen;
    return UA_OK;
}

extern "C" int ua_dev_alloc(ua_ctx *c, uint64_t bytes, void **dptr) {
    HIP_TRY(hipSetDevice(c->device));
    hipError_t e = hipMalloc(dptr, bytes);
    if (e != hipSuccess) {
        g_last_hip = e;
        return UA_ERR_NOMEM;
    }
    return UA_OK;
}

extern "C" int ua_dev_free(ua_ctx *c, void *dptr) {
    HIP_TRY(hipSetDevice(c->device));
    HIP_TRY(hipFree(dptr));
    return UA_OK;
}

extern "C" int ua_h2d(ua_ctx *c, void *dst, const void *src, uint64_t bytes) {
    HIP_TRY(hipSetDevice(c->device));
    HIP_TRY(hipMemcpyAsync(dst, src, bytes, hipMemcpyHostToDevice, c->stream));
    HIP_TRY(hipStreamSynchronize(c->stream));
    return UA_OK;
}

extern "C" int ua_d2h(ua_ctx *c, void *dst, const void *src, uint64_t bytes) {
    HIP_TRY(hipSetDevice(c->device));
    HIP_TRY(hipMemcpyAsync(dst, src, bytes, hipMemcpyDeviceToHost, c->stream));
    HIP_TRY(hipStreamSynchronize(c->stream));
    return UA_OK;
}

extern "C" int ua_sync(ua_ctx *c) {
    HIP_TRY(hipStreamSynchronize(c->stream));
    return UA_OK;
}

extern "C" int ua_stats_reset(ua_ctx *c) {
    std::lock_guard<std::recursive_mutex> g(c->mu);
    c->n_launches = 0;
    c->kernel_ms = 0.0;
    c->bytes_algo = 0;
    return UA_OK;
}

extern "C" int ua_stats_get(ua_ctx *c, uint64_t *n_launches, double *kernel_ms,
                            uint64_t *bytes_algorithmic) {
    std::lock_guard<std::recursive_mutex> g(c->mu);
    if (n_launches) *n_launches = c->n_launches;
    if (kernel_ms) *kernel_ms = c->kernel_ms;
    if (bytes_algorithmic) *bytes_algorithmic = c->bytes_algo;
    return UA_OK;
}

/* UA_RPIPE=0 disables the register-staged pipelined tile kernel (see
 * k_tiles_rp); default on for STAGE/COUNT/WRITE/DIRECT. */
static int pp_enabled() {
    static int v = -1;
    if (v < 0) {
        const char *e = getenv("UA_PP");
        v = (e && e[0]) ? (e[0] != '0') : 0;
    }
    return v;
}

static int rp_enabled() {
#if UA_PAD32
    return 0; /* k_tiles_rp commits an unpadded layout */
#else
    static int v = -1;
    if (v < 0) {
        const char *e = getenv("UA_RPIPE");
        v = (e && e[0]) ? (e[0] != '0') : 0;
    }
    return v;
#endif
}

/* persistent grid for k_tiles_rp: residency blocks (occupancy API x CUs).
 * Oversizing is harmless here (no inter-block deps in these modes). */
static u32 rp_grid(const void *kfn, u64 T) {
    static std::mutex mu;
    static std::unordered_map<const void *, u32> cache;
    u32 g;
    {
        std::lock_guard<std::mutex> lk(mu);
        auto it = cache.find(kfn);
        if (it != cache.end()) {
            g = it->second;
        } else {
            int nb = 0, ndev = 0, cus = 256;
            (void)hipOccupancyMaxActiveBlocksPerMultiprocessor(&nb, kfn, RP_BLOCK, 0);
            if (hipGetDevice(&ndev) == hipSuccess) {
                hipDeviceProp_t p;
                if (hipGetDeviceProperties(&p, ndev) == hipSuccess)
                    cus = p.multiProcessorCount;
            }
            if (nb < 1) nb = 1;
            g = (u32)nb * (u32)cus;
            cache.emplace(kfn, g);
        }
    }
    return (u64)g < T ? g : (u32)T;
}

/* ---- flat scan helper: cnt u32[n] (+1 zero sentinel at n-1 position
 * provided by caller) -> offs u64[n] exclusive scan ---- */
template <int OP, int MODE>
static void launch_tiles(ua_ctx *c, const UaDesc *descs, const u32 *tpair,
                         const u32 *ta0, u64 T, u64 *stage, u64 stride, u32 *tcnt,
                         const u64 *offs, const u64 *part,
                         const u32 *isin, u32 *isout) {
    if constexpr (MODE != MODE_LOOKBACK) {
        if (pp_enabled()) {
            u32 G = rp_grid((const void *)k_tiles_pp<OP, MODE>, T);
            hipLaunchKernelGGL((k_tiles_pp<OP, MODE>), dim3(G), dim3(UA_TBLOCK), 0,
                               c->stream, descs, tpair, ta0, T, stage, stride, tcnt,
                               offs, part, isin, isout);
            return;
        }
        if (rp_enabled()) {
            u32 G = rp_grid((const void *)k_tiles_rp<OP, MODE>, T);
            hipLaunchKernelGGL((k_tiles_rp<OP, MODE>), dim3(G), dim3(RP_BLOCK), 0,
                               c->stream, descs, tpair, ta0, T, stage, stride, tcnt,
                               offs, part);
            return;
        }
    }
#if UA_PIPE
    if constexpr (MODE != MODE_DIRECT && MODE != MODE_LOOKBACK) {
        hipLaunchKernelGGL((k_tiles_pipe<OP, MODE>), dim3((u32)((T + 1) / 2)),
                           dim3(UA_BLOCK), 0, c->stream, descs, tpair, ta0, T, stage,
                           stride, tcnt, offs, part);
        return;
    }
#endif
#if UA_PIPE2
    if constexpr (MODE != MODE_LOOKBACK) {
        u64 G = T < 1024 ? T : 1024; /* 4 WGs/CU x 256 CUs resident */
        hipLaunchKernelGGL((k_tiles_p2<OP, MODE>), dim3((u32)G), dim3(UA_BLOCK), 0,
                           c->stream, descs, tpair, ta0, T, stage, stride, tcnt, offs,
                           part);
        return;
    }
#endif
    hipLaunchKernelGGL((k_tiles<OP, MODE>), dim3((u32)T), dim3(UA_TBLOCK), 0, c->stream,
                       descs, tpair, ta0, T, stage, stride, tcnt, offs, part,
                       isin, isout);
}

/* Split flat scan: offs_dev gets chunk-local exclusive offsets, WS_PARTIAL
 * gets the scanned chunk partials; consumers combine via d_off(). */
static int run_scan(ua_ctx *c, const u32 *cnt_dev, u64 n, u64 *offs_dev) {
    u64 nchunks = (n + UA_SCAN_CHUNK - 1) / UA_SCAN_CHUNK;
    int rc = ws_reserve(c, WS_PARTIAL, (nchunks + 1) * sizeof(u64));
    if (rc) return rc;
    u64 *partials = (u64 *)c->ws[WS_PARTIAL];
    hipLaunchKernelGGL(k_scan1, dim3((u32)nchunks), dim3(UA_BLOCK), 0, c->stream,
                       cnt_dev, n, offs_dev, partials);
    hipLaunchKernelGGL(k_scan2, dim3(1), dim3(UA_BLOCK), 0, c->stream, partials, nchunks);
    return UA_OK;
}

/* ---- the batched set-algebra pipeline ---- */
static int run_batch_locked(ua_ctx *c, const ua_dpair *pairs, int n_pairs,
                            uint64_t *out_lens, int op) {
    if (n_pairs <= 0) return UA_OK;
    HIP_TRY(hipSetDevice(c->device));

    std::vector<UaDesc> descs((size_t)n_pairs);
    std::vector<u64> tb((size_t)n_pairs + 1);
    u64 total_tiles = 0;
    u64 in_bytes = 0;
    for (int p = 0; p < n_pairs; p++) {
        const ua_dpair &pr = pairs[p];
        if (pr.n >= (1ull << 31) || pr.m >= (1ull << 31)) return UA_ERR_INVALID;
        descs[p] = {pr.u, pr.n, pr.v, pr.m, pr.out, total_tiles};
        tb[p] = total_tiles;
        total_tiles += (pr.n + pr.m + UA_TILE - 1) / UA_TILE;
        in_bytes += 8 * (pr.n + pr.m);
    }
    tb[n_pairs] = total_tiles;

    int rc;
    /* descs and tb share one allocation and one upload */
    size_t descs_bytes = descs.size() * sizeof(UaDesc);
    if ((rc = ws_reserve(c, WS_DESC, descs_bytes + tb.size() * sizeof(u64)))) return rc;
    if ((rc = ws_reserve(c, WS_TPAIR, (total_tiles + 1) * sizeof(u32)))) return rc;
    if ((rc = ws_reserve(c, WS_TA0, (total_tiles + 1) * sizeof(u32)))) return rc;
    if ((rc = ws_reserve(c, WS_TCNT, (total_tiles + 1) * sizeof(u32)))) return rc;
    if ((rc = ws_reserve(c, WS_TOFF, (total_tiles + 1) * sizeof(u64)))) return rc;
    if ((rc = ws_reserve(c, WS_POUT, (size_t)n_pairs * sizeof(u64)))) return rc;

    const bool lb = (op == OP_UNION) ? (bool)UA_LOOKBACK_UNION : (bool)UA_LOOKBACK;
    u64 stage_stride = 0;
    if (!lb) {
        if (op == OP_INTERSECT) stage_stride = UA_TILE / 2;
        else if (op == OP_DIFF) stage_stride = UA_TILE;
        if (stage_stride) {
            if ((rc = ws_reserve(c, WS_STAGE, total_tiles * stage_stride * sizeof(u64)))) return rc;
            if ((rc = ws_reserve(c, WS_STAGEP, total_tiles * UA_STAGE_P * sizeof(u64)))) return rc;
        }
    }

    UaDesc *d_descs = (UaDesc *)c->ws[WS_DESC];
    u64 *d_tb = (u64 *)((u8 *)c->ws[WS_DESC] + descs_bytes);
    u32 *d_tpair = (u32 *)c->ws[WS_TPAIR];
    u32 *d_ta0 = (u32 *)c->ws[WS_TA0];
    u32 *d_tcnt = (u32 *)c->ws[WS_TCNT];
    u64 *d_toff = (u64 *)c->ws[WS_TOFF];
    u64 *d_pout = (u64 *)c->ws[WS_POUT];
    u64 *d_stage = (u64 *)c->ws[WS_STAGE];

    std::vector<u8> hostbuf(descs_bytes + tb.size() * sizeof(u64));
    memcpy(hostbuf.data(), descs.data(), descs_bytes);
    memcpy(hostbuf.data() + descs_bytes, tb.data(), tb.size() * sizeof(u64));
    HIP_TRY(hipMemcpyAsync(d_descs, hostbuf.data(), hostbuf.size(),
                           hipMemcpyHostToDevice, c->stream));
    /* zero sentinel so offs[total_tiles] = total output */
    HIP_TRY(hipMemsetAsync(d_tcnt + total_tiles, 0, sizeof(u32), c->stream));

    if (total_tiles == 0) {
        /* every pair is empty (n+m == 0) */
        HIP_TRY(hipStreamSynchronize(c->stream)); /* hostbuf is scoped */
        for (int p = 0; p < n_pairs; p++) out_lens[p] = 0;
        return UA_OK;
    }
    {
        u64 pblk = (total_tiles + UA_BLOCK - 1) / UA_BLOCK;
        hipLaunchKernelGGL(k_partition, dim3((u32)pblk), dim3(UA_BLOCK), 0, c->stream,
                           d_descs, d_tb, n_pairs, total_tiles, d_tpair, d_ta0, 0);
        hipLaunchKernelGGL(k_partition, dim3((u32)pblk), dim3(UA_BLOCK), 0, c->stream,
                           d_descs, d_tb, n_pairs, total_tiles, d_tpair, d_ta0, 1);

        if (op == OP_MERGE_ALL) {
            /* duplicate-keeping merge: output position == path position, so no
             * scan / staging / compaction / pair_out — lens are n+m */
            HIP_TRY(hipEventRecord(c->ev[0], c->stream));
            launch_tiles<OP_MERGE_ALL, MODE_DIRECT>(c, d_descs, d_tpair, d_ta0,
                                                    total_tiles, nullptr, 0, d_tcnt,
                                                    nullptr, nullptr);
            HIP_TRY(hipEventRecord(c->ev[1], c->stream));
            HIP_TRY(hipStreamSynchronize(c->stream));
            HIP_TRY(hipGetLastError());
            float ms = 0.f;
            HIP_TRY(hipEventElapsedTime(&ms, c->ev[0], c->ev[1]));
            c->kernel_ms += ms;
            c->n_launches += 1;
            for (int p = 0; p < n_pairs; p++) out_lens[p] = pairs[p].n + pairs[p].m;
            c->bytes_algo += 2 * in_bytes;
            return UA_OK;
        }

        bool two_kernels = false;
        if (lb) {
            u64 *d_lbf;
            u64 gen;
            if ((rc = lb_acquire_ws(c, total_tiles, &d_lbf, &gen))) return rc;
            HIP_TRY(hipEventRecord(c->ev[0], c->stream));
            if (op == OP_INTERSECT) {
                launch_tiles<OP_INTERSECT, MODE_LOOKBACK>(c, d_descs, d_tpair, d_ta0,
                                                          total_tiles, d_lbf, gen,
                                                          nullptr, d_pout, nullptr);
            } else if (op == OP_DIFF) {
                launch_tiles<OP_DIFF, MODE_LOOKBACK>(c, d_descs, d_tpair, d_ta0,
                                                     total_tiles, d_lbf, gen, nullptr,
                                                     d_pout, nullptr);
            } else {
                launch_tiles<OP_UNION, MODE_LOOKBACK>(c, d_descs, d_tpair, d_ta0,
                                                      total_tiles, d_lbf, gen, nullptr,
                                                      d_pout, nullptr);
            }
            HIP_TRY(hipEventRecord(c->ev[1], c->stream));
        } else {
            const u64 *prim = (!pp_enabled() && !rp_enabled() && op != OP_UNION)
                                  ? (const u64 *)c->ws[WS_STAGEP]
                                  : nullptr;
            HIP_TRY(hipEventRecord(c->ev[0], c->stream));
            if (op == OP_INTERSECT) {
                launch_tiles<OP_INTERSECT, MODE_STAGE>(c, d_descs, d_tpair, d_ta0,
                                                       total_tiles, d_stage,
                                                       stage_stride, d_tcnt, prim,
                                                       nullptr);
            } else if (op == OP_DIFF) {
                launch_tiles<OP_DIFF, MODE_STAGE>(c, d_descs, d_tpair, d_ta0,
                                                  total_tiles, d_stage, stage_stride,
                                                  d_tcnt, prim, nullptr);
            } else {
                launch_tiles<OP_UNION, MODE_COUNT>(c, d_descs, d_tpair, d_ta0,
                                                   total_tiles, nullptr, 0, d_tcnt,
                                                   nullptr, nullptr);
            }
            HIP_TRY(hipEventRecord(c->ev[1], c->stream));

            if ((rc = run_scan(c, d_tcnt, total_tiles + 1, d_toff))) return rc;
            u64 *d_part = (u64 *)c->ws[WS_PARTIAL];

            if (op == OP_UNION) {
                two_kernels = true;
                HIP_TRY(hipEventRecord(c->ev[2], c->stream));
                launch_tiles<OP_UNION, MODE_WRITE>(c, d_descs, d_tpair, d_ta0,
                                                   total_tiles, nullptr, 0, d_tcnt,
                                                   d_toff, d_part);
                HIP_TRY(hipEventRecord(c->ev[3], c->stream));
            } else {
                u64 cblk = (total_tiles + 15) / 16;
                hipLaunchKernelGGL(k_compact, dim3((u32)cblk), dim3(UA_BLOCK), 0,
                                   c->stream, d_descs, d_tpair, d_tb, d_tcnt, d_toff,
                                   d_part, d_stage, stage_stride, prim, total_tiles,
                                   op);
                u64 sblk = (total_tiles + UA_BLOCK - 1) / UA_BLOCK;
                hipLaunchKernelGGL(k_compact_small, dim3((u32)sblk), dim3(UA_BLOCK),
                                   0, c->stream, d_descs, d_tpair, d_tb, d_tcnt,
                                   d_toff, d_part, d_stage, stage_stride, prim,
                                   total_tiles, op);
            }
            u64 poutblk = ((u64)n_pairs + UA_BLOCK - 1) / UA_BLOCK;
            hipLaunchKernelGGL(k_pair_out, dim3((u32)poutblk), dim3(UA_BLOCK), 0,
                               c->stream, d_toff, d_part, d_tb, n_pairs, d_pout);
        }
        HIP_TRY(hipMemcpyAsync(out_lens, d_pout, (size_t)n_pairs * sizeof(u64),
                               hipMemcpyDeviceToHost, c->stream));
        HIP_TRY(hipStreamSynchronize(c->stream));
        HIP_TRY(hipGetLastError());
        if (lb) {
            /* pairs with zero tiles (n+m == 0) never publish a length */
            for (int p = 0; p < n_pairs; p++)
                if (pairs[p].n + pairs[p].m == 0) out_lens[p] = 0;
        }

        /* stats: HIP-event time of the dominant (tile) kernel(s) */
        float ms = 0.f;
        HIP_TRY(hipEventElapsedTime(&ms, c->ev[0], c->ev[1]));
        c->kernel_ms += ms;
        c->n_launches += 1;
        if (two_kernels) {
            float ms2 = 0.f;
            HIP_TRY(hipEventElapsedTime(&ms2, c->ev[2], c->ev[3]));
            c->kernel_ms += ms2;
            c->n_launches += 1;
        }
    }
    u64 out_elems = 0;
    for (int p = 0; p < n_pairs; p++) out_elems += out_lens[p];
    c->bytes_algo += in_bytes + 8 * out_elems;
    return UA_OK;
}

static int run_batch(ua_ctx *c, const ua_dpair *pairs, int n_pairs, uint64_t *out_lens, int op) {
    std::lock_guard<std::recursive_mutex> g(c->mu);
    return run_batch_locked(c, pairs, n_pairs, out_lens, op);
}

extern "C" int ua_intersect_batch_dev(ua_ctx *c, const ua_dpair *pairs, int n_pairs,
                                      uint64_t *out_lens) {
    return run_batch(c, pairs, n_pairs, out_lens, OP_INTERSECT);
}

/* ==================== prepared batch (repeated-query path) ==================== */

struct ua_batch {
    int n_pairs = 0;
    u64 total_tiles = 0;
    u64 in_bytes = 0;
    u64 nchunks = 0;
    void *mem = nullptr;     /* one allocation for all metadata arrays */
    u64 *d_stage = nullptr;  /* lazy; stride UA_TILE (fits intersect + diff) */
    u64 *d_stage_p = nullptr; /* lazy dense primary staging (UA_STAGE_P/tile) */
    /* set-1 buffers for the overlapped run_n (aux tail of step i on the
     * tail stream while step i+1's tile kernel fills the other set) */
    u64 *d_stage1 = nullptr;
    void *mem1 = nullptr; /* holds d_tcnt1 + d_toff1 + d_part1 */
    u32 *d_tcnt1 = nullptr;
    u64 *d_toff1 = nullptr;
    u64 *d_part1 = nullptr;
    UaDesc *d_descs = nullptr;
    u64 *d_tb = nullptr;
    u32 *d_tpair = nullptr;
    u32 *d_ta0 = nullptr;
    u32 *d_tcnt = nullptr;
    u64 *d_toff = nullptr;
    u64 *d_part = nullptr;
    u64 *d_pout = nullptr;
    u64 *d_lbf = nullptr;    /* lookback flag array [total_tiles] */
    u32 lb_gen = 0;
    /* cached per-thread merge-path splits (valid because the prepared
     * batch's inputs are immutable — same contract as the cached tile
     * partition): u16 per (tile, thread); written by the first staged /
     * lookback run, read by every later one */
    u32 *d_isplit = nullptr; /* packed lo16 = i0(s0), hi16 = i0b(smid) */
    bool i0_ready = false;
    /* A-indexed layout for the wave-register intersect (k_aisect) */
    u64 total_atiles = 0;
    u64 nchunks_a = 0;
    u32 *d_tpair_a = nullptr;
    u64 *d_tba = nullptr;
    u32 *d_bstart = nullptr;
    /* hipGraph capture of the staged pipeline (per op); out_lens land in the
     * pinned h_pout so the captured D2H copy has a fixed destination */
    hipGraphExec_t gexec[3][2] = {};
    bool no_graph = false; /* capture/replay failed on this box: stay eager */
    u64 *h_pout = nullptr;
};

static size_t align16(size_t x) { return (x + 15) & ~(size_t)15; }

extern "C" int ua_batch_create(ua_ctx *c, const ua_dpair *pairs, int n_pairs,
                               ua_batch **out) {
    std::lock_guard<std::recursive_mutex> g(c->mu);
    HIP_TRY(hipSetDevice(c->device));
    ua_batch *b = new ua_batch();
    b->n_pairs = n_pairs;
    std::vector<UaDesc> descs((size_t)std::max(n_pairs, 1));
    std::vector<u64> tb((size_t)n_pairs + 1);
    for (int p = 0; p < n_pairs; p++) {
        const ua_dpair &pr = pairs[p];
        if (pr.n >= (1ull << 31) || pr.m >= (1ull << 31)) {
            delete b;
            return UA_ERR_INVALID;
        }
        descs[p] = {pr.u, pr.n, pr.v, pr.m, pr.out, b->total_tiles};
        tb[p] = b->total_tiles;
        b->total_tiles += (pr.n + pr.m + UA_TILE - 1) / UA_TILE;
        b->in_bytes += 8 * (pr.n + pr.m);
    }
    tb[n_pairs] = b->total_tiles;
    u64 T = b->total_tiles;
    b->nchunks = (T + 1 + UA_SCAN_CHUNK - 1) / UA_SCAN_CHUNK;
    /* A-indexed tiles (wave-register intersect): tiles of UA_AT A-elems */
    std::vector<u64> tba((size_t)n_pairs + 1);
    for (int p = 0; p < n_pairs; p++) {
        tba[p] = b->total_atiles;
        b->total_atiles += (pairs[p].n + UA_AT - 1) / UA_AT;
    }
    tba[n_pairs] = b->total_atiles;
    u64 Ta = b->total_atiles;
    b->nchunks_a = (Ta + 1 + UA_SCAN_CHUNK - 1) / UA_SCAN_CHUNK;
    u64 Tmax = T > Ta ? T : Ta;
    u64 ncmax = b->nchunks > b->nchunks_a ? b->nchunks : b->nchunks_a;

    size_t o_desc = 0;
    size_t o_tb = align16(o_desc + descs.size() * sizeof(UaDesc));
    size_t o_toff = align16(o_tb + tb.size() * sizeof(u64));
    size_t o_part = align16(o_toff + (Tmax + 1) * sizeof(u64));
    size_t o_pout = align16(o_part + (ncmax + 1) * sizeof(u64));
    size_t o_tpair = align16(o_pout + (size_t)std::max(n_pairs, 1) * sizeof(u64));
    size_t o_ta0 = align16(o_tpair + (T + 1) * sizeof(u32));
    size_t o_tcnt = align16(o_ta0 + (T + 1) * sizeof(u32));
    size_t o_lbf = align16(o_tcnt + (Tmax + 1) * sizeof(u32));
    size_t o_tba = align16(o_lbf + (T + 1) * sizeof(u64));
    size_t o_tpa = align16(o_tba + tba.size() * sizeof(u64));
    size_t o_bst = align16(o_tpa + (Ta + 1) * sizeof(u32));
    size_t total_bytes = align16(o_bst + (Ta + 1) * sizeof(u32));
    hipError_t e = hipMalloc(&b->mem, total_bytes);
    if (e != hipSuccess) {
        g_last_hip = e;
        delete b;
        return UA_ERR_NOMEM;
    }
    u8 *base = (u8 *)b->mem;
    b->d_descs = (UaDesc *)(base + o_desc);
    b->d_tb = (u64 *)(base + o_tb);
    b->d_toff = (u64 *)(base + o_toff);
    b->d_part = (u64 *)(base + o_part);
    b->d_pout = (u64 *)(base + o_pout);
    b->d_tpair = (u32 *)(base + o_tpair);
    b->d_ta0 = (u32 *)(base + o_ta0);
    b->d_tcnt = (u32 *)(base + o_tcnt);
    b->d_lbf = (u64 *)(base + o_lbf);
    b->d_tba = (u64 *)(base + o_tba);
    b->d_tpair_a = (u32 *)(base + o_tpa);
    b->d_bstart = (u32 *)(base + o_bst);

    std::vector<u8> hostbuf(o_tb + tb.size() * sizeof(u64));
    memcpy(hostbuf.data(), descs.data(), descs.size() * sizeof(UaDesc));
    memcpy(hostbuf.data() + o_tb, tb.data(), tb.size() * sizeof(u64));
    HIP_TRY(hipMemcpyAsync(b->mem, hostbuf.data(), hostbuf.size(),
                           hipMemcpyHostToDevice, c->stream));
    HIP_TRY(hipMemcpyAsync(b->d_tba, tba.data(), tba.size() * sizeof(u64),
                           hipMemcpyHostToDevice, c->stream));
    HIP_TRY(hipMemsetAsync(b->d_tcnt + T, 0, sizeof(u32), c->stream));
    HIP_TRY(hipMemsetAsync(b->d_tcnt + Ta, 0, sizeof(u32), c->stream));
    HIP_TRY(hipMemsetAsync(b->d_toff, 0, sizeof(u64), c->stream));
    HIP_TRY(hipMemsetAsync(b->d_part, 0, sizeof(u64), c->stream));
    /* lookback flags + pout start zeroed (gen 0 never used; zero-tile pairs
     * never publish a length and stay 0 from here) */
    HIP_TRY(hipMemsetAsync(b->d_lbf, 0, (T + 1) * sizeof(u64), c->stream));
    HIP_TRY(hipMemsetAsync(b->d_pout, 0,
                           (size_t)std::max(n_pairs, 1) * sizeof(u64), c->stream));
    if (T > 0) {
        /* the partition depends only on the (immutable) pair contents:
         * computed once here, reused every run */
        u64 pblk = (T + UA_BLOCK - 1) / UA_BLOCK;
        hipLaunchKernelGGL(k_partition, dim3((u32)pblk), dim3(UA_BLOCK), 0, c->stream,
                           b->d_descs, b->d_tb, n_pairs, T, b->d_tpair, b->d_ta0, 0);
        hipLaunchKernelGGL(k_partition, dim3((u32)pblk), dim3(UA_BLOCK), 0, c->stream,
                           b->d_descs, b->d_tb, n_pairs, T, b->d_tpair, b->d_ta0, 1);
    }
    if (Ta > 0) {
        u64 pblk = (Ta + UA_BLOCK - 1) / UA_BLOCK;
        hipLaunchKernelGGL(k_apartition, dim3((u32)pblk), dim3(UA_BLOCK), 0,
                           c->stream, b->d_descs, b->d_tba, n_pairs, Ta,
                           b->d_tpair_a, b->d_bstart);
    }
    HIP_TRY(hipStreamSynchronize(c->stream));
    HIP_TRY(hipGetLastError());
    *out = b;
    return UA_OK;
}

extern "C" void ua_batch_destroy(ua_ctx *c, ua_batch *b) {
    if (!b) return;
    (void)hipSetDevice(c->device);
    for (int i = 0; i < 3; i++)
        for (int s = 0; s < 2; s++)
            if (b->gexec[i][s]) (void)hipGraphExecDestroy(b->gexec[i][s]);
    if (b->h_pout) (void)hipHostFree(b->h_pout);
    if (b->mem) (void)hipFree(b->mem);
    if (b->mem1) (void)hipFree(b->mem1);
    if (b->d_stage) (void)hipFree(b->d_stage);
    if (b->d_stage_p) (void)hipFree(b->d_stage_p);
    if (b->d_stage1) (void)hipFree(b->d_stage1);
    if (b->d_isplit) (void)hipFree(b->d_isplit);
    delete b;
}

#ifndef UA_GRAPH
#define UA_GRAPH 1 /* hipGraph-capture the staged pipeline's AUX TAIL
                    * (scan/compact/pair_out/D2H -> one replay).  The tile
                    * kernel stays an eager launch so its HIP-event timing
                    * (bench.py roofline leg) remains live — event-record
                    * nodes inside replayed graphs do not update events on
                    * this ROCm.  0 = eager launches */
#endif

/* the staged TILE kernel (eager, event-timed) */
/* wave-register intersect path toggle (UA_AISECT=0 disables) */
static int aisect_enabled() {
    static int v = -1;
    if (v < 0) {
        const char *e = getenv("UA_AISECT");
        v = (e && e[0]) ? (e[0] != '0') : 0;
    }
    return v;
}

/* lazy split-cache: returns (isin, isout) for this run and flips i0_ready */
static void batch_isplit(ua_batch *b, const u32 **isin, u32 **isout) {
    *isin = nullptr;
    *isout = nullptr;
    /* the experimental kernels neither store nor read the PACKED cache —
     * engaging i0_ready with them active would hand a later default run an
     * unwritten buffer */
    if (pp_enabled() || rp_enabled()) return;
    if (!b->d_isplit) {
        size_t bytes = (size_t)b->total_tiles * UA_TBLOCK * sizeof(u32);
        if (bytes == 0 || hipMalloc((void **)&b->d_isplit, bytes) != hipSuccess) {
            b->d_isplit = nullptr; /* cacheless fallback */
            (void)hipGetLastError();
            return;
        }
    }
    if (b->i0_ready) *isin = b->d_isplit;
    else {
        *isout = b->d_isplit;
        b->i0_ready = true; /* ordered: runs are stream-serialized */
    }
}

static int batch_tiles_seq(ua_ctx *c, ua_batch *b, int kop, u64 stride,
                           bool record_events, int set) {
    u64 T = b->total_tiles;
    u64 *stage = set ? b->d_stage1 : b->d_stage;
    u32 *tcnt = set ? b->d_tcnt1 : b->d_tcnt;
    const u32 *isin;
    u32 *isout;
    batch_isplit(b, &isin, &isout);
    if ((kop == OP_INTERSECT || kop == OP_DIFF) && aisect_enabled() &&
        b->total_atiles > 0) {
        u64 Ta = b->total_atiles;
        u64 blk = (Ta + (UA_BLOCK / 64) - 1) / (UA_BLOCK / 64);
        if (record_events) HIP_TRY(hipEventRecord(c->ev[0], c->stream));
        if (kop == OP_INTERSECT)
            hipLaunchKernelGGL(k_aisect<OP_INTERSECT>, dim3((u32)blk),
                               dim3(UA_BLOCK), 0, c->stream, b->d_descs,
                               b->d_tpair_a, b->d_tba, b->d_bstart, Ta, stage,
                               tcnt);
        else
            hipLaunchKernelGGL(k_aisect<OP_DIFF>, dim3((u32)blk), dim3(UA_BLOCK),
                               0, c->stream, b->d_descs, b->d_tpair_a, b->d_tba,
                               b->d_bstart, Ta, stage, tcnt);
        if (record_events) HIP_TRY(hipEventRecord(c->ev[1], c->stream));
        return UA_OK;
    }
    if (record_events) HIP_TRY(hipEventRecord(c->ev[0], c->stream));
    /* dense primary staging rides the (otherwise unused) offs arg of the
     * STAGE launch; only for the default kernel and set 0 (overlap keeps
     * the overflow-only layout, and the experimental kernels ignore it) */
    const u64 *prim = nullptr;
    if (set == 0 && !pp_enabled() && !rp_enabled() && kop != OP_UNION) {
        if (!b->d_stage_p) {
            size_t pbytes = (size_t)(T ? T : 1) * UA_STAGE_P * sizeof(u64);
            if (hipMalloc((void **)&b->d_stage_p, pbytes) != hipSuccess) {
                b->d_stage_p = nullptr;
                (void)hipGetLastError();
            }
        }
        prim = b->d_stage_p;
    }
    if (kop == OP_INTERSECT) {
        launch_tiles<OP_INTERSECT, MODE_STAGE>(c, b->d_descs, b->d_tpair, b->d_ta0,
                                               T, stage, stride, tcnt,
                                               prim, nullptr, isin, isout);
    } else if (kop == OP_DIFF) {
        launch_tiles<OP_DIFF, MODE_STAGE>(c, b->d_descs, b->d_tpair, b->d_ta0, T,
                                          stage, stride, tcnt, prim,
                                          nullptr, isin, isout);
    } else {
        launch_tiles<OP_UNION, MODE_COUNT>(c, b->d_descs, b->d_tpair, b->d_ta0, T,
                                           nullptr, 0, b->d_tcnt, nullptr, nullptr,
                                           isin, isout);
    }
    if (record_events) HIP_TRY(hipEventRecord(c->ev[1], c->stream));
    return UA_OK;
}

/* the aux tail: scan + write/compact + pair_out + D2H of lens (captured
 * into a hipGraph for intersect/diff; union's WRITE pass is a second tile
 * kernel that needs live events, so union stays eager throughout) */
static int batch_tail_seq(ua_ctx *c, ua_batch *b, int kop, u64 stride,
                          u64 *host_pout, bool record_events, int set,
                          hipStream_t st) {
    /* active tile layout: the A-indexed one when the wave-register
     * intersect produced the counts (intersect/diff with UA_AISECT) */
    bool aA = (kop != OP_UNION) && aisect_enabled() && b->total_atiles > 0;
    u64 T = aA ? b->total_atiles : b->total_tiles;
    u64 nch = aA ? b->nchunks_a : b->nchunks;
    const u32 *tpair = aA ? b->d_tpair_a : b->d_tpair;
    const u64 *tbx = aA ? b->d_tba : b->d_tb;
    u64 stride_x = aA ? (u64)UA_AT : stride;
    u32 *tcnt = set ? b->d_tcnt1 : b->d_tcnt;
    u64 *toff = set ? b->d_toff1 : b->d_toff;
    u64 *part = set ? b->d_part1 : b->d_part;
    u64 *stage = set ? b->d_stage1 : b->d_stage;
    hipLaunchKernelGGL(k_scan1, dim3((u32)nch), dim3(UA_BLOCK), 0, st,
                       tcnt, T + 1, toff, part);
    hipLaunchKernelGGL(k_scan2, dim3(1), dim3(UA_BLOCK), 0, st, part,
                       nch);
    if (kop == OP_UNION) {
        /* the count pass of this same run already stored the splits */
        const u32 *isw = (b->i0_ready && b->d_isplit) ? b->d_isplit : nullptr;
        if (record_events) HIP_TRY(hipEventRecord(c->ev[2], st));
        launch_tiles<OP_UNION, MODE_WRITE>(c, b->d_descs, b->d_tpair, b->d_ta0, T,
                                           nullptr, 0, b->d_tcnt, b->d_toff,
                                           b->d_part, isw, nullptr);
        if (record_events) HIP_TRY(hipEventRecord(c->ev[3], st));
    } else {
        const u64 *prim = (set == 0 && !pp_enabled() && !rp_enabled())
                              ? b->d_stage_p
                              : nullptr;
        hipLaunchKernelGGL(k_compact, dim3((u32)((T + 15) / 16)), dim3(UA_BLOCK), 0,
                           st, b->d_descs, tpair, tbx, tcnt, toff,
                           part, stage, stride_x, prim, T, kop);
        hipLaunchKernelGGL(k_compact_small, dim3((u32)((T + UA_BLOCK - 1) / UA_BLOCK)),
                           dim3(UA_BLOCK), 0, st, b->d_descs, tpair, tbx, tcnt,
                           toff, part, stage, stride_x, prim, T, kop);
    }
    u64 poutblk = ((u64)b->n_pairs + UA_BLOCK - 1) / UA_BLOCK;
    hipLaunchKernelGGL(k_pair_out, dim3((u32)poutblk), dim3(UA_BLOCK), 0, st,
                       toff, part, tbx, b->n_pairs, b->d_pout);
    HIP_TRY(hipMemcpyAsync(host_pout, b->d_pout, (size_t)b->n_pairs * sizeof(u64),
                           hipMemcpyDeviceToHost, st));
    return UA_OK;
}

/* one tail dispatch: graph replay when available, else eager enqueue.
 * Returns UA_OK with the tail enqueued (or already executed, for the one
 * capture-validation run). */
static int batch_tail_dispatch(ua_ctx *c, ua_batch *b, int kop, u64 stride,
                               bool record_events, int set, hipStream_t st) {
#if UA_GRAPH
    if (getenv("UA_NO_TAILGRAPH")) b->no_graph = true;
    if (kop != OP_UNION) { /* union's tail holds an event-timed kernel */
        if (!b->gexec[kop][set] && !b->no_graph) {
            /* capture the tail once and VALIDATE with a launch+sync; any
             * failure makes this batch permanently eager — self-healing,
             * never fatal.  The one validation launch IS this run's tail
             * (capture itself executes nothing). */
            hipError_t ce = hipStreamBeginCapture(st,
                                                  hipStreamCaptureModeThreadLocal);
            if (ce == hipSuccess) {
                int rc = batch_tail_seq(c, b, kop, stride, b->h_pout, false, set, st);
                hipGraph_t gr = nullptr;
                hipError_t ee = hipStreamEndCapture(st, &gr);
                if (rc == UA_OK && ee == hipSuccess && gr) {
                    hipError_t ie = hipGraphInstantiate(&b->gexec[kop][set], gr,
                                                        nullptr, nullptr, 0);
                    if (ie == hipSuccess) {
                        hipError_t le = hipGraphLaunch(b->gexec[kop][set], st);
                        if (le == hipSuccess) le = hipStreamSynchronize(st);
                        if (le == hipSuccess) {
                            (void)hipGraphDestroy(gr);
                            return UA_OK; /* tail executed */
                        }
                        if (getenv("UA_DEBUG"))
                            fprintf(stderr, "[ua] graph launch failed: %s\n",
                                    hipGetErrorName(le));
                        (void)hipGraphExecDestroy(b->gexec[kop][set]);
                        b->gexec[kop][set] = nullptr;
                    } else if (getenv("UA_DEBUG")) {
                        fprintf(stderr, "[ua] graph instantiate failed: %s\n",
                                hipGetErrorName(ie));
                        b->gexec[kop][set] = nullptr;
                    }
                    (void)hipGraphDestroy(gr);
                } else {
                    if (getenv("UA_DEBUG"))
                        fprintf(stderr, "[ua] capture failed (rc=%d, end=%s)\n", rc,
                                hipGetErrorName(ee));
                    if (gr) (void)hipGraphDestroy(gr);
                }
                b->no_graph = true;
                (void)hipGetLastError();
            } else {
                b->no_graph = true;
                (void)hipGetLastError();
            }
        } else if (b->gexec[kop][set]) {
            hipError_t le = hipGraphLaunch(b->gexec[kop][set], st);
            if (le == hipSuccess) return UA_OK;
            (void)hipGraphExecDestroy(b->gexec[kop][set]);
            b->gexec[kop][set] = nullptr;
            b->no_graph = true;
            (void)hipGetLastError();
        }
    }
#endif
    return batch_tail_seq(c, b, kop, stride, b->h_pout, record_events, set, st);
}

/* n_runs passes of one op over the prepared batch, enqueued back-to-back
 * with ONE sync at the end (the repeated-query serving shape: no host
 * round-trip between runs).  Kernel events are sampled on the LAST pass
 * (stats count it as one sampled launch). */
static int batch_run_locked(ua_ctx *c, ua_batch *b, int op, int n_runs,
                            uint64_t *out_lens) {
    HIP_TRY(hipSetDevice(c->device));
    if (b->n_pairs == 0 || n_runs <= 0) return UA_OK;
    int kop = (op == UA_OP_INTERSECT) ? OP_INTERSECT
              : (op == UA_OP_MERGE) ? OP_UNION
              : (op == UA_OP_DIFFERENCE) ? OP_DIFF
                                         : -1;
    if (kop < 0) return UA_ERR_INVALID;
    u64 T = b->total_tiles;
    if (T == 0) {
        for (int p = 0; p < b->n_pairs; p++) out_lens[p] = 0;
        return UA_OK;
    }
    const bool lb = (kop == OP_UNION) ? (bool)UA_LOOKBACK_UNION : (bool)UA_LOOKBACK;
    bool two_kernels = false;
    int rc;
    if (lb) {
        for (int it = 0; it < n_runs; it++) {
            bool rec = (it == n_runs - 1);
            if (b->lb_gen >= UA_LB_GEN_MAX) {
                HIP_TRY(hipMemsetAsync(b->d_lbf, 0, (T + 1) * sizeof(u64), c->stream));
                b->lb_gen = 0;
            }
            b->lb_gen += 1;
            u64 gen = b->lb_gen;
            const u32 *isin;
            u32 *isout;
            batch_isplit(b, &isin, &isout);
            if (rec) HIP_TRY(hipEventRecord(c->ev[0], c->stream));
            if (kop == OP_INTERSECT) {
                launch_tiles<OP_INTERSECT, MODE_LOOKBACK>(c, b->d_descs, b->d_tpair,
                                                          b->d_ta0, T, b->d_lbf, gen,
                                                          nullptr, b->d_pout, nullptr,
                                                          isin, isout);
            } else if (kop == OP_DIFF) {
                launch_tiles<OP_DIFF, MODE_LOOKBACK>(c, b->d_descs, b->d_tpair,
                                                     b->d_ta0, T, b->d_lbf, gen,
                                                     nullptr, b->d_pout, nullptr,
                                                     isin, isout);
            } else {
                launch_tiles<OP_UNION, MODE_LOOKBACK>(c, b->d_descs, b->d_tpair,
                                                      b->d_ta0, T, b->d_lbf, gen,
                                                      nullptr, b->d_pout, nullptr,
                                                      isin, isout);
            }
            if (rec) HIP_TRY(hipEventRecord(c->ev[1], c->stream));
        }
        HIP_TRY(hipMemcpyAsync(out_lens, b->d_pout, (size_t)b->n_pairs * sizeof(u64),
                               hipMemcpyDeviceToHost, c->stream));
        HIP_TRY(hipStreamSynchronize(c->stream));
        HIP_TRY(hipGetLastError());
        /* zero-tile pairs stay 0 from the create-time d_pout memset */
    } else {
        two_kernels = (kop == OP_UNION);
        u64 stride = (kop == OP_INTERSECT) ? UA_TILE / 2 : UA_TILE;
        if (kop != OP_UNION && !b->d_stage) {
            hipError_t e = hipMalloc((void **)&b->d_stage,
                                     (T ? T : 1) * UA_TILE * sizeof(u64));
            if (e != hipSuccess) {
                g_last_hip = e;
                return UA_ERR_NOMEM;
            }
        }
        if (!b->h_pout) {
            hipError_t e = hipHostMalloc((void **)&b->h_pout,
                                         (size_t)b->n_pairs * sizeof(u64));
            if (e != hipSuccess) {
                g_last_hip = e;
                return UA_ERR_NOMEM;
            }
        }
        /* overlapped serving shape (UA_OVERLAP=1, measured NEUTRAL and
         * default-off): the aux tail (scan/compact/pair_out) of run i
         * executes on the tail stream against buffer set i&1 while run
         * i+1's tile kernel fills the other set.  Parity-green, but
         * co-scheduling contention stretches the tile kernel (727->766 us)
         * and k_compact (64->90 us) by about what the overlap hides —
         * whole-step 0.813 vs 0.802 ms serial, with one box showing a
         * pathological 1.65 ms at long run counts.  Union always serial
         * (its WRITE pass is a second tile kernel with live stats
         * events). */
        const char *ovl_env = getenv("UA_OVERLAP");
        bool ovl = (kop != OP_UNION) && n_runs > 1 && c->stream_tail &&
                   ovl_env && ovl_env[0] == '1';
        if (ovl && !b->mem1) {
            size_t o_cnt1 = 0;
            size_t o_off1 = align16(o_cnt1 + (T + 1) * sizeof(u32));
            size_t o_part1 = align16(o_off1 + (T + 1) * sizeof(u64));
            size_t tot1 = align16(o_part1 + (b->nchunks + 1) * sizeof(u64));
            hipError_t e = hipMalloc(&b->mem1, tot1);
            if (e == hipSuccess)
                e = hipMalloc((void **)&b->d_stage1,
                              (T ? T : 1) * UA_TILE * sizeof(u64));
            if (e != hipSuccess) {
                if (b->mem1) { (void)hipFree(b->mem1); b->mem1 = nullptr; }
                ovl = false; /* fall back to the serial path */
            } else {
                u8 *m1 = (u8 *)b->mem1;
                b->d_tcnt1 = (u32 *)(m1 + o_cnt1);
                b->d_toff1 = (u64 *)(m1 + o_off1);
                b->d_part1 = (u64 *)(m1 + o_part1);
                HIP_TRY(hipMemsetAsync(b->d_tcnt1 + T, 0, sizeof(u32), c->stream));
            }
        } else if (ovl && !b->d_stage1) {
            ovl = false;
        }
        if (ovl) {
            for (int it = 0; it < n_runs; it++) {
                bool rec = (it == n_runs - 1);
                int s = it & 1;
                /* set s's tail from run it-2 must be done before refilling */
                HIP_TRY(hipStreamWaitEvent(c->stream, c->oev[2 + s], 0));
                if ((rc = batch_tiles_seq(c, b, kop, stride, rec, s))) return rc;
                HIP_TRY(hipEventRecord(c->oev[s], c->stream));
                HIP_TRY(hipStreamWaitEvent(c->stream_tail, c->oev[s], 0));
                if ((rc = batch_tail_dispatch(c, b, kop, stride, rec, s,
                                              c->stream_tail)))
                    return rc;
                HIP_TRY(hipEventRecord(c->oev[2 + s], c->stream_tail));
            }
            HIP_TRY(hipStreamSynchronize(c->stream_tail));
            HIP_TRY(hipStreamSynchronize(c->stream));
        } else {
            for (int it = 0; it < n_runs; it++) {
                bool rec = (it == n_runs - 1);
                if ((rc = batch_tiles_seq(c, b, kop, stride, rec, 0))) return rc;
                if ((rc = batch_tail_dispatch(c, b, kop, stride, rec, 0, c->stream)))
                    return rc;
            }
            HIP_TRY(hipStreamSynchronize(c->stream));
        }
        HIP_TRY(hipGetLastError());
        memcpy(out_lens, b->h_pout, (size_t)b->n_pairs * sizeof(u64));
    }

    float ms = 0.f;
    if (hipEventElapsedTime(&ms, c->ev[0], c->ev[1]) == hipSuccess) {
        c->kernel_ms += ms;
        c->n_launches += 1;
    }
    if (two_kernels) {
        float ms2 = 0.f;
        if (hipEventElapsedTime(&ms2, c->ev[2], c->ev[3]) == hipSuccess) {
            c->kernel_ms += ms2;
            c->n_launches += 1;
        }
    }
    u64 out_elems = 0;
    for (int p = 0; p < b->n_pairs; p++) out_elems += out_lens[p];
    c->bytes_algo += ((u64)n_runs) * b->in_bytes + ((u64)n_runs) * 8 * out_elems;
    return UA_OK;
}

extern "C" int ua_batch_run(ua_ctx *c, ua_batch *b, int op, uint64_t *out_lens) {
    std::lock_guard<std::recursive_mutex> g(c->mu);
    return batch_run_locked(c, b, op, 1, out_lens);
}

/* pipelined repeated runs: one sync for n_runs passes */
extern "C" int ua_batch_run_n(ua_ctx *c, ua_batch *b, int op, int n_runs,
                              uint64_t *out_lens) {
    std::lock_guard<std::recursive_mutex> g(c->mu);
    return batch_run_locked(c, b, op, n_runs, out_lens);
}

extern "C" int ua_merge_batch_dev(ua_ctx *c, const ua_dpair *pairs, int n_pairs,
                                  uint64_t *out_lens) {
    return run_batch(c, pairs, n_pairs, out_lens, OP_UNION);
}

extern "C" int ua_difference_batch_dev(ua_ctx *c, const ua_dpair *pairs, int n_pairs,
                                       uint64_t *out_lens) {
    return run_batch(c, pairs, n_pairs, out_lens, OP_DIFF);
}

/* duplicate-keeping batched merge (sorted-run merge; out capacity exactly n+m) */
extern "C" int ua_merge_all_batch_dev(ua_ctx *c, const ua_dpair *pairs, int n_pairs,
                                      uint64_t *out_lens) {
    return run_batch(c, pairs, n_pairs, out_lens, OP_MERGE_ALL);
}

/* ---- batched segmented sort: LDS bitonic chunks + OP_MERGE_ALL tree ---- */
extern "C" int ua_sort_segments_dev(ua_ctx *c, const ua_dseg *segs, int n_segs) {
    std::lock_guard<std::recursive_mutex> gop(c->mu);
    HIP_TRY(hipSetDevice(c->device));
    if (n_segs <= 0) return UA_OK;

    u64 max_runs = 1;
    std::vector<UaChunk> chunks;
    std::vector<int> seg_rounds(n_segs);
    for (int sgi = 0; sgi < n_segs; sgi++) {
        u64 n = segs[sgi].n;
        u64 runs = (n + UA_SORT_N - 1) / UA_SORT_N;
        if (runs > max_runs) max_runs = runs;
        int r = 0;
        for (u64 w = 1; w < runs; w <<= 1) r++;
        seg_rounds[sgi] = r;
    }
    int rounds = 0;
    for (u64 w = 1; w < max_runs; w <<= 1) rounds++;
    /* per-segment start side: a segment with r merge rounds of its own flips
     * sides r times, so start it where it will END in data — finished
     * segments then never re-copy while bigger ones keep merging */
    for (int sgi = 0; sgi < n_segs; sgi++) {
        const ua_dseg &sg = segs[sgi];
        u64 *dst = (seg_rounds[sgi] & 1) ? sg.tmp : sg.data;
        for (u64 off = 0; off < sg.n; off += UA_SORT_N) {
            u32 len = (u32)std::min<u64>(UA_SORT_N, sg.n - off);
            chunks.push_back({sg.data + off, dst + off, len, 0});
        }
    }
    if (!chunks.empty()) {
        int rc;
        if ((rc = ws_reserve(c, WS_DESC, chunks.size() * sizeof(UaChunk)))) return rc;
        HIP_TRY(hipMemcpyAsync(c->ws[WS_DESC], chunks.data(),
                               chunks.size() * sizeof(UaChunk), hipMemcpyHostToDevice,
                               c->stream));
        hipLaunchKernelGGL(k_sort_chunks, dim3((u32)chunks.size()), dim3(UA_BLOCK), 0,
                           c->stream, (const UaChunk *)c->ws[WS_DESC],
                           (u64)chunks.size());
        HIP_TRY(hipStreamSynchronize(c->stream));
    }

    u64 width = UA_SORT_N;
    for (int r = 0; r < rounds; r++) {
        std::vector<ua_dpair> prs;
        for (int sgi = 0; sgi < n_segs; sgi++) {
            if (r >= seg_rounds[sgi]) continue; /* segment already fully sorted */
            const ua_dseg &sg = segs[sgi];
            /* this segment has flipped r times from its start side */
            int side = (seg_rounds[sgi] - r) & 1; /* 1 = current runs in tmp */
            u64 *cur = side ? sg.tmp : sg.data;
            u64 *nxt = side ? sg.data : sg.tmp;
            for (u64 off = 0; off < sg.n; off += 2 * width) {
                u64 n1 = std::min<u64>(width, sg.n - off);
                u64 rem = sg.n - off - n1;
                u64 n2 = std::min<u64>(width, rem);
                /* odd trailing run pairs with an empty side = a copy */
                prs.push_back({cur + off, n1, cur + off + n1, n2, nxt + off});
            }
        }
        if (!prs.empty()) {
            std::vector<u64> lens(prs.size());
            int rc = run_batch_locked(c, prs.data(), (int)prs.size(), lens.data(),
                                      OP_MERGE_ALL);
            if (rc) return rc;
        }
        width <<= 1;
    }
    return UA_OK; /* every segment ends in data by start-side parity */
}

/* ---- batched algo.ApplyFilter (uidlist.go:21; callers worker/task.go:1403,
 * query/query.go:1431) ---- */
extern "C" int ua_apply_filter_batch_dev(ua_ctx *c, const ua_dfilter *tasks,
                                         int n_tasks, uint64_t *out_lens) {
    std::lock_guard<std::recursive_mutex> g(c->mu);
    if (n_tasks <= 0) return UA_OK;
    HIP_TRY(hipSetDevice(c->device));
    std::vector<UaFDesc> descs((size_t)n_tasks);
    std::vector<u64> tb((size_t)n_tasks + 1);
    u64 T = 0, in_bytes = 0;
    for (int p = 0; p < n_tasks; p++) {
        const ua_dfilter &f = tasks[p];
        if (f.n >= (1ull << 31)) return UA_ERR_INVALID;
        descs[p] = {f.u, f.n, f.mask, f.out, T};
        tb[p] = T;
        T += (f.n + UA_TILE - 1) / UA_TILE;
        in_bytes += 9 * f.n; /* 8 B uid + 1 B mask */
    }
    tb[n_tasks] = T;
    int rc;
    size_t descs_bytes = descs.size() * sizeof(UaFDesc);
    size_t tb_bytes = tb.size() * sizeof(u64);
    if ((rc = ws_reserve(c, WS_DESC, descs_bytes + tb_bytes))) return rc;
    if ((rc = ws_reserve(c, WS_POUT, (size_t)n_tasks * sizeof(u64)))) return rc;
    UaFDesc *d_descs = (UaFDesc *)c->ws[WS_DESC];
    u64 *d_tb = (u64 *)((u8 *)c->ws[WS_DESC] + descs_bytes);
    u64 *d_pout = (u64 *)c->ws[WS_POUT];
    std::vector<u8> hostbuf(descs_bytes + tb_bytes);
    memcpy(hostbuf.data(), descs.data(), descs_bytes);
    memcpy(hostbuf.data() + descs_bytes, tb.data(), tb_bytes);
    HIP_TRY(hipMemcpyAsync(d_descs, hostbuf.data(), hostbuf.size(),
                           hipMemcpyHostToDevice, c->stream));
    if (T > 0) {
        u64 *d_lbf;
        u64 gen;
        if ((rc = lb_acquire_ws(c, T, &d_lbf, &gen))) return rc;
        HIP_TRY(hipEventRecord(c->ev[0], c->stream));
        hipLaunchKernelGGL(k_filter, dim3((u32)T), dim3(UA_BLOCK), 0, c->stream,
                           d_descs, d_tb, n_tasks, T, d_lbf, gen, d_pout);
        HIP_TRY(hipEventRecord(c->ev[1], c->stream));
    }
    HIP_TRY(hipMemcpyAsync(out_lens, d_pout, (size_t)n_tasks * sizeof(u64),
                           hipMemcpyDeviceToHost, c->stream));
    HIP_TRY(hipStreamSynchronize(c->stream));
    HIP_TRY(hipGetLastError());
    u64 out_elems = 0;
    for (int p = 0; p < n_tasks; p++) {
        if (tasks[p].n == 0) out_lens[p] = 0; /* no tiles -> never published */
        out_elems += out_lens[p];
    }
    if (T > 0) {
        float ms = 0.f;
        HIP_TRY(hipEventElapsedTime(&ms, c->ev[0], c->ev[1]));
        c->kernel_ms += ms;
        c->n_launches += 1;
    }
    c->bytes_algo += in_bytes + 8 * out_elems;
    return UA_OK;
}

/* host-pointer algo.ApplyFilter: in-place like the reference (u is compacted
 * to *out_n kept elements); upload + device compaction + download */
extern "C" int ua_apply_filter(ua_ctx *c, uint64_t *u, uint64_t n,
                               const uint8_t *mask, uint64_t *out_n) {
    std::lock_guard<std::recursive_mutex> g(c->mu);
    HIP_TRY(hipSetDevice(c->device));
    if (n == 0) {
        *out_n = 0;
        return UA_OK;
    }
    int rc;
    if ((rc = ws_reserve(c, WS_HU, n * sizeof(u64)))) return rc;
    if ((rc = ws_reserve(c, WS_HV, n))) return rc;
    u64 *d_u = (u64 *)c->ws[WS_HU];
    u8 *d_mask = (u8 *)c->ws[WS_HV];
    HIP_TRY(hipMemcpyAsync(d_u, u, n * sizeof(u64), hipMemcpyHostToDevice, c->stream));
    HIP_TRY(hipMemcpyAsync(d_mask, mask, n, hipMemcpyHostToDevice, c->stream));
    ua_dfilter task = {d_u, n, d_mask, d_u}; /* in-place on device too */
    if ((rc = ua_apply_filter_batch_dev(c, &task, 1, out_n))) return rc;
    HIP_TRY(hipMemcpyAsync(u, d_u, *out_n * sizeof(u64), hipMemcpyDeviceToHost,
                           c->stream));
    HIP_TRY(hipStreamSynchronize(c->stream));
    return UA_OK;
}

extern "C" int ua_index_of_batch_dev(ua_ctx *c, const uint64_t *u, uint64_t n,
                                     const uint64_t *queries, uint64_t nq, int64_t *out) {
    std::lock_guard<std::recursive_mutex> g(c->mu);
    HIP_TRY(hipSetDevice(c->device));
    if (nq == 0) return UA_OK;
    u64 nblk = (nq + UA_BLOCK - 1) / UA_BLOCK;
    hipLaunchKernelGGL(k_index_of, dim3((u32)nblk), dim3(UA_BLOCK), 0, c->stream,
                       u, n, queries, nq, out);
    HIP_TRY(hipStreamSynchronize(c->stream));
    HIP_TRY(hipGetLastError());
    return UA_OK;
}

/* ---- IntersectSorted fold (uidlist.go:297): smallest-first pairwise ---- */
extern "C" int ua_intersect_k_dev(ua_ctx *c, const uint64_t *const *lists,
                                  const uint64_t *lens, int k, uint64_t *out,
                                  uint64_t *out_n) {
    std::lock_guard<std::recursive_mutex> gop(c->mu); /* whole-op: scratch reuse */
    if (k <= 0) {
        *out_n = 0;
        return UA_OK;
    }
    std::vector<int> ord(k);
    for (int i = 0; i < k; i++) ord[i] = i;
    std::stable_sort(ord.begin(), ord.end(), [&](int a, int b) { return lens[a] < lens[b]; });
    if (k == 1) {
        std::lock_guard<std::recursive_mutex> g(c->mu);
        HIP_TRY(hipSetDevice(c->device));
        HIP_TRY(hipMemcpyAsync(out, lists[0], lens[0] * sizeof(u64),
                               hipMemcpyDeviceToDevice, c->stream));
        HIP_TRY(hipStreamSynchronize(c->stream));
        *out_n = lens[0];
        return UA_OK;
    }
    u64 cap = lens[ord[0]];
    int rc;
    {
        std::lock_guard<std::recursive_mutex> g(c->mu);
        if ((rc = ws_reserve(c, WS_SCRATCH_A, (cap ? cap : 1) * sizeof(u64)))) return rc;
    }
    u64 *scratch = (u64 *)c->ws[WS_SCRATCH_A];
    ua_dpair pr;
    u64 len0 = 0;
    pr = {lists[ord[0]], lens[ord[0]], lists[ord[1]], lens[ord[1]], out};
    if ((rc = run_batch(c, &pr, 1, &len0, OP_INTERSECT))) return rc;
    for (int j = 2; j < k && len0 > 0; j++) {
        pr = {out, len0, lists[ord[j]], lens[ord[j]], scratch};
        if ((rc = run_batch(c, &pr, 1, &len0, OP_INTERSECT))) return rc;
        std::lock_guard<std::recursive_mutex> g(c->mu);
        HIP_TRY(hipSetDevice(c->device));
        HIP_TRY(hipMemcpyAsync(out, scratch, len0 * sizeof(u64),
                               hipMemcpyDeviceToDevice, c->stream));
        HIP_TRY(hipStreamSynchronize(c->stream));
    }
    *out_n = len0;
    return UA_OK;
}

/* ---- MergeSorted k-way (uidlist.go:448): pairwise union tree ---- */
extern "C" int ua_merge_k_dev(ua_ctx *c, const uint64_t *const *lists,
                              const uint64_t *lens, int k, uint64_t *out,
                              uint64_t *out_n) {
    std::lock_guard<std::recursive_mutex> gop(c->mu); /* whole-op: scratch reuse */
    if (k <= 0) {
        *out_n = 0;
        return UA_OK;
    }
    if (k == 1) {
        /* the reference's heap merge dedups even a single list
         * (TestMergeSorted9: {1,1,1} -> {1}): union with the empty list */
        ua_dpair pr = {lists[0], lens[0], lists[0], 0, out};
        return run_batch_locked(c, &pr, 1, out_n, OP_UNION);
    }
    /* Device-chained pairwise union tree: positions and tile layouts are
     * CAPACITY-based (host-known up front), while each round's actual pair
     * lengths are filled on-device from the previous round's outputs
     * (k_make_descs) — the whole log2(k)-round tree enqueues with a single
     * final sync.  Tail tiles past a pair's actual path no-op in k_tiles.
     * Odd items pair with an empty side (a dedup copy), so every item moves
     * to the other ping-pong buffer each round. */
    HIP_TRY(hipSetDevice(c->device));
    u64 total = 0;
    for (int i = 0; i < k; i++) total += lens[i];
    int rc;
    if ((rc = ws_reserve(c, WS_SCRATCH_A, (total ? total : 1) * sizeof(u64)))) return rc;
    if ((rc = ws_reserve(c, WS_SCRATCH_B, (total ? total : 1) * sizeof(u64)))) return rc;
    u64 *bufs[2] = {(u64 *)c->ws[WS_SCRATCH_A], (u64 *)c->ws[WS_SCRATCH_B]};

    /* host planning over capacities */
    struct Round {
        int nk_prev, npair;
        u64 total_tiles;
        size_t desc_off, tb_off; /* element offsets into the concat uploads */
    };
    std::vector<Round> rounds;
    std::vector<UaDesc> all_descs;
    std::vector<u64> all_tb;
    std::vector<u64> caps(lens, lens + k);
    std::vector<const u64 *> ptrs(lists, lists + k);
    u64 max_tiles = 0, cap_work = 0;
    int which = 0;
    while ((int)caps.size() > 1) {
        int nk = (int)caps.size();
        int npair = (nk + 1) / 2;
        Round r;
        r.nk_prev = nk;
        r.npair = npair;
        r.desc_off = all_descs.size();
        r.tb_off = all_tb.size();
        u64 *buf = bufs[which];
        std::vector<u64> ncaps;
        std::vector<const u64 *> nptrs;
        u64 off = 0, tiles = 0;
        for (int pj = 0; pj < npair; pj++) {
            u64 ca = caps[2 * pj];
            u64 cb = (2 * pj + 1 < nk) ? caps[2 * pj + 1] : 0;
            const u64 *pa = ptrs[2 * pj];
            const u64 *pb = (2 * pj + 1 < nk) ? ptrs[2 * pj + 1] : ptrs[2 * pj];
            all_tb.push_back(tiles);
            /* n,m are placeholders; k_make_descs overwrites from device lens */
            all_descs.push_back({pa, ca, pb, cb, buf + off, tiles});
            tiles += (ca + cb + UA_TILE - 1) / UA_TILE;
            nptrs.push_back(buf + off);
            ncaps.push_back(ca + cb);
            off += ca + cb;
            cap_work += ca + cb;
        }
        all_tb.push_back(tiles);
        r.total_tiles = tiles;
        if (tiles > max_tiles) max_tiles = tiles;
        rounds.push_back(r);
        caps.swap(ncaps);
        ptrs.swap(nptrs);
        which ^= 1;
    }
    int R = (int)rounds.size();

    /* workspace: concat descs+tb+lens ping-pong in WS_DESC; tiles in the
     * usual slots sized by the largest round */
    size_t descs_bytes = all_descs.size() * sizeof(UaDesc);
    size_t tb_bytes = all_tb.size() * sizeof(u64);
    size_t lens_bytes = (size_t)k * sizeof(u64);
    if ((rc = ws_reserve(c, WS_DESC, descs_bytes + tb_bytes + 2 * lens_bytes + 64)))
        return rc;
    u8 *wbase = (u8 *)c->ws[WS_DESC];
    UaDesc *d_descs_all = (UaDesc *)wbase;
    u64 *d_tb_all = (u64 *)(wbase + descs_bytes);
    u64 *d_lens[2] = {(u64 *)(wbase + descs_bytes + tb_bytes),
                      (u64 *)(wbase + descs_bytes + tb_bytes + lens_bytes)};
    {
        std::vector<u8> hostbuf(descs_bytes + tb_bytes + lens_bytes);
        memcpy(hostbuf.data(), all_descs.data(), descs_bytes);
        memcpy(hostbuf.data() + descs_bytes, all_tb.data(), tb_bytes);
        memcpy(hostbuf.data() + descs_bytes + tb_bytes, lens, lens_bytes);
        HIP_TRY(hipMemcpyAsync(wbase, hostbuf.data(), hostbuf.size(),
                               hipMemcpyHostToDevice, c->stream));
        HIP_TRY(hipStreamSynchronize(c->stream)); /* hostbuf is scoped */
    }
    if ((rc = ws_reserve(c, WS_TPAIR, (max_tiles + 1) * sizeof(u32)))) return rc;
    if ((rc = ws_reserve(c, WS_TA0, (max_tiles + 1) * sizeof(u32)))) return rc;
    if ((rc = ws_reserve(c, WS_TCNT, (max_tiles + 1) * sizeof(u32)))) return rc;
    if ((rc = ws_reserve(c, WS_TOFF, (max_tiles + 1) * sizeof(u64)))) return rc;
    u64 max_chunks = (max_tiles + 1 + UA_SCAN_CHUNK - 1) / UA_SCAN_CHUNK;
    if ((rc = ws_reserve(c, WS_PARTIAL, (max_chunks + 1) * sizeof(u64)))) return rc;
#if UA_LOOKBACK_UNION
    /* pre-size the lookback flag array so per-round acquires never realloc
     * mid-enqueue (hipFree would drain the stream) */
    if ((rc = ws_reserve(c, WS_LBF, (max_tiles + 1) * sizeof(u64)))) return rc;
#endif
    u32 *d_tpair = (u32 *)c->ws[WS_TPAIR];
    u32 *d_ta0 = (u32 *)c->ws[WS_TA0];
    u32 *d_tcnt = (u32 *)c->ws[WS_TCNT];
    u64 *d_toff = (u64 *)c->ws[WS_TOFF];
    u64 *d_part = (u64 *)c->ws[WS_PARTIAL];
    (void)d_tcnt;
    (void)d_toff;
    (void)d_part;

    HIP_TRY(hipEventRecord(c->ev[0], c->stream));
    int lcur = 0;
    for (int r = 0; r < R; r++) {
        const Round &rd = rounds[r];
        UaDesc *d_descs = d_descs_all + rd.desc_off;
        u64 *d_tb = d_tb_all + rd.tb_off;
        u64 T = rd.total_tiles;
        u64 mk = ((u64)rd.npair + UA_BLOCK - 1) / UA_BLOCK;
        hipLaunchKernelGGL(k_make_descs, dim3((u32)mk), dim3(UA_BLOCK), 0, c->stream,
                           d_descs, d_lens[lcur], rd.nk_prev, rd.npair);
#if UA_LOOKBACK_UNION
        /* zero-capacity pairs have no tiles and never publish a length */
        HIP_TRY(hipMemsetAsync(d_lens[lcur ^ 1], 0, (size_t)rd.npair * sizeof(u64),
                               c->stream));
        if (T > 0) {
            u64 pblk = (T + UA_BLOCK - 1) / UA_BLOCK;
            hipLaunchKernelGGL(k_partition, dim3((u32)pblk), dim3(UA_BLOCK), 0, c->stream,
                               d_descs, d_tb, rd.npair, T, d_tpair, d_ta0, 0);
            hipLaunchKernelGGL(k_partition, dim3((u32)pblk), dim3(UA_BLOCK), 0, c->stream,
                               d_descs, d_tb, rd.npair, T, d_tpair, d_ta0, 1);
            u64 *d_lbf;
            u64 gen;
            if ((rc = lb_acquire_ws(c, T, &d_lbf, &gen))) return rc;
            launch_tiles<OP_UNION, MODE_LOOKBACK>(c, d_descs, d_tpair, d_ta0, T,
                                                  d_lbf, gen, nullptr,
                                                  d_lens[lcur ^ 1], nullptr);
        }
#else
        HIP_TRY(hipMemsetAsync(d_tcnt + T, 0, sizeof(u32), c->stream));
        if (T > 0) {
            u64 pblk = (T + UA_BLOCK - 1) / UA_BLOCK;
            hipLaunchKernelGGL(k_partition, dim3((u32)pblk), dim3(UA_BLOCK), 0, c->stream,
                               d_descs, d_tb, rd.npair, T, d_tpair, d_ta0, 0);
            hipLaunchKernelGGL(k_partition, dim3((u32)pblk), dim3(UA_BLOCK), 0, c->stream,
                               d_descs, d_tb, rd.npair, T, d_tpair, d_ta0, 1);
            launch_tiles<OP_UNION, MODE_COUNT>(c, d_descs, d_tpair, d_ta0, T, nullptr,
                                               0, d_tcnt, nullptr, nullptr);
            u64 nchunks = (T + 1 + UA_SCAN_CHUNK - 1) / UA_SCAN_CHUNK;
            hipLaunchKernelGGL(k_scan1, dim3((u32)nchunks), dim3(UA_BLOCK), 0, c->stream,
                               d_tcnt, T + 1, d_toff, d_part);
            hipLaunchKernelGGL(k_scan2, dim3(1), dim3(UA_BLOCK), 0, c->stream, d_part,
                               nchunks);
            launch_tiles<OP_UNION, MODE_WRITE>(c, d_descs, d_tpair, d_ta0, T, nullptr,
                                               0, d_tcnt, d_toff, d_part);
        } else {
            HIP_TRY(hipMemsetAsync(d_toff, 0, sizeof(u64), c->stream));
            HIP_TRY(hipMemsetAsync(d_part, 0, sizeof(u64), c->stream));
        }
        u64 poutblk = ((u64)rd.npair + UA_BLOCK - 1) / UA_BLOCK;
        hipLaunchKernelGGL(k_pair_out, dim3((u32)poutblk), dim3(UA_BLOCK), 0, c->stream,
                           d_toff, d_part, d_tb, rd.npair, d_lens[lcur ^ 1]);
#endif
        lcur ^= 1;
    }
    /* final item: data in ptrs[0] (capacity position), length in d_lens[lcur][0] */
    u64 copy_grid = ((total ? total : 1) + UA_BLOCK - 1) / UA_BLOCK;
    if (copy_grid > 2048) copy_grid = 2048;
    hipLaunchKernelGGL(k_copy_len, dim3((u32)copy_grid), dim3(UA_BLOCK), 0, c->stream,
                       out, ptrs[0], d_lens[lcur]);
    u64 final_len = 0;
    HIP_TRY(hipMemcpyAsync(&final_len, d_lens[lcur], sizeof(u64), hipMemcpyDeviceToHost,
                           c->stream));
    HIP_TRY(hipStreamSynchronize(c->stream));
    HIP_TRY(hipGetLastError());
    float ms = 0.f;
    HIP_TRY(hipEventRecord(c->ev[1], c->stream));
    HIP_TRY(hipStreamSynchronize(c->stream));
    HIP_TRY(hipEventElapsedTime(&ms, c->ev[0], c->ev[1]));
    c->kernel_ms += ms;
#if UA_LOOKBACK_UNION
    c->n_launches += R;
    c->bytes_algo += 8 * 2 * cap_work; /* capacity upper bound (in + out, one pass) */
#else
    c->n_launches += 2 * R;
    c->bytes_algo += 8 * 3 * cap_work; /* capacity upper bound (union count+write) */
#endif
    *out_n = final_len;
    return UA_OK;
}

/* Pack-limit validation: k_packed decodes blocks only up to
 * num_uids <= UA_MAX_BLOCK_UIDS and deltas_len <= UA_MAX_DELTAS.  The
 * reference supports arbitrary BlockSize, so packs encoded elsewhere must be
 * REJECTED (UA_ERR_INVALID), not silently zeroed by the in-kernel guard
 * (ADVICE r01). */
__global__ __launch_bounds__(UA_BLOCK) void k_validate_pack(
    const u32 *__restrict__ num_uids, const u64 *__restrict__ delta_offs, u64 nb,
    u32 *__restrict__ viol) {
    u64 b = (u64)blockIdx.x * UA_BLOCK + threadIdx.x;
    if (b >= nb) return;
    if (num_uids[b] > UA_MAX_BLOCK_UIDS ||
        delta_offs[b + 1] - delta_offs[b] > UA_MAX_DELTAS)
        atomicOr(viol, 1u);
}

/* enqueue the validation on the ctx stream (read the flag after a sync) */
static int pack_validate_begin(ua_ctx *c, const u32 *num_uids, const u64 *delta_offs,
                               u64 nb) {
    if (!c->d_viol) {
        hipError_t e = hipMalloc((void **)&c->d_viol, sizeof(u32));
        if (e != hipSuccess) {
            g_last_hip = e;
            return UA_ERR_NOMEM;
        }
    }
    HIP_TRY(hipMemsetAsync(c->d_viol, 0, sizeof(u32), c->stream));
    if (nb) {
        u64 g = (nb + UA_BLOCK - 1) / UA_BLOCK;
        hipLaunchKernelGGL(k_validate_pack, dim3((u32)g), dim3(UA_BLOCK), 0, c->stream,
                           num_uids, delta_offs, nb, c->d_viol);
    }
    return UA_OK;
}

/* read the flag; stream must already be synchronized */
static int pack_validate_end(ua_ctx *c) {
    u32 viol = 0;
    HIP_TRY(hipMemcpy(&viol, c->d_viol, sizeof(u32), hipMemcpyDeviceToHost));
    return viol ? UA_ERR_INVALID : UA_OK;
}

/* ---- packed pipeline ---- */
static int run_packed_locked(ua_ctx *c, const ua_dpack *pk, u64 after, const u64 *v, u64 m,
                             u64 *out, u64 *out_n, int decode_only) {
    HIP_TRY(hipSetDevice(c->device));
    u64 nb = pk->n_blocks;
    if (nb == 0) {
        *out_n = 0;
        return UA_OK;
    }
    int rc;
    if ((rc = ws_reserve(c, WS_TCNT, (nb + 1) * sizeof(u32)))) return rc;
    if ((rc = ws_reserve(c, WS_TOFF, (nb + 1) * sizeof(u64)))) return rc;
    if ((rc = ws_reserve(c, WS_STAGE, nb * UA_MAX_BLOCK_UIDS * sizeof(u64)))) return rc;
    u32 *d_cnt = (u32 *)c->ws[WS_TCNT];
    u64 *d_off = (u64 *)c->ws[WS_TOFF];
    u64 *d_stage = (u64 *)c->ws[WS_STAGE];
    HIP_TRY(hipMemsetAsync(d_cnt + nb, 0, sizeof(u32), c->stream));

    if ((rc = pack_validate_begin(c, pk->num_uids, pk->delta_offs, nb))) return rc;
    u64 nwg = (nb + UA_PKW - 1) / UA_PKW;
    HIP_TRY(hipEventRecord(c->ev[0], c->stream));
    if (decode_only) {
        hipLaunchKernelGGL(k_packed<1>, dim3((u32)nwg), dim3(UA_BLOCK), 0, c->stream,
                           pk->bases, pk->num_uids, pk->delta_offs, pk->deltas, nb, after,
                           v, m, d_stage, d_cnt, (const u64 *)nullptr, 0,
                           (const ua_ptask *)nullptr);
    } else {
        hipLaunchKernelGGL(k_packed<0>, dim3((u32)nwg), dim3(UA_BLOCK), 0, c->stream,
                           pk->bases, pk->num_uids, pk->delta_offs, pk->deltas, nb, after,
                           v, m, d_stage, d_cnt, (const u64 *)nullptr, 0,
                           (const ua_ptask *)nullptr);
    }
    HIP_TRY(hipEventRecord(c->ev[1], c->stream));

    if ((rc = run_scan(c, d_cnt, nb + 1, d_off))) return rc;
    u64 *d_part = (u64 *)c->ws[WS_PARTIAL];
    hipLaunchKernelGGL(k_compact_flat, dim3((u32)((nb + 15) / 16)), dim3(UA_BLOCK), 0,
                       c->stream, out, d_cnt, d_off, d_part, d_stage,
                       (u64)UA_MAX_BLOCK_UIDS, nb);
    u64 off_nb = 0, part_nb = 0;
    HIP_TRY(hipMemcpyAsync(&off_nb, d_off + nb, sizeof(u64), hipMemcpyDeviceToHost,
                           c->stream));
    HIP_TRY(hipMemcpyAsync(&part_nb, d_part + nb / UA_SCAN_CHUNK, sizeof(u64),
                           hipMemcpyDeviceToHost, c->stream));
    HIP_TRY(hipStreamSynchronize(c->stream));
    HIP_TRY(hipGetLastError());
    *out_n = off_nb + part_nb;

    float ms = 0.f;
    HIP_TRY(hipEventElapsedTime(&ms, c->ev[0], c->ev[1]));
    c->kernel_ms += ms;
    c->n_launches += 1;
    /* algorithmic bytes: deltas + block headers + 8*(m + out) */
    u64 hdr = nb * (8 + 4 + 8);
    u64 dbytes = 0;
    HIP_TRY(hipMemcpyAsync(&dbytes, pk->delta_offs + nb, sizeof(u64),
                           hipMemcpyDeviceToHost, c->stream));
    HIP_TRY(hipStreamSynchronize(c->stream));
    if ((rc = pack_validate_end(c))) {
        *out_n = 0;
        return rc;
    }
    c->bytes_algo += hdr + dbytes + 8 * (m + *out_n);
    return UA_OK;
}

extern "C" int ua_intersect_packed_dev(ua_ctx *c, const ua_dpack *pk, uint64_t after_uid,
                                       const uint64_t *v, uint64_t m, uint64_t *out,
                                       uint64_t *out_n) {
    std::lock_guard<std::recursive_mutex> g(c->mu);
    return run_packed_locked(c, pk, after_uid, v, m, out, out_n, 0);
}

/* ---- multi-pack fan-out: one grid over every pack's blocks ---- */
extern "C" int ua_intersect_packed_batch_dev(ua_ctx *c, const uint64_t *bases,
                                             const uint32_t *num_uids,
                                             const uint64_t *delta_offs,
                                             const uint8_t *deltas,
                                             const uint64_t *pack_block_base,
                                             int n_packs, const ua_ptask *tasks,
                                             uint64_t *out_lens) {
    std::lock_guard<std::recursive_mutex> g(c->mu);
    HIP_TRY(hipSetDevice(c->device));
    if (n_packs <= 0) return UA_OK;
    u64 nb = pack_block_base[n_packs];
    int rc;
    /* upload pbb + tasks in one copy (WS_DESC is free during this call) */
    size_t pbb_bytes = ((size_t)n_packs + 1) * sizeof(u64);
    size_t task_bytes = (size_t)n_packs * sizeof(ua_ptask);
    if ((rc = ws_reserve(c, WS_DESC, pbb_bytes + task_bytes))) return rc;
    u64 *d_pbb = (u64 *)c->ws[WS_DESC];
    ua_ptask *d_tasks = (ua_ptask *)((u8 *)c->ws[WS_DESC] + pbb_bytes);
    std::vector<u8> hostbuf(pbb_bytes + task_bytes);
    memcpy(hostbuf.data(), pack_block_base, pbb_bytes);
    memcpy(hostbuf.data() + pbb_bytes, tasks, task_bytes);
    HIP_TRY(hipMemcpyAsync(c->ws[WS_DESC], hostbuf.data(), hostbuf.size(),
                           hipMemcpyHostToDevice, c->stream));

    if ((rc = ws_reserve(c, WS_TCNT, (nb + 1) * sizeof(u32)))) return rc;
    if ((rc = ws_reserve(c, WS_TOFF, (nb + 1) * sizeof(u64)))) return rc;
    if ((rc = ws_reserve(c, WS_STAGE, (nb ? nb : 1) * UA_MAX_BLOCK_UIDS * sizeof(u64))))
        return rc;
    if ((rc = ws_reserve(c, WS_POUT, (size_t)n_packs * sizeof(u64)))) return rc;
    u32 *d_cnt = (u32 *)c->ws[WS_TCNT];
    u64 *d_off_arr = (u64 *)c->ws[WS_TOFF];
    u64 *d_stage = (u64 *)c->ws[WS_STAGE];
    u64 *d_pout = (u64 *)c->ws[WS_POUT];
    HIP_TRY(hipMemsetAsync(d_cnt + nb, 0, sizeof(u32), c->stream));

    if (nb > 0) {
        if ((rc = pack_validate_begin(c, num_uids, delta_offs, nb))) return rc;
        u64 nwg = (nb + UA_PKW - 1) / UA_PKW;
        HIP_TRY(hipEventRecord(c->ev[0], c->stream));
        hipLaunchKernelGGL(k_packed<0>, dim3((u32)nwg), dim3(UA_BLOCK), 0, c->stream,
                           bases, num_uids, delta_offs, deltas, nb, (u64)0,
                           (const u64 *)nullptr, (u64)0, d_stage, d_cnt, d_pbb,
                           n_packs, d_tasks);
        HIP_TRY(hipEventRecord(c->ev[1], c->stream));
    }
    if ((rc = run_scan(c, d_cnt, nb + 1, d_off_arr))) return rc;
    u64 *d_part = (u64 *)c->ws[WS_PARTIAL];
    if (nb > 0) {
        hipLaunchKernelGGL(k_compact_pack, dim3((u32)((nb + 15) / 16)), dim3(UA_BLOCK),
                           0, c->stream, d_tasks, d_pbb, n_packs, d_cnt, d_off_arr,
                           d_part, d_stage, nb);
    }
    u64 poutblk = ((u64)n_packs + UA_BLOCK - 1) / UA_BLOCK;
    hipLaunchKernelGGL(k_pair_out, dim3((u32)poutblk), dim3(UA_BLOCK), 0, c->stream,
                       d_off_arr, d_part, d_pbb, n_packs, d_pout);
    HIP_TRY(hipMemcpyAsync(out_lens, d_pout, (size_t)n_packs * sizeof(u64),
                           hipMemcpyDeviceToHost, c->stream));
    u64 dbytes = 0;
    HIP_TRY(hipMemcpyAsync(&dbytes, delta_offs + nb, sizeof(u64),
                           hipMemcpyDeviceToHost, c->stream));
    HIP_TRY(hipStreamSynchronize(c->stream));
    HIP_TRY(hipGetLastError());
    if (nb > 0 && (rc = pack_validate_end(c))) {
        for (int p = 0; p < n_packs; p++) out_lens[p] = 0;
        return rc;
    }

    if (nb > 0) {
        float ms = 0.f;
        HIP_TRY(hipEventElapsedTime(&ms, c->ev[0], c->ev[1]));
        c->kernel_ms += ms;
        c->n_launches += 1;
    }
    u64 out_elems = 0, m_elems = 0;
    for (int p = 0; p < n_packs; p++) {
        out_elems += out_lens[p];
        m_elems += tasks[p].m;
    }
    c->bytes_algo += dbytes + nb * 20 + 8 * (m_elems + out_elems);
    return UA_OK;
}

extern "C" int ua_decode_dev(ua_ctx *c, const ua_dpack *pk, uint64_t seek_uid, uint64_t *out,
                             uint64_t *out_n) {
    std::lock_guard<std::recursive_mutex> g(c->mu);
    return run_packed_locked(c, pk, seek_uid, nullptr, 0, out, out_n, 1);
}

/* ---- prepared pack fan-out (standing query plan) ---- */

struct ua_pbatch {
    int n_packs = 0;
    u64 nb = 0;
    u64 deltas_bytes = 0;
    u64 sum_m = 0;
    const u64 *bases = nullptr;
    const u32 *nums = nullptr;
    const u64 *doffs = nullptr;
    const u8 *deltas = nullptr;
    void *mem = nullptr; /* pbb + tasks + cnt + offs + partials + pout + staging */
    u64 *d_pbb = nullptr;
    ua_ptask *d_tasks = nullptr;
    u32 *d_cnt = nullptr;
    u64 *d_offs = nullptr;
    u64 *d_part = nullptr;
    u64 *d_pout = nullptr;
    u64 *d_stage = nullptr;
    u64 nchunks = 0;
};

extern "C" int ua_pbatch_create(ua_ctx *c, const uint64_t *bases, const uint32_t *num_uids,
                                const uint64_t *delta_offs, const uint8_t *deltas,
                                const uint64_t *pack_block_base, int n_packs,
                                const ua_ptask *tasks, ua_pbatch **out) {
    std::lock_guard<std::recursive_mutex> g(c->mu);
    HIP_TRY(hipSetDevice(c->device));
    if (n_packs <= 0) return UA_ERR_INVALID;
    ua_pbatch *b = new ua_pbatch();
    b->n_packs = n_packs;
    b->nb = pack_block_base[n_packs];
    b->bases = bases;
    b->nums = num_uids;
    b->doffs = delta_offs;
    b->deltas = deltas;
    for (int p = 0; p < n_packs; p++) b->sum_m += tasks[p].m;
    b->nchunks = (b->nb + 1 + UA_SCAN_CHUNK - 1) / UA_SCAN_CHUNK;

    size_t o_pbb = 0;
    size_t o_tasks = align16(o_pbb + ((size_t)n_packs + 1) * sizeof(u64));
    size_t o_cnt = align16(o_tasks + (size_t)n_packs * sizeof(ua_ptask));
    size_t o_offs = align16(o_cnt + (b->nb + 1) * sizeof(u32));
    size_t o_part = align16(o_offs + (b->nb + 1) * sizeof(u64));
    size_t o_pout = align16(o_part + (b->nchunks + 1) * sizeof(u64));
    size_t o_stage = align16(o_pout + (size_t)n_packs * sizeof(u64));
    size_t total = align16(o_stage + (b->nb ? b->nb : 1) * UA_MAX_BLOCK_UIDS * sizeof(u64));
    hipError_t e = hipMalloc(&b->mem, total);
    if (e != hipSuccess) {
        g_last_hip = e;
        delete b;
        return UA_ERR_NOMEM;
    }
    u8 *base8 = (u8 *)b->mem;
    b->d_pbb = (u64 *)(base8 + o_pbb);
    b->d_tasks = (ua_ptask *)(base8 + o_tasks);
    b->d_cnt = (u32 *)(base8 + o_cnt);
    b->d_offs = (u64 *)(base8 + o_offs);
    b->d_part = (u64 *)(base8 + o_part);
    b->d_pout = (u64 *)(base8 + o_pout);
    b->d_stage = (u64 *)(base8 + o_stage);

    std::vector<u8> hostbuf(o_cnt);
    memcpy(hostbuf.data(), pack_block_base, ((size_t)n_packs + 1) * sizeof(u64));
    memcpy(hostbuf.data() + o_tasks, tasks, (size_t)n_packs * sizeof(ua_ptask));
    HIP_TRY(hipMemcpyAsync(b->mem, hostbuf.data(), hostbuf.size(),
                           hipMemcpyHostToDevice, c->stream));
    HIP_TRY(hipMemsetAsync(b->d_cnt + b->nb, 0, sizeof(u32), c->stream));
    u64 db = 0;
    HIP_TRY(hipMemcpyAsync(&db, delta_offs + b->nb, sizeof(u64),
                           hipMemcpyDeviceToHost, c->stream));
    int vrc = (b->nb > 0) ? pack_validate_begin(c, num_uids, delta_offs, b->nb) : UA_OK;
    HIP_TRY(hipStreamSynchronize(c->stream));
    if (vrc == UA_OK && b->nb > 0) vrc = pack_validate_end(c);
    if (vrc != UA_OK) {
        ua_pbatch_destroy(c, b);
        return vrc;
    }
    b->deltas_bytes = db;
    *out = b;
    return UA_OK;
}

extern "C" void ua_pbatch_destroy(ua_ctx *c, ua_pbatch *b) {
    if (!b) return;
    (void)hipSetDevice(c->device);
    if (b->mem) (void)hipFree(b->mem);
    delete b;
}

extern "C" int ua_pbatch_run(ua_ctx *c, ua_pbatch *b, uint64_t *out_lens) {
    std::lock_guard<std::recursive_mutex> g(c->mu);
    HIP_TRY(hipSetDevice(c->device));
    u64 nb = b->nb;
    if (nb > 0) {
        u64 nwg = (nb + UA_PKW - 1) / UA_PKW;
        HIP_TRY(hipEventRecord(c->ev[0], c->stream));
        hipLaunchKernelGGL(k_packed<0>, dim3((u32)nwg), dim3(UA_BLOCK), 0, c->stream,
                           b->bases, b->nums, b->doffs, b->deltas, nb, (u64)0,
                           (const u64 *)nullptr, (u64)0, b->d_stage, b->d_cnt,
                           b->d_pbb, b->n_packs, b->d_tasks);
        HIP_TRY(hipEventRecord(c->ev[1], c->stream));
        hipLaunchKernelGGL(k_scan1, dim3((u32)b->nchunks), dim3(UA_BLOCK), 0, c->stream,
                           b->d_cnt, nb + 1, b->d_offs, b->d_part);
        hipLaunchKernelGGL(k_scan2, dim3(1), dim3(UA_BLOCK), 0, c->stream, b->d_part,
                           b->nchunks);
        hipLaunchKernelGGL(k_compact_pack, dim3((u32)((nb + 15) / 16)), dim3(UA_BLOCK),
                           0, c->stream, b->d_tasks, b->d_pbb, b->n_packs, b->d_cnt,
                           b->d_offs, b->d_part, b->d_stage, nb);
    }
    u64 poutblk = ((u64)b->n_packs + UA_BLOCK - 1) / UA_BLOCK;
    hipLaunchKernelGGL(k_pair_out, dim3((u32)poutblk), dim3(UA_BLOCK), 0, c->stream,
                       b->d_offs, b->d_part, b->d_pbb, b->n_packs, b->d_pout);
    HIP_TRY(hipMemcpyAsync(out_lens, b->d_pout, (size_t)b->n_packs * sizeof(u64),
                           hipMemcpyDeviceToHost, c->stream));
    HIP_TRY(hipStreamSynchronize(c->stream));
    HIP_TRY(hipGetLastError());
    if (nb > 0) {
        float ms = 0.f;
        HIP_TRY(hipEventElapsedTime(&ms, c->ev[0], c->ev[1]));
        c->kernel_ms += ms;
        c->n_launches += 1;
    }
    u64 out_elems = 0;
    for (int p = 0; p < b->n_packs; p++) out_elems += out_lens[p];
    c->bytes_algo += b->deltas_bytes + nb * 20 + 8 * (b->sum_m + out_elems);
    return UA_OK;
}

/* ---- GPU codec.Encode pipeline (codec.go:393 semantics, engine layout) ---- */
extern "C" int ua_encode_dev(ua_ctx *c, const uint64_t *uids, uint64_t n,
                             uint32_t block_size, uint64_t *bases, uint32_t *num_uids,
                             uint64_t *delta_offs, uint8_t *deltas,
                             uint64_t *n_blocks_out, uint64_t *deltas_bytes_out) {
    std::lock_guard<std::recursive_mutex> g(c->mu);
    HIP_TRY(hipSetDevice(c->device));
    if (block_size > UA_MAX_BLOCK_UIDS) return UA_ERR_INVALID;
    if (n == 0) {
        HIP_TRY(hipMemsetAsync(delta_offs, 0, sizeof(u64), c->stream));
        HIP_TRY(hipStreamSynchronize(c->stream));
        *n_blocks_out = 0;
        *deltas_bytes_out = 0;
        return UA_OK;
    }
    int rc;
    if ((rc = ws_reserve(c, WS_TCNT, (n + 1) * sizeof(u32)))) return rc;
    if ((rc = ws_reserve(c, WS_TOFF, (n + 1) * sizeof(u64)))) return rc;
    if ((rc = ws_reserve(c, WS_SCRATCH_A, (n + 1) * sizeof(u64)))) return rc;
    if ((rc = ws_reserve(c, WS_SCRATCH_B, (n + 1) * sizeof(u64)))) return rc;
    u32 *d_flags = (u32 *)c->ws[WS_TCNT];
    u64 *d_offs = (u64 *)c->ws[WS_TOFF];
    u64 *d_run = (u64 *)c->ws[WS_SCRATCH_A];
    u64 *d_bstart = (u64 *)c->ws[WS_SCRATCH_B];

    u64 nblk = (n + UA_BLOCK - 1) / UA_BLOCK;
    u64 nblk1 = (n + 1 + UA_BLOCK - 1) / UA_BLOCK;
    HIP_TRY(hipMemsetAsync(d_flags + n, 0, sizeof(u32), c->stream));
    hipLaunchKernelGGL(k_enc_msb_flags, dim3((u32)nblk), dim3(UA_BLOCK), 0, c->stream,
                       uids, n, d_flags);
    if ((rc = run_scan(c, d_flags, n + 1, d_offs))) return rc;
    u64 *d_part = (u64 *)c->ws[WS_PARTIAL];
    hipLaunchKernelGGL(k_enc_mark_start, dim3((u32)nblk1), dim3(UA_BLOCK), 0, c->stream,
                       d_flags, d_offs, d_part, n, d_run);
    hipLaunchKernelGGL(k_enc_block_flags, dim3((u32)nblk), dim3(UA_BLOCK), 0, c->stream,
                       d_flags, d_offs, d_part, d_run, n, block_size);
    if ((rc = run_scan(c, d_flags, n + 1, d_offs))) return rc;
    d_part = (u64 *)c->ws[WS_PARTIAL];
    hipLaunchKernelGGL(k_enc_mark_start, dim3((u32)nblk1), dim3(UA_BLOCK), 0, c->stream,
                       d_flags, d_offs, d_part, n, d_bstart);
    u64 off_n = 0, part_n = 0;
    HIP_TRY(hipMemcpyAsync(&off_n, d_offs + n, sizeof(u64), hipMemcpyDeviceToHost, c->stream));
    HIP_TRY(hipMemcpyAsync(&part_n, d_part + n / UA_SCAN_CHUNK, sizeof(u64),
                           hipMemcpyDeviceToHost, c->stream));
    HIP_TRY(hipStreamSynchronize(c->stream));
    u64 nb = off_n + part_n;

    if ((rc = ws_reserve(c, WS_STAGE, (nb ? nb : 1) * UA_ENC_STRIDE))) return rc;
    u8 *d_stage = (u8 *)c->ws[WS_STAGE];
    HIP_TRY(hipEventRecord(c->ev[0], c->stream));
    hipLaunchKernelGGL(k_encode_blocks, dim3((u32)((nb + UA_PKW - 1) / UA_PKW)),
                       dim3(UA_BLOCK), 0, c->stream, uids, d_bstart, nb, bases, num_uids,
                       d_flags, d_stage);
    HIP_TRY(hipEventRecord(c->ev[1], c->stream));
    HIP_TRY(hipMemsetAsync(d_flags + nb, 0, sizeof(u32), c->stream));
    if ((rc = run_scan(c, d_flags, nb + 1, d_offs))) return rc;
    d_part = (u64 *)c->ws[WS_PARTIAL];
    hipLaunchKernelGGL(k_enc_finalize, dim3((u32)((nb + 1 + 3) / 4)), dim3(UA_BLOCK), 0,
                       c->stream, d_stage, d_flags, d_offs, d_part, nb, deltas, delta_offs);
    u64 db_off = 0, db_part = 0;
    HIP_TRY(hipMemcpyAsync(&db_off, d_offs + nb, sizeof(u64), hipMemcpyDeviceToHost,
                           c->stream));
    HIP_TRY(hipMemcpyAsync(&db_part, d_part + nb / UA_SCAN_CHUNK, sizeof(u64),
                           hipMemcpyDeviceToHost, c->stream));
    HIP_TRY(hipStreamSynchronize(c->stream));
    HIP_TRY(hipGetLastError());
    *n_blocks_out = nb;
    *deltas_bytes_out = db_off + db_part;

    float ms = 0.f;
    HIP_TRY(hipEventElapsedTime(&ms, c->ev[0], c->ev[1]));
    c->kernel_ms += ms;
    c->n_launches += 1;
    c->bytes_algo += 8 * n + (db_off + db_part) + nb * 20;
    return UA_OK;
}

/* ==================== host-side codec (codec.Encode restated) ==================== */

struct ua_owned_pack {
    std::vector<ua_block> blocks;
    std::vector<u8> blob; /* all deltas, contiguous */
    std::vector<size_t> blob_off;
    ua_pack view;
};

static size_t gv_encode4_cc(u8 *buf, const u32 v[4]) {
    u8 *p = buf + 1;
    u8 tag = 0;
    for (int i = 0; i < 4; i++) {
        u32 x = v[i];
        int len = 1 + (x > 0xffu) + (x > 0xffffu) + (x > 0xffffffu);
        tag = (u8)(tag | ((len - 1) << (2 * i)));
        for (int b = 0; b < len; b++) {
            *p++ = (u8)(x & 0xff);
            x >>= 8;
        }
    }
    buf[0] = tag;
    return (size_t)(p - buf);
}

/* codec.go:57 packBlock + :107 Add + :117 32-MSB/blockSize split rules */
extern "C" int ua_encode(const uint64_t *uids, uint64_t n, uint32_t block_size,
                         ua_owned_pack **out) {
    ua_owned_pack *pk = new ua_owned_pack();
    u64 i = 0;
    while (i < n) {
        /* collect one block: stop at block_size or 32-MSB change */
        u64 start = i;
        u64 base = uids[i];
        i++;
        while (i < n && (i - start) < (u64)(block_size ? block_size : 1) &&
               ((uids[i] ^ uids[i - 1]) >> 32) == 0) {
            i++;
        }
        /* blockSize==0 in Go packs after every Add -> 1-uid blocks */
        u64 cnt = i - start;
        ua_block blk;
        blk.base = base;
        blk.num_uids = (u32)cnt;
        size_t blob_start = pk->blob.size();
        u64 last = base;
        u64 off = start + 1, rem = cnt - 1;
        u8 gbuf[17];
        u32 tmp[4];
        for (;;) {
            for (int j = 0; j < 4; j++) {
                if ((u64)j >= rem) {
                    tmp[j] = 0;
                } else {
                    tmp[j] = (u32)(uids[off + (u64)j] - last);
                    last = uids[off + (u64)j];
                }
            }
            size_t sz = gv_encode4_cc(gbuf, tmp);
            pk->blob.insert(pk->blob.end(), gbuf, gbuf + sz);
            if (rem <= 4) break;
            off += 4;
            rem -= 4;
        }
        blk.deltas_len = (u32)(pk->blob.size() - blob_start);
        blk.deltas = nullptr; /* fixed up below (blob may reallocate) */
        pk->blob_off.push_back(blob_start);
        pk->blocks.push_back(blk);
    }
    for (size_t b = 0; b < pk->blocks.size(); b++)
        pk->blocks[b].deltas = pk->blob.data() + pk->blob_off[b];
    pk->view.block_size = block_size;
    pk->view.n_blocks = pk->blocks.size();
    pk->view.blocks = pk->blocks.data();
    *out = pk;
    return UA_OK;
}

extern "C" const ua_pack *ua_owned_pack_view(ua_owned_pack *pk) { return &pk->view; }
extern "C" void ua_owned_pack_free(ua_owned_pack *pk) { delete pk; }

extern "C" uint64_t ua_pack_exact_len(const ua_pack *p) {
    if (!p) return 0;
    u64 n = 0;
    for (u64 i = 0; i < p->n_blocks; i++) n += p->blocks[i].num_uids;
    return n;
}

extern "C" uint64_t ua_pack_approx_len(const ua_pack *p) {
    if (!p) return 0;
    return p->n_blocks * (u64)p->block_size;
}

extern "C" int ua_pack_flat_sizes(const ua_pack *p, uint64_t *n_blocks,
                                  uint64_t *deltas_bytes, uint64_t *total_uids) {
    if (!p) return UA_ERR_INVALID;
    u64 db = 0, tu = 0;
    for (u64 i = 0; i < p->n_blocks; i++) {
        db += p->blocks[i].deltas_len;
        tu += p->blocks[i].num_uids;
    }
    *n_blocks = p->n_blocks;
    *deltas_bytes = db;
    *total_uids = tu;
    return UA_OK;
}

extern "C" int ua_pack_flatten(const ua_pack *p, uint64_t *bases, uint32_t *num_uids,
                               uint64_t *delta_offs, uint8_t *deltas_blob) {
    if (!p) return UA_ERR_INVALID;
    u64 off = 0;
    for (u64 i = 0; i < p->n_blocks; i++) {
        const ua_block &b = p->blocks[i];
        if (b.num_uids > UA_MAX_BLOCK_UIDS || b.deltas_len > UA_MAX_DELTAS)
            return UA_ERR_INVALID;
        bases[i] = b.base;
        num_uids[i] = b.num_uids;
        delta_offs[i] = off;
        memcpy(deltas_blob + off, b.deltas, b.deltas_len);
        off += b.deltas_len;
    }
    delta_offs[p->n_blocks] = off;
    return UA_OK;
}

/* ==================== host-pointer convenience (mirrors algo signatures) ==================== */

extern "C" int64_t ua_index_of(const uint64_t *u, uint64_t n, uint64_t uid) {
    /* algo.IndexOf (uidlist.go:546): one binary search — host-side, like the
     * reference; batched GPU form is ua_index_of_batch_dev. */
    u64 lo = 0, hi = n;
    while (lo < hi) {
        u64 mid = (lo + hi) >> 1;
        if (u[mid] >= uid) hi = mid;
        else lo = mid + 1;
    }
    return (lo < n && u[lo] == uid) ? (int64_t)lo : -1;
}

static int host_pair_op(ua_ctx *c, const u64 *u, u64 n, const u64 *v, u64 m, u64 *out,
                        u64 *out_n, int op) {
    std::lock_guard<std::recursive_mutex> gop(c->mu); /* whole-op: WS_H* reuse */
    int rc;
    {
        std::lock_guard<std::recursive_mutex> g(c->mu);
        HIP_TRY(hipSetDevice(c->device));
        if ((rc = ws_reserve(c, WS_HU, (n ? n : 1) * sizeof(u64)))) return rc;
        if ((rc = ws_reserve(c, WS_HV, (m ? m : 1) * sizeof(u64)))) return rc;
        u64 cap = (op == OP_UNION) ? n + m : (op == OP_DIFF ? n : (n < m ? n : m));
        if ((rc = ws_reserve(c, WS_HOUT, (cap ? cap : 1) * sizeof(u64)))) return rc;
        HIP_TRY(hipMemcpyAsync(c->ws[WS_HU], u, n * sizeof(u64), hipMemcpyHostToDevice,
                               c->stream));
        HIP_TRY(hipMemcpyAsync(c->ws[WS_HV], v, m * sizeof(u64), hipMemcpyHostToDevice,
                               c->stream));
    }
    ua_dpair pr = {(const u64 *)c->ws[WS_HU], n, (const u64 *)c->ws[WS_HV], m,
                   (u64 *)c->ws[WS_HOUT]};
    u64 len = 0;
    if ((rc = run_batch(c, &pr, 1, &len, op))) return rc;
    {
        std::lock_guard<std::recursive_mutex> g(c->mu);
        HIP_TRY(hipMemcpyAsync(out, c->ws[WS_HOUT], len * sizeof(u64),
                               hipMemcpyDeviceToHost, c->stream));
        HIP_TRY(hipStreamSynchronize(c->stream));
    }
    *out_n = len;
    return UA_OK;
}

extern "C" int ua_intersect(ua_ctx *c, const uint64_t *u, uint64_t n, const uint64_t *v,
                            uint64_t m, uint64_t *out, uint64_t *out_n) {
    return host_pair_op(c, u, n, v, m, out, out_n, OP_INTERSECT);
}

extern "C" int ua_difference(ua_ctx *c, const uint64_t *u, uint64_t n, const uint64_t *v,
                             uint64_t m, uint64_t *out, uint64_t *out_n) {
    return host_pair_op(c, u, n, v, m, out, out_n, OP_DIFF);
}

static int host_upload_lists(ua_ctx *c, const uint64_t *const *lists, const uint64_t *lens,
                             int k, std::vector<const u64 *> &dptrs, u64 &total) {
    total = 0;
    for (int i = 0; i < k; i++) total += lens[i];
    int rc;
    if ((rc = ws_reserve(c, WS_HU, (total ? total : 1) * sizeof(u64)))) return rc;
    u64 *base = (u64 *)c->ws[WS_HU];
    u64 off = 0;
    dptrs.resize(k);
    for (int i = 0; i < k; i++) {
        HIP_TRY(hipMemcpyAsync(base + off, lists[i], lens[i] * sizeof(u64),
                               hipMemcpyHostToDevice, c->stream));
        dptrs[i] = base + off;
        off += lens[i];
    }
    HIP_TRY(hipStreamSynchronize(c->stream));
    return UA_OK;
}

extern "C" int ua_intersect_k(ua_ctx *c, const uint64_t *const *lists, const uint64_t *lens,
                              int k, uint64_t *out, uint64_t *out_n) {
    std::lock_guard<std::recursive_mutex> gop(c->mu);
    if (k <= 0) {
        *out_n = 0;
        return UA_OK;
    }
    std::vector<const u64 *> dptrs;
    u64 total;
    int rc;
    {
        std::lock_guard<std::recursive_mutex> g(c->mu);
        HIP_TRY(hipSetDevice(c->device));
        if ((rc = host_upload_lists(c, lists, lens, k, dptrs, total))) return rc;
        u64 cap = lens[0];
        for (int i = 1; i < k; i++) cap = std::min(cap, lens[i]);
        if ((rc = ws_reserve(c, WS_HOUT, (cap ? cap : 1) * sizeof(u64)))) return rc;
    }
    u64 len = 0;
    if ((rc = ua_intersect_k_dev(c, dptrs.data(), lens, k, (u64 *)c->ws[WS_HOUT], &len)))
        return rc;
    {
        std::lock_guard<std::recursive_mutex> g(c->mu);
        HIP_TRY(hipMemcpyAsync(out, c->ws[WS_HOUT], len * sizeof(u64),
                               hipMemcpyDeviceToHost, c->stream));
        HIP_TRY(hipStreamSynchronize(c->stream));
    }
    *out_n = len;
    return UA_OK;
}

extern "C" int ua_merge_k(ua_ctx *c, const uint64_t *const *lists, const uint64_t *lens,
                          int k, uint64_t *out, uint64_t *out_n) {
    std::lock_guard<std::recursive_mutex> gop(c->mu);
    if (k <= 0) {
        *out_n = 0;
        return UA_OK;
    }
    std::vector<const u64 *> dptrs;
    u64 total;
    int rc;
    {
        std::lock_guard<std::recursive_mutex> g(c->mu);
        HIP_TRY(hipSetDevice(c->device));
        if ((rc = host_upload_lists(c, lists, lens, k, dptrs, total))) return rc;
        if ((rc = ws_reserve(c, WS_HOUT, (total ? total : 1) * sizeof(u64)))) return rc;
    }
    u64 len = 0;
    if ((rc = ua_merge_k_dev(c, dptrs.data(), lens, k, (u64 *)c->ws[WS_HOUT], &len)))
        return rc;
    {
        std::lock_guard<std::recursive_mutex> g(c->mu);
        HIP_TRY(hipMemcpyAsync(out, c->ws[WS_HOUT], len * sizeof(u64),
                               hipMemcpyDeviceToHost, c->stream));
        HIP_TRY(hipStreamSynchronize(c->stream));
    }
    *out_n = len;
    return UA_OK;
}

extern "C" int ua_intersect_packed(ua_ctx *c, const ua_pack *pack, uint64_t after_uid,
                                   const uint64_t *v, uint64_t m, uint64_t *out,
                                   uint64_t *out_n) {
    std::lock_guard<std::recursive_mutex> gop(c->mu);
    if (!pack || pack->n_blocks == 0) {
        *out_n = 0;
        return UA_OK;
    }
    u64 nb, db, tu;
    int rc = ua_pack_flat_sizes(pack, &nb, &db, &tu);
    if (rc) return rc;
    std::vector<u64> bases(nb), doffs(nb + 1);
    std::vector<u32> nums(nb);
    std::vector<u8> blob(db ? db : 1);
    if ((rc = ua_pack_flatten(pack, bases.data(), nums.data(), doffs.data(), blob.data())))
        return rc;
    ua_dpack dpk;
    {
        std::lock_guard<std::recursive_mutex> g(c->mu);
        HIP_TRY(hipSetDevice(c->device));
        size_t need = nb * 8 + (nb + 1) * 8 + nb * 4 + blob.size() + 64;
        if ((rc = ws_reserve(c, WS_PACK, need))) return rc;
        u8 *w = (u8 *)c->ws[WS_PACK];
        u64 *d_bases = (u64 *)w;
        u64 *d_doffs = (u64 *)(w + nb * 8);
        u32 *d_nums = (u32 *)(w + nb * 8 + (nb + 1) * 8);
        u8 *d_blob = w + nb * 8 + (nb + 1) * 8 + nb * 4;
        HIP_TRY(hipMemcpyAsync(d_bases, bases.data(), nb * 8, hipMemcpyHostToDevice, c->stream));
        HIP_TRY(hipMemcpyAsync(d_doffs, doffs.data(), (nb + 1) * 8, hipMemcpyHostToDevice,
                               c->stream));
        HIP_TRY(hipMemcpyAsync(d_nums, nums.data(), nb * 4, hipMemcpyHostToDevice, c->stream));
        HIP_TRY(hipMemcpyAsync(d_blob, blob.data(), blob.size(), hipMemcpyHostToDevice,
                               c->stream));
        if ((rc = ws_reserve(c, WS_HV, (m ? m : 1) * sizeof(u64)))) return rc;
        HIP_TRY(hipMemcpyAsync(c->ws[WS_HV], v, m * sizeof(u64), hipMemcpyHostToDevice,
                               c->stream));
        u64 cap = std::min(tu, m);
        if ((rc = ws_reserve(c, WS_HOUT, (cap ? cap : 1) * sizeof(u64)))) return rc;
        HIP_TRY(hipStreamSynchronize(c->stream));
        dpk = {pack->block_size, nb, d_bases, d_nums, d_doffs, d_blob, tu};
    }
    u64 len = 0;
    if ((rc = ua_intersect_packed_dev(c, &dpk, after_uid, (const u64 *)c->ws[WS_HV], m,
                                      (u64 *)c->ws[WS_HOUT], &len)))
        return rc;
    {
        std::lock_guard<std::recursive_mutex> g(c->mu);
        HIP_TRY(hipMemcpyAsync(out, c->ws[WS_HOUT], len * sizeof(u64),
                               hipMemcpyDeviceToHost, c->stream));
        HIP_TRY(hipStreamSynchronize(c->stream));
    }
    *out_n = len;
    return UA_OK;
}
