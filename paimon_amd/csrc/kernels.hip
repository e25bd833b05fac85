// CDNA4 (gfx950) kernels for Paimon's merge-on-read hot path.
//
// MI355X-first design (NOT a translation of the reference's Java loops):
// the reference's LoserTree (mergetree/compact/LoserTree.java) stalls per
// key on a pointer-chasing tournament; here the same total order
// (ascending userKey, sequenceNumber, isAdd —
// SortMergeReaderWithLoserTree.java:48-75) is produced by a merge-path
// partition over k sorted runs into independent tiles, an LDS-staged
// pairwise-stable merge inside each tile, and a segmented winner reduction
// per equal-key group (DeduplicateMergeFunction.java:48-62 +
// ReducerMergeFunctionWrapper.java:53-73 + DropDeleteReader.java:53-61).
// This path is HBM-bandwidth-bound integer work: no MFMA, wide coalesced
// loads, LDS staging, grid-stride launches.
//
// Capacity limits (checked host-side): runs per section <= PMH_MAX_RUNS,
// rows per run < 2^PMH_ROW_BITS (packed winner format run:5 | row:27).

#include <hip/hip_runtime.h>

#include <cstdint>

#include "kernels.h"
#include "zstd_core.h"

#define DEV __device__ __forceinline__

namespace pmh {

// ---------------------------------------------------------------- paged cols

template <typename T>
DEV T col_load(const DevCol &c, int64_t row) {
    if (c.n_pages == 1)  // staged columns are contiguous: direct addressing
        return *reinterpret_cast<const T *>(c.addr0 + (uint64_t)row * sizeof(T));
    // general paged path: last page with start_row <= row
    int lo = 0, hi = c.n_pages - 1;
    while (lo < hi) {
        int mid = (lo + hi + 1) >> 1;
        if (c.pages[mid].start_row <= row) lo = mid;
        else hi = mid - 1;
    }
    const DevPage &pg = c.pages[lo];
    return *reinterpret_cast<const T *>(pg.addr + (uint64_t)(row - pg.start_row) * sizeof(T));
}

DEV uint64_t ukey(int64_t k) { return (uint64_t)k ^ 0x8000000000000000ull; }

// key column element (stored width 4 or 8 bytes — TINYINT..INT stage as
// INT32; the key TYPE is uniform across runs, so es is a scalar)
DEV int64_t key_at(uint64_t addr, int64_t idx, int es) {
    return es == 8 ? *reinterpret_cast<const int64_t *>(addr + idx * 8)
                   : (int64_t)*reinterpret_cast<const int32_t *>(addr +
                                                                 idx * 4);
}

// Lockstep multi-run lower/upper bound: runs the k binary searches together
// so their probe loads issue back-to-back each step (8-16x memory-level
// parallelism vs sequential searches — the partition kernel was 83%
// latency-parked on serial probes, profiles/r01_c2_pmc.md). Arrays are
// indexed only by the unrolled compile-time r, so they stay in registers.
// le=true: first index with ukey > v; le=false: first index with ukey >= v.
template <bool LE, int KM>
DEV void bound_multi(const uint64_t *addr, const int64_t *lo_in,
                     const int64_t *hi_in, int k, int kes, uint64_t v,
                     int64_t *out) {
    int64_t lo[KM], hi[KM];
#pragma unroll
    for (int r = 0; r < KM; r++) {
        if (r >= k) continue;
        lo[r] = lo_in[r];
        hi[r] = hi_in[r];
    }
    bool any = true;
    while (any) {
        any = false;
#pragma unroll
        for (int r = 0; r < KM; r++) {
            if (r >= k) continue;
            if (lo[r] < hi[r]) {
                int64_t mid = lo[r] + ((hi[r] - lo[r]) >> 1);
                uint64_t kk = ukey(key_at(addr[r], mid, kes));
                bool go = LE ? (kk <= v) : (kk < v);
                if (go) lo[r] = mid + 1;
                else hi[r] = mid;
                any |= lo[r] < hi[r];
            }
        }
    }
#pragma unroll
    for (int r = 0; r < KM; r++) {
        if (r >= k) continue;
        out[r] = lo[r];
    }
}

// --------------------------------------------------------- k_partition_wave
//
// Wave-parallel 64-ary key-domain search: ONE WAVE per tile boundary; per
// round the 64 lanes probe 64 evenly spaced pivots of the remaining key
// domain (each lane runs the lockstep multi-run count for its pivot), a
// ballot picks the bracketing pair, and domain + per-run windows narrow
// ~65x. Exact counts, so the cuts are IDENTICAL to k_partition's — but the
// dependent-load chain is ~4 rounds x log2(window) instead of ~36 x
// log2(window): the partition wall-clock is one latency chain, and this
// cuts it ~5x. Validated against a brute-force model on randomized runs
// (tests/test_tile_algorithm.py + the CPU prototype in git history).
template <int KM>
__global__ void k_partition_wave(const DevCol *keys, const int64_t *lens,
                                 int k, int64_t tile_rows, int64_t n_bounds,
                                 int64_t total_rows, int64_t stride,
                                 int32_t *cuts) {
    // stride > 1: coarse pass of the two-level partition — bounds
    // {0, stride, 2*stride, ...} u {n_bounds-1}; interior bounds follow in
    // k_partition_refine with windows clamped by the enclosing coarse cuts
    // (their probes then stay inside small, L2-resident windows: the wave
    // kernel's 64 probes/round cost ~12x the old kernel's probe TRAFFIC
    // when used for every bound — measured 2.4 ms vs 1.2 — so it only runs
    // where its short chain matters: the full-depth coarse cuts)
    const int lane = (int)(threadIdx.x & 63);
    int64_t b =
        ((int64_t)blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6)) *
        stride;
    if (b > n_bounds - 1) b = n_bounds - 1;
    int64_t D = b * tile_rows;
    if (D > total_rows) D = total_rows;
    uint64_t addr[KM];
    int64_t wlo[KM], whi[KM], pos[KM];
    const int kes = keys[0].esize;
#pragma unroll
    for (int r = 0; r < KM; r++) {
        if (r >= k) continue;
        addr[r] = keys[r].addr0;
        wlo[r] = 0;
        whi[r] = lens[r];
    }
    if (D == 0 || D >= total_rows) {
        if (lane == 0) {
#pragma unroll
            for (int r = 0; r < KM; r++) {
                if (r >= k) continue;
                cuts[b * k + r] = D == 0 ? 0 : (int32_t)lens[r];
            }
        }
        return;
    }
    uint64_t klo = ~0ull, khi = 0;
#pragma unroll
    for (int r = 0; r < KM; r++) {
        if (r >= k || whi[r] == 0) continue;
        uint64_t lo_k = ukey(key_at(addr[r], 0, kes));
        uint64_t hi_k = ukey(key_at(addr[r], whi[r] - 1, kes));
        if (lo_k < klo) klo = lo_k;
        if (hi_k > khi) khi = hi_k;
    }
    while (klo < khi) {
        const uint64_t span = khi - klo;
        uint64_t pv;
        if (span <= 63) {
            pv = klo + (uint64_t)lane;
            if (pv > khi) pv = khi;
        } else {
            // overflow-safe klo + span*(lane+1)/65, monotone in lane
            const uint64_t q = span / 65, rm = span % 65;
            pv = klo + q * (uint64_t)(lane + 1) +
                 (rm * (uint64_t)(lane + 1)) / 65;
        }
        bound_multi<true, KM>(addr, wlo, whi, k, kes, pv, pos);
        int64_t cnt = 0;
#pragma unroll
        for (int r = 0; r < KM; r++) {
            if (r >= k) continue;
            cnt += pos[r];
        }
        const uint64_t ge = __ballot(cnt >= D);
        const int f = ge ? (int)(__ffsll((unsigned long long)ge) - 1) : 64;
        const int fc = f < 64 ? f : 63;   // lane holding the new upper
        const int fm = f > 0 ? f - 1 : 0; // lane holding the new lower
        const uint64_t pv_f = __shfl(pv, fc, 64);
        const uint64_t pv_m = __shfl(pv, fm, 64);
        if (f == 64) klo = pv_f + 1;        // even the last pivot counts < D
        else if (f == 0) khi = pv_f;
        else { klo = pv_m + 1; khi = pv_f; }
#pragma unroll
        for (int r = 0; r < KM; r++) {
            if (r >= k) continue;
            const int64_t pf = __shfl(pos[r], fc, 64);
            const int64_t pm = __shfl(pos[r], fm, 64);
            if (f == 64) wlo[r] = pf;
            else if (f == 0) whi[r] = pf;
            else { wlo[r] = pm; whi[r] = pf; }
        }
    }
    // klo == v*: identical tie-take finish to k_partition (redundant across
    // lanes; lane 0 stores)
    bound_multi<false, KM>(addr, wlo, whi, k, kes, klo, pos);
    int64_t base = 0;
#pragma unroll
    for (int r = 0; r < KM; r++) {
        if (r >= k) continue;
        base += pos[r];
    }
    int64_t t = D - base;
    if (lane == 0) {
#pragma unroll
        for (int r = 0; r < KM; r++) {
            if (r >= k) continue;
            int64_t c = pos[r];
            bool has =
                (c < lens[r]) && (ukey(key_at(addr[r], c, kes)) == klo);
            if (t > 0 && has) { c++; t--; }
            cuts[b * k + r] = (int32_t)c;
        }
    }
}

// ------------------------------------------------------- k_partition_refine
//
// Level 2 of the two-level partition: each interior bound bisects inside
// the windows of its enclosing coarse cuts. Correct because a coarse cut
// is two-sided: every element below it keys <= its pivot and the pivot
// keys of bounds >= tile_rows ranks apart differ strictly (an equal-key
// group has <= k << tile_rows members, one per run), so no element outside
// the window can tie with any probed pivot in a way that changes a count
// comparison.
template <int KM>
__global__ void k_partition_refine(const DevCol *keys, const int64_t *lens,
                                   int k, int64_t tile_rows,
                                   int64_t n_bounds, int64_t total_rows,
                                   int64_t G, int32_t *cuts) {
    int64_t b = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (b >= n_bounds - 1) return;
    if (b % G == 0) return;  // coarse pass computed it
    int64_t D = b * tile_rows;  // b interior => 0 < D < total_rows
    int64_t g0 = (b / G) * G;
    int64_t g1 = g0 + G;
    if (g1 > n_bounds - 1) g1 = n_bounds - 1;
    uint64_t addr[KM];
    int64_t wlo[KM], whi[KM], pos[KM];
    const int kes = keys[0].esize;
    uint64_t klo = ~0ull, khi = 0;
#pragma unroll
    for (int r = 0; r < KM; r++) {
        if (r >= k) continue;
        addr[r] = keys[r].addr0;
        wlo[r] = cuts[g0 * k + r];
        whi[r] = cuts[g1 * k + r];
        if (wlo[r] < whi[r]) {
            uint64_t a = ukey(key_at(addr[r], wlo[r], kes));
            uint64_t z = ukey(key_at(addr[r], whi[r] - 1, kes));
            if (a < klo) klo = a;
            if (z > khi) khi = z;
        }
    }
    if (khi < klo) {  // every window empty: the coarse cuts already sum to D
#pragma unroll
        for (int r = 0; r < KM; r++) {
            if (r >= k) continue;
            cuts[b * k + r] = (int32_t)wlo[r];
        }
        return;
    }
    while (klo < khi) {
        uint64_t mid = klo + ((khi - klo) >> 1);
        bound_multi<true, KM>(addr, wlo, whi, k, kes, mid, pos);
        int64_t cnt = 0;
#pragma unroll
        for (int r = 0; r < KM; r++) {
            if (r >= k) continue;
            cnt += pos[r];
        }
        if (cnt >= D) {
            khi = mid;
#pragma unroll
            for (int r = 0; r < KM; r++) {
                if (r >= k) continue;
                whi[r] = pos[r];
            }
        } else {
            klo = mid + 1;
#pragma unroll
            for (int r = 0; r < KM; r++) {
                if (r >= k) continue;
                wlo[r] = pos[r];
            }
        }
    }
    bound_multi<false, KM>(addr, wlo, whi, k, kes, klo, pos);
    int64_t base = 0;
#pragma unroll
    for (int r = 0; r < KM; r++) {
        if (r >= k) continue;
        base += pos[r];
    }
    int64_t t = D - base;
#pragma unroll
    for (int r = 0; r < KM; r++) {
        if (r >= k) continue;
        int64_t c = pos[r];
        bool has = (c < lens[r]) && (ukey(key_at(addr[r], c, kes)) == klo);
        if (t > 0 && has) { c++; t--; }
        cuts[b * k + r] = (int32_t)c;
    }
}


// ------------------------------------------------------------ k_merge_tiles
//
// Phase 1: per tile, stage (key, seq, kind) segments in LDS, stable
// pairwise-merge into (key, run) order, mark equal-key group heads, run a
// segmented (seq, isAdd) argmax per group (groups have <= k records: one
// per run — SortedRun invariant, SortedRun.java), apply Deduplicate +
// wrapper + drop-delete rules, and emit packed winners per owned group.

struct TileSmem {
    int64_t skey[PMH_TILE_MAX];
    int64_t sseq[PMH_TILE_MAX];  // packed (sequenceNumber << 1) | isAdd:
                                 // ascending == ascending (seq, isAdd) —
                                 // the merge-order tie-break of
                                 // SortMergeReaderWithLoserTree.java:53-63.
                                 // Requires |seq| < 2^62 (Paimon sequence
                                 // numbers are non-negative counters).
    uint16_t perm[2][PMH_TILE_MAX];
    uint8_t head[PMH_TILE_MAX];    // group head flags (merged order)
    int32_t wave_tot[2 * (PMH_TILE_THREADS / 64)];
    int32_t segoff[PMH_MAX_RUNS + 1];
    int32_t seglen[PMH_MAX_RUNS];  // extended (incl. extra) lengths
    int32_t paircnt[PMH_MAX_RUNS + 1];
    int64_t predkey;
    int32_t haspred;
    int32_t mtotal;    // extended element count
    int32_t mreal;     // real element count (rank width of this tile)
    int32_t nemit;
};

// stable 2-way co-rank: number of A elements among the first d outputs of
// merge(A, B), ties take A first. A/B are perm-index sequences; key lookup
// through skey[.]. For i in [max(0,d-lb), min(d,la)), j = d-1-i is always
// in [0, lb). Advance while A[i] <= B[j] (A[i] belongs in the first d).
template <typename KT>
DEV int32_t corank(int64_t d, const uint16_t *pa, int32_t la,
                   const uint16_t *pb, int32_t lb, const KT *skey) {
    int64_t ilo = d > lb ? d - lb : 0;
    int64_t ihi = d < la ? d : la;
    while (ilo < ihi) {
        int64_t i = ilo + ((ihi - ilo) >> 1);
        int64_t j = d - 1 - i;
        if (skey[pa[i]] <= skey[pb[j]]) ilo = i + 1;
        else ihi = i;
    }
    return (int32_t)ilo;
}

DEV bool kind_is_add(uint8_t k) { return k == 0 || k == 2; }

// user-defined sequence fields (CoreOptions sequence.field;
// utils/UserDefinedSeqComparator.java:38-80, wired at
// MergeFileSplitRead.java:543-545): listed value columns compare BEFORE
// the sequence number, ascending, nulls FIRST (codegen nullIsLast=false).
DEV int useq_cmp(const DevCol *cols, const uint8_t *col_dtype, int n_cols,
                 const int16_t *ucols, int nu, int runA, int64_t rowA,
                 int runB, int64_t rowB) {
    for (int j = 0; j < nu; j++) {
        const int c = ucols[j];
        const DevCol &a = cols[runA * n_cols + c];
        const DevCol &b = cols[runB * n_cols + c];
        const uint8_t va =
            a.valid0 ? ((const uint8_t *)a.valid0)[rowA] : 1;
        const uint8_t vb =
            b.valid0 ? ((const uint8_t *)b.valid0)[rowB] : 1;
        if (va != vb) return va ? 1 : -1;  // null sorts first
        if (!va) continue;
        const int dt = col_dtype[c];
        const bool wide = dt == 4 || dt == 6;
        const int64_t x = wide ? col_load<int64_t>(a, rowA)
                               : (int64_t)col_load<int32_t>(a, rowA);
        const int64_t y = wide ? col_load<int64_t>(b, rowB)
                               : (int64_t)col_load<int32_t>(b, rowB);
        if (x != y) return x > y ? 1 : -1;
    }
    return 0;
}
// k_merge_tiles packs (seq << 1) | isAdd (the fused kernel uses the 2-bit
// ps2_* packing); helper so call sites read uniformly
DEV bool ps2m_isadd(int64_t w) { return w & 1; }

// PU=true: PartialUpdate mode — no winner reduction; emit every owned
// group's member list (ascending (seq, isAdd) order within the group) for
// the per-field overlay in k_emit_pu. v1 accepts INSERT-only streams (the
// reference's default partial-update rejects retracts,
// PartialUpdateMergeFunction.java:170-186) — retracts set err_flag.
// Separate template instantiations keep the PU path's extra registers out
// of the deduplicate kernel.
template <bool PU, bool FR>
__launch_bounds__(PMH_TILE_THREADS) __global__
void k_merge_tiles(const DevCol *keys, const DevCol *seqs, const DevCol *kinds,
                   const int64_t *lens, int k, const int32_t *cuts,
                   int64_t n_tiles, int64_t tile_rows, int flags,
                   const uint64_t *tombs, uint32_t *winners,
                   int32_t *tile_counts, uint16_t *group_start,
                   uint32_t *err_flag,
                   // full-compaction changelog (cl_entries != nullptr):
                   // per-run levels, the table's max level, and the
                   // per-group provisional changelog entries (2 slots per
                   // group; k_cl_finalize compacts them to rows)
                   const uint8_t *run_levels, int max_level,
                   uint64_t *cl_entries, int32_t *cl_counts) {
    const bool drop_delete = flags & 1;
    const bool ignore_delete = flags & 2;
    // FR = first-row engine (FirstRowMergeFunction.java:32-77): keep the
    // FIRST record per key in ascending (seq, isAdd) order; retracts throw
    // unless ignore-delete (then skipped); singleton groups bypass the merge
    // function (wrapper) exactly as deduplicate does. A template parameter so
    // the deduplicate winner walk carries no extra select.
    constexpr bool first_row = FR;
    // partial-update.remove-record-on-delete (PU mode): DELETE members are
    // legal; a group whose LAST member (max packed sseq) is a DELETE yields
    // a DELETE result and is dropped here under drop_delete. UPDATE_BEFORE
    // is rejected at staging (v1: INSERT/DELETE streams).
    const bool rrod = (flags & 16) != 0;
    // sequence groups (PartialUpdateMergeFunction.java:219-377): retracts
    // are legal members (they retract their groups); result kind is DELETE
    // only when the group has NO add member (getResult :389-397)
    const bool seqg = (flags & 32) != 0;
    // aggregation with retract-capable aggregators (FieldSumAgg.retract /
    // FieldPrimaryKeyAgg / ignore-retract wrappers): retract members are
    // legal and fold through aggregator.retract in k_emit_agg
    const bool aggr = (flags & 64) != 0;
    // ablation levels (profiling only, flags bits 8..): 1=stage,2=+merge,
    // 3=+scan, 0/absent=full. Partial levels publish a checksum so the
    // compiler cannot dead-code the ablated phases' inputs.
    const int ablate = (flags >> 8) & 0xf;
    __shared__ TileSmem sm;
    for (int64_t tile = blockIdx.x; tile < n_tiles; tile += gridDim.x) {
        const int tid = threadIdx.x;
        const int32_t *c0 = &cuts[tile * k];
        const int32_t *c1 = &cuts[(tile + 1) * k];

        // --- segment setup (extended: +1 element per run when available).
        // Parallel across the first k threads — a serial tid-0 loop of ~30
        // dependent global loads stalled the whole tile (ablation,
        // profiles/r01_c2_pmc.md); only the tiny k-entry prefix stays serial.
        __shared__ int64_t s_predcand[PMH_MAX_RUNS];
        if (tid < k) {
            int32_t a = c0[tid], b = c1[tid];
            int32_t ext = (b < (int32_t)lens[tid]) ? 1 : 0;
            sm.seglen[tid] = (b - a) + ext;
            s_predcand[tid] =
                a > 0 ? key_at(keys[tid].addr0, a - 1, keys[tid].esize)
                      : INT64_MIN;
            // reuse head[] as a tiny flag channel for "has predecessor"
            sm.head[tid] = a > 0;
        }
        __syncthreads();
        if (tid == 0) {
            int32_t off = 0;
            int32_t real = 0;
            int64_t pred = 0;
            int hp = 0;
            for (int r = 0; r < k; r++) {
                sm.segoff[r] = off;
                int32_t ext_len = sm.seglen[r];
                int32_t real_len = c1[r] - c0[r];
                real += real_len;
                off += ext_len;
                if (sm.head[r]) {
                    if (!hp || s_predcand[r] > pred) pred = s_predcand[r];
                    hp = 1;
                }
            }
            sm.segoff[k] = off;
            sm.mtotal = off;
            sm.mreal = real;
            sm.predkey = pred;
            sm.haspred = hp;
            sm.nemit = 0;
        }
        __syncthreads();
        const int32_t M = sm.mtotal;
        const int32_t Mreal = sm.mreal;
        if (Mreal == 0) {
            if (tid == 0) {
                tile_counts[tile] = 0;
                if (cl_counts) cl_counts[tile] = 0;
            }
            __syncthreads();
            continue;
        }

        // --- stage key/seq/kind segments into LDS (coalesced per run).
        // Column base addresses hoisted out of the element loop: the staged
        // columns are contiguous, and re-reading the DevCol descriptor per
        // element serialized the loads (ablation: staging was 1.5 of 3.6 ms).
        for (int r = 0; r < k; r++) {
            int32_t off = sm.segoff[r], len = sm.seglen[r];
            int64_t base = c0[r];
            const int kes = keys[r].esize;
            const uint64_t kaddr0 = keys[r].addr0 + (uint64_t)base * kes;
            const int64_t *saddr =
                reinterpret_cast<const int64_t *>(seqs[r].addr0) + base;
            const int32_t *daddr =
                reinterpret_cast<const int32_t *>(kinds[r].addr0) + base;
            const uint8_t *tb =
                tombs && tombs[r]
                    ? reinterpret_cast<const uint8_t *>(tombs[r]) + base
                    : nullptr;
            for (int32_t i = tid; i < len; i += blockDim.x) {
                sm.skey[off + i] = key_at(kaddr0, i, kes);
                int32_t kd = daddr[i];
                const bool dead = tb && tb[i];
                if (PU && rrod && !seqg && kd == 1 && !dead && err_flag)
                    atomicOr(err_flag, 1u);  // UPDATE_BEFORE: not in v1 RROD
                                             // (sequence groups accept it)
                sm.sseq[off + i] =
                    dead ? PMH_DEAD
                         : (saddr[i] << 1) | (int64_t)(kd == 0 || kd == 2);
                sm.perm[0][off + i] = (uint16_t)(off + i);
            }
        }
        __syncthreads();
        if (ablate == 1) {
            // keep staged data live; counts stay in {0,1} so the downstream
            // scan/emit of an ablated (profiling-only) run never goes OOB
            if (tid == 0) {
                int64_t x = sm.skey[M - 1] ^ sm.sseq[M - 1];
                tile_counts[tile] = (int32_t)((x ^ (x >> 32)) & 1);
            }
            __syncthreads();
            continue;
        }

        // --- pairwise stable merge, ceil(log2(k)) levels
        int cur = 0;
        for (int width = 1; width < k; width <<= 1) {
            // sequences at this level: [segoff[q*width], segoff[min((q+1)*width,k)])
            // merge pairs (2q, 2q+1)
            const int nxt = cur ^ 1;
            // per-thread chunks of 8 outputs, grid-stride over all chunks
            const int CH = 8;
            int n_chunks = (M + CH - 1) / CH;
            for (int ch = tid; ch < n_chunks; ch += blockDim.x) {
                int64_t o = (int64_t)ch * CH;  // global output rank
                int remaining = (int)(M - o < CH ? M - o : CH);
                int p = 0;
                // a chunk may span several pairs (pairs can be tiny):
                // walk pairs until the chunk's outputs are all produced
                while (remaining > 0) {
                    // pair p covers output ranks [segoff[a0], segoff[b1])
                    while ((p + 1) * 2 * width < k &&
                           sm.segoff[(p + 1) * 2 * width] <= o)
                        p++;
                    int a0 = p * 2 * width;
                    int amid = a0 + width < k ? a0 + width : k;
                    int b1 = a0 + 2 * width < k ? a0 + 2 * width : k;
                    int32_t abase = sm.segoff[a0];
                    int32_t la = sm.segoff[amid] - abase;
                    int32_t lb = sm.segoff[b1] - sm.segoff[amid];
                    int64_t d = o - abase;  // rank within pair
                    int32_t lim = la + lb - (int32_t)d;
                    if (lim <= 0) break;  // past the last pair
                    int n_out = lim < remaining ? lim : remaining;
                    const uint16_t *pa = &sm.perm[cur][abase];
                    const uint16_t *pb = &sm.perm[cur][abase + la];
                    int32_t ai = corank(d, pa, la, pb, lb, sm.skey);
                    int32_t bi = (int32_t)d - ai;
                    uint16_t *out = &sm.perm[nxt][abase + d];
                    for (int x = 0; x < n_out; x++) {
                        bool takeA;
                        if (ai >= la) takeA = false;
                        else if (bi >= lb) takeA = true;
                        else takeA = !(sm.skey[pa[ai]] > sm.skey[pb[bi]]);
                        out[x] = takeA ? pa[ai++] : pb[bi++];
                    }
                    o += n_out;
                    remaining -= n_out;
                }
            }
            cur = nxt;
            __syncthreads();
        }
        const uint16_t *mo = sm.perm[cur];
        if (ablate == 2) {
            if (tid == 0)
                tile_counts[tile] = ((int32_t)mo[M - 1] ^ (int32_t)mo[0]) & 1;
            __syncthreads();
            continue;
        }

        // --- group heads + previous-tile continuation skip
        for (int32_t i = tid; i < M; i += blockDim.x) {
            int64_t kk = sm.skey[mo[i]];
            uint8_t h = (i == 0) ? 1 : (kk != sm.skey[mo[i - 1]]);
            // records continuing the previous tile's last group are not
            // heads here (that tile consumed them as its extras)
            if (sm.haspred && kk == sm.predkey) h = 0;
            sm.head[i] = h;
        }
        __syncthreads();

        // block-wide exclusive scan (wave shuffle scans + one barrier):
        // the former Hillis-Steele block scans cost 16 __syncthreads per
        // tile and dominated the kernel's wave-parked time.
        const int lane = tid & 63;
        const int wv = tid >> 6;
        constexpr int NW = PMH_TILE_THREADS / 64;

        if constexpr (PU) {
            // --- PartialUpdate: emit owned groups' member lists
            const int32_t per =
                (Mreal + (int32_t)blockDim.x - 1) / blockDim.x;
            int32_t my_lo = tid * per;
            int32_t my_hi = my_lo + per < Mreal ? my_lo + per : Mreal;
            uint32_t *mout = &winners[tile * (tile_rows + PMH_MAX_RUNS)];
            uint16_t *gout = &group_start[tile * (tile_rows + 1)];
            bool bad_kind = false;
            int32_t g_off = 0, m_off = 0, total_g = 0, total_m = 0;
            for (int pass = 0; pass < 2; pass++) {
                int32_t ng = 0, nm = 0;
                for (int32_t i = my_lo; i < my_hi; i++) {
                    if (!sm.head[i]) continue;
                    // pre-scan the group: LIVE members only (deletion-
                    // vector tombstones stage as PMH_DEAD and never reach
                    // the merge — ApplyDeletionVectorReader semantics)
                    int32_t tail = i;
                    int32_t nlive = 0;
                    int64_t mx = INT64_MIN;
                    bool any_add = false;
                    uint16_t live1 = 0;
                    for (int32_t x = i;; x++) {
                        int64_t v = sm.sseq[mo[x]];
                        if (v != PMH_DEAD) {
                            if (nlive == 0) live1 = mo[x];
                            nlive++;
                            if (v > mx) mx = v;
                            any_add |= ps2m_isadd(v);
                        }
                        if (!(x + 1 < M && !sm.head[x + 1])) {
                            tail = x;
                            break;
                        }
                    }
                    if (nlive == 0) continue;
                    if (nlive == 1) {
                        // ReducerMergeFunctionWrapper singleton bypass
                        // (ReducerMergeFunctionWrapper.java:53-73): the lone
                        // record is served as-is (any kind, incl. retracts);
                        // a retract result drops under drop-delete
                        // (DropDeleteReader.java:53-61)
                        if (drop_delete && !ps2m_isadd(sm.sseq[live1]))
                            continue;
                    } else if (seqg && drop_delete) {
                        // result kind DELETE iff no add member
                        if (!any_add) continue;
                    } else if (rrod && drop_delete) {
                        // result kind = last member's kind (max packed
                        // sseq); DELETE results drop here (DropDeleteReader)
                        if (!ps2m_isadd(mx)) continue;
                    }
                    if (pass == 1) {
                        gout[g_off + ng] = (uint16_t)(m_off + nm);
                        // merged order within a group is (key, run); the
                        // overlay consumes records in ascending (seq, isAdd)
                        // order — insertion-sort the <= k LIVE members by
                        // the packed sseq.
                        uint16_t gm[PMH_MAX_RUNS];
                        int gn = 0;
                        for (int32_t x2 = i; x2 <= tail; x2++) {
                            uint16_t s = mo[x2];
                            if (sm.sseq[s] == PMH_DEAD) continue;
                            int y = gn;
                            while (y > 0 &&
                                   sm.sseq[gm[y - 1]] > sm.sseq[s]) {
                                gm[y] = gm[y - 1];
                                y--;
                            }
                            gm[y] = s;
                            gn++;
                        }
                        for (int x = 0; x < gn; x++) {
                            uint16_t s = gm[x];
                            // retracts only fail MULTI-record groups: the
                            // wrapper bypasses the merge function for
                            // singletons (gn == 1), so a lone retract
                            // passes; sequence groups accept retracts
                            // outright (retractWithSequenceGroup)
                            if (!rrod && !seqg && !aggr && gn > 1 &&
                                !ps2m_isadd(sm.sseq[s]))
                                bad_kind = true;
                            int r = 0;
                            while (r + 1 <= k - 1 &&
                                   sm.segoff[r + 1] <= (int32_t)s)
                                r++;
                            uint32_t grow = (uint32_t)(
                                c0[r] + ((int32_t)s - sm.segoff[r]));
                            mout[m_off + nm + x] = ((uint32_t)r << PMH_ROW_BITS) | grow;
                        }
                    }
                    ng++;
                    nm += nlive;
                }
                if (pass == 0) {
                    // dual wave scan of (ng, nm)
                    int32_t ig = ng, im = nm;
                    for (int off = 1; off < 64; off <<= 1) {
                        int32_t ug = __shfl_up(ig, off, 64);
                        int32_t um = __shfl_up(im, off, 64);
                        if (lane >= off) {
                            ig += ug;
                            im += um;
                        }
                    }
                    if (lane == 63) {
                        sm.wave_tot[wv] = ig;
                        sm.wave_tot[NW + wv] = im;
                    }
                    __syncthreads();
                    int32_t addg = 0, addm = 0;
                    total_g = 0;
                    total_m = 0;
#pragma unroll
                    for (int w = 0; w < NW; w++) {
                        if (w < wv) {
                            addg += sm.wave_tot[w];
                            addm += sm.wave_tot[NW + w];
                        }
                        total_g += sm.wave_tot[w];
                        total_m += sm.wave_tot[NW + w];
                    }
                    g_off = addg + ig - ng;
                    m_off = addm + im - nm;
                }
            }
            if (bad_kind && err_flag) atomicOr(err_flag, 1u);
            if (tid == 0) {
                tile_counts[tile] = total_g;
                gout[total_g] = (uint16_t)total_m;  // sentinel
            }
            __syncthreads();
            continue;
        }

        if (ablate == 3) {
            if (tid == 0)
                tile_counts[tile] = ((int32_t)mo[0] ^ (int32_t)sm.head[0]) & 1;
            __syncthreads();
            continue;
        }
        // --- emit winners of owned groups, in key order.
        // A group is owned iff its head is a real (rank < Mreal) head; the
        // winner — DeduplicateMergeFunction's surviving record — is found
        // inline while walking the group's <= k members (max by packed
        // (eligible, seq, isAdd)). Two passes per thread over a contiguous
        // head range (count, then write at the scanned offset).
        const int32_t per = (Mreal + (int32_t)blockDim.x - 1) / blockDim.x;
        int32_t my_lo = tid * per;
        int32_t my_hi = my_lo + per < Mreal ? my_lo + per : Mreal;
        uint32_t *wout = &winners[tile * (tile_rows + PMH_MAX_RUNS)];
        // full-compaction changelog (FullChangelogMergeFunctionWrapper.java:
        // 96-126): per live group, up to two provisional entries, written
        // INDEPENDENTLY of drop-delete (a dropped DELETE result still emits
        // DELETE(topLevelKv)). flags bit 128 = row-deduplicate pending
        // (k_cl_finalize compares values and may drop the UB/UA pair).
        const bool cl = cl_entries != nullptr;
        const bool cl_pend = (flags & 128) != 0;
        uint64_t *clout =
            cl ? &cl_entries[tile * 2 * (tile_rows + PMH_MAX_RUNS)] : nullptr;
        // map seg index -> packed (run, global row)
        auto packm = [&](uint16_t sx) -> uint64_t {
            int r = 0;
            while (r + 1 <= k - 1 && sm.segoff[r + 1] <= (int32_t)sx) r++;
            return ((uint32_t)r << PMH_ROW_BITS) |
                   (uint32_t)(c0[r] + ((int32_t)sx - sm.segoff[r]));
        };
        int32_t total = 0;
        int32_t my_off = 0;
        int32_t my_cl = 0;
        for (int pass = 0; pass < 2; pass++) {
            int32_t nloc = 0, ncl = 0;
            for (int32_t i = my_lo; i < my_hi; i++) {
                if (!sm.head[i]) continue;
                // walk the group's LIVE members (deletion-vector tombstones
                // stage as PMH_DEAD), tracking the winner by (elig, sseq):
                // deduplicate keeps the LAST record, first-row the FIRST
                uint16_t s_best = 0;
                int64_t v_best = 0;
                bool e_best = false, any_retract = false;
                int32_t nlive = 0;
                uint16_t s_top = 0;
                int n_top = 0;
                for (int32_t x = i;; x++) {
                    int64_t v = sm.sseq[mo[x]];
                    if (v != PMH_DEAD) {
                        bool e = !ignore_delete || (v & 1);
                        any_retract |= !(v & 1);
                        bool take = nlive == 0 || (e && !e_best) ||
                                    (e == e_best &&
                                     (first_row ? v < v_best : v > v_best));
                        if (take) {
                            s_best = mo[x];
                            v_best = v;
                            e_best = e;
                        }
                        if (cl) {
                            int r = 0;
                            while (r + 1 <= k - 1 &&
                                   sm.segoff[r + 1] <= (int32_t)mo[x])
                                r++;
                            if (run_levels[r] == max_level) {
                                s_top = mo[x];
                                n_top++;
                            }
                        }
                        nlive++;
                    }
                    if (!(x + 1 < M && !sm.head[x + 1])) break;
                }
                if (nlive == 0) continue;
                if (cl) {
                    if (n_top > 1 && err_flag)
                        atomicOr(err_flag, 8u);  // checkState :76-78
                    // merged = wrapper result; singletons bypass the merge
                    // function (so an ignored lone record still serves)
                    bool m_ok = nlive == 1 || e_best;
                    bool m_add = m_ok && (v_best & 1);
                    uint64_t e0 = 0, e1 = 0;
                    int cl_n = 0;
                    auto mk = [&](uint16_t sx, uint64_t kd,
                                  bool pd) -> uint64_t {
                        return packm(sx) | (kd << 32) | (1ull << 35) |
                               (pd ? (1ull << 36) : 0);
                    };
                    if (n_top == 0) {
                        if (m_add) {  // INSERT(merged) — also the singleton
                                      // "initial is add" rule (:117-119)
                            e0 = mk(s_best, 0, false);
                            cl_n = 1;
                        }
                    } else if (nlive > 1) {
                        if (!m_add) {  // DELETE(topLevelKv) (:106-107)
                            e0 = mk(s_top, 3, false);
                            cl_n = 1;
                        } else {  // UPDATE_BEFORE(top) + UPDATE_AFTER(merged)
                            e0 = mk(s_top, 1, cl_pend);
                            e1 = mk(s_best, 2, cl_pend);
                            cl_n = 2;
                        }
                    }  // singleton that IS the top level: no change (:120-123)
                    if (cl_n) {
                        if (pass == 1) {
                            clout[2 * (my_cl + ncl)] = e0;
                            clout[2 * (my_cl + ncl) + 1] = e1;
                        }
                        ncl++;
                    }
                }
                if (first_row && !ignore_delete && any_retract && nlive > 1 &&
                    err_flag)
                    atomicOr(err_flag, 2u);  // FirstRow rejects retracts
                if (!e_best && nlive > 1) continue;  // all records ignored
                if (drop_delete && !(v_best & 1)) continue;
                if (pass == 1)
                    wout[my_off + nloc] = (uint32_t)packm(s_best);
                nloc++;
            }
            if (pass == 0) {
                int32_t incl = nloc, iccl = ncl;
                for (int off = 1; off < 64; off <<= 1) {
                    int32_t up = __shfl_up(incl, off, 64);
                    int32_t uc = __shfl_up(iccl, off, 64);
                    if (lane >= off) {
                        incl += up;
                        iccl += uc;
                    }
                }
                if (lane == 63) {
                    sm.wave_tot[wv] = incl;
                    sm.wave_tot[NW + wv] = iccl;
                }
                __syncthreads();
                int32_t add = 0, addc = 0, totc = 0;
                total = 0;
#pragma unroll
                for (int w = 0; w < NW; w++) {
                    if (w < wv) {
                        add += sm.wave_tot[w];
                        addc += sm.wave_tot[NW + w];
                    }
                    total += sm.wave_tot[w];
                    totc += sm.wave_tot[NW + w];
                }
                my_off = add + incl - nloc;
                my_cl = addc + iccl - ncl;
                if (cl && tid == 0) cl_counts[tile] = totc;
            }
        }
        if (tid == 0) tile_counts[tile] = total;
        __syncthreads();
    }
}

// ------------------------------------------------------------ k_merge_emit
//
// FUSED single-pass merge + emit for the deduplicate / first-row engines:
// the partition cuts in, the merged output columns out — one kernel.
// Replaces the k_merge_tiles -> k_scan_tiles -> k_emit chain (and its
// winners round-trip through HBM) for non-member-list engines:
//
//  - tiles are taken from a global TICKET (atomic counter), so tile t-1 is
//    always owned by a workgroup that started no later than tile t's — the
//    forward-progress precondition for a decoupled-lookback prefix without
//    any dispatch-order assumption (HIP promises none, MI355X_MICROARCH.md
//    §Workgroup dispatch);
//  - each tile stages key/seq/kind in LDS, merges, walks group winners
//    (exactly k_merge_tiles' order contract), then publishes its survivor
//    count through a PACKED agent-scope atomic word (flag:2 | value:62) and
//    resolves its global output offset by wave-parallel lookback — the
//    payload rides in the atomic itself, so no separate fence choreography
//    is needed (per-XCD L2s are not coherent; agent-scope rel/acq is);
//  - emission happens from LDS: key/seq/kind come from the already-staged
//    merge arrays (full RowKind is recoverable from the packed seq word),
//    and each remaining output column is staged COALESCED per run into a
//    double-buffered LDS slab that reuses the merge arrays' space, then
//    scattered to the dense output by winner index — LDS gathers + coalesced
//    stores instead of the 4-8 B HBM gathers that left k_emit 96% latency-
//    parked (profiles/r01_final_pmc_sq.md).
//
// Packed seq word: (sequenceNumber << 2) | (isAdd << 1) | (kind >> 1).
// Ascending order == ascending (seq, isAdd) — the merge-order tie-break of
// SortMergeReaderWithLoserTree.java:53-63 (the kind>>1 bit only breaks ties
// between records with equal (seq, isAdd), which valid buckets never have —
// sequence numbers are unique). RowKind = ((w & 1) << 1) | (1 - ((w >> 1) & 1)).
// Requires |seq| < 2^61 (Paimon sequence numbers are non-negative counters).

DEV int64_t ps2_pack(int64_t seq, int32_t kd) {
    return (seq << 2) | ((int64_t)(kd == 0 || kd == 2) << 1) |
           (int64_t)((kd >> 1) & 1);
}
DEV bool ps2_isadd(int64_t w) { return (w >> 1) & 1; }
DEV int32_t ps2_kind(int64_t w) {
    return (int32_t)(((w & 1) << 1) | (1 - ((w >> 1) & 1)));
}

struct FusedSmem {
    union {
        struct {
            int64_t skey[PMH_TILE_MAX];
            int64_t sseq[PMH_TILE_MAX];
        };
        // emission phase (skey/sseq dead after the key/seq/kind emit):
        // two column-staging slabs, PMH_TILE_MAX elements of up to 8 B
        uint8_t vbuf[2][PMH_TILE_MAX * 8];
    };
    uint16_t perm[2][PMH_TILE_MAX];
    uint8_t head[PMH_TILE_MAX];
    int32_t wave_tot[PMH_TILE_THREADS / 64];
    int32_t segoff[PMH_MAX_RUNS + 1];
    int32_t seglen[PMH_MAX_RUNS];
    int64_t predcand[PMH_MAX_RUNS];
    int64_t predkey;
    int64_t s_tile;
    int64_t s_goff;
    int32_t haspred;
    int32_t mtotal;
    int32_t mreal;
    // 32-bit tile-relative key mode: real merge tiles span ~3.6k sorted
    // keys, so (max-min) almost always fits 32 bits — staged keys repack
    // in place to uint32 offsets from tile_kmin (ukey space), halving the
    // LDS bank pressure of the merge's random key compares
    uint64_t wkmin[PMH_TILE_THREADS / 64];
    uint64_t wkmax[PMH_TILE_THREADS / 64];
    uint64_t tile_kmin;
    uint32_t predkey32;
    int32_t narrow;
};

constexpr uint64_t LOOK_AGG = 1ull << 62;
constexpr uint64_t LOOK_PREFIX = 2ull << 62;
constexpr uint64_t LOOK_VAL = (1ull << 62) - 1;

template <bool FR>
__attribute__((amdgpu_waves_per_eu(4))) __launch_bounds__(PMH_TILE_THREADS, 2)
__global__
void k_merge_emit(const DevCol *keys, const DevCol *seqs, const DevCol *kinds,
                  const int64_t *lens, int k, const int32_t *cuts,
                  int64_t tile_base, int64_t tile_limit, int64_t n_tiles,
                  int64_t tile_rows, int flags, const uint64_t *tombs,
                  const DevCol *cols /* k * n_cols, run-major */,
                  const uint8_t *col_dtype, const uint8_t *col_nullable,
                  int n_cols, int key_col /* -1: composite */, int seq_col,
                  int kind_col, const int16_t *useq_cols, int n_useq,
                  uint64_t *status, uint64_t *ticket,
                  int64_t *total_out, uint32_t *dense_winners,
                  void *const *out_ptrs,
                  uint8_t *const *out_valid, uint32_t *err_flag) {
    const bool drop_delete = flags & 1;
    const bool ignore_delete = flags & 2;
    __shared__ FusedSmem sm;
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wv = tid >> 6;
    constexpr int NW = PMH_TILE_THREADS / 64;
    for (;;) {
        if (tid == 0)
            sm.s_tile = tile_base + (int64_t)__hip_atomic_fetch_add(
                ticket, 1ull, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
        __syncthreads();
        const int64_t tile = sm.s_tile;
        if (tile >= tile_limit) return;
        const int32_t *c0 = &cuts[tile * k];
        const int32_t *c1 = &cuts[(tile + 1) * k];
        int32_t my_lo = 0, my_hi = 0, my_off = 0;
        int32_t C = 0;  // this tile's survivor count
        int cur = 0;
        uint16_t *wl = nullptr;
        // one group-walk body for both passes (count, then emit): walks the
        // thread's head range applying Deduplicate/FirstRow + wrapper +
        // drop-delete rules, calling emit(n-th-winner, seg-index)
        auto seg_rr = [&](uint16_t s, int *rr, int64_t *row) {
            int r = 0;
            while (r + 1 < k && sm.segoff[r + 1] <= (int32_t)s) r++;
            *rr = r;
            *row = c0[r] + ((int32_t)s - sm.segoff[r]);
        };
        auto walk_pass = [&](int32_t lo_i, int32_t hi_i, auto emit) {
            const uint16_t *mo = sm.perm[cur];
            const int32_t M = sm.mtotal;
            int32_t nloc = 0;
            for (int32_t i = lo_i; i < hi_i; i++) {
                if (!sm.head[i]) continue;
                // walk the group's members, skipping deletion-vector
                // tombstones (PMH_DEAD) — the reference's reader never
                // shows deleted rows to the merge, so LIVE members define
                // group size for the wrapper's singleton bypass too
                uint16_t s_best = 0;
                int64_t v_best = 0;
                bool e_best = false, any_retract = false;
                int32_t nlive = 0;
                for (int32_t x = i;; x++) {
                    int64_t v = sm.sseq[mo[x]];
                    if (v != PMH_DEAD) {
                        bool e = !ignore_delete || ps2_isadd(v);
                        any_retract |= !ps2_isadd(v);
                        bool take;
                        if (nlive == 0) {
                            take = true;
                        } else if (e != e_best) {
                            take = e;
                        } else if (n_useq > 0) {
                            // (seq fields..., seq, isAdd) ascending: the
                            // fields compare before the sequence number
                            int ra, rb;
                            int64_t qa, qb;
                            seg_rr(mo[x], &ra, &qa);
                            seg_rr(s_best, &rb, &qb);
                            const int cu =
                                useq_cmp(cols, col_dtype, n_cols, useq_cols,
                                         n_useq, ra, qa, rb, qb);
                            take = FR ? (cu < 0 || (cu == 0 && v < v_best))
                                      : (cu > 0 || (cu == 0 && v > v_best));
                        } else {
                            take = FR ? v < v_best : v > v_best;
                        }
                        if (take) {
                            s_best = mo[x];
                            v_best = v;
                            e_best = e;
                        }
                        nlive++;
                    }
                    if (!(x + 1 < M && !sm.head[x + 1])) break;
                }
                if (nlive == 0) continue;
                if (FR && !ignore_delete && any_retract && nlive > 1 &&
                    err_flag)
                    atomicOr(err_flag, 2u);
                if (!e_best && nlive > 1) continue;
                if (drop_delete && !ps2_isadd(v_best)) continue;
                emit(nloc, s_best);
                nloc++;
            }
            return nloc;
        };

        // --- segment setup (k_merge_tiles' protocol: +1 extra per run)
        if (tid < k) {
            int32_t a = c0[tid], b = c1[tid];
            int32_t ext = (b < (int32_t)lens[tid]) ? 1 : 0;
            sm.seglen[tid] = (b - a) + ext;
            sm.predcand[tid] =
                a > 0 ? key_at(keys[tid].addr0, a - 1, keys[tid].esize)
                      : INT64_MIN;
            sm.head[tid] = a > 0;
        }
        __syncthreads();
        if (tid == 0) {
            int32_t off = 0, real = 0;
            int64_t pred = 0;
            int hp = 0;
            for (int r = 0; r < k; r++) {
                sm.segoff[r] = off;
                real += c1[r] - c0[r];
                off += sm.seglen[r];
                if (sm.head[r]) {
                    if (!hp || sm.predcand[r] > pred) pred = sm.predcand[r];
                    hp = 1;
                }
            }
            sm.segoff[k] = off;
            sm.mtotal = off;
            sm.mreal = real;
            sm.predkey = pred;
            sm.haspred = hp;
        }
        __syncthreads();
        const int32_t M = sm.mtotal;
        const int32_t Mreal = sm.mreal;
        if (Mreal > 0) {
            // --- stage key / packed-seq segments (coalesced per run);
            // deletion-vector tombstones stage as PMH_DEAD (the walk skips
            // them — ApplyDeletionVectorReader semantics)
            uint64_t tkmin = ~0ull, tkmax = 0;
            for (int r = 0; r < k; r++) {
                int32_t off = sm.segoff[r], len = sm.seglen[r];
                int64_t base = c0[r];
                const int kes = keys[r].esize;
                const uint64_t kaddr0 = keys[r].addr0 + (uint64_t)base * kes;
                const int64_t *saddr =
                    reinterpret_cast<const int64_t *>(seqs[r].addr0) + base;
                const int32_t *daddr =
                    reinterpret_cast<const int32_t *>(kinds[r].addr0) + base;
                const uint8_t *tb =
                    tombs && tombs[r]
                        ? reinterpret_cast<const uint8_t *>(tombs[r]) + base
                        : nullptr;
                for (int32_t i = tid; i < len; i += blockDim.x) {
                    const int64_t kv = key_at(kaddr0, i, kes);
                    const uint64_t uk = ukey(kv);
                    if (uk < tkmin) tkmin = uk;
                    if (uk > tkmax) tkmax = uk;
                    sm.skey[off + i] = kv;
                    sm.sseq[off + i] = (tb && tb[i])
                                           ? PMH_DEAD
                                           : ps2_pack(saddr[i], daddr[i]);
                    sm.perm[0][off + i] = (uint16_t)(off + i);
                }
            }
            // block-reduce the tile key range, then repack the staged keys
            // to 32-bit tile-relative offsets when they fit
            for (int off2 = 32; off2; off2 >>= 1) {
                const uint64_t a = __shfl_down(tkmin, off2, 64);
                const uint64_t b = __shfl_down(tkmax, off2, 64);
                if (a < tkmin) tkmin = a;
                if (b > tkmax) tkmax = b;
            }
            if (lane == 0) {
                sm.wkmin[wv] = tkmin;
                sm.wkmax[wv] = tkmax;
            }
            __syncthreads();
            if (tid == 0) {
                uint64_t mn = ~0ull, mx = 0;
#pragma unroll
                for (int w = 0; w < NW; w++) {
                    if (sm.wkmin[w] < mn) mn = sm.wkmin[w];
                    if (sm.wkmax[w] > mx) mx = sm.wkmax[w];
                }
                sm.tile_kmin = mn;
                sm.narrow = (mx - mn) < 0xfffffffeull;
                const uint64_t up = ukey(sm.predkey);
                sm.predkey32 =
                    (sm.haspred && up >= mn && up - mn < 0xfffffffeull)
                        ? (uint32_t)(up - mn)
                        : 0xffffffffu;
            }
            __syncthreads();
            if (sm.narrow) {
                // in-place 64->32-bit repack in two barriered halves (the
                // destination bytes of each half never overlap the other
                // half's pending reads); 4 held values per thread per half
                uint32_t *k32 = reinterpret_cast<uint32_t *>(sm.skey);
#pragma unroll
                for (int half = 0; half < 2; half++) {
                    int64_t hold[4];
#pragma unroll
                    for (int h = 0; h < 4; h++) {
                        const int32_t i =
                            tid + (half * 4 + h) * (int32_t)blockDim.x;
                        hold[h] = i < M ? sm.skey[i] : 0;
                    }
                    __syncthreads();
#pragma unroll
                    for (int h = 0; h < 4; h++) {
                        const int32_t i =
                            tid + (half * 4 + h) * (int32_t)blockDim.x;
                        if (i < M)
                            k32[i] =
                                (uint32_t)(ukey(hold[h]) - sm.tile_kmin);
                    }
                    __syncthreads();
                }
            }

            // --- pairwise stable merge + group heads, generic over the
            // staged key width (uint32 tile-relative fast path; int64
            // fallback for tiles spanning >= 2^32 of key space)
            auto merge_heads = [&](auto *skeyT, auto predk, bool haspred2) {
                for (int width = 1; width < k; width <<= 1) {
                    const int nxt = cur ^ 1;
                    const int CH = 8;
                    int n_chunks = (M + CH - 1) / CH;
                    for (int ch = tid; ch < n_chunks; ch += blockDim.x) {
                        int64_t o = (int64_t)ch * CH;
                        int remaining = (int)(M - o < CH ? M - o : CH);
                        int p = 0;
                        while (remaining > 0) {
                            while ((p + 1) * 2 * width < k &&
                                   sm.segoff[(p + 1) * 2 * width] <= o)
                                p++;
                            int a0 = p * 2 * width;
                            int amid = a0 + width < k ? a0 + width : k;
                            int b1 = a0 + 2 * width < k ? a0 + 2 * width : k;
                            int32_t abase = sm.segoff[a0];
                            int32_t la = sm.segoff[amid] - abase;
                            int32_t lb = sm.segoff[b1] - sm.segoff[amid];
                            int64_t d = o - abase;
                            int32_t lim = la + lb - (int32_t)d;
                            if (lim <= 0) break;
                            int n_out = lim < remaining ? lim : remaining;
                            const uint16_t *pa = &sm.perm[cur][abase];
                            const uint16_t *pb = &sm.perm[cur][abase + la];
                            int32_t ai = corank(d, pa, la, pb, lb, skeyT);
                            int32_t bi = (int32_t)d - ai;
                            uint16_t *out = &sm.perm[nxt][abase + d];
                            for (int x = 0; x < n_out; x++) {
                                bool takeA;
                                if (ai >= la) takeA = false;
                                else if (bi >= lb) takeA = true;
                                else takeA =
                                    !(skeyT[pa[ai]] > skeyT[pb[bi]]);
                                out[x] = takeA ? pa[ai++] : pb[bi++];
                            }
                            o += n_out;
                            remaining -= n_out;
                        }
                    }
                    cur = nxt;
                    __syncthreads();
                }
                const uint16_t *mo2 = sm.perm[cur];
                for (int32_t i = tid; i < M; i += blockDim.x) {
                    auto kk = skeyT[mo2[i]];
                    uint8_t h = (i == 0) ? 1 : (kk != skeyT[mo2[i - 1]]);
                    if (haspred2 && kk == predk) h = 0;
                    sm.head[i] = h;
                }
                __syncthreads();
            };
            if (sm.narrow)
                merge_heads(reinterpret_cast<const uint32_t *>(sm.skey),
                            sm.predkey32, sm.haspred != 0);
            else
                merge_heads(sm.skey, sm.predkey, sm.haspred != 0);

            // --- winner walk (k_merge_tiles' dedup/first-row rules).
            // Pass 0 counts winners per thread; the global offset then
            // arrives via lookback, and pass 1 (below, after the barrier)
            // emits — in split mode STRAIGHT to the output arrays.
            wl = sm.perm[cur ^ 1];
            const int32_t per =
                (Mreal + (int32_t)blockDim.x - 1) / blockDim.x;
            my_lo = tid * per;
            my_hi = my_lo + per < Mreal ? my_lo + per : Mreal;
            int32_t nloc = walk_pass(my_lo, my_hi, [](int32_t, uint16_t) {});
            {
                int32_t incl = nloc;
                for (int off = 1; off < 64; off <<= 1) {
                    int32_t up = __shfl_up(incl, off, 64);
                    if (lane >= off) incl += up;
                }
                if (lane == 63) sm.wave_tot[wv] = incl;
                __syncthreads();
                int32_t add = 0;
                C = 0;
#pragma unroll
                for (int w = 0; w < NW; w++) {
                    if (w < wv) add += sm.wave_tot[w];
                    C += sm.wave_tot[w];
                }
                my_off = add + incl - nloc;
                // publish the aggregate NOW (the emit pass only writes):
                // successors' lookbacks unblock one pass earlier. The count
                // rides in the packed atomic word, so a successor that
                // acquires the flag also gets the payload — no extra fence.
                if (tid == 0 && tile > 0)
                    __hip_atomic_store(&status[tile],
                                       LOOK_AGG | (uint64_t)C,
                                       __ATOMIC_RELEASE,
                                       __HIP_MEMORY_SCOPE_AGENT);
            }
        }

        // --- decoupled lookback for the global offset (empty tiles still
        // publish their zero count here)
        if (tid == 0 && tile > 0 && Mreal == 0)
            __hip_atomic_store(&status[tile], LOOK_AGG | (uint64_t)C,
                               __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_AGENT);
        if (wv == 0) {
            int64_t excl = 0;
            if (tile > 0) {
                int64_t look = tile - 1;
                bool done = false;
                while (!done) {
                    int64_t idx = look - lane;
                    uint64_t st = LOOK_PREFIX;  // virtual prefix 0 before t0
                    if (idx >= 0) {
                        int64_t spins = 0;
                        do {
                            st = __hip_atomic_load(&status[idx],
                                                   __ATOMIC_ACQUIRE,
                                                   __HIP_MEMORY_SCOPE_AGENT);
                        } while ((st >> 62) == 0 && ++spins < (1ll << 27));
                        if ((st >> 62) == 0) {  // bounded spin: fail loudly
                            if (err_flag) atomicOr(err_flag, 4u);
                            st = LOOK_PREFIX;
                        }
                    }
                    bool isp = (st >> 62) == 2;
                    uint64_t pm = __ballot(isp);
                    int first = pm ? (int)(__ffsll((unsigned long long)pm) - 1)
                                   : 64;
                    int64_t contrib =
                        (lane <= first) ? (int64_t)(st & LOOK_VAL) : 0;
                    for (int off = 32; off; off >>= 1)
                        contrib += __shfl_down(contrib, off, 64);
                    excl += __shfl(contrib, 0, 64);
                    if (first < 64) done = true;
                    else look -= 64;
                }
            }
            if (lane == 0) {
                __hip_atomic_store(&status[tile],
                                   LOOK_PREFIX | (uint64_t)(excl + C),
                                   __ATOMIC_RELEASE,
                                   __HIP_MEMORY_SCOPE_AGENT);
                sm.s_goff = excl;
                if (tile == n_tiles - 1) *total_out = excl + C;
            }
        }
        __syncthreads();
        if (C == 0) continue;  // nothing to emit (all threads agree: C is
                               // uniform after the pass-0 block reduction)
        const int ablate = (flags >> 12) & 0x3;  // profiling-only: 1 = skip
                                                 // value loop, 2 = skip all
                                                 // emission
        if (ablate == 2) continue;
        const int64_t goff = sm.s_goff;
        // staged keys may be 32-bit tile-relative: reconstruct the original
        // key as (tile_kmin + rel) mapped back from ukey space
        auto key_of = [&](uint16_t s) -> int64_t {
            if (sm.narrow)
                return (int64_t)((sm.tile_kmin +
                                  reinterpret_cast<const uint32_t *>(
                                      sm.skey)[s]) ^
                                 0x8000000000000000ull);
            return sm.skey[s];
        };

        // emit pass fills the LDS winner list in key order (direct global
        // writes from the walk measured SLOWER: per-thread ranges scatter,
        // while wl-indexed loops below stream perfectly coalesced)
        walk_pass(my_lo, my_hi, [&](int32_t n, uint16_t s_best) {
            wl[my_off + n] = s_best;
        });
        __syncthreads();

        // per-thread slot map over the flat element space (computed once,
        // reused by every column): slot s covers element tid + s*T
        constexpr int RMAX =
            (PMH_TILE_MAX + PMH_TILE_THREADS - 1) / PMH_TILE_THREADS;
        // packed (run << PMH_ROW_BITS) | row per slot; ~0u = dead slot
        uint32_t smap[RMAX];
#pragma unroll
        for (int s = 0; s < RMAX; s++) {
            int32_t e = tid + s * PMH_TILE_THREADS;
            smap[s] = ~0u;
            if (e < M) {
                int r = 0;
                while (r + 1 < k && sm.segoff[r + 1] <= e) r++;
                smap[s] = ((uint32_t)r << PMH_ROW_BITS) |
                          (uint32_t)(c0[r] + (e - sm.segoff[r]));
            }
        }

        // register staging: issue a column's gathers into registers FIRST
        // (no LDS write yet, so the loads stay in flight), do independent
        // work (emit the previous column from the other slab), then write
        // the registers to LDS — the vmcnt wait lands after the overlap
        // work instead of before it.
        int64_t rg[RMAX];
        auto stage_load = [&](int c, int64_t *regs) {
#pragma unroll
            for (int s = 0; s < RMAX; s++) {
                if (smap[s] == ~0u) continue;
                const DevCol &dc =
                    cols[(smap[s] >> PMH_ROW_BITS) * n_cols + c];
                const int64_t row = smap[s] & PMH_ROW_MASK;
                regs[s] = dc.esize == 8
                    ? reinterpret_cast<const int64_t *>(dc.addr0)[row]
                    : (int64_t)reinterpret_cast<const int32_t *>(
                          dc.addr0)[row];
            }
        };
        auto stage_write = [&](int c, const int64_t *regs, uint8_t *slab) {
            const int dt = col_dtype[c];
            const bool wide = dt == 4 || dt == 6;
#pragma unroll
            for (int s = 0; s < RMAX; s++) {
                if (smap[s] == ~0u) continue;
                int32_t e = tid + s * PMH_TILE_THREADS;
                if (wide)
                    reinterpret_cast<int64_t *>(slab)[e] = regs[s];
                else
                    reinterpret_cast<int32_t *>(slab)[e] = (int32_t)regs[s];
            }
        };
        auto stage_valid = [&](int c, uint8_t *slab) {
#pragma unroll
            for (int s = 0; s < RMAX; s++) {
                if (smap[s] == ~0u) continue;
                int32_t e = tid + s * PMH_TILE_THREADS;
                const DevCol &dc =
                    cols[(smap[s] >> PMH_ROW_BITS) * n_cols + c];
                slab[e] = dc.valid0
                    ? reinterpret_cast<const uint8_t *>(
                          dc.valid0)[smap[s] & PMH_ROW_MASK]
                    : 1;
            }
        };
        auto emit_col = [&](int c, const uint8_t *slab,
                            const uint8_t *vslab) {
            if (ablate == 3) return;  // profiling: staging without emission
            const int dt = col_dtype[c];
            const bool wide = dt == 4 || dt == 6;
            uint8_t *ov = (col_nullable[c] && out_valid[c]) ? out_valid[c]
                                                            : nullptr;
            for (int32_t i = tid; i < C; i += blockDim.x) {
                uint16_t s = wl[i];
                int64_t v = wide
                    ? reinterpret_cast<const int64_t *>(slab)[s]
                    : (int64_t)reinterpret_cast<const int32_t *>(slab)[s];
                switch (dt) {
                case 1: ((int8_t *)out_ptrs[c])[goff + i] = (int8_t)v; break;
                case 2: ((int16_t *)out_ptrs[c])[goff + i] =
                            (int16_t)v; break;
                case 3:
                case 5:
                case 7: ((int32_t *)out_ptrs[c])[goff + i] =
                            (int32_t)v; break;
                default: ((int64_t *)out_ptrs[c])[goff + i] = v; break;
                }
                if (ov) ov[goff + i] = vslab ? vslab[s] : 1;
            }
        };
        auto next_col = [&](int c) -> int {
            for (c++; c < n_cols; c++)
                if (c != seq_col && c != kind_col && c != key_col) return c;
            return -1;
        };

        bool any_null = false;
        for (int c = 0; c < n_cols; c++)
            if (col_nullable[c]) any_null = true;
        // emission mode (flags bit 14): 0 = direct gather per winner (high
        // MLP, no per-column barriers — winners are ~72% dense within their
        // run segments, so the 64B gather lines are mostly reused and the
        // working set stays XCD-L2-resident); 1 = LDS-staged columns
        // (coalesced loads but burst-and-wait per column: measured slower,
        // kept for A/B via PMH_FSTAGE)
        const bool staged = (flags >> 14) & 1;

        // --- overlap window 1: first column's gathers fly while the
        // key/seq/kind emit reads the merge arrays
        int cfirst = ablate == 1 ? -1 : next_col(-1);
        if (staged && cfirst >= 0 && !any_null) stage_load(cfirst, rg);

        // emit key / seq / kind straight from the merge arrays
        if (key_col >= 0) {
            const int kdt = col_dtype[key_col];
            for (int32_t i = tid; i < C; i += blockDim.x) {
                int64_t v = key_of(wl[i]);
                switch (kdt) {
                case 1: ((int8_t *)out_ptrs[key_col])[goff + i] =
                            (int8_t)v; break;
                case 2: ((int16_t *)out_ptrs[key_col])[goff + i] =
                            (int16_t)v; break;
                case 3: ((int32_t *)out_ptrs[key_col])[goff + i] =
                            (int32_t)v; break;
                default: ((int64_t *)out_ptrs[key_col])[goff + i] = v; break;
                }
            }
        }
        for (int32_t i = tid; i < C; i += blockDim.x) {
            int64_t w = sm.sseq[wl[i]];
            ((int64_t *)out_ptrs[seq_col])[goff + i] = w >> 2;
            ((int8_t *)out_ptrs[kind_col])[goff + i] = (int8_t)ps2_kind(w);
        }
        if (dense_winners) {
            // SPLIT mode: publish packed winners densely; value columns are
            // emitted by k_emit_dense (full occupancy, no LDS) right after
            // this kernel
            for (int32_t i = tid; i < C; i += blockDim.x) {
                uint16_t s = wl[i];
                int r = 0;
                while (r + 1 < k && sm.segoff[r + 1] <= (int32_t)s) r++;
                dense_winners[goff + i] =
                    ((uint32_t)r << PMH_ROW_BITS) |
                    (uint32_t)(c0[r] + ((int32_t)s - sm.segoff[r]));
            }
            __syncthreads();
            continue;
        }
        if (cfirst < 0) {
            __syncthreads();
            continue;
        }
        if (!staged) {
            // --- direct-gather emission: thread i owns winners i, i+T,
            // i+2T, ... R at a time wave-strided (coalesced stores); per
            // column R independent gathers stay in flight. No barriers, no
            // LDS traffic — overlaps freely with the other workgroup's
            // merge phase on the CU.
            constexpr int R = 4;
            for (int32_t base = 0; base < C;
                 base += (int32_t)blockDim.x * R) {
                int32_t i0 = base + tid;
                if (i0 >= C) break;
                int run[R];
                int64_t row[R];
                int32_t idx[R];
                int nr = 0;
#pragma unroll
                for (int x = 0; x < R; x++) {
                    int32_t i = i0 + x * (int32_t)blockDim.x;
                    bool live = i < C;
                    idx[x] = i;
                    if (live) {
                        nr = x + 1;
                        uint16_t s = wl[i];
                        int r = 0;
                        while (r + 1 < k && sm.segoff[r + 1] <= (int32_t)s)
                            r++;
                        run[x] = r;
                        row[x] = c0[r] + ((int32_t)s - sm.segoff[r]);
                    } else {
                        run[x] = run[0];
                        row[x] = row[0];
                    }
                }
                for (int c = cfirst; c >= 0; c = next_col(c)) {
                    const int dt = col_dtype[c];
                    uint8_t *ov = (col_nullable[c] && out_valid[c])
                                      ? out_valid[c] : nullptr;
                    if (ov) {
                        uint8_t vv[R];
#pragma unroll
                        for (int x = 0; x < R; x++) {
                            const DevCol &dc = cols[run[x] * n_cols + c];
                            vv[x] = dc.valid0
                                ? reinterpret_cast<const uint8_t *>(
                                      dc.valid0)[row[x]]
                                : 1;
                        }
#pragma unroll
                        for (int x = 0; x < R; x++)
                            if (x < nr) ov[goff + idx[x]] = vv[x];
                    }
                    if (dt == 4 || dt == 6) {
                        int64_t v[R];
#pragma unroll
                        for (int x = 0; x < R; x++)
                            v[x] = reinterpret_cast<const int64_t *>(
                                cols[run[x] * n_cols + c].addr0)[row[x]];
#pragma unroll
                        for (int x = 0; x < R; x++)
                            if (x < nr)
                                ((int64_t *)out_ptrs[c])[goff + idx[x]] =
                                    v[x];
                    } else {
                        int32_t v[R];
#pragma unroll
                        for (int x = 0; x < R; x++)
                            v[x] = reinterpret_cast<const int32_t *>(
                                cols[run[x] * n_cols + c].addr0)[row[x]];
#pragma unroll
                        for (int x = 0; x < R; x++) {
                            if (x >= nr) continue;
                            switch (dt) {
                            case 1: ((int8_t *)out_ptrs[c])[goff + idx[x]] =
                                        (int8_t)v[x]; break;
                            case 2: ((int16_t *)out_ptrs[c])[goff + idx[x]] =
                                        (int16_t)v[x]; break;
                            default: ((int32_t *)out_ptrs[c])[goff + idx[x]] =
                                        v[x]; break;
                            }
                        }
                    }
                }
            }
            __syncthreads();  // wl/segoff stay live until every thread done
            continue;
        }
        __syncthreads();  // skey/sseq die; vbuf slabs take their space
        if (any_null) {
            // nullable plans: values in slab 0, validity bytes in slab 1
            // (single-buffered; the dedup headline configs carry no nulls)
            for (int c = cfirst; c >= 0; c = next_col(c)) {
                stage_load(c, rg);
                stage_write(c, rg, sm.vbuf[0]);
                const bool nul = col_nullable[c];
                if (nul) stage_valid(c, sm.vbuf[1]);
                __syncthreads();
                emit_col(c, sm.vbuf[0], nul ? sm.vbuf[1] : nullptr);
                __syncthreads();
            }
        } else {
            // double slab + register pipeline: column c+1's gathers are in
            // flight while column c is emitted; one barrier per column
            stage_write(cfirst, rg, sm.vbuf[0]);
            int prev = cfirst, slot = 0;
            for (int c = next_col(cfirst); c >= 0; c = next_col(c)) {
                stage_load(c, rg);  // loads fly across the barrier + emit
                __syncthreads();  // slab `slot` ready; slot^1 drained
                emit_col(prev, sm.vbuf[slot], nullptr);
                stage_write(c, rg, sm.vbuf[slot ^ 1]);
                prev = c;
                slot ^= 1;
            }
            __syncthreads();
            emit_col(prev, sm.vbuf[slot], nullptr);
        }
        __syncthreads();  // all slabs drained before the next tile
    }
}


// ------------------------------------------------------------ k_emit_dense
//
// Value emission for the SPLIT fused path: k_merge_emit already wrote key/
// seq/kind and the packed winners DENSELY in output order, so this kernel
// is a pure gather — no LDS, no tile search, 60-VGPR occupancy (8 waves/
// SIMD vs the merge kernel's 4). Winners are ~72% dense inside their run
// segments, so gather lines are mostly reused and stay XCD-L2-resident.
__global__ void k_emit_dense(const DevCol *cols, const uint8_t *col_dtype,
                             const uint8_t *col_nullable, int n_cols,
                             int key_col, int seq_col, int kind_col,
                             const uint32_t *winners,
                             int64_t t0, int64_t t1,
                             const uint64_t *status,
                             void *const *out_ptrs,
                             uint8_t *const *out_valid) {
    // output range of tile chunk [t0, t1): the inclusive lookback prefixes
    // of t0-1 / t1-1 (both PREFIX-published — the chunk's merge pass
    // completed before this launch was released by its event)
    constexpr int R = 4;
    const int64_t lo =
        t0 > 0 ? (int64_t)(status[t0 - 1] & LOOK_VAL) : 0;
    const int64_t total = (int64_t)(status[t1 - 1] & LOOK_VAL);
    const int64_t span = total - lo;
    const int64_t per_block =
        (span + (int64_t)gridDim.x - 1) / (int64_t)gridDim.x;
    const int64_t slice_lo = lo + (int64_t)blockIdx.x * per_block;
    const int64_t slice_hi =
        slice_lo + per_block < total ? slice_lo + per_block : total;
    for (int64_t base = slice_lo; base < slice_hi;
         base += (int64_t)blockDim.x * R) {
        int64_t i0 = base + threadIdx.x;
        if (i0 >= slice_hi) break;
        int run[R];
        int64_t row[R], idx[R];
        int nr = 0;
#pragma unroll
        for (int x = 0; x < R; x++) {
            int64_t i = i0 + (int64_t)x * blockDim.x;
            bool live = i < slice_hi;
            idx[x] = i;
            if (live) {
                nr = x + 1;
                uint32_t packed = winners[i];
                run[x] = packed >> PMH_ROW_BITS;
                row[x] = packed & PMH_ROW_MASK;
            } else {
                run[x] = run[0];
                row[x] = row[0];
            }
        }
        for (int c = 0; c < n_cols; c++) {
            if (c == key_col || c == seq_col || c == kind_col) continue;
            const int dt = col_dtype[c];
            if (col_nullable[c] && out_valid[c]) {
                uint8_t vv[R];
#pragma unroll
                for (int x = 0; x < R; x++) {
                    const DevCol &dc = cols[run[x] * n_cols + c];
                    vv[x] = dc.valid0
                        ? reinterpret_cast<const uint8_t *>(
                              dc.valid0)[row[x]]
                        : 1;
                }
#pragma unroll
                for (int x = 0; x < R; x++)
                    if (x < nr) out_valid[c][idx[x]] = vv[x];
            }
            if (dt == 4 || dt == 6) {
                int64_t v[R];
#pragma unroll
                for (int x = 0; x < R; x++)
                    v[x] = reinterpret_cast<const int64_t *>(
                        cols[run[x] * n_cols + c].addr0)[row[x]];
#pragma unroll
                for (int x = 0; x < R; x++)
                    if (x < nr) ((int64_t *)out_ptrs[c])[idx[x]] = v[x];
            } else {
                int32_t v[R];
#pragma unroll
                for (int x = 0; x < R; x++)
                    v[x] = reinterpret_cast<const int32_t *>(
                        cols[run[x] * n_cols + c].addr0)[row[x]];
#pragma unroll
                for (int x = 0; x < R; x++) {
                    if (x >= nr) continue;
                    switch (dt) {
                    case 1: ((int8_t *)out_ptrs[c])[idx[x]] =
                                (int8_t)v[x]; break;
                    case 2: ((int16_t *)out_ptrs[c])[idx[x]] =
                                (int16_t)v[x]; break;
                    default: ((int32_t *)out_ptrs[c])[idx[x]] =
                                 v[x]; break;
                    }
                }
            }
        }
    }
}

// ------------------------------------------------------------ k_scan_tiles
// Exclusive scan of tile_counts (single workgroup, chunked).
__global__ void k_scan_tiles(const int32_t *tile_counts, int64_t n_tiles,
                             int64_t *tile_offsets, int64_t *total_out) {
    __shared__ int64_t s[PMH_TILE_THREADS];
    __shared__ int64_t running;
    if (threadIdx.x == 0) running = 0;
    __syncthreads();
    for (int64_t base = 0; base < n_tiles; base += blockDim.x) {
        int64_t i = base + threadIdx.x;
        int64_t v = i < n_tiles ? tile_counts[i] : 0;
        s[threadIdx.x] = v;
        __syncthreads();
        for (int d = 1; d < (int)blockDim.x; d <<= 1) {
            int64_t add = threadIdx.x >= d ? s[threadIdx.x - d] : 0;
            __syncthreads();
            s[threadIdx.x] += add;
            __syncthreads();
        }
        if (i < n_tiles) tile_offsets[i] = running + s[threadIdx.x] - v;
        __syncthreads();
        if (threadIdx.x == 0) running += s[blockDim.x - 1];
        __syncthreads();
    }
    if (threadIdx.x == 0) *total_out = running;
}

// ------------------------------------------------------------ k_emit
// Phase 2: gather winner rows into contiguous output columns.
// One workgroup per tile (grid-stride); thread per output row; inner loop
// over columns. Output writes are coalesced per column.
template <int R>
__global__ void k_emit(const DevCol *cols /* n_runs * n_cols, run-major */,
                       const uint8_t *col_dtype, const uint8_t *col_nullable,
                       int n_cols, int k, const uint32_t *winners,
                       const int32_t *tile_counts,
                       const int64_t *tile_offsets, int64_t n_tiles,
                       int64_t tile_rows, const int64_t *total_out,
                       void *const *out_ptrs, uint8_t *const *out_valid) {
    // Flat grid-stride over the DENSE output index space (tile_offsets is an
    // exclusive scan of tile_counts, so tile t owns output ranks
    // [off[t], off[t]+cnt[t]) contiguously). R rows per thread iteration:
    // independent gathers per column keep R x n_cols loads in flight (emit
    // was 96% latency-parked, profiles/r01_c2_pmc.md) without idling tile
    // remainders.
    // Each block owns a CONTIGUOUS output slice so its gathers stay within a
    // few tiles' source segments (~130 KB working set -> XCD-L2 resident;
    // grid-striding spread every block over the whole output and thrashed L2).
    const int64_t total = *total_out;
    const int64_t per_block =
        (total + (int64_t)gridDim.x - 1) / (int64_t)gridDim.x;
    const int64_t slice_lo = (int64_t)blockIdx.x * per_block;
    const int64_t slice_hi = slice_lo + per_block < total
                                 ? slice_lo + per_block : total;
    // one tile binary search per thread; ranks then advance monotonically
    // within the slice, so the tile cursor just walks forward. The R rows
    // per thread are WAVE-STRIDED (i = base + tid + x*blockDim), so every
    // access stays fully coalesced while keeping R loads in flight —
    // thread-contiguous R strides the lanes by R and breaks coalescing.
    int64_t t = -1;
    for (int64_t base = slice_lo; base < slice_hi;
         base += (int64_t)blockDim.x * R) {
        int64_t i_first = base + threadIdx.x;
        if (i_first >= slice_hi) break;
        if (t < 0) {  // first iteration: locate owning tile
            int64_t lo = 0, hi = n_tiles - 1;
            while (lo < hi) {
                int64_t mid = (lo + hi + 1) >> 1;
                if (tile_offsets[mid] <= i_first) lo = mid;
                else hi = mid - 1;
            }
            t = lo;
        }
        int64_t idx[R];
        int run[R];
        int64_t row[R];
        int nr = 0;
#pragma unroll
        for (int x = 0; x < R; x++) {
            int64_t i = base + threadIdx.x + (int64_t)x * blockDim.x;
            bool live = i < slice_hi;
            idx[x] = i;
            if (live) {
                nr = x + 1;
                // dense ranks: advance tile while i falls past its count
                while (t + 1 < n_tiles && tile_offsets[t + 1] <= i) t++;
                uint32_t packed =
                    winners[t * (tile_rows + PMH_MAX_RUNS) +
                            (i - tile_offsets[t])];
                run[x] = packed >> PMH_ROW_BITS;
                row[x] = packed & PMH_ROW_MASK;
            } else {  // dead lane: duplicate x=0's gather, stores are guarded
                run[x] = run[0];
                row[x] = row[0];
            }
        }
        for (int c = 0; c < n_cols; c++) {
            if (col_nullable[c] && out_valid[c]) {
                // winner's validity byte (dedup keeps the record as-is)
#pragma unroll
                for (int x = 0; x < R; x++) {
                    if (x >= nr) continue;
                    const DevCol &dc = cols[run[x] * n_cols + c];
                    out_valid[c][idx[x]] =
                        dc.valid0 ? ((const uint8_t *)dc.valid0)[row[x]] : 1;
                }
            }
            switch (col_dtype[c]) {
            case 1: {  // INT8 output from INT32-stored parquet TINYINT
                int32_t v[R];
#pragma unroll
                for (int x = 0; x < R; x++)
                    v[x] = col_load<int32_t>(cols[run[x] * n_cols + c], row[x]);
#pragma unroll
                for (int x = 0; x < R; x++)
                    if (x < nr)
                        ((int8_t *)out_ptrs[c])[idx[x]] = (int8_t)v[x];
                break;
            }
            case 2: {  // INT16 output from INT32-stored parquet SMALLINT
                int32_t v[R];
#pragma unroll
                for (int x = 0; x < R; x++)
                    v[x] = col_load<int32_t>(cols[run[x] * n_cols + c], row[x]);
#pragma unroll
                for (int x = 0; x < R; x++)
                    if (x < nr)
                        ((int16_t *)out_ptrs[c])[idx[x]] = (int16_t)v[x];
                break;
            }
            case 3:
            case 5:
            case 7: {  // string ids ride as int32
                int32_t v[R];
#pragma unroll
                for (int x = 0; x < R; x++)
                    v[x] = col_load<int32_t>(cols[run[x] * n_cols + c], row[x]);
#pragma unroll
                for (int x = 0; x < R; x++)
                    if (x < nr) ((int32_t *)out_ptrs[c])[idx[x]] = v[x];
                break;
            }
            case 4:
            case 6: {
                int64_t v[R];
#pragma unroll
                for (int x = 0; x < R; x++)
                    v[x] = col_load<int64_t>(cols[run[x] * n_cols + c], row[x]);
#pragma unroll
                for (int x = 0; x < R; x++)
                    if (x < nr) ((int64_t *)out_ptrs[c])[idx[x]] = v[x];
                break;
            }
            default: break;
            }
        }
    }
}

// ------------------------------------------------------------ k_composite
//
// Build one run's order-preserving composite key: each integer sub-key is
// biased to unsigned at its declared bit width and concatenated MSB-first
// (first key column highest), then the u64 is sign-flipped back to int64 so
// ukey() in the partition/merge recovers the unsigned order. Paimon compares
// multi-column keys lexicographically (keyComparator over the _KEY_ fields),
// which this encoding preserves exactly for integer columns whose widths sum
// to <= 64 bits. Rebuilt per pass (decode-derived).
// spec packs (shift, bits) per sub-key, 8 bits each, sub-key i at byte i.
__global__ void k_composite(const DevCol *keys, int nk, uint64_t shifts,
                            uint64_t bits, int64_t rows, int64_t *ckey) {
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < rows;
         i += stride) {
        uint64_t u = 0;
        for (int j = 0; j < nk; j++) {
            const DevCol &dc = keys[j];
            int64_t v = dc.esize == 8 ? col_load<int64_t>(dc, i)
                                      : (int64_t)col_load<int32_t>(dc, i);
            const int w = (int)((bits >> (8 * j)) & 0xff);
            const int sh = (int)((shifts >> (8 * j)) & 0xff);
            uint64_t b = ((uint64_t)v + (1ull << (w - 1))) &
                         (w >= 64 ? ~0ull : ((1ull << w) - 1));
            u |= b << sh;
        }
        ckey[i] = (int64_t)(u ^ 0x8000000000000000ull);
    }
}

// ------------------------------------------------------------ k_pack_valid
//
// Pack one run's per-column validity bytes into a u64 bitmask per row
// (bit c = column c non-null; columns with no staged nulls contribute 1).
// The PU/aggregation emit then resolves EVERY column's nullability from one
// mask load per group member instead of one byte load per (member, column) —
// the per-column walk was a serial dependent-load chain over up to n_cols
// scattered arrays and dominated the C3 emit.
__global__ void k_pack_valid(const DevCol *cols, int n_cols, int64_t rows,
                             uint64_t *mask) {
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < rows;
         i += stride) {
        uint64_t m = 0;
        for (int c = 0; c < n_cols; c++) {
            const DevCol &dc = cols[c];
            uint64_t v =
                dc.valid0 ? (uint64_t)((const uint8_t *)dc.valid0)[i] : 1u;
            m |= (v & 1) << c;
        }
        mask[i] = m;
    }
}

// ------------------------------------------------------------ k_emit_pu
//
// PartialUpdate emit: one output row per owned group; for each value column
// take the LAST member whose field is non-null (updateNonNullFields,
// PartialUpdateMergeFunction.java:188-215; members arrive in ascending
// (seq, isAdd) order). seq = last member's sequenceNumber
// (latestSequenceNumber, :188); kind = INSERT (insert-only streams, v1).
// Singleton groups return the record as-is (ReducerMergeFunctionWrapper
// bypass) — identical to the overlay for INSERT records.
// MASKS: per-row packed validity (k_pack_valid) available — one u64 load per
// member resolves all columns; false = legacy per-column byte walk (>64
// columns).
template <bool MASKS>
__global__ void k_emit_pu(const DevCol *cols, const uint8_t *col_dtype,
                          const uint8_t *col_nullable, int n_cols, int k,
                          int seq_col, int kind_col, int flags,
                          const uint32_t *members,
                          const uint16_t *group_start,
                          const int64_t *tile_offsets, int64_t n_tiles,
                          int64_t tile_rows, const int64_t *total_out,
                          uint64_t *const *run_masks, void *const *out_ptrs,
                          uint8_t *const *out_valid) {
    // remove-record-on-delete (flags bit 16): a DELETE resets the row —
    // per column, overlay only the adds NEWER than the last DELETE and fall
    // back to the DELETE record's own field (initRow semantics,
    // PartialUpdateMergeFunction.java:173-180). Result kind = DELETE when
    // the last member is the DELETE (getResult, :213-221). Groups whose result
    // DELETE were already dropped by the merge pass under drop_delete.
    const bool rrod = (flags & 16) != 0;
    const int64_t total = *total_out;
    const int64_t per_block =
        (total + (int64_t)gridDim.x - 1) / (int64_t)gridDim.x;
    const int64_t slice_lo = (int64_t)blockIdx.x * per_block;
    const int64_t slice_hi =
        slice_lo + per_block < total ? slice_lo + per_block : total;
    int64_t t = -1;
    for (int64_t i = slice_lo + threadIdx.x; i < slice_hi; i += blockDim.x) {
        if (t < 0) {  // first iteration: binary-search the owning tile; i
                      // then grows monotonically, so a forward cursor walks
            int64_t lo = 0, hi = n_tiles - 1;
            while (lo < hi) {
                int64_t mid = (lo + hi + 1) >> 1;
                if (tile_offsets[mid] <= i) lo = mid;
                else hi = mid - 1;
            }
            t = lo;
        }
        while (t + 1 < n_tiles && tile_offsets[t + 1] <= i) t++;
        int64_t g = i - tile_offsets[t];
        const uint16_t *gs = &group_start[t * (tile_rows + 1)];
        const uint32_t *mem = &members[t * (tile_rows + PMH_MAX_RUNS)];
        int32_t ms = gs[g], me = gs[g + 1];
        // cache the newest members in registers: the per-column non-null
        // walk starts at the newest record and nearly always ends within a
        // few members (group size averages ~1.3), and re-reading the member
        // list from HBM per column dominated this kernel
        uint32_t mc[4];
        uint64_t vm[4];
        const int gn = me - ms;
#pragma unroll
        for (int x = 0; x < 4; x++)
            mc[x] = x < gn ? mem[me - 1 - x] : mem[me - 1];
        if (MASKS) {
#pragma unroll
            for (int x = 0; x < 4; x++)
                vm[x] = x < gn
                            ? run_masks[mc[x] >> PMH_ROW_BITS][mc[x] & PMH_ROW_MASK]
                            : 0;
        }
        uint32_t last = mc[0];
        int lrun = last >> PMH_ROW_BITS;
        int64_t lrow = last & PMH_ROW_MASK;
        // singleton bypass serves the record's own RowKind: issue that
        // gather HERE so it overlaps the member prefetch instead of
        // stalling the column loop (measured +65% on emit_pu when loaded
        // at the kind column)
        int32_t single_kind = 0;
        if (gn == 1)
            single_kind = col_load<int32_t>(
                cols[lrun * n_cols + kind_col], lrow);
        // newest-first index of the last DELETE member; gn = none (also the
        // value when RROD is off, which neutralizes every bound below)
        int j_del = gn;
        if (rrod) {
#pragma unroll
            for (int x = 0; x < 4; x++) {
                if (x >= gn || j_del < gn) continue;
                const DevCol &dc = cols[(mc[x] >> PMH_ROW_BITS) * n_cols + kind_col];
                if (col_load<int32_t>(dc, mc[x] & PMH_ROW_MASK) == 3) j_del = x;
            }
            for (int x = 4; x < gn && j_del == gn; x++) {
                uint32_t m = mem[me - 1 - x];
                const DevCol &dc = cols[(m >> PMH_ROW_BITS) * n_cols + kind_col];
                if (col_load<int32_t>(dc, m & PMH_ROW_MASK) == 3) j_del = x;
            }
        }
        for (int c = 0; c < n_cols; c++) {
            if (c == kind_col) {
                // wrapper bypass: singletons keep their own RowKind
                ((int8_t *)out_ptrs[c])[i] = (int8_t)(
                    gn == 1 ? single_kind
                            : ((rrod && j_del == 0) ? 3 : 0));
                continue;
            }
            int64_t run = lrun, row = lrow;
            uint8_t ok = 1;
            if (col_nullable[c] && gn > 1) {
                ok = 0;
#pragma unroll
                for (int x = 0; x < 4; x++) {
                    if (ok || x >= gn || x >= j_del) continue;
                    uint32_t m = mc[x];
                    uint8_t v;
                    if (MASKS) {
                        v = (uint8_t)((vm[x] >> c) & 1);
                    } else {
                        const DevCol &dc = cols[(m >> PMH_ROW_BITS) * n_cols + c];
                        v = dc.valid0
                                ? ((const uint8_t *)dc.valid0)[m & PMH_ROW_MASK]
                                : 1;
                    }
                    if (v) {
                        run = m >> PMH_ROW_BITS;
                        row = m & PMH_ROW_MASK;
                        ok = 1;
                    }
                }
                for (int32_t x = me - 5; !ok && x >= ms; x--) {
                    if (me - 1 - x >= j_del) break;  // not newer than DELETE
                    uint32_t m = mem[x];
                    uint8_t v;
                    if (MASKS) {
                        v = (uint8_t)(
                            (run_masks[m >> PMH_ROW_BITS][m & PMH_ROW_MASK] >> c) & 1);
                    } else {
                        const DevCol &dc = cols[(m >> PMH_ROW_BITS) * n_cols + c];
                        v = dc.valid0
                                ? ((const uint8_t *)dc.valid0)[m & PMH_ROW_MASK]
                                : 1;
                    }
                    if (v) {
                        run = m >> PMH_ROW_BITS;
                        row = m & PMH_ROW_MASK;
                        ok = 1;
                    }
                }
            } else if (col_nullable[c]) {
                // singleton: the record passes through with its own validity
                if (MASKS) {
                    ok = (uint8_t)((vm[0] >> c) & 1);
                } else {
                    const DevCol &dc = cols[lrun * n_cols + c];
                    ok = dc.valid0 ? ((const uint8_t *)dc.valid0)[lrow] : 1;
                }
            }
            if (!ok && j_del < gn && col_nullable[c] && gn > 1) {
                // no add newer than the DELETE set this field: the row was
                // re-initialized from the DELETE record's value (initRow)
                uint32_t m = mem[me - 1 - j_del];
                uint8_t v;
                if (MASKS) {
                    v = (uint8_t)(
                        (run_masks[m >> PMH_ROW_BITS][m & PMH_ROW_MASK] >> c) & 1);
                } else {
                    const DevCol &dc = cols[(m >> PMH_ROW_BITS) * n_cols + c];
                    v = dc.valid0
                            ? ((const uint8_t *)dc.valid0)[m & PMH_ROW_MASK]
                            : 1;
                }
                if (v) {
                    run = m >> PMH_ROW_BITS;
                    row = m & PMH_ROW_MASK;
                    ok = 1;
                }
            }
            const DevCol &dc = cols[run * n_cols + c];
            switch (col_dtype[c]) {
            case 1:
                ((int8_t *)out_ptrs[c])[i] =
                    ok ? (int8_t)col_load<int32_t>(dc, row) : 0;
                break;
            case 2:
                ((int16_t *)out_ptrs[c])[i] =
                    ok ? (int16_t)col_load<int32_t>(dc, row) : 0;
                break;
            case 3:
            case 5:
            case 7:  // string ids ride as int32
                ((int32_t *)out_ptrs[c])[i] =
                    ok ? col_load<int32_t>(dc, row) : 0;
                break;
            case 4:
            case 6:
                ((int64_t *)out_ptrs[c])[i] =
                    ok ? col_load<int64_t>(dc, row) : 0;
                break;
            default: break;
            }
            if (out_valid[c]) out_valid[c][i] = ok;
        }
    }
}


// ---------------------------------------------------------- k_emit_pu_sg
//
// PartialUpdate with SEQUENCE GROUPS (updateWithSequenceGroup,
// PartialUpdateMergeFunction.java:219-282; retractWithSequenceGroup
// :301-377; isEmptySequenceGroup :284-299; initRow :399-407; getResult
// :389-397). Streams may carry INSERT/UPDATE_AFTER/UPDATE_BEFORE/DELETE;
// retracts act only on their sequence groups (no per-field aggregators in
// v1 — their retract support is a later round).
//
// The reference's sequential per-key fold reduces, per group, to the LAST
// PREFIX-MAX ACHIEVER over the members' sequence-field tuples (non-empty
// tuples only; lexicographic ascending, nulls FIRST; ties -> later member):
// each accepted event overwrites every field it touches, so only the last
// accepted event matters. Proven equivalent to the sequential port by
// randomized fuzz (oracle partial_update_seqgroup_model; see git history).
// col_group[c]: group id or 0xff (plain); sg_fields[g*4+j]: the group's
// sequence-field column indices; sg_nseq[g]: how many (<= 4).
__global__ void k_emit_pu_sg(const DevCol *cols, const uint8_t *col_dtype,
                             const uint8_t *col_nullable, int n_cols, int k,
                             int seq_col, int kind_col, int flags,
                             const uint8_t *col_group,
                             const int16_t *sg_fields,
                             const uint8_t *sg_nseq, int n_groups,
                             const uint32_t *members,
                             const uint16_t *group_start,
                             const int64_t *tile_offsets, int64_t n_tiles,
                             int64_t tile_rows, const int64_t *total_out,
                             uint64_t *const *run_masks,
                             void *const *out_ptrs,
                             uint8_t *const *out_valid) {
    const bool ignore_delete = flags & 2;
    const int64_t total = *total_out;
    const int64_t per_block =
        (total + (int64_t)gridDim.x - 1) / (int64_t)gridDim.x;
    const int64_t slice_lo = (int64_t)blockIdx.x * per_block;
    const int64_t slice_hi =
        slice_lo + per_block < total ? slice_lo + per_block : total;
    auto kind_of = [&](uint32_t m) -> int32_t {
        const DevCol &dc = cols[(m >> PMH_ROW_BITS) * n_cols + kind_col];
        return col_load<int32_t>(dc, m & PMH_ROW_MASK);
    };
    auto valid_of = [&](uint32_t m, int c) -> uint8_t {
        if (run_masks)
            return (uint8_t)(
                (run_masks[m >> PMH_ROW_BITS][m & PMH_ROW_MASK] >> c) & 1);
        const DevCol &dc = cols[(m >> PMH_ROW_BITS) * n_cols + c];
        return dc.valid0
                   ? ((const uint8_t *)dc.valid0)[m & PMH_ROW_MASK]
                   : 1;
    };
    auto load_of = [&](uint32_t m, int c) -> int64_t {
        const DevCol &dc = cols[(m >> PMH_ROW_BITS) * n_cols + c];
        const int dt = col_dtype[c];
        return (dt == 4 || dt == 6)
                   ? col_load<int64_t>(dc, m & PMH_ROW_MASK)
                   : (int64_t)col_load<int32_t>(dc, m & PMH_ROW_MASK);
    };
    int64_t t = -1;
    for (int64_t i = slice_lo + threadIdx.x; i < slice_hi; i += blockDim.x) {
        if (t < 0) {
            int64_t lo = 0, hi = n_tiles - 1;
            while (lo < hi) {
                int64_t mid = (lo + hi + 1) >> 1;
                if (tile_offsets[mid] <= i) lo = mid;
                else hi = mid - 1;
            }
            t = lo;
        }
        while (t + 1 < n_tiles && tile_offsets[t + 1] <= i) t++;
        const int64_t g = i - tile_offsets[t];
        const uint16_t *gs = &group_start[t * (tile_rows + 1)];
        const uint32_t *mem = &members[t * (tile_rows + PMH_MAX_RUNS)];
        const int32_t ms = gs[g], me = gs[g + 1];
        const int gn = me - ms;
        const uint32_t m0 = mem[ms];
        const int32_t kind0 = kind_of(m0);
        const bool first_retract = kind0 == 1 || kind0 == 3;
        if (gn == 1) {
            // ReducerMergeFunctionWrapper singleton bypass: the record is
            // served as-is with its own RowKind
            for (int c = 0; c < n_cols; c++) {
                if (c == kind_col) {
                    ((int8_t *)out_ptrs[c])[i] = (int8_t)kind0;
                    continue;
                }
                uint8_t ok = col_nullable[c] ? valid_of(m0, c) : 1;
                const int64_t v = ok ? load_of(m0, c) : 0;
                switch (col_dtype[c]) {
                case 1: ((int8_t *)out_ptrs[c])[i] = (int8_t)v; break;
                case 2: ((int16_t *)out_ptrs[c])[i] = (int16_t)v; break;
                case 3:
                case 5:
                case 7: ((int32_t *)out_ptrs[c])[i] = (int32_t)v; break;
                default: ((int64_t *)out_ptrs[c])[i] = v; break;
                }
                if (out_valid[c]) out_valid[c][i] = ok;
            }
            continue;
        }
        // per-group winner member (last prefix-max achiever), computed once
        // per row; group count bounded by plan validation (<= 16)
        int32_t win[16];
        bool win_r[16];
        for (int gg = 0; gg < n_groups; gg++) {
            win[gg] = -1;
            win_r[gg] = false;
        }
        bool meet_insert = false;
        uint32_t last_m = m0;
        uint32_t last_add = 0;
        bool has_add = false;
        for (int32_t x = 0; x < gn; x++) {
            const uint32_t m = mem[ms + x];
            const int32_t kd = kind_of(m);
            const bool retract = kd == 1 || kd == 3;
            if (!retract) {
                meet_insert = true;
                has_add = true;
                last_add = m;
            }
            if (!(retract && ignore_delete)) last_m = m;
            if (retract && ignore_delete) continue;
            for (int gg = 0; gg < n_groups; gg++) {
                const int ns = sg_nseq[gg];
                // empty tuples (all sequence fields null) never participate
                // (isEmptySequenceGroup)
                bool anyv = false;
                for (int j = 0; j < ns; j++)
                    anyv |= valid_of(m, sg_fields[gg * 4 + j]) != 0;
                if (!anyv) continue;
                if (win[gg] < 0) {
                    win[gg] = x;
                    win_r[gg] = retract;
                    continue;
                }
                // lexicographic tuple compare vs the current winner,
                // nulls FIRST; ties accept (later member wins)
                const uint32_t wm = mem[ms + win[gg]];
                int cmp = 0;
                for (int j = 0; j < ns && cmp == 0; j++) {
                    const int c = sg_fields[gg * 4 + j];
                    const uint8_t va = valid_of(m, c);
                    const uint8_t vb = valid_of(wm, c);
                    if (va != vb) cmp = va ? 1 : -1;  // null sorts first
                    else if (va) {
                        const int64_t a = load_of(m, c);
                        const int64_t b = load_of(wm, c);
                        if (a != b) cmp = a > b ? 1 : -1;
                    }
                }
                if (cmp >= 0) {
                    win[gg] = x;
                    win_r[gg] = retract;
                }
            }
        }
        for (int c = 0; c < n_cols; c++) {
            if (c == kind_col) {
                ((int8_t *)out_ptrs[c])[i] = meet_insert ? 0 : 3;
                continue;
            }
            if (c == seq_col) {
                ((int64_t *)out_ptrs[c])[i] = load_of(last_m, c);
                continue;
            }
            const int gg = col_group[c];
            uint32_t src_m = 0;
            uint8_t ok = 0;
            if (gg != 0xff) {
                const int ns = sg_nseq[gg];
                bool is_seq_field = false;
                for (int j = 0; j < ns; j++)
                    if (sg_fields[gg * 4 + j] == c) is_seq_field = true;
                if (win[gg] >= 0) {
                    if (is_seq_field || !win_r[gg]) {
                        src_m = mem[ms + win[gg]];
                        ok = valid_of(src_m, c);
                    }  // retract winner nulls the member fields
                } else if (first_retract) {  // initRow fallback
                    src_m = m0;
                    ok = valid_of(m0, c);
                }
            } else {
                // plain column: last non-null among ADD members, else the
                // initRow base when the first member was a retract
                if (first_retract && valid_of(m0, c)) {
                    src_m = m0;
                    ok = 1;
                }
                for (int32_t x = 0; x < gn; x++) {
                    const uint32_t m = mem[ms + x];
                    const int32_t kd = kind_of(m);
                    if (kd == 1 || kd == 3) continue;
                    if (valid_of(m, c)) {
                        src_m = m;
                        ok = 1;
                    }
                }
            }
            const int dt = col_dtype[c];
            const int64_t v = ok ? load_of(src_m, c) : 0;
            switch (dt) {
            case 1: ((int8_t *)out_ptrs[c])[i] = (int8_t)v; break;
            case 2: ((int16_t *)out_ptrs[c])[i] = (int16_t)v; break;
            case 3:
            case 5:
            case 7: ((int32_t *)out_ptrs[c])[i] = (int32_t)v; break;
            default: ((int64_t *)out_ptrs[c])[i] = v; break;
            }
            if (out_valid[c]) out_valid[c][i] = ok;
        }
    }
}

// ------------------------------------------------------------ k_emit_agg
//
// Aggregation emit (AggregateMergeFunction.java:82-125): one output row per
// owned group; each value column folds the group's members in ascending
// (seq, isAdd) order through its FieldAggregator:
//   last_non_null_value (default, :201)  — FieldLastNonNullValueAgg
//   last_value / first_value / first_non_null_value
//   sum  — FieldSumAgg (null skipped; int accumulates in 64-bit then
//          truncates to the column width — Java's addExact overflow check
//          is not replicated, documented in DESIGN.md; float/double add in
//          the column's own precision, in merge order, matching Java)
//   max / min — FieldMaxAgg/FieldMinAgg; floats compare in IEEE total order
//          on the stored bits (= Float.compare/Double.compare up to
//          non-canonical negative NaNs)
// Result: seq = last member's sequenceNumber, kind = INSERT (getResult,
// :119-125). Singleton groups bypass the merge function entirely
// (ReducerMergeFunctionWrapper.java:53-73) and pass through unchanged.
// v1 accepts INSERT-only streams; retracts were flagged by k_merge_tiles.
__device__ inline uint32_t f32_ord(int32_t b) {
    return b < 0 ? ~(uint32_t)b : ((uint32_t)b | 0x80000000u);
}
__device__ inline uint64_t f64_ord(int64_t b) {
    return b < 0 ? ~(uint64_t)b : ((uint64_t)b | 0x8000000000000000ull);
}

template <bool MASKS>
__global__ void k_emit_agg(const DevCol *cols, const uint8_t *col_dtype,
                           const uint8_t *col_nullable,
                           const uint8_t *col_agg, int n_cols, int k,
                           int seq_col, int kind_col, int flags,
                           const uint32_t *members,
                           const uint16_t *group_start,
                           const int64_t *tile_offsets, int64_t n_tiles,
                           int64_t tile_rows, const int64_t *total_out,
                           uint64_t *const *run_masks, void *const *out_ptrs,
                           uint8_t *const *out_valid) {
    // remove-record-on-delete (flags bit 16): a DELETE re-initializes the
    // row from its own value and the aggregators continue FROM those values
    // (AggregateMergeFunction.add, currentDeleteRow path) — so folds seed
    // with the last DELETE's field and consume only newer adds. first_*
    // aggregators are rejected at plan creation under this mode.
    const bool rrod = (flags & 16) != 0;
    auto valid_of = [&](uint32_t m, int c) -> uint8_t {
        if (MASKS)
            return (uint8_t)((run_masks[m >> PMH_ROW_BITS][m & PMH_ROW_MASK] >> c) & 1);
        const DevCol &dc = cols[(m >> PMH_ROW_BITS) * n_cols + c];
        return dc.valid0 ? ((const uint8_t *)dc.valid0)[m & PMH_ROW_MASK] : 1;
    };
    const int64_t total = *total_out;
    const int64_t per_block =
        (total + (int64_t)gridDim.x - 1) / (int64_t)gridDim.x;
    const int64_t slice_lo = (int64_t)blockIdx.x * per_block;
    const int64_t slice_hi =
        slice_lo + per_block < total ? slice_lo + per_block : total;
    int64_t t = -1;
    for (int64_t i = slice_lo + threadIdx.x; i < slice_hi; i += blockDim.x) {
        if (t < 0) {  // first iteration: binary-search the owning tile; i
                      // then grows monotonically, so a forward cursor walks
            int64_t lo = 0, hi = n_tiles - 1;
            while (lo < hi) {
                int64_t mid = (lo + hi + 1) >> 1;
                if (tile_offsets[mid] <= i) lo = mid;
                else hi = mid - 1;
            }
            t = lo;
        }
        while (t + 1 < n_tiles && tile_offsets[t + 1] <= i) t++;
        int64_t g = i - tile_offsets[t];
        const uint16_t *gs = &group_start[t * (tile_rows + 1)];
        const uint32_t *mem = &members[t * (tile_rows + PMH_MAX_RUNS)];
        const int32_t ms = gs[g], me = gs[g + 1];
        const int gn = me - ms;
        // register-cache the first 4 members (oldest-first) + their packed
        // masks; every access below uses CONSTANT indices after unrolling —
        // a dynamically-indexed local array spills the whole array to
        // scratch (64 B/lane measured before this shape)
        uint32_t ma[4];
        uint64_t vma[4];
#pragma unroll
        for (int x = 0; x < 4; x++)
            ma[x] = mem[ms + (x < gn ? x : gn - 1)];
        if (MASKS) {
#pragma unroll
            for (int x = 0; x < 4; x++)
                vma[x] = x < gn
                             ? run_masks[ma[x] >> PMH_ROW_BITS][ma[x] & PMH_ROW_MASK]
                             : 0;
        }
        uint32_t last = ma[0];
        uint64_t vlast = MASKS ? vma[0] : 0;
#pragma unroll
        for (int x = 1; x < 4; x++)
            if (x < gn) {
                last = ma[x];
                if (MASKS) vlast = vma[x];
            }
        if (gn > 4) {
            last = mem[me - 1];
            if (MASKS) vlast = run_masks[last >> PMH_ROW_BITS][last & PMH_ROW_MASK];
        }
        const int lrun = last >> PMH_ROW_BITS;
        const int64_t lrow = last & PMH_ROW_MASK;
        // singleton bypass kind, issued early to overlap (see k_emit_pu)
        int32_t single_kind = 0;
        if (gn == 1)
            single_kind = col_load<int32_t>(
                cols[lrun * n_cols + kind_col], lrow);
        // ascending index of the LAST DELETE member (-1 = none)
        int d_del = -1;
        uint32_t mdel = 0;
        uint64_t vdel = 0;
        if (rrod) {
#pragma unroll
            for (int x = 0; x < 4; x++) {
                if (x >= gn) continue;
                const DevCol &dc = cols[(ma[x] >> PMH_ROW_BITS) * n_cols + kind_col];
                if (col_load<int32_t>(dc, ma[x] & PMH_ROW_MASK) == 3) d_del = x;
            }
            for (int x = 4; x < gn; x++) {
                uint32_t m = mem[ms + x];
                const DevCol &dc = cols[(m >> PMH_ROW_BITS) * n_cols + kind_col];
                if (col_load<int32_t>(dc, m & PMH_ROW_MASK) == 3) d_del = x;
            }
            if (d_del >= 0) {
                mdel = mem[ms + d_del];
                if (MASKS) vdel = run_masks[mdel >> PMH_ROW_BITS][mdel & PMH_ROW_MASK];
            }
        }
        // retract mode (flags bit 64): per-member kinds drive
        // aggregator.retract; only reached when every column's aggregator
        // is retract-capable (plan validation)
        const bool aggr = (flags & 64) != 0;
        for (int c = 0; c < n_cols; c++) {
            if (c == kind_col) {
                // wrapper bypass: singletons keep their own RowKind
                ((int8_t *)out_ptrs[c])[i] = (int8_t)(
                    gn == 1 ? single_kind
                            : ((rrod && d_del == gn - 1) ? 3 : 0));
                continue;
            }
            const int dt = col_dtype[c];
            const uint8_t rawagg = col_agg[c];
            const bool ign_retract =
                (rawagg & PMH_AGG_IGNORE_RETRACT) != 0;
            int agg = rawagg & 0x7f;
            if (agg == PMH_AGG_PRIMARY_KEY)
                agg = PMH_AGG_LAST_VALUE;  // agg == retract == input
            if (gn == 1) agg = PMH_AGG_LAST_VALUE;
            auto del_valid = [&]() -> uint8_t {
                if (!col_nullable[c]) return 1;
                if (MASKS) return (uint8_t)((vdel >> c) & 1);
                const DevCol &dc = cols[(mdel >> PMH_ROW_BITS) * n_cols + c];
                return dc.valid0
                           ? ((const uint8_t *)dc.valid0)[mdel & PMH_ROW_MASK]
                           : 1;
            };
            int64_t run = lrun, row = lrow;
            uint8_t ok = 1;
            int64_t bits = 0;  // result payload (raw stored bits / int value)
            bool direct = true;  // bits not set; load from (run,row) at store
            bool handled = false;
            if (aggr && ign_retract && gn > 1 &&
                (agg == PMH_AGG_LAST_VALUE || agg == PMH_AGG_FIRST_VALUE ||
                 agg == PMH_AGG_LAST_NON_NULL ||
                 agg == PMH_AGG_FIRST_NON_NULL)) {
                // FieldIgnoreRetractAgg wrapper: retract members leave the
                // accumulator untouched, so only ADD members qualify
                auto is_rt = [&](uint32_t m) -> bool {
                    const DevCol &dc =
                        cols[(m >> PMH_ROW_BITS) * n_cols + kind_col];
                    const int32_t kd =
                        col_load<int32_t>(dc, m & PMH_ROW_MASK);
                    return kd == 1 || kd == 3;
                };
                const bool want_first = agg == PMH_AGG_FIRST_VALUE ||
                                        agg == PMH_AGG_FIRST_NON_NULL;
                const bool need_nn = agg == PMH_AGG_LAST_NON_NULL ||
                                     agg == PMH_AGG_FIRST_NON_NULL;
                ok = 0;
                for (int32_t x = 0; x < gn; x++) {
                    uint32_t m = mem[ms + x];
                    if (is_rt(m)) continue;
                    uint8_t v = col_nullable[c] ? valid_of(m, c) : 1;
                    if (need_nn && !v) continue;
                    run = m >> PMH_ROW_BITS;
                    row = m & PMH_ROW_MASK;
                    ok = need_nn ? 1 : v;
                    if (want_first) break;
                }
                handled = true;
            }
            if (!handled) switch (agg) {
            case PMH_AGG_LAST_VALUE:
                // last member as-is, its own validity (also the singleton
                // ReducerMergeFunctionWrapper bypass)
                if (col_nullable[c])
                    ok = MASKS ? (uint8_t)((vlast >> c) & 1)
                               : valid_of(last, c);
                break;
            case PMH_AGG_FIRST_VALUE: {
                run = ma[0] >> PMH_ROW_BITS;
                row = ma[0] & PMH_ROW_MASK;
                if (col_nullable[c])
                    ok = MASKS ? (uint8_t)((vma[0] >> c) & 1)
                               : valid_of(ma[0], c);
                break;
            }
            case PMH_AGG_LAST_NON_NULL:
                if (col_nullable[c] || d_del >= 0) {
                    ok = 0;
                    // newest first: uncached tail (x >= 4), then cached;
                    // only members NEWER than the last DELETE participate
                    for (int32_t x = gn - 1; !ok && x >= 4; x--) {
                        if (x <= d_del) break;
                        uint32_t m = mem[ms + x];
                        if (valid_of(m, c)) {
                            run = m >> PMH_ROW_BITS;
                            row = m & PMH_ROW_MASK;
                            ok = 1;
                        }
                    }
#pragma unroll
                    for (int x = 3; x >= 0; x--) {
                        if (ok || x >= gn || x <= d_del) continue;
                        uint8_t v = MASKS ? (uint8_t)((vma[x] >> c) & 1)
                                          : valid_of(ma[x], c);
                        if (v) {
                            run = ma[x] >> PMH_ROW_BITS;
                            row = ma[x] & PMH_ROW_MASK;
                            ok = 1;
                        }
                    }
                    if (!ok && d_del >= 0 && del_valid()) {
                        run = mdel >> PMH_ROW_BITS;  // initRow: the DELETE's field
                        row = mdel & PMH_ROW_MASK;
                        ok = 1;
                    }
                }
                break;
            case PMH_AGG_FIRST_NON_NULL:
                if (col_nullable[c]) {
                    ok = 0;
#pragma unroll
                    for (int x = 0; x < 4; x++) {
                        if (ok || x >= gn) continue;
                        uint8_t v = MASKS ? (uint8_t)((vma[x] >> c) & 1)
                                          : valid_of(ma[x], c);
                        if (v) {
                            run = ma[x] >> PMH_ROW_BITS;
                            row = ma[x] & PMH_ROW_MASK;
                            ok = 1;
                        }
                    }
                    for (int32_t x = 4; !ok && x < gn; x++) {
                        uint32_t m = mem[ms + x];
                        if (valid_of(m, c)) {
                            run = m >> PMH_ROW_BITS;
                            row = m & PMH_ROW_MASK;
                            ok = 1;
                        }
                    }
                } else {
                    run = ma[0] >> PMH_ROW_BITS;
                    row = ma[0] & PMH_ROW_MASK;
                }
                break;
            default: {  // SUM / MAX / MIN: full fold, null inputs skipped
                direct = false;
                ok = 0;
                int64_t iacc = 0;
                float facc = 0.f;
                double dacc = 0.0;
                auto fold_one = [&](uint32_t m, bool retract) {
                    const DevCol &dc = cols[(m >> PMH_ROW_BITS) * n_cols + c];
                    const int64_t r = m & PMH_ROW_MASK;
                    int64_t vb = (dt == 4 || dt == 6)
                                     ? col_load<int64_t>(dc, r)
                                     : (int64_t)col_load<int32_t>(dc, r);
                    if (!ok) {
                        ok = 1;
                        if (agg == PMH_AGG_SUM && dt == 5) {
                            facc = __int_as_float((int32_t)vb);
                            if (retract) facc = -facc;  // null acc: negate
                        } else if (agg == PMH_AGG_SUM && dt == 6) {
                            dacc = __longlong_as_double(vb);
                            if (retract) dacc = -dacc;
                        } else {
                            iacc = retract && agg == PMH_AGG_SUM ? -vb : vb;
                        }
                        return;
                    }
                    if (agg == PMH_AGG_SUM) {
                        // FieldSumAgg.agg / .retract (:60-110): subtract on
                        // retract, same width/precision rules
                        const double sgn = retract ? -1.0 : 1.0;
                        if (dt == 5)
                            facc += (float)sgn * __int_as_float((int32_t)vb);
                        else if (dt == 6)
                            dacc += sgn * __longlong_as_double(vb);
                        else
                            iacc += retract ? -vb : vb;
                    } else {
                        bool take;
                        if (dt == 5) {
                            uint32_t a = f32_ord((int32_t)iacc);
                            uint32_t b = f32_ord((int32_t)vb);
                            take = agg == PMH_AGG_MAX ? b > a : b < a;
                        } else if (dt == 6) {
                            uint64_t a = f64_ord(iacc);
                            uint64_t b = f64_ord(vb);
                            take = agg == PMH_AGG_MAX ? b > a : b < a;
                        } else {
                            take = agg == PMH_AGG_MAX ? vb > iacc : vb < iacc;
                        }
                        if (take) iacc = vb;
                    }
                };
                auto is_retract = [&](uint32_t m) -> bool {
                    if (!aggr) return false;
                    const DevCol &dc =
                        cols[(m >> PMH_ROW_BITS) * n_cols + kind_col];
                    const int32_t kd =
                        col_load<int32_t>(dc, m & PMH_ROW_MASK);
                    return kd == 1 || kd == 3;
                };
                if (d_del >= 0 && del_valid())
                    fold_one(mdel, false);  // initRow continuation
#pragma unroll
                for (int x = 0; x < 4; x++) {
                    if (x >= gn || x <= d_del) continue;
                    if (col_nullable[c] &&
                        !(MASKS ? (uint8_t)((vma[x] >> c) & 1)
                                : valid_of(ma[x], c)))
                        continue;
                    const bool rt = is_retract(ma[x]);
                    if (rt && ign_retract) continue;  // FieldIgnoreRetractAgg
                    fold_one(ma[x], rt);
                }
                for (int32_t x = 4; x < gn; x++) {
                    if (x <= d_del) continue;
                    uint32_t m = mem[ms + x];
                    if (col_nullable[c] && !valid_of(m, c)) continue;
                    const bool rt = is_retract(m);
                    if (rt && ign_retract) continue;
                    fold_one(m, rt);
                }
                if (agg == PMH_AGG_SUM && dt == 5)
                    bits = (int64_t)__float_as_int(facc);
                else if (agg == PMH_AGG_SUM && dt == 6)
                    bits = __double_as_longlong(dacc);
                else
                    bits = iacc;
                break;
            }
            }
            if (direct) {
                const DevCol &dc = cols[run * n_cols + c];
                if (ok)
                    bits = (dt == 4 || dt == 6)
                               ? col_load<int64_t>(dc, row)
                               : (int64_t)col_load<int32_t>(dc, row);
            }
            if (!ok) bits = 0;
            switch (dt) {
            case 1: ((int8_t *)out_ptrs[c])[i] = (int8_t)bits; break;
            case 2: ((int16_t *)out_ptrs[c])[i] = (int16_t)bits; break;
            case 3:
            case 5:
            case 7: ((int32_t *)out_ptrs[c])[i] = (int32_t)bits; break;
            case 4:
            case 6: ((int64_t *)out_ptrs[c])[i] = bits; break;
            default: break;
            }
            if (out_valid[c]) out_valid[c][i] = ok;
        }
    }
}

// --------------------------------------------------------- k_level_scatter
//
// Decode def-level streams (RLE/bit-packed hybrid, bit width 1) and position
// the dense PLAIN values: valid[row] = level, out[row] = dense[aux + prefix]
// (the null handling of VectorizedColumnReader.readBatch,
// paimon-format/.../reader/VectorizedColumnReader.java:143-241).
__global__ void k_level_scatter(const RleChunk *chunks, int64_t n_chunks) {
    // one WAVE per chunk, barrier-free: the host prescan bounds packed
    // chunks at <= 512 values (parquet) / <= 1040 (ORC byte-RLE groups) and
    // precomputes each chunk's dense prefix (aux), so the only running state
    // is a per-wave register popcount. The previous one-WORKGROUP-per-chunk
    // version chained 3 __syncthreads per 512 values through an LDS running
    // scan and was serialization-bound (62 ms at the 8x10M PU shape; 94 GB/s
    // effective).
    const int lane = (int)(threadIdx.x & 63);
    const int64_t wave_id =
        (int64_t)blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
    const int64_t n_waves = (int64_t)gridDim.x * (blockDim.x >> 6);
    for (int64_t cidx = wave_id; cidx < n_chunks; cidx += n_waves) {
        RleChunk ch = chunks[cidx];
        const uint8_t *dense = (const uint8_t *)ch.dense_addr;
        uint8_t *out = (uint8_t *)ch.out_addr;
        uint8_t *valid = (uint8_t *)ch.valid_addr;
        const int esize = ch.esize;
        if (ch.kind == 0) {
            if (ch.value) {  // run of non-nulls: dense block copy
                for (int32_t i = lane; i < ch.count; i += 64) {
                    valid[ch.out_start + i] = 1;
                    const uint8_t *s = dense + (ch.aux + i) * esize;
                    uint8_t *d = out + (ch.out_start + i) * esize;
                    if (esize == 4)
                        *(int32_t *)d = *(const int32_t *)s;
                    else
                        *(int64_t *)d = *(const int64_t *)s;
                }
            } else {  // run of nulls
                for (int32_t i = lane; i < ch.count; i += 64) {
                    valid[ch.out_start + i] = 0;
                    uint8_t *d = out + (ch.out_start + i) * esize;
                    if (esize == 4)
                        *(int32_t *)d = 0;
                    else
                        *(int64_t *)d = 0;
                }
            }
            continue;
        }
        // bit-packed: 1 byte per 8 levels; dense index = aux + running
        // popcount (registers only) + lower-lane ballot prefix
        const uint8_t *src = (const uint8_t *)ch.src;
        int32_t running = 0;
        for (int32_t b = 0; b < ch.count; b += 64) {
            int32_t i = b + lane;
            int bit = 0;
            if (i < ch.count) bit = (src[i >> 3] >> (i & 7)) & 1;
            uint64_t mask = __ballot(bit != 0);
            int32_t my_before =
                lane == 0 ? 0 : __popcll(mask << (64 - lane));
            if (i < ch.count) {
                valid[ch.out_start + i] = (uint8_t)bit;
                uint8_t *dp = out + (ch.out_start + i) * esize;
                if (bit) {
                    int64_t didx = ch.aux + running + my_before;
                    const uint8_t *sp = dense + didx * esize;
                    if (esize == 4)
                        *(int32_t *)dp = *(const int32_t *)sp;
                    else
                        *(int64_t *)dp = *(const int64_t *)sp;
                } else {
                    if (esize == 4)
                        *(int32_t *)dp = 0;
                    else
                        *(int64_t *)dp = 0;
                }
            }
            running += __popcll(mask);
        }
    }
}

// ------------------------------------------------------------- k_rlev2
//
// ORC RLEv2 / byte-RLE decode from host-prescanned run chunks (ORC v1 spec;
// the reference consumes this via orc-core 1.9.8 — parity pinned by the
// oracle restatement + pyarrow). One 64-lane wave per chunk
// (runs are <= 512 values); bit order is big-endian MSB-first.

DEV uint64_t be_bits(const uint8_t *src, int64_t bit_off, int width) {
    // read `width` (<= 64) bits starting at bit_off, MSB-first: two aligned
    // 8-byte loads + shift-combine (the byte-at-a-time loop cost ~5 loads
    // per value). Stream uploads are padded by 16 bytes for the window.
    uint64_t addr = (uint64_t)src + (uint64_t)(bit_off >> 3);
    uint64_t base = addr & ~7ull;  // absolute 8-byte alignment
    int off = (int)((addr - base) * 8) + (int)(bit_off & 7);  // 0..63
    uint64_t hi =
        __builtin_bswap64(*reinterpret_cast<const uint64_t *>(base));
    uint64_t lo =
        __builtin_bswap64(*reinterpret_cast<const uint64_t *>(base + 8));
    uint64_t window = off ? ((hi << off) | (lo >> (64 - off))) : hi;
    return width == 64 ? window : window >> (64 - width);
}

DEV int64_t zz_dec(uint64_t v) {
    return (int64_t)(v >> 1) ^ -(int64_t)(v & 1);
}

DEV void dense_store(void *out, int esize, int64_t idx, int64_t v) {
    if (esize == 4)
        ((int32_t *)out)[idx] = (int32_t)v;
    else
        ((int64_t *)out)[idx] = v;
}

__global__ void k_rlev2(const Rlev2Chunk *chunks, int64_t n_chunks) {
    const int wave = (int)(threadIdx.x >> 6);
    const int lane = (int)(threadIdx.x & 63);
    const int waves = (int)(blockDim.x >> 6);
    for (int64_t cidx = (int64_t)blockIdx.x * waves + wave; cidx < n_chunks;
         cidx += (int64_t)gridDim.x * waves) {
        Rlev2Chunk ch = chunks[cidx];
        void *out = (void *)ch.out_addr;
        const uint8_t *src = (const uint8_t *)ch.src;
        switch (ch.kind) {
        case 0:  // SHORT_REPEAT (value already sign-decoded by the host)
        case 4:  // BYTE_RUN
            for (int i = lane; i < ch.count; i += 64)
                dense_store(out, ch.out_esize, ch.out_start + i, ch.base);
            break;
        case 5:  // BYTE_LITERAL (ORC tinyint byte stream, sign-extended)
            for (int i = lane; i < ch.count; i += 64)
                dense_store(out, ch.out_esize, ch.out_start + i,
                            (int8_t)src[i]);
            break;
        case 1: {  // DIRECT
            for (int i = lane; i < ch.count; i += 64) {
                uint64_t raw = be_bits(src, (int64_t)i * ch.width, ch.width);
                int64_t v = ch.is_signed ? zz_dec(raw) : (int64_t)raw;
                dense_store(out, ch.out_esize, ch.out_start + i, v);
            }
            break;
        }
        case 2: {  // PATCHED_BASE: unsigned packed + base; patches on lane 0
            for (int i = lane; i < ch.count; i += 64) {
                uint64_t raw = be_bits(src, (int64_t)i * ch.width, ch.width);
                dense_store(out, ch.out_esize, ch.out_start + i,
                            ch.base + (int64_t)raw);
            }
            __builtin_amdgcn_s_waitcnt(0);  // drain this wave's stores
            if (lane == 0) {
                const uint8_t *psrc = (const uint8_t *)ch.patch_src;
                uint64_t pmask = ch.patch_pw >= 64
                                     ? ~0ull
                                     : ((1ull << ch.patch_pw) - 1);
                int64_t pos = 0, gap = 0;
                int started = 0;
                for (int pidx = 0; pidx < ch.patch_pl; pidx++) {
                    uint64_t e = be_bits(psrc, (int64_t)pidx * ch.patch_cfb,
                                         ch.patch_cfb);
                    uint64_t g = e >> ch.patch_pw;
                    uint64_t pv = e & pmask;
                    gap += (int64_t)g;
                    if (pv == 0 && g == ((1ull << ch.patch_pgw) - 1))
                        continue;
                    pos = started ? pos + gap : gap;
                    started = 1;
                    gap = 0;
                    if (pos >= 0 && pos < ch.count) {
                        uint64_t raw =
                            be_bits(src, pos * (int64_t)ch.width, ch.width) |
                            (pv << ch.width);
                        dense_store(out, ch.out_esize, ch.out_start + pos,
                                    ch.base + (int64_t)raw);
                    }
                }
            }
            break;
        }
        case 3: {  // DELTA
            if (ch.width == 0) {  // fixed delta
                for (int i = lane; i < ch.count; i += 64)
                    dense_store(out, ch.out_esize, ch.out_start + i,
                                ch.base + (int64_t)i * ch.delta);
            } else {
                // packed |deltas| for elements 2..count-1; direction =
                // sign(delta). Wave-parallel prefix over 64-lane rounds.
                if (lane == 0) {
                    dense_store(out, ch.out_esize, ch.out_start, ch.base);
                    if (ch.count > 1)
                        dense_store(out, ch.out_esize, ch.out_start + 1,
                                    ch.base + ch.delta);
                }
                int64_t run = ch.base + ch.delta;  // value at index 1
                int sign = ch.delta < 0 ? -1 : 1;
                for (int b = 0; b < ch.count - 2; b += 64) {
                    int i = b + lane;  // delta index (element 2 + i)
                    int64_t d = 0;
                    if (i < ch.count - 2)
                        d = (int64_t)be_bits(src, (int64_t)i * ch.width,
                                             ch.width);
                    // inclusive wave scan of deltas
                    int64_t acc = d;
                    for (int off = 1; off < 64; off <<= 1) {
                        int64_t up = __shfl_up(acc, off, 64);
                        if (lane >= off) acc += up;
                    }
                    if (i < ch.count - 2)
                        dense_store(out, ch.out_esize, ch.out_start + 2 + i,
                                    run + sign * acc);
                    run += sign * __shfl(acc, 63, 64);
                }
            }
            break;
        }
        default: break;
        }
    }
}


// ------------------------------------------------------------ DELTA decode
//
// Parquet DELTA_BINARY_PACKED (VectorizedDeltaBinaryPackedReader.java /
// parquet-format Encodings.md): header <block><miniblocks><count><first>,
// then blocks of [min_delta zigzag][miniblock bit widths][LSB-first packed
// deltas]. value_i = first + sum of deltas. Three batched phases: per-block
// sums (one wave per block), per-page base scan (pages are self-contained
// streams; one workgroup per page), then unpack + wave-scan + base add.

DEV uint64_t le_bits(const uint8_t *src, int64_t bit_off, int width) {
    // LSB-first window like the RLE bit-packed decoder, any width <= 64:
    // two aligned 8-byte loads + shift-combine (uploads padded by 16 B)
    uint64_t addr = (uint64_t)src + (uint64_t)(bit_off >> 3);
    uint64_t base = addr & ~7ull;
    int off = (int)((addr - base) * 8) + (int)(bit_off & 7);
    uint64_t lo = *reinterpret_cast<const uint64_t *>(base);
    uint64_t hi = *reinterpret_cast<const uint64_t *>(base + 8);
    uint64_t window = off ? ((lo >> off) | (hi << (64 - off))) : lo;
    return width == 64 ? window : window & ((1ull << width) - 1);
}

DEV int64_t delta_at(const DeltaChunk &ch, int d) {
    // miniblock of delta d and its packed offset
    const int j = d / ch.vpm;
    int64_t byte_off = 0;
    for (int q = 0; q < j; q++)
        byte_off += (int64_t)ch.vpm * ((ch.widths >> (8 * q)) & 0xff) / 8;
    const int w = (int)((ch.widths >> (8 * j)) & 0xff);
    if (w == 0) return ch.min_delta;
    const uint64_t raw = le_bits((const uint8_t *)ch.src,
                                 (int64_t)(d % ch.vpm) * w + byte_off * 8,
                                 w);
    return ch.min_delta + (int64_t)raw;
}

__global__ void k_delta_sum(const DeltaChunk *chunks, int64_t n_chunks,
                            int64_t *sums) {
    const int lane = (int)(threadIdx.x & 63);
    const int64_t wave =
        (int64_t)blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
    const int64_t n_waves = (int64_t)gridDim.x * (blockDim.x >> 6);
    for (int64_t c = wave; c < n_chunks; c += n_waves) {
        const DeltaChunk ch = chunks[c];
        int64_t acc = 0;
        for (int d = lane; d < ch.count; d += 64) acc += delta_at(ch, d);
        for (int off = 32; off; off >>= 1)
            acc += __shfl_down(acc, off, 64);
        if (lane == 0) sums[c] = acc;
    }
}

__global__ void k_delta_scan(const DeltaStream *streams, int64_t n_streams,
                             const int64_t *sums, int64_t *bases) {
    // one workgroup per stream (page): running exclusive scan over its
    // blocks' sums, seeded by the page's first value; also stores the
    // page's first value itself
    __shared__ int64_t s[256];
    __shared__ int64_t running;
    for (int64_t si = blockIdx.x; si < n_streams; si += gridDim.x) {
        const DeltaStream st = streams[si];
        if (threadIdx.x == 0) {
            running = st.first;
            if (st.out_esize == 8)
                ((int64_t *)st.out_addr)[st.out0] = st.first;
            else
                ((int32_t *)st.out_addr)[st.out0] = (int32_t)st.first;
        }
        __syncthreads();
        for (int64_t b = st.chunk_lo; b < st.chunk_hi; b += blockDim.x) {
            int64_t i = b + threadIdx.x;
            int64_t v = i < st.chunk_hi ? sums[i] : 0;
            s[threadIdx.x] = v;
            __syncthreads();
            for (int d = 1; d < (int)blockDim.x; d <<= 1) {
                int64_t add = threadIdx.x >= d ? s[threadIdx.x - d] : 0;
                __syncthreads();
                s[threadIdx.x] += add;
                __syncthreads();
            }
            if (i < st.chunk_hi) bases[i] = running + s[threadIdx.x] - v;
            __syncthreads();
            if (threadIdx.x == 0) running += s[blockDim.x - 1];
            __syncthreads();
        }
    }
}

__global__ void k_delta_emit(const DeltaChunk *chunks, int64_t n_chunks,
                             const int64_t *bases) {
    const int lane = (int)(threadIdx.x & 63);
    const int64_t wave =
        (int64_t)blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
    const int64_t n_waves = (int64_t)gridDim.x * (blockDim.x >> 6);
    for (int64_t c = wave; c < n_chunks; c += n_waves) {
        const DeltaChunk ch = chunks[c];
        int64_t run = bases[c];
        for (int b = 0; b < ch.count; b += 64) {
            const int d = b + lane;
            int64_t x = d < ch.count ? delta_at(ch, d) : 0;
            // inclusive wave scan
            int64_t acc = x;
            for (int off = 1; off < 64; off <<= 1) {
                int64_t up = __shfl_up(acc, off, 64);
                if (lane >= off) acc += up;
            }
            if (d < ch.count) {
                if (ch.out_esize == 8)
                    ((int64_t *)ch.out_addr)[ch.out_start + d] = run + acc;
                else
                    ((int32_t *)ch.out_addr)[ch.out_start + d] =
                        (int32_t)(run + acc);
            }
            run += __shfl(acc, 63, 64);
        }
    }
}

// ------------------------------------------------------------ k_rle_decode
// Decode Parquet RLE/bit-packed hybrid streams (dictionary ids, def levels)
// from host-prescanned run chunks (VectorizedRleValuesReader.java:977-1018
// wire format; the sequential varint-header walk happens on the host at
// staging time, the bulk expansion here).
__global__ void k_rle_decode(const RleChunk *chunks, int64_t n_chunks,
                             int32_t *out) {
    for (int64_t cidx = blockIdx.x; cidx < n_chunks; cidx += gridDim.x) {
        RleChunk ch = chunks[cidx];
        int bit_width = ch.bit_width;
        uint32_t mask =
            bit_width >= 32 ? 0xffffffffu : ((1u << bit_width) - 1u);
        if (ch.kind == 0) {
            for (int32_t i = threadIdx.x; i < ch.count; i += blockDim.x)
                out[ch.out_start + i] = (int32_t)ch.value;
        } else {
            const uint8_t *src = (const uint8_t *)ch.src;
            for (int32_t i = threadIdx.x; i < ch.count; i += blockDim.x) {
                int64_t g = i >> 3;
                int lane = i & 7;
                const uint8_t *gp = src + g * bit_width;
                int64_t bit_off = (int64_t)lane * bit_width;
                int64_t byte_off = bit_off >> 3;
                int shift = (int)(bit_off & 7);
                uint64_t word = 0;
                for (int b = 0; b < 5 && byte_off + b < bit_width; b++)
                    word |= (uint64_t)gp[byte_off + b] << (8 * b);
                out[ch.out_start + i] = (int32_t)((word >> shift) & mask);
            }
        }
    }
}

// ------------------------------------------------------------ k_dict_gather
template <typename T>
__global__ void k_dict_gather_t(const int32_t *ids, const T *dict, int64_t n,
                                T *out) {
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) out[i] = dict[ids[i]];
}

// endpoint seed for the single-level (k > 8) partition: write cuts for
// b = 0 (all zeros) and b = n_bounds-1 (run lengths) so k_partition_refine
// can treat [0, lens) as one full-width coarse window.
__global__ void k_partition_seed(const int64_t *lens, int k,
                                 int64_t n_bounds, int32_t *cuts) {
    int r = threadIdx.x;
    if (r >= k) return;
    cuts[r] = 0;
    cuts[(n_bounds - 1) * k + r] = (int32_t)lens[r];
}



// --------------------------------------------------------- k_zstd_compress
//
// On-GPU zstd page COMPRESSION (the write side): one wave
// per page, lane-0 serial v0 over the shared scalar encoder (greedy LZ +
// RAW literals + predefined-FSE sequences — spec-valid frames, proven
// against libzstd on the host). Page-level parallelism carries the
// throughput exactly like k_zstd_pages. Scratch: one PzEnc per job.
__global__ void k_zstd_compress(const uint8_t *src, const ZstdJob *jobs,
                                int n, uint8_t *dst, uint8_t *scratch,
                                int64_t *status) {
    // job = one 128 KB zstd BLOCK (matches are block-confined, so blocks
    // compress independently — the host stitches them into frames);
    // dst_len bit 31 carries the last-block flag
    __shared__ int32_t lhtab[4][1 << PZ_ENC_HLOG];  // 16 KB per wave
    const int wave = (int)((blockIdx.x * blockDim.x + threadIdx.x) >> 6);
    const int wiw = (threadIdx.x >> 6) & 3;
    const int lane = threadIdx.x & 63;
    const int waves = (int)((gridDim.x * blockDim.x) >> 6);
    for (int j = wave; j < n; j += waves) {
        if (lane != 0) continue;
        PzEnc *e = (PzEnc *)(scratch + (size_t)j * sizeof(PzEnc));
        e->htab = lhtab[wiw];
        int last = (int)(jobs[j].dst_len >> 31);
        int64_t cap = jobs[j].dst_len & 0x7FFFFFFF;
        status[j] = pz_encode_block(src + jobs[j].src_off, jobs[j].src_len,
                                    last, dst + jobs[j].dst_off, cap, e);
    }
}

// ------------------------------------------------------------- k_filter
//
// Value-filter pushdown for SINGLE-RUN sections (MergeFileSplitRead.java:
// 227-239: each key appears once, so dropping rows is safe; overlapping
// sections get no value filters). Failing rows mark the run's tombstone
// byte — the same "never reaches the merge" mechanism deletion vectors
// use. Comparison semantics follow LeafPredicate: a NULL field fails
// every comparison; is_null/is_not_null test validity.
__global__ void k_filter(const DevCol *cols, int n_cols,
                         const FilterTerm *terms, int n_terms, int64_t rows,
                         uint8_t *tomb) {
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         i < rows; i += stride) {
        bool pass = true;
        for (int t = 0; t < n_terms && pass; t++) {
            const FilterTerm &ft = terms[t];
            const DevCol &dc = cols[ft.col];
            bool valid =
                !dc.valid0 || ((const uint8_t *)dc.valid0)[i] != 0;
            if (ft.op == 6) {
                pass = !valid;
                continue;
            }
            if (ft.op == 7) {
                pass = valid;
                continue;
            }
            if (!valid) {
                pass = false;  // NULL fails comparisons
                continue;
            }
            int cmp;
            if (ft.is_fp) {
                double v = dc.esize == 8
                               ? *(const double *)(dc.addr0 + i * 8)
                               : (double)*(const float *)(dc.addr0 + i * 4);
                cmp = v < ft.dlit ? -1 : (v > ft.dlit ? 1 : 0);
            } else {
                int64_t v = dc.esize == 8 ? col_load<int64_t>(dc, i)
                                          : (int64_t)col_load<int32_t>(dc, i);
                cmp = v < ft.ilit ? -1 : (v > ft.ilit ? 1 : 0);
            }
            switch (ft.op) {
            case 0: pass = cmp == 0; break;
            case 1: pass = cmp != 0; break;
            case 2: pass = cmp < 0; break;
            case 3: pass = cmp <= 0; break;
            case 4: pass = cmp > 0; break;
            case 5: pass = cmp >= 0; break;
            default: pass = false; break;
            }
        }
        if (!pass) tomb[i] = 1;
    }
}

// ------------------------------------------------------------ k_cl_finalize
//
// Compact the provisional changelog entries into dense per-tile rows.
// Pending UPDATE_BEFORE/UPDATE_AFTER pairs (changelog-producer.
// row-deduplicate) compare the two records' value columns (the reference's
// RecordEqualiser over the read schema) and are dropped when equal.
// Writes the compacted rows into cl_out (same per-tile stride) and
// rewrites cl_counts[tile] from group count to ROW count.
__global__ void k_cl_finalize(const DevCol *cols, const uint8_t *col_dtype,
                              int n_cols, int first_val, int k,
                              const uint64_t *cl_entries, uint64_t *cl_out,
                              int32_t *cl_counts, int64_t n_tiles,
                              int64_t tile_rows) {
    __shared__ int32_t scan[257];
    const int64_t stride2 = 2 * (tile_rows + PMH_MAX_RUNS);
    for (int64_t tile = blockIdx.x; tile < n_tiles; tile += gridDim.x) {
        const uint64_t *in = &cl_entries[tile * stride2];
        uint64_t *out = &cl_out[tile * stride2];
        int32_t ng = cl_counts[tile];
        const int tid = threadIdx.x;
        const int32_t per = (ng + (int32_t)blockDim.x - 1) / blockDim.x;
        int32_t lo = tid * per;
        int32_t hi = lo + per < ng ? lo + per : ng;
        // verdict per group: 2 bits (number of surviving rows, 0/1/2)
        uint32_t verd[8] = {0, 0, 0, 0, 0, 0, 0, 0};  // per <= 128 groups
        int32_t nrows = 0;
        for (int32_t g = lo; g < hi; g++) {
            uint64_t e0 = in[2 * g], e1 = in[2 * g + 1];
            int nv = ((e0 >> 35) & 1) + ((e1 >> 35) & 1);
            if ((e0 >> 36) & 1) {  // pending pair: value-equality check
                uint32_t a = (uint32_t)e0, b = (uint32_t)e1;
                int ra = a >> PMH_ROW_BITS, rb = b >> PMH_ROW_BITS;
                int64_t xa = a & PMH_ROW_MASK, xb = b & PMH_ROW_MASK;
                bool eq = true;
                for (int c = first_val; eq && c < n_cols; c++) {
                    const DevCol &da = cols[ra * n_cols + c];
                    const DevCol &db = cols[rb * n_cols + c];
                    uint8_t va = da.valid0
                                     ? ((const uint8_t *)da.valid0)[xa]
                                     : 1;
                    uint8_t vb = db.valid0
                                     ? ((const uint8_t *)db.valid0)[xb]
                                     : 1;
                    if (va != vb) {
                        eq = false;
                    } else if (va) {
                        if (da.esize == 8
                                ? (col_load<int64_t>(da, xa) !=
                                   col_load<int64_t>(db, xb))
                                : (col_load<int32_t>(da, xa) !=
                                   col_load<int32_t>(db, xb)))
                            eq = false;
                    }
                }
                if (eq) nv = 0;
            }
            int li = g - lo;
            verd[li >> 4] |= (uint32_t)nv << (2 * (li & 15));
            nrows += nv;
        }
        // block exclusive scan of nrows (group order preserved)
        scan[tid + 1] = nrows;
        if (tid == 0) scan[0] = 0;
        __syncthreads();
        for (int off = 1; off < (int)blockDim.x; off <<= 1) {
            int32_t v = tid + 1 > off ? scan[tid + 1 - off] : 0;
            __syncthreads();
            scan[tid + 1] += v;
            __syncthreads();
        }
        int32_t o = scan[tid];
        for (int32_t g = lo; g < hi; g++) {
            int li = g - lo;
            int nv = (verd[li >> 4] >> (2 * (li & 15))) & 3;
            if (nv >= 1) out[o++] = in[2 * g];
            if (nv == 2) out[o++] = in[2 * g + 1];
        }
        __syncthreads();
        if (tid == (int)blockDim.x - 1) cl_counts[tile] = o;
        __syncthreads();
    }
}

// ---------------------------------------------------------------- k_cl_emit
//
// Gather the compacted changelog rows into the changelog output columns:
// key/seq/value columns come from the source record; the kind column is
// the changelog RowKind carried in the entry (INSERT / UPDATE_BEFORE /
// UPDATE_AFTER / DELETE).
__global__ void k_cl_emit(const DevCol *cols, const uint8_t *col_dtype,
                          const uint8_t *col_nullable, int n_cols,
                          int kind_col, const uint64_t *cl_rows,
                          const int32_t *cl_counts,
                          const int64_t *cl_offsets, int64_t n_tiles,
                          int64_t tile_rows, void *const *out_ptrs,
                          uint8_t *const *out_valid) {
    const int64_t stride2 = 2 * (tile_rows + PMH_MAX_RUNS);
    for (int64_t tile = blockIdx.x; tile < n_tiles; tile += gridDim.x) {
        const uint64_t *in = &cl_rows[tile * stride2];
        const int32_t n = cl_counts[tile];
        const int64_t ob = cl_offsets[tile];
        for (int32_t i = threadIdx.x; i < n; i += blockDim.x) {
            uint64_t e = in[i];
            uint32_t m = (uint32_t)e;
            int run = m >> PMH_ROW_BITS;
            int64_t row = m & PMH_ROW_MASK;
            int8_t kd = (int8_t)((e >> 32) & 7);
            int64_t o = ob + i;
            for (int c = 0; c < n_cols; c++) {
                const DevCol &dc = cols[run * n_cols + c];
                if (col_nullable[c] && out_valid[c])
                    out_valid[c][o] =
                        dc.valid0 ? ((const uint8_t *)dc.valid0)[row] : 1;
                if (c == kind_col) {
                    ((int8_t *)out_ptrs[c])[o] = kd;
                    continue;
                }
                switch (col_dtype[c]) {
                case 1:
                    ((int8_t *)out_ptrs[c])[o] =
                        (int8_t)col_load<int32_t>(dc, row);
                    break;
                case 2:
                    ((int16_t *)out_ptrs[c])[o] =
                        (int16_t)col_load<int32_t>(dc, row);
                    break;
                case 3:
                case 5:
                case 7:
                    ((int32_t *)out_ptrs[c])[o] = col_load<int32_t>(dc, row);
                    break;
                case 4:
                case 6:
                    ((int64_t *)out_ptrs[c])[o] = col_load<int64_t>(dc, row);
                    break;
                default:
                    break;
                }
            }
        }
        __syncthreads();
    }
}

// ------------------------------------------------------------ k_zstd_pages
//
// On-GPU zstd page decompression: one WAVEFRONT per parquet
// page frame. The scalar RFC 8878 core (zstd_core.h — fuzz-pinned against
// libzstd on the host) does all header/table parsing; the wave supplies the
// parallelism where the bytes move:
//  - raw/RLE blocks and raw/RLE literals: 64-lane copies/fills;
//  - huffman literals: the 4 interleaved streams decode on 4 lanes in SIMD
//    lockstep, each with a register 64-bit bit-window (one 8-byte refill
//    per ~7 symbols instead of 6 dependent byte loads per symbol), LUT in
//    LDS;
//  - sequence execution: 64-lane literal copies and match copies (overlap
//    handled with the period trick: byte j of a match reads
//    dst[start - off + j % off]), __threadfence_block between dependent
//    phases (same-CU L1 visibility);
//  - sequence DECODE stays serial on lane 0 (FSE state chain), buffered 64
//    sequences at a time through LDS.
// Scratch per job: 128 KB literals buffer + decode context.

// per-wave LDS workspace
struct ZWaveLds {
    PzHuf hlut[1 << PZ_HUF_LOG_MAX];  // 4 KB huffman LUT
    uint32_t ll[64], ml[64];          // sequence ring (lane 0 -> wave)
    uint64_t off[64];
    int32_t stat[4];                  // per-stream literal decode status
    int64_t flag;                     // lane-0 -> wave broadcast / error
};

// backward bit reader with a register window (device hot loops)
struct ZBack {
    const uint8_t *buf;
    int64_t bp;    // absolute bit position of the next read's top
    uint64_t win;  // bits [wlo, wlo+64)
    int64_t wlo;
    DEV void init(const uint8_t *b, int64_t sentinel_bp) {
        buf = b;
        bp = sentinel_bp;
        wlo = INT64_MIN / 2;
        win = 0;
    }
    DEV void refill(int64_t lo) {
        int64_t byte0 = lo >> 3;  // floor, works for negative lo
        wlo = byte0 << 3;
        uint64_t v = 0;
#pragma unroll
        for (int i = 0; i < 8; i++) {
            int64_t by = byte0 + i;
            v |= (uint64_t)(by >= 0 ? buf[by] : (uint8_t)0) << (8 * i);
        }
        win = v;
    }
    DEV uint32_t peek(int n) {  // bits [bp-n, bp), n <= 57
        int64_t lo = bp - n;
        if (lo < wlo || bp > wlo + 64) refill(lo);
        return (uint32_t)((win >> (lo - wlo)) &
                          ((n >= 32) ? 0xFFFFFFFFu : ((1u << n) - 1u)));
    }
    DEV uint32_t peek_abs(int n) {  // bits [bp, bp+n) — bp pre-decremented
        if (n == 0) return 0;
        if (bp < wlo || bp + n > wlo + 64) refill(bp);
        return (uint32_t)((win >> (bp - wlo)) &
                          ((n >= 32) ? 0xFFFFFFFFu : ((1u << n) - 1u)));
    }
};

// lane-0 sequence decode over the register bit-window (the scalar
// pz_seq_next re-gathers 6 bytes per field read; ZBack refills once per
// ~56 bits — the "next lever" for k_zstd_pages' match-heavy frames)
static DEV int zdev_seq_next(ZBack &rb, const PzFse *llT, const PzFse *ofT,
                             const PzFse *mlT, uint32_t &sll, uint32_t &sof,
                             uint32_t &sml, uint64_t rep[3], int last,
                             PzSeq *out) {
    int llc = llT[sll].sym;
    int ofc = ofT[sof].sym;
    int mlc = mlT[sml].sym;
    if (llc > 35 || mlc > 52 || ofc > 31) return PZ_ERR_SEQ;
    uint64_t ofv;
    if (ofc > 0) {
        rb.bp -= ofc;
        ofv = ((uint64_t)1 << ofc) + rb.peek_abs(ofc);
    } else {
        ofv = 1;
    }
    int mb = pz_ml_bits(mlc);
    rb.bp -= mb;
    uint32_t ml = pz_ml_base(mlc) + rb.peek_abs(mb);
    int lb = pz_ll_bits(llc);
    rb.bp -= lb;
    uint32_t ll = pz_ll_base(llc) + rb.peek_abs(lb);
    uint64_t off;
    if (ofv > 3) {
        off = ofv - 3;
        rep[2] = rep[1];
        rep[1] = rep[0];
        rep[0] = off;
    } else {
        int idx = (int)ofv - 1 + (ll == 0 ? 1 : 0);
        if (idx == 0) {
            off = rep[0];
        } else if (idx == 1) {
            off = rep[1];
            rep[1] = rep[0];
            rep[0] = off;
        } else if (idx == 2) {
            off = rep[2];
            rep[2] = rep[1];
            rep[1] = rep[0];
            rep[0] = off;
        } else {
            off = rep[0] - 1;
            if (off == 0) return PZ_ERR_OFFSET;
            rep[2] = rep[1];
            rep[1] = rep[0];
            rep[0] = off;
        }
    }
    out->ll = ll;
    out->ml = ml;
    out->off = off;
    if (!last) {
        rb.bp -= llT[sll].nbits;
        sll = llT[sll].base + rb.peek_abs(llT[sll].nbits);
        rb.bp -= mlT[sml].nbits;
        sml = mlT[sml].base + rb.peek_abs(mlT[sml].nbits);
        rb.bp -= ofT[sof].nbits;
        sof = ofT[sof].base + rb.peek_abs(ofT[sof].nbits);
        if (rb.bp < 0) return PZ_ERR_SEQ;
    }
    return 0;
}

static DEV void z_fence() { __threadfence_block(); }

// decode one frame with one wave. Returns decompressed size or PZ_ERR_*.
static DEV int64_t zdev_page(const uint8_t *src, int64_t slen, uint8_t *dst,
                             int64_t dcap, uint8_t *lit, PzCtx *cx,
                             ZWaveLds *L) {
    const int lane = threadIdx.x & 63;
    PzFrame F;
    int fh = pz_parse_frame(src, slen, &F);  // all lanes, redundant+uniform
    if (fh < 0) return fh;
    int64_t sp = fh, dp = 0;
    int htl = -1;
    PzSeqState st;  // lane 0 only (sequence states + repeat offsets)
    st.rep[0] = 1;
    st.rep[1] = 4;
    st.rep[2] = 8;
    if (lane == 0) {
        cx->htl = -1;
        cx->have_ll = cx->have_of = cx->have_ml = 0;
    }
    int last = 0;
    while (!last) {
        if (sp + 3 > slen) return PZ_ERR_SRC_SMALL;
        uint32_t bh = (uint32_t)src[sp] | ((uint32_t)src[sp + 1] << 8) |
                      ((uint32_t)src[sp + 2] << 16);
        sp += 3;
        last = bh & 1;
        int btype = (bh >> 1) & 3;
        int64_t bsize = bh >> 3;
        if (btype == 0) {  // raw block
            if (sp + bsize > slen || dp + bsize > dcap)
                return PZ_ERR_SRC_SMALL;
            for (int64_t j = lane; j < bsize; j += 64)
                dst[dp + j] = src[sp + j];
            z_fence();
            sp += bsize;
            dp += bsize;
            continue;
        }
        if (btype == 1) {  // RLE block
            if (sp + 1 > slen || dp + bsize > dcap) return PZ_ERR_SRC_SMALL;
            uint8_t v = src[sp];
            for (int64_t j = lane; j < bsize; j += 64) dst[dp + j] = v;
            z_fence();
            sp += 1;
            dp += bsize;
            continue;
        }
        if (btype != 2 || bsize > slen - sp) return PZ_ERR_BLOCK;
        const uint8_t *bb = src + sp;
        int64_t bn = bsize;
        sp += bsize;
        PzLits Lh;
        int rc = pz_parse_lits(bb, bn, &Lh);  // uniform on all lanes
        if (rc < 0) return rc;
        int64_t pos = Lh.hdr;
        int64_t nlit = Lh.regen;
        if (nlit > PZ_BLOCK_MAX) return PZ_ERR_LITERALS;
        if (Lh.type == 0) {  // raw literals
            if (pos + nlit > bn) return PZ_ERR_SRC_SMALL;
            for (int64_t j = lane; j < nlit; j += 64) lit[j] = bb[pos + j];
            pos += nlit;
        } else if (Lh.type == 1) {  // RLE literals
            if (pos + 1 > bn) return PZ_ERR_SRC_SMALL;
            uint8_t v = bb[pos];
            for (int64_t j = lane; j < nlit; j += 64) lit[j] = v;
            pos += 1;
        } else {  // huffman literals (2 = new tree, 3 = treeless)
            if (pos + Lh.comp > bn) return PZ_ERR_SRC_SMALL;
            const uint8_t *hp = bb + pos;
            int64_t hn = Lh.comp;
            int64_t off = 0;
            if (Lh.type == 2) {
                if (lane == 0) {
                    int nw;
                    int used = pz_huf_read_weights(hp, hn, cx->weights, &nw,
                                                   cx->wksp64, cx->norm);
                    int tl = used < 0
                                 ? used
                                 : pz_huf_build(cx->weights, nw, L->hlut);
                    L->flag = used < 0 ? used
                                       : (tl < 0 ? tl
                                                 : (((int64_t)tl << 32) |
                                                    (uint32_t)used));
                }
                z_fence();
                int64_t f = L->flag;
                if (f < 0) return f;
                htl = (int)(f >> 32);
                off = (int64_t)(uint32_t)f;
            }
            if (htl < 0) return PZ_ERR_HUFFMAN;
            // stream layout: 1 stream, or 4 with a 6-byte jump table
            int ns = Lh.n_streams;
            const uint8_t *s0 = hp + off;
            int64_t szs[4], outs[4], oofs[4], sofs[4];
            if (ns == 1) {
                szs[0] = hn - off;
                outs[0] = nlit;
                oofs[0] = 0;
                sofs[0] = 0;
            } else {
                if (hn - off < 6) return PZ_ERR_SRC_SMALL;
                int64_t s1 = s0[0] | ((int64_t)s0[1] << 8);
                int64_t s2 = s0[2] | ((int64_t)s0[3] << 8);
                int64_t s3 = s0[4] | ((int64_t)s0[5] << 8);
                int64_t s4 = (hn - off - 6) - s1 - s2 - s3;
                if (s4 <= 0) return PZ_ERR_LITERALS;
                int64_t q = (nlit + 3) / 4;
                if (3 * q > nlit) return PZ_ERR_LITERALS;
                s0 += 6;
                szs[0] = s1;
                szs[1] = s2;
                szs[2] = s3;
                szs[3] = s4;
                outs[0] = outs[1] = outs[2] = q;
                outs[3] = nlit - 3 * q;
                sofs[0] = 0;
                sofs[1] = s1;
                sofs[2] = s1 + s2;
                sofs[3] = s1 + s2 + s3;
                oofs[0] = 0;
                oofs[1] = q;
                oofs[2] = 2 * q;
                oofs[3] = 3 * q;
            }
            if (lane < ns) {
                const uint8_t *sb = s0 + sofs[lane];
                int64_t want = outs[lane];
                int64_t bp0 = pz_back_init(sb, szs[lane]);
                int rcs = 0;
                if (bp0 < 0) {
                    rcs = PZ_ERR_HUFFMAN;
                } else {
                    ZBack rb;
                    rb.init(sb, bp0);
                    uint8_t *out = lit + oofs[lane];
                    for (int64_t i = 0; i < want; i++) {
                        uint32_t idx = rb.peek(htl);
                        PzHuf e = L->hlut[idx];
                        out[i] = e.sym;
                        rb.bp -= e.nbits;
                        if (rb.bp < 0) {
                            rcs = PZ_ERR_HUFFMAN;
                            break;
                        }
                    }
                    if (!rcs && rb.bp != 0) rcs = PZ_ERR_HUFFMAN;
                }
                L->stat[lane] = rcs;
            }
            z_fence();
            for (int i = 0; i < ns; i++)
                if (L->stat[i] < 0) return L->stat[i];
            pos += hn;
        }
        z_fence();  // literals visible to the whole wave
        // ---------------- sequences
        if (pos >= bn) return PZ_ERR_SEQ;
        const uint8_t *sq = bb + pos;
        int64_t sn = bn - pos;
        int b0 = sq[0];
        int nseq;
        int64_t so = 1;
        if (b0 < 128) {
            nseq = b0;
        } else if (b0 < 255) {
            if (sn < 2) return PZ_ERR_SRC_SMALL;
            nseq = ((b0 - 128) << 8) + sq[1];
            so = 2;
        } else {
            if (sn < 3) return PZ_ERR_SRC_SMALL;
            nseq = sq[1] + (sq[2] << 8) + 0x7F00;
            so = 3;
        }
        if (nseq == 0) {  // literals only
            if (dp + nlit > dcap) return PZ_ERR_DST_SMALL;
            for (int64_t j = lane; j < nlit; j += 64) dst[dp + j] = lit[j];
            z_fence();
            dp += nlit;
            continue;
        }
        if (so >= sn) return PZ_ERR_SRC_SMALL;
        int modes = sq[so++];
        if (modes & 3) return PZ_ERR_SEQ;
        if (lane == 0) {  // FSE table builds + stream init, serial
            int64_t f = 0;
            int used = pz_seq_table(sq + so, sn - so, (modes >> 6) & 3, 0, 9,
                                    35, cx->llT, &cx->ll_al, cx->have_ll,
                                    cx->norm);
            if (used >= 0) {
                cx->have_ll = 1;
                int64_t so2 = so + used;
                int u2 = pz_seq_table(sq + so2, sn - so2, (modes >> 4) & 3, 1,
                                      8, 31, cx->ofT, &cx->of_al,
                                      cx->have_of, cx->norm);
                if (u2 >= 0) {
                    cx->have_of = 1;
                    so2 += u2;
                    int u3 = pz_seq_table(sq + so2, sn - so2,
                                          (modes >> 2) & 3, 2, 9, 52,
                                          cx->mlT, &cx->ml_al, cx->have_ml,
                                          cx->norm);
                    if (u3 >= 0) {
                        cx->have_ml = 1;
                        so2 += u3;
                        f = so2 - so;
                    } else {
                        f = u3;
                    }
                } else {
                    f = u2;
                }
            } else {
                f = used;
            }
            if (f >= 0) {
                uint64_t k0 = st.rep[0], k1 = st.rep[1], k2 = st.rep[2];
                int rci = pz_seq_init(sq + so + f, sn - so - f, cx->ll_al,
                                      cx->of_al, cx->ml_al, &st);
                st.rep[0] = k0;  // repeat offsets persist across blocks
                st.rep[1] = k1;
                st.rep[2] = k2;
                if (rci < 0) f = rci;
            }
            L->flag = f;
        }
        z_fence();
        {
            int64_t f = L->flag;
            if (f < 0) return f;
            so += f;
        }
        const uint8_t *bs = sq + so;
        int64_t lpos = 0;
        int done = 0;
        ZBack rb;  // lane-0 register bit-window over the sequence stream
        if (lane == 0) {
            rb.init(bs, st.bp);
            rb.bp = st.bp;
        }
        while (done < nseq) {
            int cnt = nseq - done < 64 ? nseq - done : 64;
            if (lane == 0) {
                int64_t f = 0;
                for (int i = 0; i < cnt; i++) {
                    PzSeq q;
                    int rcq = zdev_seq_next(rb, cx->llT, cx->ofT, cx->mlT,
                                            st.s_ll, st.s_of, st.s_ml,
                                            st.rep,
                                            done + i == nseq - 1, &q);
                    if (rcq < 0) {
                        f = rcq;
                        break;
                    }
                    L->ll[i] = q.ll;
                    L->ml[i] = q.ml;
                    L->off[i] = q.off;
                }
                if (done + cnt >= nseq) st.bp = rb.bp;  // final check uses it
                L->flag = f;
            }
            z_fence();
            if (L->flag < 0) return L->flag;
            for (int i = 0; i < cnt; i++) {
                int64_t ll = L->ll[i], ml = L->ml[i];
                int64_t mo = (int64_t)L->off[i];
                if (lpos + ll > nlit) return PZ_ERR_SEQ;
                if (dp + ll + ml > dcap) return PZ_ERR_DST_SMALL;
                for (int64_t j = lane; j < ll; j += 64)
                    dst[dp + j] = lit[lpos + j];
                z_fence();
                dp += ll;
                lpos += ll;
                if (mo > dp) return PZ_ERR_OFFSET;
                if (mo >= ml) {  // non-overlapping: plain parallel copy
                    for (int64_t j = lane; j < ml; j += 64)
                        dst[dp + j] = dst[dp + j - mo];
                } else {  // overlapping: period-mo pattern replication
                    for (int64_t j = lane; j < ml; j += 64)
                        dst[dp + j] = dst[dp - mo + (j % mo)];
                }
                z_fence();
                dp += ml;
            }
            done += cnt;
        }
        if (lane == 0) L->flag = st.bp == 0 ? 0 : PZ_ERR_SEQ;
        z_fence();
        if (L->flag < 0) return L->flag;
        int64_t rem = nlit - lpos;
        if (rem < 0 || dp + rem > dcap) return PZ_ERR_DST_SMALL;
        for (int64_t j = lane; j < rem; j += 64) dst[dp + j] = lit[lpos + j];
        z_fence();
        dp += rem;
    }
    if (F.content_size >= 0 && dp != F.content_size) return PZ_ERR_CORRUPT;
    return dp;
}

__launch_bounds__(256) __global__
void k_zstd_pages(const uint8_t *src, const ZstdJob *jobs, int n,
                  uint8_t *dst, uint8_t *scratch, int64_t *status) {
    __shared__ ZWaveLds lds[4];
    const int wave = (int)((blockIdx.x * blockDim.x + threadIdx.x) >> 6);
    const int wiw = (threadIdx.x >> 6) & 3;
    const int waves = (int)((gridDim.x * blockDim.x) >> 6);
    for (int j = wave; j < n; j += waves) {
        uint8_t *lit = scratch + (size_t)j * PZ_SLOT;
        PzCtx *cx = (PzCtx *)(lit + PZ_BLOCK_MAX);
        int64_t r = zdev_page(src + jobs[j].src_off, jobs[j].src_len,
                              dst + jobs[j].dst_off, jobs[j].dst_len, lit,
                              cx, &lds[wiw]);
        if ((threadIdx.x & 63) == 0) status[j] = r;
    }
}

// emit-family x-block count: 4096 measured ~5% faster than 2048 at C2
// (smaller contiguous output slices balance the tail); PMH_EMIT_BLOCKS
// is the A/B override.
static int emit_grid() {
    static int xblocks = 0;
    if (xblocks <= 0) {
        const char *b = getenv("PMH_EMIT_BLOCKS");
        xblocks = b ? atoi(b) : 0;
        if (xblocks <= 0) xblocks = 4096;
    }
    return xblocks;
}

// ---------------------------------------------------------------- launchers


extern "C" {

hipError_t pmh_launch_partition(const DevCol *keys, const int64_t *lens, int k,
                                int64_t tile_rows, int64_t n_bounds,
                                int64_t total_rows, int32_t *cuts,
                                hipStream_t stream) {
    // k <= 8 two-level: wave-coarse cuts every PMH_COARSE_G tiles (64-ary
    // domain search, ~4 probe rounds deep), then windowed refine for the
    // interior bounds (probes confined to the enclosing coarse windows).
    // k > 8: the per-lane window state of the wave kernel spills at KM >= 16
    // and measured SLOWER than one-level bisection (16x20M: 3.9ms two-level
    // vs 3.2ms one-level, same box — DESIGN.md §7), so seed the endpoints
    // and run the refine kernel with a single full-width window per bound.
    const bool two_level = k <= 8;
    const int64_t G = two_level
                          ? (n_bounds > PMH_COARSE_G + 1 ? PMH_COARSE_G : 1)
                          : (n_bounds > 1 ? n_bounds - 1 : 1);
    const int64_t n_coarse = (n_bounds - 2) / G + 2;  // {0,G,..} u {last}
    int cblocks = (int)((n_coarse + 3) / 4);  // one wave per coarse bound
    int rblocks = (int)((n_bounds + 127) / 128);
    auto launch = [&](auto coarse, auto refiner) {
        if (two_level)
            hipLaunchKernelGGL(coarse, dim3(cblocks), dim3(256), 0, stream,
                               keys, lens, k, tile_rows, n_bounds, total_rows,
                               G, cuts);
        else
            hipLaunchKernelGGL(k_partition_seed, dim3(1), dim3(64), 0,
                               stream, lens, k, n_bounds, cuts);
        if (G > 1)
            hipLaunchKernelGGL(refiner, dim3(rblocks), dim3(128), 0, stream,
                               keys, lens, k, tile_rows, n_bounds,
                               total_rows, G, cuts);
    };
    if (k <= 4)
        launch(k_partition_wave<4>, k_partition_refine<4>);
    else if (k <= 8)
        launch(k_partition_wave<8>, k_partition_refine<8>);
    else if (k <= 16)
        launch(k_partition_wave<16>, k_partition_refine<16>);
    else
        launch(k_partition_wave<PMH_MAX_RUNS>,
               k_partition_refine<PMH_MAX_RUNS>);
    return hipGetLastError();
}

hipError_t pmh_launch_merge_tiles(const DevCol *keys, const DevCol *seqs,
                                  const DevCol *kinds, const int64_t *lens,
                                  int k, const int32_t *cuts, int64_t n_tiles,
                                  int64_t tile_rows, int flags,
                                  const uint64_t *tombs,
                                  uint32_t *winners, int32_t *tile_counts,
                                  uint16_t *group_start, uint32_t *err_flag,
                                  const uint8_t *run_levels, int max_level,
                                  uint64_t *cl_entries, int32_t *cl_counts,
                                  hipStream_t stream) {
    int blocks = n_tiles < 4096 ? (int)n_tiles : 4096;
    const bool pu = flags & 4, fr = flags & 8;
    auto launch = [&](auto kern) {
        hipLaunchKernelGGL(kern, dim3(blocks), dim3(PMH_TILE_THREADS), 0,
                           stream, keys, seqs, kinds, lens, k, cuts, n_tiles,
                           tile_rows, flags, tombs, winners, tile_counts,
                           group_start, err_flag, run_levels, max_level,
                           cl_entries, cl_counts);
    };
    if (pu)
        launch(k_merge_tiles<true, false>);  // PU/agg never first-row
    else if (fr)
        launch(k_merge_tiles<false, true>);
    else
        launch(k_merge_tiles<false, false>);
    return hipGetLastError();
}

hipError_t pmh_launch_emit_pu(const DevCol *cols, const uint8_t *col_dtype,
                              const uint8_t *col_nullable, int n_cols, int k,
                              int seq_col, int kind_col, int flags,
                              const uint32_t *members,
                              const uint16_t *group_start,
                              const int64_t *tile_offsets, int64_t n_tiles,
                              int64_t tile_rows, const int64_t *total_out,
                              uint64_t *const *run_masks,
                              void *const *out_ptrs,
                              uint8_t *const *out_valid, hipStream_t stream) {
    if (run_masks)
        hipLaunchKernelGGL(k_emit_pu<true>, dim3(emit_grid()), dim3(256), 0, stream,
                           cols, col_dtype, col_nullable, n_cols, k, seq_col,
                           kind_col, flags, members, group_start,
                           tile_offsets, n_tiles, tile_rows, total_out,
                           run_masks, out_ptrs, out_valid);
    else
        hipLaunchKernelGGL(k_emit_pu<false>, dim3(emit_grid()), dim3(256), 0, stream,
                           cols, col_dtype, col_nullable, n_cols, k, seq_col,
                           kind_col, flags, members, group_start,
                           tile_offsets, n_tiles, tile_rows, total_out,
                           run_masks, out_ptrs, out_valid);
    return hipGetLastError();
}

hipError_t pmh_launch_composite(const DevCol *keys, int nk, uint64_t shifts,
                                uint64_t bits, int64_t rows, int64_t *ckey,
                                hipStream_t stream) {
    hipLaunchKernelGGL(k_composite, dim3(1024), dim3(256), 0, stream, keys,
                       nk, shifts, bits, rows, ckey);
    return hipGetLastError();
}

hipError_t pmh_launch_pack_valid(const DevCol *cols, int n_cols, int64_t rows,
                                 uint64_t *mask, hipStream_t stream) {
    hipLaunchKernelGGL(k_pack_valid, dim3(1024), dim3(256), 0, stream, cols,
                       n_cols, rows, mask);
    return hipGetLastError();
}

hipError_t pmh_launch_emit_pu_sg(
    const DevCol *cols, const uint8_t *col_dtype,
    const uint8_t *col_nullable, int n_cols, int k, int seq_col,
    int kind_col, int flags, const uint8_t *col_group,
    const int16_t *sg_fields, const uint8_t *sg_nseq, int n_groups,
    const uint32_t *members, const uint16_t *group_start,
    const int64_t *tile_offsets, int64_t n_tiles, int64_t tile_rows,
    const int64_t *total_out, uint64_t *const *run_masks,
    void *const *out_ptrs, uint8_t *const *out_valid, hipStream_t stream) {
    hipLaunchKernelGGL(k_emit_pu_sg, dim3(emit_grid()), dim3(256), 0, stream, cols,
                       col_dtype, col_nullable, n_cols, k, seq_col, kind_col,
                       flags, col_group, sg_fields, sg_nseq, n_groups,
                       members, group_start, tile_offsets, n_tiles,
                       tile_rows, total_out, run_masks, out_ptrs, out_valid);
    return hipGetLastError();
}

hipError_t pmh_launch_emit_agg(const DevCol *cols, const uint8_t *col_dtype,
                               const uint8_t *col_nullable,
                               const uint8_t *col_agg, int n_cols, int k,
                               int seq_col, int kind_col, int flags,
                               const uint32_t *members,
                               const uint16_t *group_start,
                               const int64_t *tile_offsets, int64_t n_tiles,
                               int64_t tile_rows, const int64_t *total_out,
                               uint64_t *const *run_masks,
                               void *const *out_ptrs,
                               uint8_t *const *out_valid,
                               hipStream_t stream) {
    if (run_masks)
        hipLaunchKernelGGL(k_emit_agg<true>, dim3(emit_grid()), dim3(256), 0, stream,
                           cols, col_dtype, col_nullable, col_agg, n_cols, k,
                           seq_col, kind_col, flags, members, group_start,
                           tile_offsets, n_tiles, tile_rows, total_out,
                           run_masks, out_ptrs, out_valid);
    else
        hipLaunchKernelGGL(k_emit_agg<false>, dim3(emit_grid()), dim3(256), 0,
                           stream, cols, col_dtype, col_nullable, col_agg,
                           n_cols, k, seq_col, kind_col, flags, members,
                           group_start, tile_offsets, n_tiles, tile_rows,
                           total_out, run_masks, out_ptrs, out_valid);
    return hipGetLastError();
}

hipError_t pmh_launch_level_scatter(const RleChunk *chunks, int64_t n_chunks,
                                    hipStream_t stream) {
    int blocks = n_chunks < 4096 ? (int)(n_chunks ? n_chunks : 1) : 4096;
    hipLaunchKernelGGL(k_level_scatter, dim3(blocks), dim3(PMH_TILE_THREADS),
                       0, stream, chunks, n_chunks);
    return hipGetLastError();
}

hipError_t pmh_launch_merge_emit(const DevCol *keys, const DevCol *seqs,
                                 const DevCol *kinds, const int64_t *lens,
                                 int k, const int32_t *cuts,
                                 int64_t tile_base, int64_t tile_limit,
                                 int64_t n_tiles,
                                 int64_t tile_rows, int flags,
                                 const uint64_t *tombs,
                                 const DevCol *cols, const uint8_t *col_dtype,
                                 const uint8_t *col_nullable, int n_cols,
                                 int key_col, int seq_col, int kind_col,
                                 const int16_t *useq_cols, int n_useq,
                                 uint64_t *status, uint64_t *ticket,
                                 int64_t *total_out, uint32_t *dense_winners,
                                 void *const *out_ptrs,
                                 uint8_t *const *out_valid,
                                 uint32_t *err_flag, hipStream_t stream) {
    // persistent workgroups: 2 resident per CU (LDS-bound) x 256 CUs; the
    // ticket (zeroed before this launch) hands out the chunk's tiles in
    // order, so any residency is safe
    const int64_t chunk = tile_limit - tile_base;
    int blocks = chunk < 512 ? (int)chunk : 512;
    const bool fr = flags & 8;
    auto launch = [&](auto kern) {
        hipLaunchKernelGGL(kern, dim3(blocks), dim3(PMH_TILE_THREADS), 0,
                           stream, keys, seqs, kinds, lens, k, cuts,
                           tile_base, tile_limit, n_tiles, tile_rows, flags,
                           tombs, cols, col_dtype, col_nullable,
                           n_cols, key_col, seq_col, kind_col, useq_cols,
                           n_useq, status, ticket,
                           total_out, dense_winners, out_ptrs, out_valid,
                           err_flag);
    };
    if (fr)
        launch(k_merge_emit<true>);
    else
        launch(k_merge_emit<false>);
    return hipGetLastError();
}

hipError_t pmh_launch_emit_dense(const DevCol *cols,
                                 const uint8_t *col_dtype,
                                 const uint8_t *col_nullable, int n_cols,
                                 int key_col, int seq_col, int kind_col,
                                 const uint32_t *winners,
                                 int64_t t0, int64_t t1,
                                 const uint64_t *status,
                                 void *const *out_ptrs,
                                 uint8_t *const *out_valid,
                                 hipStream_t stream) {
    hipLaunchKernelGGL(k_emit_dense, dim3(emit_grid()), dim3(256), 0, stream, cols,
                       col_dtype, col_nullable, n_cols, key_col, seq_col,
                       kind_col, winners, t0, t1, status, out_ptrs,
                       out_valid);
    return hipGetLastError();
}

hipError_t pmh_launch_scan_tiles(const int32_t *tile_counts, int64_t n_tiles,
                                 int64_t *tile_offsets, int64_t *total_out,
                                 hipStream_t stream) {
    hipLaunchKernelGGL(k_scan_tiles, dim3(1), dim3(PMH_TILE_THREADS), 0,
                       stream, tile_counts, n_tiles, tile_offsets, total_out);
    return hipGetLastError();
}

hipError_t pmh_launch_emit(const DevCol *cols, const uint8_t *col_dtype,
                           const uint8_t *col_nullable, int n_cols, int k,
                           const uint32_t *winners,
                           const int32_t *tile_counts,
                           const int64_t *tile_offsets, int64_t n_tiles,
                           int64_t tile_rows, const int64_t *total_out,
                           void *const *out_ptrs, uint8_t *const *out_valid,
                           hipStream_t stream) {
    // R = rows per thread iteration (independent gathers in flight);
    // PMH_EMIT_R sweeps 4/6/8 (4 = measured default)
    static int r_rows = 0;
    if (r_rows <= 0) {
        const char *e = getenv("PMH_EMIT_R");
        r_rows = e ? atoi(e) : 4;
        if (r_rows != 6 && r_rows != 8) r_rows = 4;
    }
    auto launch = [&](auto kern) {
        hipLaunchKernelGGL(kern, dim3(emit_grid()), dim3(256), 0, stream,
                           cols, col_dtype, col_nullable, n_cols, k, winners,
                           tile_counts, tile_offsets, n_tiles, tile_rows,
                           total_out, out_ptrs, out_valid);
    };
    if (r_rows == 6)
        launch(k_emit<6>);
    else if (r_rows == 8)
        launch(k_emit<8>);
    else
        launch(k_emit<4>);
    return hipGetLastError();
}



hipError_t pmh_launch_filter(const DevCol *cols, int n_cols,
                             const FilterTerm *terms, int n_terms,
                             int64_t rows, uint8_t *tomb,
                             hipStream_t stream) {
    int64_t want = (rows + 255) / 256;
    int blocks = want < 4096 ? (int)(want ? want : 1) : 4096;
    hipLaunchKernelGGL(k_filter, dim3(blocks), dim3(256), 0, stream, cols,
                       n_cols, terms, n_terms, rows, tomb);
    return hipGetLastError();
}

hipError_t pmh_launch_cl_finalize(const DevCol *cols,
                                  const uint8_t *col_dtype, int n_cols,
                                  int first_val, int k,
                                  const uint64_t *cl_entries,
                                  uint64_t *cl_out, int32_t *cl_counts,
                                  int64_t n_tiles, int64_t tile_rows,
                                  hipStream_t stream) {
    int blocks = n_tiles < 4096 ? (int)n_tiles : 4096;
    hipLaunchKernelGGL(k_cl_finalize, dim3(blocks), dim3(256), 0, stream,
                       cols, col_dtype, n_cols, first_val, k, cl_entries,
                       cl_out, cl_counts, n_tiles, tile_rows);
    return hipGetLastError();
}

hipError_t pmh_launch_cl_emit(const DevCol *cols, const uint8_t *col_dtype,
                              const uint8_t *col_nullable, int n_cols,
                              int kind_col, const uint64_t *cl_rows,
                              const int32_t *cl_counts,
                              const int64_t *cl_offsets, int64_t n_tiles,
                              int64_t tile_rows, void *const *out_ptrs,
                              uint8_t *const *out_valid,
                              hipStream_t stream) {
    int blocks = n_tiles < 4096 ? (int)n_tiles : 4096;
    hipLaunchKernelGGL(k_cl_emit, dim3(blocks), dim3(256), 0, stream, cols,
                       col_dtype, col_nullable, n_cols, kind_col, cl_rows,
                       cl_counts, cl_offsets, n_tiles, tile_rows, out_ptrs,
                       out_valid);
    return hipGetLastError();
}

hipError_t pmh_launch_zstd_compress(const uint8_t *src,
                                    const ZstdJob *jobs, int n,
                                    uint8_t *dst, uint8_t *scratch,
                                    int64_t *status, hipStream_t stream) {
    int want = (n + 3) / 4;
    int blocks = want < 4096 ? (want ? want : 1) : 4096;
    hipLaunchKernelGGL(k_zstd_compress, dim3(blocks), dim3(256), 0, stream,
                       src, jobs, n, dst, scratch, status);
    return hipGetLastError();
}

hipError_t pmh_launch_zstd_pages(const uint8_t *src, const ZstdJob *jobs,
                                 int n, uint8_t *dst, uint8_t *scratch,
                                 int64_t *status, hipStream_t stream) {
    int want = (n + 3) / 4;  // 4 waves (pages) per 256-thread block
    int blocks = want < 4096 ? (want ? want : 1) : 4096;
    hipLaunchKernelGGL(k_zstd_pages, dim3(blocks), dim3(256), 0, stream, src,
                       jobs, n, dst, scratch, status);
    return hipGetLastError();
}

hipError_t pmh_launch_rlev2(const Rlev2Chunk *chunks, int64_t n_chunks,
                            hipStream_t stream) {
    int waves_per_block = 4;  // 256 threads
    int64_t want = (n_chunks + waves_per_block - 1) / waves_per_block;
    int blocks = want < 4096 ? (int)(want ? want : 1) : 4096;
    hipLaunchKernelGGL(k_rlev2, dim3(blocks), dim3(256), 0, stream, chunks,
                       n_chunks);
    return hipGetLastError();
}


hipError_t pmh_launch_delta_sum(const DeltaChunk *chunks, int64_t n_chunks,
                                int64_t *sums, hipStream_t stream) {
    int64_t want = (n_chunks + 3) / 4;
    int blocks = want < 4096 ? (int)(want ? want : 1) : 4096;
    hipLaunchKernelGGL(k_delta_sum, dim3(blocks), dim3(256), 0, stream,
                       chunks, n_chunks, sums);
    return hipGetLastError();
}

hipError_t pmh_launch_delta_scan(const DeltaStream *streams,
                                 int64_t n_streams, const int64_t *sums,
                                 int64_t *bases, hipStream_t stream) {
    int blocks = n_streams < 2048 ? (int)(n_streams ? n_streams : 1) : 2048;
    hipLaunchKernelGGL(k_delta_scan, dim3(blocks), dim3(256), 0, stream,
                       streams, n_streams, sums, bases);
    return hipGetLastError();
}

hipError_t pmh_launch_delta_emit(const DeltaChunk *chunks, int64_t n_chunks,
                                 const int64_t *bases, hipStream_t stream) {
    int64_t want = (n_chunks + 3) / 4;
    int blocks = want < 4096 ? (int)(want ? want : 1) : 4096;
    hipLaunchKernelGGL(k_delta_emit, dim3(blocks), dim3(256), 0, stream,
                       chunks, n_chunks, bases);
    return hipGetLastError();
}

hipError_t pmh_launch_rle_decode(const RleChunk *chunks, int64_t n_chunks,
                                 int32_t *out, hipStream_t stream) {
    int blocks = n_chunks < 4096 ? (int)(n_chunks ? n_chunks : 1) : 4096;
    hipLaunchKernelGGL(k_rle_decode, dim3(blocks), dim3(256), 0, stream,
                       chunks, n_chunks, out);
    return hipGetLastError();
}

hipError_t pmh_launch_dict_gather(const int32_t *ids, const void *dict,
                                  int64_t n, void *out, int esize,
                                  hipStream_t stream) {
    int threads = 256;
    int64_t want = (n + threads - 1) / threads;
    int blocks = want < 4096 ? (int)(want ? want : 1) : 4096;
    if (esize == 4)
        hipLaunchKernelGGL(k_dict_gather_t<int32_t>, dim3(blocks),
                           dim3(threads), 0, stream, ids,
                           (const int32_t *)dict, n, (int32_t *)out);
    else if (esize == 8)
        hipLaunchKernelGGL(k_dict_gather_t<int64_t>, dim3(blocks),
                           dim3(threads), 0, stream, ids,
                           (const int64_t *)dict, n, (int64_t *)out);
    else
        return hipErrorInvalidValue;
    return hipGetLastError();
}

}  // extern "C"

}  // namespace pmh
