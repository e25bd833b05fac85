/* ORACLE — CPU restatement of Apache Paimon's merge-on-read hot path.
 *
 * TEST INFRASTRUCTURE ONLY. This library is the parity checker (and the
 * reported cpu_baseline timed by bench.py); it is never the product path.
 * Only tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg may
 * call it.
 *
 * Each function restates the algorithm of the reference file:line it cites
 * (apache/paimon @ 2026-08-21). The restatement is pinned against:
 *   - the reference's own test vectors (SortMergeReaderTestBase.java:60-88,
 *     MergeFunctionTestUtils.java:35-133) ported into tests/,
 *   - pypaimon (the reference's own Python implementation,
 *     paimon-python/pypaimon/read/reader/sort_merge_reader.py), run in the
 *     build container to generate committed golden vectors (tests/golden/).
 *
 * Build: oracle/Makefile -> oracle/libpaimon_oracle.so (gcc -O2).
 */

#include <stdint.h>
#include <stdlib.h>
#include <string.h>
#include <pthread.h>

/* ---------------------------------------------------------------- *
 * RowKind byte encoding: paimon-api/.../types/RowKind.java:35-56.
 * isAdd() = INSERT(0) | UPDATE_AFTER(2); isRetract() = 1 | 3.
 * ---------------------------------------------------------------- */
static inline int kind_is_add(int8_t k) { return k == 0 || k == 2; }

/* ================================================================ *
 * LoserTree<KeyValue> — restatement of
 * paimon-core/.../mergetree/compact/LoserTree.java:45-356.
 *
 * A leaf here is a cursor over one sorted run held as columnar arrays
 * (key/seq/kind), standing in for LeafIterator over a RecordReader
 * (LoserTree.java:252-325). Batch boundaries do not affect the merged
 * order, so the cursor reads straight through the run.
 * ================================================================ */

typedef enum { /* LoserTree.java:329-345 */
    LOSER_WITH_NEW_KEY = 0,
    LOSER_WITH_SAME_KEY = 1,
    LOSER_POPPED = 2,
    WINNER_WITH_NEW_KEY = 3, /* isWinner() for states >= 3 */
    WINNER_WITH_SAME_KEY = 4,
    WINNER_POPPED = 5
} lt_state;

static inline int state_is_winner(lt_state s) { return s >= WINNER_WITH_NEW_KEY; }

typedef struct {
    const int64_t *key;
    const int64_t *seq;
    const int8_t *kind;
    int64_t len;
    int64_t pos;     /* cursor; pos >= len <=> kv == null (endOfInput) */
    int has_kv;      /* kv != null */
    int first_same_key_index; /* LeafIterator.firstSameKeyIndex */
    lt_state state;
} lt_leaf;

typedef struct {
    int size;
    int *tree;       /* int[size]; tree[0] = overall winner leaf index */
    lt_leaf *leaves;
    int initialized;
} loser_tree;

/* firstComparator wiring: SortMergeReaderWithLoserTree.java:48
 *   (e1, e2) -> userKeyComparator.compare(e2.key(), e1.key())
 * with the null guard of LoserTree.java:70-72 (null loses).
 * compare(parent, child) > 0  <=>  parent wins (smaller key).       */
static int first_cmp(const loser_tree *t, int a_leaf, int b_leaf) {
    const lt_leaf *a = &t->leaves[a_leaf], *b = &t->leaves[b_leaf];
    if (!a->has_kv) return -1;
    if (!b->has_kv) return 1;
    int64_t ka = a->key[a->pos], kb = b->key[b->pos];
    return (kb > ka) - (kb < ka);
}

/* secondComparator: SortMergeReaderWithLoserTree.java:53-63 (no
 * user-defined sequence fields): descending-compare so the SMALLER
 * (sequenceNumber, isAdd) wins; DELETE pops before INSERT on seq ties. */
static int second_cmp(const loser_tree *t, int a_leaf, int b_leaf) {
    const lt_leaf *a = &t->leaves[a_leaf], *b = &t->leaves[b_leaf];
    if (!a->has_kv) return -1;
    if (!b->has_kv) return 1;
    int64_t sa = a->seq[a->pos], sb = b->seq[b->pos];
    if (sa != sb) return (sb > sa) - (sb < sa);
    int aa = kind_is_add(a->kind[a->pos]), ab = kind_is_add(b->kind[b->pos]);
    return ab - aa; /* Boolean.compare(e2.isAdd, e1.isAdd) */
}

/* LeafIterator.advanceIfAvailable, LoserTree.java:305-325 */
static void leaf_advance(lt_leaf *l) {
    l->first_same_key_index = -1;
    l->state = WINNER_WITH_NEW_KEY;
    if (l->has_kv || l->pos < 0) {
        l->pos++;
    }
    l->has_kv = l->pos < l->len;
}

/* adjustWithSameWinnerKey, LoserTree.java:180-204 */
static void adjust_same_winner_key(loser_tree *t, int index, int parent_leaf,
                                   int winner_leaf) {
    lt_leaf *p = &t->leaves[parent_leaf], *w = &t->leaves[winner_leaf];
    switch (p->state) {
    case LOSER_WITH_SAME_KEY: {
        int second = second_cmp(t, parent_leaf, winner_leaf);
        if (second > 0) {
            p->state = WINNER_WITH_SAME_KEY;
            w->state = LOSER_WITH_SAME_KEY;
            if (p->first_same_key_index == -1) p->first_same_key_index = index;
        } else {
            if (w->first_same_key_index == -1) w->first_same_key_index = index;
        }
        return;
    }
    case LOSER_WITH_NEW_KEY:
    case LOSER_POPPED:
        return;
    default:
        abort();
    }
}

/* adjustWithNewWinnerKey, LoserTree.java:206-245 */
static void adjust_new_winner_key(loser_tree *t, int index, int parent_leaf,
                                  int winner_leaf) {
    lt_leaf *p = &t->leaves[parent_leaf], *w = &t->leaves[winner_leaf];
    switch (p->state) {
    case LOSER_WITH_NEW_KEY: {
        int first = first_cmp(t, parent_leaf, winner_leaf);
        if (first == 0) {
            int second = second_cmp(t, parent_leaf, winner_leaf);
            if (second < 0) {
                p->state = LOSER_WITH_SAME_KEY;
                if (w->first_same_key_index == -1) w->first_same_key_index = index;
            } else {
                w->state = LOSER_WITH_SAME_KEY;
                p->state = WINNER_WITH_NEW_KEY;
                if (p->first_same_key_index == -1) p->first_same_key_index = index;
            }
        } else if (first > 0) {
            p->state = WINNER_WITH_NEW_KEY;
            w->state = LOSER_WITH_NEW_KEY;
        }
        return;
    }
    case LOSER_POPPED:
        p->state = WINNER_POPPED;
        p->first_same_key_index = -1;
        w->state = LOSER_WITH_NEW_KEY;
        return;
    default:
        abort(); /* LOSER_WITH_SAME_KEY unreachable, LoserTree.java:228-233 */
    }
}

/* adjust, LoserTree.java:126-158 */
static void lt_adjust(loser_tree *t, int winner) {
    for (int parent = (winner + t->size) / 2; parent > 0 && winner >= 0;
         parent /= 2) {
        lt_leaf *wn = &t->leaves[winner];
        if (t->tree[parent] == -1) {
            wn->state = LOSER_WITH_NEW_KEY; /* tree initialization */
        } else {
            int parent_leaf = t->tree[parent];
            switch (wn->state) {
            case WINNER_WITH_NEW_KEY:
                adjust_new_winner_key(t, parent, parent_leaf, winner);
                break;
            case WINNER_WITH_SAME_KEY:
                adjust_same_winner_key(t, parent, parent_leaf, winner);
                break;
            case WINNER_POPPED:
                if (wn->first_same_key_index < 0) {
                    parent = -1; /* fast path: leave loop after swap check */
                } else {
                    parent = wn->first_same_key_index;
                    lt_leaf *pn = &t->leaves[t->tree[parent]];
                    wn->state = LOSER_POPPED;
                    pn->state = WINNER_WITH_SAME_KEY;
                }
                break;
            default:
                abort();
            }
        }
        if (parent >= 0 && !state_is_winner(t->leaves[winner].state)) {
            int tmp = winner;
            winner = t->tree[parent];
            t->tree[parent] = tmp;
        } else if (parent < 0) {
            /* WINNER_POPPED fast path with no same key: Java's loop ends
             * because parent=-1 fails `parent > 0`; no swap occurs. */
            break;
        }
    }
    t->tree[0] = winner;
}

static void lt_init(loser_tree *t, int n_runs, const int64_t **keys,
                    const int64_t **seqs, const int8_t **kinds,
                    const int64_t *lens) {
    t->size = n_runs;
    t->tree = malloc(sizeof(int) * n_runs);
    t->leaves = malloc(sizeof(lt_leaf) * n_runs);
    for (int i = 0; i < n_runs; i++) {
        lt_leaf *l = &t->leaves[i];
        l->key = keys[i];
        l->seq = seqs[i];
        l->kind = kinds ? kinds[i] : NULL;
        l->len = lens[i];
        l->pos = -1;
        l->has_kv = 0;
        l->first_same_key_index = -1;
        l->state = WINNER_WITH_NEW_KEY;
    }
    /* initializeIfNeeded, LoserTree.java:83-92 */
    for (int i = 0; i < n_runs; i++) t->tree[i] = -1;
    for (int i = n_runs - 1; i >= 0; i--) {
        leaf_advance(&t->leaves[i]);
        lt_adjust(t, i);
    }
    t->initialized = 1;
}

static void lt_free(loser_tree *t) {
    free(t->tree);
    free(t->leaves);
}

/* adjustForNextLoop, LoserTree.java:95-102 */
static void lt_adjust_for_next_loop(loser_tree *t) {
    int w = t->tree[0];
    while (t->leaves[w].state == WINNER_POPPED) {
        leaf_advance(&t->leaves[w]);
        lt_adjust(t, w);
        w = t->tree[0];
    }
}

/* popWinner, LoserTree.java:105-114: returns leaf index or -1 */
static int lt_pop_winner(loser_tree *t, int64_t *row_out) {
    int w = t->tree[0];
    lt_leaf *l = &t->leaves[w];
    if (l->state == WINNER_POPPED) return -1;
    if (!l->has_kv) return -1; /* all runs exhausted */
    *row_out = l->pos;
    l->state = WINNER_POPPED; /* LeafIterator.pop() */
    lt_adjust(t, w);
    return w;
}

static int lt_peek_has_winner(loser_tree *t) {
    lt_leaf *l = &t->leaves[t->tree[0]];
    return l->state != WINNER_POPPED && l->has_kv;
}

/* ================================================================ *
 * Public oracle entry points
 * ================================================================ */

/* Full merged order with key-group boundaries — the output contract of
 * SortMergeReaderWithLoserTree.SortMergeIterator (…:97-122): the stream is
 * ascending (userKey, sequenceNumber, isAdd); out_head[i]=1 marks the first
 * record of each equal-key group. Returns total records emitted. */
int64_t pmo_merge_order(int n_runs, const int64_t **keys, const int64_t **seqs,
                        const int8_t **kinds, const int64_t *lens,
                        int32_t *out_run, int64_t *out_row, uint8_t *out_head) {
    loser_tree t;
    lt_init(&t, n_runs, keys, seqs, kinds, lens);
    int64_t n = 0;
    for (;;) {
        lt_adjust_for_next_loop(&t);
        int64_t row;
        int w = lt_pop_winner(&t, &row);
        if (w < 0) break;
        out_run[n] = w;
        out_row[n] = row;
        out_head[n] = 1;
        n++;
        while (lt_peek_has_winner(&t)) {
            w = lt_pop_winner(&t, &row);
            out_run[n] = w;
            out_row[n] = row;
            out_head[n] = 0;
            n++;
        }
    }
    lt_free(&t);
    return n;
}

/* Deduplicate merge-on-read:
 *   SortMergeReaderWithLoserTree + ReducerMergeFunctionWrapper
 *   (ReducerMergeFunctionWrapper.java:53-73: singleton groups bypass the
 *   merge function — same result for Deduplicate) + DeduplicateMergeFunction
 *   (DeduplicateMergeFunction.java:48-62: keep last KV of the group;
 *   with ignore_delete, retract records are skipped) + optional
 *   DropDeleteReader (DropDeleteReader.java:53-61: drop !isAdd results).
 * Emits (run, row) of each surviving record. Returns count. */
int64_t pmo_merge_dedup(int n_runs, const int64_t **keys, const int64_t **seqs,
                        const int8_t **kinds, const int64_t *lens,
                        int ignore_delete, int drop_delete,
                        int32_t *out_run, int64_t *out_row) {
    loser_tree t;
    lt_init(&t, n_runs, keys, seqs, kinds, lens);
    int64_t n = 0;
    for (;;) {
        lt_adjust_for_next_loop(&t);
        int64_t row;
        int w = lt_pop_winner(&t, &row);
        if (w < 0) break;
        /* group accumulation: wrapper.add per record */
        int latest_run = -1;
        int64_t latest_row = -1;
        int count = 0;
        int first_run = w;
        int64_t first_row = row;
        do {
            count++;
            if (!(ignore_delete && !kind_is_add(t.leaves[w].kind[row]))) {
                latest_run = w;
                latest_row = row;
            }
            if (!lt_peek_has_winner(&t)) break;
            w = lt_pop_winner(&t, &row);
        } while (w >= 0);
        /* ReducerMergeFunctionWrapper: a singleton group returns the input
         * unchanged (merge function not called) — with ignore_delete a
         * singleton retract is still returned (wrapper bypass). */
        if (count == 1) {
            latest_run = first_run;
            latest_row = first_row;
        }
        if (latest_run < 0) continue; /* all records ignored */
        if (drop_delete && !kind_is_add(t.leaves[latest_run].kind[latest_row]))
            continue;
        out_run[n] = latest_run;
        out_row[n] = latest_row;
        n++;
    }
    lt_free(&t);
    return n;
}

/* ================================================================ *
 * Parquet RLE / bit-packed hybrid decoder — restatement of
 * paimon-format/.../reader/VectorizedRleValuesReader.java:
 *   group loop readNextGroup :977-1018 (LEB128 varint header;
 *   header&1==0 -> RLE run of (header>>>1) values; header&1==1 ->
 *   (header>>>1)*8 bit-packed values, little-endian bit order),
 *   RLE literal readIntLittleEndianPaddedOnBitWidth :950-974,
 *   bitWidth==0 -> implicit zeros :116-120.
 * in: the raw stream (after any 4-byte length prefix / 1-byte bit width
 * byte — the caller strips those per :102-114). Decodes exactly
 * num_values int32 values. Returns number decoded, or -1 on overrun.
 * ================================================================ */
int64_t pmo_rle_bp_decode(const uint8_t *in, int64_t in_len, int bit_width,
                          int64_t num_values, int32_t *out) {
    int64_t p = 0, n = 0;
    if (bit_width == 0) {
        memset(out, 0, sizeof(int32_t) * num_values);
        return num_values;
    }
    int byte_width = (bit_width + 7) / 8;
    uint32_t mask = bit_width == 32 ? 0xffffffffu : ((1u << bit_width) - 1);
    while (n < num_values) {
        /* LEB128 unsigned varint header */
        uint64_t header = 0;
        int shift = 0;
        for (;;) {
            if (p >= in_len) return -1;
            uint8_t b = in[p++];
            header |= (uint64_t)(b & 0x7f) << shift;
            if (!(b & 0x80)) break;
            shift += 7;
        }
        if ((header & 1) == 0) {
            /* RLE run */
            int64_t count = (int64_t)(header >> 1);
            if (p + byte_width > in_len) return -1;
            uint32_t v = 0;
            for (int i = 0; i < byte_width; i++) v |= (uint32_t)in[p + i] << (8 * i);
            p += byte_width;
            if (count > num_values - n) count = num_values - n; /* final run may
                overhang: reader stops at num_values (readBatch loop :218-262) */
            for (int64_t i = 0; i < count; i++) out[n + i] = (int32_t)v;
            n += count;
        } else {
            /* bit-packed groups of 8 values, bit_width bytes per group,
             * little-endian bit order (Packer.LITTLE_ENDIAN, :132) */
            int64_t groups = (int64_t)(header >> 1);
            for (int64_t g = 0; g < groups && n < num_values; g++) {
                if (p + bit_width > in_len) return -1;
                uint64_t bitpos = 0;
                for (int i = 0; i < 8 && n < num_values; i++) {
                    uint64_t bit_off = (uint64_t)i * bit_width;
                    uint64_t byte_off = bit_off >> 3;
                    int bit_shift = (int)(bit_off & 7);
                    uint64_t word = 0;
                    /* gather up to 5 bytes to cover bit_width<=32 + shift */
                    for (int bidx = 0; bidx < 5 && (byte_off + bidx) < (uint64_t)bit_width; bidx++)
                        word |= (uint64_t)in[p + byte_off + bidx] << (8 * bidx);
                    out[n++] = (int32_t)((word >> bit_shift) & mask);
                    (void)bitpos;
                }
                p += bit_width;
            }
        }
    }
    return n;
}

/* Simple throughput-baseline variant of the dedup merge used for
 * cpu_baseline timing (same algorithm, no index outputs gathered by the
 * caller): returns merged surviving-row count. */
int64_t pmo_merge_dedup_count(int n_runs, const int64_t **keys,
                              const int64_t **seqs, const int8_t **kinds,
                              const int64_t *lens, int ignore_delete,
                              int drop_delete) {
    int64_t total = 0;
    for (int i = 0; i < n_runs; i++) total += lens[i];
    int32_t *orun = malloc(sizeof(int32_t) * (total ? total : 1));
    int64_t *orow = malloc(sizeof(int64_t) * (total ? total : 1));
    int64_t n = pmo_merge_dedup(n_runs, keys, seqs, kinds, lens, ignore_delete,
                                drop_delete, orun, orow);
    free(orun);
    free(orow);
    return n;
}

/* N-thread variant of the dedup merge for the cpu_baseline (BASELINE.md:
 * "single-thread and N-thread (one thread per bucket)"; C2 is one bucket,
 * so the threads split the KEY SPACE instead — per-thread two-sided cuts
 * via upper_bound of split keys, so an equal-key group never spans a
 * boundary and per-slice loser trees compose exactly). Reported baseline
 * only; never on the product path. */
typedef struct {
    int n_runs;
    const int64_t **keys;
    const int64_t **seqs;
    const int8_t **kinds;
    int64_t lo[64];  /* per-run slice bounds */
    int64_t hi[64];
    int ignore_delete, drop_delete;
    int64_t out;
} mt_slice;

static int64_t ub_key(const int64_t *a, int64_t n, int64_t v) {
    int64_t lo = 0, hi = n;
    while (lo < hi) {
        int64_t mid = lo + ((hi - lo) >> 1);
        if (a[mid] <= v) lo = mid + 1;
        else hi = mid;
    }
    return lo;
}

static void *mt_worker(void *arg) {
    mt_slice *s = (mt_slice *)arg;
    const int64_t *keys[64];
    const int64_t *seqs[64];
    const int8_t *kinds[64];
    int64_t lens[64];
    for (int r = 0; r < s->n_runs; r++) {
        keys[r] = s->keys[r] + s->lo[r];
        seqs[r] = s->seqs[r] + s->lo[r];
        kinds[r] = s->kinds[r] + s->lo[r];
        lens[r] = s->hi[r] - s->lo[r];
    }
    s->out = pmo_merge_dedup_count(s->n_runs, keys, seqs, kinds, lens,
                                   s->ignore_delete, s->drop_delete);
    return NULL;
}

int64_t pmo_merge_dedup_count_mt(int n_runs, const int64_t **keys,
                                 const int64_t **seqs, const int8_t **kinds,
                                 const int64_t *lens, int ignore_delete,
                                 int drop_delete, int n_threads) {
    if (n_threads < 1) n_threads = 1;
    if (n_threads > 64) n_threads = 64;
    if (n_runs > 64) return -1;
    /* split keys: quantiles of the largest run's key column */
    int big = 0;
    for (int r = 1; r < n_runs; r++)
        if (lens[r] > lens[big]) big = r;
    if (lens[big] == 0) return 0;
    mt_slice *sl = calloc(n_threads, sizeof(mt_slice));
    pthread_t *th = malloc(sizeof(pthread_t) * n_threads);
    for (int t = 0; t < n_threads; t++) {
        sl[t].n_runs = n_runs;
        sl[t].keys = keys;
        sl[t].seqs = seqs;
        sl[t].kinds = kinds;
        sl[t].ignore_delete = ignore_delete;
        sl[t].drop_delete = drop_delete;
        for (int r = 0; r < n_runs; r++) {
            if (t == 0) sl[t].lo[r] = 0;
            else sl[t].lo[r] = sl[t - 1].hi[r];
            if (t == n_threads - 1) sl[t].hi[r] = lens[r];
            else {
                int64_t split = keys[big][(lens[big] * (t + 1)) / n_threads];
                sl[t].hi[r] = ub_key(keys[r], lens[r], split);
            }
        }
    }
    for (int t = 0; t < n_threads; t++) pthread_create(&th[t], NULL, mt_worker, &sl[t]);
    int64_t total = 0;
    for (int t = 0; t < n_threads; t++) {
        pthread_join(th[t], NULL);
        total += sl[t].out;
    }
    free(th);
    free(sl);
    return total;
}

/* ================================================================ *
 * ORC RLEv2 integer decoder — restatement of the published ORC v1
 * specification ("Run Length Encoding version 2") as implemented by the
 * orc-core 1.9.8 dependency of the reference (RunLengthIntegerReaderV2;
 * the dep is NOT vendored under /root/reference — SURVEY.md §8c — so
 * parity is pinned via pyarrow.orc round trips + synthetic KATs).
 * Sub-encodings by the top 2 bits of the first byte:
 *   00 SHORT_REPEAT, 01 DIRECT, 10 PATCHED_BASE, 11 DELTA.
 * ================================================================ */

static const int pmo_fbs[32] = {1,  2,  3,  4,  5,  6,  7,  8,
                                9,  10, 11, 12, 13, 14, 15, 16,
                                17, 18, 19, 20, 21, 22, 23, 24,
                                26, 28, 30, 32, 40, 48, 56, 64};

static int pmo_closest_fixed_bits(int n) {
    for (int i = 0; i < 32; i++)
        if (pmo_fbs[i] >= n) return pmo_fbs[i];
    return 64;
}

typedef struct {
    const uint8_t *p;
    const uint8_t *end;
    int err;
} pmo_bs;

static uint64_t bs_byte(pmo_bs *b) {
    if (b->p >= b->end) {
        b->err = 1;
        return 0;
    }
    return *b->p++;
}

static uint64_t bs_be(pmo_bs *b, int nbytes) {
    uint64_t v = 0;
    for (int i = 0; i < nbytes; i++) v = (v << 8) | bs_byte(b);
    return v;
}

static uint64_t bs_uvarint(pmo_bs *b) {
    uint64_t v = 0;
    int shift = 0;
    for (;;) {
        uint64_t x = bs_byte(b);
        v |= (x & 0x7F) << shift;
        if (!(x & 0x80)) return v;
        shift += 7;
    }
}

static int64_t bs_svarint(pmo_bs *b) {
    uint64_t v = bs_uvarint(b);
    return (int64_t)(v >> 1) ^ -(int64_t)(v & 1);
}

/* big-endian bit-packed reader */
typedef struct {
    pmo_bs *b;
    uint64_t cur;
    int bits_left;
} pmo_br;

static uint64_t br_read(pmo_br *r, int width) {
    uint64_t v = 0;
    int need = width;
    while (need > 0) {
        if (r->bits_left == 0) {
            r->cur = bs_byte(r->b);
            r->bits_left = 8;
        }
        int take = need < r->bits_left ? need : r->bits_left;
        v = (v << take) |
            ((r->cur >> (r->bits_left - take)) & ((1ull << take) - 1));
        r->bits_left -= take;
        need -= take;
    }
    return v;
}

static int64_t pmo_zz(uint64_t v) {
    return (int64_t)(v >> 1) ^ -(int64_t)(v & 1);
}

/* Decode exactly n values; returns n or -1 on malformed input. */
int64_t pmo_orc_rlev2_decode(const uint8_t *in, int64_t len, int64_t n,
                             int is_signed, int64_t *out) {
    pmo_bs b = {in, in + len, 0};
    int64_t cnt = 0;
    while (cnt < n && !b.err) {
        uint64_t first = bs_byte(&b);
        int enc = (int)(first >> 6) & 3;
        if (enc == 0) { /* SHORT_REPEAT */
            int w = ((int)(first >> 3) & 7) + 1;
            int rep = ((int)first & 7) + 3;
            uint64_t raw = bs_be(&b, w);
            int64_t v = is_signed ? pmo_zz(raw) : (int64_t)raw;
            for (int i = 0; i < rep && cnt < n; i++) out[cnt++] = v;
        } else if (enc == 1) { /* DIRECT */
            int width = pmo_fbs[(first >> 1) & 0x1f];
            int count = (int)(((first & 1) << 8) | bs_byte(&b)) + 1;
            pmo_br r = {&b, 0, 0};
            for (int i = 0; i < count; i++) {
                uint64_t raw = br_read(&r, width);
                if (cnt < n)
                    out[cnt++] = is_signed ? pmo_zz(raw) : (int64_t)raw;
            }
        } else if (enc == 2) { /* PATCHED_BASE */
            int width = pmo_fbs[(first >> 1) & 0x1f];
            int count = (int)(((first & 1) << 8) | bs_byte(&b)) + 1;
            uint64_t third = bs_byte(&b), fourth = bs_byte(&b);
            int bw = ((int)(third >> 5) & 7) + 1;
            int pw = pmo_fbs[third & 0x1f];
            int pgw = ((int)(fourth >> 5) & 7) + 1;
            int pl = (int)fourth & 0x1f;
            uint64_t braw = bs_be(&b, bw);
            uint64_t smask = 1ull << (bw * 8 - 1);
            int64_t base = (braw & smask) ? -(int64_t)(braw & ~smask)
                                          : (int64_t)braw;
            if (count > 512 || pl > 32) return -1;
            uint64_t vals[512];
            pmo_br r = {&b, 0, 0};
            for (int i = 0; i < count; i++) vals[i] = br_read(&r, width);
            int cfb = pmo_closest_fixed_bits(pw + pgw);
            uint64_t patches[32];
            pmo_br r2 = {&b, 0, 0};
            for (int i = 0; i < pl; i++) patches[i] = br_read(&r2, cfb);
            uint64_t pmask = (pw == 64) ? ~0ull : ((1ull << pw) - 1);
            /* apply patches: positions are cumulative gaps from position 0;
             * an entry with gap == 2^pgw-1 and patch == 0 only extends the
             * gap (verified against pyarrow-written streams) */
            int64_t gap = 0, pos = 0;
            int started = 0;
            for (int pidx = 0; pidx < pl; pidx++) {
                uint64_t g = patches[pidx] >> pw;
                uint64_t pv = patches[pidx] & pmask;
                gap += (int64_t)g;
                if (pv == 0 && g == ((1ull << pgw) - 1)) continue;
                pos = started ? pos + gap : gap;
                started = 1;
                gap = 0;
                if (pos < 0 || pos >= count) return -1;
                vals[pos] |= pv << width;
            }
            for (int i = 0; i < count && cnt < n; i++)
                out[cnt++] = base + (int64_t)vals[i];
        } else { /* DELTA */
            int wcode = (first >> 1) & 0x1f;
            int width = wcode == 0 ? 0 : pmo_fbs[wcode];
            int count = (int)(((first & 1) << 8) | bs_byte(&b)) + 1;
            int64_t base = is_signed ? bs_svarint(&b)
                                     : (int64_t)bs_uvarint(&b);
            int64_t delta = bs_svarint(&b);
            if (cnt < n) out[cnt] = base;
            cnt++;
            int64_t prev = base;
            if (width == 0) {
                for (int i = 1; i < count; i++) {
                    prev += delta;
                    if (cnt < n) out[cnt] = prev;
                    cnt++;
                }
            } else {
                prev = base + delta;
                if (count > 1) {
                    if (cnt < n) out[cnt] = prev;
                    cnt++;
                }
                pmo_br r = {&b, 0, 0};
                for (int i = 2; i < count; i++) {
                    uint64_t d = br_read(&r, width);
                    prev += delta < 0 ? -(int64_t)d : (int64_t)d;
                    if (cnt < n) out[cnt] = prev;
                    cnt++;
                }
            }
        }
    }
    return b.err ? -1 : cnt;
}

/* ORC boolean run: outer byte-RLE (header h: h>=0 -> h+3 copies of next
 * byte; h<0 -> -h literal bytes) over MSB-first bit-packed bytes. Used by
 * PRESENT streams. Emits one byte (0/1) per value. */
int64_t pmo_orc_boolrle_decode(const uint8_t *in, int64_t len, int64_t n,
                               uint8_t *out) {
    pmo_bs b = {in, in + len, 0};
    int64_t cnt = 0;
    while (cnt < n && !b.err) {
        int8_t h = (int8_t)bs_byte(&b);
        if (b.err) break;
        if (h >= 0) { /* run of h+3 repeated bytes */
            uint8_t v = (uint8_t)bs_byte(&b);
            for (int i = 0; i < h + 3; i++)
                for (int bit = 7; bit >= 0 && cnt < n; bit--)
                    out[cnt++] = (v >> bit) & 1;
        } else {
            for (int i = 0; i < -(int)h; i++) {
                uint8_t v = (uint8_t)bs_byte(&b);
                for (int bit = 7; bit >= 0 && cnt < n; bit--)
                    out[cnt++] = (v >> bit) & 1;
            }
        }
    }
    return b.err ? -1 : cnt;
}
