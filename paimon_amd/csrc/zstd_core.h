// Scalar zstd (RFC 8878) frame-decoding core, shared by the GPU page
// decoder (k_zstd_pages in zstd_dev.hip — wave-parallel orchestration around
// these scalar pieces) and by a serial host decoder (pz_decode_frame_scalar)
// that the CPU tests fuzz against libzstd. This is an independent
// restatement of the PUBLISHED format (RFC 8878 / zstd format.md), written
// for the subset parquet pages use: single frame, no dictionary, optional
// content checksum (skipped, not verified). Replaces nothing in the
// reference repo directly — apache/paimon delegates page decompression to
// the parquet-java/aircompressor zstd codecs
// (ParquetCompressionCodecFactory); the drop-in behaviour pinned by tests is
// "bytes out == ZSTD_decompress bytes out".
//
// Layout conventions:
//  - forward streams (FSE table descriptions, headers) read bits LSB-first
//    from ascending byte positions;
//  - backward streams (huffman literals, FSE weight pairs, sequences) are
//    addressed by an absolute descending bit position: bit i of the buffer
//    is (buf[i>>3] >> (i&7)) & 1, a read of n bits at position p returns
//    bits [p, p+n) with bit p as the LSB — the highest (first-consumed)
//    bit lands in the MSB of the result, which is exactly the canonical
//    huffman / FSE "top bits" order. Reads below position 0 zero-fill
//    (the format's "assumed zero" tail rule).
#pragma once

#include <stdint.h>
#include <string.h>

#if defined(__HIPCC__) || defined(__HIP_DEVICE_COMPILE__)
#define PZHD static inline __host__ __device__
#else
#define PZHD static inline
#endif

// error codes (negative); positive/zero returns are counts or byte sizes
#define PZ_ERR_MAGIC (-1)
#define PZ_ERR_HEADER (-2)
#define PZ_ERR_BLOCK (-3)
#define PZ_ERR_LITERALS (-4)
#define PZ_ERR_HUFFMAN (-5)
#define PZ_ERR_FSE (-6)
#define PZ_ERR_SEQ (-7)
#define PZ_ERR_OFFSET (-8)
#define PZ_ERR_DST_SMALL (-9)
#define PZ_ERR_SRC_SMALL (-10)
#define PZ_ERR_CORRUPT (-11)

#define PZ_HUF_LOG_MAX 11       // max huffman code length (literals)
#define PZ_FSE_LOG_MAX 9        // max accuracy: LL 9, ML 9, OF 8, weights 6
#define PZ_BLOCK_MAX (128 * 1024)

PZHD int pz_highbit(uint32_t v) {  // floor(log2(v)), v != 0
#if defined(__HIP_DEVICE_COMPILE__)
    return 31 - __clz(v);
#elif defined(__GNUC__)
    return 31 - __builtin_clz(v);
#else
    int r = 0;
    while (v >>= 1) r++;
    return r;
#endif
}

// ------------------------------------------------------------- bit readers

// forward LSB-first reader: bits [*bit, *bit+n) of buf, zero-fill past len
PZHD uint32_t pz_fwd_bits(const uint8_t *buf, int64_t len, uint64_t *bit,
                          int n) {
    uint64_t v = 0;
    for (int i = 0; i < n; i++) {
        uint64_t b = *bit + i;
        uint64_t byte = b >> 3;
        uint32_t o = byte < (uint64_t)len ? (buf[byte] >> (b & 7)) & 1 : 0;
        v |= (uint64_t)o << i;
    }
    *bit += n;
    return (uint32_t)v;
}

// backward-stream extraction: n (<= 32) bits at absolute position p;
// positions below 0 read as 0
PZHD uint32_t pz_bits_at(const uint8_t *buf, int64_t p, int n) {
    if (n == 0) return 0;
    uint64_t v = 0;
    // gather up to 7 bytes covering [p, p+n)
    int64_t first = p >> 3;  // may be negative
    for (int i = 0; i < 6; i++) {
        int64_t byte = first + i;
        if (byte >= 0) v |= (uint64_t)buf[byte] << (8 * i);
    }
    int sh = (int)(p - first * 8);  // 0..7 even for negative p
    v >>= sh;
    return (uint32_t)(v & ((n >= 32) ? 0xFFFFFFFFu : ((1u << n) - 1u)));
}

// init a backward stream over [buf, buf+len): bit position of the sentinel
// (the highest set bit of the last byte); returns -1 if invalid
PZHD int64_t pz_back_init(const uint8_t *buf, int64_t len) {
    if (len <= 0 || buf[len - 1] == 0) return -1;
    return (len - 1) * 8 + pz_highbit(buf[len - 1]);
}

// ------------------------------------------------------------------- FSE

typedef struct {
    uint8_t sym;
    uint8_t nbits;
    uint16_t base;
} PzFse;

// read a normalized count table description (forward stream).
// norm[0..maxsv_cap) zero-filled then populated; returns bytes consumed or
// error. out_nsym = number of symbols present (last index + 1), out_al =
// accuracy log.
PZHD int pz_fse_read_ncount(const uint8_t *src, int64_t slen, int al_cap,
                            int16_t *norm, int maxsv_cap, int *out_nsym,
                            int *out_al) {
    for (int i = 0; i < maxsv_cap; i++) norm[i] = 0;
    uint64_t bit = 0;
    int al = 5 + (int)pz_fwd_bits(src, slen, &bit, 4);
    if (al > al_cap) return PZ_ERR_FSE;
    int remaining = (1 << al) + 1;
    int threshold = 1 << al;
    int nb = al + 1;
    int charnum = 0;
    int prev0 = 0;
    while (remaining > 1 && charnum < maxsv_cap) {
        if (prev0) {
            // runs of zero-probability symbols: 2-bit repeat counts,
            // value 3 continues
            for (;;) {
                uint32_t v = pz_fwd_bits(src, slen, &bit, 2);
                if (v == 3) {
                    charnum += 3;
                    if (charnum >= maxsv_cap) return PZ_ERR_FSE;
                } else {
                    charnum += (int)v;
                    break;
                }
            }
            prev0 = 0;
            if (charnum >= maxsv_cap) break;
        }
        int maxv = (2 * threshold - 1) - remaining;
        uint64_t peek_pos = bit;
        uint32_t raw = pz_fwd_bits(src, slen, &peek_pos, nb);
        int count;
        if ((int)(raw & (threshold - 1)) < maxv) {
            count = (int)(raw & (threshold - 1));
            bit += nb - 1;
        } else {
            count = (int)(raw & (2 * threshold - 1));
            if (count >= threshold) count -= maxv;
            bit += nb;
        }
        count--;  // -1 encodes "less than 1" probability
        remaining -= count < 0 ? -count : count;
        norm[charnum++] = (int16_t)count;
        prev0 = (count == 0);
        while (remaining < threshold) {
            nb--;
            threshold >>= 1;
        }
    }
    if (remaining != 1) return PZ_ERR_FSE;
    int64_t bytes = (int64_t)((bit + 7) >> 3);
    if (bytes > slen) return PZ_ERR_SRC_SMALL;
    *out_nsym = charnum;
    *out_al = al;
    return (int)bytes;
}

// build the FSE decoding table (size 1<<al) from normalized counts
PZHD int pz_fse_build(const int16_t *norm, int nsym, int al, PzFse *table) {
    int size = 1 << al;
    int high = size - 1;
    uint16_t next[256];
    if (nsym > 256) return PZ_ERR_FSE;
    for (int s = 0; s < nsym; s++) {
        if (norm[s] == -1) {
            table[high--].sym = (uint8_t)s;
            next[s] = 1;
        } else {
            next[s] = (uint16_t)norm[s];
        }
    }
    int step = (size >> 1) + (size >> 3) + 3;
    int pos = 0;
    for (int s = 0; s < nsym; s++) {
        for (int i = 0; i < norm[s]; i++) {
            table[pos].sym = (uint8_t)s;
            do {
                pos = (pos + step) & (size - 1);
            } while (pos > high);
        }
    }
    if (pos != 0) return PZ_ERR_FSE;
    for (int i = 0; i < size; i++) {
        int s = table[i].sym;
        uint16_t x = next[s]++;
        if (x == 0) return PZ_ERR_FSE;
        int nbits = al - pz_highbit(x);
        table[i].nbits = (uint8_t)nbits;
        table[i].base = (uint16_t)(((uint32_t)x << nbits) - size);
    }
    return 0;
}

// ---------------------------------------------------------------- huffman

typedef struct {
    uint8_t sym;
    uint8_t nbits;
} PzHuf;

// read a huffman tree description at src; fills weights[0..255] (explicit
// weights only; *out_nw = count INCLUDING the implicit last symbol slot is
// NOT added here). Returns bytes consumed or error.
PZHD int pz_huf_read_weights(const uint8_t *src, int64_t avail,
                             uint8_t *weights, int *out_nw, PzFse *wksp64,
                             int16_t *norm_wksp /* >= 256 */) {
    if (avail < 1) return PZ_ERR_SRC_SMALL;
    int h = src[0];
    if (h >= 128) {  // direct 4-bit weights
        int nw = h - 127;
        int64_t bytes = 1 + (nw + 1) / 2;
        if (bytes > avail) return PZ_ERR_SRC_SMALL;
        for (int i = 0; i < nw; i++) {
            uint8_t b = src[1 + i / 2];
            weights[i] = (i & 1) ? (b & 0xF) : (b >> 4);
        }
        *out_nw = nw;
        return (int)bytes;
    }
    // FSE-compressed weights: h = compressed size; two interleaved states
    int64_t clen = h;
    if (1 + clen > avail) return PZ_ERR_SRC_SMALL;
    const uint8_t *cs = src + 1;
    int nsym = 0, al = 0;
    int used = pz_fse_read_ncount(cs, clen, 6, norm_wksp, 256, &nsym, &al);
    if (used < 0) return used;
    int rc = pz_fse_build(norm_wksp, nsym, al, wksp64);
    if (rc < 0) return rc;
    const uint8_t *bs = cs + used;
    int64_t blen = clen - used;
    int64_t bp = pz_back_init(bs, blen);
    if (bp < 0) return PZ_ERR_HUFFMAN;
    bp -= al;
    uint32_t s1 = pz_bits_at(bs, bp, al);
    bp -= al;
    uint32_t s2 = pz_bits_at(bs, bp, al);
    int nw = 0;
    for (;;) {
        if (nw >= 255) return PZ_ERR_HUFFMAN;
        weights[nw++] = wksp64[s1].sym;
        bp -= wksp64[s1].nbits;
        s1 = wksp64[s1].base + pz_bits_at(bs, bp, wksp64[s1].nbits);
        if (bp < 0) {
            if (nw >= 255) return PZ_ERR_HUFFMAN;
            weights[nw++] = wksp64[s2].sym;
            break;
        }
        if (nw >= 255) return PZ_ERR_HUFFMAN;
        weights[nw++] = wksp64[s2].sym;
        bp -= wksp64[s2].nbits;
        s2 = wksp64[s2].base + pz_bits_at(bs, bp, wksp64[s2].nbits);
        if (bp < 0) {
            if (nw >= 255) return PZ_ERR_HUFFMAN;
            weights[nw++] = wksp64[s1].sym;
            break;
        }
    }
    *out_nw = nw;
    return 1 + (int)clen;
}

// build the single-level huffman decoding LUT (2^tl entries) from explicit
// weights; the LAST symbol's weight is implicit. Returns table log or error.
PZHD int pz_huf_build(const uint8_t *weights, int nw, PzHuf *lut) {
    uint32_t sum = 0;
    for (int i = 0; i < nw; i++) {
        if (weights[i] > PZ_HUF_LOG_MAX) return PZ_ERR_HUFFMAN;
        if (weights[i]) sum += 1u << (weights[i] - 1);
    }
    if (sum == 0) return PZ_ERR_HUFFMAN;
    int tl = pz_highbit(sum) + 1;
    if (tl > PZ_HUF_LOG_MAX) return PZ_ERR_HUFFMAN;
    uint32_t rest = (1u << tl) - sum;
    // rest must be a power of two; implicit last weight completes the sum
    if (rest == 0 || (rest & (rest - 1))) return PZ_ERR_HUFFMAN;
    uint8_t wlast = (uint8_t)(pz_highbit(rest) + 1);
    // cumulative start position per weight: lower weights (longer codes)
    // occupy lower LUT indices; same weight in symbol order
    uint32_t count[PZ_HUF_LOG_MAX + 2];
    for (int w = 0; w <= PZ_HUF_LOG_MAX + 1; w++) count[w] = 0;
    for (int i = 0; i < nw; i++) count[weights[i]]++;
    count[wlast]++;
    uint32_t start[PZ_HUF_LOG_MAX + 2];
    uint32_t acc = 0;
    for (int w = 1; w <= tl; w++) {
        start[w] = acc;
        acc += count[w] << (w - 1);
    }
    if (acc != (1u << tl)) return PZ_ERR_HUFFMAN;
    int nsym = nw + 1;
    for (int s = 0; s < nsym; s++) {
        uint8_t w = s < nw ? weights[s] : wlast;
        if (!w) continue;
        uint32_t cells = 1u << (w - 1);
        uint8_t nbits = (uint8_t)(tl + 1 - w);
        for (uint32_t u = 0; u < cells; u++) {
            lut[start[w] + u].sym = (uint8_t)s;
            lut[start[w] + u].nbits = nbits;
        }
        start[w] += cells;
    }
    return tl;
}

// decode one backward huffman stream into out[0..want); returns 0 or error
PZHD int pz_huf_stream(const uint8_t *bs, int64_t blen, const PzHuf *lut,
                       int tl, uint8_t *out, int64_t want) {
    int64_t bp = pz_back_init(bs, blen);
    if (bp < 0) return PZ_ERR_HUFFMAN;
    for (int64_t i = 0; i < want; i++) {
        uint32_t idx = pz_bits_at(bs, bp - tl, tl);
        out[i] = lut[idx].sym;
        bp -= lut[idx].nbits;
        if (bp < 0) return PZ_ERR_HUFFMAN;
    }
    return bp == 0 ? 0 : PZ_ERR_HUFFMAN;
}

// ----------------------------------------------------- sequence code tables

// literals-length code -> baseline / extra bits (RFC 8878 table)
PZHD uint32_t pz_ll_base(int code) {
    static const uint32_t b[36] = {
        0,  1,  2,  3,  4,  5,  6,  7,  8,   9,   10,  11,   12,   13,
        14, 15, 16, 18, 20, 22, 24, 28, 32,  40,  48,  64,   128,  256,
        512, 1024, 2048, 4096, 8192, 16384, 32768, 65536};
    return b[code];
}
PZHD int pz_ll_bits(int code) {
    static const uint8_t b[36] = {0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0,
                                  0, 0, 0, 0, 1, 1, 1, 1, 2, 2, 3, 3,
                                  4, 6, 7, 8, 9, 10, 11, 12, 13, 14, 15, 16};
    return b[code];
}
// match-length code -> baseline / extra bits
PZHD uint32_t pz_ml_base(int code) {
    static const uint32_t b[53] = {
        3,  4,  5,  6,  7,  8,  9,  10, 11, 12, 13, 14, 15, 16,
        17, 18, 19, 20, 21, 22, 23, 24, 25, 26, 27, 28, 29, 30,
        31, 32, 33, 34, 35, 37, 39, 41, 43, 47, 51, 59, 67, 83,
        99, 131, 259, 515, 1027, 2051, 4099, 8195, 16387, 32771, 65539};
    return b[code];
}
PZHD int pz_ml_bits(int code) {
    static const uint8_t b[53] = {0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0,  0,  0, 0,
                                  0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0,  0,  0, 0,
                                  0, 0, 0, 0, 1, 1, 1, 1, 2, 2, 3,  3,  4, 4,
                                  5, 7, 8, 9, 10, 11, 12, 13, 14, 15, 16};
    return b[code];
}

// predefined FSE distributions (RFC 8878 §3.1.1.3.2.2)
PZHD int pz_fse_predef(int which /*0=LL,1=OF,2=ML*/, int16_t *norm,
                       int *nsym, int *al) {
    static const int16_t LL[36] = {4, 3, 2, 2, 2, 2, 2, 2, 2, 2, 2, 2,
                                   2, 1, 1, 1, 2, 2, 2, 2, 2, 2, 2, 2,
                                   2, 3, 2, 1, 1, 1, 1, 1, -1, -1, -1, -1};
    static const int16_t OF[29] = {1, 1, 1, 1, 1, 1, 2, 2, 2, 1,
                                   1, 1, 1, 1, 1, 1, 1, 1, 1, 1,
                                   1, 1, 1, 1, -1, -1, -1, -1, -1};
    static const int16_t ML[53] = {1, 4, 3, 2, 2, 2, 2, 2, 2, 1, 1,
                                   1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1,
                                   1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1,
                                   1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1,
                                   1, 1, -1, -1, -1, -1, -1, -1, -1};
    const int16_t *t;
    int n, a;
    if (which == 0) {
        t = LL; n = 36; a = 6;
    } else if (which == 1) {
        t = OF; n = 29; a = 5;
    } else {
        t = ML; n = 53; a = 6;
    }
    for (int i = 0; i < n; i++) norm[i] = t[i];
    *nsym = n;
    *al = a;
    return 0;
}

// ------------------------------------------------------- sequence decoding

typedef struct {
    uint32_t ll, ml;
    uint64_t off;  // resolved match offset (0 => literals-only final run N/A)
} PzSeq;

typedef struct {
    uint32_t s_ll, s_of, s_ml;
    int64_t bp;
    uint64_t rep[3];
} PzSeqState;

// init the backward sequence stream and the three states (order LL, OF, ML)
PZHD int pz_seq_init(const uint8_t *bs, int64_t blen, int ll_al, int of_al,
                     int ml_al, PzSeqState *st) {
    int64_t bp = pz_back_init(bs, blen);
    if (bp < 0) return PZ_ERR_SEQ;
    bp -= ll_al;
    st->s_ll = pz_bits_at(bs, bp, ll_al);
    bp -= of_al;
    st->s_of = pz_bits_at(bs, bp, of_al);
    bp -= ml_al;
    st->s_ml = pz_bits_at(bs, bp, ml_al);
    st->bp = bp;
    return 0;
}

// decode the next sequence; `last` = no state update after it.
// Returns 0 or error. Offsets resolve through the 3-slot repeat history.
PZHD int pz_seq_next(const uint8_t *bs, const PzFse *llT, const PzFse *ofT,
                     const PzFse *mlT, PzSeqState *st, int last, PzSeq *out) {
    int llc = llT[st->s_ll].sym;
    int ofc = ofT[st->s_of].sym;
    int mlc = mlT[st->s_ml].sym;
    if (llc > 35 || mlc > 52 || ofc > 31) return PZ_ERR_SEQ;
    // extra bits read order: OF, ML, LL
    uint64_t ofv;
    if (ofc > 0) {
        st->bp -= ofc;
        // offset extra bits may exceed 32 conceptually but ofc <= 31
        ofv = ((uint64_t)1 << ofc) + pz_bits_at(bs, st->bp, ofc);
    } else {
        ofv = 1;
    }
    int mb = pz_ml_bits(mlc);
    st->bp -= mb;
    uint32_t ml = pz_ml_base(mlc) + pz_bits_at(bs, st->bp, mb);
    int lb = pz_ll_bits(llc);
    st->bp -= lb;
    uint32_t ll = pz_ll_base(llc) + pz_bits_at(bs, st->bp, lb);
    // repeat-offset resolution (RFC 8878 §3.1.1.5)
    uint64_t off;
    if (ofv > 3) {
        off = ofv - 3;
        st->rep[2] = st->rep[1];
        st->rep[1] = st->rep[0];
        st->rep[0] = off;
    } else {
        // index into the repeat history; literals_length == 0 shifts the
        // meaning by one (value 3 then means "rep1 - 1")
        int idx = (int)ofv - 1 + (ll == 0 ? 1 : 0);
        if (idx == 0) {
            off = st->rep[0];  // history unchanged
        } else if (idx == 1) {
            off = st->rep[1];  // swap rep1/rep2
            st->rep[1] = st->rep[0];
            st->rep[0] = off;
        } else if (idx == 2) {
            off = st->rep[2];  // rotate rep3 to front
            st->rep[2] = st->rep[1];
            st->rep[1] = st->rep[0];
            st->rep[0] = off;
        } else {
            off = st->rep[0] - 1;  // ll == 0 and offset value 3
            if (off == 0) return PZ_ERR_OFFSET;
            st->rep[2] = st->rep[1];
            st->rep[1] = st->rep[0];
            st->rep[0] = off;
        }
    }
    out->ll = ll;
    out->ml = ml;
    out->off = off;
    // state updates (order LL, ML, OF), skipped for the last sequence
    if (!last) {
        st->bp -= llT[st->s_ll].nbits;
        st->s_ll = llT[st->s_ll].base +
                   pz_bits_at(bs, st->bp, llT[st->s_ll].nbits);
        st->bp -= mlT[st->s_ml].nbits;
        st->s_ml = mlT[st->s_ml].base +
                   pz_bits_at(bs, st->bp, mlT[st->s_ml].nbits);
        st->bp -= ofT[st->s_of].nbits;
        st->s_of = ofT[st->s_of].base +
                   pz_bits_at(bs, st->bp, ofT[st->s_of].nbits);
        if (st->bp < 0) return PZ_ERR_SEQ;
    }
    return 0;
}

// --------------------------------------------------- section header parsing

typedef struct {
    int type;        // 0 raw, 1 RLE, 2 compressed, 3 treeless
    int64_t regen;   // regenerated literals size
    int64_t comp;    // compressed payload size (tree + streams), types 2/3
    int hdr;         // header bytes consumed
    int n_streams;   // 1 or 4 (types 2/3)
} PzLits;

PZHD int pz_parse_lits(const uint8_t *p, int64_t avail, PzLits *L) {
    if (avail < 1) return PZ_ERR_SRC_SMALL;
    int b0 = p[0];
    L->type = b0 & 3;
    int sf = (b0 >> 2) & 3;
    if (L->type <= 1) {  // raw / RLE
        if (sf == 0 || sf == 2) {
            L->regen = b0 >> 3;
            L->hdr = 1;
        } else if (sf == 1) {
            if (avail < 2) return PZ_ERR_SRC_SMALL;
            L->regen = (b0 >> 4) | ((int64_t)p[1] << 4);
            L->hdr = 2;
        } else {
            if (avail < 3) return PZ_ERR_SRC_SMALL;
            L->regen = (b0 >> 4) | ((int64_t)p[1] << 4) |
                       ((int64_t)p[2] << 12);
            L->hdr = 3;
        }
        L->comp = L->type == 0 ? L->regen : 1;
        L->n_streams = 1;
    } else {  // compressed / treeless
        if (sf == 0) {
            if (avail < 3) return PZ_ERR_SRC_SMALL;
            L->regen = (b0 >> 4) | (((int64_t)p[1] & 0x3F) << 4);
            L->comp = (p[1] >> 6) | ((int64_t)p[2] << 2);
            L->hdr = 3;
            L->n_streams = 1;
        } else if (sf == 1) {
            if (avail < 3) return PZ_ERR_SRC_SMALL;
            L->regen = (b0 >> 4) | (((int64_t)p[1] & 0x3F) << 4);
            L->comp = (p[1] >> 6) | ((int64_t)p[2] << 2);
            L->hdr = 3;
            L->n_streams = 4;
        } else if (sf == 2) {
            if (avail < 4) return PZ_ERR_SRC_SMALL;
            L->regen = (b0 >> 4) | ((int64_t)p[1] << 4) |
                       (((int64_t)p[2] & 3) << 12);
            L->comp = (p[2] >> 2) | ((int64_t)p[3] << 6);
            L->hdr = 4;
            L->n_streams = 4;
        } else {
            if (avail < 5) return PZ_ERR_SRC_SMALL;
            L->regen = (b0 >> 4) | ((int64_t)p[1] << 4) |
                       (((int64_t)p[2] & 0x3F) << 12);
            L->comp = (p[2] >> 6) | ((int64_t)p[3] << 2) |
                      ((int64_t)p[4] << 10);
            L->hdr = 5;
            L->n_streams = 4;
        }
        if (L->regen > PZ_BLOCK_MAX) return PZ_ERR_LITERALS;
    }
    return 0;
}

// frame header: returns header size (incl. magic) or error; sets content
// size (-1 if unknown) and whether a 4-byte content checksum trails the
// last block
typedef struct {
    int64_t content_size;  // -1 = not stated
    int has_checksum;
    int hdr;  // bytes incl. magic
} PzFrame;

PZHD int pz_parse_frame(const uint8_t *p, int64_t avail, PzFrame *F) {
    if (avail < 5) return PZ_ERR_SRC_SMALL;
    uint32_t magic = (uint32_t)p[0] | ((uint32_t)p[1] << 8) |
                     ((uint32_t)p[2] << 16) | ((uint32_t)p[3] << 24);
    if (magic != 0xFD2FB528u) return PZ_ERR_MAGIC;
    int d = p[4];
    int fcs_flag = d >> 6;
    int single = (d >> 5) & 1;
    F->has_checksum = (d >> 2) & 1;
    int did_flag = d & 3;
    if ((d >> 3) & 1) return PZ_ERR_HEADER;  // reserved bit
    int pos = 5;
    if (!single) pos += 1;  // window descriptor
    static const int didb[4] = {0, 1, 2, 4};
    pos += didb[did_flag];
    int fcsb = fcs_flag == 0 ? (single ? 1 : 0) : (1 << fcs_flag);
    if (pos + fcsb > avail) return PZ_ERR_SRC_SMALL;
    if (fcsb == 0) {
        F->content_size = -1;
    } else {
        uint64_t v = 0;
        for (int i = 0; i < fcsb; i++) v |= (uint64_t)p[pos + i] << (8 * i);
        if (fcsb == 2) v += 256;
        F->content_size = (int64_t)v;
        pos += fcsb;
        F->hdr = pos;
        return pos;
    }
    F->hdr = pos;
    return pos;
}

// build one of the LL/OF/ML tables per its 2-bit mode. Returns bytes
// consumed from p (0 for predefined/RLE-with-byte handled inside/repeat)
// or error. have_prev: a table from an earlier block exists (repeat mode).
PZHD int pz_seq_table(const uint8_t *p, int64_t avail, int mode, int which,
                      int al_cap, int max_sym, PzFse *table, int *al,
                      int have_prev, int16_t *norm) {
    if (mode == 0) {  // predefined
        int nsym, a;
        pz_fse_predef(which, norm, &nsym, &a);
        *al = a;
        int rc = pz_fse_build(norm, nsym, a, table);
        return rc < 0 ? rc : 0;
    }
    if (mode == 1) {  // RLE: 1 byte = the only symbol; 0-bit state machine
        if (avail < 1) return PZ_ERR_SRC_SMALL;
        if (p[0] > max_sym) return PZ_ERR_SEQ;
        table[0].sym = p[0];
        table[0].nbits = 0;
        table[0].base = 0;
        *al = 0;
        return 1;
    }
    if (mode == 2) {  // FSE-described
        int nsym, a;
        int used = pz_fse_read_ncount(p, avail, al_cap, norm, max_sym + 1,
                                      &nsym, &a);
        if (used < 0) return used;
        int rc = pz_fse_build(norm, nsym, a, table);
        if (rc < 0) return rc;
        *al = a;
        return used;
    }
    // repeat: keep previous table
    return have_prev ? 0 : PZ_ERR_SEQ;
}

// ------------------------------------------------ whole-frame serial decode

// decode context: tables persist across blocks (treeless literals, repeat
// FSE modes, repeat offsets). ~11 KB; the GPU kernel places one per page in
// scratch, the host decoder on its stack/heap.
typedef struct {
    PzHuf hlut[1 << PZ_HUF_LOG_MAX];
    PzFse llT[512], mlT[512], ofT[256], wksp64[64];
    int16_t norm[256];
    uint8_t weights[256];
    int htl;  // -1 = no huffman table yet
    int ll_al, of_al, ml_al;
    int have_ll, have_of, have_ml;
    uint64_t rep[3];
} PzCtx;

// per-page scratch slot for the GPU kernel: literals buffer + context
#define PZ_SLOT \
    ((size_t)PZ_BLOCK_MAX + ((sizeof(PzCtx) + 255) & ~(size_t)255))

// Serial zstd frame decode (RFC 8878 subset: no dictionary; content
// checksum skipped). Returns decompressed size or a PZ_ERR_* code.
// litbuf must hold PZ_BLOCK_MAX bytes.
PZHD int64_t pz_decode_frame(const uint8_t *src, int64_t slen, uint8_t *dst,
                             int64_t dcap, uint8_t *litbuf, PzCtx *cx) {
    PzFrame F;
    int fh = pz_parse_frame(src, slen, &F);
    if (fh < 0) return fh;
    int64_t sp = fh, dp = 0;
    cx->htl = -1;
    cx->have_ll = cx->have_of = cx->have_ml = 0;
    cx->rep[0] = 1;
    cx->rep[1] = 4;
    cx->rep[2] = 8;
    int last = 0;
    while (!last) {
        if (sp + 3 > slen) return PZ_ERR_SRC_SMALL;
        uint32_t bh = (uint32_t)src[sp] | ((uint32_t)src[sp + 1] << 8) |
                      ((uint32_t)src[sp + 2] << 16);
        sp += 3;
        last = bh & 1;
        int btype = (bh >> 1) & 3;
        int64_t bsize = bh >> 3;
        if (btype == 0) {  // raw
            if (sp + bsize > slen || dp + bsize > dcap)
                return PZ_ERR_SRC_SMALL;
            for (int64_t i = 0; i < bsize; i++) dst[dp + i] = src[sp + i];
            sp += bsize;
            dp += bsize;
            continue;
        }
        if (btype == 1) {  // RLE: bsize = regenerated size, 1 content byte
            if (sp + 1 > slen || dp + bsize > dcap) return PZ_ERR_SRC_SMALL;
            uint8_t v = src[sp++];
            for (int64_t i = 0; i < bsize; i++) dst[dp + i] = v;
            dp += bsize;
            continue;
        }
        if (btype != 2 || bsize > slen - sp) return PZ_ERR_BLOCK;
        // ---------------- compressed block
        const uint8_t *bb = src + sp;
        int64_t bn = bsize;
        sp += bsize;
        PzLits L;
        int rc = pz_parse_lits(bb, bn, &L);
        if (rc < 0) return rc;
        int64_t pos = L.hdr;
        int64_t nlit = L.regen;
        if (nlit > PZ_BLOCK_MAX) return PZ_ERR_LITERALS;
        if (L.type == 0) {  // raw literals
            if (pos + nlit > bn) return PZ_ERR_SRC_SMALL;
            for (int64_t i = 0; i < nlit; i++) litbuf[i] = bb[pos + i];
            pos += nlit;
        } else if (L.type == 1) {  // RLE literals
            if (pos + 1 > bn) return PZ_ERR_SRC_SMALL;
            uint8_t v = bb[pos++];
            for (int64_t i = 0; i < nlit; i++) litbuf[i] = v;
        } else {  // huffman (2 = with tree, 3 = treeless)
            if (pos + L.comp > bn) return PZ_ERR_SRC_SMALL;
            const uint8_t *hp = bb + pos;
            int64_t hn = L.comp;
            int64_t off = 0;
            if (L.type == 2) {
                int nw;
                int used = pz_huf_read_weights(hp, hn, cx->weights, &nw,
                                               cx->wksp64, cx->norm);
                if (used < 0) return used;
                int tl = pz_huf_build(cx->weights, nw, cx->hlut);
                if (tl < 0) return tl;
                cx->htl = tl;
                off = used;
            }
            if (cx->htl < 0) return PZ_ERR_HUFFMAN;
            if (L.n_streams == 1) {
                rc = pz_huf_stream(hp + off, hn - off, cx->hlut, cx->htl,
                                   litbuf, nlit);
                if (rc < 0) return rc;
            } else {
                if (hn - off < 6) return PZ_ERR_SRC_SMALL;
                int64_t s1 = hp[off] | ((int64_t)hp[off + 1] << 8);
                int64_t s2 = hp[off + 2] | ((int64_t)hp[off + 3] << 8);
                int64_t s3 = hp[off + 4] | ((int64_t)hp[off + 5] << 8);
                int64_t s4 = (hn - off - 6) - s1 - s2 - s3;
                if (s4 <= 0) return PZ_ERR_LITERALS;
                int64_t q = (nlit + 3) / 4;
                if (3 * q > nlit) return PZ_ERR_LITERALS;
                const uint8_t *sp1 = hp + off + 6;
                const int64_t ss[4] = {s1, s2, s3, s4};
                const int64_t qs[4] = {q, q, q, nlit - 3 * q};
                int64_t so_ = 0, qo = 0;
                for (int i = 0; i < 4; i++) {
                    rc = pz_huf_stream(sp1 + so_, ss[i], cx->hlut, cx->htl,
                                       litbuf + qo, qs[i]);
                    if (rc < 0) return rc;
                    so_ += ss[i];
                    qo += qs[i];
                }
            }
            pos += hn;
        }
        // ---------------- sequences
        if (pos >= bn) return PZ_ERR_SEQ;
        const uint8_t *sq = bb + pos;
        int64_t sn = bn - pos;
        int b0 = sq[0];
        int nseq;
        int so = 1;
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
            for (int64_t i = 0; i < nlit; i++) dst[dp + i] = litbuf[i];
            dp += nlit;
            continue;
        }
        if (so >= sn) return PZ_ERR_SRC_SMALL;
        int modes = sq[so++];
        if (modes & 3) return PZ_ERR_SEQ;  // reserved bits
        int used = pz_seq_table(sq + so, sn - so, (modes >> 6) & 3, 0, 9, 35,
                                cx->llT, &cx->ll_al, cx->have_ll, cx->norm);
        if (used < 0) return used;
        so += used;
        cx->have_ll = 1;
        used = pz_seq_table(sq + so, sn - so, (modes >> 4) & 3, 1, 8, 31,
                            cx->ofT, &cx->of_al, cx->have_of, cx->norm);
        if (used < 0) return used;
        so += used;
        cx->have_of = 1;
        used = pz_seq_table(sq + so, sn - so, (modes >> 2) & 3, 2, 9, 52,
                            cx->mlT, &cx->ml_al, cx->have_ml, cx->norm);
        if (used < 0) return used;
        so += used;
        cx->have_ml = 1;
        PzSeqState st;
        st.rep[0] = cx->rep[0];
        st.rep[1] = cx->rep[1];
        st.rep[2] = cx->rep[2];
        rc = pz_seq_init(sq + so, sn - so, cx->ll_al, cx->of_al, cx->ml_al,
                         &st);
        if (rc < 0) return rc;
        int64_t lpos = 0;
        for (int i = 0; i < nseq; i++) {
            PzSeq q;
            rc = pz_seq_next(sq + so, cx->llT, cx->ofT, cx->mlT, &st,
                             i == nseq - 1, &q);
            if (rc < 0) return rc;
            if (lpos + q.ll > nlit) return PZ_ERR_SEQ;
            if (dp + q.ll + q.ml > dcap) return PZ_ERR_DST_SMALL;
            for (uint32_t j = 0; j < q.ll; j++) dst[dp + j] = litbuf[lpos + j];
            dp += q.ll;
            lpos += q.ll;
            if ((int64_t)q.off > dp) return PZ_ERR_OFFSET;
            for (uint32_t j = 0; j < q.ml; j++)
                dst[dp + j] = dst[dp + j - q.off];
            dp += q.ml;
        }
        if (st.bp != 0) return PZ_ERR_SEQ;
        cx->rep[0] = st.rep[0];
        cx->rep[1] = st.rep[1];
        cx->rep[2] = st.rep[2];
        int64_t rem = nlit - lpos;
        if (rem < 0 || dp + rem > dcap) return PZ_ERR_DST_SMALL;
        for (int64_t i = 0; i < rem; i++) dst[dp + i] = litbuf[lpos + i];
        dp += rem;
    }
    if (F.content_size >= 0 && dp != F.content_size) return PZ_ERR_CORRUPT;
    return dp;
}

// ================================================================ ENCODER
//
// Simplified zstd COMPRESSOR producing spec-valid frames any zstd decoder
// reads (RFC 8878): greedy hash-table LZ matching, RAW literals, sequences
// entropy-coded with the PREDEFINED FSE distributions (no table
// descriptions, no huffman literals — simplicity and wave-friendliness
// over the last few percent of ratio). The write-side counterpart of the
// decoder above; parquet page compression for the compaction write-back
// (the reference delegates to parquet-java's zstd codec — the contract is
// only "a valid zstd frame of these bytes").

// forward LSB-first bit writer; closing appends the '1' sentinel bit
typedef struct {
    uint8_t *out;
    int64_t cap;
    int64_t pos;   // bytes written
    uint64_t acc;
    int nbits;
} PzBitW;

PZHD void pzw_init(PzBitW *w, uint8_t *out, int64_t cap) {
    w->out = out;
    w->cap = cap;
    w->pos = 0;
    w->acc = 0;
    w->nbits = 0;
}

PZHD int pzw_add(PzBitW *w, uint64_t v, int n) {
    if (n == 0) return 0;
    w->acc |= (v & ((n >= 64) ? ~0ull : ((1ull << n) - 1))) << w->nbits;
    w->nbits += n;
    while (w->nbits >= 8) {
        if (w->pos >= w->cap) return PZ_ERR_DST_SMALL;
        w->out[w->pos++] = (uint8_t)w->acc;
        w->acc >>= 8;
        w->nbits -= 8;
    }
    return 0;
}

PZHD int64_t pzw_close(PzBitW *w) {  // returns bytes or error
    if (pzw_add(w, 1, 1) < 0) return PZ_ERR_DST_SMALL;
    if (w->nbits > 0) {
        if (w->pos >= w->cap) return PZ_ERR_DST_SMALL;
        w->out[w->pos++] = (uint8_t)w->acc;
        w->acc = 0;
        w->nbits = 0;
    }
    return w->pos;
}

// FSE encode tables (FSE_buildCTable over the predefined distributions)
typedef struct {
    int32_t deltaNbBits;    // (maxBitsOut << 16) - minStatePlus
    int32_t deltaFindState;
} PzFseCSym;

typedef struct {
    uint16_t stateTable[512];  // max predefined table size is 64; 512 is
                               // roomy for any AL <= 9
    PzFseCSym sym[64];
    int tableLog;
} PzFseC;

PZHD int pz_fse_build_ctable(const int16_t *norm, int nsym, int al,
                             PzFseC *ct) {
    const int size = 1 << al;
    ct->tableLog = al;
    // spread (same placement as the decode table)
    uint8_t spread[512];
    int high = size - 1;
    for (int s = 0; s < nsym; s++)
        if (norm[s] == -1) spread[high--] = (uint8_t)s;
    int step = (size >> 1) + (size >> 3) + 3;
    int pos = 0;
    for (int s = 0; s < nsym; s++) {
        for (int i = 0; i < norm[s]; i++) {
            spread[pos] = (uint8_t)s;
            do {
                pos = (pos + step) & (size - 1);
            } while (pos > high);
        }
    }
    if (pos != 0) return PZ_ERR_FSE;
    // cumulative start per symbol
    int cumul[64];
    int total = 0;
    for (int s = 0; s < nsym; s++) {
        cumul[s] = total;
        total += norm[s] == -1 ? 1 : norm[s];
    }
    if (total != size) return PZ_ERR_FSE;
    // state table: cells in table order fill each symbol's slots
    for (int u = 0; u < size; u++) {
        int s = spread[u];
        ct->stateTable[cumul[s]++] = (uint16_t)(size + u);
    }
    // per-symbol transforms
    total = 0;
    for (int s = 0; s < nsym; s++) {
        int freq = norm[s] == -1 ? 1 : norm[s];
        if (freq == 0) {
            ct->sym[s].deltaNbBits = ((al + 1) << 16) - (1 << al);
            ct->sym[s].deltaFindState = 0;
            continue;
        }
        if (freq == 1) {
            ct->sym[s].deltaNbBits = (al << 16) - (1 << al);
            ct->sym[s].deltaFindState = total - 1;
            total += 1;
        } else {
            int maxBitsOut = al - pz_highbit((uint32_t)(freq - 1));
            int minStatePlus = freq << maxBitsOut;
            ct->sym[s].deltaNbBits = (maxBitsOut << 16) - minStatePlus;
            ct->sym[s].deltaFindState = total - freq;
            total += freq;
        }
    }
    return 0;
}

typedef struct {
    uint32_t value;
} PzFseCState;

PZHD void pz_fse_cinit(PzFseCState *st, const PzFseC *ct, int sym) {
    uint32_t nbBitsOut =
        (uint32_t)(ct->sym[sym].deltaNbBits + (1 << 15)) >> 16;
    uint32_t v = (nbBitsOut << 16) - (uint32_t)ct->sym[sym].deltaNbBits;
    st->value =
        ct->stateTable[(v >> nbBitsOut) + ct->sym[sym].deltaFindState];
}

PZHD int pz_fse_cenc(PzBitW *w, PzFseCState *st, const PzFseC *ct,
                     int sym) {
    uint32_t nbBitsOut =
        (uint32_t)(st->value + (uint32_t)ct->sym[sym].deltaNbBits) >> 16;
    int rc = pzw_add(w, st->value, (int)nbBitsOut);
    st->value = ct->stateTable[(st->value >> nbBitsOut) +
                               ct->sym[sym].deltaFindState];
    return rc;
}

PZHD int pz_fse_cflush(PzBitW *w, const PzFseCState *st, const PzFseC *ct) {
    return pzw_add(w, st->value, ct->tableLog);
}

// code mapping (inverse of the baseline tables above)
PZHD int pz_ll_code(uint32_t ll) {
    if (ll <= 15) return (int)ll;
    for (int c = 35; c >= 16; c--)
        if (ll >= pz_ll_base(c)) return c;
    return 16;
}

PZHD int pz_ml_code(uint32_t ml) {  // ml >= 3
    if (ml <= 34) return (int)(ml - 3);
    for (int c = 52; c >= 32; c--)
        if (ml >= pz_ml_base(c)) return c;
    return 32;
}

// per-call encoder scratch (one per wave/job): sequence lists + tables;
// the hash table is caller-provided (the GPU kernel places it in LDS —
// the match loop is a serial dependent-load chain and LDS latency is the
// difference between ~0.3 and several MB/s per wave)
#define PZ_ENC_HLOG 12
#define PZ_ENC_MAXSEQ (PZ_BLOCK_MAX / 4 + 16)
typedef struct {
    int32_t *htab;  // [1 << PZ_ENC_HLOG]
    uint32_t s_ll[PZ_ENC_MAXSEQ];
    uint32_t s_ml[PZ_ENC_MAXSEQ];
    uint32_t s_of[PZ_ENC_MAXSEQ];
    PzFseC ct_ll, ct_of, ct_ml;
    int16_t norm[64];
} PzEnc;

PZHD uint32_t pz_read32(const uint8_t *p) {
    return (uint32_t)p[0] | ((uint32_t)p[1] << 8) | ((uint32_t)p[2] << 16) |
           ((uint32_t)p[3] << 24);
}

PZHD uint32_t pz_hash32(uint32_t v) {
    return (v * 2654435761u) >> (32 - PZ_ENC_HLOG);
}

// Encode ONE block (3-byte header + body) of <= PZ_BLOCK_MAX input bytes.
// Matches are confined to the block (its own hash table), so blocks
// compress INDEPENDENTLY — the GPU kernel runs one wave per block and the
// host concatenates them into a frame. Returns block bytes or PZ_ERR_*.
PZHD int64_t pz_encode_block(const uint8_t *src, int64_t bn, int last,
                             uint8_t *dst, int64_t cap, PzEnc *e) {
    int nsym, al, rc;
    pz_fse_predef(0, e->norm, &nsym, &al);
    if ((rc = pz_fse_build_ctable(e->norm, nsym, al, &e->ct_ll)) < 0)
        return rc;
    pz_fse_predef(1, e->norm, &nsym, &al);
    if ((rc = pz_fse_build_ctable(e->norm, nsym, al, &e->ct_of)) < 0)
        return rc;
    pz_fse_predef(2, e->norm, &nsym, &al);
    if ((rc = pz_fse_build_ctable(e->norm, nsym, al, &e->ct_ml)) < 0)
        return rc;
    for (int i = 0; i < (1 << PZ_ENC_HLOG); i++) e->htab[i] = -1;
    const int64_t bp = 0;
    int64_t dp = 0;
    {
        // ---- greedy match pass over [bp, bp+bn)
        int nseq = 0;
        int64_t lit_start = bp;   // start of pending literals
        int64_t lit_total = 0;    // literal bytes of this block
        int64_t p = bp;
        const int64_t limit = bp + bn - 4;  // need 4 bytes to match
        while (p <= limit) {
            uint32_t v = pz_read32(src + p);
            uint32_t h = pz_hash32(v);
            int64_t cand = e->htab[h];
            e->htab[h] = (int32_t)p;
            if (cand >= 0 && pz_read32(src + (int64_t)cand) == v &&
                p - cand <= (1 << 27)) {
                // extend
                int64_t m = 4;
                int64_t maxm = bp + bn - p;
                while (m < maxm && src[cand + m] == src[p + m]) m++;
                uint32_t ll = (uint32_t)(p - lit_start);
                if (nseq >= PZ_ENC_MAXSEQ - 1) break;  // overflow: literals
                e->s_ll[nseq] = ll;
                e->s_ml[nseq] = (uint32_t)m;
                e->s_of[nseq] = (uint32_t)(p - cand);
                nseq++;
                lit_total += ll;
                // seed a few hashes inside the match (cheap index upkeep)
                int64_t q = p + 1;
                int64_t qe = p + m - 3;
                for (; q < qe; q += 13)
                    e->htab[pz_hash32(pz_read32(src + q))] = (int32_t)q;
                p += m;
                lit_start = p;
            } else {
                p++;
            }
        }
        int64_t tail_lits = bp + bn - lit_start;
        lit_total += tail_lits;
        // ---- size the compressed form: lit header (<=3) + lits + seq
        // header (<=3) + modes 1 + bitstream (bounded below); fall back to
        // a RAW block unless compressed is smaller
        // literals section header: raw literals, size_format by size
        uint8_t lhdr[3];
        int lhn;
        if (lit_total < 32) {
            lhdr[0] = (uint8_t)(lit_total << 3);
            lhn = 1;
        } else if (lit_total < 4096) {
            lhdr[0] = (uint8_t)(((lit_total & 0xF) << 4) | (1 << 2));
            lhdr[1] = (uint8_t)(lit_total >> 4);
            lhn = 2;
        } else {
            lhdr[0] = (uint8_t)(((lit_total & 0xF) << 4) | (3 << 2));
            lhdr[1] = (uint8_t)(lit_total >> 4);
            lhdr[2] = (uint8_t)(lit_total >> 12);
            lhn = 3;
        }
        // assemble into a bounded region after the 3-byte block header;
        // if anything overflows the raw size, emit RAW instead
        int64_t bh_pos = dp;
        if (dp + 3 > cap) return PZ_ERR_DST_SMALL;
        dp += 3;
        int64_t body = dp;
        int64_t raw_budget = bn;  // compressed must beat raw
        int ok = 1;
        if (body + lhn + lit_total + 4 - dp <= raw_budget &&
            body + lhn + lit_total + 16 <= cap) {
            // literals
            for (int i = 0; i < lhn; i++) dst[dp++] = lhdr[i];
            int64_t ls = bp;
            for (int s = 0; s < nseq; s++) {
                for (uint32_t i = 0; i < e->s_ll[s]; i++)
                    dst[dp++] = src[ls + i];
                ls += e->s_ll[s] + e->s_ml[s];
            }
            for (int64_t i = 0; i < tail_lits; i++)
                dst[dp++] = src[lit_start + i];
            // sequences header
            if (nseq < 128) {
                dst[dp++] = (uint8_t)nseq;
            } else if (nseq < 0x7F00) {
                dst[dp++] = (uint8_t)((nseq >> 8) + 128);
                dst[dp++] = (uint8_t)nseq;
            } else {
                dst[dp++] = 255;
                dst[dp++] = (uint8_t)(nseq - 0x7F00);
                dst[dp++] = (uint8_t)((nseq - 0x7F00) >> 8);
            }
            if (nseq > 0) {
                dst[dp++] = 0;  // modes: predefined / predefined / predef
                PzBitW w;
                pzw_init(&w, dst + dp, cap - dp);
                int lastq = nseq - 1;
                int llc = pz_ll_code(e->s_ll[lastq]);
                int mlc = pz_ml_code(e->s_ml[lastq]);
                uint32_t ofb = e->s_of[lastq] + 3;
                int ofc = pz_highbit(ofb);
                PzFseCState sml, sof, sll;
                pz_fse_cinit(&sml, &e->ct_ml, mlc);
                pz_fse_cinit(&sof, &e->ct_of, ofc);
                pz_fse_cinit(&sll, &e->ct_ll, llc);
                rc = 0;
                rc |= pzw_add(&w, e->s_ll[lastq] - pz_ll_base(llc),
                              pz_ll_bits(llc));
                rc |= pzw_add(&w, e->s_ml[lastq] - pz_ml_base(mlc),
                              pz_ml_bits(mlc));
                rc |= pzw_add(&w, ofb - (1u << ofc), ofc);
                for (int s = nseq - 2; s >= 0 && rc == 0; s--) {
                    llc = pz_ll_code(e->s_ll[s]);
                    mlc = pz_ml_code(e->s_ml[s]);
                    ofb = e->s_of[s] + 3;
                    ofc = pz_highbit(ofb);
                    rc |= pz_fse_cenc(&w, &sof, &e->ct_of, ofc);
                    rc |= pz_fse_cenc(&w, &sml, &e->ct_ml, mlc);
                    rc |= pz_fse_cenc(&w, &sll, &e->ct_ll, llc);
                    rc |= pzw_add(&w, e->s_ll[s] - pz_ll_base(llc),
                                  pz_ll_bits(llc));
                    rc |= pzw_add(&w, e->s_ml[s] - pz_ml_base(mlc),
                                  pz_ml_bits(mlc));
                    rc |= pzw_add(&w, ofb - (1u << ofc), ofc);
                }
                rc |= pz_fse_cflush(&w, &sml, &e->ct_ml);
                rc |= pz_fse_cflush(&w, &sof, &e->ct_of);
                rc |= pz_fse_cflush(&w, &sll, &e->ct_ll);
                int64_t wbytes = rc == 0 ? pzw_close(&w) : -1;
                if (wbytes < 0) {
                    ok = 0;
                } else {
                    dp += wbytes;
                }
            }
            if (ok && dp - body < raw_budget) {
                uint32_t bh = (uint32_t)(last ? 1 : 0) | (2u << 1) |
                              ((uint32_t)(dp - body) << 3);
                dst[bh_pos] = (uint8_t)bh;
                dst[bh_pos + 1] = (uint8_t)(bh >> 8);
                dst[bh_pos + 2] = (uint8_t)(bh >> 16);
            } else {
                ok = 0;
            }
        } else {
            ok = 0;
        }
        if (!ok) {  // RAW block
            dp = body;
            if (dp + bn > cap) return PZ_ERR_DST_SMALL;
            for (int64_t i = 0; i < bn; i++) dst[dp + i] = src[bp + i];
            dp += bn;
            uint32_t bh =
                (uint32_t)(last ? 1 : 0) | (0u << 1) | ((uint32_t)bn << 3);
            dst[bh_pos] = (uint8_t)bh;
            dst[bh_pos + 1] = (uint8_t)(bh >> 8);
            dst[bh_pos + 2] = (uint8_t)(bh >> 16);
        }
    }
    return dp;
}

// frame header for an n-byte single-segment frame (magic + FHD + 4-byte
// content size); returns header bytes
PZHD int pz_frame_header(uint8_t *dst, int64_t n) {
    dst[0] = 0x28;
    dst[1] = 0xB5;
    dst[2] = 0x2F;
    dst[3] = 0xFD;
    dst[4] = (2 << 6) | (1 << 5);  // FCS 4 bytes, single segment
    dst[5] = (uint8_t)n;
    dst[6] = (uint8_t)(n >> 8);
    dst[7] = (uint8_t)(n >> 16);
    dst[8] = (uint8_t)(n >> 24);
    return 9;
}

// Encode one frame serially (host path / tests): header + independent
// blocks. Worst case output n + (n >> 10) + 64.
PZHD int64_t pz_encode_frame(const uint8_t *src, int64_t n, uint8_t *dst,
                             int64_t cap, PzEnc *e) {
    if (cap < 16) return PZ_ERR_DST_SMALL;
    int64_t dp = pz_frame_header(dst, n);
    if (n == 0) {
        if (dp + 3 > cap) return PZ_ERR_DST_SMALL;
        dst[dp++] = 1;  // last, raw, size 0
        dst[dp++] = 0;
        dst[dp++] = 0;
        return dp;
    }
    int64_t bp = 0;
    while (bp < n) {
        int64_t bn = n - bp < PZ_BLOCK_MAX ? n - bp : PZ_BLOCK_MAX;
        int last = bp + bn >= n;
        int64_t r = pz_encode_block(src + bp, bn, last, dst + dp, cap - dp,
                                    e);
        if (r < 0) return r;
        dp += r;
        bp += bn;
    }
    return dp;
}
