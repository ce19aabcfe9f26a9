/* CPU oracle #2 — ORACLE, TEST INFRASTRUCTURE ONLY (see oracle/__init__.py).
 *
 * Independent scalar C restatement of the ENTIRE hot path the build
 * replaces: parquet bytes -> footer (thrift compact) -> page walk ->
 * LZ4_RAW page decompression -> PLAIN / RLE_DICTIONARY / RLE def-level /
 * DELTA_BINARY_PACKED decode -> conjunctive predicate evaluation ->
 * hash group-by with count/sum/min/max.
 *
 * What it restates (reference file:line):
 *  - the parquet dialect produced by src/parseable/streams.rs:705-780
 *    (row groups of 262,144 rows, LZ4_RAW codec, DELTA_BINARY_PACKED
 *    time column, dictionary columns with PLAIN fallback, data page v1)
 *    — the *format* itself is the parquet-format spec, implemented by the
 *    reference's pinned `parquet` crate 58.1.0 (Cargo.lock; not vendored
 *    under /root/reference, so this follows the published spec and is
 *    pinned against pyarrow + the committed golden vectors instead);
 *  - scan-time semantics of src/query/mod.rs:291-372 + the DataFusion 53
 *    operators it drives: injected `p_timestamp >= lo AND < hi`
 *    (query/mod.rs:829-888), SQL 3-valued predicate logic, GROUP BY with
 *    NULL groups, count(*) vs count(col), i64 sums, LIKE '%x%' substring.
 *
 * Single file, no dependencies beyond libc. Built by oracle/Makefile.
 * Also used (timed) as bench.py's cpu_baseline kind="port" sample.
 */
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <stdint.h>
#include <inttypes.h>
#include <math.h>

static void die(const char *msg) { fprintf(stderr, "cpu_ref: %s\n", msg); exit(1); }

/* set during accumulation: which agg slots hold f64 values */
static int g_agg_is_f64[16];

/* ----------------------------------------------------------------- */
/* thrift compact protocol reader                                     */
/* ----------------------------------------------------------------- */
typedef struct { const uint8_t *p, *end; } TR;

static uint8_t tr_u8(TR *r) { if (r->p >= r->end) die("thrift eof"); return *r->p++; }
static uint64_t tr_varint(TR *r) {
    uint64_t v = 0; int sh = 0;
    for (;;) { uint8_t b = tr_u8(r); v |= (uint64_t)(b & 0x7f) << sh; if (!(b & 0x80)) return v; sh += 7; }
}
static int64_t tr_zigzag(TR *r) { uint64_t v = tr_varint(r); return (int64_t)(v >> 1) ^ -(int64_t)(v & 1); }

/* field header: returns wire type (0 = stop), sets *fid */
static int tr_field(TR *r, int16_t *fid) {
    uint8_t b = tr_u8(r);
    if (b == 0) return 0;
    int delta = b >> 4, t = b & 0xf;
    if (delta) *fid += delta; else *fid = (int16_t)tr_zigzag(r);
    return t;
}
static void tr_skip(TR *r, int t);
static void tr_skip_struct(TR *r) {
    int16_t fid = 0;
    for (;;) { int t = tr_field(r, &fid); if (!t) return; tr_skip(r, t); }
}
static void tr_list_head(TR *r, int *etype, uint32_t *n) {
    uint8_t h = tr_u8(r); *etype = h & 0xf; *n = h >> 4;
    if (*n == 15) *n = (uint32_t)tr_varint(r);
}
static void tr_skip(TR *r, int t) {
    switch (t) {
    case 1: case 2: break;                    /* bool true/false in field header */
    case 3: tr_u8(r); break;                  /* byte */
    case 4: case 5: case 6: tr_zigzag(r); break;
    case 7: r->p += 8; break;                 /* double */
    case 8: { uint64_t n = tr_varint(r); r->p += n; break; } /* binary */
    case 9: case 10: { int et; uint32_t n; tr_list_head(r, &et, &n);
                       for (uint32_t i = 0; i < n; i++) tr_skip(r, et); break; }
    case 12: tr_skip_struct(r); break;
    default: die("thrift bad type");
    }
    if (r->p > r->end) die("thrift overrun");
}
static void tr_binary(TR *r, const uint8_t **s, uint32_t *len) {
    uint64_t n = tr_varint(r); *s = r->p; *len = (uint32_t)n; r->p += n;
    if (r->p > r->end) die("thrift binary overrun");
}

/* ----------------------------------------------------------------- */
/* parquet metadata model (only what the hot path needs)              */
/* ----------------------------------------------------------------- */
enum { PT_BOOLEAN = 0, PT_INT32 = 1, PT_INT64 = 2, PT_INT96 = 3, PT_FLOAT = 4,
       PT_DOUBLE = 5, PT_BYTE_ARRAY = 6, PT_FLBA = 7 };
enum { ENC_PLAIN = 0, ENC_PLAIN_DICT = 2, ENC_RLE = 3, ENC_DELTA_BP = 5,
       ENC_RLE_DICT = 8 };
enum { CODEC_UNCOMPRESSED = 0, CODEC_SNAPPY = 1, CODEC_LZ4_RAW = 7 };
enum { PAGE_DATA = 0, PAGE_INDEX = 1, PAGE_DICT = 2, PAGE_DATA_V2 = 3 };

typedef struct {
    char name[256];
    int phys_type;        /* PT_* */
    int optional;         /* repetition OPTIONAL -> max_def_level 1 */
} SchemaCol;

typedef struct {
    int64_t data_page_offset, dict_page_offset; /* -1 if none */
    int64_t total_compressed_size, num_values;
    int codec;
    int schema_idx;
} ChunkMeta;

typedef struct {
    int64_t num_rows;
    ChunkMeta *chunks;    /* n_cols entries, schema order */
} RowGroupMeta;

typedef struct {
    SchemaCol *cols; int n_cols;
    RowGroupMeta *rgs; int n_rgs;
    int64_t num_rows;
} FileMeta;

static void parse_schema(TR *r, FileMeta *fm) {
    int et; uint32_t n; tr_list_head(r, &et, &n);
    fm->cols = calloc(n, sizeof(SchemaCol));
    fm->n_cols = 0;
    for (uint32_t i = 0; i < n; i++) {
        int16_t fid = 0; int t;
        int phys = -1, rep = 0, nchild = 0; char name[256] = {0};
        for (;;) {
            t = tr_field(r, &fid); if (!t) break;
            switch (fid) {
            case 1: phys = (int)tr_zigzag(r); break;
            case 3: rep = (int)tr_zigzag(r); break;
            case 4: { const uint8_t *s; uint32_t l; tr_binary(r, &s, &l);
                      if (l > 255) l = 255; memcpy(name, s, l); name[l] = 0; break; }
            case 5: nchild = (int)tr_zigzag(r); break;
            default: tr_skip(r, t);
            }
        }
        if (i == 0) { (void)nchild; continue; }  /* root group element */
        SchemaCol *c = &fm->cols[fm->n_cols++];
        strcpy(c->name, name);
        c->phys_type = phys;
        c->optional = (rep == 1);
    }
}

static void parse_column_meta(TR *r, ChunkMeta *cm, FileMeta *fm) {
    int16_t fid = 0; int t;
    cm->dict_page_offset = -1;
    for (;;) {
        t = tr_field(r, &fid); if (!t) break;
        switch (fid) {
        case 3: { /* path_in_schema: list<string> — flat schema: 1 element */
            int et; uint32_t n; tr_list_head(r, &et, &n);
            char name[256] = {0};
            for (uint32_t i = 0; i < n; i++) {
                const uint8_t *s; uint32_t l; tr_binary(r, &s, &l);
                if (i == 0) { if (l > 255) l = 255; memcpy(name, s, l); name[l] = 0; }
            }
            cm->schema_idx = -1;
            for (int i = 0; i < fm->n_cols; i++)
                if (!strcmp(fm->cols[i].name, name)) { cm->schema_idx = i; break; }
            break; }
        case 4: cm->codec = (int)tr_zigzag(r); break;
        case 5: cm->num_values = tr_zigzag(r); break;
        case 7: cm->total_compressed_size = tr_zigzag(r); break;
        case 9: cm->data_page_offset = tr_zigzag(r); break;
        case 11: cm->dict_page_offset = tr_zigzag(r); break;
        default: tr_skip(r, t);
        }
    }
}

static void parse_footer(const uint8_t *buf, size_t len, FileMeta *fm) {
    if (len < 12 || memcmp(buf + len - 4, "PAR1", 4)) die("not parquet");
    uint32_t flen; memcpy(&flen, buf + len - 8, 4);
    if (flen + 8 > len) die("footer too big");
    TR r = { buf + len - 8 - flen, buf + len - 8 };
    int16_t fid = 0; int t;
    for (;;) {
        t = tr_field(&r, &fid); if (!t) break;
        switch (fid) {
        case 2: parse_schema(&r, fm); break;
        case 3: fm->num_rows = tr_zigzag(&r); break;
        case 4: { /* row_groups */
            int et; uint32_t n; tr_list_head(&r, &et, &n);
            fm->rgs = calloc(n, sizeof(RowGroupMeta));
            fm->n_rgs = (int)n;
            for (uint32_t g = 0; g < n; g++) {
                RowGroupMeta *rg = &fm->rgs[g];
                rg->chunks = calloc(fm->n_cols, sizeof(ChunkMeta));
                int16_t fid2 = 0; int t2;
                for (;;) {
                    t2 = tr_field(&r, &fid2); if (!t2) break;
                    if (fid2 == 1) { /* columns */
                        int et2; uint32_t nc; tr_list_head(&r, &et2, &nc);
                        for (uint32_t c = 0; c < nc; c++) {
                            ChunkMeta tmp; memset(&tmp, 0, sizeof(tmp));
                            int16_t fid3 = 0; int t3;
                            for (;;) {
                                t3 = tr_field(&r, &fid3); if (!t3) break;
                                if (fid3 == 3) parse_column_meta(&r, &tmp, fm);
                                else tr_skip(&r, t3);
                            }
                            if (tmp.schema_idx >= 0 && tmp.schema_idx < fm->n_cols)
                                rg->chunks[tmp.schema_idx] = tmp;
                        }
                    } else if (fid2 == 3) rg->num_rows = tr_zigzag(&r);
                    else tr_skip(&r, t2);
                }
            }
            break; }
        default: tr_skip(&r, t);
        }
    }
}

/* ----------------------------------------------------------------- */
/* page header                                                        */
/* ----------------------------------------------------------------- */
typedef struct {
    int type;                     /* PAGE_* */
    int32_t uncomp_size, comp_size;
    int32_t num_values;           /* data or dict */
    int encoding;                 /* data-page value encoding / dict encoding */
    int def_encoding;
    /* v2 */
    int32_t num_nulls, def_len, rep_len; int v2_compressed;
} PageHdr;

static void parse_page_header(TR *r, PageHdr *ph) {
    memset(ph, 0, sizeof(*ph));
    ph->v2_compressed = 1;
    int16_t fid = 0; int t;
    for (;;) {
        t = tr_field(r, &fid); if (!t) break;
        switch (fid) {
        case 1: ph->type = (int)tr_zigzag(r); break;
        case 2: ph->uncomp_size = (int32_t)tr_zigzag(r); break;
        case 3: ph->comp_size = (int32_t)tr_zigzag(r); break;
        case 5: { /* DataPageHeader */
            int16_t f2 = 0; int t2;
            for (;;) {
                t2 = tr_field(r, &f2); if (!t2) break;
                switch (f2) {
                case 1: ph->num_values = (int32_t)tr_zigzag(r); break;
                case 2: ph->encoding = (int)tr_zigzag(r); break;
                case 3: ph->def_encoding = (int)tr_zigzag(r); break;
                default: tr_skip(r, t2);
                }
            }
            break; }
        case 7: { /* DictionaryPageHeader */
            int16_t f2 = 0; int t2;
            for (;;) {
                t2 = tr_field(r, &f2); if (!t2) break;
                switch (f2) {
                case 1: ph->num_values = (int32_t)tr_zigzag(r); break;
                case 2: ph->encoding = (int)tr_zigzag(r); break;
                default: tr_skip(r, t2);
                }
            }
            break; }
        case 8: { /* DataPageHeaderV2 */
            ph->type = PAGE_DATA_V2;
            int16_t f2 = 0; int t2;
            for (;;) {
                t2 = tr_field(r, &f2); if (!t2) break;
                switch (f2) {
                case 1: ph->num_values = (int32_t)tr_zigzag(r); break;
                case 2: ph->num_nulls = (int32_t)tr_zigzag(r); break;
                case 4: ph->encoding = (int)tr_zigzag(r); break;
                case 5: ph->def_len = (int32_t)tr_zigzag(r); break;
                case 6: ph->rep_len = (int32_t)tr_zigzag(r); break;
                case 7: ph->v2_compressed = (t2 == 1); break;
                default: tr_skip(r, t2);
                }
            }
            break; }
        default: tr_skip(r, t);
        }
    }
}

/* ----------------------------------------------------------------- */
/* LZ4 raw block decompression (scalar restatement of the published   */
/* LZ4 block format; parquet LZ4_RAW = one raw block per page)        */
/* ----------------------------------------------------------------- */
static int lz4_decompress(const uint8_t *src, size_t src_len,
                          uint8_t *dst, size_t dst_cap) {
    const uint8_t *sp = src, *send = src + src_len;
    uint8_t *dp = dst, *dend = dst + dst_cap;
    while (sp < send) {
        uint8_t token = *sp++;
        size_t lit = token >> 4;
        if (lit == 15) { uint8_t b; do { if (sp >= send) return -1; b = *sp++; lit += b; } while (b == 255); }
        if (sp + lit > send || dp + lit > dend) return -1;
        memcpy(dp, sp, lit); sp += lit; dp += lit;
        if (sp >= send) break;              /* last sequence: literals only */
        if (sp + 2 > send) return -1;
        size_t off = sp[0] | ((size_t)sp[1] << 8); sp += 2;
        if (off == 0 || (size_t)(dp - dst) < off) return -1;
        size_t mlen = (token & 0xf);
        if (mlen == 15) { uint8_t b; do { if (sp >= send) return -1; b = *sp++; mlen += b; } while (b == 255); }
        mlen += 4;
        if (dp + mlen > dend) return -1;
        const uint8_t *mp = dp - off;
        for (size_t i = 0; i < mlen; i++) dp[i] = mp[i];  /* overlap = repeat */
        dp += mlen;
    }
    return (int)(dp - dst);
}

/* decompress a page; returns pointer (into dst or src for uncompressed) */
static int snappy_decompress(const uint8_t *src, size_t comp, uint8_t *dst,
                             size_t cap) {
    size_t s = 0, d = 0;
    uint64_t ulen = 0; int sh = 0;
    for (;;) {
        if (s >= comp) return -1;
        uint8_t b = src[s++];
        ulen |= (uint64_t)(b & 0x7f) << sh;
        if (!(b & 0x80)) break;
        sh += 7;
    }
    if (ulen > cap) return -1;
    while (s < comp) {
        uint8_t tag = src[s++];
        int type = tag & 3;
        if (type == 0) {
            size_t len = (size_t)(tag >> 2) + 1;
            if (len > 60) {
                int extra = (int)len - 60;
                if (s + extra > comp) return -1;
                len = 0;
                for (int i = 0; i < extra; i++) len |= (size_t)src[s + i] << (8 * i);
                len += 1;
                s += extra;
            }
            if (s + len > comp || d + len > cap) return -1;
            memcpy(dst + d, src + s, len);
            s += len; d += len;
        } else {
            size_t ml, off;
            if (type == 1) {
                ml = ((tag >> 2) & 7) + 4;
                if (s >= comp) return -1;
                off = ((size_t)(tag >> 5) << 8) | src[s++];
            } else if (type == 2) {
                ml = (size_t)(tag >> 2) + 1;
                if (s + 2 > comp) return -1;
                off = src[s] | ((size_t)src[s + 1] << 8);
                s += 2;
            } else {
                ml = (size_t)(tag >> 2) + 1;
                if (s + 4 > comp) return -1;
                off = src[s] | ((size_t)src[s + 1] << 8) |
                      ((size_t)src[s + 2] << 16) | ((size_t)src[s + 3] << 24);
                s += 4;
            }
            if (off == 0 || off > d || d + ml > cap) return -1;
            const uint8_t *mp = dst + d - off;
            for (size_t i = 0; i < ml; i++) dst[d + i] = mp[i];
            d += ml;
        }
    }
    return (int)d;
}

static const uint8_t *page_payload(int codec, const uint8_t *src, int32_t comp,
                                   uint8_t *dst, int32_t uncomp) {
    if (codec == CODEC_UNCOMPRESSED || comp == uncomp) {
        /* arrow-rs/parquet-cpp store the page raw when compression did not
           shrink it only in v2; for v1 LZ4_RAW always compressed — but a
           comp==uncomp page is by construction ambiguous; try LZ4 first */
        if (codec == CODEC_UNCOMPRESSED) return src;
    }
    if (codec == CODEC_LZ4_RAW) {
        int n = lz4_decompress(src, (size_t)comp, dst, (size_t)uncomp);
        if (n == (int)uncomp) return dst;
        if (comp == uncomp) return src;     /* stored raw */
        die("lz4 decode failed");
    }
    if (codec == CODEC_SNAPPY) {
        int n = snappy_decompress(src, (size_t)comp, dst, (size_t)uncomp);
        if (n == (int)uncomp) return dst;
        die("snappy decode failed");
    }
    die("unsupported codec");
    return NULL;
}

/* ----------------------------------------------------------------- */
/* RLE / bit-packed hybrid decode (LSB-first), parquet-format spec    */
/* ----------------------------------------------------------------- */
static const uint8_t *rle_decode(const uint8_t *p, const uint8_t *end,
                                 int bit_width, int32_t n, int32_t *out) {
    int32_t got = 0;
    int byte_w = (bit_width + 7) / 8;
    while (got < n) {
        if (p >= end) die("rle eof");
        uint64_t hdr = 0; int sh = 0;
        for (;;) { uint8_t b = *p++; hdr |= (uint64_t)(b & 0x7f) << sh; if (!(b & 0x80)) break; sh += 7; }
        if (hdr & 1) {                       /* bit-packed groups of 8 */
            int32_t groups = (int32_t)(hdr >> 1);
            int32_t cnt = groups * 8;
            uint64_t acc = 0; int nbits = 0;
            for (int32_t i = 0; i < cnt; i++) {
                while (nbits < bit_width) {
                    if (p < end) acc |= (uint64_t)(*p++) << nbits;
                    nbits += 8;
                }
                int32_t v = (int32_t)(acc & ((bit_width == 32) ? 0xffffffffu : ((1u << bit_width) - 1)));
                acc >>= bit_width; nbits -= bit_width;
                if (got < n) out[got++] = v;   /* tail of last group padded */
            }
        } else {                             /* RLE run */
            int32_t cnt = (int32_t)(hdr >> 1);
            uint32_t v = 0;
            for (int b = 0; b < byte_w; b++) { if (p >= end) die("rle val eof"); v |= (uint32_t)(*p++) << (8 * b); }
            if (cnt > n - got) cnt = n - got;  /* be tolerant of overlong run */
            for (int32_t i = 0; i < cnt; i++) out[got++] = (int32_t)v;
        }
    }
    return p;
}

/* ----------------------------------------------------------------- */
/* DELTA_BINARY_PACKED (i64) decode, parquet-format spec              */
/* ----------------------------------------------------------------- */
static uint64_t dv_varint(const uint8_t **pp) {
    uint64_t v = 0; int sh = 0;
    for (;;) { uint8_t b = *(*pp)++; v |= (uint64_t)(b & 0x7f) << sh; if (!(b & 0x80)) return v; sh += 7; }
}
static int64_t dv_zigzag(const uint8_t **pp) { uint64_t v = dv_varint(pp); return (int64_t)(v >> 1) ^ -(int64_t)(v & 1); }

static void delta_bp_decode(const uint8_t *p, const uint8_t *end,
                            int32_t n_expect, int64_t *out) {
    uint64_t block_size = dv_varint(&p);
    uint64_t mini_per_block = dv_varint(&p);
    uint64_t total = dv_varint(&p);
    int64_t value = dv_zigzag(&p);
    if ((int64_t)total != n_expect && n_expect >= 0) { /* page may hold fewer */ }
    uint64_t per_mini = block_size / mini_per_block;
    int64_t emitted = 0;
    if (total == 0) return;
    out[emitted++] = value;
    while ((uint64_t)emitted < total) {
        int64_t min_delta = dv_zigzag(&p);
        uint8_t bw[256];
        if (mini_per_block > 256) die("too many miniblocks");
        for (uint64_t m = 0; m < mini_per_block; m++) bw[m] = *p++;
        for (uint64_t m = 0; m < mini_per_block && (uint64_t)emitted < total; m++) {
            int bit_width = bw[m];
            uint64_t acc = 0; int nbits = 0;
            const uint8_t *mp = p;
            p += (per_mini * bit_width + 7) / 8;
            if (p > end + 16) die("delta overrun");
            for (uint64_t i = 0; i < per_mini; i++) {
                uint64_t d = 0;
                if (bit_width) {
                    while (nbits < bit_width) { acc |= (uint64_t)(*mp++) << nbits; nbits += 8; }
                    d = (bit_width == 64) ? acc : (acc & ((1ULL << bit_width) - 1));
                    acc >>= bit_width; nbits -= bit_width;
                }
                if ((uint64_t)emitted < total) {
                    value += min_delta + (int64_t)d;
                    out[emitted++] = value;
                }
            }
        }
    }
}

/* ----------------------------------------------------------------- */
/* column chunk decode -> in-memory column                            */
/* ----------------------------------------------------------------- */
typedef struct { const uint8_t *p; uint32_t len; } StrRef;

typedef struct {
    int phys_type;
    int64_t n;               /* rows in row group */
    uint8_t *valid;          /* 1 per row */
    int64_t *i64;            /* PT_INT64/INT32 (widened) */
    double *f64;
    StrRef *str;             /* PT_BYTE_ARRAY */
    /* backing buffers to free */
    uint8_t **bufs; int n_bufs;
} Col;

static void col_free(Col *c) {
    free(c->valid); free(c->i64); free(c->f64); free(c->str);
    for (int i = 0; i < c->n_bufs; i++) free(c->bufs[i]);
    free(c->bufs);
}
static uint8_t *col_buf(Col *c, size_t sz) {
    uint8_t *b = malloc(sz ? sz : 1);
    c->bufs = realloc(c->bufs, sizeof(void *) * (c->n_bufs + 1));
    c->bufs[c->n_bufs++] = b;
    return b;
}

/* decode one column chunk of one row group */
static void decode_chunk(const uint8_t *file, ChunkMeta *cm, SchemaCol *sc,
                         int64_t rg_rows, Col *out) {
    memset(out, 0, sizeof(*out));
    out->phys_type = sc->phys_type;
    out->n = rg_rows;
    out->valid = malloc(rg_rows);
    memset(out->valid, 1, rg_rows);
    if (sc->phys_type == PT_INT64 || sc->phys_type == PT_INT32)
        out->i64 = malloc(sizeof(int64_t) * rg_rows);
    else if (sc->phys_type == PT_DOUBLE || sc->phys_type == PT_FLOAT)
        out->f64 = malloc(sizeof(double) * rg_rows);
    else if (sc->phys_type == PT_BYTE_ARRAY)
        out->str = malloc(sizeof(StrRef) * rg_rows);
    else die("unsupported physical type");

    int64_t start = cm->dict_page_offset >= 0 && cm->dict_page_offset < cm->data_page_offset
                        ? cm->dict_page_offset : cm->data_page_offset;
    const uint8_t *p = file + start;
    const uint8_t *chunk_end = file + start + cm->total_compressed_size;

    /* dictionary (decoded on first DICT page) */
    StrRef *dict_str = NULL; int64_t *dict_i64 = NULL; double *dict_f64 = NULL;
    int32_t dict_n = 0;

    int64_t row = 0;          /* rows (levels) consumed */
    int32_t *idx_buf = malloc(sizeof(int32_t) * 65536);
    size_t idx_cap = 65536;
    int32_t *def_buf = malloc(sizeof(int32_t) * 65536);
    size_t def_cap = 65536;

    while (row < rg_rows && p < chunk_end) {
        TR r = { p, chunk_end };
        PageHdr ph; parse_page_header(&r, &ph);
        const uint8_t *payload_src = r.p;
        p = r.p + ph.comp_size;

        uint8_t *dst = col_buf(out, ph.uncomp_size);
        if (ph.type == PAGE_DICT) {
            const uint8_t *d = page_payload(cm->codec, payload_src, ph.comp_size, dst, ph.uncomp_size);
            dict_n = ph.num_values;
            if (sc->phys_type == PT_BYTE_ARRAY) {
                dict_str = (StrRef *)col_buf(out, sizeof(StrRef) * dict_n);
                const uint8_t *q = d;
                for (int32_t i = 0; i < dict_n; i++) {
                    uint32_t l; memcpy(&l, q, 4); q += 4;
                    dict_str[i].p = q; dict_str[i].len = l; q += l;
                }
            } else if (sc->phys_type == PT_INT64) {
                dict_i64 = (int64_t *)col_buf(out, sizeof(int64_t) * dict_n);
                memcpy(dict_i64, d, 8 * (size_t)dict_n);
            } else if (sc->phys_type == PT_INT32) {
                dict_i64 = (int64_t *)col_buf(out, sizeof(int64_t) * dict_n);
                for (int32_t i = 0; i < dict_n; i++) { int32_t v; memcpy(&v, d + 4 * i, 4); dict_i64[i] = v; }
            } else if (sc->phys_type == PT_DOUBLE) {
                dict_f64 = (double *)col_buf(out, sizeof(double) * dict_n);
                memcpy(dict_f64, d, 8 * (size_t)dict_n);
            }
            continue;
        }
        if (ph.type != PAGE_DATA) die("only data page v1 + dict supported (v2 unexpected from this writer)");

        const uint8_t *d = page_payload(cm->codec, payload_src, ph.comp_size, dst, ph.uncomp_size);
        const uint8_t *dend = d + ph.uncomp_size;
        int32_t nv = ph.num_values;          /* rows incl. nulls (flat schema) */

        if ((size_t)nv > def_cap) { def_cap = nv; def_buf = realloc(def_buf, sizeof(int32_t) * def_cap); }
        if ((size_t)nv > idx_cap) { idx_cap = nv; idx_buf = realloc(idx_buf, sizeof(int32_t) * idx_cap); }

        int32_t present = nv;
        if (sc->optional) {
            /* v1: [u32 len][RLE/bit-packed hybrid, bit_width=1] (max_def=1) */
            uint32_t dl; memcpy(&dl, d, 4);
            rle_decode(d + 4, d + 4 + dl, 1, nv, def_buf);
            d += 4 + dl;
            present = 0;
            for (int32_t i = 0; i < nv; i++) present += def_buf[i];
        } else {
            for (int32_t i = 0; i < nv; i++) def_buf[i] = 1;
        }

        switch (ph.encoding) {
        case ENC_RLE_DICT: case ENC_PLAIN_DICT: {
            int bit_width = *d++;
            rle_decode(d, dend, bit_width, present, idx_buf);
            int32_t k = 0;
            for (int32_t i = 0; i < nv; i++) {
                if (def_buf[i]) {
                    int32_t ix = idx_buf[k++];
                    if (ix < 0 || ix >= dict_n) die("dict index out of range");
                    if (out->str) out->str[row + i] = dict_str[ix];
                    else if (out->i64) out->i64[row + i] = dict_i64[ix];
                    else out->f64[row + i] = dict_f64[ix];
                } else out->valid[row + i] = 0;
            }
            break; }
        case ENC_PLAIN: {
            const uint8_t *q = d;
            for (int32_t i = 0; i < nv; i++) {
                if (!def_buf[i]) { out->valid[row + i] = 0; continue; }
                if (sc->phys_type == PT_INT64) { int64_t v; memcpy(&v, q, 8); q += 8; out->i64[row + i] = v; }
                else if (sc->phys_type == PT_INT32) { int32_t v; memcpy(&v, q, 4); q += 4; out->i64[row + i] = v; }
                else if (sc->phys_type == PT_DOUBLE) { double v; memcpy(&v, q, 8); q += 8; out->f64[row + i] = v; }
                else { uint32_t l; memcpy(&l, q, 4); q += 4; out->str[row + i].p = q; out->str[row + i].len = l; q += l; }
            }
            break; }
        case ENC_DELTA_BP: {
            if (sc->phys_type != PT_INT64 && sc->phys_type != PT_INT32) die("delta_bp on non-int");
            int64_t *vals = malloc(sizeof(int64_t) * present);
            delta_bp_decode(d, dend, present, vals);
            int32_t k = 0;
            for (int32_t i = 0; i < nv; i++) {
                if (def_buf[i]) out->i64[row + i] = vals[k++];
                else out->valid[row + i] = 0;
            }
            free(vals);
            break; }
        default: die("unsupported data encoding");
        }
        row += nv;
    }
    if (row != rg_rows) die("row count mismatch in chunk");
    free(idx_buf); free(def_buf);
}

/* ----------------------------------------------------------------- */
/* query model                                                        */
/* ----------------------------------------------------------------- */
enum { OP_EQ, OP_NE, OP_LT, OP_LE, OP_GT, OP_GE, OP_BETWEEN, OP_CONTAINS };
enum { AGG_COUNT_STAR, AGG_COUNT, AGG_SUM, AGG_MIN, AGG_MAX, AGG_AVG };

typedef struct {
    char col[256]; int op;
    int is_str; char lit_s[512];
    int64_t lo, hi;              /* int literal / between bounds */
    double flit; int is_f64;
} Pred;

typedef struct { int op; char col[256]; } Agg;

typedef struct {
    char *files[4096]; int n_files;
    Pred preds[64]; int n_preds;
    char group_by[8][256]; int n_group;
    Agg aggs[16]; int n_aggs;
    int64_t t_lo, t_hi; int has_time;
} Query;

/* ----------------------------------------------------------------- */
/* group-by hash table: key = concatenated key strings                */
/* ----------------------------------------------------------------- */
typedef struct {
    char *key;                   /* encoded key (NUL-joined, \1 marks NULL) */
    uint32_t klen;               /* full key length (keys embed NULs) */
    int64_t cnt[16];             /* per-agg count of accumulated values */
    int64_t i64v[16];
    double f64v[16];
    double f64c[16];      /* Neumaier compensation for f64 sums */
} Group;

typedef struct { Group *slots; uint32_t cap, n; } HashTab;

static uint64_t fnv1a(const char *s, size_t n) {
    uint64_t h = 1469598103934665603ULL;
    for (size_t i = 0; i < n; i++) { h ^= (uint8_t)s[i]; h *= 1099511628211ULL; }
    return h;
}
static Group *ht_get(HashTab *ht, const char *key, size_t klen) {
    if (ht->n * 2 >= ht->cap) {
        uint32_t ncap = ht->cap ? ht->cap * 2 : 1024;
        Group *ns = calloc(ncap, sizeof(Group));
        for (uint32_t i = 0; i < ht->cap; i++) {
            if (!ht->slots[i].key) continue;
            /* hash the FULL stored length: multi-key buffers embed NUL
               separators, so strlen() truncated and split groups after the
               first rehash (caught by the g_c5 7k-group fixture) */
            uint64_t h = fnv1a(ht->slots[i].key, ht->slots[i].klen);
            uint32_t j = h & (ncap - 1);
            while (ns[j].key) j = (j + 1) & (ncap - 1);
            ns[j] = ht->slots[i];
        }
        free(ht->slots); ht->slots = ns; ht->cap = ncap;
    }
    uint64_t h = fnv1a(key, klen);
    uint32_t j = h & (ht->cap - 1);
    for (;;) {
        if (!ht->slots[j].key) {
            ht->slots[j].key = malloc(klen);
            memcpy(ht->slots[j].key, key, klen);
            ht->slots[j].klen = (uint32_t)klen;
            ht->n++;
            return &ht->slots[j];
        }
        if (ht->slots[j].klen == (uint32_t)klen &&
            !memcmp(ht->slots[j].key, key, klen))
            return &ht->slots[j];
        j = (j + 1) & (ht->cap - 1);
    }
}

/* ----------------------------------------------------------------- */
/* executor                                                           */
/* ----------------------------------------------------------------- */
static int col_index(FileMeta *fm, const char *name) {
    for (int i = 0; i < fm->n_cols; i++)
        if (!strcmp(fm->cols[i].name, name)) return i;
    return -1;
}

static int str_contains(const uint8_t *hay, uint32_t hlen, const char *needle, uint32_t nlen) {
    if (nlen == 0) return 1;
    if (hlen < nlen) return 0;
    for (uint32_t i = 0; i + nlen <= hlen; i++)
        if (hay[i] == (uint8_t)needle[0] && !memcmp(hay + i, needle, nlen)) return 1;
    return 0;
}

static int cmp_res_int(int64_t a, int64_t b) { return a < b ? -1 : (a > b ? 1 : 0); }
static int cmp_res_f64(double a, double b) { return a < b ? -1 : (a > b ? 1 : 0); }
static int cmp_res_str(const uint8_t *a, uint32_t al, const char *b, uint32_t bl) {
    uint32_t m = al < bl ? al : bl;
    int c = memcmp(a, b, m);
    if (c) return c < 0 ? -1 : 1;
    return al < bl ? -1 : (al > bl ? 1 : 0);
}
static int op_match(int op, int c) {
    switch (op) {
    case OP_EQ: return c == 0;
    case OP_NE: return c != 0;
    case OP_LT: return c < 0;
    case OP_LE: return c <= 0;
    case OP_GT: return c > 0;
    case OP_GE: return c >= 0;
    }
    return 0;
}

static void run_query(Query *q) {
    HashTab ht = {0};

    for (int fi = 0; fi < q->n_files; fi++) {
        FILE *fh = fopen(q->files[fi], "rb");
        if (!fh) die("cannot open file");
        fseek(fh, 0, SEEK_END); long fsz = ftell(fh); fseek(fh, 0, SEEK_SET);
        uint8_t *buf = malloc(fsz);
        if (fread(buf, 1, fsz, fh) != (size_t)fsz) die("short read");
        fclose(fh);
        FileMeta fm; memset(&fm, 0, sizeof(fm));
        parse_footer(buf, fsz, &fm);

        /* which columns are needed */
        int need[256]; int n_need = 0;
        int ci_time = -1;
        int pred_ci[64], group_ci[8], agg_ci[16];
        #define NEED(name) ({ int _i = col_index(&fm, name); if (_i < 0) die("no such column"); \
                              int _f = -1; for (int _j = 0; _j < n_need; _j++) if (need[_j] == _i) _f = _j; \
                              if (_f < 0) { need[n_need] = _i; _f = n_need++; } _f; })
        if (q->has_time) ci_time = NEED("p_timestamp");
        for (int i = 0; i < q->n_preds; i++) pred_ci[i] = NEED(q->preds[i].col);
        for (int i = 0; i < q->n_group; i++) group_ci[i] = NEED(q->group_by[i]);
        for (int i = 0; i < q->n_aggs; i++)
            agg_ci[i] = (q->aggs[i].op == AGG_COUNT_STAR) ? -1 : NEED(q->aggs[i].col);

        for (int g = 0; g < fm.n_rgs; g++) {
            RowGroupMeta *rg = &fm.rgs[g];
            Col cols[256];
            for (int j = 0; j < n_need; j++)
                decode_chunk(buf, &rg->chunks[need[j]], &fm.cols[need[j]], rg->num_rows, &cols[j]);

            char keybuf[4096];
            for (int64_t r = 0; r < rg->num_rows; r++) {
                /* injected time range: ts >= lo && ts < hi (query/mod.rs:829-888) */
                if (q->has_time) {
                    Col *tc = &cols[ci_time];
                    if (!tc->valid[r]) continue;
                    int64_t ts = tc->i64[r];
                    if (ts < q->t_lo || ts >= q->t_hi) continue;
                }
                int ok = 1;
                for (int i = 0; i < q->n_preds && ok; i++) {
                    Pred *pd = &q->preds[i];
                    Col *c = &cols[pred_ci[i]];
                    if (!c->valid[r]) { ok = 0; break; }   /* NULL never matches */
                    if (pd->op == OP_CONTAINS) {
                        ok = str_contains(c->str[r].p, c->str[r].len, pd->lit_s, (uint32_t)strlen(pd->lit_s));
                    } else if (pd->op == OP_BETWEEN) {
                        int64_t v = c->i64[r];
                        ok = (v >= pd->lo && v <= pd->hi);
                    } else if (pd->is_str) {
                        ok = op_match(pd->op, cmp_res_str(c->str[r].p, c->str[r].len, pd->lit_s, (uint32_t)strlen(pd->lit_s)));
                    } else if (pd->is_f64) {
                        ok = op_match(pd->op, cmp_res_f64(c->f64[r], pd->flit));
                    } else {
                        ok = op_match(pd->op, cmp_res_int(c->i64[r], pd->lo));
                    }
                }
                if (!ok) continue;

                /* group key */
                size_t kl = 0;
                for (int i = 0; i < q->n_group; i++) {
                    Col *c = &cols[group_ci[i]];
                    if (!c->valid[r]) keybuf[kl++] = 1;    /* NULL marker */
                    else {
                        keybuf[kl++] = 2;
                        memcpy(keybuf + kl, c->str[r].p, c->str[r].len);
                        kl += c->str[r].len;
                    }
                    keybuf[kl++] = 0;
                }
                if (!q->n_group) keybuf[kl++] = 0;
                Group *gr = ht_get(&ht, keybuf, kl ? kl : 1);

                for (int i = 0; i < q->n_aggs; i++) {
                    Agg *a = &q->aggs[i];
                    if (a->op == AGG_COUNT_STAR) { gr->cnt[i]++; continue; }
                    Col *c = &cols[agg_ci[i]];
                    if (!c->valid[r]) continue;
                    if (a->op == AGG_COUNT) { gr->cnt[i]++; continue; }
                    if (c->f64) {
                        g_agg_is_f64[i] = 1;
                        double v = c->f64[r];
                        if (!gr->cnt[i]) { gr->f64v[i] = v; gr->f64c[i] = 0.0; }
                        else if (a->op == AGG_SUM || a->op == AGG_AVG) {
                            /* Neumaier compensated sum (order-dependent but
                               ~exact; final gate vs the oracle is rtol) */
                            double s2 = gr->f64v[i] + v;
                            if (fabs(gr->f64v[i]) >= fabs(v))
                                gr->f64c[i] += (gr->f64v[i] - s2) + v;
                            else
                                gr->f64c[i] += (v - s2) + gr->f64v[i];
                            gr->f64v[i] = s2;
                        }
                        else if (a->op == AGG_MIN) { if (v < gr->f64v[i]) gr->f64v[i] = v; }
                        else if (a->op == AGG_MAX) { if (v > gr->f64v[i]) gr->f64v[i] = v; }
                        gr->cnt[i]++;
                    } else {
                        int64_t v = c->i64[r];
                        if (!gr->cnt[i]) { gr->i64v[i] = v; }
                        else if (a->op == AGG_SUM || a->op == AGG_AVG) gr->i64v[i] += v;
                        else if (a->op == AGG_MIN) { if (v < gr->i64v[i]) gr->i64v[i] = v; }
                        else if (a->op == AGG_MAX) { if (v > gr->i64v[i]) gr->i64v[i] = v; }
                        gr->cnt[i]++;
                    }
                }
            }
            for (int j = 0; j < n_need; j++) col_free(&cols[j]);
        }
        free(fm.cols);
        for (int g = 0; g < fm.n_rgs; g++) free(fm.rgs[g].chunks);
        free(fm.rgs);
        free(buf);
    }

    /* collect, sort by key (NULL last within each key position) */
    Group **rows = malloc(sizeof(Group *) * (ht.n ? ht.n : 1));
    uint32_t nr = 0;
    for (uint32_t i = 0; i < ht.cap; i++)
        if (ht.slots[i].key) rows[nr++] = &ht.slots[i];

    /* key encoding: per key: [1]=NULL | [2]<bytes>, then NUL. \1 < \2 would
       sort NULL first; we want NULL LAST, so compare with marker swapped. */
    int cmp(const void *a, const void *b) {
        const char *ka = (*(Group **)a)->key, *kb = (*(Group **)b)->key;
        for (;;) {
            unsigned char ma = *ka++, mb = *kb++;
            if (ma == 0 && mb == 0) return 0;
            if (ma != mb) return (ma == 1) ? 1 : (mb == 1) ? -1 : (ma < mb ? -1 : 1);
            if (ma == 1) { /* both NULL: next key */ ka++; kb++; continue; }
            /* both present: compare strings up to NUL */
            while (*ka && *kb && *ka == *kb) { ka++; kb++; }
            unsigned char ca = *ka, cb = *kb;
            if (ca != cb) return ca < cb ? -1 : 1;
            ka++; kb++;
        }
    }
    qsort(rows, nr, sizeof(Group *), cmp);

    /* print: keys then agg values, tab-separated; NULL = \N */

    if (nr == 0 && q->n_group == 0) {
        /* aggregate over empty input */
        for (int i = 0; i < q->n_aggs; i++) {
            if (i) putchar('\t');
            if (q->aggs[i].op == AGG_COUNT_STAR || q->aggs[i].op == AGG_COUNT) putchar('0');
            else printf("\\N");
        }
        putchar('\n');
    }
    for (uint32_t r = 0; r < nr; r++) {
        Group *gr = rows[r];
        const char *k = gr->key;
        for (int i = 0; i < q->n_group; i++) {
            if (i) putchar('\t');
            if ((unsigned char)*k == 1) { printf("\\N"); k += 2; }
            else { k++; fputs(k, stdout); k += strlen(k) + 1; }
        }
        for (int i = 0; i < q->n_aggs; i++) {
            if (i || q->n_group) putchar('\t');
            Agg *a = &q->aggs[i];
            if (a->op == AGG_COUNT_STAR || a->op == AGG_COUNT) printf("%" PRId64, gr->cnt[i]);
            else if (!gr->cnt[i]) printf("\\N");
            else if (a->op == AGG_AVG)
                printf("%.17g", (g_agg_is_f64[i] ? gr->f64v[i] + gr->f64c[i]
                                                 : (double)gr->i64v[i]) /
                                    (double)gr->cnt[i]);
            else if (g_agg_is_f64[i])
                printf("%.17g", q->aggs[i].op == AGG_SUM
                                    ? gr->f64v[i] + gr->f64c[i] : gr->f64v[i]);
            else printf("%" PRId64, gr->i64v[i]);
        }
        putchar('\n');
    }
    free(rows);
}

/* ----------------------------------------------------------------- */
/* CLI                                                                */
/* ----------------------------------------------------------------- */
int main(int argc, char **argv) {
    Query q; memset(&q, 0, sizeof(q));
    for (int i = 1; i < argc; i++) {
        if (!strcmp(argv[i], "--time")) {
            sscanf(argv[++i], "%" SCNd64 ",%" SCNd64, &q.t_lo, &q.t_hi);
            q.has_time = 1;
        } else if (!strcmp(argv[i], "--group-by")) {
            char *s = argv[++i], *tok;
            while ((tok = strsep(&s, ","))) strcpy(q.group_by[q.n_group++], tok);
        } else if (!strcmp(argv[i], "--agg")) {
            char *s = argv[++i];
            Agg *a = &q.aggs[q.n_aggs++];
            if (!strcmp(s, "count_star")) a->op = AGG_COUNT_STAR;
            else {
                char *colon = strchr(s, ':');
                *colon = 0;
                if (!strcmp(s, "count")) a->op = AGG_COUNT;
                else if (!strcmp(s, "sum")) a->op = AGG_SUM;
                else if (!strcmp(s, "avg")) a->op = AGG_AVG;
                else if (!strcmp(s, "min")) a->op = AGG_MIN;
                else if (!strcmp(s, "max")) a->op = AGG_MAX;
                else die("bad agg");
                strcpy(a->col, colon + 1);
            }
        } else if (!strcmp(argv[i], "--pred")) {
            /* op:col:kind:literal  (kind = i64|f64|str; between: i64 lo,hi) */
            char *s = argv[++i];
            Pred *pd = &q.preds[q.n_preds++];
            char *p1 = strchr(s, ':'); *p1++ = 0;
            char *p2 = strchr(p1, ':'); *p2++ = 0;
            char *p3 = strchr(p2, ':'); *p3++ = 0;
            if (!strcmp(s, "eq")) pd->op = OP_EQ;
            else if (!strcmp(s, "ne")) pd->op = OP_NE;
            else if (!strcmp(s, "lt")) pd->op = OP_LT;
            else if (!strcmp(s, "le")) pd->op = OP_LE;
            else if (!strcmp(s, "gt")) pd->op = OP_GT;
            else if (!strcmp(s, "ge")) pd->op = OP_GE;
            else if (!strcmp(s, "between")) pd->op = OP_BETWEEN;
            else if (!strcmp(s, "contains")) pd->op = OP_CONTAINS;
            else die("bad pred op");
            strcpy(pd->col, p1);
            if (!strcmp(p2, "str")) { pd->is_str = 1; strcpy(pd->lit_s, p3); }
            else if (!strcmp(p2, "f64")) { pd->is_f64 = 1; pd->flit = atof(p3); }
            else {
                if (pd->op == OP_BETWEEN) sscanf(p3, "%" SCNd64 ",%" SCNd64, &pd->lo, &pd->hi);
                else pd->lo = strtoll(p3, NULL, 10);
            }
            if (pd->op == OP_CONTAINS) { pd->is_str = 1; strcpy(pd->lit_s, p3); }
        } else {
            q.files[q.n_files++] = argv[i];
        }
    }
    if (!q.n_files || !q.n_aggs) die("usage: cpu_ref [--time lo,hi] [--group-by a,b] --agg op[:col] [--pred op:col:kind:lit] files...");
    run_query(&q);
    return 0;
}
