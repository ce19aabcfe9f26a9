// gfx950 (CDNA4, MI355X) kernels for the Parseable hot path:
// LZ4_RAW page decompression, parquet RLE/dictionary + DELTA_BINARY_PACKED
// decode into device-resident column arrays, predicate masks, and hash
// group-by with count/sum/min/max.
//
// Design notes (HBM-bound integer/byte work — no MFMA, per the north star):
//  - wavefront = 64; one wave per page for the inherently-serial-per-page
//    decode chains (LZ4 token stream, RLE run headers), with all 64 lanes
//    doing the data movement of each run/sequence in parallel;
//  - parsing is LANE-REDUNDANT (all lanes read the same few header bytes —
//    a broadcast L1 load) instead of shfl choreography;
//  - LZ4 match copies read the last-64-KiB window from an LDS ring, so no
//    global store->load ordering waits are needed inside a page;
//  - row-parallel kernels (comparisons, aggregation) are grid-stride with
//    coalesced accesses; aggregation uses a per-block LDS table when the
//    group table fits, flushed once per block with global atomics.
#include <hip/hip_runtime.h>
#include <cstring>
#include <rocprim/device/device_radix_sort.hpp>
#include "dev_types.h"
#include "meta.h"

namespace gpuq {

#define WAVE 64

#define LZ4_RING 16384  // LDS window for the serial fallback resolver
#define LZ4_IN 4096     // LDS input window for the token parse

// ------------------------------------------------------------------
// Segment-parallel LZ4 (v7): the host's load-time structure walk
// (meta.cpp lz4_walk) splits each page into <=16KB-output segments at
// sequence boundaries; segments decompress IN PARALLEL, each wave
// assembling its segment in LDS (in-segment matches read the linear LDS
// image — no ring, no modulo) and flushing coalesced. Matches reaching
// before the segment start were deferred by the host as backref records;
// any in-segment match influenced by such a gap was deferred
// transitively, so phase-1 bytes under a backref dst are simply
// overwritten by phase 2 (k_lz4_backrefs), which resolves each page's
// records in dst order.
#define SEG_MAX 8192   // 8K output/segment: 12 blocks/CU in phase 1
__global__ void __launch_bounds__(WAVE)
k_lz4_seg(const uint8_t* __restrict__ raw, uint8_t* __restrict__ dec,
          const DevSeg* __restrict__ segs, int n, int32_t* __restrict__ d_error) {
  __shared__ uint8_t seg[SEG_MAX];
  __shared__ uint8_t inbuf[LZ4_IN + 256];
  int si = blockIdx.x;
  if (si >= n) return;
  const DevSeg sg = segs[si];
  const uint8_t* src = raw + sg.src_off;
  uint8_t* dst = dec + sg.dst_off;
  const int lane = threadIdx.x;

  if (sg.raw) {
    for (uint32_t i = lane * 16u; i < sg.out_len; i += WAVE * 16u) {
      uint32_t rem = sg.out_len - i;
      if (rem >= 16 && (((uintptr_t)(src + i)) & 15) == 0 && (((uintptr_t)(dst + i)) & 15) == 0)
        *(uint4*)(dst + i) = *(const uint4*)(src + i);
      else
        for (uint32_t b = 0; b < 16 && i + b < sg.out_len; b++) dst[i + b] = src[i + b];
    }
    return;
  }

  const uint32_t comp = sg.comp_len, uncomp = sg.out_len;
  uint32_t in_base = 0;
  bool in_valid = false;
  auto refill = [&](uint32_t pos) {
    in_base = pos & ~15u;
    uint32_t v[17];
#pragma unroll
    for (int k = 0; k < 17; k++)
      __builtin_memcpy(&v[k], src + in_base + lane * 4u + (uint32_t)k * (WAVE * 4u), 4);
#pragma unroll
    for (int k = 0; k < 17; k++)
      *(uint32_t*)&inbuf[lane * 4u + (uint32_t)k * (WAVE * 4u)] = v[k];
    __builtin_amdgcn_wave_barrier();
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    in_valid = true;
  };
  auto inb = [&](uint32_t pos) -> uint8_t {
    if (!in_valid || pos - in_base >= LZ4_IN) refill(pos);
    return inbuf[pos - in_base];
  };
  const uint32_t* in32 = (const uint32_t*)inbuf;
  uint32_t s = 0, d = 0;   // d: OUTPUT offset relative to segment start
  bool bad = false;
  const bool big = sg.big != 0;
  while (s < comp && d < uncomp) {
    if (!in_valid || s - in_base >= LZ4_IN) refill(s);
    uint32_t rel = s - in_base;
    uint32_t w[6];
#pragma unroll
    for (int k = 0; k < 6; k++) w[k] = in32[(rel >> 2) + k];
    uint32_t sub = rel & 3;
    auto gb = [&](uint32_t j) {
      uint32_t t = sub + j;
      return (w[t >> 2] >> ((t & 3) * 8)) & 0xffu;
    };
    uint32_t token = gb(0);
    uint32_t lit = token >> 4;
    uint32_t off, ml;
    if (lit < 15 && !big) {
      if (s + 1 + lit > comp || d + lit > uncomp) { bad = true; break; }
      if ((uint32_t)lane < lit)
        seg[d + lane] = (uint8_t)gb(1 + lane);
      __builtin_amdgcn_wave_barrier();
      s += 1 + lit; d += lit;
      if (s >= comp) break;
      if (s + 2 > comp) { bad = true; break; }
      off = gb(1 + lit) | (gb(2 + lit) << 8);
      s += 2;
      ml = token & 0xf;
      if (ml == 15) {
        uint32_t b;
        do { if (s >= comp) { bad = true; break; } b = inb(s); s++; ml += b; } while (b == 255);
        if (bad) break;
      }
    } else {
      // long literal (or BIG segment): chunked copy through the window
      s++;
      if (lit == 15) {
        uint32_t b;
        do { if (s >= comp) { bad = true; break; } b = inb(s); s++; lit += b; } while (b == 255);
        if (bad) break;
      }
      if (s + lit > comp || d + lit > uncomp) { bad = true; break; }
      uint32_t doneL = 0;
      while (doneL < lit) {
        if (!in_valid || (s + doneL) - in_base >= LZ4_IN) refill(s + doneL);
        uint32_t avail = LZ4_IN - ((s + doneL) - in_base);
        uint32_t chunk = min(lit - doneL, avail);
        if (big) {
          // stream to global (no LDS image for giant sequences)
          const uint8_t* lsrc = &inbuf[(s + doneL) - in_base];
          uint32_t base = d + doneL;
          for (uint32_t i = lane; i < chunk; i += WAVE) dst[base + i] = lsrc[i];
        } else {
          const uint8_t* lsrc = &inbuf[(s + doneL) - in_base];
          uint32_t base = d + doneL;
          for (uint32_t i = lane; i < chunk; i += WAVE) seg[base + i] = lsrc[i];
        }
        doneL += chunk;
      }
      __builtin_amdgcn_wave_barrier();
      s += lit; d += lit;
      if (s >= comp) break;
      if (s + 2 > comp) { bad = true; break; }
      off = inb(s) | ((uint32_t)inb(s + 1) << 8);
      s += 2;
      ml = token & 0xf;
      if (ml == 15) {
        uint32_t b2;
        do { if (s >= comp) { bad = true; break; } b2 = inb(s); s++; ml += b2; } while (b2 == 255);
        if (bad) break;
      }
    }
    ml += 4;
    if (d + ml > uncomp) { bad = true; break; }
    if (big || off > d) {
      // deferred by the host (backref) — leave the gap for phase 2
      d += ml;
      continue;
    }
    // in-segment match: linear LDS image, no wrap
    {
      uint32_t done = 0;
      __builtin_amdgcn_wave_barrier();
      while (done < ml) {
        uint32_t chunk = min(ml - done, off);
        uint32_t sbase = d + done - off;
        uint32_t dbase = d + done;
        for (uint32_t i = lane; i < chunk; i += WAVE)
          seg[dbase + i] = seg[sbase + i];
        __builtin_amdgcn_wave_barrier();
        done += chunk;
      }
      d += ml;
    }
  }
  if (bad || d != uncomp) {
    if (lane == 0) atomicExch(d_error, ERR_LZ4);
    return;
  }
  if (!big) {
    // coalesced flush LDS segment -> global (dst-aligned u32 body; segment
    // boundaries within a page are arbitrary byte offsets)
    __builtin_amdgcn_wave_barrier();
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    uint32_t head = (uint32_t)((4 - ((uintptr_t)dst & 3)) & 3);
    if (head > uncomp) head = uncomp;
    for (uint32_t i = lane; i < head; i += WAVE) dst[i] = seg[i];
    uint32_t body = (uncomp - head) & ~3u;
    for (uint32_t i = lane * 4u; i < body; i += WAVE * 4u) {
      uint32_t t = head + i;
      uint32_t sh = (t & 3) * 8;
      const uint32_t* s32 = (const uint32_t*)seg;
      uint32_t v = s32[t >> 2] >> sh;
      if (sh) v |= s32[(t >> 2) + 1] << (32 - sh);
      *(uint32_t*)(dst + t) = v;
    }
    for (uint32_t i = head + body + lane; i < uncomp; i += WAVE) dst[i] = seg[i];
  }
}

// phase 2: resolve each page's deferred matches in dst order, one wave per
// page. On pattern-heavy pages (PLAIN int columns) the transitive gap rule
// defers nearly every match, so this must NOT pay a memory drain per
// record: the wave keeps a sliding 16 KiB LDS image of the output window
// (invariant: slot b & MASK holds byte b for b in [wend-16K, wend)).
// Sources are always window-resident when off + len <= 16K (periodicity:
// out[dst+i] == window[src + i mod off]); record writes go through the
// window AND global, and single-wave LDS ordering replaces all drains.
// Window advances load only bytes this wave never wrote (later dst gaps
// get overwritten in LDS before any later record reads them).
// literal-resolved record resolvers: the host composed every deferred
// match's pattern down to PHASE-1 source pieces, so all records are
// mutually independent — one launch, no ordering. out[dst + rep*off +
// pat_off + j] = dec[piece.src + j].
__global__ void k_brres_lane(uint8_t* __restrict__ dec,
                             const DevBrRes* __restrict__ recs,
                             const DevPiece* __restrict__ pieces, int64_t n) {
  int64_t r = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (r >= n) return;
  const DevBrRes rec = recs[r];
  uint32_t pat_off = 0;
  for (uint32_t k = 0; k < rec.piece_n; k++) {
    const DevPiece pc = pieces[rec.piece_start + k];
    for (uint32_t base = pat_off; base < rec.len; base += rec.off) {
      uint32_t m = min(pc.len, rec.len - base);
      for (uint32_t j = 0; j < m; j++)
        dec[rec.dst + base + j] = dec[pc.src + j];
    }
    pat_off += pc.len;
  }
}
// inlined-pattern resolved matches: pure writes (records are dst-sorted,
// adjacent lanes write adjacent regions)
__global__ void k_brres_inl(uint8_t* __restrict__ dec,
                            const DevBrInl* __restrict__ recs, int64_t n) {
  int64_t r = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (r >= n) return;
  const DevBrInl R = recs[r];
  const uint64_t dst = R.meta & ((1ull << 40) - 1);
  const uint32_t len = (uint32_t)(R.meta >> 40) & 0xfff;
  const uint32_t period = (uint32_t)(R.meta >> 52);
  uint8_t* o = dec + dst;
  for (uint32_t j = 0; j < len; j++)
    o[j] = (uint8_t)(R.pat >> ((j % period) * 8));
}
__global__ void __launch_bounds__(WAVE)
k_brres_wave(uint8_t* __restrict__ dec, const DevBrRes* __restrict__ recs,
             const DevPiece* __restrict__ pieces, int n) {
  int r = blockIdx.x;
  if (r >= n) return;
  const DevBrRes rec = recs[r];
  uint32_t pat_off = 0;
  for (uint32_t k = 0; k < rec.piece_n; k++) {
    const DevPiece pc = pieces[rec.piece_start + k];
    for (uint32_t base = pat_off; base < rec.len; base += rec.off) {
      uint32_t m = min(pc.len, rec.len - base);
      for (uint32_t j = threadIdx.x; j < m; j += WAVE)
        dec[rec.dst + base + j] = dec[pc.src + j];
    }
    pat_off += pc.len;
  }
}

// litpar literal copies: one record per sequence's literal run (host walk,
// meta.cpp lz4_walk litpar mode). Records are independent; short runs get a
// lane each (records are dst-sorted, so adjacent lanes touch adjacent
// memory), long runs a wave.
__global__ void k_lit_lane(const uint8_t* __restrict__ raw,
                           uint8_t* __restrict__ dec,
                           const DevLit* __restrict__ lits, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  const DevLit L = lits[i];
  const uint8_t* s = raw + (L.a & ((1ull << 40) - 1));
  const uint32_t len = (uint32_t)(L.a >> 40);
  uint8_t* o = dec + L.b;
  uint32_t k = 0;
  for (; k + 8 <= len; k += 8) {
    uint64_t w;
    __builtin_memcpy(&w, s + k, 8);
    __builtin_memcpy(o + k, &w, 8);
  }
  for (; k < len; k++) o[k] = s[k];
}
__global__ void __launch_bounds__(WAVE)
k_lit_wave(const uint8_t* __restrict__ raw, uint8_t* __restrict__ dec,
           const DevLit* __restrict__ lits, int n) {
  if (blockIdx.x >= (unsigned)n) return;
  const DevLit L = lits[blockIdx.x];
  const uint8_t* s = raw + (L.a & ((1ull << 40) - 1));
  const uint32_t len = (uint32_t)(L.a >> 40);
  uint8_t* o = dec + L.b;
  uint32_t nw = len / 8;
  for (uint32_t w = threadIdx.x; w < nw; w += WAVE) {
    uint64_t v;
    __builtin_memcpy(&v, s + w * 8, 8);
    __builtin_memcpy(o + w * 8, &v, 8);
  }
  uint32_t t = nw * 8 + threadIdx.x;
  if (t < len) o[t] = s[t];
}

#define BR_WIN 16384
__global__ void __launch_bounds__(WAVE)
k_lz4_backrefs(uint8_t* __restrict__ dec, const DevBr* __restrict__ brs,
               const DevPageBr* __restrict__ pages, int n) {
  __shared__ uint8_t win[BR_WIN];
  int pi = blockIdx.x;
  if (pi >= n) return;
  const DevPageBr pb = pages[pi];
  const int lane = threadIdx.x;
  uint64_t page0 = pb.count ? brs[pb.start].dst : 0;  // absolute anchor
  // window covers [wend-16K, wend); start it just below the first record's
  // dst so the first advance pulls in the preceding (phase-1) bytes
  uint64_t wend = page0 > BR_WIN ? page0 - BR_WIN : 0;
  for (uint32_t r = 0; r < pb.count; r++) {
    const DevBr br = brs[pb.start + r];
    uint64_t off = br.dst - br.src;
    if (off + br.len > BR_WIN / 2) {  // window lookahead halves coverage
      // rare far/huge record: chunked global copy with drains
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      uint32_t done = 0;
      while (done < br.len) {
        uint32_t chunk = (uint32_t)min((uint64_t)(br.len - done), off);
        for (uint32_t i = lane; i < chunk; i += WAVE)
          dec[br.dst + done + i] = dec[br.src + done + i];
        done += chunk;
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
      // the global writes were not mirrored in the window: reload any
      // overlap between [dst, dst+len) and the current window
      {
        uint64_t wlo = wend > BR_WIN ? wend - BR_WIN : 0;
        uint64_t a = br.dst > wlo ? br.dst : wlo;
        uint64_t b = (br.dst + br.len) < wend ? (br.dst + br.len) : wend;
        if (a < b) {
          for (uint64_t x = a + lane; x < b; x += WAVE)
            win[x & (BR_WIN - 1)] = dec[x];
          __builtin_amdgcn_wave_barrier();
          asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        }
      }
      continue;
    }
    // advance the window in half-window steps (a per-record advance would
    // reintroduce a drain per record); lookahead bytes may be unresolved
    // gaps — later records overwrite their LDS slots before any read
    uint64_t need_end = br.dst + br.len;
    if (need_end > wend) {
      uint64_t target = need_end + BR_WIN / 2;
      uint64_t new_w0 = target > BR_WIN ? target - BR_WIN : 0;
      uint64_t load_from = wend > new_w0 ? wend : new_w0;
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      for (uint64_t b = load_from + lane; b < target; b += WAVE)
        win[b & (BR_WIN - 1)] = dec[b];
      __builtin_amdgcn_wave_barrier();
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      wend = target;
    }
    // sources resident: src + (i mod off) in [wend-16K, dst) always;
    // dst slots never alias source slots (the whole span fits the window)
    uint32_t off32 = (uint32_t)off;
    uint32_t pow2 = (off32 & (off32 - 1)) == 0;
    for (uint32_t i = lane; i < br.len; i += WAVE) {
      uint32_t j = pow2 ? (i & (off32 - 1)) : (i % off32);
      uint8_t v = win[(br.src + j) & (BR_WIN - 1)];
      dec[br.dst + i] = v;
      win[(br.dst + i) & (BR_WIN - 1)] = v;
    }
    __builtin_amdgcn_wave_barrier();
  }
}

// ------------------------------------------------------------------
// serial readers (lane-redundant or lane0) for slow paths
// ------------------------------------------------------------------
struct SerialRle {
  const uint8_t* p;
  const uint8_t* end;
  int bit_width, byte_w;
  // run state
  uint32_t run_left = 0;
  bool packed = false;
  uint32_t rle_val = 0;
  uint64_t bit_acc = 0;
  int bit_cnt = 0;
  uint32_t pending = 0;
  bool has_pending = false;
  __device__ SerialRle(const uint8_t* p_, const uint8_t* end_, int bw)
      : p(p_), end(end_), bit_width(bw), byte_w((bw + 7) / 8) {}
  __device__ void unread(uint32_t v) { pending = v; has_pending = true; }
  __device__ uint32_t varint() {
    uint64_t v = 0; int sh = 0;
    for (;;) {
      uint8_t b = (p < end) ? *p++ : 0;
      v |= (uint64_t)(b & 0x7f) << sh;
      if (!(b & 0x80)) return (uint32_t)v;
      sh += 7;
    }
  }
  __device__ uint32_t next() {
    if (has_pending) { has_pending = false; return pending; }
    if (!run_left) {
      uint32_t hdr = varint();
      if (hdr & 1) {
        packed = true;
        run_left = (hdr >> 1) * 8;
        bit_acc = 0; bit_cnt = 0;
      } else {
        packed = false;
        run_left = hdr >> 1;
        rle_val = 0;
        for (int b = 0; b < byte_w; b++)
          rle_val |= (uint32_t)((p < end) ? *p++ : 0) << (8 * b);
      }
    }
    run_left--;
    if (!packed) return rle_val;
    while (bit_cnt < bit_width) {
      bit_acc |= (uint64_t)((p < end) ? *p++ : 0) << bit_cnt;
      bit_cnt += 8;
    }
    uint32_t v = (uint32_t)(bit_acc & ((bit_width >= 32) ? 0xffffffffull
                                                         : ((1ull << bit_width) - 1)));
    bit_acc >>= bit_width;
    bit_cnt -= bit_width;
    return v;
  }
};

// def-level helpers: v1 page payload = [u32 len][RLE runs] when optional.
// Returns pointer to values; *all_valid set when a single RLE(1) run covers
// the page (the overwhelmingly common case for Parseable streams).
__device__ inline const uint8_t* def_levels(const DevPage& pg, const uint8_t* payload,
                                            const uint8_t** def_start, uint32_t* def_len,
                                            bool* all_valid) {
  if (!pg.optional) { *def_start = nullptr; *def_len = 0; *all_valid = true; return payload; }
  uint32_t dl;
  memcpy(&dl, payload, 4);
  *def_start = payload + 4;
  *def_len = dl;
  // quick probe: single RLE run of value 1 covering all values?
  const uint8_t* q = payload + 4;
  uint64_t hdr = 0; int sh = 0;
  for (;;) {
    uint8_t b = *q++;
    hdr |= (uint64_t)(b & 0x7f) << sh;
    if (!(b & 0x80)) break;
    sh += 7;
  }
  *all_valid = (!(hdr & 1)) && ((hdr >> 1) >= pg.num_values) && (*q == 1);
  return payload + 4 + dl;
}

// ------------------------------------------------------------------
// Wave-parallel definition-level decode for null-bearing pages: decodes
// the RLE/bit-packed def stream into (a) valid bytes, (b) a dense->row
// mapping rowof[row0 + k] = row of the k-th present value, and (c) the
// page's present count. Value decoders then run their PARALLEL paths over
// the dense stream and scatter through rowof — replacing the serial lane0
// fallback that cost ~8ms on a 262k-row 60%-null page.
// LDS: row bitmap 32KB (262,144 rows max) + 512 tile counts.
#define DEF_MAX_ROWS 262144
__global__ void __launch_bounds__(WAVE)
k_def_levels(const uint8_t* __restrict__ dec, const DevPage* __restrict__ pages,
             const int32_t* __restrict__ ids, int n,
             uint8_t* __restrict__ valid, uint8_t* __restrict__ null_mask,
             uint32_t* __restrict__ rowof, uint32_t* __restrict__ rankout,
             uint32_t* __restrict__ present, int32_t* d_error) {
  __shared__ uint64_t bits[DEF_MAX_ROWS / 64];
  __shared__ uint32_t tile_base[DEF_MAX_ROWS / 512];
  int pi = blockIdx.x;
  if (pi >= n) return;
  const DevPage pg = pages[ids[pi]];
  const int lane = threadIdx.x;
  if (pg.num_values > DEF_MAX_ROWS) {
    if (lane == 0) atomicExch(d_error, ERR_PAGE);
    return;
  }
  const uint8_t* payload = dec + pg.dst_off;
  {
    const uint8_t* ds; uint32_t dl_probe; bool av;
    def_levels(pg, payload, &ds, &dl_probe, &av);
    if (av) return;  // all-valid page: consumers take the direct fast path
  }
  uint32_t dl = 0;
  const uint8_t* p = payload;
  if (pg.optional) {
    memcpy(&dl, payload, 4);
    p = payload + 4;
  }
  const uint32_t nv = pg.num_values, row0 = pg.row_start;
  uint32_t nwords = (nv + 63) / 64;
  for (uint32_t w = lane; w < nwords; w += WAVE) bits[w] = 0;
  __builtin_amdgcn_wave_barrier();
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

  if (!pg.optional) {
    for (uint32_t w = lane; w < nwords; w += WAVE) bits[w] = ~0ull;
  } else {
    // lane-redundant run walk; parallel fills (def bit_width == 1)
    const uint8_t* end = p + dl;
    uint32_t v = 0;
    while (v < nv && p < end) {
      uint64_t hdr = 0; int sh = 0;
      for (;;) {
        uint8_t b = *p++;
        hdr |= (uint64_t)(b & 0x7f) << sh;
        if (!(b & 0x80)) break;
        sh += 7;
      }
      if (hdr & 1) {
        uint32_t groups = (uint32_t)(hdr >> 1);   // 8 values per byte
        for (uint32_t g = lane; g < groups; g += WAVE) {
          uint8_t byte = p[g];
          uint32_t base = v + g * 8;
          if (byte && base < nv) {
            // base need not be 8-aligned (an RLE run of arbitrary length may
            // precede) — clip to nv and spill across the word boundary
            uint64_t bb = byte;
            if (base + 8 > nv) bb &= (1ull << (nv - base)) - 1;
            uint32_t w0 = base >> 6, sh = base & 63;
            if (bb << sh)
              atomicOr((unsigned long long*)&bits[w0],
                       (unsigned long long)(bb << sh));
            if (sh && (bb >> (64 - sh)))
              atomicOr((unsigned long long*)&bits[w0 + 1],
                       (unsigned long long)(bb >> (64 - sh)));
          }
        }
        p += groups;
        uint32_t add = groups * 8;
        v += (add > nv - v) ? (nv - v) : add;
      } else {
        uint32_t cnt = (uint32_t)(hdr >> 1);
        uint8_t val = *p++;
        if (cnt > nv - v) cnt = nv - v;
        if (val) {
          // set bits [v, v+cnt)
          for (uint32_t w = lane; w * 64 < v + cnt; w += WAVE) {
            uint32_t lo = w * 64, hi = lo + 64;
            if (hi <= v || lo >= v + cnt) continue;
            uint64_t m = ~0ull;
            if (v > lo) m &= ~0ull << (v - lo);
            if (v + cnt < hi) m &= ~0ull >> (hi - (v + cnt));
            atomicOr((unsigned long long*)&bits[w], (unsigned long long)m);
          }
        }
        v += cnt;
      }
      __builtin_amdgcn_wave_barrier();
    }
    // mask tail bits beyond nv
    if (lane == 0 && (nv & 63)) bits[nv >> 6] &= (~0ull >> (64 - (nv & 63)));
  }
  __builtin_amdgcn_wave_barrier();
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

  // tile counts (8 words = 512 rows per tile) + serial scan
  uint32_t ntiles = (nv + 511) / 512;
  for (uint32_t t = lane; t < ntiles; t += WAVE) {
    uint32_t c = 0;
    for (uint32_t w = t * 8; w < (t + 1) * 8 && w < nwords; w++)
      c += (uint32_t)__popcll(bits[w]);
    tile_base[t] = c;
  }
  __builtin_amdgcn_wave_barrier();
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  if (lane == 0) {
    uint32_t run = 0;
    for (uint32_t t = 0; t < ntiles; t++) {
      uint32_t c = tile_base[t];
      tile_base[t] = run;
      run += c;
    }
    present[ids[pi]] = run;
  }
  __builtin_amdgcn_wave_barrier();
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

  // emit: one WORD per wave iteration, one ROW per lane — valid/rank stores
  // are 64/256 consecutive bytes per instruction (the per-lane tile walk
  // this replaces scattered byte stores 512 rows apart); rank within the
  // word from a lane-masked popcount, word prefix from tile_base + at most
  // 7 sibling-word popcounts.
  const uint64_t lmask = (1ull << lane) - 1;
  for (uint32_t w = 0; w < nwords; w++) {
    uint64_t b = bits[w];
    uint32_t base = tile_base[w >> 3];
    for (uint32_t ww = w & ~7u; ww < w; ww++)
      base += (uint32_t)__popcll(bits[ww]);
    uint32_t r = w * 64 + lane;
    if (r >= nv) break;
    uint8_t ok = (uint8_t)((b >> lane) & 1);
    valid[row0 + r] = ok;
    if (null_mask && !ok) null_mask[row0 + r] = 0;  // NULL never matches
    uint32_t rk = base + (uint32_t)__popcll(b & lmask);
    if (rankout) rankout[row0 + r] = rk;  // dense index (valid rows)
    if (rowof && ok) rowof[row0 + rk] = row0 + r;
  }
}

// ------------------------------------------------------------------
// dictionary-index decode (RLE/bit-packed hybrid), one wave per page.
// Emit policy via template: remap-to-gid, gather-i64, gather-f64, LUT-mask.
// ------------------------------------------------------------------
struct EmitGid {
  const int32_t* remap;
  int32_t* out;        // row-aligned gid, 0 = NULL
  uint8_t* valid;      // optional validity bytes (for COUNT(utf8_col))
  __device__ void operator()(uint32_t row, uint32_t idx) const {
    out[row] = remap[idx];
    if (valid) valid[row] = 1;
  }
  __device__ void null_at(uint32_t row) const { out[row] = 0; if (valid) valid[row] = 0; }
};
struct EmitDictI64 {
  const int64_t* dictv;
  int64_t* out;
  uint8_t* valid;
  __device__ void operator()(uint32_t row, uint32_t idx) const { out[row] = dictv[idx]; if (valid) valid[row] = 1; }
  __device__ void null_at(uint32_t row) const { if (valid) valid[row] = 0; }
};
struct EmitDictMask {
  const uint8_t* lut;  // per-local-dict-id pred result
  uint8_t* mask;
  __device__ void operator()(uint32_t row, uint32_t idx) const { mask[row] &= lut[idx]; }
  __device__ void null_at(uint32_t row) const { mask[row] = 0; }  // NULL never matches
};

// mode 0 (DIRECT): all-valid pages only — decode straight to the final
// row-aligned arrays. mode 1 (SCRATCH): null-bearing pages only — decode
// the dense stream to scratch at [row0+k]; a separate expansion kernel
// then writes the final arrays COALESCED through the k_def_levels rank map
// (a scatter here would cost a read-modify-write line fetch per value).
// Workgroups carry DP_WAVES waves per page: every wave walks the (cheap,
// lane-redundant) run headers, but runs are unpacked round-robin by wave —
// at 1 B-row scale a page-per-wave launch was ~1.9k waves, far too few to
// hide latency on the unpack loads.
#define DP_WAVES 4
template <class Emit>
__device__ void dict_page_decode(const DevPage& pg, const uint8_t* payload, Emit emit,
                                 const uint32_t* __restrict__ present,
                                 int32_t page_id, int mode, int32_t* d_error) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const uint8_t* def_start; uint32_t def_len; bool all_valid;
  const uint8_t* vals = def_levels(pg, payload, &def_start, &def_len, &all_valid);
  uint32_t nv = pg.num_values;
  const uint32_t row0 = pg.row_start;
  if (mode == 0) {
    if (!all_valid) return;
  } else {
    if (all_valid) return;
    nv = present[page_id];
  }

  {
    int bw = *vals++;
    auto target = [&](uint32_t k) { return row0 + k; };
    if (bw == 0) {  // all values are dict id 0
      for (uint32_t i = threadIdx.x; i < nv; i += WAVE * DP_WAVES)
        emit(target(i), 0);
      return;
    }
    // lane-redundant run-header walk; data movement parallel per run,
    // runs striped across the block's waves
    const uint8_t* p = vals;
    const uint32_t dict_n = pg.dict_n;
    uint32_t v = 0;
    uint32_t rc = 0;
    while (v < nv) {
      // varint header (redundant on all lanes)
      uint64_t hdr = 0; int sh = 0;
      for (;;) {
        uint8_t b = *p++;
        hdr |= (uint64_t)(b & 0x7f) << sh;
        if (!(b & 0x80)) break;
        sh += 7;
      }
      if (hdr & 1) {
        uint32_t groups = (uint32_t)(hdr >> 1);
        // one VALUE per lane (adjacent lanes -> adjacent rows): the emit
        // stores coalesce into full lines. The group-of-8-per-lane layout
        // this replaces made every store instruction span 64 half-used
        // lines (PMC: 2.6x write amplification); the redundant group-byte
        // reads (8 lanes share a group) stay in L1/L2.
        uint32_t run_vals = groups * 8;
        if (run_vals > nv - v) run_vals = nv - v;
        if ((rc++ % DP_WAVES) == (uint32_t)wave)
        for (uint32_t i = lane; i < run_vals; i += WAVE) {
          uint32_t g = i >> 3;
          int k = (int)(i & 7);
          const uint8_t* q = p + (size_t)g * bw;
          uint32_t mask_v = (bw >= 32) ? 0xffffffffu : ((1u << bw) - 1);
          // one unaligned 8B load replaces the byte loop: bits beyond the
          // group's bw bytes are never selected (k*bw+bw <= 64 => the used
          // window lies inside q[0..8)); arena padding covers the over-read
          uint32_t idx;
          if (k * bw + bw <= 64) {
            uint64_t acc;
            __builtin_memcpy(&acc, q, 8);
            idx = (uint32_t)(acc >> (k * bw)) & mask_v;
          } else {  // value straddles the first u64 window: shifted re-read
            uint64_t acc2;
            __builtin_memcpy(&acc2, q + (k * bw) / 8, 8);
            idx = (uint32_t)(acc2 >> ((k * bw) % 8)) & mask_v;
          }
          if (idx >= dict_n) { atomicExch(d_error, ERR_DICT_RANGE); idx = 0; }
          emit(target(v + i), idx);
        }
        p += (size_t)groups * bw;
        uint32_t add = groups * 8;
        v += (add > nv - v) ? (nv - v) : add;
      } else {
        uint32_t cnt = (uint32_t)(hdr >> 1);
        uint32_t val = 0;
        int byte_w = (bw + 7) / 8;
        for (int b = 0; b < byte_w; b++) val |= (uint32_t)p[b] << (8 * b);
        p += byte_w;
        if (cnt > nv - v) cnt = nv - v;
        if (val >= dict_n) { if (threadIdx.x == 0) atomicExch(d_error, ERR_DICT_RANGE); val = 0; }
        if ((rc++ % DP_WAVES) == (uint32_t)wave)
        for (uint32_t i = lane; i < cnt; i += WAVE) emit(target(v + i), val);
        v += cnt;
      }
    }
  }
}

template <class Emit>
__global__ void __launch_bounds__(WAVE * DP_WAVES)
k_dict_pages(const uint8_t* __restrict__ dec, const DevPage* __restrict__ pages,
             const int32_t* __restrict__ ids, int n, Emit emit,
             const uint32_t* present, int mode, int32_t* d_error) {
  int pi = blockIdx.x;
  if (pi >= n) return;
  DevPage pg = pages[ids[pi]];
  Emit e = emit;
  e.advance(pg);  // per-page aux pool offsets
  dict_page_decode(pg, dec + pg.dst_off, e, present, ids[pi], mode, d_error);
}

// wrappers adding per-page aux advance
struct EmitGidP : EmitGid {
  const int32_t* pool;
  __device__ void advance(const DevPage& pg) { remap = pool + pg.aux; }
};
struct EmitDictI64P : EmitDictI64 {
  const int64_t* pool;
  __device__ void advance(const DevPage& pg) { dictv = pool + pg.aux_val; }
};
struct EmitDictMaskP : EmitDictMask {
  const uint8_t* pool;
  __device__ void advance(const DevPage& pg) { lut = pool + pg.aux_lut; }
};
struct EmitLutD {  // scratch mode: STORE the per-value LUT byte densely
  const uint8_t* pool;
  const uint8_t* lut;
  uint8_t* scr;
  __device__ void operator()(uint32_t row, uint32_t idx) const { scr[row] = lut[idx]; }
  __device__ void null_at(uint32_t) const {}
  __device__ void advance(const DevPage& pg) { lut = pool + pg.aux_lut; }
};

// ------------------------------------------------------------------
// FUSED count path: GROUP BY <single dict column> + count(*) only, no
// predicates — decode RLE/bit-packed indices straight into a per-block
// LDS histogram over global key ids; no gid materialization, no mask,
// no separate aggregation pass. Table layout matches k_agg
// (slots = 1 + 2*n_aggs, every agg is COUNT_STAR).
// ------------------------------------------------------------------
#define FUSED_MAX_GROUPS 8192
__global__ void __launch_bounds__(WAVE)
k_dict_count(const uint8_t* __restrict__ dec, const DevPage* __restrict__ pages,
             const int32_t* __restrict__ ids, int n,
             const int32_t* __restrict__ remap_pool,
             uint64_t* __restrict__ table, int32_t n_groups, int n_aggs,
             int32_t* d_error) {
  extern __shared__ uint64_t hist[];   // n_groups counters
  const int lane = threadIdx.x;
  for (int g = lane; g < n_groups; g += WAVE) hist[g] = 0;
  __syncthreads();

  // several pages per block: pages n mapped gridDim-strided
  for (int pi = blockIdx.x; pi < n; pi += gridDim.x) {
    DevPage pg = pages[ids[pi]];
    const int32_t* remap = remap_pool + pg.aux;
    const uint8_t* payload = dec + pg.dst_off;
    const uint8_t* def_start; uint32_t def_len; bool all_valid;
    const uint8_t* vals = def_levels(pg, payload, &def_start, &def_len, &all_valid);
    const uint32_t nv = pg.num_values;
    const uint32_t dict_n = pg.dict_n;
    if (!all_valid) {
      if (lane == 0) {
        SerialRle def(def_start, def_start + def_len, 1);
        int bw = *vals++;
        SerialRle idx(vals, payload + pg.uncomp_size, bw);
        for (uint32_t r = 0; r < nv; r++) {
          if (def.next()) {
            uint32_t ix = bw ? idx.next() : 0;
            if (ix >= dict_n) { atomicExch(d_error, ERR_DICT_RANGE); ix = 0; }
            atomicAdd((unsigned long long*)&hist[remap[ix]], 1ull);
          } else {
            atomicAdd((unsigned long long*)&hist[0], 1ull);  // NULL group
          }
        }
      }
      continue;
    }
    int bw = *vals++;
    if (bw == 0) {
      if (lane == 0) atomicAdd((unsigned long long*)&hist[remap[0]], (unsigned long long)nv);
      continue;
    }
    const uint8_t* p = vals;
    uint32_t v = 0;
    while (v < nv) {
      uint64_t hdr = 0; int sh = 0;
      for (;;) {
        uint8_t b = *p++;
        hdr |= (uint64_t)(b & 0x7f) << sh;
        if (!(b & 0x80)) break;
        sh += 7;
      }
      if (hdr & 1) {
        uint32_t groups = (uint32_t)(hdr >> 1);
        for (uint32_t g = lane; g < groups; g += WAVE) {
          const uint8_t* q = p + (size_t)g * bw;
          uint64_t acc = 0;
          for (int b = 0; b < bw && b < 8; b++) acc |= (uint64_t)q[b] << (8 * b);
          uint32_t base = v + g * 8;
          uint32_t mask_v = (bw >= 32) ? 0xffffffffu : ((1u << bw) - 1);
          for (int k = 0; k < 8; k++) {
            uint32_t idx;
            if (k * bw + bw <= 64) {
              idx = (uint32_t)(acc >> (k * bw)) & mask_v;
            } else {
              uint64_t acc2 = 0;
              const uint8_t* q2 = q + (k * bw) / 8;
              int shift = (k * bw) % 8;
              for (int b = 0; b < 8; b++) acc2 |= (uint64_t)q2[b] << (8 * b);
              idx = (uint32_t)(acc2 >> shift) & mask_v;
            }
            if (base + k < nv) {
              if (idx >= dict_n) { atomicExch(d_error, ERR_DICT_RANGE); idx = 0; }
              atomicAdd((unsigned long long*)&hist[remap[idx]], 1ull);
            }
          }
        }
        p += (size_t)groups * bw;
        uint32_t add = groups * 8;
        v += (add > nv - v) ? (nv - v) : add;
      } else {
        uint32_t cnt = (uint32_t)(hdr >> 1);
        uint32_t val = 0;
        int byte_w = (bw + 7) / 8;
        for (int b = 0; b < byte_w; b++) val |= (uint32_t)p[b] << (8 * b);
        p += byte_w;
        if (cnt > nv - v) cnt = nv - v;
        if (val >= dict_n) { if (lane == 0) atomicExch(d_error, ERR_DICT_RANGE); val = 0; }
        if (lane == 0) atomicAdd((unsigned long long*)&hist[remap[val]], (unsigned long long)cnt);
        v += cnt;
      }
    }
  }
  __syncthreads();
  // flush: presence + every (COUNT_STAR) agg count slot
  int slots = 1 + 2 * n_aggs;
  for (int g = lane; g < n_groups; g += WAVE) {
    uint64_t h = hist[g];
    if (!h) continue;
    atomicAdd((unsigned long long*)&table[(int64_t)g * slots], (unsigned long long)h);
    for (int a = 0; a < n_aggs; a++)
      atomicAdd((unsigned long long*)&table[(int64_t)g * slots + 2 + 2 * a],
                (unsigned long long)h);
  }
}

// ------------------------------------------------------------------
// PLAIN i64 / f64 pages -> row-aligned arrays
// ------------------------------------------------------------------
__global__ void __launch_bounds__(WAVE)
k_plain_fixed(const uint8_t* __restrict__ dec, const DevPage* __restrict__ pages,
              const int32_t* __restrict__ ids, int n,
              int64_t* __restrict__ out, uint8_t* __restrict__ valid,
              const uint32_t* __restrict__ present, int mode,
              int32_t* d_error) {
  int pi = blockIdx.x;
  if (pi >= n) return;
  const DevPage pg = pages[ids[pi]];
  const int lane = threadIdx.x;
  const uint8_t* def_start; uint32_t def_len; bool all_valid;
  const uint8_t* vals = def_levels(pg, dec + pg.dst_off, &def_start, &def_len, &all_valid);
  const uint32_t row0 = pg.row_start;
  if (mode == 0) {
    if (!all_valid) return;
    uint32_t nv = pg.num_values;
    for (uint32_t i = lane; i < nv; i += WAVE) {
      int64_t v;
      memcpy(&v, vals + (size_t)i * 8, 8);
      out[row0 + i] = v;
      if (valid) valid[row0 + i] = 1;
    }
  } else {
    if (all_valid) return;
    uint32_t nd = present[ids[pi]];   // dense to scratch; expansion follows
    for (uint32_t i = lane; i < nd; i += WAVE) {
      int64_t v;
      memcpy(&v, vals + (size_t)i * 8, 8);
      out[row0 + i] = v;
    }
  }
}

// ------------------------------------------------------------------
// expansion: null-bearing pages only — coalesced row-side pass writing the
// final arrays from the dense scratch through the k_def_levels rank map.
// mode 0: i64 value (valid written by k_def_levels); mode 1: gid i32
// (NULL -> 0); mode 2: predicate mask AND (NULL never matches).
// ------------------------------------------------------------------
__global__ void __launch_bounds__(WAVE)
k_expand(const uint8_t* __restrict__ dec, const DevPage* __restrict__ pages,
         const int32_t* __restrict__ ids, int n,
         const uint8_t* __restrict__ scr, const uint32_t* __restrict__ rank,
         const uint8_t* __restrict__ valid, uint8_t* __restrict__ out,
         int mode) {
  int pi = blockIdx.x;
  if (pi >= n) return;
  const DevPage pg = pages[ids[pi]];
  const uint8_t* def_start; uint32_t def_len; bool all_valid;
  def_levels(pg, dec + pg.dst_off, &def_start, &def_len, &all_valid);
  if (all_valid) return;
  const uint32_t row0 = pg.row_start, nv = pg.num_values;
  if (mode == 0) {
    const int64_t* s = (const int64_t*)scr;
    int64_t* o = (int64_t*)out;
    for (uint32_t r = threadIdx.x; r < nv; r += WAVE)
      if (valid[row0 + r]) o[row0 + r] = s[row0 + rank[row0 + r]];
  } else if (mode == 1) {
    const int32_t* s = (const int32_t*)scr;
    int32_t* o = (int32_t*)out;
    for (uint32_t r = threadIdx.x; r < nv; r += WAVE)
      o[row0 + r] = valid[row0 + r] ? s[row0 + rank[row0 + r]] : 0;
  } else {
    for (uint32_t r = threadIdx.x; r < nv; r += WAVE)
      out[row0 + r] &= valid[row0 + r] ? scr[row0 + rank[row0 + r]] : 0;
  }
}

// ------------------------------------------------------------------
// Raw-byte utf8 machinery (hash group-by / predicates / min-max over
// columns with PLAIN-fallback pages — the dict-only gid path cannot cover
// them). A row's string is a STRREF: the absolute dec-arena offset of its
// [u32 len][bytes] record — dict entries and PLAIN values both live in the
// arena (dict pages of hash columns are decompressed on device too).
// ------------------------------------------------------------------

// PLAIN byte-array pages: the host walked the [len][bytes] chain at plan
// time and left one absolute strref per (non-null) value in the pool;
// emit them row-aligned (mode 0 direct / mode 1 dense-scratch + k_expand,
// exactly the k_plain_fixed contract).
__global__ void __launch_bounds__(WAVE)
k_pool_vals(const uint8_t* __restrict__ dec, const DevPage* __restrict__ pages,
            const int32_t* __restrict__ ids, int n,
            const int64_t* __restrict__ pool,
            int64_t* __restrict__ out, uint8_t* __restrict__ valid,
            const uint32_t* __restrict__ present, int mode) {
  int pi = blockIdx.x;
  if (pi >= n) return;
  const DevPage pg = pages[ids[pi]];
  const int lane = threadIdx.x;
  const uint8_t* def_start; uint32_t def_len; bool all_valid;
  def_levels(pg, dec + pg.dst_off, &def_start, &def_len, &all_valid);
  const uint32_t row0 = pg.row_start;
  const int64_t* src = pool + pg.aux_val;
  if (mode == 0) {
    if (!all_valid) return;
    for (uint32_t i = lane; i < pg.num_values; i += WAVE) {
      out[row0 + i] = src[i];
      if (valid) valid[row0 + i] = 1;
    }
  } else {
    if (all_valid) return;
    uint32_t nd = present[ids[pi]];
    for (uint32_t i = lane; i < nd; i += WAVE) out[row0 + i] = src[i];
  }
}

__device__ inline uint32_t ref_len(const uint8_t* dec, uint64_t ref) {
  uint32_t l;
  __builtin_memcpy(&l, dec + ref, 4);
  return l;
}
__device__ inline bool ref_eq(const uint8_t* dec, uint64_t a, uint64_t b) {
  if (a == b) return true;
  uint32_t la = ref_len(dec, a), lb = ref_len(dec, b);
  if (la != lb) return false;
  const uint8_t* pa = dec + a + 4;
  const uint8_t* pb = dec + b + 4;
  uint32_t i = 0;
  for (; i + 8 <= la; i += 8) {
    uint64_t wa, wb;
    __builtin_memcpy(&wa, pa + i, 8);
    __builtin_memcpy(&wb, pb + i, 8);
    if (wa != wb) return false;
  }
  for (; i < la; i++)
    if (pa[i] != pb[i]) return false;
  return true;
}
// lexicographic byte order (utf8 min/max semantics): a < b
__device__ inline int ref_cmp(const uint8_t* dec, uint64_t a, uint64_t b) {
  if (a == b) return 0;
  uint32_t la = ref_len(dec, a), lb = ref_len(dec, b);
  const uint8_t* pa = dec + a + 4;
  const uint8_t* pb = dec + b + 4;
  uint32_t n = la < lb ? la : lb;
  for (uint32_t i = 0; i < n; i++) {
    if (pa[i] != pb[i]) return pa[i] < pb[i] ? -1 : 1;
  }
  return la == lb ? 0 : (la < lb ? -1 : 1);
}

__device__ inline uint64_t ref_hash(const uint8_t* dec, uint64_t ref) {
  uint32_t len = ref_len(dec, ref);
  const uint8_t* p = dec + ref + 4;
  uint64_t h = 0xcbf29ce484222325ull ^ len;
  uint32_t i = 0;
  for (; i + 8 <= len; i += 8) {
    uint64_t w;
    __builtin_memcpy(&w, p + i, 8);
    h = (h ^ w) * 0x100000001b3ull;
    h ^= h >> 29;
  }
  uint64_t tail = 0;
  for (uint32_t k = 0; i < len; i++, k += 8) tail |= (uint64_t)p[i] << k;
  h = (h ^ tail) * 0x100000001b3ull;
  h ^= h >> 32;
  return h;
}

#define HREF_EMPTY (~0ull)

// pass 1: claim distinct strings. Open addressing, linear probe; the slot
// key is the FIRST strref seen for the string (byte-equality dedups dict
// entries vs PLAIN occurrences of the same value). Claimers draw a dense
// gid from *counter and record gid2ref; hgids stores are completed before
// pass 2 runs (kernel boundary), so lookup never spins.
__global__ void k_hash_build(const uint8_t* __restrict__ dec,
                             const int64_t* __restrict__ refs,
                             const uint8_t* __restrict__ valid, int64_t n_rows,
                             uint64_t* __restrict__ hkeys,
                             int32_t* __restrict__ hgids, int clog2,
                             uint32_t* counter, uint64_t* __restrict__ gid2ref,
                             int32_t gid_cap, int32_t* d_error) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const uint64_t mask = (1ull << clog2) - 1;
  for (; i < n_rows; i += stride) {
    if (valid && !valid[i]) continue;
    uint64_t ref = (uint64_t)refs[i];
    uint64_t slot = ref_hash(dec, ref) & mask;
    for (uint32_t probe = 0; ; probe++) {
      if (probe > (1u << clog2)) { atomicExch(d_error, ERR_HASH_PROBE); return; }
      uint64_t old = atomicCAS((unsigned long long*)&hkeys[slot],
                               (unsigned long long)HREF_EMPTY,
                               (unsigned long long)ref);
      if (old == HREF_EMPTY) {
        uint32_t g = atomicAdd(counter, 1u);
        if ((int32_t)g >= gid_cap) { atomicExch(d_error, ERR_HASH_CAP); return; }
        gid2ref[g] = ref;
        hgids[slot] = (int32_t)g;
        break;
      }
      if (ref_eq(dec, old, ref)) break;
      slot = (slot + 1) & mask;
    }
  }
}

// pass 2: resolve every row's gid (1-based; 0 = NULL group)
__global__ void k_hash_lookup(const uint8_t* __restrict__ dec,
                              const int64_t* __restrict__ refs,
                              const uint8_t* __restrict__ valid, int64_t n_rows,
                              const uint64_t* __restrict__ hkeys,
                              const int32_t* __restrict__ hgids, int clog2,
                              int32_t* __restrict__ out_gid) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const uint64_t mask = (1ull << clog2) - 1;
  for (; i < n_rows; i += stride) {
    if (valid && !valid[i]) { out_gid[i] = 0; continue; }
    uint64_t ref = (uint64_t)refs[i];
    uint64_t slot = ref_hash(dec, ref) & mask;
    for (;;) {
      uint64_t k = hkeys[slot];
      if (k == HREF_EMPTY) { out_gid[i] = 0; break; }  // unreachable
      if (ref_eq(dec, k, ref)) { out_gid[i] = hgids[slot] + 1; break; }
      slot = (slot + 1) & mask;
    }
  }
}

// pair cascade: when the dense per-key product space exceeds the group cap,
// keys combine two at a time — (combined_so_far, next_gid) packs into one
// u64 (each side < 2^32), claimed with a single-word CAS into the shared
// table. Exact (key compare is ==), no spinning (two-pass like the string
// hash). Mirrors DataFusion's row-hash over arbitrary key tuples.
__device__ inline uint64_t mix64(uint64_t x) {
  x ^= x >> 33;
  x *= 0xff51afd7ed558ccdull;
  x ^= x >> 33;
  x *= 0xc4ceb9fe1a85ec53ull;
  x ^= x >> 33;
  return x;
}
__global__ void k_pair_build(const int32_t* __restrict__ a,
                             const int32_t* __restrict__ b, int64_t n_rows,
                             uint64_t* __restrict__ hkeys,
                             int32_t* __restrict__ hgids, int clog2,
                             uint32_t* counter, uint64_t* __restrict__ gid2pair,
                             int32_t gid_cap, int32_t* d_error) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const uint64_t mask = (1ull << clog2) - 1;
  for (; i < n_rows; i += stride) {
    uint64_t key = ((uint64_t)(uint32_t)a[i] << 32) | (uint32_t)b[i];
    uint64_t slot = mix64(key) & mask;
    for (uint32_t probe = 0;; probe++) {
      if (probe > (1u << clog2)) { atomicExch(d_error, ERR_PAIR_PROBE); return; }
      uint64_t old = atomicCAS((unsigned long long*)&hkeys[slot],
                               (unsigned long long)HREF_EMPTY,
                               (unsigned long long)key);
      if (old == HREF_EMPTY) {
        uint32_t g = atomicAdd(counter, 1u);
        if ((int32_t)g >= gid_cap) { atomicExch(d_error, ERR_PAIR_CAP); return; }
        gid2pair[g] = key;
        hgids[slot] = (int32_t)g;
        break;
      }
      if (old == key) break;
      slot = (slot + 1) & mask;
    }
  }
}
__global__ void k_pair_lookup(const int32_t* __restrict__ a,
                              const int32_t* __restrict__ b, int64_t n_rows,
                              const uint64_t* __restrict__ hkeys,
                              const int32_t* __restrict__ hgids, int clog2,
                              int32_t* __restrict__ out) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const uint64_t mask = (1ull << clog2) - 1;
  for (; i < n_rows; i += stride) {
    uint64_t key = ((uint64_t)(uint32_t)a[i] << 32) | (uint32_t)b[i];
    uint64_t slot = mix64(key) & mask;
    for (;;) {
      uint64_t k = hkeys[slot];
      if (k == key) { out[i] = hgids[slot]; break; }
      if (k == HREF_EMPTY) { out[i] = 0; break; }  // unreachable
      slot = (slot + 1) & mask;
    }
  }
}

// numeric (i64) group keys: dense gids via the same two-pass claim/lookup
// structure (the reference's engine groups by any column; utf8 keys get
// strref hashing, numeric keys hash the 64-bit value itself). The all-ones
// bit pattern is both a valid value (-1) and the EMPTY sentinel, so rows
// with value -1 claim a dedicated slot at index 2^clog2 instead of probing.
// f64 keys group by VALUE equality under DataFusion's row-format
// normalization: -0.0 folds into +0.0 and every NaN into one canonical
// bit pattern, so bit-equality matches SQL group semantics.
__device__ inline uint64_t numkey_norm(uint64_t key, int is_f64) {
  if (!is_f64) return key;
  if ((key & 0x7fffffffffffffffull) == 0) return 0;              // -0.0 -> +0.0
  if ((key & 0x7ff0000000000000ull) == 0x7ff0000000000000ull &&
      (key & 0x000fffffffffffffull) != 0)
    return 0x7ff8000000000000ull;                                // canonical NaN
  return key;
}

__global__ void k_numhash_build(const int64_t* __restrict__ vals,
                                const uint8_t* __restrict__ valid,
                                int64_t n_rows, uint64_t* __restrict__ hkeys,
                                int32_t* __restrict__ hgids, int clog2,
                                uint32_t* counter,
                                uint64_t* __restrict__ gid2key,
                                int32_t gid_cap, int is_f64, int32_t* d_error) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const uint64_t mask = (1ull << clog2) - 1;
  const uint64_t cap_slot = 1ull << clog2;
  for (; i < n_rows; i += stride) {
    if (valid && !valid[i]) continue;
    uint64_t key = numkey_norm((uint64_t)vals[i], is_f64);
    if (key == HREF_EMPTY) {  // value -1: the dedicated overflow slot
      uint64_t old = atomicCAS((unsigned long long*)&hkeys[cap_slot],
                               (unsigned long long)HREF_EMPTY, 0ull);
      if (old == HREF_EMPTY) {
        uint32_t g = atomicAdd(counter, 1u);
        if ((int32_t)g >= gid_cap) { atomicExch(d_error, ERR_HASH_CAP); return; }
        gid2key[g] = key;
        hgids[cap_slot] = (int32_t)g;
      }
      continue;
    }
    uint64_t slot = mix64(key) & mask;
    for (uint32_t probe = 0;; probe++) {
      if (probe > (1u << clog2)) { atomicExch(d_error, ERR_HASH_PROBE); return; }
      uint64_t old = atomicCAS((unsigned long long*)&hkeys[slot],
                               (unsigned long long)HREF_EMPTY,
                               (unsigned long long)key);
      if (old == HREF_EMPTY) {
        uint32_t g = atomicAdd(counter, 1u);
        if ((int32_t)g >= gid_cap) { atomicExch(d_error, ERR_HASH_CAP); return; }
        gid2key[g] = key;
        hgids[slot] = (int32_t)g;
        break;
      }
      if (old == key) break;
      slot = (slot + 1) & mask;
    }
  }
}
__global__ void k_numhash_lookup(const int64_t* __restrict__ vals,
                                 const uint8_t* __restrict__ valid,
                                 int64_t n_rows,
                                 const uint64_t* __restrict__ hkeys,
                                 const int32_t* __restrict__ hgids, int clog2,
                                 int is_f64, int32_t* __restrict__ out_gid) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const uint64_t mask = (1ull << clog2) - 1;
  const uint64_t cap_slot = 1ull << clog2;
  for (; i < n_rows; i += stride) {
    if (valid && !valid[i]) { out_gid[i] = 0; continue; }
    uint64_t key = numkey_norm((uint64_t)vals[i], is_f64);
    if (key == HREF_EMPTY) { out_gid[i] = hgids[cap_slot] + 1; continue; }
    uint64_t slot = mix64(key) & mask;
    for (;;) {
      uint64_t k = hkeys[slot];
      if (k == key) { out_gid[i] = hgids[slot] + 1; break; }
      if (k == HREF_EMPTY) { out_gid[i] = 0; break; }  // unreachable
      slot = (slot + 1) & mask;
    }
  }
}

// string predicates over row strrefs (mixed dict/PLAIN chunks where the
// per-dict-entry LUT cannot cover the PLAIN pages). op: CmpMode, or -1 for
// CONTAINS (byte substring).
__global__ void k_cmp_str(const uint8_t* __restrict__ dec,
                          const int64_t* __restrict__ refs,
                          const uint8_t* __restrict__ valid,
                          const uint8_t* __restrict__ lit, uint32_t lit_len,
                          int op, uint8_t* __restrict__ mask, int64_t n_rows) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n_rows; i += stride) {
    if (valid && !valid[i]) { mask[i] = 0; continue; }
    uint64_t ref = (uint64_t)refs[i];
    uint32_t len = ref_len(dec, ref);
    const uint8_t* s = dec + ref + 4;
    bool ok;
    if (op < 0) {  // CONTAINS
      ok = false;
      if (lit_len == 0) ok = true;
      else if (len >= lit_len) {
        uint8_t c0 = lit[0];
        for (uint32_t j = 0; j + lit_len <= len && !ok; j++) {
          if (s[j] != c0) continue;
          uint32_t k = 1;
          while (k < lit_len && s[j + k] == lit[k]) k++;
          ok = (k == lit_len);
        }
      }
    } else {
      uint32_t n = len < lit_len ? len : lit_len;
      int c = 0;
      for (uint32_t j = 0; j < n && !c; j++)
        c = s[j] < lit[j] ? -1 : (s[j] > lit[j] ? 1 : 0);
      if (!c) c = len == lit_len ? 0 : (len < lit_len ? -1 : 1);
      switch (op) {
        case CMP_EQ: ok = c == 0; break;
        case CMP_NE: ok = c != 0; break;
        case CMP_LT: ok = c < 0; break;
        case CMP_LE: ok = c <= 0; break;
        case CMP_GT: ok = c > 0; break;
        default: ok = c >= 0; break;
      }
    }
    if (!ok) mask[i] = 0;
  }
}

// export-side string fetch: lengths then packed bytes for a strref list
__global__ void k_ref_lens(const uint8_t* __restrict__ dec,
                           const uint64_t* __restrict__ refs, int64_t n,
                           uint32_t* __restrict__ lens) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) lens[i] = ref_len(dec, refs[i]);
}
__global__ void k_ref_gather(const uint8_t* __restrict__ dec,
                             const uint64_t* __restrict__ refs,
                             const uint64_t* __restrict__ offs, int64_t n,
                             uint8_t* __restrict__ out) {
  int64_t i = blockIdx.x;
  if (i >= n) return;
  uint64_t ref = refs[i];
  uint32_t len = ref_len(dec, ref);
  const uint8_t* s = dec + ref + 4;
  uint8_t* d = out + offs[i];
  for (uint32_t j = threadIdx.x; j < len; j += blockDim.x) d[j] = s[j];
}

// ------------------------------------------------------------------
// DELTA_BINARY_PACKED i64 -> row-aligned (one wave per page).
// Phase A: lane-redundant block-header walk storing per-miniblock
//   (data offset, bit width, block min_delta ref); Phase B: parallel
//   per-miniblock delta sums; Phase C: serial scan of miniblock sums;
//   Phase D: parallel value reconstruction.
// LDS budget: 8192 miniblocks (262,144 values @ 32/miniblock).
// ------------------------------------------------------------------
#define MAX_MB 8192
#define MAX_BLK 2048
#define DELTA_T 256
__global__ void __launch_bounds__(DELTA_T)
k_delta_i64(const uint8_t* __restrict__ dec, const DevPage* __restrict__ pages,
            const int32_t* __restrict__ ids, int n,
            int64_t* __restrict__ out, uint8_t* __restrict__ valid,
            int32_t* d_error) {
  // LDS budget (fits 160 KiB/CU): 32K off + 8K bw + 16K md + 64K sum = 120K.
  // mb_sum is re-used in place as the post-scan starting value; the block id
  // of miniblock m is m / mpb (uniform miniblocks per block).
  __shared__ uint32_t mb_off[MAX_MB];     // payload-relative offset of miniblock data
  __shared__ uint8_t mb_bw[MAX_MB];
  __shared__ int64_t blk_md[MAX_BLK];     // min_delta per block
  __shared__ int64_t mb_sum[MAX_MB];      // phase B: delta sums; phase C: start values

  int pi = blockIdx.x;
  if (pi >= n) return;
  const DevPage pg = pages[ids[pi]];
  const int lane = threadIdx.x;
  const uint8_t* def_start; uint32_t def_len; bool all_valid;
  const uint8_t* payload = dec + pg.dst_off;
  const uint8_t* vals = def_levels(pg, payload, &def_start, &def_len, &all_valid);

  if (!all_valid) {
    // nulls in the time column: lane0 fully serial (never hit by Parseable
    // streams — p_timestamp is always set by ingest, event/format/mod.rs:167)
    if (lane != 0) return;
    SerialRle def(def_start, def_start + def_len, 1);
    const uint8_t* q = vals;
    auto rv = [&]() { uint64_t v = 0; int sh = 0; for (;;) { uint8_t b = *q++; v |= (uint64_t)(b & 0x7f) << sh; if (!(b & 0x80)) return v; sh += 7; } };
    auto rz = [&]() { uint64_t v = rv(); return (int64_t)(v >> 1) ^ -(int64_t)(v & 1); };
    uint64_t blk = rv(), mpb = rv(), total = rv();
    int64_t value = rz();
    uint64_t per_mini = blk / mpb;
    uint64_t emitted = 0;
    int64_t cur_md = 0; uint8_t bws[256]; uint64_t mb = 0, in_mb = 0;
    uint64_t acc = 0; int nbits = 0;
    for (uint32_t r = 0; r < pg.num_values; r++) {
      if (!def.next()) { if (valid) valid[pg.row_start + r] = 0; continue; }
      int64_t v;
      if (emitted == 0) v = value;
      else {
        if (((emitted - 1) % (per_mini * mpb)) == 0) {  // new block
          cur_md = rz();
          for (uint64_t m = 0; m < mpb; m++) bws[m] = *q++;
          mb = 0; in_mb = 0; acc = 0; nbits = 0;
        }
        int bw = bws[mb];
        uint64_t d = 0;
        if (bw) {
          while (nbits < bw) { acc |= (uint64_t)(*q++) << nbits; nbits += 8; }
          d = (bw >= 64) ? acc : (acc & ((1ull << bw) - 1));
          acc >>= bw; nbits -= bw;
        }
        value += cur_md + (int64_t)d;
        v = value;
        if (++in_mb == per_mini) { in_mb = 0; mb++; acc = 0; nbits = 0; }
      }
      emitted++;
      out[pg.row_start + r] = v;
      if (valid) valid[pg.row_start + r] = 1;
    }
    (void)total;
    return;
  }

  // ---- fast path: no nulls ----
  // Phase A (lane-redundant): header + block walk
  const uint8_t* q = vals;
  auto rv = [&]() { uint64_t v = 0; int sh = 0; for (;;) { uint8_t b = *q++; v |= (uint64_t)(b & 0x7f) << sh; if (!(b & 0x80)) return v; sh += 7; } };
  auto rz = [&]() { uint64_t v = rv(); return (int64_t)(v >> 1) ^ -(int64_t)(v & 1); };
  uint64_t blk_size = rv(), mpb = rv(), total = rv();
  int64_t first = rz();
  uint64_t per_mini = blk_size / mpb;
  if (per_mini % 8 || total > pg.num_values || mpb > 256) {
    if (lane == 0) atomicExch(d_error, ERR_DELTA);
    return;
  }
  uint64_t n_deltas = total ? total - 1 : 0;
  uint32_t n_mb = (uint32_t)((n_deltas + per_mini - 1) / per_mini);
  uint32_t n_blk = (uint32_t)((n_mb + mpb - 1) / mpb);
  if (n_mb > MAX_MB || n_blk > MAX_BLK) {
    if (lane == 0) atomicExch(d_error, ERR_DELTA);
    return;
  }
  // walk blocks redundantly; every lane records into LDS identically
  {
    uint32_t mb = 0;
    for (uint32_t b = 0; b < n_blk; b++) {
      int64_t md = rz();
      if (lane == 0) blk_md[b] = md;
      const uint8_t* bws = q;
      q += mpb;
      for (uint64_t m = 0; m < mpb && mb < n_mb; m++, mb++) {
        if (lane == 0) {
          mb_off[mb] = (uint32_t)(q - vals);
          mb_bw[mb] = bws[m];
        }
        q += (per_mini * bws[m]) / 8;
      }
    }
  }
  __syncthreads();

  // Phase B: per-miniblock delta sums (parallel over miniblocks)
  for (uint32_t m = lane; m < n_mb; m += DELTA_T) {
    const uint8_t* p = vals + mb_off[m];
    int bw = mb_bw[m];
    int64_t md = blk_md[m / (uint32_t)mpb];
    uint64_t cnt = per_mini;
    if ((uint64_t)(m + 1) * per_mini > n_deltas) cnt = n_deltas - (uint64_t)m * per_mini;
    int64_t s = 0;
    uint64_t acc = 0; int nbits = 0;
    for (uint64_t i = 0; i < cnt; i++) {
      uint64_t dv = 0;
      if (bw) {
        while (nbits < bw) { acc |= (uint64_t)(*p++) << nbits; nbits += 8; }
        dv = (bw >= 64) ? acc : (acc & ((1ull << bw) - 1));
        acc >>= bw; nbits -= bw;
      }
      s += md + (int64_t)dv;
    }
    mb_sum[m] = s;
  }
  __syncthreads();

  // Phase C: serial exclusive scan of miniblock sums -> starting value
  if (lane == 0) {
    int64_t run = first;
    for (uint32_t m = 0; m < n_mb; m++) { int64_t s = mb_sum[m]; mb_sum[m] = run; run += s; }
  }
  __syncthreads();

  // Phase D: reconstruct values
  if (total) {
    if (lane == 0) { out[pg.row_start] = first; if (valid) valid[pg.row_start] = 1; }
  }
  for (uint32_t m = lane; m < n_mb; m += DELTA_T) {
    const uint8_t* p = vals + mb_off[m];
    int bw = mb_bw[m];
    int64_t md = blk_md[m / (uint32_t)mpb];
    uint64_t cnt = per_mini;
    if ((uint64_t)(m + 1) * per_mini > n_deltas) cnt = n_deltas - (uint64_t)m * per_mini;
    int64_t v = mb_sum[m];
    uint64_t acc = 0; int nbits = 0;
    uint64_t base = 1 + (uint64_t)m * per_mini;  // value index of first delta output
    for (uint64_t i = 0; i < cnt; i++) {
      uint64_t dv = 0;
      if (bw) {
        while (nbits < bw) { acc |= (uint64_t)(*p++) << nbits; nbits += 8; }
        dv = (bw >= 64) ? acc : (acc & ((1ull << bw) - 1));
        acc >>= bw; nbits -= bw;
      }
      v += md + (int64_t)dv;
      out[pg.row_start + base + i] = v;
      if (valid) valid[pg.row_start + base + i] = 1;
    }
  }
}

// ------------------------------------------------------------------
#define CWIN 16384
#define CTHREADS 256
#define CQMAX 4096
__device__ inline uint32_t pat_full(uint8_t c0) { return 0x01010101u * c0; }
// window-parallel CONTAINS over PLAIN byte-array pages: one block per
// DevCWin. The host decompressed each page ONCE at load time, walked the
// [u32 len][bytes] chain, and emitted value-aligned <=16KB windows plus a
// u16 start offset per value — so the kernel has NO serial spine at all:
// stage window -> all 256 threads sweep needle candidates (zero-byte trick,
// verified from byte 0) into a match-start bitmap -> thread-parallel
// per-value bitmap range checks. Null rows are zeroed by the preceding
// k_def_levels launch (null_mask).
__global__ void __launch_bounds__(CTHREADS)
k_contains_win(const uint8_t* __restrict__ dec,
               const DevCWin* __restrict__ wins, int n,
               const DevPage* __restrict__ pages,
               const uint16_t* __restrict__ starts_pool,
               const uint8_t* __restrict__ needle, int nlen,
               const uint32_t* __restrict__ rowof, uint8_t* __restrict__ mask) {
  __shared__ uint8_t win[CWIN];
  __shared__ uint32_t bm[CWIN / 32];
  __shared__ uint16_t q[CQMAX];   // candidate queue (register presweep)
  __shared__ uint32_t qn;
  __shared__ int shit;
  if (blockIdx.x >= (unsigned)n) return;
  const DevCWin W = wins[blockIdx.x];
  const DevPage pg = pages[W.page_id];
  const uint32_t row0 = pg.row_start;
  bool direct = true;  // row mapping: probe mirrors the value decoders
  if (pg.optional) {
    const uint8_t* ds; uint32_t dl; bool av;
    def_levels(pg, dec + pg.dst_off, &ds, &dl, &av);
    direct = av;
  }
  const uint8_t* src = dec + W.src;
  if (W.nbytes > CWIN) {  // single oversized value: strided global scan
    uint32_t vl;
    __builtin_memcpy(&vl, src, 4);
    if (threadIdx.x == 0) shit = (nlen == 0);
    __syncthreads();
    if (nlen && vl >= (uint32_t)nlen) {
      for (uint32_t j = threadIdx.x; j + (uint32_t)nlen <= vl && !shit;
           j += CTHREADS) {
        int k = 0;
        while (k < nlen && src[4 + j + k] == needle[k]) k++;
        if (k == nlen) shit = 1;
      }
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      uint32_t di = W.dense0;
      mask[direct ? row0 + di : rowof[row0 + di]] &= (uint8_t)(shit != 0);
    }
    return;
  }
  if (threadIdx.x == 0) qn = 0;
  __syncthreads();
  {  // stage the full window unconditionally (arena is padded by CWIN) and
     // presweep needle candidates while the words are STILL IN REGISTERS —
     // the verify pass then touches only queued positions instead of
     // re-reading the whole window from LDS
    uint32_t v[CWIN / (CTHREADS * 4)];
#pragma unroll
    for (int k = 0; k < CWIN / (CTHREADS * 4); k++)
      __builtin_memcpy(&v[k], src + threadIdx.x * 4u + (uint32_t)k * (CTHREADS * 4u), 4);
#pragma unroll
    for (int k = 0; k < CWIN / (CTHREADS * 4); k++)
      *(uint32_t*)&win[threadIdx.x * 4u + (uint32_t)k * (CTHREADS * 4u)] = v[k];
    if (nlen) {
      const uint32_t pat = 0x01010101u * needle[0];
#pragma unroll
      for (int k = 0; k < CWIN / (CTHREADS * 4); k++) {
        uint32_t p = threadIdx.x * 4u + (uint32_t)k * (CTHREADS * 4u);
        if (p >= W.nbytes) continue;
        uint32_t x = v[k] ^ pat;
        uint32_t cand = (x - 0x01010101u) & ~x & 0x80808080u;
        while (cand) {
          int b = (__builtin_ctz(cand)) >> 3;
          cand &= cand - 1;
          uint32_t qi = atomicAdd(&qn, 1u);
          if (qi < CQMAX) q[qi] = (uint16_t)(p + b);
        }
      }
    }
  }
  for (uint32_t i = threadIdx.x; i < CWIN / 32; i += CTHREADS) bm[i] = 0;
  __syncthreads();
  if (nlen) {
    if (qn <= CQMAX) {
      for (uint32_t i = threadIdx.x; i < qn; i += CTHREADS) {
        uint32_t pos = q[i];
        if (pos + nlen <= CWIN) {
          int k = 0;  // verify from 0: the borrow trick has false positives
          while (k < nlen && win[pos + k] == needle[k]) k++;
          if (k == nlen) atomicOr(&bm[pos >> 5], 1u << (pos & 31));
        }
      }
    } else {  // queue overflow (pathological needle[0] density): full sweep
      for (uint32_t p = threadIdx.x * 4u; p < W.nbytes; p += CTHREADS * 4u) {
        uint32_t w = *(const uint32_t*)&win[p];
        uint32_t x = w ^ pat_full(needle[0]);
        uint32_t cand = (x - 0x01010101u) & ~x & 0x80808080u;
        while (cand) {
          int b = (__builtin_ctz(cand)) >> 3;
          cand &= cand - 1;
          uint32_t pos = p + b;
          if (pos + nlen <= CWIN) {
            int k = 0;
            while (k < nlen && win[pos + k] == needle[k]) k++;
            if (k == nlen) atomicOr(&bm[pos >> 5], 1u << (pos & 31));
          }
        }
      }
    }
  }
  __syncthreads();
  const uint32_t* win32 = (const uint32_t*)win;
  for (uint32_t i = threadIdx.x; i < W.n_values; i += CTHREADS) {
    uint32_t o = starts_pool[W.starts + i];
    uint32_t sh = (o & 3) * 8;
    uint32_t vl = win32[o >> 2] >> sh;
    if (sh) vl |= win32[(o >> 2) + 1] << (32 - sh);
    uint8_t hit = 0;
    if (nlen == 0) {
      hit = 1;
    } else if (vl >= (uint32_t)nlen) {
      uint32_t lo = o + 4, hi = o + 4 + vl - nlen;  // inclusive starts
      uint32_t w0 = lo >> 5, w1 = hi >> 5;
      if (w0 == w1) {
        uint32_t m = (hi - lo == 31) ? ~0u
                                     : (((1u << (hi - lo + 1)) - 1) << (lo & 31));
        hit = (bm[w0] & m) != 0;
      } else {
        uint32_t m0 = ~0u << (lo & 31);
        uint32_t m1 = ((hi & 31) == 31) ? ~0u : ((1u << ((hi & 31) + 1)) - 1);
        hit = ((bm[w0] & m0) != 0) | ((bm[w1] & m1) != 0);
        for (uint32_t w = w0 + 1; w < w1 && !hit; w++) hit |= (bm[w] != 0);
      }
    }
    uint32_t di = W.dense0 + i;
    mask[direct ? row0 + di : rowof[row0 + di]] &= hit;
  }
}

__global__ void k_cmp_i64(const int64_t* __restrict__ col,
                          const uint8_t* __restrict__ valid,
                          int64_t lo, int64_t hi, int mode, int hi_exclusive,
                          int is_f64, uint8_t* __restrict__ mask, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    uint8_t ok = valid ? valid[i] : 1;
    if (ok) {
      if (is_f64) {
        double v = __longlong_as_double((long long)col[i]);
        double flo = __longlong_as_double((long long)lo);
        double fhi = __longlong_as_double((long long)hi);
        switch (mode) {
          case CMP_EQ: ok = (v == flo); break;
          case CMP_NE: ok = (v != flo); break;
          case CMP_LT: ok = (v < flo); break;
          case CMP_LE: ok = (v <= flo); break;
          case CMP_GT: ok = (v > flo); break;
          case CMP_GE: ok = (v >= flo); break;
          case CMP_RANGE: ok = (v >= flo) && (hi_exclusive ? (v < fhi) : (v <= fhi)); break;
        }
      } else {
        int64_t v = col[i];
        switch (mode) {
          case CMP_EQ: ok = (v == lo); break;
          case CMP_NE: ok = (v != lo); break;
          case CMP_LT: ok = (v < lo); break;
          case CMP_LE: ok = (v <= lo); break;
          case CMP_GT: ok = (v > lo); break;
          case CMP_GE: ok = (v >= lo); break;
          case CMP_RANGE: ok = (v >= lo) && (hi_exclusive ? (v < hi) : (v <= hi)); break;
        }
      }
    }
    mask[i] &= ok;
  }
}

// ------------------------------------------------------------------
// Projection scans (ORDER BY p_timestamp DESC LIMIT k — the console's
// default query; output-ordering contract stream_schema_provider.rs:181-204):
// wave-ballot stream compaction of selected (sort-key, row) pairs, then a
// device radix sort (rocPRIM) and a gather of the k winners.
// ------------------------------------------------------------------
__global__ void k_compact_selected(const uint8_t* __restrict__ mask,
                                   const int64_t* __restrict__ key_col,
                                   int64_t n_rows,
                                   int64_t* __restrict__ out_keys,
                                   uint32_t* __restrict__ out_rows,
                                   unsigned long long* __restrict__ counter) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n_rows; i += stride) {
    bool sel = mask ? (mask[i] != 0) : true;
    // wave-level ballot + prefix: one atomicAdd per wave, lanes write at
    // base + their popcount rank
    unsigned long long ballot = __ballot(sel);
    int lane = threadIdx.x & 63;
    unsigned long long base = 0;
    int cnt = __popcll(ballot);
    if (cnt) {
      if (lane == __ffsll((unsigned long long)ballot) - 1)
        base = atomicAdd(counter, (unsigned long long)cnt);
      base = __shfl(base, __ffsll((unsigned long long)ballot) - 1);
      if (sel) {
        int rank = __popcll(ballot & ((1ull << lane) - 1));
        out_keys[base + rank] = key_col[i];
        out_rows[base + rank] = (uint32_t)i;
      }
    }
  }
}

__global__ void k_gather_i64(const uint32_t* __restrict__ rows, int64_t k,
                             const int64_t* __restrict__ col,
                             int64_t* __restrict__ out) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < k) out[i] = col[rows[i]];
}
__global__ void k_gather_i32(const uint32_t* __restrict__ rows, int64_t k,
                             const int32_t* __restrict__ col,
                             int32_t* __restrict__ out) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < k) out[i] = col[rows[i]];
}
__global__ void k_gather_u8(const uint32_t* __restrict__ rows, int64_t k,
                            const uint8_t* __restrict__ col,
                            uint8_t* __restrict__ out) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < k) out[i] = col[rows[i]];
}

// ------------------------------------------------------------------
// DATE_BIN group key: gid = 1 + (v - first_bin_origin)/stride for valid
// rows, 0 for NULL (query/mod.rs:665-735 semantics: bins are
// origin-aligned windows of stride ms)
// ------------------------------------------------------------------
__global__ void k_bin_i64(const int64_t* __restrict__ col,
                          const uint8_t* __restrict__ valid,
                          int64_t origin, int64_t stride, int64_t min_idx,
                          int32_t nbins, int32_t* __restrict__ out, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride_t = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride_t) {
    if (valid && !valid[i]) { out[i] = 0; continue; }
    int64_t v = col[i] - origin;
    int64_t idx = (v >= 0) ? v / stride : -((-v + stride - 1) / stride);
    idx -= min_idx;
    if (idx < 0) idx = 0;
    if (idx >= nbins) idx = nbins - 1;
    out[i] = (int32_t)(idx + 1);
  }
}

// ------------------------------------------------------------------
// aggregation: grid-stride, per-block LDS table (when it fits) flushed
// with global atomics.
// ------------------------------------------------------------------
__device__ inline void atomic_min_i64(uint64_t* addr, int64_t val) {
  int64_t old = (int64_t)*addr;
  while (val < old) {
    uint64_t prev = atomicCAS((unsigned long long*)addr, (unsigned long long)old,
                              (unsigned long long)val);
    if ((int64_t)prev == old) break;
    old = (int64_t)prev;
  }
}
__device__ inline void atomic_max_i64(uint64_t* addr, int64_t val) {
  int64_t old = (int64_t)*addr;
  while (val > old) {
    uint64_t prev = atomicCAS((unsigned long long*)addr, (unsigned long long)old,
                              (unsigned long long)val);
    if ((int64_t)prev == old) break;
    old = (int64_t)prev;
  }
}
__device__ inline void atomic_min_f64(uint64_t* addr, double val) {
  uint64_t old = *addr;
  for (;;) {
    double cur = __longlong_as_double((long long)old);
    if (!(val < cur)) break;
    uint64_t prev = atomicCAS((unsigned long long*)addr, (unsigned long long)old,
                              (unsigned long long)__double_as_longlong(val));
    if (prev == old) break;
    old = prev;
  }
}
__device__ inline void atomic_max_f64(uint64_t* addr, double val) {
  uint64_t old = *addr;
  for (;;) {
    double cur = __longlong_as_double((long long)old);
    if (!(val > cur)) break;
    uint64_t prev = atomicCAS((unsigned long long*)addr, (unsigned long long)old,
                              (unsigned long long)__double_as_longlong(val));
    if (prev == old) break;
    old = prev;
  }
}
__device__ inline void atomic_add_f64(uint64_t* addr, double val) {
  uint64_t old = *addr;
  for (;;) {
    double cur = __longlong_as_double((long long)old);
    uint64_t desired = (uint64_t)__double_as_longlong(cur + val);
    uint64_t prev = atomicCAS((unsigned long long*)addr, (unsigned long long)old,
                              (unsigned long long)desired);
    if (prev == old) break;
    old = prev;
  }
}
// LDS variants (shared-memory atomics)
__device__ inline void atomic_min_i64_s(uint64_t* addr, int64_t val) {
  int64_t old = (int64_t)*addr;
  while (val < old) {
    uint64_t prev = atomicCAS((unsigned long long*)addr, (unsigned long long)old,
                              (unsigned long long)val);
    if ((int64_t)prev == old) break;
    old = (int64_t)prev;
  }
}

// ---- exact f64 SUM: 256-bit fixed-point superaccumulator (dev_types.h) ----
// Decompose d into 4 sign-extended two's-complement limbs, lsb = 2^-160.
// Window: mantissa lsb >= 2^-160 and |x| < 2^61 (p <= 168 keeps 35 bits of
// headroom above the largest mantissa bit for 2^34-row partitions).
// Returns false when d is outside the window (inf/nan/denormal/huge).
__device__ inline bool acc256_decompose(double d, uint64_t a[4]) {
  uint64_t bits = (uint64_t)__double_as_longlong(d);
  uint64_t mant = bits & ((1ull << 52) - 1);
  int exp = (int)((bits >> 52) & 0x7ff);
  if (exp == 0x7ff) return false;  // inf / nan
  if (exp) mant |= 1ull << 52;
  int e = (exp ? exp : 1) - 1075;
  int p = e + 160;  // bit position of the mantissa lsb in the 256-bit field
  if (p < 0 || p > 168) return false;
  a[0] = a[1] = a[2] = a[3] = 0;
  int idx = p >> 6, sh = p & 63;
  a[idx] = mant << sh;
  if (sh) a[idx + 1] |= mant >> (64 - sh);
  if (bits >> 63) {  // negative: two's complement of the 256-bit value
    uint64_t c = 1;
#pragma unroll
    for (int i = 0; i < 4; i++) {
      a[i] = ~a[i] + c;
      c = (c && a[i] == 0) ? 1 : 0;
    }
  }
  return true;
}

// limb-wise atomic add with carry propagation (exact mod 2^256 — two's
// complement keeps mixed-sign accumulation correct; the window bound keeps
// the true sum inside +-2^255 so the dropped final carry never matters)
__device__ inline void acc256_add(uint64_t* l, const uint64_t a[4]) {
  uint64_t carry = 0;
#pragma unroll
  for (int i = 0; i < 4; i++) {
    uint64_t add = a[i] + carry;
    uint64_t c2 = (add < carry) ? 1ull : 0ull;  // a[i]+carry wrapped (add==0)
    if (add) {
      uint64_t old = atomicAdd((unsigned long long*)&l[i],
                               (unsigned long long)add);
      if (old + add < old) c2 = 1;
    }
    carry = c2;
  }
}

// utf8 min/max over strrefs (hash-mode columns): CAS keep-the-winner loop
// with lexicographic byte compare against the dec arena. Converges like the
// numeric CAS min/max — after the winner settles, the compare short-circuits
// and no atomic issues.
__device__ inline void atomic_minmax_str(const uint8_t* dec, uint64_t* addr,
                                         uint64_t ref, bool want_min) {
  uint64_t old = *addr;
  for (;;) {
    if (old != HREF_EMPTY) {
      int c = ref_cmp(dec, ref, old);
      if (want_min ? (c >= 0) : (c <= 0)) break;
    }
    uint64_t prev = atomicCAS((unsigned long long*)addr,
                              (unsigned long long)old,
                              (unsigned long long)ref);
    if (prev == old) break;
    old = prev;
  }
}

template <bool USE_LDS>
__global__ void __launch_bounds__(256)
k_agg(AggArgs a) {
  extern __shared__ uint64_t lt[];
  const int slots = 1 + 2 * a.n_aggs;
  const int64_t tsz = (int64_t)a.n_groups * slots;
  const int64_t fsz = (int64_t)a.fsum_n * a.n_groups * 4;
  uint64_t* tab;
  uint64_t* fs;
  if (USE_LDS) {
    tab = lt;
    fs = lt + tsz;
    for (int64_t i = threadIdx.x; i < tsz + fsz; i += blockDim.x) {
      uint64_t init = 0;
      if (i < tsz) {
        int s = (int)(i % slots);
        if (s > 0 && ((s - 1) & 1) == 0) {  // value slot
          int ai = (s - 1) / 2;
          int k = a.agg_kind[ai];
          if (k == AGGK_MIN_I64 || k == AGGK_MIN_RANK) init = (uint64_t)INT64_MAX;
          else if (k == AGGK_MAX_I64 || k == AGGK_MAX_RANK) init = (uint64_t)INT64_MIN;
          else if (k == AGGK_MIN_F64) init = (uint64_t)0x7ff0000000000000ull;   // +inf
          else if (k == AGGK_MAX_F64) init = (uint64_t)0xfff0000000000000ull;   // -inf
          else if (k == AGGK_MIN_STR || k == AGGK_MAX_STR) init = HREF_EMPTY;
        }
      }
      lt[i] = init;
    }
    __syncthreads();
  } else {
    tab = a.table;  // pre-initialized by k_init_table
    fs = a.fsum;    // zeroed before launch
  }

  // four rows per thread per iteration: the per-row chain (mask -> gid ->
  // val loads -> LDS atomics) is latency-bound (PMC: fetch exactly
  // algorithmic at ~1.6 TB/s); unrolling forces four independent load
  // chains in flight
  const int64_t tid0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride1 = (int64_t)gridDim.x * blockDim.x;
  const int64_t n4 = a.n_rows & ~3ll;
  for (int64_t i4 = tid0 * 4; i4 < n4; i4 += stride1 * 4) {
    uint32_t m4 = 0x01010101u;
    if (a.mask) __builtin_memcpy(&m4, a.mask + i4, 4);
    int32_t gs[4];
#pragma unroll
    for (int r = 0; r < 4; r++) {
      gs[r] = 0;
      if (!((m4 >> (r * 8)) & 0xff)) { gs[r] = -1; continue; }
      for (int k = 0; k < a.n_keys; k++)
        gs[r] = gs[r] * a.key_size[k] + a.key_gid[k][i4 + r];
    }
#pragma unroll
    for (int r = 0; r < 4; r++) {
      if (gs[r] < 0) continue;
      const int64_t i = i4 + r;
      uint64_t* row = tab + (int64_t)gs[r] * slots;
      atomicAdd((unsigned long long*)&row[0], 1ull);
      for (int ai = 0; ai < a.n_aggs; ai++) {
        int k = a.agg_kind[ai];
        if (k == AGGK_COUNT_STAR) continue;
        if (a.agg_valid[ai] && !a.agg_valid[ai][i]) continue;
        if (k == AGGK_COUNT) {
          if (!a.cnt_skip[ai])
            atomicAdd((unsigned long long*)&row[1 + 2 * ai + 1], 1ull);
          continue;
        }
        int64_t v = a.agg_val[ai][i];
        uint64_t* vs = &row[1 + 2 * ai];
        switch (k) {
          case AGGK_SUM_I64: atomicAdd((unsigned long long*)vs, (unsigned long long)v); break;
          case AGGK_SUM_F64: {
            double d = __longlong_as_double((long long)v);
            if (d != 0.0) {
              uint64_t acc[4];
              if (!acc256_decompose(d, acc)) {
                if (a.err) atomicExch(a.err, ERR_FSUM_RANGE);
              } else {
                acc256_add(fs + ((int64_t)a.fsum_idx[ai] * a.n_groups + gs[r]) * 4,
                           acc);
              }
            }
            break;
          }
          case AGGK_MIN_I64: case AGGK_MIN_RANK: atomic_min_i64(vs, v); break;
          case AGGK_MAX_I64: case AGGK_MAX_RANK: atomic_max_i64(vs, v); break;
          case AGGK_MIN_F64: atomic_min_f64(vs, __longlong_as_double((long long)v)); break;
          case AGGK_MAX_F64: atomic_max_f64(vs, __longlong_as_double((long long)v)); break;
          case AGGK_MIN_STR: atomic_minmax_str(a.dec, vs, (uint64_t)v, true); break;
          case AGGK_MAX_STR: atomic_minmax_str(a.dec, vs, (uint64_t)v, false); break;
        }
        if (!a.cnt_skip[ai])
          atomicAdd((unsigned long long*)&row[1 + 2 * ai + 1], 1ull);
      }
    }
  }
  // tail rows (n_rows % 4)
  int64_t i = n4 + tid0;
  int64_t stride = stride1;
  for (; i < a.n_rows; i += stride) {
    if (a.mask && !a.mask[i]) continue;
    int32_t g = 0;
    for (int k = 0; k < a.n_keys; k++)
      g = g * a.key_size[k] + a.key_gid[k][i];
    uint64_t* row = tab + (int64_t)g * slots;
    atomicAdd((unsigned long long*)&row[0], 1ull);
    for (int ai = 0; ai < a.n_aggs; ai++) {
      int k = a.agg_kind[ai];
      if (k == AGGK_COUNT_STAR) continue;  // == presence (export reads slot 0)
      if (a.agg_valid[ai] && !a.agg_valid[ai][i]) continue;
      if (k == AGGK_COUNT) {  // validity only; no value array needed
        if (!a.cnt_skip[ai])
          atomicAdd((unsigned long long*)&row[1 + 2 * ai + 1], 1ull);
        continue;
      }
      int64_t v = a.agg_val[ai][i];
      uint64_t* vs = &row[1 + 2 * ai];
      switch (k) {
        case AGGK_SUM_I64: atomicAdd((unsigned long long*)vs, (unsigned long long)v); break;
        case AGGK_SUM_F64: {
          double d = __longlong_as_double((long long)v);
          if (d != 0.0) {
            uint64_t acc[4];
            if (!acc256_decompose(d, acc)) {
              if (a.err) atomicExch(a.err, ERR_FSUM_RANGE);
            } else {
              acc256_add(fs + ((int64_t)a.fsum_idx[ai] * a.n_groups + g) * 4,
                         acc);
            }
          }
          break;
        }
        case AGGK_MIN_I64: case AGGK_MIN_RANK: atomic_min_i64(vs, v); break;
        case AGGK_MAX_I64: case AGGK_MAX_RANK: atomic_max_i64(vs, v); break;
        case AGGK_MIN_F64: atomic_min_f64(vs, __longlong_as_double((long long)v)); break;
        case AGGK_MAX_F64: atomic_max_f64(vs, __longlong_as_double((long long)v)); break;
        case AGGK_MIN_STR: atomic_minmax_str(a.dec, vs, (uint64_t)v, true); break;
        case AGGK_MAX_STR: atomic_minmax_str(a.dec, vs, (uint64_t)v, false); break;
      }
      if (!a.cnt_skip[ai])
        atomicAdd((unsigned long long*)&row[1 + 2 * ai + 1], 1ull);
    }
  }

  if (USE_LDS) {
    __syncthreads();
    // flush block table to global
    for (int64_t t = threadIdx.x; t < tsz; t += blockDim.x) {
      int s = (int)(t % slots);
      uint64_t v = tab[t];
      uint64_t* g = &a.table[t];
      if (s == 0) { if (v) atomicAdd((unsigned long long*)g, (unsigned long long)v); continue; }
      if (((s - 1) & 1) == 1) { if (v) atomicAdd((unsigned long long*)g, (unsigned long long)v); continue; }
      int ai = (s - 1) / 2;
      switch (a.agg_kind[ai]) {
        case AGGK_COUNT_STAR: case AGGK_COUNT:
          if (v) atomicAdd((unsigned long long*)g, (unsigned long long)v); break;
        case AGGK_SUM_I64:
          if (v) atomicAdd((unsigned long long*)g, (unsigned long long)v); break;
        case AGGK_SUM_F64: break;  // carried exactly in the fsum flush below
        case AGGK_MIN_I64: case AGGK_MIN_RANK:
          if ((int64_t)v != INT64_MAX) atomic_min_i64(g, (int64_t)v); break;
        case AGGK_MAX_I64: case AGGK_MAX_RANK:
          if ((int64_t)v != INT64_MIN) atomic_max_i64(g, (int64_t)v); break;
        case AGGK_MIN_F64: {
          double d = __longlong_as_double((long long)v);
          if (d != __builtin_inf()) atomic_min_f64(g, d);
          break;
        }
        case AGGK_MAX_F64: {
          double d = __longlong_as_double((long long)v);
          if (d != -__builtin_inf()) atomic_max_f64(g, d);
          break;
        }
        case AGGK_MIN_STR:
          if (v != HREF_EMPTY) atomic_minmax_str(a.dec, g, v, true);
          break;
        case AGGK_MAX_STR:
          if (v != HREF_EMPTY) atomic_minmax_str(a.dec, g, v, false);
          break;
      }
    }
    // flush the block's exact superaccumulators (limb-wise with carries)
    for (int64_t t = threadIdx.x; t < (int64_t)a.fsum_n * a.n_groups;
         t += blockDim.x) {
      uint64_t* src = fs + t * 4;
      if (src[0] | src[1] | src[2] | src[3]) {
        uint64_t v4[4] = {src[0], src[1], src[2], src[3]};
        acc256_add(a.fsum + t * 4, v4);
      }
    }
  }
}

// Specialized aggregation for the headline shape: ONE key, aggregates =
// {COUNT_STAR [, one i64 sum/min/max over a null-free column]} — ~1/3 the
// instructions of the general kernel (no per-agg dispatch, no validity
// reads, vectorized 16 B gid loads). Dispatch in launch_agg.
#define AGGF_NONE -1
template <bool USE_LDS, int K1>
__global__ void __launch_bounds__(256)
k_agg_fast(AggArgs a) {
  extern __shared__ uint64_t lt[];
  const int slots = 1 + 2 * a.n_aggs;
  const int64_t tsz = (int64_t)a.n_groups * slots;
  uint64_t* tab;
  if (USE_LDS) {
    tab = lt;
    for (int64_t i = threadIdx.x; i < tsz; i += blockDim.x) {
      uint64_t init = 0;
      int sl = (int)(i % slots);
      if (sl == 3) {  // the K1 agg's value slot (agg index 1)
        if (K1 == AGGK_MIN_I64) init = (uint64_t)INT64_MAX;
        else if (K1 == AGGK_MAX_I64) init = (uint64_t)INT64_MIN;
      }
      lt[i] = init;
    }
    __syncthreads();
  } else {
    tab = a.table;
  }
  const int32_t* __restrict__ gid = a.key_gid[0];
  const int64_t* __restrict__ val = a.agg_val[K1 == AGGF_NONE ? 0 : 1];
  const int64_t tid0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride4 = (int64_t)gridDim.x * blockDim.x * 4;
  const int64_t n4 = a.n_rows & ~3ll;
  for (int64_t i4 = tid0 * 4; i4 < n4; i4 += stride4) {
    uint32_t m4 = 0x01010101u;
    if (a.mask) __builtin_memcpy(&m4, a.mask + i4, 4);
    int32_t g4[4];
    __builtin_memcpy(g4, gid + i4, 16);
    int64_t v4[4];
    if (K1 != AGGF_NONE) __builtin_memcpy(v4, val + i4, 32);
#pragma unroll
    for (int r = 0; r < 4; r++) {
      if (!((m4 >> (r * 8)) & 0xff)) continue;
      uint64_t* row = tab + (int64_t)g4[r] * slots;
      atomicAdd((unsigned long long*)&row[0], 1ull);
      if (K1 == AGGK_SUM_I64)
        atomicAdd((unsigned long long*)&row[3], (unsigned long long)v4[r]);
      else if (K1 == AGGK_MIN_I64)
        atomic_min_i64(&row[3], v4[r]);
      else if (K1 == AGGK_MAX_I64)
        atomic_max_i64(&row[3], v4[r]);
    }
  }
  for (int64_t i = n4 + tid0; i < a.n_rows; i += stride4 / 4) {
    if (a.mask && !a.mask[i]) continue;
    uint64_t* row = tab + (int64_t)gid[i] * slots;
    atomicAdd((unsigned long long*)&row[0], 1ull);
    if (K1 != AGGF_NONE) {
      int64_t v = val[i];
      if (K1 == AGGK_SUM_I64)
        atomicAdd((unsigned long long*)&row[3], (unsigned long long)v);
      else if (K1 == AGGK_MIN_I64) atomic_min_i64(&row[3], v);
      else if (K1 == AGGK_MAX_I64) atomic_max_i64(&row[3], v);
    }
  }
  if (USE_LDS) {
    __syncthreads();
    for (int64_t t = threadIdx.x; t < tsz; t += blockDim.x) {
      int sl = (int)(t % slots);
      uint64_t v = tab[t];
      uint64_t* g = &a.table[t];
      if (sl == 0) {
        if (v) atomicAdd((unsigned long long*)g, (unsigned long long)v);
      } else if (sl == 3 && K1 != AGGF_NONE) {
        if (K1 == AGGK_SUM_I64) {
          if (v) atomicAdd((unsigned long long*)g, (unsigned long long)v);
        } else if (K1 == AGGK_MIN_I64) {
          if ((int64_t)v != INT64_MAX) atomic_min_i64(g, (int64_t)v);
        } else if ((int64_t)v != INT64_MIN) {
          atomic_max_i64(g, (int64_t)v);
        }
      }
    }
  }
}

__global__ void k_init_table(uint64_t* table, int32_t n_groups, int n_aggs,
                             const int32_t* agg_kind) {
  int slots = 1 + 2 * n_aggs;
  int64_t tsz = (int64_t)n_groups * slots;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < tsz; i += stride) {
    int s = (int)(i % slots);
    uint64_t init = 0;
    if (s > 0 && ((s - 1) & 1) == 0) {
      int k = agg_kind[(s - 1) / 2];
      if (k == AGGK_MIN_I64 || k == AGGK_MIN_RANK) init = (uint64_t)INT64_MAX;
      else if (k == AGGK_MAX_I64 || k == AGGK_MAX_RANK) init = (uint64_t)INT64_MIN;
      else if (k == AGGK_MIN_F64) init = 0x7ff0000000000000ull;
      else if (k == AGGK_MAX_F64) init = 0xfff0000000000000ull;
      else if (k == AGGK_MIN_STR || k == AGGK_MAX_STR) init = ~0ull;
    }
    table[i] = init;
  }
}

// ------------------------------------------------------------------
// host-side launchers (called from gpuq.cpp, same TU set)
// ------------------------------------------------------------------
void launch_lz4_seg(hipStream_t st, const uint8_t* raw, uint8_t* dec,
                    const DevSeg* segs, int n, int32_t* d_err) {
  if (n) hipLaunchKernelGGL(k_lz4_seg, dim3(n), dim3(WAVE), 0, st, raw, dec, segs, n, d_err);
}
void launch_lit_lane(hipStream_t st, const uint8_t* raw, uint8_t* dec,
                     const DevLit* lits, int64_t n) {
  if (n) hipLaunchKernelGGL(k_lit_lane, dim3((int)((n + 255) / 256)), dim3(256), 0, st, raw, dec, lits, n);
}
void launch_lit_wave(hipStream_t st, const uint8_t* raw, uint8_t* dec,
                     const DevLit* lits, int n) {
  if (n) hipLaunchKernelGGL(k_lit_wave, dim3(n), dim3(WAVE), 0, st, raw, dec, lits, n);
}
void launch_lz4_backrefs(hipStream_t st, uint8_t* dec, const DevBr* brs,
                         const DevPageBr* pages, int n) {
  if (n) hipLaunchKernelGGL(k_lz4_backrefs, dim3(n), dim3(WAVE), 0, st, dec, brs, pages, n);
}
void launch_brres_lane(hipStream_t st, uint8_t* dec, const DevBrRes* recs,
                       const DevPiece* pieces, int64_t n) {
  if (n) hipLaunchKernelGGL(k_brres_lane, dim3((int)((n + 255) / 256)), dim3(256), 0, st, dec, recs, pieces, n);
}
void launch_brres_inl(hipStream_t st, uint8_t* dec, const DevBrInl* recs,
                      int64_t n) {
  if (n) hipLaunchKernelGGL(k_brres_inl, dim3((int)((n + 255) / 256)), dim3(256), 0, st, dec, recs, n);
}
void launch_brres_wave(hipStream_t st, uint8_t* dec, const DevBrRes* recs,
                       const DevPiece* pieces, int n) {
  if (n) hipLaunchKernelGGL(k_brres_wave, dim3(n), dim3(WAVE), 0, st, dec, recs, pieces, n);
}
void launch_contains_win(hipStream_t st, const uint8_t* dec,
                         const DevCWin* wins, int n, const DevPage* pages,
                         const uint16_t* starts_pool, const uint8_t* needle,
                         int nlen, const uint32_t* rowof, uint8_t* mask) {
  if (n) hipLaunchKernelGGL(k_contains_win, dim3(n), dim3(CTHREADS), 0, st, dec, wins, n, pages, starts_pool, needle, nlen, rowof, mask);
}
void launch_def_levels(hipStream_t st, const uint8_t* dec, const DevPage* pages,
                       const int32_t* ids, int n, uint8_t* valid,
                       uint8_t* null_mask, uint32_t* rowof, uint32_t* rank,
                       uint32_t* present, int32_t* d_err) {
  if (n) hipLaunchKernelGGL(k_def_levels, dim3(n), dim3(WAVE), 0, st, dec, pages, ids, n, valid, null_mask, rowof, rank, present, d_err);
}
void launch_dict_gid(hipStream_t st, const uint8_t* dec, const DevPage* pages,
                     const int32_t* ids, int n, const int32_t* remap_pool,
                     int32_t* out, uint8_t* valid, const uint32_t* present,
                     int mode, int32_t* d_err) {
  if (!n) return;
  EmitGidP e{}; e.pool = remap_pool; e.out = out; e.valid = valid;
  hipLaunchKernelGGL(k_dict_pages<EmitGidP>, dim3(n), dim3(WAVE * DP_WAVES), 0, st, dec, pages, ids, n, e, present, mode, d_err);
}
void launch_dict_i64(hipStream_t st, const uint8_t* dec, const DevPage* pages,
                     const int32_t* ids, int n, const int64_t* dictv_pool,
                     int64_t* out, uint8_t* valid, const uint32_t* present,
                     int mode, int32_t* d_err) {
  if (!n) return;
  EmitDictI64P e{}; e.pool = dictv_pool; e.out = out; e.valid = valid;
  hipLaunchKernelGGL(k_dict_pages<EmitDictI64P>, dim3(n), dim3(WAVE * DP_WAVES), 0, st, dec, pages, ids, n, e, present, mode, d_err);
}
void launch_dict_mask(hipStream_t st, const uint8_t* dec, const DevPage* pages,
                      const int32_t* ids, int n, const uint8_t* lut_pool,
                      uint8_t* mask, int32_t* d_err) {
  if (!n) return;
  EmitDictMaskP e{}; e.pool = lut_pool; e.mask = mask;
  hipLaunchKernelGGL(k_dict_pages<EmitDictMaskP>, dim3(n), dim3(WAVE * DP_WAVES), 0, st, dec, pages, ids, n, e, (const uint32_t*)nullptr, 0, d_err);
}
void launch_dict_lut_scr(hipStream_t st, const uint8_t* dec,
                         const DevPage* pages, const int32_t* ids, int n,
                         const uint8_t* lut_pool, uint8_t* scr,
                         const uint32_t* present, int32_t* d_err) {
  if (!n) return;
  EmitLutD e{}; e.pool = lut_pool; e.scr = scr;
  hipLaunchKernelGGL(k_dict_pages<EmitLutD>, dim3(n), dim3(WAVE * DP_WAVES), 0, st, dec, pages, ids, n, e, present, 1, d_err);
}
void launch_plain_fixed(hipStream_t st, const uint8_t* dec, const DevPage* pages,
                        const int32_t* ids, int n, int64_t* out, uint8_t* valid,
                        const uint32_t* present, int mode, int32_t* d_err) {
  if (n) hipLaunchKernelGGL(k_plain_fixed, dim3(n), dim3(WAVE), 0, st, dec, pages, ids, n, out, valid, present, mode, d_err);
}
void launch_expand(hipStream_t st, const uint8_t* dec, const DevPage* pages,
                   const int32_t* ids, int n, const uint8_t* scr,
                   const uint32_t* rank, const uint8_t* valid, uint8_t* out,
                   int mode) {
  if (n) hipLaunchKernelGGL(k_expand, dim3(n), dim3(WAVE), 0, st, dec, pages, ids, n, scr, rank, valid, out, mode);
}
void launch_delta_i64(hipStream_t st, const uint8_t* dec, const DevPage* pages,
                      const int32_t* ids, int n, int64_t* out, uint8_t* valid,
                      int32_t* d_err) {
  if (n) hipLaunchKernelGGL(k_delta_i64, dim3(n), dim3(DELTA_T), 0, st, dec, pages, ids, n, out, valid, d_err);
}
void launch_cmp_i64(hipStream_t st, const int64_t* col, const uint8_t* valid,
                    int64_t lo, int64_t hi, int mode, int hi_excl, int is_f64,
                    uint8_t* mask, int64_t n) {
  int blocks = (int)((n + 255) / 256);
  if (blocks > 4096) blocks = 4096;
  if (n) hipLaunchKernelGGL(k_cmp_i64, dim3(blocks), dim3(256), 0, st, col, valid, lo, hi, mode, hi_excl, is_f64, mask, n);
}
void launch_compact(hipStream_t st, const uint8_t* mask, const int64_t* key_col,
                    int64_t n_rows, int64_t* out_keys, uint32_t* out_rows,
                    unsigned long long* counter) {
  int blocks = (int)((n_rows + 255) / 256);
  if (blocks > 4096) blocks = 4096;
  if (n_rows)
    hipLaunchKernelGGL(k_compact_selected, dim3(blocks), dim3(256), 0, st,
                       mask, key_col, n_rows, out_keys, out_rows, counter);
}
void launch_gather_i64(hipStream_t st, const uint32_t* rows, int64_t k,
                       const int64_t* col, int64_t* out) {
  if (k) hipLaunchKernelGGL(k_gather_i64, dim3((int)((k + 255) / 256)), dim3(256), 0, st, rows, k, col, out);
}
void launch_gather_i32(hipStream_t st, const uint32_t* rows, int64_t k,
                       const int32_t* col, int32_t* out) {
  if (k) hipLaunchKernelGGL(k_gather_i32, dim3((int)((k + 255) / 256)), dim3(256), 0, st, rows, k, col, out);
}
void launch_gather_u8(hipStream_t st, const uint32_t* rows, int64_t k,
                      const uint8_t* col, uint8_t* out) {
  if (k) hipLaunchKernelGGL(k_gather_u8, dim3((int)((k + 255) / 256)), dim3(256), 0, st, rows, k, col, out);
}

void launch_bin_i64(hipStream_t st, const int64_t* col, const uint8_t* valid,
                    int64_t origin, int64_t stride, int64_t min_idx,
                    int32_t nbins, int32_t* out, int64_t n) {
  int blocks = (int)((n + 255) / 256);
  if (blocks > 4096) blocks = 4096;
  if (n) hipLaunchKernelGGL(k_bin_i64, dim3(blocks), dim3(256), 0, st, col, valid, origin, stride, min_idx, nbins, out, n);
}

void launch_init_table(hipStream_t st, uint64_t* table, int32_t n_groups,
                       int n_aggs, const int32_t* d_agg_kind) {
  int64_t tsz = (int64_t)n_groups * (1 + 2 * n_aggs);
  int blocks = (int)((tsz + 255) / 256);
  if (blocks > 4096) blocks = 4096;
  hipLaunchKernelGGL(k_init_table, dim3(blocks), dim3(256), 0, st, table, n_groups, n_aggs, d_agg_kind);
}
void launch_dict_count(hipStream_t st, const uint8_t* dec, const DevPage* pages,
                       const int32_t* ids, int n, const int32_t* remap_pool,
                       uint64_t* table, int32_t n_groups, int n_aggs,
                       int32_t* d_err) {
  if (!n) return;
  int blocks = n < 16384 ? n : 16384;
  size_t lds = (size_t)n_groups * 8;
  hipLaunchKernelGGL(k_dict_count, dim3(blocks), dim3(WAVE), lds, st,
                     dec, pages, ids, n, remap_pool, table, n_groups, n_aggs, d_err);
}

void launch_agg(hipStream_t st, const AggArgs& a) {
  size_t lds = ((size_t)a.n_groups * (1 + 2 * a.n_aggs) +
                (size_t)a.fsum_n * a.n_groups * 4) * 8;
  int blocks = (int)((a.n_rows + 255) / 256);
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  // fast path: one key, {COUNT_STAR [, one null-free i64 sum/min/max]}
  // (the c2 headline shape); validity reads and the per-agg dispatch
  // vanish, gid/val loads vectorize
  bool fast = a.n_keys == 1 && a.n_aggs >= 1 && a.n_aggs <= 2 &&
              a.agg_kind[0] == AGGK_COUNT_STAR && a.fsum_n == 0;
  int k1 = AGGF_NONE;
  if (fast && a.n_aggs == 2) {
    k1 = a.agg_kind[1];
    fast = (k1 == AGGK_SUM_I64 || k1 == AGGK_MIN_I64 || k1 == AGGK_MAX_I64) &&
           a.cnt_skip[1] && a.agg_valid[1] == nullptr &&
           a.agg_val[1] != nullptr;
  }
  // 64 KiB still admits 2 blocks/CU and avoids the global-atomic cliff on
  // ~1000-group tables (c2: GROUP BY host was 100x slower via global atomics)
  bool use_lds = lds <= 64 * 1024;
  if (fast) {
    auto launch = [&](auto kern) {
      hipLaunchKernelGGL(kern, dim3(blocks), dim3(256), use_lds ? lds : 0, st, a);
    };
    if (use_lds) {
      switch (k1) {
        case AGGK_SUM_I64: launch(k_agg_fast<true, AGGK_SUM_I64>); break;
        case AGGK_MIN_I64: launch(k_agg_fast<true, AGGK_MIN_I64>); break;
        case AGGK_MAX_I64: launch(k_agg_fast<true, AGGK_MAX_I64>); break;
        default: launch(k_agg_fast<true, AGGF_NONE>); break;
      }
    } else {
      switch (k1) {
        case AGGK_SUM_I64: launch(k_agg_fast<false, AGGK_SUM_I64>); break;
        case AGGK_MIN_I64: launch(k_agg_fast<false, AGGK_MIN_I64>); break;
        case AGGK_MAX_I64: launch(k_agg_fast<false, AGGK_MAX_I64>); break;
        default: launch(k_agg_fast<false, AGGF_NONE>); break;
      }
    }
    return;
  }
  if (use_lds) {
    hipLaunchKernelGGL(k_agg<true>, dim3(blocks), dim3(256), lds, st, a);
  } else {
    hipLaunchKernelGGL(k_agg<false>, dim3(blocks), dim3(256), 0, st, a);
  }
}

void launch_pool_vals(hipStream_t st, const uint8_t* dec, const DevPage* pages,
                      const int32_t* ids, int n, const int64_t* pool,
                      int64_t* out, uint8_t* valid, const uint32_t* present,
                      int mode) {
  if (n) hipLaunchKernelGGL(k_pool_vals, dim3(n), dim3(WAVE), 0, st, dec,
                            pages, ids, n, pool, out, valid, present, mode);
}
void launch_hash_build(hipStream_t st, const uint8_t* dec, const int64_t* refs,
                       const uint8_t* valid, int64_t n_rows, uint64_t* hkeys,
                       int32_t* hgids, int clog2, uint32_t* counter,
                       uint64_t* gid2ref, int32_t gid_cap, int32_t* d_err) {
  if (!n_rows) return;
  int blocks = (int)((n_rows + 255) / 256);
  if (blocks > 8192) blocks = 8192;
  hipLaunchKernelGGL(k_hash_build, dim3(blocks), dim3(256), 0, st, dec, refs,
                     valid, n_rows, hkeys, hgids, clog2, counter, gid2ref,
                     gid_cap, d_err);
}
void launch_hash_lookup(hipStream_t st, const uint8_t* dec, const int64_t* refs,
                        const uint8_t* valid, int64_t n_rows,
                        const uint64_t* hkeys, const int32_t* hgids, int clog2,
                        int32_t* out_gid) {
  if (!n_rows) return;
  int blocks = (int)((n_rows + 255) / 256);
  if (blocks > 8192) blocks = 8192;
  hipLaunchKernelGGL(k_hash_lookup, dim3(blocks), dim3(256), 0, st, dec, refs,
                     valid, n_rows, hkeys, hgids, clog2, out_gid);
}
void launch_pair_build(hipStream_t st, const int32_t* a, const int32_t* b,
                       int64_t n_rows, uint64_t* hkeys, int32_t* hgids,
                       int clog2, uint32_t* counter, uint64_t* gid2pair,
                       int32_t gid_cap, int32_t* d_err) {
  if (!n_rows) return;
  int blocks = (int)((n_rows + 255) / 256);
  if (blocks > 8192) blocks = 8192;
  hipLaunchKernelGGL(k_pair_build, dim3(blocks), dim3(256), 0, st, a, b,
                     n_rows, hkeys, hgids, clog2, counter, gid2pair, gid_cap,
                     d_err);
}
void launch_pair_lookup(hipStream_t st, const int32_t* a, const int32_t* b,
                        int64_t n_rows, const uint64_t* hkeys,
                        const int32_t* hgids, int clog2, int32_t* out) {
  if (!n_rows) return;
  int blocks = (int)((n_rows + 255) / 256);
  if (blocks > 8192) blocks = 8192;
  hipLaunchKernelGGL(k_pair_lookup, dim3(blocks), dim3(256), 0, st, a, b,
                     n_rows, hkeys, hgids, clog2, out);
}
void launch_numhash_build(hipStream_t st, const int64_t* vals,
                          const uint8_t* valid, int64_t n_rows,
                          uint64_t* hkeys, int32_t* hgids, int clog2,
                          uint32_t* counter, uint64_t* gid2key,
                          int32_t gid_cap, int is_f64, int32_t* d_err) {
  if (!n_rows) return;
  int blocks = (int)((n_rows + 255) / 256);
  if (blocks > 8192) blocks = 8192;
  hipLaunchKernelGGL(k_numhash_build, dim3(blocks), dim3(256), 0, st, vals,
                     valid, n_rows, hkeys, hgids, clog2, counter, gid2key,
                     gid_cap, is_f64, d_err);
}
void launch_numhash_lookup(hipStream_t st, const int64_t* vals,
                           const uint8_t* valid, int64_t n_rows,
                           const uint64_t* hkeys, const int32_t* hgids,
                           int clog2, int is_f64, int32_t* out_gid) {
  if (!n_rows) return;
  int blocks = (int)((n_rows + 255) / 256);
  if (blocks > 8192) blocks = 8192;
  hipLaunchKernelGGL(k_numhash_lookup, dim3(blocks), dim3(256), 0, st, vals,
                     valid, n_rows, hkeys, hgids, clog2, is_f64, out_gid);
}
void launch_cmp_str(hipStream_t st, const uint8_t* dec, const int64_t* refs,
                    const uint8_t* valid, const uint8_t* lit, uint32_t lit_len,
                    int op, uint8_t* mask, int64_t n_rows) {
  if (!n_rows) return;
  int blocks = (int)((n_rows + 255) / 256);
  if (blocks > 8192) blocks = 8192;
  hipLaunchKernelGGL(k_cmp_str, dim3(blocks), dim3(256), 0, st, dec, refs,
                     valid, lit, lit_len, op, mask, n_rows);
}
void launch_ref_lens(hipStream_t st, const uint8_t* dec, const uint64_t* refs,
                     int64_t n, uint32_t* lens) {
  if (n) hipLaunchKernelGGL(k_ref_lens, dim3((int)((n + 255) / 256)), dim3(256),
                            0, st, dec, refs, n, lens);
}
void launch_ref_gather(hipStream_t st, const uint8_t* dec, const uint64_t* refs,
                       const uint64_t* offs, int64_t n, uint8_t* out) {
  if (n) hipLaunchKernelGGL(k_ref_gather, dim3((int)n), dim3(64), 0, st, dec,
                            refs, offs, n, out);
}

// device radix sort of (key,row) pairs, descending (AMD rocPRIM — native
// header library, not a CUDA shim). Returns needed temp bytes when
// d_temp == nullptr.
size_t sort_pairs_desc(hipStream_t st, void* d_temp, size_t temp_bytes,
                       const int64_t* keys_in, int64_t* keys_out,
                       const uint32_t* rows_in, uint32_t* rows_out,
                       int64_t n) {
  size_t need = 0;
  (void)rocprim::radix_sort_pairs_desc(nullptr, need, keys_in, keys_out,
                                       rows_in, rows_out, (size_t)n, 0, 64, st);
  if (!d_temp) return need;
  (void)rocprim::radix_sort_pairs_desc(d_temp, temp_bytes, keys_in, keys_out,
                                       rows_in, rows_out, (size_t)n, 0, 64, st);
  return need;
}

}  // namespace gpuq
