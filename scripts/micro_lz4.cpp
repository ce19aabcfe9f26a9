// Per-page LZ4 decompression timing on REAL parquet pages (from a generated
// c1 file: host=zipf-10bit match-heavy, latency=18bit ~incompressible,
// level=3bit highly repetitive). Reports per-page cycles + implied GB/s.
// Build+run on the GPU box:
//   hipcc --offload-arch=gfx950 -O3 -std=c++17 -I. scripts/micro_lz4.cpp \
//     parseable_amd/csrc/meta.cpp -o /tmp/ml && /tmp/ml <parquet-file>
#include <hip/hip_runtime.h>
#include "parseable_amd/csrc/meta.h"
#include <cstdio>
#include <fcntl.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <unistd.h>
#include <vector>
#include <string>
#include <algorithm>

using namespace gpuq;
#define WAVE 64
#define LZ4_RING 16384
#define LZ4_IN 4096

struct Pg { uint64_t src_off, dst_off; uint32_t comp, uncomp; };

// v4 kernel (copy of the production structure) + cycle instrumentation
__global__ void __launch_bounds__(WAVE)
k_lz4(const uint8_t* __restrict__ raw, uint8_t* __restrict__ dec,
      const Pg* pages, int n, uint64_t* cycles, int* err) {
  uint64_t c_parse = 0, c_lit = 0, c_match = 0, c_refill = 0, n_seq = 0;
  uint64_t tmark;
#define MARK() tmark = __builtin_amdgcn_s_memtime()
#define ACC(x) x += __builtin_amdgcn_s_memtime() - tmark
  __shared__ uint8_t ring[LZ4_RING];
  __shared__ uint8_t inbuf[LZ4_IN + 256];
  int pi = blockIdx.x;
  if (pi >= n) return;
  const Pg pg = pages[pi];
  const uint8_t* src = raw + pg.src_off;
  uint8_t* dst = dec + pg.dst_off;
  const int lane = threadIdx.x;
  uint64_t t0 = __builtin_amdgcn_s_memtime();

  const uint32_t comp = pg.comp, uncomp = pg.uncomp;
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
  uint32_t s = 0, d = 0;
  bool bad = false;
  while (s < comp && d < uncomp) {
    MARK();
    if (!in_valid || s - in_base >= LZ4_IN) refill(s);
    ACC(c_refill); MARK(); n_seq++;
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
    ACC(c_parse); MARK();
    uint32_t lit = token >> 4;
    uint32_t off, ml;
    if (lit < 15) {
      if (s + 1 + lit > comp || d + lit > uncomp) { bad = true; break; }
      const uint8_t* lsrc = &inbuf[rel + 1];
      for (uint32_t i = lane; i < lit; i += WAVE) {
        uint8_t v = lsrc[i];
        dst[d + i] = v;
        ring[(d + i) & (LZ4_RING - 1)] = v;
      }
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
      s++;
      uint32_t b;
      do { if (s >= comp) { bad = true; break; } b = inb(s); s++; lit += b; } while (b == 255);
      if (bad) break;
      if (s + lit > comp || d + lit > uncomp) { bad = true; break; }
      uint32_t doneL = 0;
      while (doneL < lit) {
        if (!in_valid || (s + doneL) - in_base >= LZ4_IN) refill(s + doneL);
        uint32_t avail = LZ4_IN - ((s + doneL) - in_base);
        uint32_t chunk = min(lit - doneL, avail);
        const uint8_t* lsrc = &inbuf[(s + doneL) - in_base];
        uint32_t base = d + doneL;
        for (uint32_t i = lane; i < chunk; i += WAVE) {
          uint8_t v = lsrc[i];
          dst[base + i] = v;
          ring[(base + i) & (LZ4_RING - 1)] = v;
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
    if (off == 0 || off > d) { bad = true; break; }
    ACC(c_lit); MARK();
    ml += 4;
    if (d + ml > uncomp) { bad = true; break; }
    uint32_t done = 0;
    if (off > LZ4_RING / 2) {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      while (done < ml) {
        uint32_t chunk = min(ml - done, off);
        for (uint32_t i = lane; i < chunk; i += WAVE) {
          uint8_t v = dst[d + done - off + i];
          dst[d + done + i] = v;
          ring[(d + done + i) & (LZ4_RING - 1)] = v;
        }
        done += chunk;
        if (done < ml) asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
    } else {
      while (done < ml) {
        uint32_t chunk;
        __builtin_amdgcn_wave_barrier();
        if (off < WAVE) {
          chunk = min(ml - done, (uint32_t)LZ4_RING - off);
          for (uint32_t i = lane; i < chunk; i += WAVE) {
            uint8_t v = ring[(d + done - off + (i % off)) & (LZ4_RING - 1)];
            dst[d + done + i] = v;
            ring[(d + done + i) & (LZ4_RING - 1)] = v;
          }
        } else {
          chunk = min(ml - done, min(off, (uint32_t)LZ4_RING - off));
          for (uint32_t i = lane; i < chunk; i += WAVE) {
            uint8_t v = ring[(d + done - off + i) & (LZ4_RING - 1)];
            dst[d + done + i] = v;
            ring[(d + done + i) & (LZ4_RING - 1)] = v;
          }
        }
        done += chunk;
      }
      __builtin_amdgcn_wave_barrier();
    }
    d += ml;
    ACC(c_match);
  }
  if ((bad || d != uncomp) && lane == 0) *err = 1;
  if (lane == 0) {
    cycles[pi] = __builtin_amdgcn_s_memtime() - t0;
    cycles[n + pi * 5 + 0] = c_refill;
    cycles[n + pi * 5 + 1] = c_parse;
    cycles[n + pi * 5 + 2] = c_lit;
    cycles[n + pi * 5 + 3] = c_match;
    cycles[n + pi * 5 + 4] = n_seq;
  }
}

int main(int argc, char** argv) {
  int fd = open(argv[1], O_RDONLY);
  struct stat st; fstat(fd, &st);
  const uint8_t* data = (const uint8_t*)mmap(0, st.st_size, PROT_READ, MAP_PRIVATE, fd, 0);
  FileMeta fm = parse_footer(data, st.st_size);
  auto& rg = fm.row_groups[0];
  // collect data pages per column
  std::vector<Pg> pages;
  std::vector<std::string> names;
  uint64_t dst_off = 0;
  std::vector<uint8_t> rawbuf;
  for (size_t ci = 0; ci < fm.columns.size(); ci++) {
    auto& cm = rg.chunks[ci];
    auto pis = walk_pages(data, cm, rg.num_rows);
    for (auto& pi : pis) {
      if (pi.type != PAGE_DATA && pi.type != PAGE_DICT) continue;
      Pg p;
      p.src_off = rawbuf.size();
      rawbuf.insert(rawbuf.end(), data + pi.payload_off, data + pi.payload_off + pi.comp_size);
      p.comp = pi.comp_size;
      p.uncomp = pi.uncomp_size;
      p.dst_off = dst_off;
      dst_off += pi.uncomp_size + 64;
      pages.push_back(p);
      names.push_back(fm.columns[ci].name + (pi.type == PAGE_DICT ? "/dict" : "/data"));
    }
  }
  rawbuf.resize(rawbuf.size() + 8192);
  int rep = argc > 2 ? atoi(argv[2]) : 1;
  {
    int base_n = (int)pages.size();
    uint64_t base_dst = dst_off;
    for (int r = 1; r < rep; r++)
      for (int i = 0; i < base_n; i++) {
        Pg p = pages[i];
        p.dst_off += base_dst * r;
        pages.push_back(p);
        names.push_back(names[i]);
      }
    dst_off *= rep;
  }
  int n = (int)pages.size();
  printf("%d pages (rep=%d)\n", n, rep);
  uint8_t *d_raw, *d_dec;
  Pg* d_pages; uint64_t* d_cyc; int* d_err;
  hipMalloc(&d_raw, rawbuf.size()); hipMemcpy(d_raw, rawbuf.data(), rawbuf.size(), hipMemcpyHostToDevice);
  hipMalloc(&d_dec, dst_off);
  hipMalloc(&d_pages, sizeof(Pg) * n); hipMemcpy(d_pages, pages.data(), sizeof(Pg) * n, hipMemcpyHostToDevice);
  hipMalloc(&d_cyc, 8 * n * 6); hipMalloc(&d_err, 4); hipMemset(d_err, 0, 4);
  hipLaunchKernelGGL(k_lz4, dim3(n), dim3(WAVE), 0, 0, d_raw, d_dec, d_pages, n, d_cyc, d_err);  // warm
  hipDeviceSynchronize();
  hipEvent_t ea, eb; hipEventCreate(&ea); hipEventCreate(&eb);
  hipEventRecord(ea);
  hipLaunchKernelGGL(k_lz4, dim3(n), dim3(WAVE), 0, 0, d_raw, d_dec, d_pages, n, d_cyc, d_err);
  hipEventRecord(eb); hipEventSynchronize(eb);
  float wall; hipEventElapsedTime(&wall, ea, eb);
  uint64_t tot_uncomp = 0; for (auto& p : pages) tot_uncomp += p.uncomp;
  printf("WALL %.3f ms for %.1f MB uncomp -> %.1f GB/s aggregate\n", wall, tot_uncomp/1e6, tot_uncomp/(wall/1e3)/1e9);
  std::vector<uint64_t> cyc(n * 6);
  hipMemcpy(cyc.data(), d_cyc, 8 * n * 6, hipMemcpyDeviceToHost);
  int err = 0; hipMemcpy(&err, d_err, 4, hipMemcpyDeviceToHost);
  printf("err=%d\n", err);
  std::vector<uint64_t> sorted_c(cyc);
  std::sort(sorted_c.begin(), sorted_c.end());
  printf("page cycles p50=%lu p90=%lu p99=%lu max=%lu\n",
         (unsigned long)sorted_c[n/2], (unsigned long)sorted_c[(int)(n*0.9)],
         (unsigned long)sorted_c[(int)(n*0.99)], (unsigned long)sorted_c[n-1]);
  if (rep == 1) {
    std::vector<int> order(n);
    for (int i = 0; i < n; i++) order[i] = i;
    std::sort(order.begin(), order.end(), [&](int a, int b) { return cyc[a] > cyc[b]; });
    for (int oi = 0; oi < n && oi < 20; oi++) {
      int i = order[oi];
      double us = cyc[i] / 2.4e3;
      printf("%-14s comp=%8u uncomp=%8u cycles=%10lu (%8.1f us) refill=%lu parse=%lu lit=%lu match=%lu nseq=%lu cyc/seq=%lu\n",
             names[i].c_str(), pages[i].comp, pages[i].uncomp,
             (unsigned long)cyc[i], us,
             (unsigned long)cyc[n + i * 5 + 0], (unsigned long)cyc[n + i * 5 + 1],
             (unsigned long)cyc[n + i * 5 + 2], (unsigned long)cyc[n + i * 5 + 3],
             (unsigned long)cyc[n + i * 5 + 4],
             (unsigned long)(cyc[i] / (cyc[n + i * 5 + 4] ? cyc[n + i * 5 + 4] : 1)));
    }
  }
  return 0;
}
