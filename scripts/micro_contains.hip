// Microbenchmark: isolate the phases of the LIKE-'%x%' page kernel to find
// where the time goes. Synthetic pages of length-prefixed values (20-120B).
// Build+run on the GPU box:
//   hipcc --offload-arch=gfx950 -O3 scripts/micro_contains.hip -o /tmp/mc && /tmp/mc
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <vector>

#define WAVE 64
#define CWIN 16384

struct Page { uint64_t off; uint32_t bytes; uint32_t nv; };

// A: window copy only
__global__ void __launch_bounds__(WAVE) k_copy(const uint8_t* dec, const Page* pages, int n, uint32_t* sink) {
  __shared__ uint8_t win[CWIN];
  int pi = blockIdx.x; if (pi >= n) return;
  Page pg = pages[pi];
  const uint8_t* vals = dec + pg.off;
  uint32_t acc = 0;
  for (uint32_t walk = 0; walk < pg.bytes; walk += CWIN) {
    uint32_t wb = min((uint32_t)CWIN, pg.bytes - walk + 8);
    for (uint32_t i = threadIdx.x * 4u; i < wb; i += WAVE * 4u) {
      uint32_t v; __builtin_memcpy(&v, vals + walk + i, 4);
      *(uint32_t*)&win[i] = v;
    }
    __syncthreads();
    acc += win[threadIdx.x];
    __syncthreads();
  }
  if (acc == 0xdeadbeef) sink[0] = acc;
}

// B: copy + lane0 serial walk
__global__ void __launch_bounds__(WAVE) k_walk(const uint8_t* dec, const Page* pages, int n, uint32_t* sink) {
  __shared__ uint8_t win[CWIN];
  __shared__ uint32_t offs[2049];
  __shared__ uint32_t ctrl[2];
  int pi = blockIdx.x; if (pi >= n) return;
  Page pg = pages[pi];
  const uint8_t* vals = dec + pg.off;
  uint32_t done = 0, walk = 0;
  while (done < pg.nv) {
    uint32_t wb = min((uint32_t)CWIN, pg.bytes - walk + 8);
    for (uint32_t i = threadIdx.x * 4u; i < wb; i += WAVE * 4u) {
      uint32_t v; __builtin_memcpy(&v, vals + walk + i, 4);
      *(uint32_t*)&win[i] = v;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      uint32_t w = 0, cnt = 0;
      while (done + cnt < pg.nv && cnt < 2048) {
        if (w + 4 > CWIN) break;
        uint32_t l; __builtin_memcpy(&l, &win[w], 4);
        if (w + 4 + l > CWIN) break;
        offs[cnt] = w;
        w += 4 + l; cnt++;
      }
      ctrl[0] = cnt; ctrl[1] = w;
    }
    __syncthreads();
    done += ctrl[0]; walk += ctrl[1];
    if (ctrl[0] == 0) break;
    __syncthreads();
  }
  if (done == 0xdeadbeef) sink[0] = done;
}

// C: full (copy + walk + scan)
__global__ void __launch_bounds__(WAVE) k_full(const uint8_t* dec, const Page* pages, int n,
                                               const uint8_t* needle, int nlen, uint8_t* mask) {
  __shared__ uint8_t win[CWIN];
  __shared__ uint32_t offs[2049];
  __shared__ uint32_t ctrl[2];
  int pi = blockIdx.x; if (pi >= n) return;
  Page pg = pages[pi];
  const uint8_t* vals = dec + pg.off;
  uint32_t done = 0, walk = 0;
  while (done < pg.nv) {
    uint32_t wb = min((uint32_t)CWIN, pg.bytes - walk + 8);
    for (uint32_t i = threadIdx.x * 4u; i < wb; i += WAVE * 4u) {
      uint32_t v; __builtin_memcpy(&v, vals + walk + i, 4);
      *(uint32_t*)&win[i] = v;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      uint32_t w = 0, cnt = 0;
      while (done + cnt < pg.nv && cnt < 2048) {
        if (w + 4 > CWIN) break;
        uint32_t l; __builtin_memcpy(&l, &win[w], 4);
        if (w + 4 + l > CWIN) break;
        offs[cnt] = w;
        w += 4 + l; cnt++;
      }
      ctrl[0] = cnt; ctrl[1] = w;
    }
    __syncthreads();
    uint32_t bn = ctrl[0];
    for (uint32_t i = threadIdx.x; i < bn; i += WAVE) {
      uint32_t o = offs[i];
      uint32_t vl; __builtin_memcpy(&vl, &win[o], 4);
      const uint8_t* s = &win[o + 4];
      uint8_t hit = 0;
      uint8_t c0 = needle[0];
      for (uint32_t j = 0; j + nlen <= vl; j++) {
        if (s[j] == c0) {
          uint32_t k = 1;
          while (k < (uint32_t)nlen && s[j + k] == needle[k]) k++;
          if (k == (uint32_t)nlen) { hit = 1; break; }
        }
      }
      mask[pg.off / 78 + done + i] &= hit;  // approx row index; perf only
    }
    __syncthreads();
    done += bn; walk += ctrl[1];
    if (bn == 0) break;
  }
}

// E: bitmap sweep — lane0 walks offsets while lanes 1..63 sweep the window
// for needle matches (word-parallel candidate detection, ~4%% verify rate),
// then all lanes check their values' position ranges against the bitmap.
__global__ void __launch_bounds__(WAVE) k_bitmap(const uint8_t* dec, const Page* pages, int n,
                                                 const uint8_t* needle, int nlen, uint8_t* mask) {
  __shared__ uint8_t win[CWIN];
  __shared__ uint32_t offs[2049];
  __shared__ uint32_t ctrl[2];
  __shared__ uint32_t bm[CWIN / 32];     // match-start bitmap
  int pi = blockIdx.x; if (pi >= n) return;
  Page pg = pages[pi];
  const uint8_t* vals = dec + pg.off;
  const uint8_t c0 = needle[0];
  uint32_t done = 0, walk = 0;
  while (done < pg.nv) {
    uint32_t wb = min((uint32_t)CWIN, pg.bytes - walk + 8);
    for (uint32_t i = threadIdx.x * 4u; i < wb; i += WAVE * 4u) {
      uint32_t v; __builtin_memcpy(&v, vals + walk + i, 4);
      *(uint32_t*)&win[i] = v;
    }
    for (uint32_t i = threadIdx.x; i < CWIN / 32; i += WAVE) bm[i] = 0;
    __syncthreads();
    if (threadIdx.x == 0) {
      // serial offset walk (concurrent with the sweep on other lanes)
      uint32_t w = 0, cnt = 0;
      while (done + cnt < pg.nv && cnt < 2048) {
        if (w + 4 > CWIN) break;
        uint32_t l; __builtin_memcpy(&l, &win[w], 4);
        if (w + 4 + l > CWIN) break;
        offs[cnt] = w;
        w += 4 + l; cnt++;
      }
      ctrl[0] = cnt; ctrl[1] = w;
    } else {
      // word-parallel candidate sweep over the window (63 lanes)
      uint32_t lane = threadIdx.x - 1;
      for (uint32_t p = lane * 4u; p + 4 <= wb; p += (WAVE - 1) * 4u) {
        uint32_t w = *(const uint32_t*)&win[p];
        // find bytes equal to c0
        uint32_t x = w ^ (0x01010101u * c0);
        uint32_t cand = (x - 0x01010101u) & ~x & 0x80808080u;
        while (cand) {
          int b = (__builtin_ctz(cand)) >> 3;
          cand &= cand - 1;
          uint32_t pos = p + b;
          if (pos + nlen <= CWIN) {
            int k = 1;
            while (k < nlen && win[pos + k] == needle[k]) k++;
            if (k == nlen) atomicOr(&bm[pos >> 5], 1u << (pos & 31));
          }
        }
      }
    }
    __syncthreads();
    uint32_t bn = ctrl[0];
    for (uint32_t i = threadIdx.x; i < bn; i += WAVE) {
      uint32_t o = offs[i];
      uint32_t vl; __builtin_memcpy(&vl, &win[o], 4);
      uint8_t hit = 0;
      if (vl >= (uint32_t)nlen) {
        uint32_t lo = o + 4, hi = o + 4 + vl - nlen;  // inclusive match-start range
        uint32_t w0 = lo >> 5, w1 = hi >> 5;
        if (w0 == w1) {
          uint32_t m = (hi - lo == 31) ? ~0u : (((1u << (hi - lo + 1)) - 1) << (lo & 31));
          hit = (bm[w0] & m) != 0;
        } else {
          uint32_t m0 = ~0u << (lo & 31);
          uint32_t m1 = (hi & 31) == 31 ? ~0u : ((1u << ((hi & 31) + 1)) - 1);
          hit = (bm[w0] & m0) || (bm[w1] & m1);
          for (uint32_t w = w0 + 1; w < w1 && !hit; w++) hit |= bm[w] != 0;
        }
      }
      mask[pg.off / 78 + done + i] &= hit;
    }
    __syncthreads();
    done += bn; walk += ctrl[1];
    if (bn == 0) break;
  }
}

// F: bitmap sweep with 256-thread blocks (4 waves per page): copy and sweep
// 4x wider; walk still on thread 0 but latency overlaps other waves' work.
#define CTHREADS 256
__global__ void __launch_bounds__(CTHREADS) k_bitmap4(const uint8_t* dec, const Page* pages, int n,
                                                      const uint8_t* needle, int nlen, uint8_t* mask) {
  __shared__ uint8_t win[CWIN];
  __shared__ uint32_t offs[2049];
  __shared__ uint32_t ctrl[2];
  __shared__ uint32_t bm[CWIN / 32];
  int pi = blockIdx.x; if (pi >= n) return;
  Page pg = pages[pi];
  const uint8_t* vals = dec + pg.off;
  const uint8_t c0 = needle[0];
  uint32_t done = 0, walk = 0;
  while (done < pg.nv) {
    uint32_t wb = min((uint32_t)CWIN, pg.bytes - walk + 8);
    for (uint32_t i = threadIdx.x * 4u; i < wb; i += CTHREADS * 4u) {
      uint32_t v; __builtin_memcpy(&v, vals + walk + i, 4);
      *(uint32_t*)&win[i] = v;
    }
    for (uint32_t i = threadIdx.x; i < CWIN / 32; i += CTHREADS) bm[i] = 0;
    __syncthreads();
    if (threadIdx.x == 0) {
      uint32_t w = 0, cnt = 0;
      while (done + cnt < pg.nv && cnt < 2048) {
        if (w + 4 > CWIN) break;
        uint32_t l; __builtin_memcpy(&l, &win[w], 4);
        if (w + 4 + l > CWIN) break;
        offs[cnt] = w;
        w += 4 + l; cnt++;
      }
      ctrl[0] = cnt; ctrl[1] = w;
    } else {
      uint32_t lane = threadIdx.x - 1;
      for (uint32_t p = lane * 4u; p + 4 <= wb; p += (CTHREADS - 1) * 4u) {
        uint32_t w = *(const uint32_t*)&win[p];
        uint32_t x = w ^ (0x01010101u * c0);
        uint32_t cand = (x - 0x01010101u) & ~x & 0x80808080u;
        while (cand) {
          int b = (__builtin_ctz(cand)) >> 3;
          cand &= cand - 1;
          uint32_t pos = p + b;
          if (pos + nlen <= CWIN) {
            int k = 1;
            while (k < nlen && win[pos + k] == needle[k]) k++;
            if (k == nlen) atomicOr(&bm[pos >> 5], 1u << (pos & 31));
          }
        }
      }
    }
    __syncthreads();
    uint32_t bn = ctrl[0];
    for (uint32_t i = threadIdx.x; i < bn; i += CTHREADS) {
      uint32_t o = offs[i];
      uint32_t vl; __builtin_memcpy(&vl, &win[o], 4);
      uint8_t hit = 0;
      if (vl >= (uint32_t)nlen) {
        uint32_t lo = o + 4, hi = o + 4 + vl - nlen;
        uint32_t w0 = lo >> 5, w1 = hi >> 5;
        if (w0 == w1) {
          uint32_t m = (hi - lo == 31) ? ~0u : (((1u << (hi - lo + 1)) - 1) << (lo & 31));
          hit = (bm[w0] & m) != 0;
        } else {
          uint32_t m0 = ~0u << (lo & 31);
          uint32_t m1 = (hi & 31) == 31 ? ~0u : ((1u << ((hi & 31) + 1)) - 1);
          hit = (bm[w0] & m0) || (bm[w1] & m1);
          for (uint32_t w = w0 + 1; w < w1 && !hit; w++) hit |= bm[w] != 0;
        }
      }
      mask[pg.off / 78 + done + i] &= hit;
    }
    __syncthreads();
    done += bn; walk += ctrl[1];
    if (bn == 0) break;
  }
}

// D: direct-global variant (old design: lane0 walks global, lanes scan global)
__global__ void __launch_bounds__(WAVE) k_global(const uint8_t* dec, const Page* pages, int n,
                                                 const uint8_t* needle, int nlen, uint8_t* mask) {
  __shared__ uint32_t offs[1025];
  int pi = blockIdx.x; if (pi >= n) return;
  Page pg = pages[pi];
  const uint8_t* vals = dec + pg.off;
  uint32_t walk = 0;
  for (uint32_t b0 = 0; b0 < pg.nv; b0 += 1024) {
    uint32_t bn = min(1024u, pg.nv - b0);
    if (threadIdx.x == 0) {
      uint32_t w = walk;
      for (uint32_t i = 0; i < bn; i++) {
        offs[i] = w;
        uint32_t l; __builtin_memcpy(&l, vals + w, 4);
        w += 4 + l;
      }
      offs[bn] = w;
      walk = w;
    }
    __syncthreads();
    for (uint32_t i = threadIdx.x; i < bn; i += WAVE) {
      uint32_t o = offs[i];
      uint32_t vl; __builtin_memcpy(&vl, vals + o, 4);
      const uint8_t* s = vals + o + 4;
      uint8_t hit = 0;
      uint8_t c0 = needle[0];
      for (uint32_t j = 0; j + nlen <= vl; j++) {
        if (s[j] == c0) {
          uint32_t k = 1;
          while (k < (uint32_t)nlen && s[j + k] == needle[k]) k++;
          if (k == (uint32_t)nlen) { hit = 1; break; }
        }
      }
      mask[pg.off / 78 + b0 + i] &= hit;
    }
    __syncthreads();
  }
}

int main() {
  const int NPAGES = 2048;
  const int NV = 13000;            // values per page, ~74B avg -> ~1MB page
  srand(42);
  std::vector<uint8_t> host;
  std::vector<Page> pages(NPAGES);
  host.reserve((size_t)NPAGES * NV * 80);
  for (int p = 0; p < NPAGES; p++) {
    pages[p].off = host.size();
    pages[p].nv = NV;
    for (int v = 0; v < NV; v++) {
      uint32_t l = 20 + rand() % 101;
      uint32_t le = l;
      host.insert(host.end(), (uint8_t*)&le, (uint8_t*)&le + 4);
      for (uint32_t j = 0; j < l; j++) host.push_back('a' + rand() % 26);
    }
    pages[p].bytes = (uint32_t)(host.size() - pages[p].off);
  }
  host.resize(host.size() + 64);
  printf("total bytes: %.2f GB\n", host.size() / 1e9);

  uint8_t *d_dec, *d_mask, *d_needle;
  Page* d_pages;
  uint32_t* d_sink;
  hipMalloc(&d_dec, host.size());
  hipMemcpy(d_dec, host.data(), host.size(), hipMemcpyHostToDevice);
  hipMalloc(&d_pages, sizeof(Page) * NPAGES);
  hipMemcpy(d_pages, pages.data(), sizeof(Page) * NPAGES, hipMemcpyHostToDevice);
  hipMalloc(&d_mask, (size_t)NPAGES * NV + (1 << 24));
  hipMalloc(&d_needle, 16);
  hipMemcpy(d_needle, "error", 6, hipMemcpyHostToDevice);
  hipMalloc(&d_sink, 4);

  auto time_it = [&](const char* name, auto&& launch) {
    hipEvent_t a, b;
    hipEventCreate(&a); hipEventCreate(&b);
    launch();  // warm
    hipDeviceSynchronize();
    hipEventRecord(a);
    for (int i = 0; i < 3; i++) launch();
    hipEventRecord(b);
    hipEventSynchronize(b);
    float ms; hipEventElapsedTime(&ms, a, b);
    printf("%-10s %8.3f ms/iter  (%.1f GB/s of value bytes)\n", name, ms / 3,
           host.size() / (ms / 3 / 1e3) / 1e9);
  };
  time_it("copy", [&]{ hipLaunchKernelGGL(k_copy, dim3(NPAGES), dim3(WAVE), 0, 0, d_dec, d_pages, NPAGES, d_sink); });
  time_it("copy+walk", [&]{ hipLaunchKernelGGL(k_walk, dim3(NPAGES), dim3(WAVE), 0, 0, d_dec, d_pages, NPAGES, d_sink); });
  time_it("full", [&]{ hipLaunchKernelGGL(k_full, dim3(NPAGES), dim3(WAVE), 0, 0, d_dec, d_pages, NPAGES, d_needle, 5, d_mask); });
  time_it("global", [&]{ hipLaunchKernelGGL(k_global, dim3(NPAGES), dim3(WAVE), 0, 0, d_dec, d_pages, NPAGES, d_needle, 5, d_mask); });
  time_it("bitmap", [&]{ hipLaunchKernelGGL(k_bitmap, dim3(NPAGES), dim3(WAVE), 0, 0, d_dec, d_pages, NPAGES, d_needle, 5, d_mask); });
  time_it("bitmap4", [&]{ hipLaunchKernelGGL(k_bitmap4, dim3(NPAGES), dim3(CTHREADS), 0, 0, d_dec, d_pages, NPAGES, d_needle, 5, d_mask); });
  // correctness cross-check full vs bitmap on real masks
  {
    size_t msz = (size_t)NPAGES * NV + (1 << 24);
    std::vector<uint8_t> m1(msz), m2(msz);
    hipMemset(d_mask, 1, msz); hipDeviceSynchronize();
    hipLaunchKernelGGL(k_full, dim3(NPAGES), dim3(WAVE), 0, 0, d_dec, d_pages, NPAGES, d_needle, 5, d_mask);
    hipMemcpy(m1.data(), d_mask, msz, hipMemcpyDeviceToHost);
    hipMemset(d_mask, 1, msz); hipDeviceSynchronize();
    hipLaunchKernelGGL(k_bitmap, dim3(NPAGES), dim3(WAVE), 0, 0, d_dec, d_pages, NPAGES, d_needle, 5, d_mask);
    hipMemcpy(m2.data(), d_mask, msz, hipMemcpyDeviceToHost);
    printf("bitmap == full: %s\n", m1 == m2 ? "YES" : "NO");
    size_t hits = 0; for (auto b : m1) hits += b;
    printf("(hit bytes m1: %zu)\n", hits);
    // locate first mismatch and dump the value
    for (size_t i = 0; i < msz; i++) {
      if (m1[i] != m2[i]) {
        printf("first mismatch at mask[%zu]: full=%d bitmap=%d\n", i, m1[i], m2[i]);
        // find which page/value: search pages
        for (int p = 0; p < NPAGES; p++) {
          size_t base = pages[p].off / 78;
          if (i >= base && i < base + NV) {
            size_t vidx = i - base;
            printf("  page %d value %zu\n", p, vidx);
            // walk host data to that value
            size_t w = pages[p].off;
            for (size_t v = 0; v < vidx; v++) {
              uint32_t l; memcpy(&l, &host[w], 4); w += 4 + l;
            }
            uint32_t l; memcpy(&l, &host[w], 4);
            printf("  len=%u bytes='%.*s'\n", l, (int)l, (const char*)&host[w + 4]);
            break;
          }
        }
        break;
      }
    }
    // also compare bitmap4
    hipMemset(d_mask, 1, msz); hipDeviceSynchronize();
    hipLaunchKernelGGL(k_bitmap4, dim3(NPAGES), dim3(CTHREADS), 0, 0, d_dec, d_pages, NPAGES, d_needle, 5, d_mask);
    hipMemcpy(m2.data(), d_mask, msz, hipMemcpyDeviceToHost);
    printf("bitmap4 == full: %s\n", m1 == m2 ? "YES" : "NO");
  }
  return 0;
}
