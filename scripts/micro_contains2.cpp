// Run the PRODUCTION k_bytes_contains on (a) real c3 message pages and
// (b) synthetic pages, to localize the production-vs-microbench gap.
// Build on the GPU box:
//   hipcc --offload-arch=gfx950 -O3 -std=c++17 -I. scripts/micro_contains2.cpp \
//     parseable_amd/csrc/kernels.hip parseable_amd/csrc/meta.cpp -o /tmp/mc2
//   /tmp/mc2 <c3-parquet-file>
#include <hip/hip_runtime.h>
#include "parseable_amd/csrc/dev_types.h"
#include "parseable_amd/csrc/kernels_api.h"
#include "parseable_amd/csrc/meta.h"
#include <cstdio>
#include <cstring>
#include <fcntl.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <unistd.h>
#include <vector>
#include <string>

using namespace gpuq;

static void run_case(const char* name, const std::vector<uint8_t>& dec_host,
                     const std::vector<DevPage>& pages, int64_t n_rows) {
  int n = (int)pages.size();
  uint8_t *d_dec, *d_mask, *d_needle;
  DevPage* d_pages;
  int32_t *d_ids, *d_err;
  hipMalloc(&d_dec, dec_host.size());
  hipMemcpy(d_dec, dec_host.data(), dec_host.size(), hipMemcpyHostToDevice);
  hipMalloc(&d_pages, sizeof(DevPage) * n);
  hipMemcpy(d_pages, pages.data(), sizeof(DevPage) * n, hipMemcpyHostToDevice);
  std::vector<int32_t> ids(n);
  for (int i = 0; i < n; i++) ids[i] = i;
  hipMalloc(&d_ids, 4 * n);
  hipMemcpy(d_ids, ids.data(), 4 * n, hipMemcpyHostToDevice);
  hipMalloc(&d_mask, n_rows + 16);
  hipMemset(d_mask, 1, n_rows + 16);
  hipMalloc(&d_needle, 16);
  hipMemcpy(d_needle, "error", 6, hipMemcpyHostToDevice);
  hipMalloc(&d_err, 4);
  hipMemset(d_err, 0, 4);

  launch_bytes_contains(0, d_dec, d_pages, d_ids, n, d_needle, 5, d_mask, d_err);
  hipDeviceSynchronize();
  hipEvent_t a, b;
  hipEventCreate(&a); hipEventCreate(&b);
  hipEventRecord(a);
  for (int i = 0; i < 3; i++)
    launch_bytes_contains(0, d_dec, d_pages, d_ids, n, d_needle, 5, d_mask, d_err);
  hipEventRecord(b);
  hipEventSynchronize(b);
  float ms;
  hipEventElapsedTime(&ms, a, b);
  int err = 0;
  hipMemcpy(&err, d_err, 4, hipMemcpyDeviceToHost);
  int64_t hits = 0;
  {
    std::vector<uint8_t> m(n_rows);
    hipMemcpy(m.data(), d_mask, n_rows, hipMemcpyDeviceToHost);
    for (auto v : m) hits += v;
  }
  printf("%-10s pages=%4d rows=%8ld bytes=%.2f GB  %8.3f ms/iter (%6.1f GB/s) err=%d hits=%ld\n",
         name, n, (long)n_rows, dec_host.size() / 1e9, ms / 3,
         dec_host.size() / (ms / 3 / 1e3) / 1e9, err, (long)hits);
  hipFree(d_dec); hipFree(d_pages); hipFree(d_ids); hipFree(d_mask);
  hipFree(d_needle); hipFree(d_err);
}

int main(int argc, char** argv) {
  // --- case A: real c3 message pages (decompressed on host) ---
  if (argc > 1) {
    int fd = open(argv[1], O_RDONLY);
    struct stat st; fstat(fd, &st);
    const uint8_t* data = (const uint8_t*)mmap(0, st.st_size, PROT_READ, MAP_PRIVATE, fd, 0);
    FileMeta fm = parse_footer(data, st.st_size);
    int ci = fm.col_index("message");
    auto& rg = fm.row_groups[0];
    auto& cm = rg.chunks[ci];
    auto pis = walk_pages(data, cm, rg.num_rows);
    std::vector<uint8_t> dec;
    std::vector<DevPage> pages;
    int64_t rows = 0;
    int reps = 40;  // replicate to production scale (~2000 pages)
    for (int rep = 0; rep < reps; rep++)
      for (auto& pi : pis) {
        if (pi.type != PAGE_DATA || pi.encoding != ENC_PLAIN) continue;
        size_t off = dec.size();
        dec.resize(off + pi.uncomp_size + 16384 + 64);
        int nn = lz4_decompress_host(data + pi.payload_off, pi.comp_size,
                                     dec.data() + off, pi.uncomp_size);
        if (nn != pi.uncomp_size) { printf("host lz4 fail\n"); return 1; }
        dec.resize(off + pi.uncomp_size);
        DevPage dp{};
        dp.dst_off = off;
        dp.uncomp_size = pi.uncomp_size;
        dp.num_values = pi.num_values;
        dp.row_start = (uint32_t)rows;
        dp.optional = 1;
        pages.push_back(dp);
        rows += pi.num_values;
      }
    dec.resize(dec.size() + 16384 + 64);
    run_case("real-c3", dec, pages, rows);
  }
  // --- case B: synthetic pages like micro_contains (no def levels) ---
  {
    srand(42);
    std::vector<uint8_t> dec;
    std::vector<DevPage> pages;
    int64_t rows = 0;
    for (int p = 0; p < 2000; p++) {
      size_t off = dec.size();
      uint32_t nv = 13000;
      for (uint32_t v = 0; v < nv; v++) {
        uint32_t l = 20 + rand() % 101;
        dec.insert(dec.end(), (uint8_t*)&l, (uint8_t*)&l + 4);
        for (uint32_t j = 0; j < l; j++) dec.push_back('a' + rand() % 26);
      }
      DevPage dp{};
      dp.dst_off = off;
      dp.uncomp_size = (uint32_t)(dec.size() - off);
      dp.num_values = nv;
      dp.row_start = (uint32_t)rows;
      dp.optional = 0;
      pages.push_back(dp);
      rows += nv;
    }
    dec.resize(dec.size() + 16384 + 64);
    run_case("synthetic", dec, pages, rows);
  }
  return 0;
}
