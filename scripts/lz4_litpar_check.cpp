// Host-side validation of the litpar LZ4 plan (meta.cpp lz4_walk litpar
// mode): for every LZ4_RAW data page of the given parquet files, build the
// litpar plan, apply it the way the GPU kernels would (k_lit_* literal
// copies, then k_brres_* periodic-pattern resolution from pieces) and
// compare byte-for-byte with the scalar decoder. Build:
//   g++ -O2 -std=c++17 -I. scripts/lz4_litpar_check.cpp \
//       parseable_amd/csrc/meta.cpp -o /tmp/litpar_check
#include "parseable_amd/csrc/meta.h"
#include <cstdio>
#include <cstring>
#include <fcntl.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <unistd.h>
#include <vector>
using namespace gpuq;

int main(int argc, char** argv) {
  long pages = 0, litpar_pages = 0, fb = 0, bad = 0;
  long lits = 0, res = 0, pieces = 0;
  for (int a = 1; a < argc; a++) {
    int fd = open(argv[a], O_RDONLY);
    struct stat sb;
    fstat(fd, &sb);
    auto* data =
        (const uint8_t*)mmap(0, sb.st_size, PROT_READ, MAP_PRIVATE, fd, 0);
    FileMeta fm = parse_footer(data, sb.st_size);
    for (auto& rg : fm.row_groups)
      for (auto& cm : rg.chunks) {
        if (cm.codec != CODEC_LZ4_RAW) continue;
        for (auto& pi : walk_pages(data, cm, rg.num_rows)) {
          if (pi.comp_size <= 0 || pi.uncomp_size <= 0) continue;
          const uint8_t* praw = data + pi.payload_off;
          std::vector<uint8_t> want(pi.uncomp_size);
          if (lz4_decompress_host(praw, pi.comp_size, want.data(),
                                  want.size()) != pi.uncomp_size)
            continue;  // stored-raw page
          pages++;
          Lz4Plan lp;
          try {
            lp = lz4_walk(praw, pi.comp_size, pi.uncomp_size, 8192, true);
          } catch (const std::exception& e) {
            printf("walk threw: %s\n", e.what());
            bad++;
            continue;
          }
          if (lp.fallback) {
            fb++;
            continue;
          }
          litpar_pages++;
          lits += lp.lits.size();
          res += lp.resolved.size();
          pieces += lp.pieces.size();
          // phase 1: literal copies (any order)
          std::vector<uint8_t> got(pi.uncomp_size, 0xcd);
          for (auto& L : lp.lits) memcpy(&got[L.dst], praw + L.src, L.len);
          // cross-check the host pattern-inline path (gpuq.cpp lit_bytes):
          // every piece's bytes must be reachable from the COMPRESSED stream
          // through the lits map and equal the decompressed literal bytes
          long inl_mismatch = 0;
          for (auto& r : lp.resolved) {
            for (uint32_t k = 0; k < r.piece_n; k++) {
              const Lz4Piece& pc = lp.pieces[r.piece_start + k];
              size_t lo = 0, hi = lp.lits.size();
              while (lo < hi) {
                size_t mid = (lo + hi) / 2;
                if (lp.lits[mid].dst <= pc.src) lo = mid + 1;
                else hi = mid;
              }
              if (lo == 0) { inl_mismatch++; continue; }
              const Lz4Lit& L = lp.lits[lo - 1];
              if (pc.src < L.dst || pc.src + pc.len > L.dst + L.len) {
                inl_mismatch++;
                continue;
              }
              if (memcmp(praw + L.src + (pc.src - L.dst), &want[pc.src],
                         pc.len) != 0)
                inl_mismatch++;
            }
          }
          if (inl_mismatch) {
            printf("INLINE SOURCE MISMATCH x%ld\n", inl_mismatch);
            bad++;
          }
          // phase 2: resolved records (any order) — pattern from pieces,
          // applied periodically, exactly as k_brres does
          for (auto& r : lp.resolved) {
            std::vector<uint8_t> pat;
            for (uint32_t k = 0; k < r.piece_n; k++) {
              const Lz4Piece& pc = lp.pieces[r.piece_start + k];
              pat.insert(pat.end(), &got[pc.src], &got[pc.src] + pc.len);
            }
            uint32_t period = r.off < r.len ? r.off : r.len;
            if (pat.size() != period) {
              printf("pattern size mismatch %zu vs %u\n", pat.size(), period);
              bad++;
              continue;
            }
            for (uint32_t i = 0; i < r.len; i++)
              got[r.dst + i] = pat[i % period];
          }
          if (memcmp(got.data(), want.data(), pi.uncomp_size) != 0) {
            bad++;
            int first = -1;
            for (int i = 0; i < pi.uncomp_size; i++)
              if (got[i] != want[i]) { first = i; break; }
            printf("MISMATCH page uncomp=%d first_diff=%d\n", pi.uncomp_size,
                   first);
          }
        }
      }
    munmap((void*)data, sb.st_size);
    close(fd);
  }
  printf("pages=%ld litpar=%ld fallback=%ld bad=%ld  lits=%ld res=%ld pieces=%ld\n",
         pages, litpar_pages, fb, bad, lits, res, pieces);
  return bad != 0;
}
