// membench — HBM read-bandwidth probe for (granule, stride) patterns.
// Answers: what does gfx950 HBM deliver for G-byte pieces strided S bytes
// (the jerasure packet layout makes the bitmatrix kernel read 256 B
// pieces at 2 KiB stride), vs contiguous streaming? Standalone; writes
// one JSON line per pattern.
//   hipcc --offload-arch=gfx950 -O3 tools/membench.hip -o gpurun_out/membench
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>

#define HT(x) do { hipError_t e = (x); if (e != hipSuccess) { \
  fprintf(stderr, "HIP %s\n", hipGetErrorString(e)); exit(1); } } while (0)

typedef uint32_t v4u __attribute__((ext_vector_type(4)));

// Each block reads `pieces` pieces of `gran` bytes, consecutive pieces
// `stride` bytes apart (gran <= stride); XOR-reduces to defeat DCE.
__global__ __launch_bounds__(256) void read_pattern(
    const uint8_t* __restrict__ buf, long n_blocks_work, long gran,
    long stride, long pieces, long big, uint32_t* __restrict__ sink) {
  // big > 0: pieces are (j, c) with j in 8 chunk regions `big` apart and
  // c in pieces/8 packets `stride` apart — the real k=8 staging shape
  v4u acc = {0, 0, 0, 0};
  const long vecs_per_piece = gran >> 4;
  const long items = pieces * vecs_per_piece;
  const long cpj = pieces / 8;
  for (long w = blockIdx.x; w < n_blocks_work; w += gridDim.x) {
    const uint8_t* base = buf + (big ? w * stride : w * stride * pieces);
    // all pieces fetched concurrently, 16 B per lane within a piece —
    // the same item shape as the bitmatrix kernel's staging phase
    for (long t = threadIdx.x; t < items; t += blockDim.x) {
      const long p = t / vecs_per_piece, v = t - p * vecs_per_piece;
      const long off = big ? (p / cpj) * big + (p - (p / cpj) * cpj) * stride
                           : p * stride;
      const v4u d = __builtin_nontemporal_load(
          reinterpret_cast<const v4u*>(base + off + (v << 4)));
      acc.x ^= d.x; acc.y ^= d.y; acc.z ^= d.z; acc.w ^= d.w;
    }
  }
  if (acc.x + acc.y + acc.z + acc.w == 0xdeadbeefu)
    sink[threadIdx.x] = acc.x;  // never true for random fill
}

int main() {
  const size_t total = 8ull << 30;  // 8 GiB working set
  uint8_t* d;
  uint32_t* sink;
  HT(hipMalloc(&d, total));
  HT(hipMalloc(&sink, 1024 * 4));
  HT(hipMemset(d, 0x5a, total));
  hipEvent_t e0, e1;
  HT(hipEventCreate(&e0));
  HT(hipEventCreate(&e1));

  struct Pat { long gran, stride, pieces, big; const char* name; };
  const long MB = 1 << 20;
  Pat pats[] = {
      {2048, 2048, 8, 0, "contig-2k"},
      {1024, 2048, 8, 0, "1k-of-2k"},
      {512, 2048, 8, 0, "512-of-2k"},
      {256, 2048, 8, 0, "256-of-2k"},   // the bitmatrix q=256 pattern
      {128, 2048, 8, 0, "128-of-2k"},
      {256, 4096, 8, 0, "256-of-4k"},
      {65536, 65536, 1, 0, "contig-64k"},
      {256, 2048, 64, MB, "k8-256-of-2k"},   // real staging shape q=256
      {512, 2048, 64, MB, "k8-512-of-2k"},
      {2048, 2048, 64, MB, "k8-contig-2k"},
  };
  for (auto& p : pats) {
    const long unit = p.stride * p.pieces;
    long n_units = (long)(total / unit);
    if (p.big) {
      n_units = (long)((total - 8 * p.big) / p.stride);
      if (n_units > (256 << 10)) n_units = 256 << 10;
    }
    const double bytes = (double)n_units * p.gran * p.pieces;
    // warm
    hipLaunchKernelGGL(read_pattern, dim3(8192), dim3(256), 0, 0, d,
                       n_units, p.gran, p.stride, p.pieces, p.big, sink);
    HT(hipDeviceSynchronize());
    HT(hipEventRecord(e0));
    for (int r = 0; r < 3; r++)
      hipLaunchKernelGGL(read_pattern, dim3(8192), dim3(256), 0, 0, d,
                         n_units, p.gran, p.stride, p.pieces, p.big, sink);
    HT(hipEventRecord(e1));
    HT(hipEventSynchronize(e1));
    float ms = 0;
    HT(hipEventElapsedTime(&ms, e0, e1));
    printf("{\"pattern\": \"%s\", \"gran\": %ld, \"stride\": %ld, "
           "\"read_GBs\": %.0f}\n",
           p.name, p.gran, p.stride, 3 * bytes / (ms * 1e6));
    fflush(stdout);
  }
  return 0;
}
