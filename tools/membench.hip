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

// ---- differential ladder: from pure reads to the full bitmatrix shape.
// MODE 0: glds staging + barrier + LDS-gather compute, result XOR-sunk
//         (no global stores)
// MODE 1: MODE 0 + nontemporal 16 B stores in the real scattered pattern
// MODE 2: MODE 1 but plain (temporal) stores
// MODE 3: MODE 1 volume, but stores to a CONTIGUOUS per-block region
//         (granularity probe: same bytes, ideal locality)
// MODE 4: output rows accumulated in LDS across the superword's 8
//         windows, flushed as contiguous 2 KiB packets (the candidate
//         kernel design)
// Shape fixed at k=8 w=8 q=256 pkt=2048 m=3 (the bench default); ops
// table synthesised on-device with the same per-row count (32).
template <int MODE>
__global__ __launch_bounds__(256, 2) void stage_pattern(
    const uint8_t* __restrict__ bufbase, long n_sw, int wpb,
    uint32_t* __restrict__ sink) {
  const uint8_t* buf = bufbase + (size_t)blockIdx.y * (11u << 20);
  uint8_t* obuf = const_cast<uint8_t*>(buf) + (8u << 20);
  constexpr int K = 8, W = 8, Q = 256, PKT = 2048, M = 3;
  constexpr int VQ = Q / 16, ITEMS = K * W * VQ, ROWS = M * W;
  __shared__ uint8_t s_data[K * W * Q];
  __shared__ uint8_t s_out[MODE == 4 ? ROWS * PKT : 16];
  __shared__ uint16_t s_ops[ROWS * 32];
  for (int t = threadIdx.x; t < ROWS * 32; t += blockDim.x)
    s_ops[t] = (uint16_t)(((t >> 5) * 7 + (t & 31) * 13 + ((t & 31) >> 2)) %
               (K * W));
  const long chunk_bytes = 1 << 20;
  v4u acc_sink = {0, 0, 0, 0};
  for (long it = 0; it < wpb; it++) {
    const long wt = (long)blockIdx.x * wpb + it;
    if (wt >= n_sw * (PKT / Q)) break;
    const int win = (int)(wt % (PKT / Q));
    const long sw = wt / (PKT / Q);
    const long sw_off = sw * (long)W * PKT + (long)win * Q;
    __syncthreads();
    {
      const int lane = threadIdx.x & 63;
      const int nwaves = blockDim.x >> 6;
      for (int t0 = (int)(threadIdx.x >> 6) * 64; t0 < ITEMS;
           t0 += nwaves * 64) {
        const int t = t0 + lane;
        const int jc = t >> 4;
        const int v = t & 15;
        const int j = jc >> 3, c = jc & 7;
        const uint8_t* src = buf + (long)j * chunk_bytes + sw_off +
                             (long)c * PKT + (long)v * 16;
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) uint32_t*)src,
            (__attribute__((address_space(3))) uint32_t*)(s_data +
                                                          (size_t)t0 * 16),
            16, 0, 2);
      }
    }
    __syncthreads();
    for (int t = threadIdx.x; t < ROWS * VQ; t += blockDim.x) {
      const int r = t >> 4;
      const int v = t & 15;
      v4u acc = {0, 0, 0, 0};
      for (int o = 0; o < 32; o++) {
        const int jc = s_ops[r * 32 + o];
        const v4u d = *reinterpret_cast<const v4u*>(
            s_data + (size_t)jc * Q + (size_t)v * 16);
        acc.x ^= d.x; acc.y ^= d.y; acc.z ^= d.z; acc.w ^= d.w;
      }
      if (MODE == 0) {
        acc_sink.x ^= acc.x; acc_sink.y ^= acc.y;
        acc_sink.z ^= acc.z; acc_sink.w ^= acc.w;
      } else if (MODE == 3) {
        v4u* dst = reinterpret_cast<v4u*>(
            obuf + ((long)blockIdx.x * wpb + it) * (ROWS * Q) + (long)t * 16);
        __builtin_nontemporal_store(acc, dst);
      } else if (MODE == 4) {
        *reinterpret_cast<v4u*>(s_out + (size_t)r * PKT + (size_t)win * Q +
                                (size_t)v * 16) = acc;
      } else {
        v4u* dst = reinterpret_cast<v4u*>(
            obuf + (long)(r >> 3) * chunk_bytes + sw * (long)W * PKT +
            (long)(r & 7) * PKT + (long)win * Q + (long)v * 16);
        if (MODE == 1)
          __builtin_nontemporal_store(acc, dst);
        else
          *dst = acc;
      }
    }
    if (MODE == 4 && win == (PKT / Q) - 1) {
      // superword complete: flush all rows as contiguous 2 KiB packets
      __syncthreads();
      for (int t = threadIdx.x; t < ROWS * (PKT / 16); t += blockDim.x) {
        const int r = t / (PKT / 16);
        const int v = t - r * (PKT / 16);
        const v4u val = *reinterpret_cast<const v4u*>(
            s_out + (size_t)r * PKT + (size_t)v * 16);
        __builtin_nontemporal_store(
            val, reinterpret_cast<v4u*>(obuf + (long)(r >> 3) * chunk_bytes +
                                        sw * (long)W * PKT +
                                        (long)(r & 7) * PKT + (long)v * 16));
      }
    }
  }
  if (acc_sink.x + acc_sink.y + acc_sink.z + acc_sink.w == 0xdeadbeefu)
    sink[threadIdx.x] = acc_sink.x;
}

// Pure 3-read:1-write streaming mix (out = a ^ b ^ c), registers only —
// the achievable ceiling for the bitmatrix kernel's ~73/27 R/W mix.
__global__ __launch_bounds__(256) void xor3_kernel(
    const uint8_t* __restrict__ a, const uint8_t* __restrict__ b,
    const uint8_t* __restrict__ c, uint8_t* __restrict__ o, long n_vecs) {
  for (long p = (long)blockIdx.x * blockDim.x + threadIdx.x; p < n_vecs;
       p += (long)gridDim.x * blockDim.x) {
    const v4u va = __builtin_nontemporal_load(
        reinterpret_cast<const v4u*>(a) + p);
    const v4u vb = __builtin_nontemporal_load(
        reinterpret_cast<const v4u*>(b) + p);
    const v4u vc = __builtin_nontemporal_load(
        reinterpret_cast<const v4u*>(c) + p);
    v4u vo;
    vo.x = va.x ^ vb.x ^ vc.x; vo.y = va.y ^ vb.y ^ vc.y;
    vo.z = va.z ^ vb.z ^ vc.z; vo.w = va.w ^ vb.w ^ vc.w;
    __builtin_nontemporal_store(vo, reinterpret_cast<v4u*>(o) + p);
  }
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
  // ---- ladder runs: 1 MiB chunks, 64 superwords/chunk, many stripes
  {
    const long n_sw = 64;       // one chunk's superwords (1 MiB chunks)
    const long stripes = 512;   // 11 MiB arena per stripe within 8 GiB
    const int wpb = 8;
    const long windows = n_sw * 8;
    const unsigned blocks_per_stripe = (unsigned)(windows / wpb);
    const dim3 grid(blocks_per_stripe, (unsigned)stripes);
    // per launch: read 16 KiB + write 6 KiB per window
    const double rd = (double)stripes * windows * 16384;
    const double wr = (double)stripes * windows * 6144;
    for (int mode = 0; mode < 5; mode++) {
      HT(hipDeviceSynchronize());
      HT(hipEventRecord(e0));
      for (int r = 0; r < 3; r++) {
        switch (mode) {
          case 0:
            hipLaunchKernelGGL(stage_pattern<0>, grid, dim3(256), 0, 0, d,
                               n_sw, wpb, sink);
            break;
          case 1:
            hipLaunchKernelGGL(stage_pattern<1>, grid, dim3(256), 0, 0, d,
                               n_sw, wpb, sink);
            break;
          case 2:
            hipLaunchKernelGGL(stage_pattern<2>, grid, dim3(256), 0, 0, d,
                               n_sw, wpb, sink);
            break;
          case 3:
            hipLaunchKernelGGL(stage_pattern<3>, grid, dim3(256), 0, 0, d,
                               n_sw, wpb, sink);
            break;
          case 4:
            hipLaunchKernelGGL(stage_pattern<4>, grid, dim3(256), 0, 0, d,
                               n_sw, wpb, sink);
            break;
        }
      }
      HT(hipEventRecord(e1));
      HT(hipEventSynchronize(e1));
      float ms = 0;
      HT(hipEventElapsedTime(&ms, e0, e1));
      const double bytes = mode == 0 ? rd : rd + wr;
      printf("{\"ladder_mode\": %d, \"GBs\": %.0f, \"read_GBs_equiv\":"
             " %.0f}\n", mode, 3 * bytes / (ms * 1e6),
             3 * rd / (ms * 1e6));
      fflush(stdout);
    }
  }
  {
    // xor3: 2 GiB per stream
    const long sz = 2ll << 30;
    const long n_vecs = sz >> 4;
    HT(hipDeviceSynchronize());
    HT(hipEventRecord(e0));
    for (int r = 0; r < 3; r++)
      hipLaunchKernelGGL(xor3_kernel, dim3(8192), dim3(256), 0, 0, d,
                         d + sz, d + 2 * sz, d + 3 * sz, n_vecs);
    HT(hipEventRecord(e1));
    HT(hipEventSynchronize(e1));
    float ms = 0;
    HT(hipEventElapsedTime(&ms, e0, e1));
    printf("{\"pattern\": \"xor3-streaming\", \"GBs\": %.0f}\n",
           3 * 4.0 * sz / (ms * 1e6));
  }
  return 0;
}
