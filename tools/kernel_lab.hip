// Standalone microbenchmark lab for the histogram-accumulation design on
// gfx950. Measures LDS-atomic variants of the (row, feature) -> bin
// accumulate that dominates GBT training (SQ counters show the LDS array
// ~100% busy under scattered ds_add_f32).
//
// Build: hipcc --offload-arch=gfx950 -O3 tools/kernel_lab.hip -o tools/kernel_lab
// Run (GPU box): ./tools/kernel_lab
#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <vector>

#define CHECK(x) do { hipError_t e = (x); if (e != hipSuccess) { \
  printf("HIP error %s @%d\n", hipGetErrorString(e), __LINE__); exit(1);} } while (0)

constexpr int kBins = 256;
constexpr int kBlock = 256;

// Common frame: each block owns a row chunk for one feature; LDS hist for
// one slot; every variant performs the same loads, different accumulates.
template <int MODE>
__global__ void hist_variant(const uint8_t* __restrict__ bins,
                             const float2* __restrict__ gh,
                             const int32_t* __restrict__ node_ids,
                             float* __restrict__ out, int64_t N,
                             int64_t rows_per_block, float gscale,
                             float hscale) {
  __shared__ __attribute__((aligned(16))) float lh[kBins * 4];
  unsigned long long* lh64 = reinterpret_cast<unsigned long long*>(lh);
  for (int i = threadIdx.x; i < kBins * 4; i += blockDim.x) lh[i] = 0.f;
  __syncthreads();
  const int f = blockIdx.x;
  const int64_t row0 = (int64_t)blockIdx.y * rows_per_block;
  const int64_t row1 = min(row0 + rows_per_block, N);
  const uint8_t* fb = bins + (int64_t)f * N;
  const int64_t stride = blockDim.x;
  int64_t i = row0 + threadIdx.x;
  const int64_t bulk = row1 - 3 * stride;
  for (; i < bulk; i += 4 * stride) {
    int nid[4]; float2 v[4]; uint8_t b[4];
#pragma unroll
    for (int u = 0; u < 4; ++u) nid[u] = node_ids[i + u * stride];
#pragma unroll
    for (int u = 0; u < 4; ++u) v[u] = gh[i + u * stride];
#pragma unroll
    for (int u = 0; u < 4; ++u) b[u] = fb[i + u * stride];
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      if (nid[u] < 0) continue;
      const int bin = b[u];
      if (MODE == 0) {          // 3x f32 atomics (baseline)
        float* p = lh + bin * 3;
        atomicAdd(p, v[u].x);
        atomicAdd(p + 1, v[u].y);
        atomicAdd(p + 2, (v[u].y != 0.f) ? 1.f : 0.f);
      } else if (MODE == 1) {   // 1x f32 atomic (scaling check)
        atomicAdd(lh + bin, v[u].x);
      } else if (MODE == 2) {   // f32 g + u64 (h fixed 40b | count<<40)
        atomicAdd(lh + bin, v[u].x);
        const unsigned long long hq =
            (unsigned long long)(v[u].y * hscale + 0.5f);
        const unsigned long long pk =
            hq | ((unsigned long long)(v[u].y != 0.f) << 40);
        atomicAdd(lh64 + kBins + bin, pk);
      } else if (MODE == 3) {   // single u64: g 32b | h 20b | count 12b
        const unsigned long long gq =
            (unsigned long long)((v[u].x + 2.f) * gscale + 0.5f);
        const unsigned long long hq =
            (unsigned long long)(v[u].y * hscale + 0.5f);
        const unsigned long long pk =
            hq | (gq << 20) | (1ull << 52);
        atomicAdd(lh64 + bin, pk);
      } else if (MODE == 4) {   // 2x f32 (g,h; no count)
        float* p = lh + bin * 2;
        atomicAdd(p, v[u].x);
        atomicAdd(p + 1, v[u].y);
      } else if (MODE == 5) {   // no atomics: loads + compute floor
        lh[threadIdx.x] += v[u].x * (bin + 1);
      } else if (MODE == 6) {   // 3x global f32 atomics (L2 resident)
        float* p = out + ((int64_t)f * kBins + bin) * 3;
        atomicAdd(p, v[u].x);
        atomicAdd(p + 1, v[u].y);
        atomicAdd(p + 2, (v[u].y != 0.f) ? 1.f : 0.f);
      } else if (MODE == 7) {   // 1x f64 atomic (g as double)
        atomicAdd(reinterpret_cast<double*>(lh64 + bin), (double)v[u].x);
      } else if (MODE == 8) {   // PRODUCTION: f64 g + u64(h fixed | cnt<<44)
        atomicAdd(reinterpret_cast<double*>(lh64) + bin * 2, (double)v[u].x);
        const unsigned long long hq =
            (unsigned long long)(v[u].y * 16777216.0f + 0.5f);
        atomicAdd(lh64 + bin * 2 + 1,
                  hq | ((unsigned long long)(v[u].y != 0.f) << 44));
      } else if (MODE == 9) {   // global u64 atomics spread (merge candidate)
        unsigned long long* p = reinterpret_cast<unsigned long long*>(out) +
                                ((int64_t)f * kBins + bin);
        atomicAdd(p, 1ull);
      } else if (MODE == 10) {  // global f64 atomics spread (merge candidate)
        atomicAdd(reinterpret_cast<double*>(out) + ((int64_t)f * kBins + bin),
                  (double)v[u].x);
      }
    }
  }
  __syncthreads();
  // fold LDS into out so nothing is optimized away
  for (int k = threadIdx.x; k < kBins * 3; k += blockDim.x) {
    const float vv = lh[k];
    if (vv != 0.f && MODE != 6)
      atomicAdd(&out[((int64_t)f * kBins + k / 3) * 3 + k % 3], vv);
  }
}

template <int MODE>
float run_mode(const uint8_t* bins, const float2* gh, const int32_t* nids,
               float* out, int64_t N, int F, int iters) {
  const int chunks = 8192 / F;
  const int64_t rpb = (N + chunks - 1) / chunks;
  dim3 grid(F, chunks);
  // warmup
  hipLaunchKernelGGL(HIP_KERNEL_NAME(hist_variant<MODE>), grid, dim3(kBlock),
                     0, 0, bins, gh, nids, out, N, rpb, 1e6f, 65536.f);
  CHECK(hipDeviceSynchronize());
  hipEvent_t a, b;
  hipEventCreate(&a);
  hipEventCreate(&b);
  hipEventRecord(a);
  for (int it = 0; it < iters; ++it)
    hipLaunchKernelGGL(HIP_KERNEL_NAME(hist_variant<MODE>), grid,
                       dim3(kBlock), 0, 0, bins, gh, nids, out, N, rpb, 1e6f,
                       65536.f);
  hipEventRecord(b);
  CHECK(hipEventSynchronize(b));
  float ms = 0;
  hipEventElapsedTime(&ms, a, b);
  return ms / iters;
}

int main() {
  const int64_t N = 11000000;
  const int F = 28;
  uint8_t* bins;
  float2* gh;
  int32_t* nids;
  float* out;
  CHECK(hipMalloc(&bins, (size_t)F * N));
  CHECK(hipMalloc(&gh, N * sizeof(float2)));
  CHECK(hipMalloc(&nids, N * sizeof(int32_t)));
  CHECK(hipMalloc(&out, (size_t)F * kBins * 4 * sizeof(float)));
  // host init (pseudo-random)
  {
    std::vector<uint8_t> hb((size_t)F * N);
    std::vector<float2> hg(N);
    std::vector<int32_t> hn(N);
    unsigned s = 12345;
    for (size_t i = 0; i < hb.size(); ++i) {
      s = s * 1664525u + 1013904223u;
      hb[i] = (s >> 24) & 255;
    }
    for (int64_t i = 0; i < N; ++i) {
      s = s * 1664525u + 1013904223u;
      hg[i] = make_float2((float)(s & 1023) / 1024.f - 0.5f,
                          0.25f * (float)((s >> 10) & 1023) / 1024.f);
      hn[i] = 0;
    }
    CHECK(hipMemcpy(bins, hb.data(), hb.size(), hipMemcpyHostToDevice));
    CHECK(hipMemcpy(gh, hg.data(), N * sizeof(float2),
                    hipMemcpyHostToDevice));
    CHECK(hipMemcpy(nids, hn.data(), N * sizeof(int32_t),
                    hipMemcpyHostToDevice));
  }
  const double visits = (double)F * N;
  struct Row { const char* name; float ms; };
  auto report = [&](const char* name, float ms) {
    printf("%-34s %8.2f ms  %7.2f Gvisits/s\n", name, ms,
           visits / ms / 1e6);
    fflush(stdout);
  };
  report("3x ds_add_f32 (baseline)", run_mode<0>(bins, gh, nids, out, N, F, 5));
  report("1x ds_add_f32", run_mode<1>(bins, gh, nids, out, N, F, 5));
  report("f32 g + u64(h|cnt)", run_mode<2>(bins, gh, nids, out, N, F, 5));
  report("1x u64(g|h|cnt)", run_mode<3>(bins, gh, nids, out, N, F, 5));
  report("2x ds_add_f32 (no count)", run_mode<4>(bins, gh, nids, out, N, F, 5));
  report("no atomics (load floor)", run_mode<5>(bins, gh, nids, out, N, F, 5));
  report("3x global f32 atomics", run_mode<6>(bins, gh, nids, out, N, F, 2));
  report("1x ds_add_f64", run_mode<7>(bins, gh, nids, out, N, F, 5));
  report("f64 g + u64(h|cnt) [prod]", run_mode<8>(bins, gh, nids, out, N, F, 5));
  report("1x global u64 atomic", run_mode<9>(bins, gh, nids, out, N, F, 2));
  report("1x global f64 atomic", run_mode<10>(bins, gh, nids, out, N, F, 2));
  return 0;
}
