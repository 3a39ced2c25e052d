// Latency-ceiling probe for the walk kernel's access pattern.
//
// The walk is a per-lane dependent chain: each element crossing needs the
// current tet's 144-B record (4 planes + neighbors) before the next tet is
// known.  This kernel strips away all geometry and measures the maximum
// dependent-hop rate the chip sustains at the walk's own launch shape
// (1024 blocks x 256 threads, ~7 waves/SIMD): one 32-B record read + one
// 16-B next-pointer read per hop over an L3-resident table.
// Build: hipcc --offload-arch=gfx950 -O3 -shared -fPIC tools/chase.hip -o chase.so
#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>

namespace {

__global__ void k_fill(int32_t *next, double *rec, int64_t n, uint64_t seed) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n * 4;
       i += stride) {
    uint64_t x = (uint64_t)i * 6364136223846793005ull + seed;
    x ^= x >> 33;
    x *= 0xff51afd7ed558ccdull;
    x ^= x >> 33;
    next[i] = (int32_t)(x % (uint64_t)n);
    rec[i * 4] = (double)(x & 0xffff);
    rec[i * 4 + 1] = 1.0;
    rec[i * 4 + 2] = 2.0;
    rec[i * 4 + 3] = 3.0;
  }
}

__global__ void k_chase(const int32_t *__restrict__ next,
                        const double *__restrict__ rec, int64_t n, int hops,
                        double *__restrict__ sink) {
  const int64_t lane = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  uint32_t idx = (uint32_t)(((uint64_t)lane * 2654435761ull) % (uint64_t)n);
  double acc = 0.0;
  int f = 0;
  for (int h = 0; h < hops; ++h) {
    const int64_t base = (int64_t)idx * 4;
    // one 32-B piece of the record (like one plane) + the 4 neighbor ids
    acc += rec[base * 4 + f * 4] + rec[base * 4 + f * 4 + 1];
    const int32_t n0 = next[base + ((f + 0) & 3)];
    const int32_t n1 = next[base + ((f + 1) & 3)];
    f = ((uint32_t)n0 ^ (uint32_t)h) & 3;
    idx = (uint32_t)((f & 1) ? n1 : n0);
  }
  if (acc == 1e301) sink[lane] = acc;
}

// Variant doing the WALK's actual per-hop work: read the full 128-B plane
// record + 16-B neighbor row, 8 fp64 plane evaluations (4 planes x 2
// endpoints), cross-multiplied face selection, one fp64 division, and
// (mode 1) one fp64 atomicAdd into a 1M-entry tally.
__global__ void k_chase_walklike(const int32_t *__restrict__ next,
                                 const double *__restrict__ rec, int64_t n,
                                 int hops, int do_atomic,
                                 double *__restrict__ tally,
                                 double *__restrict__ sink) {
  const int64_t lane = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  uint32_t idx = (uint32_t)(((uint64_t)lane * 2654435761ull) % (uint64_t)n);
  const double ox = 0.1 + (lane & 7) * 0.01, oy = 0.2, oz = 0.3;
  const double dx = 0.9, dy = 0.8, dz = 0.7;
  double t_cur = 0.0;
  double acc = 0.0;
  for (int h = 0; h < hops; ++h) {
    const int64_t base = (int64_t)idx * 4;
    const double *pl = rec + base * 4;
    double num_best = 2.0, den_best = 1.0;
    int ef = 0;
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      const double vd =
          pl[f * 4] * dx + pl[f * 4 + 1] * dy + pl[f * 4 + 2] * dz - pl[f * 4 + 3];
      const double vo =
          pl[f * 4] * ox + pl[f * 4 + 1] * oy + pl[f * 4 + 2] * oz - pl[f * 4 + 3];
      const double den = vo - vd;
      if (den > 0.0 && vo * den_best < num_best * den) {
        num_best = vo < 0 ? -vo : vo;
        den_best = den;
        ef = f;
      }
    }
    const double t = num_best / (den_best + 1.0);
    t_cur = t > t_cur ? t_cur : t; // keep t bounded, data-dependent
    if (do_atomic == 1) atomicAdd(&tally[idx], t);
    if (do_atomic == 2)
      atomicAdd((unsigned long long *)&tally[idx],
                (unsigned long long)(t * 1048576.0));
    acc += t;
    idx = (uint32_t)next[base + ef];
  }
  if (acc == 1e301) sink[lane] = acc + t_cur;
}

// Two independent chains per lane (ILP-2): doubles memory-level
// parallelism per lane at the cost of ~2x register state.  Probes whether
// the real walk kernel should interleave two particles per lane.
__global__ void k_chase_walklike2(const int32_t *__restrict__ next,
                                  const double *__restrict__ rec, int64_t n,
                                  int hops, int do_atomic,
                                  double *__restrict__ tally,
                                  double *__restrict__ sink) {
  const int64_t lane = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  uint32_t idxA = (uint32_t)(((uint64_t)lane * 2654435761ull) % (uint64_t)n);
  uint32_t idxB = (uint32_t)(((uint64_t)(lane + 9173) * 40503ull) % (uint64_t)n);
  const double ox = 0.1 + (lane & 7) * 0.01, oy = 0.2, oz = 0.3;
  const double dx = 0.9, dy = 0.8, dz = 0.7;
  double accA = 0.0, accB = 0.0;
  for (int h = 0; h < hops; ++h) {
    const int64_t baseA = (int64_t)idxA * 4, baseB = (int64_t)idxB * 4;
    const double *plA = rec + baseA * 4, *plB = rec + baseB * 4;
    double numA = 2.0, denA = 1.0, numB = 2.0, denB = 1.0;
    int efA = 0, efB = 0;
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      const double vdA = plA[f * 4] * dx + plA[f * 4 + 1] * dy +
                         plA[f * 4 + 2] * dz - plA[f * 4 + 3];
      const double voA = plA[f * 4] * ox + plA[f * 4 + 1] * oy +
                         plA[f * 4 + 2] * oz - plA[f * 4 + 3];
      const double dA = voA - vdA;
      if (dA > 0.0 && voA * denA < numA * dA) {
        numA = voA < 0 ? -voA : voA;
        denA = dA;
        efA = f;
      }
      const double vdB = plB[f * 4] * dx + plB[f * 4 + 1] * dy +
                         plB[f * 4 + 2] * dz - plB[f * 4 + 3];
      const double voB = plB[f * 4] * ox + plB[f * 4 + 1] * oy +
                         plB[f * 4 + 2] * oz - plB[f * 4 + 3];
      const double dB = voB - vdB;
      if (dB > 0.0 && voB * denB < numB * dB) {
        numB = voB < 0 ? -voB : voB;
        denB = dB;
        efB = f;
      }
    }
    const double tA = numA / (denA + 1.0), tB = numB / (denB + 1.0);
    if (do_atomic) {
      atomicAdd(&tally[idxA], tA);
      atomicAdd(&tally[idxB], tB);
    }
    accA += tA;
    accB += tB;
    idxA = (uint32_t)next[baseA + efA];
    idxB = (uint32_t)next[baseB + efB];
  }
  if (accA == 1e301) sink[lane] = accA + accB;
}

} // namespace

extern "C" double chase_walklike2_bench(int64_t n, int hops, int blocks,
                                        int threads, int reps, int do_atomic) {
  int32_t *next = nullptr;
  double *rec = nullptr, *sink = nullptr, *tally = nullptr;
  if (hipMalloc(&next, n * 4 * sizeof(int32_t)) != hipSuccess) return -1;
  if (hipMalloc(&rec, n * 16 * sizeof(double)) != hipSuccess) return -1;
  if (hipMalloc(&sink, (int64_t)blocks * threads * sizeof(double)) != hipSuccess)
    return -1;
  if (hipMalloc(&tally, n * sizeof(double)) != hipSuccess) return -1;
  (void)hipMemset(tally, 0, n * sizeof(double));
  k_fill<<<1024, 256>>>(next, rec, n, 0x9e3779b97f4a7c15ull);
  (void)hipDeviceSynchronize();
  hipEvent_t a, b;
  (void)hipEventCreate(&a);
  (void)hipEventCreate(&b);
  k_chase_walklike2<<<blocks, threads>>>(next, rec, n, hops, do_atomic, tally, sink);
  (void)hipDeviceSynchronize();
  (void)hipEventRecord(a, 0);
  for (int r = 0; r < reps; ++r)
    k_chase_walklike2<<<blocks, threads>>>(next, rec, n, hops, do_atomic, tally, sink);
  (void)hipEventRecord(b, 0);
  (void)hipEventSynchronize(b);
  float ms = 0;
  (void)hipEventElapsedTime(&ms, a, b);
  const double hops_total = 2.0 * (double)blocks * threads * hops * reps;
  (void)hipFree(next);
  (void)hipFree(rec);
  (void)hipFree(sink);
  (void)hipFree(tally);
  (void)hipEventDestroy(a);
  (void)hipEventDestroy(b);
  return hops_total / (ms * 1e-3);
}

extern "C" double chase_walklike_bench(int64_t n, int hops, int blocks,
                                       int threads, int reps, int do_atomic) {
  int32_t *next = nullptr;
  double *rec = nullptr, *sink = nullptr, *tally = nullptr;
  if (hipMalloc(&next, n * 4 * sizeof(int32_t)) != hipSuccess) return -1;
  if (hipMalloc(&rec, n * 16 * sizeof(double)) != hipSuccess) return -1;
  if (hipMalloc(&sink, (int64_t)blocks * threads * sizeof(double)) != hipSuccess)
    return -1;
  if (hipMalloc(&tally, n * sizeof(double)) != hipSuccess) return -1;
  (void)hipMemset(tally, 0, n * sizeof(double));
  k_fill<<<1024, 256>>>(next, rec, n, 0x9e3779b97f4a7c15ull);
  (void)hipDeviceSynchronize();
  hipEvent_t a, b;
  (void)hipEventCreate(&a);
  (void)hipEventCreate(&b);
  k_chase_walklike<<<blocks, threads>>>(next, rec, n, hops, do_atomic, tally, sink);
  (void)hipDeviceSynchronize();
  (void)hipEventRecord(a, 0);
  for (int r = 0; r < reps; ++r)
    k_chase_walklike<<<blocks, threads>>>(next, rec, n, hops, do_atomic, tally, sink);
  (void)hipEventRecord(b, 0);
  (void)hipEventSynchronize(b);
  float ms = 0;
  (void)hipEventElapsedTime(&ms, a, b);
  const double hops_total = (double)blocks * threads * hops * reps;
  (void)hipFree(next);
  (void)hipFree(rec);
  (void)hipFree(sink);
  (void)hipFree(tally);
  (void)hipEventDestroy(a);
  (void)hipEventDestroy(b);
  return hops_total / (ms * 1e-3);
}

extern "C" double chase_bench(int64_t n, int hops, int blocks, int threads,
                              int reps) {
  int32_t *next = nullptr;
  double *rec = nullptr, *sink = nullptr;
  if (hipMalloc(&next, n * 4 * sizeof(int32_t)) != hipSuccess) return -1;
  if (hipMalloc(&rec, n * 16 * sizeof(double)) != hipSuccess) return -1;
  if (hipMalloc(&sink, (int64_t)blocks * threads * sizeof(double)) != hipSuccess)
    return -1;
  k_fill<<<1024, 256>>>(next, rec, n, 0x9e3779b97f4a7c15ull);
  (void)hipDeviceSynchronize();
  hipEvent_t a, b;
  (void)hipEventCreate(&a);
  (void)hipEventCreate(&b);
  k_chase<<<blocks, threads>>>(next, rec, n, hops, sink); // warm
  (void)hipDeviceSynchronize();
  (void)hipEventRecord(a, 0);
  for (int r = 0; r < reps; ++r) k_chase<<<blocks, threads>>>(next, rec, n, hops, sink);
  (void)hipEventRecord(b, 0);
  (void)hipEventSynchronize(b);
  float ms = 0;
  (void)hipEventElapsedTime(&ms, a, b);
  const double hops_total = (double)blocks * threads * hops * reps;
  (void)hipFree(next);
  (void)hipFree(rec);
  (void)hipFree(sink);
  (void)hipEventDestroy(a);
  (void)hipEventDestroy(b);
  return hops_total / (ms * 1e-3); // dependent hops per second
}
