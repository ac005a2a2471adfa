/* membw_probe — isolate why the scan kernel plateaus at ~1.9 TB/s.
 * Variants over the same 1.68 GB working set (4 f64-ish streams):
 *   A: pure grid-stride streaming read (double2/lane), no LDS, no barriers
 *   B: A + 33 KB dummy LDS per block (same blocks/CU as the scan kernel)
 *   C: chunked LDS round trip: load 1024-row chunk -> ds_write -> barrier ->
 *      ds_read+sum -> barrier (the scan kernel's skeleton)
 *   D: C + register-staged prefetch of the next chunk (the T14 split)
 * Build: hipcc --offload-arch=gfx950 -O3 membw_probe.hip -o membw_probe
 */
#include <hip/hip_runtime.h>
#include <cstdio>
#include <vector>

#define WG 256
#define CHUNK 1024
#define GAS __attribute__((address_space(1)))

typedef double double2_t __attribute__((ext_vector_type(2)));
typedef int int2_t __attribute__((ext_vector_type(2)));

__global__ __launch_bounds__(WG, 2) void k_streamA(
    const double *__restrict__ a, const double *__restrict__ b,
    const double *__restrict__ c, const double *__restrict__ d,
    long long n, double *out) {
  const GAS double2_t *A = (const GAS double2_t *)(uintptr_t)a;
  const GAS double2_t *B = (const GAS double2_t *)(uintptr_t)b;
  const GAS double2_t *C = (const GAS double2_t *)(uintptr_t)c;
  const GAS double2_t *D = (const GAS double2_t *)(uintptr_t)d;
  long long i = blockIdx.x * (long long)WG + threadIdx.x;
  long long stride = gridDim.x * (long long)WG;
  double2_t s = {0, 0};
  for (; i < n / 2; i += stride) {
    double2_t x = A[i], y = B[i], z = C[i], w = D[i];
    s.x += x.x + y.x + z.x + w.x;
    s.y += x.y + y.y + z.y + w.y;
  }
  double v = s.x + s.y;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  if ((threadIdx.x & 63) == 0 && v != 0.0) atomicAdd(out, v);
}

template <int LDSKB, int STAGED, int TILE_CHUNKS = 1>
__global__ __launch_bounds__(WG, 2) void k_chunked(
    const double *__restrict__ a, const double *__restrict__ b,
    const double *__restrict__ c, const double *__restrict__ d,
    long long n, double *out) {
  __shared__ __attribute__((aligned(16))) double lds[LDSKB * 128]; /* KB -> doubles */
  const GAS double2_t *src[4] = {
    (const GAS double2_t *)(uintptr_t)a, (const GAS double2_t *)(uintptr_t)b,
    (const GAS double2_t *)(uintptr_t)c, (const GAS double2_t *)(uintptr_t)d };
  const int tid = threadIdx.x;
  long long nchunks = n / CHUNK;
  double acc = 0.0;
  double2_t st[4][2];
  /* TILE_CHUNKS>1: contiguous multi-chunk tiles per block (the engine's
   * mapping); 1: fine-grained grid-stride (chip-wide sequential sweep) */
  long long ntiles = nchunks / TILE_CHUNKS;
  for (long long tt = blockIdx.x; tt < ntiles; tt += gridDim.x) {
  long long t0 = tt * TILE_CHUNKS;
  if (STAGED) {
#pragma unroll
    for (int cc = 0; cc < 4; cc++)
#pragma unroll
      for (int p = 0; p < 2; p++)
        st[cc][p] = src[cc][t0 * (CHUNK / 2) + tid + p * WG];
  }
  for (long long t = t0; t < t0 + TILE_CHUNKS; t += 1) {
    if (STAGED) {
#pragma unroll
      for (int cc = 0; cc < 4; cc++)
#pragma unroll
        for (int p = 0; p < 2; p++)
          ((double2_t *)(lds + cc * CHUNK))[tid + p * WG] = st[cc][p];
    } else {
#pragma unroll
      for (int cc = 0; cc < 4; cc++)
#pragma unroll
        for (int p = 0; p < 2; p++)
          ((double2_t *)(lds + cc * CHUNK))[tid + p * WG] =
              src[cc][t * (CHUNK / 2) + tid + p * WG];
    }
    __syncthreads();
    long long tn = t + 1;
    if (STAGED && tn < nchunks) {
#pragma unroll
      for (int cc = 0; cc < 4; cc++)
#pragma unroll
        for (int p = 0; p < 2; p++)
          st[cc][p] = src[cc][tn * (CHUNK / 2) + tid + p * WG];
    }
    /* row phase: 4 passes, read every stream from LDS, cheap filter+sum */
#pragma unroll 2
    for (int k = 0; k < CHUNK / WG; k++) {
      int r = tid + k * WG;
      double ship = lds[3 * CHUNK + r];
      int alive = ship > 0.1 && ship < 0.9;
      double v = lds[0 * CHUNK + r] * lds[2 * CHUNK + r];
      acc += (alive && lds[1 * CHUNK + r] < 2.0) ? v : 0.0;
    }
    __syncthreads();
  }
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) acc += __shfl_down(acc, off, 64);
  if ((threadIdx.x & 63) == 0 && acc != 0.0) atomicAdd(out, acc);
}

static double bench(void (*launch)(int, const double *, const double *,
                                   const double *, const double *, long long,
                                   double *, hipStream_t),
                    int grid, const double *a, const double *b, const double *c,
                    const double *d, long long n, double *out) {
  hipEvent_t e0, e1;
  hipEventCreate(&e0); hipEventCreate(&e1);
  launch(grid, a, b, c, d, n, out, 0);   /* warmup */
  hipDeviceSynchronize();
  hipEventRecord(e0);
  for (int i = 0; i < 5; i++) launch(grid, a, b, c, d, n, out, 0);
  hipEventRecord(e1);
  hipEventSynchronize(e1);
  float ms;
  hipEventElapsedTime(&ms, e0, e1);
  hipEventDestroy(e0); hipEventDestroy(e1);
  return ms / 5.0;
}

/* F: grouped shape — 7 streams (40 B/row: 4 f64 + i32 + 2 i16), staged,
 * slot from the two i16 streams, 6 aggregate values, 8x7 predicated adds */
__global__ __launch_bounds__(WG, 2) void k_groupedF(
    const double *__restrict__ a, const double *__restrict__ b,
    const double *__restrict__ c, const double *__restrict__ d,
    long long n, double *out) {
  __shared__ __attribute__((aligned(16))) double lds[7 * CHUNK];
  const GAS double2_t *src[4] = {
    (const GAS double2_t *)(uintptr_t)a, (const GAS double2_t *)(uintptr_t)b,
    (const GAS double2_t *)(uintptr_t)c, (const GAS double2_t *)(uintptr_t)d };
  /* reuse stream a's tail as the i32+2xi16 streams (8B per row covers all) */
  const GAS int2_t *isrc = (const GAS int2_t *)(uintptr_t)b;
  const int tid = threadIdx.x;
  long long nchunks = n / CHUNK;
  double sums[8][6]; double rc[8];
#pragma unroll
  for (int s2 = 0; s2 < 8; s2++) { rc[s2] = 0;
#pragma unroll
    for (int a2 = 0; a2 < 6; a2++) sums[s2][a2] = 0; }
  double2_t st[4][2]; int2_t sti[2];
  long long ntilesF = nchunks / 16;
  for (long long tt = blockIdx.x; tt < ntilesF; tt += gridDim.x) {
    long long t0 = tt * 16;
#pragma unroll
    for (int cc = 0; cc < 4; cc++)
#pragma unroll
      for (int p = 0; p < 2; p++)
        st[cc][p] = src[cc][t0 * (CHUNK / 2) + tid + p * WG];
    sti[0] = isrc[t0 * (CHUNK / 2) + tid];
    sti[1] = isrc[t0 * (CHUNK / 2) + tid + WG];
    for (long long t = t0; t < t0 + 16; t++) {
#pragma unroll
      for (int cc = 0; cc < 4; cc++)
#pragma unroll
        for (int p = 0; p < 2; p++)
          ((double2_t *)(lds + cc * CHUNK))[tid + p * WG] = st[cc][p];
      /* widen the int streams into 3 more LDS images (i32 + 2 dicts) */
#pragma unroll
      for (int p = 0; p < 2; p++) {
        int2_t v = sti[p];
        double2_t y0; y0.x = (double)v.x; y0.y = (double)v.y;
        ((double2_t *)(lds + 4 * CHUNK))[tid + p * WG] = y0;
        double2_t y1; y1.x = (double)(v.x & 3); y1.y = (double)(v.y & 3);
        ((double2_t *)(lds + 5 * CHUNK))[tid + p * WG] = y1;
        double2_t y2; y2.x = (double)((v.x >> 2) & 1); y2.y = (double)((v.y >> 2) & 1);
        ((double2_t *)(lds + 6 * CHUNK))[tid + p * WG] = y2;
      }
      __syncthreads();
      long long tn = t + 1;
      if (tn < t0 + 16) {
#pragma unroll
        for (int cc = 0; cc < 4; cc++)
#pragma unroll
          for (int p = 0; p < 2; p++)
            st[cc][p] = src[cc][tn * (CHUNK / 2) + tid + p * WG];
        sti[0] = isrc[tn * (CHUNK / 2) + tid];
        sti[1] = isrc[tn * (CHUNK / 2) + tid + WG];
      }
#pragma unroll 4
      for (int k = 0; k < CHUNK / WG; k++) {
        int r = tid + k * WG;
        double ship = lds[4 * CHUNK + r];
        int ok = ship > 100.0;
        int slot = (int)lds[5 * CHUNK + r] * 2 + (int)lds[6 * CHUNK + r];
        double q = lds[0 * CHUNK + r], ep = lds[1 * CHUNK + r],
               di = lds[2 * CHUNK + r], tx = lds[3 * CHUNK + r];
        double va[6] = { q, ep, ep * (1 - di), ep * (1 - di) * (1 + tx), di, 1.0 };
#pragma unroll
        for (int s2 = 0; s2 < 8; s2++) {
          int ms = ok && slot == s2;
          rc[s2] += ms ? 1.0 : 0.0;
#pragma unroll
          for (int a2 = 0; a2 < 6; a2++) sums[s2][a2] += ms ? va[a2] : 0.0;
        }
      }
      __syncthreads();
    }
  }
  double acc = 0;
#pragma unroll
  for (int s2 = 0; s2 < 8; s2++) { acc += rc[s2];
#pragma unroll
    for (int a2 = 0; a2 < 6; a2++) acc += sums[s2][a2]; }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) acc += __shfl_down(acc, off, 64);
  if ((threadIdx.x & 63) == 0 && acc != 0.0) atomicAdd(out, acc);
}

#define LAUNCHER(name, kexpr)                                                 \
  static void name(int grid, const double *a, const double *b,                \
                   const double *c, const double *d, long long n,             \
                   double *out, hipStream_t s) {                              \
    hipLaunchKernelGGL(kexpr, dim3(grid), dim3(WG), 0, s, a, b, c, d, n, out); \
  }
LAUNCHER(la, k_streamA)
LAUNCHER(lb, (k_chunked<33, 0>))
LAUNCHER(lc, (k_chunked<33, 1>))
LAUNCHER(ld, (k_chunked<33, 0, 16>))
LAUNCHER(le, (k_chunked<33, 1, 16>))
LAUNCHER(lf, k_groupedF)

int main(int argc, char **argv) {
  long long n = 60LL * 1000 * 1000;   /* rows */
  n = (n / CHUNK) * CHUNK;
  double *a, *b, *c, *d, *out;
  hipMalloc(&a, n * 8); hipMalloc(&b, n * 8);
  hipMalloc(&c, n * 8); hipMalloc(&d, n * 8);
  hipMalloc(&out, 8);
  hipMemset(a, 0x11, n * 8); hipMemset(b, 0x12, n * 8);
  hipMemset(c, 0x13, n * 8); hipMemset(d, 0x14, n * 8);
  double bytes = 4.0 * n * 8;
  for (int grid : {1024, 2048, 4096}) {
    double msA = bench(la, grid, a, b, c, d, n, out);
    double msB = bench(lb, grid, a, b, c, d, n, out);
    double msC = bench(lc, grid, a, b, c, d, n, out);
    double msD = bench(ld, grid, a, b, c, d, n, out);
    double msE = bench(le, grid, a, b, c, d, n, out);
    double msF = bench(lf, grid, a, b, c, d, n, out);
    printf("grid=%d  A(stream)=%.0fGB/s  B(chunk-stride)=%.0fGB/s  C(staged-stride)=%.0fGB/s  "
           "D(16chunk-tile)=%.0fGB/s  E(16tile-staged)=%.0fGB/s  "
           "F(groupedQ1,40B/row)=%.0fGB/s\n", grid,
           bytes / msA / 1e6, bytes / msB / 1e6, bytes / msC / 1e6,
           bytes / msD / 1e6, bytes / msE / 1e6,
           (40.0 * n) / msF / 1e6);
  }
  return 0;
}
