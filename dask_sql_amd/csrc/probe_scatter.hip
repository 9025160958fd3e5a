// probe_scatter — standalone experiment: what does a bucket-major record
// scatter cost on gfx950, and does a TWO-PASS (64x64) partition beat the
// one-pass high-fanout scatter? (The write wall behind both the groupby
// scatter and the radix-join scatter: 16-B appends to many open buckets.)
// Build: make probe_scatter; run on the GPU box.
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdint>
#include <vector>
#define BLOCK 256
#define HIP_TRY(x) do { hipError_t e = (x); if (e) { printf("ERR %s\n", hipGetErrorString(e)); return 1; } } while (0)

__device__ __forceinline__ uint64_t mix64(uint64_t x) {
  x += 0x9E3779B97F4A7C15ull; x ^= x >> 30; x *= 0xBF58476D1CE4E5B9ull;
  x ^= x >> 27; x *= 0x94D049BB133111EBull; x ^= x >> 31; return x;
}
__device__ __forceinline__ void block_range(int64_t n, int64_t& lo, int64_t& hi) {
  int64_t per = (n + gridDim.x - 1) / gridDim.x;
  lo = (int64_t)blockIdx.x * per;
  int64_t h = lo + per;
  hi = h < n ? h : n;
  if (lo > n) lo = n;
}

// hist over NB buckets, shift chooses which bits
__global__ void k_hist(const uint64_t* keys, int64_t n, int nb, int shift,
                       int64_t* hist) {
  extern __shared__ char smem[];
  uint32_t* s_hist = (uint32_t*)smem;
  for (int i = threadIdx.x; i < nb; i += BLOCK) s_hist[i] = 0;
  __syncthreads();
  int64_t lo, hi; block_range(n, lo, hi);
  for (int64_t r = lo + threadIdx.x; r < hi; r += BLOCK)
    atomicAdd(&s_hist[(int)((mix64(keys[r]) >> shift) & (nb - 1))], 1u);
  __syncthreads();
  for (int i = threadIdx.x; i < nb; i += BLOCK)
    hist[(int64_t)blockIdx.x * nb + i] = (int64_t)s_hist[i];
}

// serial-ish scan on device (test harness; nb*grid small)
__global__ void k_scan(const int64_t* hist, int grid, int nb, int64_t* bases,
                       int64_t* out /* [grid][nb] absolute */) {
  // one thread per bucket: sum column, then prefix across buckets on t0
  __shared__ int64_t tot[8192];
  for (int b = threadIdx.x; b < nb; b += 1024) {
    int64_t s = 0;
    for (int g = 0; g < grid; g++) s += hist[(int64_t)g * nb + b];
    tot[b] = s;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    int64_t run = 0;
    for (int b = 0; b < nb; b++) { bases[b] = run; run += tot[b]; }
    bases[nb] = run;
  }
  __syncthreads();
  for (int b = threadIdx.x; b < nb; b += 1024) {
    int64_t run = bases[b];
    for (int g = 0; g < grid; g++) {
      out[(int64_t)g * nb + b] = run;
      run += hist[(int64_t)g * nb + b];
    }
  }
}

// scatter 16-B records (key, payload) bucket-major
__global__ void k_scat(const uint64_t* keys, const uint64_t* pay, int64_t n,
                       int nb, int shift, const int64_t* abs_off,
                       ulonglong2* out) {
  extern __shared__ char smem[];
  int64_t* s_off = (int64_t*)smem;
  for (int i = threadIdx.x; i < nb; i += blockDim.x)
    s_off[i] = abs_off[(int64_t)blockIdx.x * nb + i];
  __syncthreads();
  int64_t lo, hi; block_range(n, lo, hi);
  for (int64_t r = lo + threadIdx.x; r < hi; r += blockDim.x) {
    uint64_t k = keys[r];
    int b = (int)((mix64(k) >> shift) & (nb - 1));
    int64_t o = (int64_t)atomicAdd((unsigned long long*)&s_off[b], 1ull);
    out[o] = ulonglong2{k, pay[r]};
  }
}

// pass2: records in (record key, payload) form, sub-partition each coarse
// bucket by lower bits; blocks are assigned per coarse bucket
__global__ void k_scat2(const ulonglong2* recs, const int64_t* cb_bases,
                        int ncoarse, int nfine, int shift2,
                        const int64_t* abs_off /* [blocks_per][ncoarse][nfine] */,
                        int blocks_per, ulonglong2* out) {
  extern __shared__ char smem[];
  int64_t* s_off = (int64_t*)smem;
  int cb = blockIdx.x / blocks_per;
  int sub = blockIdx.x % blocks_per;
  for (int i = threadIdx.x; i < nfine; i += blockDim.x)
    s_off[i] = abs_off[((int64_t)sub * ncoarse + cb) * nfine + i];
  __syncthreads();
  int64_t c0 = cb_bases[cb], c1 = cb_bases[cb + 1];
  int64_t span = c1 - c0, chunk = (span + blocks_per - 1) / blocks_per;
  int64_t q0 = c0 + (int64_t)sub * chunk;
  int64_t q1 = q0 + chunk < c1 ? q0 + chunk : c1;
  for (int64_t r = q0 + threadIdx.x; r < q1; r += blockDim.x) {
    ulonglong2 rec = recs[r];
    int f = (int)((mix64(rec.x) >> shift2) & (nfine - 1));
    int64_t o = (int64_t)atomicAdd((unsigned long long*)&s_off[f], 1ull);
    out[o] = rec;
  }
}

// hist for pass2 per (sub-block, coarse, fine)
__global__ void k_hist2(const ulonglong2* recs, const int64_t* cb_bases,
                        int ncoarse, int nfine, int shift2, int blocks_per,
                        int64_t* hist) {
  extern __shared__ char smem[];
  uint32_t* s_hist = (uint32_t*)smem;
  int cb = blockIdx.x / blocks_per;
  int sub = blockIdx.x % blocks_per;
  for (int i = threadIdx.x; i < nfine; i += blockDim.x) s_hist[i] = 0;
  __syncthreads();
  int64_t c0 = cb_bases[cb], c1 = cb_bases[cb + 1];
  int64_t span = c1 - c0, chunk = (span + blocks_per - 1) / blocks_per;
  int64_t q0 = c0 + (int64_t)sub * chunk;
  int64_t q1 = q0 + chunk < c1 ? q0 + chunk : c1;
  for (int64_t r = q0 + threadIdx.x; r < q1; r += blockDim.x)
    atomicAdd(&s_hist[(int)((mix64(recs[r].x) >> shift2) & (nfine - 1))], 1u);
  __syncthreads();
  for (int i = threadIdx.x; i < nfine; i += blockDim.x)
    hist[((int64_t)sub * ncoarse + cb) * nfine + i] = (int64_t)s_hist[i];
}

// scan for pass2 layout: global base of fine bucket (cb,f) = coarse base +
// prefix inside cb; absolute offsets per (sub, cb, f)
__global__ void k_scan2(const int64_t* hist, int blocks_per, int ncoarse,
                        int nfine, const int64_t* cb_bases, int64_t* out) {
  int cb = blockIdx.x;
  __shared__ int64_t tot[4096];
  for (int f = threadIdx.x; f < nfine; f += blockDim.x) {
    int64_t s = 0;
    for (int g = 0; g < blocks_per; g++)
      s += hist[((int64_t)g * ncoarse + cb) * nfine + f];
    tot[f] = s;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    int64_t run = cb_bases[cb];
    for (int f = 0; f < nfine; f++) { int64_t t = tot[f]; tot[f] = run; run += t; }
  }
  __syncthreads();
  for (int f = threadIdx.x; f < nfine; f += blockDim.x) {
    int64_t run = tot[f];
    for (int g = 0; g < blocks_per; g++) {
      out[((int64_t)g * ncoarse + cb) * nfine + f] = run;
      run += hist[((int64_t)g * ncoarse + cb) * nfine + f];
    }
  }
}

__global__ void k_fill(uint64_t* keys, uint64_t* pay, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * BLOCK + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * BLOCK;
  for (; i < n; i += stride) { keys[i] = mix64(i) % 10000000; pay[i] = i; }
}

static float timeit(hipEvent_t a, hipEvent_t b) {
  float ms; hipEventElapsedTime(&ms, a, b); return ms;
}

int main() {
  const int64_t N = 100'000'000;
  uint64_t *keys, *pay;
  ulonglong2 *out1, *out2;
  HIP_TRY(hipMalloc(&keys, N * 8));
  HIP_TRY(hipMalloc(&pay, N * 8));
  HIP_TRY(hipMalloc(&out1, N * 16));
  HIP_TRY(hipMalloc(&out2, N * 16));
  int grid = 2048;
  hipLaunchKernelGGL(k_fill, dim3(grid), dim3(BLOCK), 0, 0, keys, pay, N);
  int64_t *hist, *bases, *abs_off;
  HIP_TRY(hipMalloc(&hist, (int64_t)grid * 8192 * 8));
  HIP_TRY(hipMalloc(&bases, 8200 * 8));
  HIP_TRY(hipMalloc(&abs_off, (int64_t)grid * 8192 * 8));
  hipEvent_t ev[8];
  for (auto& e : ev) hipEventCreate(&e);

  // ---- experiment A: one-pass scatter at nb = 256..4096
  for (int nb : {256, 1024, 4096}) {
    hipEventRecord(ev[0]);
    hipLaunchKernelGGL(k_hist, dim3(grid), dim3(BLOCK), nb * 4, 0, keys, N,
                       nb, 0, hist);
    hipEventRecord(ev[1]);
    hipLaunchKernelGGL(k_scan, dim3(1), dim3(1024), 0, 0, hist, grid, nb,
                       bases, abs_off);
    hipEventRecord(ev[2]);
    hipLaunchKernelGGL(k_scat, dim3(grid), dim3(1024), nb * 8, 0, keys, pay,
                       N, nb, 0, abs_off, out1);
    hipEventRecord(ev[3]);
    HIP_TRY(hipDeviceSynchronize());
    printf("1-pass nb=%4d: hist %.3f scan %.3f scat %.3f ms (scat %.2f GB/s eff %.1f)\n",
           nb, timeit(ev[0], ev[1]), timeit(ev[1], ev[2]), timeit(ev[2], ev[3]),
           (N * 24.0 / 1e9) / (timeit(ev[2], ev[3]) / 1e3),
           (N * 24.0 / 1e9) / (timeit(ev[2], ev[3]) / 1e3));
  }

  // ---- experiment B: two-pass 64 x 64 = 4096
  {
    int nc = 64, nf = 64, bp = 32;  // blocks per coarse bucket in pass 2
    hipEventRecord(ev[0]);
    hipLaunchKernelGGL(k_hist, dim3(grid), dim3(BLOCK), nc * 4, 0, keys, N,
                       nc, 6, hist);  // coarse = bits 6..11
    hipLaunchKernelGGL(k_scan, dim3(1), dim3(1024), 0, 0, hist, grid, nc,
                       bases, abs_off);
    hipEventRecord(ev[1]);
    hipLaunchKernelGGL(k_scat, dim3(grid), dim3(1024), nc * 8, 0, keys, pay,
                       N, nc, 6, abs_off, out1);
    hipEventRecord(ev[2]);
    // pass 2
    hipLaunchKernelGGL(k_hist2, dim3(nc * bp), dim3(1024), nf * 4, 0, out1,
                       bases, nc, nf, 0, bp, hist);
    hipLaunchKernelGGL(k_scan2, dim3(nc), dim3(1024), 0, 0, hist, bp, nc,
                       nf, bases, abs_off);
    hipEventRecord(ev[3]);
    hipLaunchKernelGGL(k_scat2, dim3(nc * bp), dim3(1024), nf * 8, 0, out1,
                       bases, nc, nf, 0, abs_off, bp, out2);
    hipEventRecord(ev[4]);
    HIP_TRY(hipDeviceSynchronize());
    float t1 = timeit(ev[1], ev[2]), t2 = timeit(ev[3], ev[4]);
    printf("2-pass 64x64: hist1+scan %.3f scat1 %.3f hist2+scan2 %.3f scat2 %.3f ms | total %.3f\n",
           timeit(ev[0], ev[1]), t1, timeit(ev[2], ev[3]), t2,
           timeit(ev[0], ev[4]));
    printf("  scat1 %.2f GB/s  scat2 %.2f GB/s (32 B streams each)\n",
           (N * 24.0 / 1e9) / (t1 / 1e3), (N * 32.0 / 1e9) / (t2 / 1e3));
  }
  return 0;
}
