// probe_groupby — standalone microbench isolating the k_groupby_global cost
// components on gfx950: streaming scan, hash-table probe loads, atomic RMW
// throughput to a table-sized working set. Informs the aggregate-kernel
// design. Build: make probe (in this dir); run on the GPU box.
#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstdio>
#include <vector>

#define BLOCK 256
#define GRID 2048
#define CHECK(x) do { hipError_t e=(x); if(e){printf("ERR %s\n", hipGetErrorString(e)); return 1;} } while(0)

__device__ __forceinline__ uint64_t mix64(uint64_t x) {
  x += 0x9E3779B97F4A7C15ull; x ^= x >> 30; x *= 0xBF58476D1CE4E5B9ull;
  x ^= x >> 27; x *= 0x94D049BB133111EBull; x ^= x >> 31; return x;
}

__global__ void k_stream(const int64_t* key, const double* val, int64_t n,
                         double* sink) {
  int64_t i = (int64_t)blockIdx.x * BLOCK + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * BLOCK;
  double s = 0; int64_t k = 0;
  for (; i < n; i += stride) { k += key[i]; s += val[i]; }
  if (s == 12345.678 && k == 42) *sink = s;  // keep live
}

// one f64 atomic per row to a mixed slot
__global__ void k_atomic1(const int64_t* key, const double* val, int64_t n,
                          double* table, int64_t mask) {
  int64_t i = (int64_t)blockIdx.x * BLOCK + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * BLOCK;
  for (; i < n; i += stride) {
    int64_t s = (int64_t)(mix64((uint64_t)key[i]) & mask);
    unsafeAtomicAdd(&table[s], val[i]);
  }
}

// two atomics per row (sum + count) — the C2 inner loop's RMW load
__global__ void k_atomic2(const int64_t* key, const double* val, int64_t n,
                          double* table, unsigned long long* cnt,
                          int64_t mask) {
  int64_t i = (int64_t)blockIdx.x * BLOCK + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * BLOCK;
  for (; i < n; i += stride) {
    int64_t s = (int64_t)(mix64((uint64_t)key[i]) & mask);
    unsafeAtomicAdd(&table[s], val[i]);
    atomicAdd(&cnt[s], 1ull);
  }
}

// full emulation: CAS-claim probe + 2 atomics (what k_groupby_global does)
__global__ void k_probe2(const int64_t* key, const double* val, int64_t n,
                         uint64_t* tkeys, double* table,
                         unsigned long long* cnt, int64_t mask) {
  int64_t i = (int64_t)blockIdx.x * BLOCK + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * BLOCK;
  for (; i < n; i += stride) {
    if (val[i] >= 0.5) continue;
    uint64_t c = (uint64_t)key[i];
    int64_t s = (int64_t)(mix64(c) & mask);
    while (true) {
      uint64_t k = tkeys[s];
      if (k == c) break;
      if (k == ~0ull) {
        unsigned long long old = atomicCAS((unsigned long long*)&tkeys[s],
                                           ~0ull, (unsigned long long)c);
        if (old == ~0ull || old == c) break;
      }
      s = (s + 1) & mask;
    }
    unsafeAtomicAdd(&table[s], val[i]);
    atomicAdd(&cnt[s], 1ull);
  }
}

// probe-only (claimed table): isolates the random-load component
__global__ void k_probeonly(const int64_t* key, const double* val, int64_t n,
                            const uint64_t* tkeys, int64_t mask,
                            unsigned long long* sink) {
  int64_t i = (int64_t)blockIdx.x * BLOCK + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * BLOCK;
  unsigned long long acc = 0;
  for (; i < n; i += stride) {
    uint64_t c = (uint64_t)key[i];
    int64_t s = (int64_t)(mix64(c) & mask);
    while (true) {
      uint64_t k = tkeys[s];
      if (k == c || k == ~0ull) break;
      s = (s + 1) & mask;
    }
    acc += s;
  }
  if (acc == 1) *sink = acc;
}

// per-wave sorted-segment combine: ballot-match duplicate keys in the wave,
// leader does one atomic pair (value summed via DPP-less shfl loop)
__global__ void k_atomic2_wavecomb(const int64_t* key, const double* val,
                                   int64_t n, double* table,
                                   unsigned long long* cnt, int64_t mask) {
  int64_t i = (int64_t)blockIdx.x * BLOCK + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * BLOCK;
  for (; i < n; i += stride) {
    int64_t s = (int64_t)(mix64((uint64_t)key[i]) & mask);
    double v = val[i];
    // match_any emulation over 64 lanes
    uint64_t peers = 0;
    {
      uint64_t active = __ballot(true);
      uint64_t m = active;
      peers = 0;
      for (;;) {
        int lead = __ffsll((unsigned long long)m) - 1;
        if (lead < 0) break;
        int64_t ls = __shfl(s, lead, 64);
        uint64_t same = __ballot(ls == s) & m;
        if (ls == s) { peers = same; break; }
        m &= ~same;
      }
    }
    int lane = threadIdx.x & 63;
    int leader = __ffsll((unsigned long long)peers) - 1;
    // sum within peer group via shfl reduction over the group mask
    double sum = v;
    uint64_t rest = peers & ~(1ull << lane);
    while (rest) {
      int src = __ffsll((unsigned long long)rest) - 1;
      double other = __shfl(v, src, 64);
      if (lane == leader) sum += other;
      rest &= rest - 1;
    }
    if (lane == leader) {
      unsafeAtomicAdd(&table[s], sum);
      atomicAdd(&cnt[s], (unsigned long long)__popcll(peers));
    }
  }
}

static float timeit(const char* name, void (*launch)(), int iters = 5) {
  hipEvent_t a, b;
  hipEventCreate(&a); hipEventCreate(&b);
  launch();  // warm
  hipEventRecord(a);
  for (int i = 0; i < iters; i++) launch();
  hipEventRecord(b);
  hipEventSynchronize(b);
  float ms; hipEventElapsedTime(&ms, a, b);
  printf("%-22s %8.3f ms\n", name, ms / iters);
  hipEventDestroy(a); hipEventDestroy(b);
  return ms / iters;
}

int64_t N = 100'000'000;
int64_t G = 1'000'000;
int64_t SLOTS = 2'097'152;
int64_t *d_key; double *d_val; uint64_t *d_tkeys; double *d_table;
unsigned long long *d_cnt, *d_sink;

int main(int argc, char** argv) {
  if (argc > 1) G = atoll(argv[1]);
  while (SLOTS < 2 * G) SLOTS <<= 1;
  if (SLOTS > 2 * G && SLOTS / 2 >= 2 * G) SLOTS = SLOTS;  // pow2 ≥ 2G
  printf("N=%lld G=%lld SLOTS=%lld\n", (long long)N, (long long)G,
         (long long)SLOTS);
  std::vector<int64_t> key(N);
  std::vector<double> val(N);
  uint64_t st = 7;
  for (int64_t i = 0; i < N; i++) {
    st = st * 6364136223846793005ull + 1442695040888963407ull;
    key[i] = (int64_t)((st >> 16) % (uint64_t)G);
    val[i] = (double)((st >> 11) & 0xFFFFF) / (double)0x100000;
  }
  CHECK(hipMalloc(&d_key, N * 8)); CHECK(hipMalloc(&d_val, N * 8));
  CHECK(hipMalloc(&d_tkeys, SLOTS * 8)); CHECK(hipMalloc(&d_table, SLOTS * 8));
  CHECK(hipMalloc(&d_cnt, SLOTS * 8)); CHECK(hipMalloc(&d_sink, 8));
  CHECK(hipMemcpy(d_key, key.data(), N * 8, hipMemcpyHostToDevice));
  CHECK(hipMemcpy(d_val, val.data(), N * 8, hipMemcpyHostToDevice));
  CHECK(hipMemset(d_table, 0, SLOTS * 8));
  CHECK(hipMemset(d_cnt, 0, SLOTS * 8));

  float t;
  t = timeit("stream 16B/row", []() {
    hipLaunchKernelGGL(k_stream, dim3(GRID), dim3(BLOCK), 0, 0, d_key, d_val,
                       N, (double*)d_sink); });
  printf("  -> %.2f TB/s\n", 16.0 * N / t / 1e9);
  timeit("atomic1 f64", []() {
    hipLaunchKernelGGL(k_atomic1, dim3(GRID), dim3(BLOCK), 0, 0, d_key, d_val,
                       N, d_table, SLOTS - 1); });
  timeit("atomic2 f64+u64", []() {
    hipLaunchKernelGGL(k_atomic2, dim3(GRID), dim3(BLOCK), 0, 0, d_key, d_val,
                       N, d_table, d_cnt, SLOTS - 1); });
  timeit("atomic2 wave-combine", []() {
    hipLaunchKernelGGL(k_atomic2_wavecomb, dim3(GRID), dim3(BLOCK), 0, 0,
                       d_key, d_val, N, d_table, d_cnt, SLOTS - 1); });
  CHECK(hipMemset(d_tkeys, 0xFF, SLOTS * 8));
  timeit("probe2 (full emul)", []() {
    hipLaunchKernelGGL(k_probe2, dim3(GRID), dim3(BLOCK), 0, 0, d_key, d_val,
                       N, d_tkeys, d_table, d_cnt, SLOTS - 1); });
  timeit("probe-only (warm)", []() {
    hipLaunchKernelGGL(k_probeonly, dim3(GRID), dim3(BLOCK), 0, 0, d_key,
                       d_val, N, d_tkeys, SLOTS - 1, d_sink); });
  CHECK(hipDeviceSynchronize());
  return 0;
}
