// Are bucket-ordered CAS inserts faster than random ones on gfx950?
// Decides whether radix-partitioned hash builds are worth building.
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdint>

__host__ __device__ static inline uint64_t sm(uint64_t x) {
  x += 0x9e3779b97f4a7c15ull;
  x = (x ^ (x >> 30)) * 0xbf58476d1ce4e5b9ull;
  x = (x ^ (x >> 27)) * 0x94d049bb133111ebull;
  return x ^ (x >> 31);
}

// generate keys so that key i's slot is ordered (bucketed) or random
__global__ void k_gen_keys(int64_t n, uint64_t nslots, int bucketed,
                           unsigned long long* keys) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    if (bucketed) {
      // keys sorted by target slot: slot grows with i
      uint64_t slot = (uint64_t)((__uint128_t)i * nslots / n);
      keys[i] = (slot << 20) | (sm(i) & 0xFFFFF) | 1; // distinct-ish
    } else {
      keys[i] = sm(i) | 1;
    }
  }
}

__global__ void k_insert(const unsigned long long* keys, int64_t n,
                         unsigned long long* tkeys, unsigned long long* tpay,
                         uint64_t nslots, int bucketed) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    unsigned long long k = keys[i];
    uint64_t pos = bucketed ? (k >> 20) % nslots : (sm(k) & (nslots - 1));
    for (;;) {
      unsigned long long prev = atomicCAS(&tkeys[pos], 0ull, k);
      if (prev == 0 || prev == k) { tpay[pos] = k * 3; break; }
      pos = (pos + 1) & (nslots - 1);
    }
  }
}

int main() {
  const int64_t n = 14600000;          // ~Q3 build size
  const uint64_t nslots = 32ull << 20; // 32M slots = 256MB keys
  unsigned long long *keys, *tk, *tp;
  (void)hipMalloc(&keys, n * 8);
  (void)hipMalloc(&tk, nslots * 8);
  (void)hipMalloc(&tp, nslots * 8);
  for (int bucketed = 0; bucketed <= 1; bucketed++) {
    hipLaunchKernelGGL(k_gen_keys, dim3(2048), dim3(256), 0, 0, n, nslots, bucketed, keys);
    for (int rep = 0; rep < 3; rep++) {
      (void)hipMemset(tk, 0, nslots * 8);
      hipEvent_t a, b; (void)hipEventCreate(&a); (void)hipEventCreate(&b);
      (void)hipEventRecord(a, 0);
      hipLaunchKernelGGL(k_insert, dim3(2048), dim3(256), 0, 0, keys, n, tk, tp, nslots, bucketed);
      (void)hipEventRecord(b, 0);
      (void)hipEventSynchronize(b);
      float ms = 0; (void)hipEventElapsedTime(&ms, a, b);
      printf("bucketed=%d rep=%d: %.3f ms (%.1f M inserts/s)\n",
             bucketed, rep, ms, n / ms / 1000.0);
    }
  }
  return 0;
}
