// Minimal reproducer for the qty-column corruption seen in k_gen_lineitem:
// several formulations of 1 + splitmix64(...) % 50, plus the raw hash.
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdint>

__host__ __device__ static inline uint64_t sm(uint64_t x) {
  x += 0x9e3779b97f4a7c15ull;
  x = (x ^ (x >> 30)) * 0xbf58476d1ce4e5b9ull;
  x = (x ^ (x >> 27)) * 0x94d049bb133111ebull;
  return x ^ (x >> 31);
}
__host__ __device__ static inline uint64_t rnd(uint64_t seed, uint64_t tab,
                                               uint64_t row, uint64_t slot) {
  return sm(seed ^ (tab << 48) ^ (slot << 40) ^ row);
}

__global__ void k(uint64_t* out, int n) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  uint64_t u = (uint64_t)i;
  uint64_t h = rnd(42, 0x4c49ull, u, 0);
  out[i*8+0] = h;                                  // raw hash
  out[i*8+1] = (uint64_t)(int64_t)(1 + h % 50) * 100;   // original form
  out[i*8+2] = 1 + h % 50;                         // no cast/mul
  { int32_t m = (int32_t)(h % 50); out[i*8+3] = (uint64_t)(int64_t)((1+m)*100); } // restructured
  { uint64_t q = h / 50; out[i*8+4] = 1 + (h - q*50); }  // explicit div
  out[i*8+5] = h % 50ull;                          // ull literal
  out[i*8+6] = (uint64_t)(h % 121);                // the %121 used by ship
  out[i*8+7] = (uint64_t)(h % 11);                 // the %11 used by disc
}

int main() {
  const int n = 8;
  uint64_t *d, h[n*8];
  (void)hipMalloc(&d, sizeof(h));
  hipLaunchKernelGGL(k, dim3(1), dim3(64), 0, 0, d, n);
  (void)hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
  for (int i = 0; i < n; i++) {
    uint64_t hh = rnd(42, 0x4c49ull, i, 0);
    printf("row %d host: h=%016llx m50=%llu\n", i,
           (unsigned long long)hh, (unsigned long long)(hh % 50));
    printf("row %d dev : h=%016llx orig=%llu plain=%llu restr=%llu div=%llu ull=%llu m121=%llu m11=%llu\n",
           i, (unsigned long long)h[i*8+0], (unsigned long long)h[i*8+1],
           (unsigned long long)h[i*8+2], (unsigned long long)h[i*8+3],
           (unsigned long long)h[i*8+4], (unsigned long long)h[i*8+5],
           (unsigned long long)h[i*8+6], (unsigned long long)h[i*8+7]);
  }
  return 0;
}
