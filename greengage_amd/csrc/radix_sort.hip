/*
 * LSB radix sort for (uint64 key, uint64 payload) pairs — the engine's
 * general ORDER BY operator (reference semantics: nodeSort.c:48 +
 * tuplesort.c qsort/mk-sort paths; the bounded-LIMIT case uses the
 * histogram top-k select in engine_abi.cpp instead, mirroring the
 * reference's bounded-heap switch at tuplesort.c:1360–1377).
 *
 * Classic 3-phase LSB passes, 8 bits per pass over the requested byte
 * range: per-block histogram → exclusive scan over (digit, block) →
 * rank-stable scatter.  Stability across passes gives multi-key orders
 * by sorting least-significant key first (tuplesort_mk's column-at-a-
 * time idea, tuplesort_mk.c).  Descending orders are expressed by
 * pre-inverting the key bytes (host helper in engine_abi.cpp).
 *
 * MI355X notes: 256-thread blocks; per-block LDS histograms (256 × u32)
 * privatized ×4 replicas to spread bank traffic; scatter computes each
 * lane's in-block rank from an LDS digit-count prefix (wave ballot per
 * digit would serialize 256 digits — counting via shared atomics is
 * faster at this digit width).
 */
#include <hip/hip_runtime.h>

#include "engine_internal.h"

namespace gg
{

static constexpr int RS_THREADS = 256;
static constexpr int RS_RADIX = 256;
/* items each block owns per pass (grid sized from this) */
static constexpr int RS_TILE = 4096;

__global__ __launch_bounds__(RS_THREADS)
void k_rs_hist(const unsigned long long *__restrict__ keys, int64_t n,
	       int shift, unsigned int *__restrict__ block_hist)
{
	__shared__ unsigned int h[RS_RADIX];

	for (int i = threadIdx.x; i < RS_RADIX; i += blockDim.x)
		h[i] = 0;
	__syncthreads();

	int64_t base = (int64_t) blockIdx.x * RS_TILE;
	int64_t end = min(base + (int64_t) RS_TILE, n);

	for (int64_t i = base + threadIdx.x; i < end; i += blockDim.x)
		atomicAdd(&h[(keys[i] >> shift) & 0xff], 1u);
	__syncthreads();
	/* layout: hist[digit][block] so the global exclusive scan gives
	 * stable digit-major, block-minor offsets */
	for (int d = threadIdx.x; d < RS_RADIX; d += blockDim.x)
		block_hist[(size_t) d * gridDim.x + blockIdx.x] = h[d];
}

/* single-block exclusive scan of the (digit-major) histogram table:
 * chunks of 1024 swept in order; inside a chunk, wave-level shuffle
 * scan + cross-wave LDS partials (16 waves) */
__global__ __launch_bounds__(1024)
void k_rs_scan(unsigned int *__restrict__ hist, int64_t len)
{
	__shared__ unsigned long long carry;
	__shared__ unsigned int wavesum[16];

	if (threadIdx.x == 0)
		carry = 0;
	__syncthreads();
	for (int64_t base = 0; base < len; base += blockDim.x)
	{
		int64_t i = base + threadIdx.x;
		unsigned int v = (i < len) ? hist[i] : 0;
		int lane = (int) (threadIdx.x & 63);
		int wave = (int) (threadIdx.x >> 6);

		/* wave-inclusive scan */
		unsigned int inc = v;

		for (int off = 1; off < 64; off <<= 1)
		{
			unsigned int up = __shfl_up(inc, off, 64);

			if (lane >= off)
				inc += up;
		}
		if (lane == 63)
			wavesum[wave] = inc;
		__syncthreads();
		/* wave 0 scans the 16 wave sums */
		if (threadIdx.x < 16)
		{
			unsigned int w = wavesum[threadIdx.x];
			unsigned int winc = w;

			for (int off = 1; off < 16; off <<= 1)
			{
				unsigned int up = __shfl_up(winc, off, 64);

				if ((int) threadIdx.x >= off)
					winc += up;
			}
			wavesum[threadIdx.x] = winc - w;	/* exclusive */
		}
		__syncthreads();
		{
			unsigned int excl = inc - v + wavesum[wave];

			if (i < len)
				hist[i] = (unsigned int) (carry + excl);
		}
		__syncthreads();
		if (threadIdx.x == 1023)
			carry += (unsigned long long) (inc + wavesum[wave]);
		__syncthreads();
	}
}

__global__ __launch_bounds__(RS_THREADS)
void k_rs_scatter(const unsigned long long *__restrict__ keys,
		  const unsigned long long *__restrict__ pay, int64_t n,
		  int shift, const unsigned int *__restrict__ hist,
		  unsigned long long *__restrict__ out_keys,
		  unsigned long long *__restrict__ out_pay)
{
	/* per-digit running cursor for this block (stable: items are
	 * processed in index order by a single wavefront-ordered sweep) */
	__shared__ unsigned int cursor[RS_RADIX];

	for (int i = threadIdx.x; i < RS_RADIX; i += blockDim.x)
		cursor[i] = hist[(size_t) i * gridDim.x + blockIdx.x];
	__syncthreads();

	int64_t base = (int64_t) blockIdx.x * RS_TILE;
	int64_t end = min(base + (int64_t) RS_TILE, n);

	/* one wave processes contiguous chunks in order; lanes claim
	 * stable ranks inside the chunk via prefix ballot per digit */
	for (int64_t chunk = base; chunk < end; chunk += blockDim.x)
	{
		int64_t i = chunk + threadIdx.x;
		unsigned long long k = 0, p = 0;
		int digit = -1;

		if (i < end)
		{
			k = keys[i];
			if (pay)
				p = pay[i];
			digit = (int) ((k >> shift) & 0xff);
		}
		/* wave-sequential stable ranking: waves run one at a
		 * time over the shared cursor (cheap at 4 waves).
		 * Same-digit lane mask built from 8 bit-ballots (HIP has
		 * no __match_any); lanes past the end vote with an
		 * invalid mask and never write. */
		for (int w = 0; w < (int) (blockDim.x / 64); w++)
		{
			if ((int) (threadIdx.x / 64) == w)
			{
				unsigned long long valid = __ballot(digit >= 0);
				unsigned long long same = valid;

				for (int b = 0; b < 8; b++)
				{
					unsigned long long vote =
						__ballot((digit >> b) & 1);

					same &= ((digit >> b) & 1)
						? vote : ~vote;
				}
				if (digit >= 0)
				{
					int lane = (int) (threadIdx.x & 63);
					int rank = __popcll(same &
							    ((1ull << lane) - 1));
					int leader =
						__ffsll((long long) same) - 1;
					unsigned int lbase = 0;

					if (lane == leader)
						lbase = atomicAdd(
							&cursor[digit],
							(unsigned int)
							__popcll(same));
					lbase = __shfl(lbase, leader, 64);
					{
						unsigned int dst = lbase + rank;

						out_keys[dst] = k;
						if (pay)
							out_pay[dst] = p;
					}
				}
			}
			__syncthreads();
		}
	}
}

hipError_t
launch_radix_sort_pass(hipStream_t s, const unsigned long long *keys,
		       const unsigned long long *pay, int64_t n, int shift,
		       unsigned int *block_hist, int nblocks,
		       unsigned long long *out_keys,
		       unsigned long long *out_pay)
{
	hipLaunchKernelGGL(k_rs_hist, dim3(nblocks), dim3(RS_THREADS), 0, s,
			   keys, n, shift, block_hist);
	hipLaunchKernelGGL(k_rs_scan, dim3(1), dim3(1024), 0, s, block_hist,
			   (int64_t) RS_RADIX * nblocks);
	hipLaunchKernelGGL(k_rs_scatter, dim3(nblocks), dim3(RS_THREADS), 0,
			   s, keys, pay, n, shift, block_hist, out_keys,
			   out_pay);
	return hipGetLastError();
}

int radix_sort_nblocks(int64_t n)
{
	int64_t b = (n + RS_TILE - 1) / RS_TILE;

	return (int) (b < 1 ? 1 : b);
}

}				/* namespace gg */
