/*
 * Dense-orderkey fast path for the Q3/Q5 orders side: when o_orderkey
 * is (near-)dense — the guarded max(key) <= 8x rows check, hash-table
 * fallback otherwise — the orders hash table collapses to direct-map
 * arrays indexed by orderkey:
 *
 *   pay[okey]  u64: (orderdate u32 | shippriority << 32) for Q3,
 *                   c_nationkey for Q5; GG_PAY_ABSENT = no such order
 *   rev[okey]  u64: scale-4 revenue accumulator (group slot; group key
 *                   ≡ join key, execHHashagg.c:456 semantics)
 *
 * Build becomes plain stores (no CAS); the probe keeps the blocked
 * Bloom filter in front (rejects ~97% of lineitem rows before touching
 * the GB-sized arrays) and replaces the linear-probe walk with one
 * load.  Join/aggregate results are identical to the hash path by
 * construction (same row sets, same integer sums).
 *
 * GG_PAY_ABSENT (all-ones) cannot collide with a real payload: Q3
 * payload dates pass o_orderdate < 1995-03-15 (DateADT ≈ -1753) so the
 * low word is never 0xFFFFFFFF (= date -1 = 1999-12-31); Q5 payloads
 * are nationkeys 0..24.
 */
#include <hip/hip_runtime.h>

#include "../../include/gg_pg_hash.h"
#include "../../include/gg_checksum.h"
#include <cstdlib>

#include "engine_internal.h"

namespace gg
{

static constexpr int DN_THREADS = 256;
static constexpr int DN_MAX_BLOCKS = 2048;

static inline int dn_grid(int64_t n)
{
	int64_t b = (n + DN_THREADS - 1) / DN_THREADS;

	return (int) (b < 1 ? 1 : (b > DN_MAX_BLOCKS ? DN_MAX_BLOCKS : b));
}

/* grid for XCD-chunked kernels: multiple of 8 so each XCD gets the
 * same number of blocks */
__attribute__((unused)) static inline int dn_grid8(int64_t n)
{
	int g = dn_grid(n);

	return (g + 7) & ~7;
}

/* Contiguous per-XCD chunking (microarch guide: block b runs on XCD
 * b % 8, each XCD has a private 4 MiB L2).  Both lineitem and the
 * dense orderkey-indexed arrays are orderkey-ordered, so giving XCD x
 * the contiguous row range [x*chunk, (x+1)*chunk) makes its slice of
 * the membership bitmap (2.3 MB at SF100) L2-resident and keeps the
 * rev[] atomics XCD-local — a grid-stride loop instead drags the whole
 * 18.75 MB bitmap through every XCD's L2. */
struct DnChunk
{
	int64_t base;		/* first index of this block's lane 0 */
	int64_t hi;		/* this XCD chunk's end (exclusive) */
	int64_t stride;		/* per-iteration advance */
};

__device__ inline DnChunk dn_chunk(int64_t n)
{
	const int xcd = blockIdx.x & 7;
	const int lb = blockIdx.x >> 3;
	const int nlb = gridDim.x >> 3;	/* grid is a multiple of 8 */
	const int64_t chunk = (n + 7) >> 3;
	const int64_t lo = (int64_t) xcd * chunk;
	DnChunk c;

	c.hi = lo + chunk < n ? lo + chunk : n;
	c.base = lo + (int64_t) lb * blockDim.x;
	c.stride = (int64_t) nlb * blockDim.x;
	return c;
}

/* Quad variant: each lane owns 4 CONSECUTIVE rows so the column
 * streams are read with 16-byte vector loads (the microarch guide
 * measures 8-B NT accesses at only 0.54–0.70x the 16-B rate; a 4-B
 * int32 stream is worse still).  Chunk starts are 4-row aligned so
 * longlong2/int4 loads stay 16-B aligned. */
__device__ inline DnChunk dn_chunk4(int64_t n)
{
	const int xcd = blockIdx.x & 7;
	const int lb = blockIdx.x >> 3;
	const int nlb = gridDim.x >> 3;
	const int64_t chunk = (((n + 7) >> 3) + 3) & ~(int64_t) 3;
	const int64_t lo = (int64_t) xcd * chunk;
	DnChunk c;

	c.hi = lo + chunk < n ? lo + chunk : n;
	if (c.hi < lo)
		c.hi = lo;
	c.base = lo + (int64_t) lb * blockDim.x * 4;
	c.stride = (int64_t) nlb * blockDim.x * 4;
	return c;
}

/* clang ext_vector types (HIP_vector_type is a class the nontemporal
 * builtin rejects) */
typedef long long gg_ll2 __attribute__((ext_vector_type(2)));
typedef int gg_i4 __attribute__((ext_vector_type(4)));
typedef unsigned long long gg_ull2 __attribute__((ext_vector_type(2)));

__device__ inline gg_ll2 dn_ld2(const int64_t *p)
{
	return __builtin_nontemporal_load((const gg_ll2 *) p);
}

__device__ inline gg_i4 dn_ld4x32(const int32_t *p)
{
	return __builtin_nontemporal_load((const gg_i4 *) p);
}

__device__ inline int64_t dn_ld64(const int64_t *p)
{
	return __builtin_nontemporal_load(p);
}
__device__ inline int32_t dn_ld32(const int32_t *p)
{
	return __builtin_nontemporal_load(p);
}

__device__ inline void
dn_bloom_insert(unsigned long long *bloom, uint64_t words, int64_t key)
{
	uint32_t h = gg_hashint8(key);
	uint32_t h2 = h * 0x9E3779B1u;
	uint64_t w = (uint64_t) (h >> 6) & (words - 1);

	atomicOr(&bloom[w], (1ull << (h2 & 63)) | (1ull << ((h2 >> 6) & 63)));
}

/* one returning atomicAdd per wave (see kernels.hip wave_append) */
__device__ inline unsigned long long
dn_wave_append(unsigned long long *ctr, bool take)
{
	unsigned long long mask = __ballot(take);
	int lane = (int) (threadIdx.x & 63);
	unsigned long long base = 0;

	if (mask)
	{
		int leader = __ffsll((long long) mask) - 1;

		if (lane == leader)
			base = atomicAdd(ctr,
					 (unsigned long long) __popcll(mask));
		base = __shfl(base, leader, 64);
	}
	return base + (unsigned long long) __popcll(mask &
						    ((1ull << lane) - 1));
}

__device__ inline bool
dn_bloom_maybe(const unsigned long long *__restrict__ bloom, uint64_t words,
	       int64_t key)
{
	uint32_t h = gg_hashint8(key);
	uint32_t h2 = h * 0x9E3779B1u;
	uint64_t w = (uint64_t) (h >> 6) & (words - 1);
	unsigned long long m = (1ull << (h2 & 63)) |
		(1ull << ((h2 >> 6) & 63));

	return (bloom[w] & m) == m;
}

/* Exact 1-bit-per-key membership over DENSE orderkeys: 150M keys =
 * 18.75 MB, and lineitem is orderkey-ordered so probe reads stream
 * sequentially — unlike a hashed Bloom of the same size, whose random
 * reads thrash the XCD-local L2s.  No false positives either. */
__device__ inline void
dn_bit_set(unsigned long long *bm, int64_t key)
{
	atomicOr(&bm[key >> 6], 1ull << (key & 63));
}

/* Wave-aggregated bitmap OR: with near-sequential keys a whole wave
 * lands in 1–2 bitmap words, so per-lane atomics serialize.  Combine
 * per distinct word with ballot + an or-reduction, then one atomic by
 * the leader.  Must be executed by ALL lanes of the wave (pass
 * active=false for lanes with nothing to set). */
__device__ inline void
dn_bit_set_wave_mask(unsigned long long *bm, unsigned long long w,
		     unsigned long long mask, bool active)
{
	unsigned long long alive = __ballot(active);
	int lane = (int) (threadIdx.x & 63);

	if (!active)
		w = ~0ull;
	while (alive)
	{
		int leader = __ffsll((long long) alive) - 1;
		unsigned long long lw = __shfl(w, leader, 64);
		bool same = active && (w == lw);
		unsigned long long v = same ? mask : 0;

		for (int off = 32; off; off >>= 1)
			v |= __shfl_xor(v, off, 64);
		if (lane == leader)
			atomicOr(&bm[lw], v);
		alive &= ~__ballot(same);
	}
}

__device__ inline void
dn_bit_set_wave(unsigned long long *bm, int64_t key, bool active)
{
	dn_bit_set_wave_mask(bm, (unsigned long long) (key >> 6),
			     active ? (1ull << (key & 63)) : 0, active);
}

__device__ inline bool
dn_bit_test(const unsigned long long *__restrict__ bm, int64_t key)
{
	return (bm[key >> 6] >> (key & 63)) & 1;
}


/* One global atomic per BLOCK instead of per wave: wave leaders stage
 * partials in LDS, thread 0 flushes.  Per-wave flushes to a single hot
 * word serialize at ~88 atomics/µs (the dequeue wall): 2048 blocks x 4
 * waves x 5 stat words measured ~460 µs of pure epilogue in the stats
 * kernel. */
__device__ inline void
dn_block_add(unsigned long long *lds4, unsigned long long v,
	     unsigned long long *target)
{
	for (int off = 32; off; off >>= 1)
		v += __shfl_down(v, off, 64);
	if ((threadIdx.x & 63) == 0)
		lds4[threadIdx.x >> 6] = v;
	__syncthreads();
	if (threadIdx.x == 0)
	{
		unsigned long long s = 0;

		for (int i = 0; i < (int) (blockDim.x >> 6); i++)
			s += lds4[i];
		if (s)
			atomicAdd(target, s);
	}
	__syncthreads();
}

/* Q3 orders build: date filter + customer membership → pay store.
 * Round-2 sweep note: both a 4-way strided unroll and a quad 16-B
 * vector-load layout measured SLOWER than this simple stride loop
 * (1.44 ms vs 1.30 ms — the wave-aggregated bitmap OR dominates and
 * extra unroll state just raises register pressure), so round 1's
 * layout stands (profiles/r02c sweep). */
__global__ __launch_bounds__(DN_THREADS, 8)
void k_dn_build_orders(const int64_t *__restrict__ okey,
		  const int64_t *__restrict__ ckey,
		  const int32_t *__restrict__ odate,
		  const int32_t *__restrict__ prio, int64_t n, int32_t cutoff,
		  const unsigned long long *__restrict__ cust_keys,
		  uint64_t cust_slots,
		  const unsigned long long *__restrict__ cust_bloom,
		  uint64_t cust_bwords,
		  const unsigned long long *__restrict__ cust_bits,
		  int64_t cust_dlen,
		  unsigned long long *__restrict__ pr, int64_t dense_len,
		  unsigned long long *__restrict__ bloom, uint64_t bwords,
		  unsigned long long *match_count)
{
	const int64_t stride = (int64_t) gridDim.x * blockDim.x;
	unsigned long long matches = 0;
	/* round the per-lane range up so every lane of a wave executes
	 * the same number of iterations (the wave-aggregated bitmap OR
	 * below needs a convergent wave) */
	const int64_t i0 = (int64_t) blockIdx.x * blockDim.x + threadIdx.x;
	const int64_t n_up = ((n + stride - 1) / stride) * stride;

	for (int64_t i = i0; i < n_up; i += stride)
	{
		bool ok = i < n;
		int32_t d = 0;
		int64_t ck = 0, k = -1;

		if (ok)
		{
			d = dn_ld32(&odate[i]);
			ck = dn_ld64(&ckey[i]);
			k = dn_ld64(&okey[i]);
			ok = d < cutoff;
		}
		if (ok)
		{
			if (cust_bits)
				ok = (ck >= 0 && ck < cust_dlen &&
				      ((cust_bits[ck >> 6] >> (ck & 63)) & 1));
			else
			{
				uint32_t h = gg_hashint8(ck);

				if (cust_bloom)
				{
					uint32_t h2 = h * 0x9E3779B1u;
					uint64_t w = (uint64_t) (h >> 6) &
						(cust_bwords - 1);
					unsigned long long m =
						(1ull << (h2 & 63)) |
						(1ull << ((h2 >> 6) & 63));

					ok = (cust_bloom[w] & m) == m;
				}
				if (ok)
				{
					uint64_t pos = (uint64_t) h &
						(cust_slots - 1);

					for (;;)
					{
						unsigned long long v =
							cust_keys[pos];

						if (v ==
						    (unsigned long long) ck)
							break;
						if (v == 0)
						{
							ok = false;
							break;
						}
						pos = (pos + 1) &
							(cust_slots - 1);
					}
				}
			}
		}
		ok = ok && k >= 0 && k < dense_len;
		if (ok)
		{
			matches++;
			/* NT store: skip the read-for-ownership on lines we
			 * only partially fill (~39% of keys pass the date
			 * filter) */
			__builtin_nontemporal_store(
				(unsigned long long) (uint32_t) d |
				((unsigned long long) (uint32_t) prio[i]
				 << 32), &pr[2 * k + 1]);
		}
		dn_bit_set_wave(bloom, k, ok);
	}
	{
		__shared__ unsigned long long lsum[4];

		dn_block_add(lsum, matches, match_count);
	}
}

hipError_t
launch_dn_build_orders(hipStream_t s, const int64_t *okey,
		       const int64_t *ckey, const int32_t *odate,
		       const int32_t *prio, int64_t n, int32_t cutoff,
		       DeviceHashTable cust,
		       const unsigned long long *cust_bits,
		       int64_t cust_dlen, unsigned long long *pr,
		       int64_t dense_len, unsigned long long *bloom,
		       uint64_t bwords, unsigned long long *match_count)
{
	hipLaunchKernelGGL(k_dn_build_orders, dim3(dn_grid(n)),
			   dim3(DN_THREADS), 0, s, okey, ckey, odate, prio, n,
			   cutoff, cust.keys, cust.nslots, cust.bloom,
			   cust.bloom_words, cust_bits, cust_dlen, pr,
			   dense_len, bloom, bwords, match_count);
	return hipGetLastError();
}

/* exchange-path insert of received (okey, pay) rows */
__global__ void
k_dn_insert_orders(const int64_t *__restrict__ okey,
		   const int64_t *__restrict__ rowpay, int64_t n,
		   unsigned long long *__restrict__ pr, int64_t dense_len,
		   unsigned long long *__restrict__ bloom, uint64_t bwords)
{
	const int64_t stride = (int64_t) gridDim.x * blockDim.x;

	for (int64_t i = (int64_t) blockIdx.x * blockDim.x + threadIdx.x;
	     i < n; i += stride)
	{
		int64_t k = okey[i];

		if (k < 0 || k >= dense_len)
			continue;
		pr[2 * k + 1] = (unsigned long long) rowpay[i];
		dn_bit_set(bloom, k);
	}
}

hipError_t
launch_dn_insert_orders(hipStream_t s, const int64_t *okey,
			const int64_t *rowpay, int64_t n,
			unsigned long long *pr, int64_t dense_len,
			unsigned long long *bloom, uint64_t bwords)
{
	hipLaunchKernelGGL(k_dn_insert_orders, dim3(dn_grid(n)),
			   dim3(DN_THREADS), 0, s, okey, rowpay, n, pr,
			   dense_len, bloom, bwords);
	return hipGetLastError();
}

/* Q3 lineitem probe against the dense orders map.  Round-1 PMC showed
 * this kernel at ~3.5 TB/s effective with only ~12 B/row of sequential
 * traffic (okey+shipdate): the limiter was memory-level parallelism,
 * not occupancy.  Two measured layouts, both XCD-chunked so each XCD's
 * bitmap/rev slice stays in its private L2: QUAD = 4 consecutive rows
 * per lane via 16-B vector NT loads; strided = 4-way unrolled strided
 * streams.  GG_Q3_PROBE_VAR selects (sweep evidence in profiles/). */
template <int QUAD, int WAVES>
__global__ __launch_bounds__(DN_THREADS, WAVES)
void k_dn_probe_lineitem_t(const int64_t *__restrict__ okey,
			   const int32_t *__restrict__ shipdate,
			   const int64_t *__restrict__ price,
			   const int64_t *__restrict__ disc, int64_t n,
			   int32_t cutoff,
			   unsigned long long *__restrict__ pr,
			   int64_t dense_len,
			   const unsigned long long *__restrict__ bloom,
			   uint64_t bwords, unsigned long long *join_rows,
			   unsigned long long *__restrict__ surv,
			   uint64_t region,
			   unsigned long long *__restrict__ counts,
			   unsigned long long *ovf)
{
	/* per-block survivor region: group creators append via an LDS
	 * counter (a single global counter saturates at ~88 returning
	 * atomics/µs — measured 13.7 ms probe when tried) */
	__shared__ unsigned long long lsurv;

	if (threadIdx.x == 0)
		lsurv = 0;
	__syncthreads();

	unsigned long long joined = 0;
	unsigned long long *mine = surv + (uint64_t) blockIdx.x * region;

	auto body = [&](int64_t i, int32_t sd, int64_t k)
	{
		if (sd <= cutoff)	/* qual: l_shipdate > cutoff */
			return;
		if (k < 0 || k >= dense_len)
			return;
		if (!dn_bit_test(bloom, k))	/* exact: no false positives */
			return;
		joined++;
		{
			unsigned long long r4 = (unsigned long long)
				(price[i] * (100 - disc[i]));
			unsigned long long old = atomicAdd(&pr[2 * k], r4);

			/* group creation (execHHashagg find-or-create):
			 * exactly one adder sees the 0→nonzero transition,
			 * so the survivor regions get each group once — the
			 * top-k then gathers ~ngroups entries instead of
			 * sweeping the GB-sized dense array (zero-sum
			 * groups stay invisible, as rev==0 always meant) */
			if (old == 0 && r4 != 0)
			{
				unsigned long long at =
					atomicAdd(&lsurv, 1ull);

				if (at < region)
					mine[at] = (unsigned long long) k;
			}
		}
	};

	if (QUAD)
	{
		DnChunk c = dn_chunk4(n);
		const int64_t S = c.stride;

		for (int64_t r = c.base + (int64_t) threadIdx.x * 4;
		     r < c.hi; r += S)
		{
			if (r + 3 < c.hi)
			{
				gg_i4 sd = dn_ld4x32(&shipdate[r]);
				gg_ll2 k01 = dn_ld2(&okey[r]);
				gg_ll2 k23 = dn_ld2(&okey[r + 2]);

				body(r, sd.x, k01.x);
				body(r + 1, sd.y, k01.y);
				body(r + 2, sd.z, k23.x);
				body(r + 3, sd.w, k23.y);
			}
			else
				for (int64_t j = r; j < c.hi; j++)
					body(j, dn_ld32(&shipdate[j]),
					     dn_ld64(&okey[j]));
		}
	}
	else if (QUAD == 2)
	{
		/* 8-way strided unroll: 16 loads in flight per lane */
		DnChunk c = dn_chunk(n);
		const int64_t S = c.stride;
		int64_t i = c.base + threadIdx.x;

		for (; i + 7 * S < c.hi; i += 8 * S)
		{
			int32_t sd[8];
			int64_t k[8];

			for (int u = 0; u < 8; u++)
			{
				sd[u] = dn_ld32(&shipdate[i + u * S]);
				k[u] = dn_ld64(&okey[i + u * S]);
			}
			for (int u = 0; u < 8; u++)
				body(i + u * S, sd[u], k[u]);
		}
		for (; i < c.hi; i += S)
			body(i, dn_ld32(&shipdate[i]), dn_ld64(&okey[i]));
	}
	else
	{
		DnChunk c = dn_chunk(n);
		const int64_t S = c.stride;
		int64_t i = c.base + threadIdx.x;

		for (; i + 3 * S < c.hi; i += 4 * S)
		{
			int32_t sd0 = dn_ld32(&shipdate[i]);
			int32_t sd1 = dn_ld32(&shipdate[i + S]);
			int32_t sd2 = dn_ld32(&shipdate[i + 2 * S]);
			int32_t sd3 = dn_ld32(&shipdate[i + 3 * S]);
			int64_t k0 = dn_ld64(&okey[i]);
			int64_t k1 = dn_ld64(&okey[i + S]);
			int64_t k2 = dn_ld64(&okey[i + 2 * S]);
			int64_t k3 = dn_ld64(&okey[i + 3 * S]);

			body(i, sd0, k0);
			body(i + S, sd1, k1);
			body(i + 2 * S, sd2, k2);
			body(i + 3 * S, sd3, k3);
		}
		for (; i < c.hi; i += S)
			body(i, dn_ld32(&shipdate[i]), dn_ld64(&okey[i]));
	}
	{
		__shared__ unsigned long long lsum[4];

		dn_block_add(lsum, joined, join_rows);
	}
	__syncthreads();
	if (threadIdx.x == 0)
	{
		counts[blockIdx.x] = lsurv < region ? lsurv : region;
		if (lsurv > region)
			atomicOr(ovf, 1ull);
	}
}

/* sweep-selected default: strided streams at 8 blocks/CU (see
 * profiles/r02 probe sweep) */
static int dn_probe_var(const char *env, int dflt)
{
	const char *v = getenv(env);

	return (v && *v) ? atoi(v) : dflt;
}

static int
dn_grid_env(int64_t n, const char *env)
{
	const char *v = getenv(env);

	if (v && *v)
	{
		int g = atoi(v);

		if (g >= 1 && g <= 65535)
		{
			int64_t need = (n + DN_THREADS - 1) / DN_THREADS;

			return (int) (need < g ? (need < 1 ? 1 : need) : g);
		}
	}
	return dn_grid(n);
}

/* host-visible probe grid (exec_q3 sizes the survivor regions by it) */
int dn_probe_grid(int64_t n)
{
	return (dn_grid_env(n, "GG_Q3_PROBE_GRID") + 7) & ~7;
}

hipError_t
launch_dn_probe_lineitem(hipStream_t s, const int64_t *okey,
			 const int32_t *shipdate, const int64_t *price,
			 const int64_t *disc, int64_t n, int32_t cutoff,
			 unsigned long long *pr,
			 int64_t dense_len, unsigned long long *bloom,
			 uint64_t bwords, unsigned long long *join_rows,
			 unsigned long long *surv, uint64_t region,
			 unsigned long long *counts, unsigned long long *ovf,
			 int grid)
{
	dim3 gg(grid), bb(DN_THREADS);

	switch (dn_probe_var("GG_Q3_PROBE_VAR", 0))
	{
		default:
		case 0:
			hipLaunchKernelGGL((k_dn_probe_lineitem_t<0, 8>), gg,
					   bb, 0, s, okey, shipdate, price,
					   disc, n, cutoff, pr,
					   dense_len, bloom, bwords,
					   join_rows, surv, region, counts,
					   ovf);
			break;
		case 1:
			hipLaunchKernelGGL((k_dn_probe_lineitem_t<1, 4>), gg,
					   bb, 0, s, okey, shipdate, price,
					   disc, n, cutoff, pr,
					   dense_len, bloom, bwords,
					   join_rows, surv, region, counts,
					   ovf);
			break;
		case 2:
			hipLaunchKernelGGL((k_dn_probe_lineitem_t<1, 8>), gg,
					   bb, 0, s, okey, shipdate, price,
					   disc, n, cutoff, pr,
					   dense_len, bloom, bwords,
					   join_rows, surv, region, counts,
					   ovf);
			break;
		case 3:
			hipLaunchKernelGGL((k_dn_probe_lineitem_t<0, 4>), gg,
					   bb, 0, s, okey, shipdate, price,
					   disc, n, cutoff, pr,
					   dense_len, bloom, bwords,
					   join_rows, surv, region, counts,
					   ovf);
			break;
		case 4:
			hipLaunchKernelGGL((k_dn_probe_lineitem_t<2, 8>), gg,
					   bb, 0, s, okey, shipdate, price,
					   disc, n, cutoff, pr,
					   dense_len, bloom, bwords,
					   join_rows, surv, region, counts,
					   ovf);
			break;
		case 5:
			hipLaunchKernelGGL((k_dn_probe_lineitem_t<2, 4>), gg,
					   bb, 0, s, okey, shipdate, price,
					   disc, n, cutoff, pr,
					   dense_len, bloom, bwords,
					   join_rows, surv, region, counts,
					   ovf);
			break;
	}
	return hipGetLastError();
}

/* ---- Q5 u8 dense orders map ----------------------------------------
 * The Q5 orders payload is only the customer's nation (0..24), so the
 * dense map is a byte array (value 255 = no matching order): 8x less
 * random probe traffic than the u64 map, and it doubles as its own
 * membership filter, so the probe needs no Bloom reads at all. */

__global__ __launch_bounds__(DN_THREADS, 8)
void k_dn_build_orders_q5_u8(const int64_t *__restrict__ okey,
			const int64_t *__restrict__ ckey,
			const int32_t *__restrict__ odate, int64_t n,
			int32_t date_lo, int32_t date_hi,
			const unsigned long long *__restrict__ cust_keys,
			const unsigned long long *__restrict__ cust_pay,
			uint64_t cust_slots,
			const uint8_t *__restrict__ cust_dense,
			int64_t cust_dlen, uint8_t *__restrict__ pay8,
			int64_t dense_len, unsigned long long *match_count)
{
	const int64_t stride = (int64_t) gridDim.x * blockDim.x;
	unsigned long long matches = 0;

	for (int64_t i = (int64_t) blockIdx.x * blockDim.x + threadIdx.x;
	     i < n; i += stride)
	{
		int32_t d = dn_ld32(&odate[i]);
		int64_t ck = dn_ld64(&ckey[i]);
		int64_t k = dn_ld64(&okey[i]);
		unsigned long long nat = 0;
		bool ok;

		if (d < date_lo || d >= date_hi)
			continue;
		if (cust_dense)
		{
			ok = (ck >= 0 && ck < cust_dlen &&
			      cust_dense[ck] != 255);
			if (ok)
				nat = cust_dense[ck];
		}
		else
		{
			uint32_t h = gg_hashint8(ck);
			uint64_t pos = (uint64_t) h & (cust_slots - 1);

			ok = false;
			for (;;)
			{
				unsigned long long v = cust_keys[pos];

				if (v == (unsigned long long) ck)
				{
					nat = cust_pay[pos];
					ok = true;
					break;
				}
				if (v == 0)
					break;
				pos = (pos + 1) & (cust_slots - 1);
			}
		}
		if (!ok || k < 0 || k >= dense_len)
			continue;
		matches++;
		__builtin_nontemporal_store((uint8_t) nat, &pay8[k]);
	}
	{
		__shared__ unsigned long long lsum[4];

		dn_block_add(lsum, matches, match_count);
	}
}

hipError_t
launch_dn_build_orders_q5_u8(hipStream_t s, const int64_t *okey,
			     const int64_t *ckey, const int32_t *odate,
			     int64_t n, int32_t date_lo, int32_t date_hi,
			     DeviceHashTable cust, const uint8_t *cust_dense,
			     int64_t cust_dlen, uint8_t *pay8,
			     int64_t dense_len,
			     unsigned long long *match_count)
{
	hipLaunchKernelGGL(k_dn_build_orders_q5_u8, dim3(dn_grid(n)),
			   dim3(DN_THREADS), 0, s, okey, ckey, odate, n,
			   date_lo, date_hi, cust.keys, cust.payload,
			   cust.nslots, cust_dense, cust_dlen, pay8,
			   dense_len, match_count);
	return hipGetLastError();
}

/* exchange-path insert of received (okey, nation) rows */
__global__ void
k_dn_insert_orders_q5_u8(const int64_t *__restrict__ okey,
			 const int64_t *__restrict__ rownat, int64_t n,
			 uint8_t *__restrict__ pay8, int64_t dense_len)
{
	const int64_t stride = (int64_t) gridDim.x * blockDim.x;

	for (int64_t i = (int64_t) blockIdx.x * blockDim.x + threadIdx.x;
	     i < n; i += stride)
	{
		int64_t k = okey[i];

		if (k < 0 || k >= dense_len)
			continue;
		pay8[k] = (uint8_t) rownat[i];
	}
}

hipError_t
launch_dn_insert_orders_q5_u8(hipStream_t s, const int64_t *okey,
			      const int64_t *rownat, int64_t n,
			      uint8_t *pay8, int64_t dense_len)
{
	hipLaunchKernelGGL(k_dn_insert_orders_q5_u8, dim3(dn_grid(n)),
			   dim3(DN_THREADS), 0, s, okey, rownat, n, pay8,
			   dense_len);
	return hipGetLastError();
}

template <int QUAD, int WAVES>
__global__ __launch_bounds__(DN_THREADS, WAVES)
void k_dn_probe_lineitem_q5_u8_t(const int64_t *__restrict__ okey,
			  const int64_t *__restrict__ skey,
			  const int64_t *__restrict__ price,
			  const int64_t *__restrict__ disc, int64_t n,
			  const uint8_t *__restrict__ pay8,
			  int64_t dense_len,
			  const uint8_t *__restrict__ supp_dense,
			  int64_t supp_dlen,
			  unsigned long long *__restrict__ acc /* [25][3] */,
			  unsigned long long *join_rows)
{
	__shared__ unsigned long long lds[25][2];

	for (int i = threadIdx.x; i < 50; i += blockDim.x)
		((unsigned long long *) lds)[i] = 0;
	__syncthreads();

	unsigned long long joined = 0;

	/* filter order: the orders map passes ~3% of rows (date range x
	 * in-region customer) and its reads are nearly sequential
	 * (lineitem is orderkey-ordered; XCD chunking keeps each XCD's
	 * map slice local), so check it before touching any other column
	 * — skey/price/disc lines are then only fetched for passing
	 * lanes.  4 consecutive rows per lane → 16-B vector loads. */
	auto body = [&](int64_t i, int64_t k)
	{
		unsigned snat, onat;

		if (k < 0 || k >= dense_len)
			return;
		onat = pay8[k];
		if (onat == 255)
			return;
		int64_t sk = skey[i];

		if (sk < 0 || sk >= supp_dlen)
			return;
		snat = supp_dense[sk];
		if (snat != onat)	/* 255 (absent) never equals 0..24 */
			return;
		joined++;
		{
			unsigned long long rev4 = (unsigned long long)
				(price[i] * (100 - disc[i]));

			atomicAdd(&lds[snat][0], 1ull);
			atomicAdd(&lds[snat][1], rev4);
		}
	};

	if (QUAD)
	{
		DnChunk c = dn_chunk4(n);
		const int64_t S = c.stride;

		for (int64_t r = c.base + (int64_t) threadIdx.x * 4;
		     r < c.hi; r += S)
		{
			if (r + 3 < c.hi)
			{
				gg_ll2 k01 = dn_ld2(&okey[r]);
				gg_ll2 k23 = dn_ld2(&okey[r + 2]);

				body(r, k01.x);
				body(r + 1, k01.y);
				body(r + 2, k23.x);
				body(r + 3, k23.y);
			}
			else
				for (int64_t j = r; j < c.hi; j++)
					body(j, dn_ld64(&okey[j]));
		}
	}
	else if (QUAD == 2)
	{
		DnChunk c = dn_chunk(n);
		const int64_t S = c.stride;
		int64_t i = c.base + threadIdx.x;

		for (; i + 7 * S < c.hi; i += 8 * S)
		{
			int64_t k[8];

			for (int u = 0; u < 8; u++)
				k[u] = dn_ld64(&okey[i + u * S]);
			for (int u = 0; u < 8; u++)
				body(i + u * S, k[u]);
		}
		for (; i < c.hi; i += S)
			body(i, dn_ld64(&okey[i]));
	}
	else
	{
		DnChunk c = dn_chunk(n);
		const int64_t S = c.stride;
		int64_t i = c.base + threadIdx.x;

		for (; i + 3 * S < c.hi; i += 4 * S)
		{
			int64_t k0 = dn_ld64(&okey[i]);
			int64_t k1 = dn_ld64(&okey[i + S]);
			int64_t k2 = dn_ld64(&okey[i + 2 * S]);
			int64_t k3 = dn_ld64(&okey[i + 3 * S]);

			body(i, k0);
			body(i + S, k1);
			body(i + 2 * S, k2);
			body(i + 3 * S, k3);
		}
		for (; i < c.hi; i += S)
			body(i, dn_ld64(&okey[i]));
	}
	for (int off = 32; off; off >>= 1)
		joined += __shfl_down(joined, off, 64);
	if ((threadIdx.x & 63) == 0 && joined)
		atomicAdd(join_rows, joined);
	__syncthreads();
	for (int nat = threadIdx.x; nat < 25; nat += blockDim.x)
	{
		unsigned long long c = lds[nat][0];
		unsigned long long r = lds[nat][1];

		if (!c)
			continue;
		atomicAdd(&acc[nat * 3 + 0], c);
		{
			unsigned long long old =
				atomicAdd(&acc[nat * 3 + 1], r);
			if (old + r < old)
				atomicAdd(&acc[nat * 3 + 2], 1ull);
		}
	}
}

hipError_t
launch_dn_probe_lineitem_q5_u8(hipStream_t s, const int64_t *okey,
			       const int64_t *skey, const int64_t *price,
			       const int64_t *disc, int64_t n,
			       const uint8_t *pay8, int64_t dense_len,
			       const uint8_t *supp_dense, int64_t supp_dlen,
			       unsigned long long *acc,
			       unsigned long long *join_rows)
{
	dim3 gg((dn_grid_env(n, "GG_Q5_PROBE_GRID") + 7) & ~7);
	dim3 bb(DN_THREADS);

	switch (dn_probe_var("GG_Q5_PROBE_VAR", 0))
	{
		default:
		case 0:
			hipLaunchKernelGGL((k_dn_probe_lineitem_q5_u8_t<0, 8>),
					   gg, bb, 0, s, okey, skey, price,
					   disc, n, pay8, dense_len,
					   supp_dense, supp_dlen, acc,
					   join_rows);
			break;
		case 1:
			hipLaunchKernelGGL((k_dn_probe_lineitem_q5_u8_t<1, 4>),
					   gg, bb, 0, s, okey, skey, price,
					   disc, n, pay8, dense_len,
					   supp_dense, supp_dlen, acc,
					   join_rows);
			break;
		case 2:
			hipLaunchKernelGGL((k_dn_probe_lineitem_q5_u8_t<1, 8>),
					   gg, bb, 0, s, okey, skey, price,
					   disc, n, pay8, dense_len,
					   supp_dense, supp_dlen, acc,
					   join_rows);
			break;
		case 3:
			hipLaunchKernelGGL((k_dn_probe_lineitem_q5_u8_t<0, 4>),
					   gg, bb, 0, s, okey, skey, price,
					   disc, n, pay8, dense_len,
					   supp_dense, supp_dlen, acc,
					   join_rows);
			break;
		case 4:
			hipLaunchKernelGGL((k_dn_probe_lineitem_q5_u8_t<2, 8>),
					   gg, bb, 0, s, okey, skey, price,
					   disc, n, pay8, dense_len,
					   supp_dense, supp_dlen, acc,
					   join_rows);
			break;
		case 5:
			hipLaunchKernelGGL((k_dn_probe_lineitem_q5_u8_t<2, 4>),
					   gg, bb, 0, s, okey, skey, price,
					   disc, n, pay8, dense_len,
					   supp_dense, supp_dlen, acc,
					   join_rows);
			break;
	}
	return hipGetLastError();
}

/* Two-pass Q5 probe.  Pass A streams only okey + the u8 orders map
 * and compacts the ~3%% of rows with a matching order into PER-BLOCK
 * regions of the output (positions from an LDS counter — no global
 * atomics: a single returning global counter saturates at ~88
 * atomics/us, which made a wave-append version 45x slower).  Pass B
 * gathers skey/price/disc only for the survivors (indices ascending
 * within a region, so lines still coalesce) and finishes the
 * supplier-nation join + aggregation; gather block b walks compact
 * region b.
 */
__global__ __launch_bounds__(DN_THREADS, 8)
void k_dn_q5_compact(const int64_t *__restrict__ okey, int64_t n,
		     const uint8_t *__restrict__ pay8, int64_t dense_len,
		     int64_t region, /* per-block output capacity */
		     unsigned long long *__restrict__ out /* idx<<8|nat */ ,
		     unsigned long long *__restrict__ counts /* [grid] */ )
{
	__shared__ unsigned long long lcnt;

	if (threadIdx.x == 0)
		lcnt = 0;
	__syncthreads();

	const int64_t stride = (int64_t) gridDim.x * blockDim.x;
	unsigned long long *mine = out + (int64_t) blockIdx.x * region;

	for (int64_t i = (int64_t) blockIdx.x * blockDim.x + threadIdx.x;
	     i < n; i += stride)
	{
		int64_t k = dn_ld64(&okey[i]);

		if (k < 0 || k >= dense_len)
			continue;
		unsigned nat = pay8[k];

		if (nat == 255)
			continue;
		unsigned long long at = atomicAdd(&lcnt, 1ull);

		if (at < (unsigned long long) region)
			mine[at] = ((unsigned long long) i << 8) | nat;
	}
	__syncthreads();
	if (threadIdx.x == 0)
		counts[blockIdx.x] = lcnt;
}

__global__ __launch_bounds__(DN_THREADS, 8)
void k_dn_q5_gather(const unsigned long long *__restrict__ comp,
		    const unsigned long long *__restrict__ counts,
		    int64_t region, int64_t nregions,
		    const int64_t *__restrict__ skey,
		    const int64_t *__restrict__ price,
		    const int64_t *__restrict__ disc,
		    const uint8_t *__restrict__ supp_dense,
		    int64_t supp_dlen,
		    unsigned long long *__restrict__ acc /* [25][3] */ ,
		    unsigned long long *join_rows,
		    unsigned long long *overflow)
{
	__shared__ unsigned long long lds[25][2];

	for (int i = threadIdx.x; i < 50; i += blockDim.x)
		((unsigned long long *) lds)[i] = 0;
	__syncthreads();

	unsigned long long joined = 0;

	for (int64_t b = blockIdx.x; b < nregions; b += gridDim.x)
	{
		unsigned long long m = counts[b];

		if (m > (unsigned long long) region)
		{
			if (threadIdx.x == 0)
				atomicOr(overflow, 1ull);
			m = (unsigned long long) region;
		}
		const unsigned long long *seg = comp + b * region;

		for (unsigned long long j = threadIdx.x; j < m;
		     j += blockDim.x)
		{
			unsigned long long e = seg[j];
			int64_t i = (int64_t) (e >> 8);
			unsigned onat = (unsigned) (e & 0xFF);
			int64_t sk = skey[i];

			if (sk < 0 || sk >= supp_dlen)
				continue;
			unsigned snat = supp_dense[sk];

			if (snat != onat)
				continue;
			joined++;
			{
				unsigned long long rev4 =
					(unsigned long long)
					(price[i] * (100 - disc[i]));

				atomicAdd(&lds[onat][0], 1ull);
				atomicAdd(&lds[onat][1], rev4);
			}
		}
	}
	for (int off = 32; off; off >>= 1)
		joined += __shfl_down(joined, off, 64);
	if ((threadIdx.x & 63) == 0 && joined)
		atomicAdd(join_rows, joined);
	__syncthreads();
	for (int nat = threadIdx.x; nat < 25; nat += blockDim.x)
	{
		unsigned long long c = lds[nat][0];
		unsigned long long r = lds[nat][1];

		if (!c)
			continue;
		atomicAdd(&acc[nat * 3 + 0], c);
		{
			unsigned long long old =
				atomicAdd(&acc[nat * 3 + 1], r);

			if (old + r < old)
				atomicAdd(&acc[nat * 3 + 2], 1ull);
		}
	}
}

hipError_t
launch_dn_q5_compact(hipStream_t s, const int64_t *okey, int64_t n,
		     const uint8_t *pay8, int64_t dense_len, int64_t region,
		     unsigned long long *out, unsigned long long *counts)
{
	hipLaunchKernelGGL(k_dn_q5_compact, dim3(dn_grid(n)),
			   dim3(DN_THREADS), 0, s, okey, n, pay8, dense_len,
			   region, out, counts);
	return hipGetLastError();
}

hipError_t
launch_dn_q5_gather(hipStream_t s, const unsigned long long *comp,
		    const unsigned long long *counts, int64_t region,
		    int64_t nregions, const int64_t *skey,
		    const int64_t *price, const int64_t *disc,
		    const uint8_t *supp_dense, int64_t supp_dlen,
		    unsigned long long *acc, unsigned long long *join_rows,
		    unsigned long long *overflow)
{
	hipLaunchKernelGGL(k_dn_q5_gather, dim3((int) nregions),
			   dim3(DN_THREADS), 0, s, comp, counts, region,
			   nregions, skey, price, disc, supp_dense,
			   supp_dlen, acc, join_rows, overflow);
	return hipGetLastError();
}

/* ---- survivor-list top-k (v3) --------------------------------------
 * The probe records each group's dense index once (0→nonzero rev
 * transition), so the top-k (the reference's bounded-heap switch,
 * tuplesort.c:1360–1377) never sweeps the GB-sized dense array:
 * stats+hist, threshold, collect and clear all walk the ~ngroups
 * survivor entries.  The histogram uses a 16-bit monotonic
 * exponent-mantissa code, so no max pre-pass is needed. */

/* monotonic 16-bit code of a u64 (r > 0): 6-bit exponent bucket + 10
 * mantissa bits; order-preserving, max code 55 295 < 65 536 */
__device__ inline unsigned dn_code16(unsigned long long r)
{
	int e = 63 - __clzll(r);

	if (e <= 9)
		return (unsigned) r;
	return (unsigned) (((unsigned) (e - 9) << 10) |
			   ((r >> (e - 10)) & 1023));
}

/* smallest u64 whose code is `code` (threshold decode) */
__device__ inline unsigned long long dn_code16_lo(unsigned code)
{
	if (code < 1024)
		return code;
	{
		unsigned b = code >> 10;
		unsigned m = code & 1023;

		return (1024ull + m) << (b - 1);
	}
}

/* Survivor-list top-k: the probe appended each group's dense index
 * once (0→nonzero transition of its rev[] slot), so every stage walks
 * ~ngroups entries (~10 MB of gathers at SF100) instead of sweeping
 * the 1.2 GB dense array — the r02c PMC showed those sweeps were
 * latency-bound at 2.1 TB/s. */

/* gather stats + 16-bit exponent-mantissa histogram over the
 * per-block survivor regions the probe filled */
__global__ __launch_bounds__(DN_THREADS, 8)
void k_dn_q3_stats_surv(const unsigned long long *__restrict__ surv,
			const unsigned long long *__restrict__ counts,
			uint64_t region, int64_t nregions,
			const unsigned long long *__restrict__ pr,
			unsigned long long *__restrict__ out5,
			unsigned int *__restrict__ hist64k)
{
	unsigned long long ng = 0, revsum = 0, carry = 0, ck = 0, mx = 0;

	for (int64_t b = blockIdx.x; b < nregions; b += gridDim.x)
	{
	const unsigned long long *seg = surv + (uint64_t) b * region;
	unsigned long long m = counts[b];

	for (unsigned long long i = threadIdx.x; i < m; i += blockDim.x)
	{
		int64_t k = (int64_t) seg[i];
		/* rev and pay interleave as a 16-B pair: one cache line
		 * per survivor instead of two (this kernel is cold-
		 * random-gather bound) */
		gg_ull2 rp = *(const gg_ull2 *) &pr[2 * k];
		unsigned long long r = rp.x;

		if (!r)
			continue;	/* zero-sum group: invisible, as the
					 * dense-array sweep always treated it */
		ng++;
		mx = max(mx, r);
		{
			unsigned long long old = revsum;

			revsum += r;
			carry += (revsum < old);
		}
		/* SUBSET-sampled histogram: every 16th survivor.  The
		 * threshold walk still targets k, and k sampled values
		 * >= thr imply >= k TRUE values >= thr (samples are a
		 * subset), so the threshold is always admissible; the
		 * candidate set just inflates ~16x k (collected exactly,
		 * host-sorts the tail).  This cuts the global hist
		 * atomics 16x — they dominated this kernel (537 µs vs
		 * the 25 µs the same gathers cost in collect_surv). */
		if ((i & 15) == 0)
			atomicAdd(&hist64k[dn_code16(r)], 1u);
		{
			unsigned long long p = rp.y;
			int32_t date = (int32_t) (uint32_t) p;
			int32_t prio = (int32_t) (uint32_t) (p >> 32);

			ck += gg_group_hash((unsigned long long) k, r, 0,
					    date, prio);
		}
	}
	}
	for (int off = 32; off; off >>= 1)
	{
		unsigned long long orev = revsum;

		ng += __shfl_down(ng, off, 64);
		revsum += __shfl_down(revsum, off, 64);
		carry += __shfl_down(carry, off, 64) + (revsum < orev);
		ck += __shfl_down(ck, off, 64);
		mx = max(mx, __shfl_down(mx, off, 64));
	}
	/* block-level combine: ONE set of global atomics per block */
	{
		__shared__ unsigned long long ls[4][5];
		int w = threadIdx.x >> 6;

		if ((threadIdx.x & 63) == 0)
		{
			ls[w][0] = ng;
			ls[w][1] = revsum;
			ls[w][2] = carry;
			ls[w][3] = ck;
			ls[w][4] = mx;
		}
		__syncthreads();
		if (threadIdx.x == 0)
		{
			unsigned long long bng = 0, brev = 0, bcar = 0,
				bck = 0, bmx = 0;

			for (int i = 0; i < (int) (blockDim.x >> 6); i++)
			{
				unsigned long long orev = brev;

				bng += ls[i][0];
				brev += ls[i][1];
				bcar += ls[i][2] + (brev < orev);
				bck += ls[i][3];
				bmx = max(bmx, ls[i][4]);
			}
			if (bng)
			{
				atomicAdd(&out5[0], bng);
				{
					unsigned long long old =
						atomicAdd(&out5[1], brev);

					if (old + brev < old)
						atomicAdd(&out5[2], 1ull);
					atomicAdd(&out5[2], bcar);
				}
				atomicAdd(&out5[3], bck);
				atomicMax(&out5[4], bmx);
			}
		}
	}
}

hipError_t
launch_dn_q3_stats_surv(hipStream_t s, const unsigned long long *surv,
			const unsigned long long *counts, uint64_t region,
			int64_t nregions, const unsigned long long *pr,
			unsigned long long *out5, unsigned int *hist64k)
{
	int g = (int) (nregions < 1 ? 1 :
		       (nregions > DN_MAX_BLOCKS ? DN_MAX_BLOCKS : nregions));

	hipLaunchKernelGGL(k_dn_q3_stats_surv, dim3(g),
			   dim3(DN_THREADS), 0, s, surv, counts, region,
			   nregions, pr, out5, hist64k);
	return hipGetLastError();
}

/* one block: walk the 16-bit-code histogram from the top until >= k
 * candidates; emit the decoded revenue threshold */
__global__ __launch_bounds__(1024)
void k_dn_q3_threshold2(const unsigned int *__restrict__ hist64k, int64_t k,
			unsigned long long *__restrict__ out_thr)
{
	__shared__ unsigned long long partial[1024];
	unsigned long long s = 0;

	for (int b = 0; b < 64; b++)
		s += hist64k[threadIdx.x * 64 + b];
	partial[threadIdx.x] = s;
	__syncthreads();
	if (threadIdx.x == 0)
	{
		unsigned long long cum = 0;
		int chunk = 1023;

		for (; chunk > 0; chunk--)
		{
			if (cum + partial[chunk] >= (unsigned long long) k)
				break;
			cum += partial[chunk];
		}
		{
			int b = 63;

			for (; b > 0; b--)
			{
				unsigned long long c =
					hist64k[chunk * 64 + b];

				if (cum + c >= (unsigned long long) k)
					break;
				cum += c;
			}
			{
				unsigned long long thr = dn_code16_lo(
					(unsigned) (chunk * 64 + b));

				*out_thr = thr ? thr : 1;
			}
		}
	}
}

hipError_t
launch_dn_q3_threshold2(hipStream_t s, const unsigned int *hist64k, int64_t k,
			unsigned long long *out_thr)
{
	hipLaunchKernelGGL(k_dn_q3_threshold2, dim3(1), dim3(1024), 0, s,
			   hist64k, k, out_thr);
	return hipGetLastError();
}

/* collect candidates >= threshold from the survivor regions */
__global__ __launch_bounds__(DN_THREADS, 8)
void k_dn_q3_collect_surv(const unsigned long long *__restrict__ surv,
			  const unsigned long long *__restrict__ counts,
			  uint64_t region, int64_t nregions,
			  const unsigned long long *__restrict__ pr,
			  const unsigned long long *__restrict__ thr_ptr,
			  gg_q3_result_row *__restrict__ out,
			  unsigned long long *out_count, uint64_t cap)
{
	const unsigned long long threshold = *thr_ptr;

	for (int64_t b = blockIdx.x; b < nregions; b += gridDim.x)
	{
		const unsigned long long *seg = surv + (uint64_t) b * region;
		unsigned long long m = counts[b];

		for (unsigned long long i = threadIdx.x; i < m;
		     i += blockDim.x)
		{
			int64_t k = (int64_t) seg[i];
			gg_ull2 rp = *(const gg_ull2 *) &pr[2 * k];
			unsigned long long r = rp.x;
			bool take = (r != 0 && r >= threshold);
			unsigned long long idx =
				dn_wave_append(out_count, take);

			if (!take || idx >= cap)
				continue;
			{
				unsigned long long p = rp.y;

				out[idx].orderkey = k;
				out[idx].rev_lo = r;
				out[idx].rev_hi = 0;
				out[idx].orderdate = (int32_t) (uint32_t) p;
				out[idx].shippriority =
					(int32_t) (uint32_t) (p >> 32);
			}
		}
	}
}

hipError_t
launch_dn_q3_collect_surv(hipStream_t s, const unsigned long long *surv,
			  const unsigned long long *counts, uint64_t region,
			  int64_t nregions, const unsigned long long *pr,
			  const unsigned long long *thr_ptr,
			  gg_q3_result_row *out,
			  unsigned long long *out_count, uint64_t cap)
{
	int g = (int) (nregions < 1 ? 1 :
		       (nregions > DN_MAX_BLOCKS ? DN_MAX_BLOCKS : nregions));

	hipLaunchKernelGGL(k_dn_q3_collect_surv, dim3(g),
			   dim3(DN_THREADS), 0, s, surv, counts, region,
			   nregions, pr, thr_ptr, out, out_count, cap);
	return hipGetLastError();
}

/* zero exactly the rev[] entries this pass touched, so the next
 * execute skips the full dense memset */
__global__ __launch_bounds__(DN_THREADS, 8)
void k_dn_q3_clear_surv(const unsigned long long *__restrict__ surv,
			const unsigned long long *__restrict__ counts,
			uint64_t region, int64_t nregions,
			unsigned long long *__restrict__ pr)
{
	for (int64_t b = blockIdx.x; b < nregions; b += gridDim.x)
	{
		const unsigned long long *seg = surv + (uint64_t) b * region;
		unsigned long long m = counts[b];

		for (unsigned long long i = threadIdx.x; i < m;
		     i += blockDim.x)
			pr[2 * seg[i]] = 0;
	}
}

hipError_t
launch_dn_q3_clear_surv(hipStream_t s, const unsigned long long *surv,
			const unsigned long long *counts, uint64_t region,
			int64_t nregions, unsigned long long *pr)
{
	int g = (int) (nregions < 1 ? 1 :
		       (nregions > DN_MAX_BLOCKS ? DN_MAX_BLOCKS : nregions));

	hipLaunchKernelGGL(k_dn_q3_clear_surv, dim3(g),
			   dim3(DN_THREADS), 0, s, surv, counts, region,
			   nregions, pr);
	return hipGetLastError();
}

/* stats (+max) over the dense group arrays (orderkey = index) */
__global__ __launch_bounds__(DN_THREADS, 8)
void k_dn_q3_stats(const unsigned long long *__restrict__ pr,
	      int64_t dense_len, unsigned long long *out5)
{
	const int64_t stride = (int64_t) gridDim.x * blockDim.x;
	unsigned long long ng = 0, revsum = 0, carry = 0, ck = 0, mx = 0;

	/* fallback full sweep over the interleaved (rev, pay) pairs —
	 * 16-B NT loads, 2 per iteration for memory-level parallelism */
	auto body = [&](int64_t i, unsigned long long r, unsigned long long p)
	{
		if (!r)
			return;
		ng++;
		mx = max(mx, r);
		{
			unsigned long long old = revsum;

			revsum += r;
			carry += (revsum < old);
		}
		{
			int32_t date = (int32_t) (uint32_t) p;
			int32_t prio = (int32_t) (uint32_t) (p >> 32);

			ck += gg_group_hash(i, r, 0, date, prio);
		}
	};
	const int64_t i0 = (int64_t) blockIdx.x * blockDim.x + threadIdx.x;
	int64_t i = i0;

	for (; i + stride < dense_len; i += 2 * stride)
	{
		gg_ull2 a = __builtin_nontemporal_load(
			(const gg_ull2 *) &pr[2 * i]);
		gg_ull2 b = __builtin_nontemporal_load(
			(const gg_ull2 *) &pr[2 * (i + stride)]);

		body(i, a.x, a.y);
		body(i + stride, b.x, b.y);
	}
	for (; i < dense_len; i += stride)
		body(i, pr[2 * i], pr[2 * i + 1]);
	for (int off = 32; off; off >>= 1)
	{
		unsigned long long orev = revsum;

		ng += __shfl_down(ng, off, 64);
		revsum += __shfl_down(revsum, off, 64);
		carry += __shfl_down(carry, off, 64) + (revsum < orev);
		ck += __shfl_down(ck, off, 64);
		mx = max(mx, __shfl_down(mx, off, 64));
	}
	/* block-level combine: ONE set of global atomics per block */
	{
		__shared__ unsigned long long ls[4][5];
		int w = threadIdx.x >> 6;

		if ((threadIdx.x & 63) == 0)
		{
			ls[w][0] = ng;
			ls[w][1] = revsum;
			ls[w][2] = carry;
			ls[w][3] = ck;
			ls[w][4] = mx;
		}
		__syncthreads();
		if (threadIdx.x == 0)
		{
			unsigned long long bng = 0, brev = 0, bcar = 0,
				bck = 0, bmx = 0;

			for (int i = 0; i < (int) (blockDim.x >> 6); i++)
			{
				unsigned long long orev = brev;

				bng += ls[i][0];
				brev += ls[i][1];
				bcar += ls[i][2] + (brev < orev);
				bck += ls[i][3];
				bmx = max(bmx, ls[i][4]);
			}
			if (bng)
			{
				atomicAdd(&out5[0], bng);
				{
					unsigned long long old =
						atomicAdd(&out5[1], brev);

					if (old + brev < old)
						atomicAdd(&out5[2], 1ull);
					atomicAdd(&out5[2], bcar);
				}
				atomicAdd(&out5[3], bck);
				atomicMax(&out5[4], bmx);
			}
		}
	}
}

hipError_t
launch_dn_q3_stats(hipStream_t s, const unsigned long long *pr,
		   int64_t dense_len, unsigned long long *out5)
{
	hipLaunchKernelGGL(k_dn_q3_stats, dim3(dn_grid(dense_len)),
			   dim3(DN_THREADS), 0, s, pr, dense_len, out5);
	return hipGetLastError();
}

__global__ void
k_dn_q3_hist(const unsigned long long *__restrict__ pr, int64_t dense_len,
	     const unsigned long long *__restrict__ stats5,
	     unsigned int *__restrict__ hist64k)
{
	const int64_t stride = (int64_t) gridDim.x * blockDim.x;
	int shift = 0;
	unsigned long long maxrev = stats5[4];

	while ((maxrev >> shift) > 65535)
		shift++;
	for (int64_t i = (int64_t) blockIdx.x * blockDim.x + threadIdx.x;
	     i < dense_len; i += stride)
	{
		unsigned long long r = pr[2 * i];

		if (!r)
			continue;
		{
			unsigned long long bin = r >> shift;

			if (bin > 65535)
				bin = 65535;
			atomicAdd(&hist64k[bin], 1u);
		}
	}
}

hipError_t
launch_dn_q3_hist(hipStream_t s, const unsigned long long *pr,
		  int64_t dense_len, const unsigned long long *stats5,
		  unsigned int *hist64k)
{
	hipLaunchKernelGGL(k_dn_q3_hist, dim3(dn_grid(dense_len)),
			   dim3(DN_THREADS), 0, s, pr, dense_len, stats5,
			   hist64k);
	return hipGetLastError();
}

__global__ void
k_dn_q3_collect(const unsigned long long *__restrict__ pr,
		int64_t dense_len,
		const unsigned long long *__restrict__ thr_ptr,
		gg_q3_result_row *__restrict__ out,
		unsigned long long *out_count, uint64_t cap)
{
	const int64_t stride = (int64_t) gridDim.x * blockDim.x;
	const unsigned long long threshold = *thr_ptr;

	for (int64_t i = (int64_t) blockIdx.x * blockDim.x + threadIdx.x;
	     i < dense_len; i += stride)
	{
		unsigned long long r = pr[2 * i];
		bool take = (r != 0 && r >= threshold);
		unsigned long long idx = dn_wave_append(out_count, take);

		if (!take || idx >= cap)
			continue;
		{
			unsigned long long p = pr[2 * i + 1];

			out[idx].orderkey = i;
			out[idx].rev_lo = r;
			out[idx].rev_hi = 0;
			out[idx].orderdate = (int32_t) (uint32_t) p;
			out[idx].shippriority = (int32_t) (uint32_t) (p >> 32);
		}
	}
}

hipError_t
launch_dn_q3_collect(hipStream_t s, const unsigned long long *pr,
		     int64_t dense_len,
		     const unsigned long long *thr_ptr, gg_q3_result_row *out,
		     unsigned long long *out_count, uint64_t cap)
{
	hipLaunchKernelGGL(k_dn_q3_collect, dim3(dn_grid(dense_len)),
			   dim3(DN_THREADS), 0, s, pr, dense_len,
			   thr_ptr, out, out_count, cap);
	return hipGetLastError();
}

}				/* namespace gg */
