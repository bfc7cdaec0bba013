/*
 * Generic compiled-plan kernels (engine_abi.h "generalized pipeline
 * descriptor"): ONE build kernel and ONE fused scan/filter/join/agg
 * kernel execute any descriptor-shaped query — no query-specific code.
 *
 * Reference semantics implemented:
 *   predicates    execScan.c:110–214 + execQual.c:6260 (conjunctive
 *                 range quals; NULL input → qual not true)
 *   semi joins    nodeHash.c:88 (build) / nodeHashjoin.c:78 (probe);
 *                 NULL join keys never match (nodeHash.c:1070–1077)
 *   group-by      execHHashagg.c:905 find-or-create; NULL keys group
 *                 together (:531)
 *   transitions   nodeAgg.c:413–460 strict rules: SUM/COUNT(col) skip
 *                 NULL inputs, COUNT(*) does not
 *   agg exprs     products with (100±col) under numeric.c:1735 mul_var
 *                 scale addition (scaled-int exact arithmetic)
 *
 * Layouts: group table = open addressing, EMPTY = INT64_MIN bit
 * pattern; hash semi-set stores key ^ 0x8000…, 0 = empty (build keys
 * of INT64_MIN rejected at compile); dense build sides collapse to a
 * 1-bit membership bitmap under the same max(key) <= 8x rows guard as
 * the named pipelines.
 */
#include <hip/hip_runtime.h>
#include <cstdlib>

#include "../../include/gg_pg_hash.h"
#include "engine_internal.h"

namespace gg
{

static constexpr int PL_THREADS = 256;
static constexpr int PL_MAX_BLOCKS = 2048;

static inline int pl_grid(int64_t n)
{
	int64_t b = (n + PL_THREADS - 1) / PL_THREADS;

	return (int) (b < 1 ? 1 : (b > PL_MAX_BLOCKS ? PL_MAX_BLOCKS : b));
}

template <int NT>
__device__ static inline int64_t pl_ld(const void *c, int w, int64_t i)
{
	if (NT)
		switch (w)
		{
			case 1:
				return __builtin_nontemporal_load(
					&((const uint8_t *) c)[i]);
			case 4:
				return __builtin_nontemporal_load(
					&((const int32_t *) c)[i]);
			default:
				return __builtin_nontemporal_load(
					&((const int64_t *) c)[i]);
		}
	switch (w)
	{
		case 1:
			return ((const uint8_t *) c)[i];
		case 4:
			return ((const int32_t *) c)[i];
		default:
			return ((const int64_t *) c)[i];
	}
}

template <int NT>
__device__ static inline bool
pl_preds_pass(const PlanPredDev *preds, int npreds, int64_t i)
{
	for (int p = 0; p < npreds; p++)
	{
		if (preds[p].nulls && preds[p].nulls[i])
			return false;	/* NULL comparison is not true */
		{
			int64_t v = pl_ld<NT>(preds[p].col, preds[p].width, i);

			if (v < preds[p].lo || v >= preds[p].hi)
				return false;
		}
	}
	return true;
}

/* 4 rows at once, predicate loads UNCONDITIONAL: short-circuiting per
 * row turns the scan into dependent sparse gathers (measured 2.3 TB/s
 * on Q6); streaming every predicate column and ANDing the results
 * keeps the loads wide and independent like the named pipelines. */
template <int NT>
__device__ static inline void
pl_preds_pass4(const PlanPredDev *preds, int npreds, int64_t i, int64_t S,
	       bool ok[4])
{
	ok[0] = ok[1] = ok[2] = ok[3] = true;
	for (int p = 0; p < npreds; p++)
	{
		const PlanPredDev &P = preds[p];
		int64_t v0 = pl_ld<NT>(P.col, P.width, i);
		int64_t v1 = pl_ld<NT>(P.col, P.width, i + S);
		int64_t v2 = pl_ld<NT>(P.col, P.width, i + 2 * S);
		int64_t v3 = pl_ld<NT>(P.col, P.width, i + 3 * S);

		ok[0] &= v0 >= P.lo && v0 < P.hi;
		ok[1] &= v1 >= P.lo && v1 < P.hi;
		ok[2] &= v2 >= P.lo && v2 < P.hi;
		ok[3] &= v3 >= P.lo && v3 < P.hi;
		if (P.nulls)
		{
			ok[0] &= !P.nulls[i];
			ok[1] &= !P.nulls[i + S];
			ok[2] &= !P.nulls[i + 2 * S];
			ok[3] &= !P.nulls[i + 3 * S];
		}
	}
}

/* ---- semi-join build ------------------------------------------------ */

__global__ __launch_bounds__(PL_THREADS, 4)
void k_plan_build(PlanBuildDev B)
{
	const int64_t stride = (int64_t) gridDim.x * blockDim.x;

	for (int64_t i = (int64_t) blockIdx.x * blockDim.x + threadIdx.x;
	     i < B.n; i += stride)
	{
		if (!pl_preds_pass<0>(B.preds, B.npreds, i))
			continue;
		if (B.knulls && B.knulls[i])
			continue;	/* NULL build key never matches */
		{
			int64_t k = pl_ld<0>(B.key, B.kw, i);

			if (B.bits)
			{
				if (k >= 0 && k < B.dlen)
					atomicOr(&B.bits[k >> 6],
						 1ull << (k & 63));
			}
			else
			{
				unsigned long long t = (unsigned long long) k
					^ 0x8000000000000000ull;
				uint64_t pos = (uint64_t) gg_hashint8(k) &
					(B.hslots - 1);

				for (uint64_t it = 0; it < B.hslots; it++)
				{
					unsigned long long cur =
						B.hkeys[pos];

					if (cur == t)
						break;
					if (cur == 0)
					{
						unsigned long long prev =
							atomicCAS(&B.hkeys[pos],
								  0ull, t);

						if (prev == 0 || prev == t)
							break;
						continue;
					}
					pos = (pos + 1) & (B.hslots - 1);
				}
			}
		}
	}
}

hipError_t launch_plan_build(hipStream_t s, const PlanBuildDev &b)
{
	hipLaunchKernelGGL(k_plan_build, dim3(pl_grid(b.n)),
			   dim3(PL_THREADS), 0, s, b);
	return hipGetLastError();
}

/* ---- fused scan + filter + semi-joins + group agg ------------------- */

__device__ static inline bool
pl_joins_pass(const PlanJoinDev *joins, int njoins, int64_t i)
{
	for (int j = 0; j < njoins; j++)
	{
		const PlanJoinDev &J = joins[j];

		if (J.pnulls && J.pnulls[i])
			return false;	/* NULL probe key never matches */
		{
			int64_t k = pl_ld<0>(J.pkey, J.width, i);

			if (J.bits)
			{
				if (k < 0 || k >= J.dlen ||
				    !((J.bits[k >> 6] >> (k & 63)) & 1))
					return false;
			}
			else
			{
				unsigned long long t = (unsigned long long) k
					^ 0x8000000000000000ull;
				uint64_t pos = (uint64_t) gg_hashint8(k) &
					(J.hslots - 1);
				bool hit = false;

				for (;;)
				{
					unsigned long long cur =
						J.hkeys[pos];

					if (cur == t)
					{
						hit = true;
						break;
					}
					if (cur == 0)
						break;
					pos = (pos + 1) & (J.hslots - 1);
				}
				if (!hit)
					return false;
			}
		}
	}
	return true;
}

/* group code; INT64_MIN bit pattern = empty sentinel (never a code) */
__device__ static inline long long
pl_group_code(const PlanDev &P, int64_t i)
{
	if (P.ngroup == 0)
		return 0;
	if (P.ngroup == 2)
	{
		/* two char1 columns; NULL encodes as 256 so NULL groups
		 * stay distinct per column (execHHashagg.c:531) */
		long long e0 = (P.gnulls[0] && P.gnulls[0][i]) ? 256
			: pl_ld<0>(P.gcol[0], P.gwidth[0], i);
		long long e1 = (P.gnulls[1] && P.gnulls[1][i]) ? 256
			: pl_ld<0>(P.gcol[1], P.gwidth[1], i);

		return e0 * 512 + e1;
	}
	if (P.gnulls[0] && P.gnulls[0][i])
		return GG_PLAN_NULL_KEY;
	return pl_ld<0>(P.gcol[0], P.gwidth[0], i);
}

/* strict-transition aggregate input (nodeAgg.c:413) */
__device__ static inline bool
pl_agg_val(const PlanAggDev &a, int64_t i, __int128 *out)
{
	if (a.kind == 0)	/* COUNT(*) */
	{
		*out = 1;
		return true;
	}
	for (int f = 0; f < a.nf; f++)
		if (a.nulls[f] && a.nulls[f][i])
			return false;	/* strict: skip NULL input */
	if (a.kind == 1)	/* COUNT(col) */
	{
		*out = 1;
		return true;
	}
	{
		__int128 v = 1;

		for (int f = 0; f < a.nf; f++)
		{
			int64_t x = pl_ld<0>(a.col[f], a.width[f], i);

			if (a.mod[f] == 1)
				x = 100 - x;
			else if (a.mod[f] == 2)
				x = 100 + x;
			v *= x;
		}
		*out = v;
		return true;
	}
}

static constexpr unsigned long long PL_EMPTY = 0x8000000000000000ull;

__device__ static inline int64_t
pl_slot(const PlanDev &P, long long code)
{
	uint64_t pos = (uint64_t) gg_hashint8(code) & (P.nslots - 1);

	for (uint64_t it = 0; it < P.nslots; it++)
	{
		unsigned long long cur = P.tkeys[pos];

		if (cur == (unsigned long long) code)
			return (int64_t) pos;
		if (cur == PL_EMPTY)
		{
			unsigned long long prev =
				atomicCAS(&P.tkeys[pos], PL_EMPTY,
					  (unsigned long long) code);

			if (prev == PL_EMPTY ||
			    prev == (unsigned long long) code)
				return (int64_t) pos;
			continue;
		}
		pos = (pos + 1) & (P.nslots - 1);
	}
	return -1;
}

__device__ static inline void
pl_atomic_add128(unsigned long long *lo, unsigned long long *hi, __int128 v)
{
	unsigned long long vlo = (unsigned long long) v;
	unsigned long long vhi = (unsigned long long) (v >> 64);
	unsigned long long old = atomicAdd(lo, vlo);

	if (old + vlo < old)
		vhi++;
	if (vhi)
		atomicAdd(hi, vhi);
}

template <int NT>
__global__ __launch_bounds__(PL_THREADS, 4)
void k_plan_scan_agg(PlanDev P)
{
	const int64_t stride = (int64_t) gridDim.x * blockDim.x;

	if (P.ngroup == 0)
	{
		/* global-group fast path: register accumulators, one
		 * wave-reduced atomic flush per wave (the Q1/sumprice
		 * pattern — keeps e.g. Q6 at streaming rate); 4-way
		 * strided rows with unconditional predicate streams */
		unsigned long long alo[GG_PLAN_MAX_AGGS] = {};
		long long ahi[GG_PLAN_MAX_AGGS] = {};

		auto tail = [&](int64_t i)
		{
			if (!pl_preds_pass<NT>(P.preds, P.npreds, i))
				return;
			if (!pl_joins_pass(P.joins, P.njoins, i))
				return;
			for (int a = 0; a < P.naggs; a++)
			{
				__int128 v;

				if (!pl_agg_val(P.aggs[a], i, &v))
					continue;
				{
					unsigned long long vlo =
						(unsigned long long) v;
					unsigned long long old = alo[a];

					alo[a] += vlo;
					ahi[a] += (long long) (v >> 64) +
						(alo[a] < old);
				}
			}
		};
		int64_t i = (int64_t) blockIdx.x * blockDim.x + threadIdx.x;

		for (; i + 3 * stride < P.n; i += 4 * stride)
		{
			bool ok[4];

			pl_preds_pass4<NT>(P.preds, P.npreds, i, stride, ok);
			for (int j = 0; j < 4; j++)
			{
				int64_t r = i + j * stride;

				if (!ok[j])
					continue;
				if (!pl_joins_pass(P.joins, P.njoins, r))
					continue;
				for (int a = 0; a < P.naggs; a++)
				{
					__int128 v;

					if (!pl_agg_val(P.aggs[a], r, &v))
						continue;
					{
						unsigned long long vlo =
							(unsigned long long) v;
						unsigned long long old =
							alo[a];

						alo[a] += vlo;
						ahi[a] += (long long)
							(v >> 64) +
							(alo[a] < old);
					}
				}
			}
		}
		for (; i < P.n; i += stride)
			tail(i);
		for (int a = 0; a < P.naggs; a++)
		{
			for (int off = 32; off; off >>= 1)
			{
				unsigned long long olo = alo[a];

				alo[a] += __shfl_down(alo[a], off, 64);
				ahi[a] += __shfl_down(ahi[a], off, 64) +
					(long long) (alo[a] < olo);
			}
			if ((threadIdx.x & 63) == 0 && (alo[a] || ahi[a]))
			{
				__int128 v = ((__int128) ahi[a] << 64) |
					(__int128) alo[a];

				pl_atomic_add128(&P.tvals[2 * a],
						 &P.tvals[2 * a + 1], v);
			}
		}
		/* mark slot 0 used so compaction emits the group */
		if (blockIdx.x == 0 && threadIdx.x == 0)
			P.tkeys[0] = 0;
		return;
	}

	/* Per-block LDS group cache: low-cardinality group-bys (Q1's 6
	 * groups, nation's 25) otherwise hammer a handful of global
	 * accumulator words with per-row atomics — the ~88 atomics/µs
	 * hot-word wall.  64-slot LDS open addressing absorbs the per-row
	 * transitions; overflow rows (cardinality > ~64 per block) fall
	 * back to the global table, and the LDS slots flush once per
	 * block (execHHashagg.c:456 find-or-create, two-level). */
	constexpr int LSLOTS = 32;
	constexpr int LREPL = 8;	/* replicas split the per-word LDS
					 * atomic serialization 8 ways (the
					 * k_q1_agg replica pattern) */
	constexpr long long LEMPTY = (long long) 0x8000000000000000ull;
	__shared__ long long lkeys[LSLOTS];
	__shared__ unsigned long long
		lvals[LREPL][LSLOTS][2 * GG_PLAN_MAX_AGGS];

	for (int s = threadIdx.x; s < LSLOTS; s += blockDim.x)
		lkeys[s] = LEMPTY;
	for (int s = threadIdx.x;
	     s < LREPL * LSLOTS * 2 * GG_PLAN_MAX_AGGS; s += blockDim.x)
		((unsigned long long *) lvals)[s] = 0;
	__syncthreads();

	const int lrep = (int) (threadIdx.x & (LREPL - 1));

	auto global_row = [&](int64_t i, long long code)
	{
		int64_t slot = pl_slot(P, code);

		if (slot < 0)
		{
			atomicOr(P.err, 1ull);
			return;
		}
		for (int a = 0; a < P.naggs; a++)
		{
			__int128 v;

			if (!pl_agg_val(P.aggs[a], i, &v))
				continue;
			pl_atomic_add128(
				&P.tvals[(slot * P.naggs + a) * 2],
				&P.tvals[(slot * P.naggs + a) * 2 + 1], v);
		}
	};
	auto grouped_row = [&](int64_t i)
	{
		if (!pl_joins_pass(P.joins, P.njoins, i))
			return;
		{
			long long code = pl_group_code(P, i);
			int ls = -1;
			uint32_t pos = (uint32_t) gg_hashint8(code) &
				(LSLOTS - 1);

			for (int probe = 0; probe < 8; probe++)
			{
				long long cur = lkeys[pos];

				if (cur == code)
				{
					ls = (int) pos;
					break;
				}
				if (cur == LEMPTY)
				{
					long long prev = atomicCAS(
						(unsigned long long *)
						&lkeys[pos],
						(unsigned long long) LEMPTY,
						(unsigned long long) code);

					if (prev == LEMPTY || prev == code)
					{
						ls = (int) pos;
						break;
					}
					continue;
				}
				pos = (pos + 1) & (LSLOTS - 1);
			}
			if (ls < 0)
			{
				global_row(i, code);	/* cache full */
				return;
			}
			for (int a = 0; a < P.naggs; a++)
			{
				__int128 v;

				if (!pl_agg_val(P.aggs[a], i, &v))
					continue;
				{
					unsigned long long vlo =
						(unsigned long long) v;
					unsigned long long vhi =
						(unsigned long long) (v >> 64);
					unsigned long long old = atomicAdd(
						&lvals[lrep][ls][2 * a], vlo);

					if (old + vlo < old)
						vhi++;
					if (vhi)
						atomicAdd(&lvals[lrep][ls]
							  [2 * a + 1], vhi);
				}
			}
		}
	};
	int64_t i = (int64_t) blockIdx.x * blockDim.x + threadIdx.x;

	for (; i + 3 * stride < P.n; i += 4 * stride)
	{
		bool ok[4];

		pl_preds_pass4<NT>(P.preds, P.npreds, i, stride, ok);
		for (int j = 0; j < 4; j++)
			if (ok[j])
				grouped_row(i + j * stride);
	}
	for (; i < P.n; i += stride)
		if (pl_preds_pass<NT>(P.preds, P.npreds, i))
			grouped_row(i);

	/* flush the block's LDS groups into the global table */
	__syncthreads();
	for (int s = threadIdx.x; s < LSLOTS; s += blockDim.x)
	{
		if (lkeys[s] == LEMPTY)
			continue;
		{
			int64_t slot = pl_slot(P, lkeys[s]);

			if (slot < 0)
			{
				atomicOr(P.err, 1ull);
				continue;
			}
			for (int a = 0; a < P.naggs; a++)
			{
				unsigned long long lo = 0, hi = 0;

				for (int rr = 0; rr < LREPL; rr++)
				{
					unsigned long long rl =
						lvals[rr][s][2 * a];
					unsigned long long o = lo;

					lo += rl;
					hi += lvals[rr][s][2 * a + 1] +
						(lo < o);
				}

				if (!lo && !hi)
					continue;
				{
					unsigned long long old = atomicAdd(
						&P.tvals[(slot * P.naggs + a)
							 * 2], lo);

					if (old + lo < old)
						hi++;
					if (hi)
						atomicAdd(
							&P.tvals[(slot *
								  P.naggs + a)
								 * 2 + 1], hi);
				}
			}
		}
	}
}

hipError_t launch_plan_scan_agg(hipStream_t s, const PlanDev &p)
{
	const char *nt = getenv("GG_PLAN_NT");

	if (nt && nt[0] == '1')
		hipLaunchKernelGGL((k_plan_scan_agg<1>), dim3(pl_grid(p.n)),
				   dim3(PL_THREADS), 0, s, p);
	else
		hipLaunchKernelGGL((k_plan_scan_agg<0>), dim3(pl_grid(p.n)),
				   dim3(PL_THREADS), 0, s, p);
	return hipGetLastError();
}

/* compact used slots into rows of (code, naggs x (lo,hi)) */
__global__ __launch_bounds__(PL_THREADS, 4)
void k_plan_compact(const unsigned long long *__restrict__ tkeys,
		    const unsigned long long *__restrict__ tvals,
		    uint64_t nslots, int naggs,
		    unsigned long long *__restrict__ out,
		    unsigned long long *out_n, uint64_t cap)
{
	const int64_t stride = (int64_t) gridDim.x * blockDim.x;
	const int rowsz = 1 + 2 * naggs;

	for (int64_t s = (int64_t) blockIdx.x * blockDim.x + threadIdx.x;
	     s < (int64_t) nslots; s += stride)
	{
		unsigned long long k = tkeys[s];
		bool take = k != PL_EMPTY;
		unsigned long long mask = __ballot(take);
		int lane = (int) (threadIdx.x & 63);
		unsigned long long base = 0;

		if (mask)
		{
			int leader = __ffsll((long long) mask) - 1;

			if (lane == leader)
				base = atomicAdd(out_n,
						 (unsigned long long)
						 __popcll(mask));
			base = __shfl(base, leader, 64);
		}
		if (!take)
			continue;
		{
			unsigned long long idx = base +
				(unsigned long long) __popcll(
					mask & ((1ull << lane) - 1));

			if (idx >= cap)
				continue;
			{
				unsigned long long *row = out + idx * rowsz;

				row[0] = k;
				for (int a = 0; a < 2 * naggs; a++)
					row[1 + a] =
						tvals[s * 2 * naggs + a];
			}
		}
	}
}

hipError_t launch_plan_compact(hipStream_t s,
			       const unsigned long long *tkeys,
			       const unsigned long long *tvals,
			       uint64_t nslots, int naggs,
			       unsigned long long *out,
			       unsigned long long *out_n, uint64_t cap)
{
	hipLaunchKernelGGL(k_plan_compact, dim3(pl_grid((int64_t) nslots)),
			   dim3(PL_THREADS), 0, s, tkeys, tvals, nslots,
			   naggs, out, out_n, cap);
	return hipGetLastError();
}

}				/* namespace gg */
