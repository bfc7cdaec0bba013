/*
 * Runtime kernel specialization for compiled plans (hipRTC).
 *
 * The interpreted generic kernel (plan.hip) pays for its generality in
 * control flow: objdump shows ~6000 instructions with ~790 branches
 * and ~420 s_waitcnt — per-row divergent walks that keep loads from
 * batching (measured 27 G rows/s on a Q1-shaped plan vs 171 G for the
 * hand-specialized kernel).  At gg_engine_compile_plan time we instead
 * GENERATE a gfx950 kernel with the plan's SHAPE baked in — predicate
 * count and column widths, aggregate expression trees, NULL-flag
 * presence, group mode — and JIT it with hipRTC; bounds/constants stay
 * runtime arguments so the same binary serves every execute.  This is
 * the GPU analog of expression compilation the reference executor
 * never had (its ExecQual is interpreted per row, execQual.c:6260).
 *
 * Fallback: any RTC failure leaves the pipeline on the interpreted
 * kernel (stat row path_plan_interp); GG_PLAN_RTC=0 disables JIT.
 */
#include <hip/hiprtc.h>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <string>
#include <vector>

#include "engine_internal.h"

namespace gg
{

struct PlanRtc
{
	hipModule_t mod = nullptr;
	hipFunction_t fn = nullptr;

	~PlanRtc()
	{
		if (mod)
			(void) hipModuleUnload(mod);
	}
};

/* device-side struct definitions, embedded verbatim so the generated
 * source sees the exact layouts engine_internal.h defines (guarded by
 * a sizeof static_assert) */
static const char *STRUCT_DEFS = R"GG(
typedef long long int64_t;
typedef unsigned long long uint64_t;
typedef unsigned int uint32_t;
typedef int int32_t;
typedef unsigned char uint8_t;
typedef signed char int8_t;

struct PlanPredDev
{
	const void *col;
	const uint8_t *nulls;
	int width;
	int64_t lo, hi;
};

struct PlanJoinDev
{
	const void *pkey;
	const uint8_t *pnulls;
	int width;
	const unsigned long long *bits;
	int64_t dlen;
	const unsigned long long *hkeys;
	uint64_t hslots;
};

struct PlanAggDev
{
	int kind;
	int nf;
	const void *col[3];
	const uint8_t *nulls[3];
	int width[3];
	int8_t mod[3];
};

struct PlanDev
{
	int64_t n;
	int npreds, njoins, naggs, ngroup;
	PlanPredDev preds[8];
	PlanJoinDev joins[2];
	const void *gcol[2];
	const uint8_t *gnulls[2];
	int gwidth[2];
	unsigned long long *tkeys;
	unsigned long long *tvals;
	uint64_t nslots;
	unsigned long long *err;
	PlanAggDev aggs[8];
};

/* reference hashint8 (hashfunc.c:52 -> hash_any 32-bit path) */
__device__ inline uint32_t gg_hash_uint32(uint32_t k)
{
	uint32_t a = 0x9e3779b9u + 4 + 3923095u;
	uint32_t b = a, c = a;

	a += k;
	c ^= b; c -= ((b << 14) | (b >> 18));
	a ^= c; a -= ((c << 11) | (c >> 21));
	b ^= a; b -= ((a << 25) | (a >> 7));
	c ^= b; c -= ((b << 16) | (b >> 16));
	a ^= c; a -= ((c << 4) | (c >> 28));
	b ^= a; b -= ((a << 14) | (a >> 18));
	c ^= b; c -= ((b << 24) | (b >> 8));
	return c;
}

__device__ inline uint32_t gg_hashint8(int64_t v)
{
	uint32_t lohalf = (uint32_t) v;
	uint32_t hihalf = (uint32_t) ((uint64_t) v >> 32);

	lohalf ^= (v >= 0) ? hihalf : ~hihalf;
	return gg_hash_uint32(lohalf);
}
)GG";

/* emit a typed, width-baked load; nt = nontemporal streaming load
 * (k_q1_agg measured +11% from NT on its scan columns — the data is
 * touched once per pass, so L2 retention only evicts useful lines) */
static void emit_ld(std::string &s, const char *dst, const char *ptr,
		    int width, const char *idx, bool nt = false)
{
	char buf[256];
	const char *ty = width == 1 ? "const uint8_t *"
		: width == 4 ? "const int32_t *" : "const int64_t *";

	if (nt && width != 1)
		std::snprintf(buf, sizeof(buf),
			      "\t\tint64_t %s = (int64_t) "
			      "__builtin_nontemporal_load(&((%s) %s)[%s]);\n",
			      dst, ty, ptr, idx);
	else
		std::snprintf(buf, sizeof(buf),
			      "\t\tint64_t %s = (int64_t) ((%s) %s)[%s];\n",
			      dst, ty, ptr, idx);
	s += buf;
}

/* generate + compile; returns GG_OK with *out set, or non-OK (caller
 * falls back to the interpreted kernel) */
gg_status
plan_rtc_compile(const PlanDev &D, bool has_gnull0, bool has_gnull1,
		 std::shared_ptr<void> *out, const long long *bake,
		 int nbake, bool fast)
{
	const char *dis = getenv("GG_PLAN_RTC");

	if (dis && dis[0] == '0')
		return fail(GG_ENOTSUP, "rtc disabled");
	if (nbake <= 0)
		fast = false;	/* the u64 fast tier exists only baked */

	/* NT loads: measured FLAT while the row loop carried the dead
	 * fallback (4.90 vs 4.96 ms, r02k) but +12% once the loop got
	 * lean (4.20 -> 3.76 ms, r02nt2) — same physics as k_q1_agg's
	 * +11%.  Default ON for baked kernels; GG_PLAN_RTC_NT=0 reverts. */
	const char *ntv = getenv("GG_PLAN_RTC_NT");
	bool nt = nbake > 0 && !(ntv && ntv[0] == '0');

	std::string s = STRUCT_DEFS;
	char buf[512];

	/* baked variant: more replicas by default — the accumulator
	 * array has only nbake slots, so per-word contention is higher
	 * and LDS is cheap (k_q1_agg makes the same trade) */
	int lrepl = nbake > 0 ? 16 : 8;
	const char *lr = getenv("GG_PLAN_LREPL");

	if (lr && atoi(lr) >= 1 && atoi(lr) <= 32)
		lrepl = atoi(lr);
	/* LDS budget: LREPL x slots x 2*naggs u64 must stay within
	 * the per-block share; shrink replicas for wide agg lists */
	{
		int slots = nbake > 0 ? nbake : 32;

		while (lrepl > 1 &&
		       (size_t) lrepl * slots * 2 * D.naggs * 8 > 64 * 1024)
			lrepl /= 2;
	}
	std::snprintf(buf, sizeof(buf),
		      "static_assert(sizeof(PlanDev) == %zu, \"layout\");\n"
		      "#define LSLOTS 32\n#define LREPL %d\n"
		      "#define LEMPTY ((long long) 0x8000000000000000ull)\n",
		      sizeof(PlanDev), lrepl);
	s += buf;
	/* fast (bounds-proven) aggregate values are exact in modular u64
	 * arithmetic — the true per-row value is non-negative < 2^61, so
	 * wrapping intermediates preserve the result and the int128
	 * multiplies (3-4 u64 muls each) collapse to one mul per factor */
	s += fast ? "#define VAT unsigned long long\n"
		  : "#define VAT __int128\n";
	{
		bool nn = true;

		for (int a = 0; a < D.naggs; a++)
			for (int f = 0; f < D.aggs[a].nf; f++)
				if (D.aggs[a].nulls[f])
					nn = false;
		/* strict-transition flags vanish at compile time when
		 * no aggregate input is nullable (the common case) */
		s += nn ? "#define AOK(a) true\n"
			: "#define AOK(a) (aok[a])\n";
	}

	{
		/* waves/SIMD occupancy hint; 4 matches the hand kernels,
		 * sweepable (GG_PLAN_WAVES) — fewer waves = more VGPRs
		 * per wave for the register-heavy baked tier */
		int waves = 4;
		const char *wv = getenv("GG_PLAN_WAVES");

		if (wv && atoi(wv) >= 1 && atoi(wv) <= 8)
			waves = atoi(wv);
		std::snprintf(buf, sizeof(buf),
			      "extern \"C\" __global__ "
			      "__launch_bounds__(256, %d)\n"
			      "void plan_kernel(PlanDev P)\n{\n"
			      "\tconst int64_t stride = "
			      "(int64_t) gridDim.x * blockDim.x;\n", waves);
		s += buf;
	}

	/* ---- per-row evaluation as a macro-free inline sequence ---- */
	std::string rowfn;

	rowfn += "\tauto row_pass = [&](int64_t i) -> bool\n\t{\n";
	for (int p = 0; p < D.npreds; p++)
	{
		char d[32], ptr[64];

		std::snprintf(d, sizeof(d), "v%d", p);
		std::snprintf(ptr, sizeof(ptr), "P.preds[%d].col", p);
		emit_ld(rowfn, d, ptr, D.preds[p].width, "i", nt);
		if (D.preds[p].nulls)
		{
			std::snprintf(buf, sizeof(buf),
				      "\t\tif (P.preds[%d].nulls[i]) return false;\n", p);
			rowfn += buf;
		}
		std::snprintf(buf, sizeof(buf),
			      "\t\tif (v%d < P.preds[%d].lo || v%d >= P.preds[%d].hi) return false;\n",
			      p, p, p, p);
		rowfn += buf;
	}
	for (int j = 0; j < D.njoins; j++)
	{
		if (D.joins[j].pnulls)
		{
			std::snprintf(buf, sizeof(buf),
				      "\t\tif (P.joins[%d].pnulls[i]) return false;\n", j);
			rowfn += buf;
		}
		char d[32], ptr[64];

		std::snprintf(d, sizeof(d), "k%d", j);
		std::snprintf(ptr, sizeof(ptr), "P.joins[%d].pkey", j);
		emit_ld(rowfn, d, ptr, D.joins[j].width, "i");
		if (D.joins[j].bits)
		{
			std::snprintf(buf, sizeof(buf),
				      "\t\tif (k%d < 0 || k%d >= P.joins[%d].dlen ||\n"
				      "\t\t    !((P.joins[%d].bits[k%d >> 6] >> (k%d & 63)) & 1))\n"
				      "\t\t\treturn false;\n", j, j, j, j, j, j);
			rowfn += buf;
		}
		else
		{
			std::snprintf(buf, sizeof(buf),
				      "\t\t{\n"
				      "\t\t\tunsigned long long t = (unsigned long long) k%d ^ 0x8000000000000000ull;\n"
				      "\t\t\tuint64_t pos = (uint64_t) gg_hashint8(k%d) & (P.joins[%d].hslots - 1);\n"
				      "\t\t\tfor (;;) {\n"
				      "\t\t\t\tunsigned long long cur = P.joins[%d].hkeys[pos];\n"
				      "\t\t\t\tif (cur == t) break;\n"
				      "\t\t\t\tif (cur == 0) return false;\n"
				      "\t\t\t\tpos = (pos + 1) & (P.joins[%d].hslots - 1);\n"
				      "\t\t\t}\n\t\t}\n", j, j, j, j, j);
			rowfn += buf;
		}
	}
	rowfn += "\t\treturn true;\n\t};\n";
	s += rowfn;

	/* agg input values.  Factor-column loads are DEDUPLICATED at
	 * generation time: the same device pointer can appear in many
	 * aggregates (Q1 reads price in three sums) and clang cannot
	 * prove P.aggs[i].col[f] aliases across aggs, so without this
	 * the kernel issued 13 per-row VMEM loads where the
	 * hand-written k_q1_agg issues 7 (measured via offline ISA
	 * dump, tools/rtc_dump_local.cpp). */
	std::string aggfn = "\tauto agg_vals = [&](int64_t i, VAT *av, bool *aok)\n\t{\n";
	std::vector<std::pair<const void *, int>> uniq;
	auto var_of = [&](int a, int f) -> int
	{
		const void *c = D.aggs[a].col[f];
		int w = D.aggs[a].width[f];

		for (size_t u = 0; u < uniq.size(); u++)
			if (uniq[u].first == c && uniq[u].second == w)
				return (int) u;
		uniq.push_back({c, w});
		{
			char d[32], ptr[64];

			std::snprintf(d, sizeof(d), "xu%d",
				      (int) uniq.size() - 1);
			std::snprintf(ptr, sizeof(ptr),
				      "P.aggs[%d].col[%d]", a, f);
			emit_ld(aggfn, d, ptr, w, "i", nt);
		}
		return (int) uniq.size() - 1;
	};

	/* pass 1: one load per distinct (pointer, width) */
	for (int a = 0; a < D.naggs; a++)
		if (D.aggs[a].kind == 2)
			for (int f = 0; f < D.aggs[a].nf; f++)
				(void) var_of(a, f);
	/* pass 1b: each distinct (var, 100±) term once — (100-disc)
	 * feeds two of Q1's sums */
	{
		std::vector<std::pair<int, int>> mods_done;

		for (int a = 0; a < D.naggs; a++)
			if (D.aggs[a].kind == 2)
				for (int f = 0; f < D.aggs[a].nf; f++)
				{
					int m = D.aggs[a].mod[f];

					if (!m)
						continue;
					int u = var_of(a, f);
					bool seen = false;

					for (auto &q : mods_done)
						if (q.first == u &&
						    q.second == m)
							seen = true;
					if (seen)
						continue;
					mods_done.push_back({u, m});
					std::snprintf(buf, sizeof(buf),
						      "\t\tint64_t xm%d_%d = 100 %c xu%d;\n",
						      u, m,
						      m == 1 ? '-' : '+', u);
					aggfn += buf;
				}
	}
	/* pass 2: per-agg strict flags + products over the shared vars */
	for (int a = 0; a < D.naggs; a++)
	{
		if (D.aggs[a].kind == 0)
		{
			std::snprintf(buf, sizeof(buf),
				      "\t\tav[%d] = 1; aok[%d] = true;\n", a, a);
			aggfn += buf;
			continue;
		}
		std::snprintf(buf, sizeof(buf), "\t\taok[%d] = true;\n", a);
		aggfn += buf;
		for (int f = 0; f < D.aggs[a].nf; f++)
			if (D.aggs[a].nulls[f])
			{
				std::snprintf(buf, sizeof(buf),
					      "\t\tif (P.aggs[%d].nulls[%d][i]) aok[%d] = false;\n",
					      a, f, a);
				aggfn += buf;
			}
		if (D.aggs[a].kind == 1)
		{
			std::snprintf(buf, sizeof(buf), "\t\tav[%d] = 1;\n", a);
			aggfn += buf;
			continue;
		}
		std::snprintf(buf, sizeof(buf), "\t\t{ VAT v = 1;\n");
		aggfn += buf;
		for (int f = 0; f < D.aggs[a].nf; f++)
		{
			int u = var_of(a, f);

			if (D.aggs[a].mod[f])
				std::snprintf(buf, sizeof(buf),
					      "\t\tv *= (VAT) xm%d_%d;\n",
					      u, (int) D.aggs[a].mod[f]);
			else
				std::snprintf(buf, sizeof(buf),
					      "\t\tv *= (VAT) xu%d;\n", u);
			aggfn += buf;
		}
		std::snprintf(buf, sizeof(buf), "\t\tav[%d] = v; }\n", a);
		aggfn += buf;
	}
	aggfn += "\t};\n";
	s += aggfn;

	/* NAGGS as a real constant for array bounds */
	s += std::string("\tconstexpr int NA = ") +
		std::to_string(D.naggs) + ";\n";

	if (D.ngroup == 0)
	{
		s += R"GG(
	unsigned long long alo[NA] = {};
	long long ahi[NA] = {};

	for (int64_t i = (int64_t) blockIdx.x * blockDim.x + threadIdx.x;
	     i < P.n; i += stride)
	{
		if (!row_pass(i))
			continue;
		__int128 av[NA];
		bool aok[NA];

		agg_vals(i, av, aok);
		for (int a = 0; a < NA; a++)
		{
			if (!AOK(a))
				continue;
			unsigned long long vlo = (unsigned long long) av[a];
			unsigned long long old = alo[a];

			alo[a] += vlo;
			ahi[a] += (long long) (av[a] >> 64) + (alo[a] < old);
		}
	}
	for (int a = 0; a < NA; a++)
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
			unsigned long long old =
				atomicAdd(&P.tvals[2 * a], alo[a]);
			unsigned long long hi = (unsigned long long) ahi[a] +
				(old + alo[a] < old ? 1ull : 0ull);

			if (hi)
				atomicAdd(&P.tvals[2 * a + 1], hi);
		}
	}
	if (blockIdx.x == 0 && threadIdx.x == 0)
		P.tkeys[0] = 0;
}
)GG";
	}
	else
	{
		/* group code expression */
		std::string gcode;

		if (D.ngroup == 2)
		{
			gcode = "\t\tlong long e0 = ";
			gcode += has_gnull0 ? "(P.gnulls[0][i] ? 256 : (long long) ((const uint8_t *) P.gcol[0])[i]);\n"
					    : "(long long) ((const uint8_t *) P.gcol[0])[i];\n";
			gcode += "\t\tlong long e1 = ";
			gcode += has_gnull1 ? "(P.gnulls[1][i] ? 256 : (long long) ((const uint8_t *) P.gcol[1])[i]);\n"
					    : "(long long) ((const uint8_t *) P.gcol[1])[i];\n";
			gcode += "\t\tlong long code = e0 * 512 + e1;\n";
		}
		else
		{
			const char *ld = D.gwidth[0] == 1
				? "(long long) ((const uint8_t *) P.gcol[0])[i]"
				: D.gwidth[0] == 4
				? "(long long) ((const int32_t *) P.gcol[0])[i]"
				: "((const int64_t *) P.gcol[0])[i]";

			gcode = "\t\tlong long code = ";
			if (has_gnull0)
			{
				gcode += "P.gnulls[0][i] ? (long long) "
					 "0x8000000000000001ull : (";
				gcode += ld;
				gcode += ");\n";
			}
			else
			{
				gcode += ld;
				gcode += ";\n";
			}
		}

		if (nbake > 0)
		{
			/* ---- baked-codes variant ----------------------
			 * The group set was observed on a prior execute
			 * of this plan (tables are immutable, so it is
			 * exact for repeats).  Codes become a constexpr
			 * compare chain -> direct LDS index: no key
			 * probe, no CAS on the hot path.  A row whose
			 * code is not baked (impossible for repeats,
			 * kept for correctness) takes the global-table
			 * path, and group PRESENCE is tracked in a
			 * per-thread bit mask so groups whose sums are
			 * all zero still materialize. */
			s += "\tconstexpr int NG = " +
				std::to_string(nbake) + ";\n"
				"\tconstexpr long long GC[NG] = {";
			for (int q = 0; q < nbake; q++)
			{
				std::snprintf(buf, sizeof(buf), "%s%lldll",
					      q ? ", " : "", bake[q]);
				s += buf;
			}
			s += "};\n";
			/* LW = u64 words per accumulator.  fast (=proven
			 * by plan.cpp's interval bound: every agg value
			 * non-negative and bound*rows_per_block < 2^61)
			 * uses ONE word and fire-and-forget LDS adds —
			 * the returned-value carry check otherwise makes
			 * every add a RETURNING atomic, which serializes
			 * at the ~88 returning-atomics/us hot-word wall
			 * (measured: baked 2-word = generic 5.5 ms;
			 * k_q1_agg's budget-proven 1-word form = 3.6 ms
			 * on the same shape). */
			s += fast ? "#define LW 1\n" : "#define LW 2\n";
			/* odd per-replica stride: spreads replicas across
			 * LDS banks (k_q1_agg's Q1_STRIDE trick) */
			s += "#define LPAD ((NG * LW * NA) | 1)\n";
			s += R"GG(
	__shared__ unsigned long long lvals[LREPL * LPAD];
	__shared__ unsigned int btouch;

	for (int q = threadIdx.x; q < LREPL * LPAD; q += blockDim.x)
		lvals[q] = 0;
	if (threadIdx.x == 0)
		btouch = 0;
	__syncthreads();
	const int lrep = (int) (threadIdx.x & (LREPL - 1));
	unsigned int tmask = 0;
	unsigned int nmiss = 0;

	auto gslot = [&](long long code) -> int64_t
	{
		uint64_t pos = (uint64_t) gg_hashint8(code) & (P.nslots - 1);

		for (uint64_t it = 0; it < P.nslots; it++)
		{
			unsigned long long cur = P.tkeys[pos];

			if (cur == (unsigned long long) code)
				return (int64_t) pos;
			if (cur == 0x8000000000000000ull)
			{
				unsigned long long prev = atomicCAS(
					&P.tkeys[pos], 0x8000000000000000ull,
					(unsigned long long) code);

				if (prev == 0x8000000000000000ull ||
				    prev == (unsigned long long) code)
					return (int64_t) pos;
				continue;
			}
			pos = (pos + 1) & (P.nslots - 1);
		}
		return -1;
	};

	for (int64_t i = (int64_t) blockIdx.x * blockDim.x + threadIdx.x;
	     i < P.n; i += stride)
	{
		if (!row_pass(i))
			continue;
)GG";
			s += gcode;
			s += R"GG(
		VAT av[NA];
		bool aok[NA];

		agg_vals(i, av, aok);
		int g = -1;
#pragma unroll
		for (int q = 0; q < NG; q++)
			if (code == GC[q])
				g = q;
		if (g >= 0)
		{
			unsigned long long *mine = &lvals[lrep * LPAD];

			tmask |= 1u << g;
			for (int a = 0; a < NA; a++)
			{
				if (!AOK(a))
					continue;
#if LW == 1
				atomicAdd(&mine[g * NA + a],
					  (unsigned long long) av[a]);
#else
				unsigned long long vlo =
					(unsigned long long) av[a];
				unsigned long long vhi =
					(unsigned long long) (av[a] >> 64);
				unsigned long long old = atomicAdd(
					&mine[(g * NA + a) * 2], vlo);

				if (old + vlo < old)
					vhi++;
				if (vhi)
					atomicAdd(&mine[(g * NA + a) * 2 + 1],
						  vhi);
#endif
			}
		}
		else
			/* a code outside the baked set is impossible on
			 * re-execution over immutable tables; if it ever
			 * happens, flag it and let the host re-run the
			 * generic kernel (keeping the full global-table
			 * path INSIDE this loop cost ~35% extra
			 * instructions for a dead branch) */
			nmiss++;
	}
	if (nmiss)
		atomicOr(P.err, 2ull);
	if (tmask)
		atomicOr(&btouch, tmask);
	__syncthreads();
	for (int q = threadIdx.x; q < NG; q += blockDim.x)
	{
		if (!((btouch >> q) & 1u))
			continue;
		int64_t slot = gslot(GC[q]);

		if (slot < 0) { atomicOr(P.err, 1ull); continue; }
		for (int a = 0; a < NA; a++)
		{
#if LW == 1
			unsigned long long lo = 0, hi = 0;

			for (int rr = 0; rr < LREPL; rr++)
				lo += lvals[rr * LPAD + q * NA + a];
#else
			unsigned long long lo = 0, hi = 0;

			for (int rr = 0; rr < LREPL; rr++)
			{
				unsigned long long rl =
					lvals[rr * LPAD + (q * NA + a) * 2];
				unsigned long long o = lo;

				lo += rl;
				hi += lvals[rr * LPAD +
					    (q * NA + a) * 2 + 1] + (lo < o);
			}
#endif
			if (!lo && !hi)
				continue;
			unsigned long long old =
				atomicAdd(&P.tvals[(slot * NA + a) * 2], lo);

			if (old + lo < old)
				hi++;
			if (hi)
				atomicAdd(&P.tvals[(slot * NA + a) * 2 + 1],
					  hi);
		}
	}
}
)GG";
			goto compile;
		}
		s += R"GG(
	__shared__ long long lkeys[LSLOTS];
	__shared__ unsigned long long lvals[LREPL][LSLOTS][2 * NA];

	for (int q = threadIdx.x; q < LSLOTS; q += blockDim.x)
		lkeys[q] = LEMPTY;
	for (int q = threadIdx.x; q < LREPL * LSLOTS * 2 * NA;
	     q += blockDim.x)
		((unsigned long long *) lvals)[q] = 0;
	__syncthreads();
	const int lrep = (int) (threadIdx.x & (LREPL - 1));

	auto gslot = [&](long long code) -> int64_t
	{
		uint64_t pos = (uint64_t) gg_hashint8(code) & (P.nslots - 1);

		for (uint64_t it = 0; it < P.nslots; it++)
		{
			unsigned long long cur = P.tkeys[pos];

			if (cur == (unsigned long long) code)
				return (int64_t) pos;
			if (cur == 0x8000000000000000ull)
			{
				unsigned long long prev = atomicCAS(
					&P.tkeys[pos], 0x8000000000000000ull,
					(unsigned long long) code);

				if (prev == 0x8000000000000000ull ||
				    prev == (unsigned long long) code)
					return (int64_t) pos;
				continue;
			}
			pos = (pos + 1) & (P.nslots - 1);
		}
		return -1;
	};

	for (int64_t i = (int64_t) blockIdx.x * blockDim.x + threadIdx.x;
	     i < P.n; i += stride)
	{
		if (!row_pass(i))
			continue;
)GG";
		s += gcode;
		s += R"GG(
		__int128 av[NA];
		bool aok[NA];

		agg_vals(i, av, aok);
		int ls = -1;
		uint32_t pos = (uint32_t) gg_hashint8(code) & (LSLOTS - 1);

		for (int probe = 0; probe < 8; probe++)
		{
			long long cur = lkeys[pos];

			if (cur == code) { ls = (int) pos; break; }
			if (cur == LEMPTY)
			{
				long long prev = atomicCAS(
					(unsigned long long *) &lkeys[pos],
					(unsigned long long) LEMPTY,
					(unsigned long long) code);

				if (prev == LEMPTY || prev == code)
				{ ls = (int) pos; break; }
				continue;
			}
			pos = (pos + 1) & (LSLOTS - 1);
		}
		if (ls >= 0)
		{
			for (int a = 0; a < NA; a++)
			{
				if (!AOK(a))
					continue;
				unsigned long long vlo =
					(unsigned long long) av[a];
				unsigned long long vhi =
					(unsigned long long) (av[a] >> 64);
				unsigned long long old = atomicAdd(
					&lvals[lrep][ls][2 * a], vlo);

				if (old + vlo < old)
					vhi++;
				if (vhi)
					atomicAdd(&lvals[lrep][ls][2 * a + 1],
						  vhi);
			}
		}
		else
		{
			int64_t slot = gslot(code);

			if (slot < 0) { atomicOr(P.err, 1ull); continue; }
			for (int a = 0; a < NA; a++)
			{
				if (!AOK(a))
					continue;
				unsigned long long vlo =
					(unsigned long long) av[a];
				unsigned long long vhi =
					(unsigned long long) (av[a] >> 64);
				unsigned long long old = atomicAdd(
					&P.tvals[(slot * NA + a) * 2], vlo);

				if (old + vlo < old)
					vhi++;
				if (vhi)
					atomicAdd(&P.tvals[(slot * NA + a) * 2 + 1],
						  vhi);
			}
		}
	}
	__syncthreads();
	for (int q = threadIdx.x; q < LSLOTS; q += blockDim.x)
	{
		if (lkeys[q] == LEMPTY)
			continue;
		int64_t slot = gslot(lkeys[q]);

		if (slot < 0) { atomicOr(P.err, 1ull); continue; }
		for (int a = 0; a < NA; a++)
		{
			unsigned long long lo = 0, hi = 0;

			for (int rr = 0; rr < LREPL; rr++)
			{
				unsigned long long rl = lvals[rr][q][2 * a];
				unsigned long long o = lo;

				lo += rl;
				hi += lvals[rr][q][2 * a + 1] + (lo < o);
			}
			if (!lo && !hi)
				continue;
			unsigned long long old =
				atomicAdd(&P.tvals[(slot * NA + a) * 2], lo);

			if (old + lo < old)
				hi++;
			if (hi)
				atomicAdd(&P.tvals[(slot * NA + a) * 2 + 1],
					  hi);
		}
	}
}
)GG";
	}

	/* ---- compile ---- */
compile:;
	const char *dump = getenv("GG_PLAN_RTC_DUMP");

	if (dump && dump[0])
	{
		FILE *f = fopen(dump, nbake > 0 ? "a" : "w");

		if (f)
		{
			fprintf(f, "/* ==== nbake=%d ==== */\n%s",
				nbake, s.c_str());
			fclose(f);
		}
	}
	hiprtcProgram prog;

	if (hiprtcCreateProgram(&prog, s.c_str(), "gg_plan.cu", 0, nullptr,
				nullptr) != HIPRTC_SUCCESS)
		return fail(GG_ENOTSUP, "hiprtcCreateProgram failed");
	const char *opts[] = {"--offload-arch=gfx950", "-O3", "-std=c++17"};
	hiprtcResult cr = hiprtcCompileProgram(prog, 3, opts);

	if (cr != HIPRTC_SUCCESS)
	{
		size_t ls = 0;

		hiprtcGetProgramLogSize(prog, &ls);
		std::string log(ls, 0);
		hiprtcGetProgramLog(prog, &log[0]);
		hiprtcDestroyProgram(&prog);
		return fail(GG_ENOTSUP, "plan rtc compile failed: %.300s",
			    log.c_str());
	}
	size_t csz = 0;

	hiprtcGetCodeSize(prog, &csz);
	std::vector<char> code(csz);
	hiprtcGetCode(prog, code.data());
	hiprtcDestroyProgram(&prog);

	auto rtc = std::make_shared<PlanRtc>();

	if (hipModuleLoadData(&rtc->mod, code.data()) != hipSuccess)
		return fail(GG_ENOTSUP, "plan rtc module load failed");
	if (hipModuleGetFunction(&rtc->fn, rtc->mod, "plan_kernel")
	    != hipSuccess)
		return fail(GG_ENOTSUP, "plan rtc function lookup failed");
	*out = rtc;
	return GG_OK;
}

gg_status
plan_rtc_launch(hipStream_t s, const std::shared_ptr<void> &h, PlanDev P,
		int grid, int block)
{
	PlanRtc *rtc = (PlanRtc *) h.get();
	size_t size = sizeof(P);
	void *config[] = {HIP_LAUNCH_PARAM_BUFFER_POINTER, &P,
			  HIP_LAUNCH_PARAM_BUFFER_SIZE, &size,
			  HIP_LAUNCH_PARAM_END};

	GG_HIP(hipModuleLaunchKernel(rtc->fn, grid, 1, 1, block, 1, 1, 0, s,
				     nullptr, config));
	return GG_OK;
}

}				/* namespace gg */
