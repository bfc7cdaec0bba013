/*
 * Generic compiled-plan host side: gg_engine_compile_plan resolves a
 * gg_plan_desc (engine_abi.h "generalized pipeline descriptor")
 * against registered device columns; exec_plan runs the generic
 * build/scan-agg/compact kernels (plan.hip) and combines partial
 * group states across segments exactly like the reference's two-stage
 * aggregation (cdbgroup.c:1245 / nodeAgg.c:2130–2142: exact partials,
 * one combine).
 *
 * Also here: gg_engine_table_set_nulls (nullable scan surface) and
 * gg_engine_hash_groupby_i64_n (NULL-aware general group-by,
 * execHHashagg.c:531 NULL-key grouping + nodeAgg.c:413 strict
 * transitions).
 */
#include <algorithm>
#include <cstring>
#include <map>
#include <vector>

#include "engine_internal.h"

namespace gg
{

gg_status comm_allgather_u64(const void *dev_send, void *dev_recv,
			     size_t count);
bool comm_ready();

#define GG_TRY(expr) \
	do { gg_status _s = (expr); if (_s != GG_OK) return _s; } while (0)

typedef __int128 i128;

struct PlanResolved
{
	PlanDev dev;		/* device pointers resolved; table ptrs live
				 * in the engine registry */
	struct BJoin
	{
		PlanBuildDev bd;	/* preds/key resolved; bits/hkeys
					 * filled per execute (scratch) */
		int64_t dlen;		/* >0: dense bitmap path */
		uint64_t hslots;	/* else: hash set */
	};
	std::vector<BJoin> joins;
	int64_t scan_rows = 0;
	int64_t pred_bytes_per_row = 0;
	/* hipRTC-specialized kernel (nullptr = interpreted fallback);
	 * compiled lazily on first execute, when the join structures'
	 * shape (bitmap vs hash set) is known */
	std::shared_ptr<void> rtc;
	bool rtc_tried = false;
	/* second-stage specialization: after the first execute the
	 * (small) group set is known exactly — tables are immutable —
	 * and gets baked into a direct-indexed kernel */
	std::shared_ptr<void> rtc_baked;
	bool bake_tried = false;
};

static int coltype_width(gg_coltype t)
{
	switch (t)
	{
		case GG_COL_CHAR1:
			return 1;
		case GG_COL_INT32:
			return 4;
		default:
			return 8;
	}
}

static gg_status
resolve_pred(Table *t, const gg_plan_pred *in, PlanPredDev *out)
{
	Table::Col *c = t->find(in->col ? in->col : "");

	if (!c)
		return fail(GG_EINVAL, "plan: no column '%s' in table '%s'",
			    in->col ? in->col : "(null)", t->name.c_str());
	out->col = c->dev;
	out->nulls = (const uint8_t *) c->nulls;
	out->width = coltype_width(c->type);
	out->lo = in->lo;
	out->hi = in->hi;
	return GG_OK;
}

extern "C" gg_status
gg_engine_compile_plan(const gg_plan_desc *d, gg_pipeline *out)
{
	Engine &e = engine();

	if (!e.inited)
		return fail(GG_ESTATE, "engine not initialized");
	if (!d || !out)
		return fail(GG_EINVAL, "null plan args");
	Table *scan = engine_table(d->scan.table);

	if (!scan)
		return fail(GG_EINVAL, "plan: bad driving table handle");
	if (d->scan.npreds < 0 || d->scan.npreds > GG_PLAN_MAX_PREDS ||
	    d->njoins < 0 || d->njoins > GG_PLAN_MAX_JOINS ||
	    d->naggs < 1 || d->naggs > GG_PLAN_MAX_AGGS ||
	    d->ngroup < 0 || d->ngroup > 2)
		return fail(GG_ENOTSUP, "plan: shape out of v1 bounds "
			    "(fall back to standard_ExecutorRun)");

	auto pr = std::make_shared<PlanResolved>();
	PlanResolved *R = pr.get();

	std::memset(&R->dev, 0, sizeof(R->dev));
	R->dev.n = scan->nrows;
	R->scan_rows = scan->nrows;
	R->dev.npreds = d->scan.npreds;
	for (int i = 0; i < d->scan.npreds; i++)
	{
		gg_status st = resolve_pred(scan, &d->scan.preds[i],
					    &R->dev.preds[i]);

		if (st != GG_OK)
			return st;
		R->pred_bytes_per_row += R->dev.preds[i].width;
	}

	R->dev.njoins = d->njoins;
	for (int j = 0; j < d->njoins; j++)
	{
		const gg_plan_join *J = &d->joins[j];
		Table *bt = engine_table(J->build.table);

		if (!bt)
			return fail(GG_EINVAL, "plan: bad build table (join %d)",
				    j);
		if (J->build.npreds < 0 ||
		    J->build.npreds > GG_PLAN_MAX_PREDS)
			return fail(GG_ENOTSUP, "plan: join %d pred count", j);
		Table::Col *bk = bt->find(J->build_key ? J->build_key : "");
		Table::Col *pk = scan->find(J->probe_key ? J->probe_key : "");

		if (!bk || !pk)
			return fail(GG_EINVAL,
				    "plan: join %d key column missing", j);
		PlanResolved::BJoin bj{};

		bj.bd.key = bk->dev;
		bj.bd.knulls = (const uint8_t *) bk->nulls;
		bj.bd.kw = coltype_width(bk->type);
		bj.bd.n = bt->nrows;
		bj.bd.npreds = J->build.npreds;
		for (int i = 0; i < J->build.npreds; i++)
		{
			gg_status st = resolve_pred(bt, &J->build.preds[i],
						    &bj.bd.preds[i]);

			if (st != GG_OK)
				return st;
		}
		R->joins.push_back(bj);
		R->dev.joins[j].pkey = pk->dev;
		R->dev.joins[j].pnulls = (const uint8_t *) pk->nulls;
		R->dev.joins[j].width = coltype_width(pk->type);
		R->pred_bytes_per_row += R->dev.joins[j].width;
	}

	R->dev.ngroup = d->ngroup;
	for (int g = 0; g < d->ngroup; g++)
	{
		Table::Col *c = scan->find(d->group_cols[g]
					   ? d->group_cols[g] : "");

		if (!c)
			return fail(GG_EINVAL, "plan: group column missing");
		if (d->ngroup == 2 && c->type != GG_COL_CHAR1)
			return fail(GG_ENOTSUP,
				    "plan: 2-column group keys must be char1");
		R->dev.gcol[g] = c->dev;
		R->dev.gnulls[g] = (const uint8_t *) c->nulls;
		R->dev.gwidth[g] = coltype_width(c->type);
		R->pred_bytes_per_row += R->dev.gwidth[g];
	}

	R->dev.naggs = d->naggs;
	for (int a = 0; a < d->naggs; a++)
	{
		const gg_plan_agg *A = &d->aggs[a];
		PlanAggDev *D = &R->dev.aggs[a];

		if (A->kind < 0 || A->kind > 2)
			return fail(GG_ENOTSUP, "plan: agg %d kind", a);
		if (A->kind == GG_AGG_COUNT_STAR)
			D->nf = 0;
		else if (A->nfactors < 1 || A->nfactors > 3 ||
			 (A->kind == GG_AGG_COUNT_COL && A->nfactors != 1))
			return fail(GG_ENOTSUP, "plan: agg %d factors", a);
		else
			D->nf = A->nfactors;
		D->kind = A->kind;
		for (int f = 0; f < D->nf; f++)
		{
			Table::Col *c = scan->find(A->col[f] ? A->col[f]
						   : "");

			if (!c)
				return fail(GG_EINVAL,
					    "plan: agg %d column missing", a);
			if (A->mod[f] < 0 || A->mod[f] > 2)
				return fail(GG_ENOTSUP,
					    "plan: agg %d factor mod", a);
			D->col[f] = c->dev;
			D->nulls[f] = (const uint8_t *) c->nulls;
			D->width[f] = coltype_width(c->type);
			D->mod[f] = A->mod[f];
			R->pred_bytes_per_row += D->width[f];
		}
	}

	Pipeline *p = new Pipeline();

	p->desc = gg_pipeline_desc{};
	p->desc.kind = (gg_pipeline_kind) GG_PIPE_PLAN_INTERNAL;
	p->plan = pr;
	e.pipelines.push_back(p);
	*out = (gg_pipeline) (e.pipelines.size() - 1);
	return GG_OK;
}

static uint64_t next_pow2_pl(uint64_t v)
{
	uint64_t p = 1;

	while (p < v)
		p <<= 1;
	return p;
}

/* nslots for the generic group table: supports up to ~nslots/2 groups */
static constexpr uint64_t PL_NSLOTS = 1ull << 18;
static constexpr unsigned long long PL_EMPTY_HOST = 0x8000000000000000ull;

gg_status
exec_plan(Pipeline *p, void *arena, size_t bytes, size_t *written)
{
	Engine &e = engine();
	PlanResolved *R = (PlanResolved *) p->plan.get();
	int nseg = e.cfg.n_segments;

	if (!R)
		return fail(GG_ESTATE, "not a compiled plan");
	unsigned long long *ctr = (unsigned long long *) p->sget("ctr", 8);

	if (!ctr)
		return fail(GG_ENOMEM, "plan scratch");

	/* 1. build the semi-join structures (sizing guard as in the
	 * named pipelines: dense bitmap iff max(key) <= 8x rows) */
	for (size_t j = 0; j < R->joins.size(); j++)
	{
		PlanResolved::BJoin &B = R->joins[j];
		char nm[32];
		Timed tm(e.stream);

		if (B.bd.kw == 8 && !B.dlen && !B.hslots)
		{
			unsigned long long maxk = 0;

			GG_TRY(engine_cached_max_i64(
				e, p, (const int64_t *) B.bd.key, B.bd.n,
				ctr, &maxk));
			if (B.bd.n > 0 && maxk > 0 &&
			    maxk <= (unsigned long long) (8 * B.bd.n + 16))
				B.dlen = (int64_t) maxk + 1;
		}
		if (!B.dlen && !B.hslots)
			B.hslots = next_pow2_pl(2 * (uint64_t) B.bd.n + 2);
		std::snprintf(nm, sizeof(nm), "plan.j%zu", j);
		if (B.dlen)
		{
			size_t words = (size_t) (B.dlen / 64 + 2);
			unsigned long long *bits = (unsigned long long *)
				p->sget(nm, words * 8);

			if (!bits)
				return fail(GG_ENOMEM, "plan join bitmap");
			GG_HIP(hipMemsetAsync(bits, 0, words * 8, e.stream));
			B.bd.bits = bits;
			B.bd.dlen = B.dlen;
			B.bd.hkeys = nullptr;
			B.bd.hslots = 0;
		}
		else
		{
			unsigned long long *hk = (unsigned long long *)
				p->sget(nm, B.hslots * 8);

			if (!hk)
				return fail(GG_ENOMEM, "plan join set");
			GG_HIP(hipMemsetAsync(hk, 0, B.hslots * 8, e.stream));
			B.bd.bits = nullptr;
			B.bd.dlen = 0;
			B.bd.hkeys = hk;
			B.bd.hslots = B.hslots;
		}
		GG_HIP(launch_plan_build(e.stream, B.bd));
		{
			double ms = tm.stop();
			KernelStatAcc &st = p->stat("plan_build");

			st.launches++;
			st.total_ms += ms;
			st.rows_in += B.bd.n;
		}
		R->dev.joins[j].bits = B.bd.bits;
		R->dev.joins[j].dlen = B.bd.dlen;
		R->dev.joins[j].hkeys = B.bd.hkeys;
		R->dev.joins[j].hslots = B.bd.hslots;
		p->stat(B.dlen ? "path_plan_join_bitmap"
			: "path_plan_join_hash").launches++;
	}

	/* 2. group table + fused scan */
	uint64_t nslots = R->dev.ngroup == 0 ? 1
		: (R->dev.ngroup == 2 ? 1 << 19 : PL_NSLOTS);
	int naggs = R->dev.naggs;
	unsigned long long *tkeys = (unsigned long long *)
		p->sget("plan.tkeys", nslots * 8);
	unsigned long long *tvals = (unsigned long long *)
		p->sget("plan.tvals", nslots * (size_t) naggs * 16);
	unsigned long long *derr = (unsigned long long *)
		p->sget("plan.err", 8);

	if (!tkeys || !tvals || !derr)
		return fail(GG_ENOMEM, "plan group table");
	{
		Timed tm(e.stream);

		GG_HIP(launch_fill_u64(e.stream, tkeys, nslots,
				       PL_EMPTY_HOST));
		GG_HIP(hipMemsetAsync(tvals, 0,
				      nslots * (size_t) naggs * 16,
				      e.stream));
		GG_HIP(hipMemsetAsync(derr, 0, 8, e.stream));
		R->dev.tkeys = tkeys;
		R->dev.tvals = tvals;
		R->dev.nslots = nslots;
		R->dev.err = derr;
		if (!R->rtc_tried)
		{
			/* specialize now: bitmap-vs-hash join shape is
			 * resolved; NULL presence and widths are baked */
			gg_status rs = plan_rtc_compile(
				R->dev, R->dev.gnulls[0] != nullptr,
				R->dev.gnulls[1] != nullptr, &R->rtc);

			R->rtc_tried = true;
			p->stat(rs == GG_OK ? "path_plan_rtc"
				: "path_plan_interp").launches++;
		}
		if (R->rtc)
		{
			int64_t g = (R->dev.n + 255) / 256;

			if (g > 2048)
				g = 2048;
			if (g < 1)
				g = 1;
			GG_TRY(plan_rtc_launch(
				e.stream,
				R->rtc_baked ? R->rtc_baked : R->rtc,
				R->dev, (int) g, 256));
		}
		else
			GG_HIP(launch_plan_scan_agg(e.stream, R->dev));
		{
			double ms = tm.stop();
			KernelStatAcc &st = p->stat("plan_scan_agg");

			st.launches++;
			st.total_ms += ms;
			st.rows_in += R->scan_rows;
			st.hbm_bytes += R->scan_rows *
				R->pred_bytes_per_row;
		}
	}

	/* 3. compact + host sort */
	uint64_t cap = nslots;
	int rowsz = 1 + 2 * naggs;
	unsigned long long *dout = (unsigned long long *)
		p->sget("plan.out", cap * (size_t) rowsz * 8);
	unsigned long long ng = 0, herr = 0;

	if (!dout)
		return fail(GG_ENOMEM, "plan out");
	GG_HIP(hipMemsetAsync(ctr, 0, 8, e.stream));
	GG_HIP(launch_plan_compact(e.stream, tkeys, tvals, nslots, naggs,
				   dout, ctr, cap));
	GG_HIP(hipStreamSynchronize(e.stream));
	GG_HIP(hipMemcpy(&herr, derr, 8, hipMemcpyDeviceToHost));
	if (herr == 2ull && R->rtc_baked)
	{
		/* the baked kernel saw a group code outside its baked
		 * set (impossible over immutable tables; defensive) —
		 * drop the baked binary and redo this execute on the
		 * generic kernel */
		R->rtc_baked.reset();
		p->stat("path_plan_bake_miss").launches++;
		GG_HIP(launch_fill_u64(e.stream, tkeys, nslots,
				       PL_EMPTY_HOST));
		GG_HIP(hipMemsetAsync(tvals, 0,
				      nslots * (size_t) naggs * 16,
				      e.stream));
		GG_HIP(hipMemsetAsync(derr, 0, 8, e.stream));
		{
			int64_t g = (R->dev.n + 255) / 256;

			if (g > 2048)
				g = 2048;
			if (g < 1)
				g = 1;
			GG_TRY(plan_rtc_launch(e.stream, R->rtc, R->dev,
					       (int) g, 256));
		}
		GG_HIP(hipMemsetAsync(ctr, 0, 8, e.stream));
		GG_HIP(launch_plan_compact(e.stream, tkeys, tvals, nslots,
					   naggs, dout, ctr, cap));
		GG_HIP(hipStreamSynchronize(e.stream));
		GG_HIP(hipMemcpy(&herr, derr, 8, hipMemcpyDeviceToHost));
	}
	if (herr)
		return fail(GG_EINVAL,
			    "plan: group cardinality exceeds %llu "
			    "(generic-path v1 cap)",
			    (unsigned long long) nslots / 2);
	GG_HIP(hipMemcpy(&ng, ctr, 8, hipMemcpyDeviceToHost));
	if (ng > cap)
		return fail(GG_ESTATE, "plan compact overflow");

	std::vector<unsigned long long> rows((size_t) ng * rowsz);

	if (ng)
		GG_HIP(hipMemcpy(rows.data(), dout, rows.size() * 8,
				 hipMemcpyDeviceToHost));

	/* 4. combine partial states across segments (2-stage agg:
	 * exact partials + one combine) */
	if (nseg > 1)
	{
		if (!comm_ready())
			return fail(GG_ESTATE,
				    "multi-segment plan without comm");
		/* max group count, then padded allgather of rows */
		std::vector<unsigned long long> cnts(nseg);
		unsigned long long *g = (unsigned long long *)
			p->sget("plan.cntg", (1 + (size_t) nseg) * 8);

		if (!g)
			return fail(GG_ENOMEM, "plan comm scratch");
		GG_HIP(hipMemcpy(g, &ng, 8, hipMemcpyHostToDevice));
		GG_TRY(comm_allgather_u64(g, g + 1, 1));
		GG_HIP(hipMemcpy(cnts.data(), g + 1, (size_t) nseg * 8,
				 hipMemcpyDeviceToHost));
		uint64_t mx = 0;

		for (int r = 0; r < nseg; r++)
			mx = std::max(mx, (uint64_t) cnts[r]);
		size_t per = (size_t) mx * rowsz;

		if (per)
		{
			unsigned long long *ag = (unsigned long long *)
				p->sget("plan.allg",
					(per + per * (size_t) nseg) * 8);

			if (!ag)
				return fail(GG_ENOMEM, "plan allgather");
			GG_HIP(hipMemsetAsync(ag, 0, per * 8, e.stream));
			if (ng)
				GG_HIP(hipMemcpyAsync(ag, dout,
						      (size_t) ng * rowsz * 8,
						      hipMemcpyDeviceToDevice,
						      e.stream));
			GG_HIP(hipStreamSynchronize(e.stream));
			GG_TRY(comm_allgather_u64(ag, ag + per, per));
			std::vector<unsigned long long> all(per *
							    (size_t) nseg);

			GG_HIP(hipMemcpy(all.data(), ag + per,
					 all.size() * 8,
					 hipMemcpyDeviceToHost));

			std::map<long long,
				 std::vector<i128>> merged;

			for (int r = 0; r < nseg; r++)
				for (uint64_t i = 0; i < cnts[r]; i++)
				{
					const unsigned long long *row =
						&all[per * (size_t) r +
						     i * rowsz];
					auto &acc = merged[(long long)
							   row[0]];

					if (acc.empty())
						acc.assign(naggs, 0);
					for (int a = 0; a < naggs; a++)
						acc[a] += ((i128)
							   (int64_t)
							   row[2 + 2 * a]
							   << 64) |
							(i128) row[1 + 2 * a];
				}
			rows.clear();
			for (auto &kv : merged)
			{
				rows.push_back((unsigned long long)
					       kv.first);
				for (int a = 0; a < naggs; a++)
				{
					rows.push_back((unsigned long long)
						       (uint64_t)
						       kv.second[a]);
					rows.push_back((unsigned long long)
						       (uint64_t)
						       (kv.second[a] >> 64));
				}
			}
			ng = rows.size() / rowsz;
		}
	}

	/* 5. sort by decoded keys and materialize the arena */
	std::vector<size_t> order(ng);

	for (size_t i = 0; i < ng; i++)
		order[i] = i;
	auto keys_of = [&](size_t i, int64_t *k0, int64_t *k1)
	{
		long long code = (long long) rows[i * rowsz];

		if (R->dev.ngroup == 2)
		{
			long long e0 = code / 512, e1 = code % 512;

			*k0 = e0 == 256 ? GG_PLAN_NULL_KEY : e0;
			*k1 = e1 == 256 ? GG_PLAN_NULL_KEY : e1;
		}
		else
		{
			*k0 = code;
			*k1 = 0;
		}
	};
	std::sort(order.begin(), order.end(),
		  [&](size_t a, size_t b)
	{
		int64_t a0, a1, b0, b1;

		keys_of(a, &a0, &a1);
		keys_of(b, &b0, &b1);
		return a0 != b0 ? a0 < b0 : a1 < b1;
	});

	size_t need = 16 + (size_t) ng * (16 + 16 * (size_t) naggs);

	if (bytes < need)
		return fail(GG_EINVAL, "plan arena too small (%zu < %zu)",
			    bytes, need);
	{
		uint8_t *w = (uint8_t *) arena;
		int64_t ng64 = (int64_t) ng;
		int32_t na32 = naggs, ngc32 = R->dev.ngroup;

		std::memcpy(w, &ng64, 8);
		std::memcpy(w + 8, &na32, 4);
		std::memcpy(w + 12, &ngc32, 4);
		w += 16;
		for (size_t oi = 0; oi < ng; oi++)
		{
			size_t i = order[oi];
			int64_t k0, k1;

			keys_of(i, &k0, &k1);
			std::memcpy(w, &k0, 8);
			std::memcpy(w + 8, &k1, 8);
			w += 16;
			std::memcpy(w, &rows[i * rowsz + 1],
				    16 * (size_t) naggs);
			w += 16 * (size_t) naggs;
		}
	}
	*written = need;

	/* second-stage bake: with the group set observed (and exact —
	 * registered tables are immutable), recompile the RTC kernel
	 * with the codes as a constexpr compare chain so later
	 * executes of this plan skip key probing entirely.  Gated on a
	 * small group count (register/LDS budget) and on the first
	 * RTC compile having succeeded. */
	{
		int bake_max = 8;
		const char *bm = getenv("GG_PLAN_BAKE");

		if (bm && bm[0] == '0')
			bake_max = 0;
		if (!R->bake_tried && R->rtc && R->dev.ngroup > 0 &&
		    ng > 0 && (int64_t) ng <= bake_max)
		{
			std::vector<long long> codes(ng);

			for (size_t i = 0; i < ng; i++)
				codes[i] = (long long) rows[i * rowsz];

			/* overflow-budget proof for the fire-and-forget
			 * LDS tier: interval-bound every aggregate value
			 * from cached column min/max (immutable data).
			 * All values must be non-negative and
			 * bound * rows_per_block must clear 2^61 with a
			 * 4x margin (long double keeps int64 exact; the
			 * margin absorbs product rounding). */
			bool fastok = true;
			int64_t grid = (R->dev.n + 255) / 256;

			if (grid > 2048)
				grid = 2048;
			if (grid < 1)
				grid = 1;
			{
				long double rows_pb = (long double)
					R->dev.n / (long double) grid + 256.0L;

				for (int a = 0;
				     fastok && a < R->dev.naggs; a++)
				{
					const PlanAggDev &A = R->dev.aggs[a];
					long double lo = 1.0L, hi = 1.0L;

					if (A.kind != 2)
						continue;	/* count: [1,1] */
					for (int f = 0;
					     fastok && f < A.nf; f++)
					{
						long long mn, mx;

						if (engine_col_minmax(
							e, A.col[f],
							A.width[f],
							R->dev.n, &mn,
							&mx) != GG_OK)
						{
							fastok = false;
							break;
						}
						long double a0 = (long double) mn;
						long double a1 = (long double) mx;

						if (A.mod[f] == 1)
						{
							long double t0 = 100.0L - a1;
							long double t1 = 100.0L - a0;

							a0 = t0;
							a1 = t1;
						}
						else if (A.mod[f] == 2)
						{
							a0 = 100.0L + a0;
							a1 = 100.0L + a1;
						}
						{
							long double c[4] = {
								lo * a0, lo * a1,
								hi * a0, hi * a1};
							long double nl = c[0],
								nh = c[0];

							for (int q = 1; q < 4; q++)
							{
								if (c[q] < nl)
									nl = c[q];
								if (c[q] > nh)
									nh = c[q];
							}
							lo = nl;
							hi = nh;
						}
					}
					if (lo < 0.0L ||
					    hi * rows_pb >=
					    2305843009213693952.0L) /* 2^61 */
						fastok = false;
				}
			}
			gg_status rs = plan_rtc_compile(
				R->dev, R->dev.gnulls[0] != nullptr,
				R->dev.gnulls[1] != nullptr, &R->rtc_baked,
				codes.data(), (int) ng, fastok);

			R->bake_tried = true;
			if (rs == GG_OK)
				p->stat(fastok ? "path_plan_rtc_baked_fast"
					: "path_plan_rtc_baked").launches++;
		}
	}
	return GG_OK;
}

/* ---- nullable columns ---- */

extern "C" gg_status
gg_engine_table_set_nulls(gg_table t, const char *col,
			  const uint8_t *host_nulls)
{
	Engine &e = engine();

	if (!e.inited)
		return fail(GG_ESTATE, "engine not initialized");
	Table *tab = engine_table(t);

	if (!tab)
		return fail(GG_EINVAL, "bad table handle");
	Table::Col *c = tab->find(col ? col : "");

	if (!c)
		return fail(GG_EINVAL, "no column '%s'", col ? col : "");
	if (!host_nulls)
		return fail(GG_EINVAL, "null flags pointer");
	if (!c->nulls)
		GG_HIP(hipMalloc(&c->nulls, (size_t) tab->nrows));
	GG_HIP(hipMemcpy(c->nulls, host_nulls, (size_t) tab->nrows,
			 hipMemcpyHostToDevice));
	return GG_OK;
}

/* ---- NULL-aware general hash group-by ----
 * Reuses the in-memory group-by kernels twice: pass A aggregates
 * NULL-mapped keys with NULL vals as 0 (sums exact since strict SUM
 * adds nothing for NULLs), pass B counts only the non-NULL vals (the
 * strict COUNT side); groups existing only through NULL vals keep
 * count 0 (a seen key creates its group regardless, execHHashagg
 * find-or-create). */
extern "C" gg_status
gg_engine_hash_groupby_i64_n(const int64_t *keys, const uint8_t *key_nulls,
			     const int64_t *vals, const uint8_t *val_nulls,
			     int64_t n, int64_t *out_keys,
			     int64_t *out_sums, int64_t *out_counts,
			     int64_t cap, int64_t *out_ngroups)
{
	if (!keys || !vals || n < 0 || !out_keys || !out_sums ||
	    !out_counts || !out_ngroups)
		return fail(GG_EINVAL, "bad groupby_n args");
	std::vector<int64_t> k2((size_t) n), v2((size_t) n);

	for (int64_t i = 0; i < n; i++)
	{
		if (keys[i] == INT64_MIN)
			return fail(GG_EINVAL,
				    "group key INT64_MIN unsupported");
		k2[i] = (key_nulls && key_nulls[i]) ? GG_PLAN_NULL_KEY
			: keys[i];
		v2[i] = (val_nulls && val_nulls[i]) ? 0 : vals[i];
	}
	GG_TRY(gg_engine_hash_groupby_i64(k2.data(), v2.data(), n, out_keys,
					  out_sums, out_counts, cap,
					  out_ngroups));
	if (!val_nulls)
		return GG_OK;

	/* pass B: non-NULL counts via a filtered second group-by */
	std::vector<int64_t> kb, vb;

	kb.reserve((size_t) n);
	for (int64_t i = 0; i < n; i++)
		if (!val_nulls[i])
		{
			kb.push_back(k2[i]);
			vb.push_back(0);
		}
	std::vector<int64_t> bk(kb.size() + 1), bs(kb.size() + 1),
		bc(kb.size() + 1);
	int64_t nb = 0;

	if (!kb.empty())
		GG_TRY(gg_engine_hash_groupby_i64(kb.data(), vb.data(),
						  (int64_t) kb.size(),
						  bk.data(), bs.data(),
						  bc.data(),
						  (int64_t) bk.size(), &nb));
	/* both outputs are sorted by key: merge counts (default 0) */
	{
		int64_t j = 0;

		for (int64_t i = 0; i < *out_ngroups; i++)
		{
			while (j < nb && bk[j] < out_keys[i])
				j++;
			out_counts[i] = (j < nb && bk[j] == out_keys[i])
				? bc[j] : 0;
		}
	}
	return GG_OK;
}

}				/* namespace gg */
