/*
 * CPU oracle — see oracle.h for the semantics contract and reference
 * citations.  TEST INFRASTRUCTURE ONLY: linked by tests/ and by
 * bench.py's cpu_baseline leg, never by the product.
 *
 * Parallel variants use OpenMP; all accumulation is integer addition
 * (commutative/associative), so results are bit-identical for any
 * thread count — the same property that makes the GPU results
 * shard- and schedule-independent.
 */
#include <stdio.h>
#include <stdlib.h>
#include <stddef.h>
#include <string.h>

#include "../include/gg_pg_hash.h"
#include "../include/gg_pgdate.h"
#include "../include/gg_gen.h"
#include "../include/gg_checksum.h"
#include "oracle.h"

#ifdef _OPENMP
#include <omp.h>
#endif

typedef __int128 i128;
typedef unsigned __int128 u128;

static inline i128 mk128(uint64_t lo, int64_t hi)
{
	return ((i128) hi << 64) | (i128) lo;
}

static inline void split128(i128 v, uint64_t *lo, int64_t *hi)
{
	*lo = (uint64_t) (u128) v;
	*hi = (int64_t) (v >> 64);
}

static void set_threads(int nthreads)
{
#ifdef _OPENMP
	if (nthreads > 0)
		omp_set_num_threads(nthreads);
#else
	(void) nthreads;
#endif
}

/* ---------------- Q1 ---------------- */

/* group slot for (returnflag, linestatus); -1 on unexpected bytes */
static inline int q1_slot(uint8_t rf, uint8_t ls)
{
	int fi, si;

	switch (rf)
	{
		case 'A': fi = 0; break;
		case 'N': fi = 1; break;
		case 'R': fi = 2; break;
		default: return -1;
	}
	switch (ls)
	{
		case 'F': si = 0; break;
		case 'O': si = 1; break;
		default: return -1;
	}
	return fi * 2 + si;
}

typedef struct q1_acc
{
	int64_t count[GG_Q1_NGROUPS];
	int64_t sum_qty[GG_Q1_NGROUPS];
	int64_t sum_base[GG_Q1_NGROUPS];
	int64_t sum_dcol[GG_Q1_NGROUPS];
	i128 sum_disc[GG_Q1_NGROUPS];
	i128 sum_charge[GG_Q1_NGROUPS];
} q1_acc;

/*
 * Per-row Q1 transition (nodeAgg.c:393–860 semantics on scaled ints):
 *   disc_price scale4 = price_c * (100 - disc_c)   [mul_var: dscale 2+2]
 *   charge scale6     = disc_price * (100 + tax_c) [dscale 4+2]
 */
static inline void q1_update(q1_acc *a, int g, int64_t qty_c, int64_t price_c,
			     int64_t disc_c, int64_t tax_c)
{
	int64_t disc4 = price_c * (100 - disc_c);
	i128 charge6 = (i128) disc4 * (100 + tax_c);

	a->count[g] += 1;
	a->sum_qty[g] += qty_c;
	a->sum_base[g] += price_c;
	a->sum_dcol[g] += disc_c;
	a->sum_disc[g] += disc4;
	a->sum_charge[g] += charge6;
}

static void q1_acc_merge(q1_acc *dst, const q1_acc *src)
{
	for (int g = 0; g < GG_Q1_NGROUPS; g++)
	{
		dst->count[g] += src->count[g];
		dst->sum_qty[g] += src->sum_qty[g];
		dst->sum_base[g] += src->sum_base[g];
		dst->sum_dcol[g] += src->sum_dcol[g];
		dst->sum_disc[g] += src->sum_disc[g];
		dst->sum_charge[g] += src->sum_charge[g];
	}
}

static void q1_acc_out(const q1_acc *a, gg_q1_group out[GG_Q1_NGROUPS])
{
	for (int g = 0; g < GG_Q1_NGROUPS; g++)
	{
		out[g].count = a->count[g];
		out[g].sum_qty_c = a->sum_qty[g];
		out[g].sum_base_c = a->sum_base[g];
		out[g].sum_dcol_c = a->sum_dcol[g];
		split128(a->sum_disc[g], &out[g].disc_lo, &out[g].disc_hi);
		split128(a->sum_charge[g], &out[g].charge_lo, &out[g].charge_hi);
	}
}

int
gg_oracle_q1_arrays(const int32_t *shipdate, const uint8_t *rflag,
		    const uint8_t *lstatus, const int64_t *qty_c,
		    const int64_t *price_c, const int64_t *disc_c,
		    const int64_t *tax_c, int64_t nrows, int32_t cutoff,
		    gg_q1_group out[GG_Q1_NGROUPS], int nthreads)
{
	q1_acc total;
	int bad = 0;

	memset(&total, 0, sizeof(total));
	set_threads(nthreads);

#pragma omp parallel
	{
		q1_acc local;

		memset(&local, 0, sizeof(local));
#pragma omp for schedule(static) reduction(|:bad)
		for (int64_t i = 0; i < nrows; i++)
		{
			int g;

			if (shipdate[i] > cutoff)	/* qual: l_shipdate <= cutoff */
				continue;
			g = q1_slot(rflag[i], lstatus[i]);
			if (g < 0)
			{
				bad |= 1;
				continue;
			}
			q1_update(&local, g, qty_c[i], price_c[i], disc_c[i],
				  tax_c[i]);
		}
#pragma omp critical
		q1_acc_merge(&total, &local);
	}

	q1_acc_out(&total, out);
	return bad ? -1 : 0;
}

static int
q1_synth_range(uint64_t seed, int64_t row_lo, int64_t row_hi, int32_t cutoff,
	       int32_t nseg, int32_t seg, gg_q1_group out[GG_Q1_NGROUPS],
	       int nthreads)
{
	q1_acc total;
	int bad = 0;

	memset(&total, 0, sizeof(total));
	set_threads(nthreads);

#pragma omp parallel
	{
		q1_acc local;

		memset(&local, 0, sizeof(local));
#pragma omp for schedule(static) reduction(|:bad)
		for (int64_t i = row_lo; i < row_hi; i++)
		{
			gg_lineitem_row r;
			int g;

			gg_gen_lineitem(seed, i, &r);
			if (nseg > 1 &&
			    gg_cdbhash_segment_int8(r.l_orderkey, nseg) != seg)
				continue;
			if (r.l_shipdate > cutoff)
				continue;
			g = q1_slot(r.l_returnflag, r.l_linestatus);
			if (g < 0)
			{
				bad |= 1;
				continue;
			}
			q1_update(&local, g, r.l_quantity_c,
				  r.l_extendedprice_c, r.l_discount_c,
				  r.l_tax_c);
		}
#pragma omp critical
		q1_acc_merge(&total, &local);
	}

	q1_acc_out(&total, out);
	return bad ? -1 : 0;
}

int
gg_oracle_q1_synth(uint64_t seed, int64_t sf, int64_t row_lo, int64_t row_hi,
		   int32_t cutoff, gg_q1_group out[GG_Q1_NGROUPS],
		   int nthreads)
{
	if (row_hi < 0)
		row_hi = gg_n_lineitem(sf);
	return q1_synth_range(seed, row_lo, row_hi, cutoff, 1, 0, out,
			      nthreads);
}

int
gg_oracle_q1_synth_segment(uint64_t seed, int64_t sf, int32_t nseg,
			   int32_t seg, int32_t cutoff,
			   gg_q1_group out[GG_Q1_NGROUPS], int nthreads)
{
	return q1_synth_range(seed, 0, gg_n_lineitem(sf), cutoff, nseg, seg,
			      out, nthreads);
}

/* ---------------- Q1, Volcano restatement (the "cpu-ref" leg) --------
 * Models the reference's tuple-at-a-time executor for the Q1 slice
 * (BASELINE.md cpu-ref): one ExecProcNode-style INDIRECT call per
 * tuple (execProcnode.c:925), slot deform into a Datum array
 * (slot_deform_tuple heaptuple.c:1119), an interpreted expression
 * walk per row for the qual (ExecQual execQual.c:6260) and each agg
 * input (ExecTargetList :6369), a per-row group hash + bucket-chain
 * compare (execHHashagg.c:157 calc_hash_value / :456
 * lookup_agg_hash_entry), and one indirect transition-function call
 * per agg per row (invoke_agg_trans_func nodeAgg.c:413) — the
 * structure whose per-row costs the batched GPU engine removes.
 * Results are bit-identical to the vectorized oracle leg. */

enum
{
	VE_VAR,			/* slot[attno] */
	VE_CONST,
	VE_LE,			/* l <= r */
	VE_SUB100_RHS,		/* l * (100 - r): mul_var scale 2+2 */
	VE_ADD100_RHS		/* l * (100 + r): mul_var scale 4+2 */
};

typedef struct vexpr
{
	int op;
	int attno;
	int64_t cval;
	const struct vexpr *l, *r;
} vexpr;

static int64_t
vexpr_eval(const vexpr *e, const int64_t *slot)
{
	switch (e->op)
	{
		case VE_VAR:
			return slot[e->attno];
		case VE_CONST:
			return e->cval;
		case VE_LE:
			return vexpr_eval(e->l, slot) <= vexpr_eval(e->r, slot);
		case VE_SUB100_RHS:
			return vexpr_eval(e->l, slot) *
				(100 - vexpr_eval(e->r, slot));
		case VE_ADD100_RHS:
			return vexpr_eval(e->l, slot) *
				(100 + vexpr_eval(e->r, slot));
	}
	return 0;
}

/* slot layout (deformed tuple) */
enum { VA_QTY, VA_PRICE, VA_DISC, VA_TAX, VA_SHIPDATE, VA_RFLAG, VA_LSTATUS,
       VA_NATTS };

typedef struct vscan
{
	uint64_t seed;
	int64_t next, hi;
	const vexpr *qual;
	int64_t slot[VA_NATTS];
} vscan;

/* SeqNext + ExecQual loop (execScan.c:110–214): returns the deformed
 * slot of the next qual-passing tuple, NULL at end of scan */
static const int64_t *
vscan_next(void *p)
{
	vscan *s = (vscan *) p;

	while (s->next < s->hi)
	{
		gg_lineitem_row r;

		gg_gen_lineitem(s->seed, s->next++, &r);
		s->slot[VA_QTY] = r.l_quantity_c;
		s->slot[VA_PRICE] = r.l_extendedprice_c;
		s->slot[VA_DISC] = r.l_discount_c;
		s->slot[VA_TAX] = r.l_tax_c;
		s->slot[VA_SHIPDATE] = r.l_shipdate;
		s->slot[VA_RFLAG] = r.l_returnflag;
		s->slot[VA_LSTATUS] = r.l_linestatus;
		if (vexpr_eval(s->qual, s->slot))
			return s->slot;
	}
	return NULL;
}

/* per-agg transition state + indirect transition fn (nodeAgg.c:413) */
typedef void (*vtrans_fn)(void *state, int64_t datum);

static void vt_count(void *s, int64_t d) { (void) d; *(int64_t *) s += 1; }
static void vt_sum64(void *s, int64_t d) { *(int64_t *) s += d; }
static void vt_sum128(void *s, int64_t d) { *(i128 *) s += d; }

typedef struct vagg
{
	vtrans_fn fn;
	const vexpr *arg;	/* NULL for count(*) */
	size_t state_off;	/* into the group entry */
} vagg;

typedef struct vgroup
{
	uint8_t rf, ls;
	int used;
	int64_t count, sum_qty, sum_base, sum_dcol;
	i128 sum_disc, sum_charge;
	struct vgroup *next;	/* bucket chain */
} vgroup;

#define VHASH_NBUCKETS 16

typedef struct vhash
{
	vgroup *buckets[VHASH_NBUCKETS];
	vgroup pool[GG_Q1_NGROUPS + 2];
	int npool;
} vhash;

/* calc_hash_value (execHHashagg.c:157): FunctionCall1-per-key hashes
 * combined with the rotate-left-xor the reference applies per column */
static vgroup *
vhash_lookup(vhash *h, uint8_t rf, uint8_t ls)
{
	uint32_t hv = gg_hashchar((char) rf);

	hv = ((hv << 1) | (hv >> 31)) ^ gg_hashchar((char) ls);
	{
		vgroup *g = h->buckets[hv & (VHASH_NBUCKETS - 1)];

		for (; g; g = g->next)
			if (g->rf == rf && g->ls == ls)
				return g;
		if (h->npool >= GG_Q1_NGROUPS + 2)
			return NULL;
		g = &h->pool[h->npool++];
		memset(g, 0, sizeof(*g));
		g->rf = rf;
		g->ls = ls;
		g->used = 1;
		g->next = h->buckets[hv & (VHASH_NBUCKETS - 1)];
		h->buckets[hv & (VHASH_NBUCKETS - 1)] = g;
		return g;
	}
}

static int
q1_volcano_range(uint64_t seed, int64_t row_lo, int64_t row_hi,
		 int32_t cutoff, q1_acc *acc)
{
	/* plan setup (ExecutorStart analog): qual + agg input exprs */
	vexpr v_ship = {VE_VAR, VA_SHIPDATE, 0, NULL, NULL};
	vexpr v_cut = {VE_CONST, 0, cutoff, NULL, NULL};
	vexpr qual = {VE_LE, 0, 0, &v_ship, &v_cut};
	vexpr v_qty = {VE_VAR, VA_QTY, 0, NULL, NULL};
	vexpr v_price = {VE_VAR, VA_PRICE, 0, NULL, NULL};
	vexpr v_disc = {VE_VAR, VA_DISC, 0, NULL, NULL};
	vexpr v_tax = {VE_VAR, VA_TAX, 0, NULL, NULL};
	vexpr e_disc4 = {VE_SUB100_RHS, 0, 0, &v_price, &v_disc};
	vexpr e_charge6 = {VE_ADD100_RHS, 0, 0, &e_disc4, &v_tax};
	vagg aggs[6] = {
		{vt_count, NULL, offsetof(vgroup, count)},
		{vt_sum64, &v_qty, offsetof(vgroup, sum_qty)},
		{vt_sum64, &v_price, offsetof(vgroup, sum_base)},
		{vt_sum64, &v_disc, offsetof(vgroup, sum_dcol)},
		{vt_sum128, &e_disc4, offsetof(vgroup, sum_disc)},
		{vt_sum128, &e_charge6, offsetof(vgroup, sum_charge)},
	};
	vscan scan;
	vhash hash;
	/* indirect child call per pull, as ExecProcNode dispatches */
	const int64_t *(*volatile child)(void *) = vscan_next;

	memset(&hash, 0, sizeof(hash));
	scan.seed = seed;
	scan.next = row_lo;
	scan.hi = row_hi;
	scan.qual = &qual;

	for (;;)
	{
		const int64_t *slot = child(&scan);

		if (!slot)
			break;
		{
			vgroup *g = vhash_lookup(&hash,
						 (uint8_t) slot[VA_RFLAG],
						 (uint8_t) slot[VA_LSTATUS]);

			if (!g)
				return -1;
			for (int a = 0; a < 6; a++)
			{
				int64_t d = aggs[a].arg
					? vexpr_eval(aggs[a].arg, slot) : 0;

				aggs[a].fn((char *) g + aggs[a].state_off, d);
			}
		}
	}
	for (int i = 0; i < hash.npool; i++)
	{
		vgroup *g = &hash.pool[i];
		int s = q1_slot(g->rf, g->ls);

		if (s < 0)
			return -1;
		acc->count[s] += g->count;
		acc->sum_qty[s] += g->sum_qty;
		acc->sum_base[s] += g->sum_base;
		acc->sum_dcol[s] += g->sum_dcol;
		acc->sum_disc[s] += g->sum_disc;
		acc->sum_charge[s] += g->sum_charge;
	}
	return 0;
}

int
gg_oracle_q1_volcano_synth(uint64_t seed, int64_t sf, int64_t row_lo,
			   int64_t row_hi, int32_t cutoff,
			   gg_q1_group out[GG_Q1_NGROUPS], int nthreads)
{
	q1_acc total;
	int bad = 0;

	if (row_hi < 0)
		row_hi = gg_n_lineitem(sf);
	memset(&total, 0, sizeof(total));
	set_threads(nthreads);
	if (nthreads == 1)
		bad = q1_volcano_range(seed, row_lo, row_hi, cutoff, &total);
	else
	{
		/* parallel variant: one Volcano pipeline per thread over a
		 * contiguous row range (1 segment per worker), partials
		 * merged — the reference's N-segments-on-one-host shape */
#pragma omp parallel reduction(|:bad)
		{
			q1_acc local;
			int64_t nth = 1, tid = 0;

			memset(&local, 0, sizeof(local));
#ifdef _OPENMP
			nth = omp_get_num_threads();
			tid = omp_get_thread_num();
#endif
			{
				int64_t n = row_hi - row_lo;
				int64_t lo = row_lo + n * tid / nth;
				int64_t hi = row_lo + n * (tid + 1) / nth;

				bad |= q1_volcano_range(seed, lo, hi, cutoff,
							&local);
			}
#pragma omp critical
			q1_acc_merge(&total, &local);
		}
	}
	q1_acc_out(&total, out);
	return bad ? -1 : 0;
}

/* ---------------- numeric formatting / AVG ---------------- */

/*
 * Decimal digit string (LSD first) of a u128; returns digit count (>=1).
 */
static int
u128_digits(u128 v, int digs[48])
{
	int nd = 0;

	if (v == 0)
	{
		digs[0] = 0;
		return 1;
	}
	while (v)
	{
		digs[nd++] = (int) (v % 10);
		v /= 10;
	}
	return nd;
}

void
gg_numeric_to_str(uint64_t lo, int64_t hi, int scale, char *buf)
{
	i128 v = mk128(lo, hi);
	u128 mag;
	int digs[48], nd, p = 0;

	if (v < 0)
	{
		buf[p++] = '-';
		mag = (u128) (-v);
	}
	else
		mag = (u128) v;

	nd = u128_digits(mag, digs);

	{
		int intdigs = nd - scale;

		if (intdigs <= 0)
			buf[p++] = '0';
		else
			for (int i = nd - 1; i >= scale; i--)
				buf[p++] = (char) ('0' + digs[i]);
		if (scale > 0)
		{
			buf[p++] = '.';
			for (int i = scale - 1; i >= 0; i--)
				buf[p++] = (char) ('0' + (i < nd ? digs[i] : 0));
		}
	}
	buf[p] = '\0';
}

/*
 * Normalized NBASE weight and first digit of value = mag × 10^-scale
 * (numeric.c:7144 select_div_scale's view of a NumericVar: digits in
 * base 10000 aligned to the decimal point; weight counts NBASE digits
 * left of the point minus one).
 */
static void
nbase_norm(u128 mag, int scale, int *weight, int *firstdigit)
{
	int digs[48], nd, pmax, g;

	if (mag == 0)
	{
		*weight = 0;
		*firstdigit = 0;
		return;
	}
	nd = u128_digits(mag, digs);
	/* decimal position of digs[i] is (i - scale); most significant: */
	pmax = nd - 1 - scale;
	/* NBASE group of decimal position p is floor(p/4) */
	g = (pmax >= 0) ? pmax / 4 : -((-pmax + 3) / 4);
	{
		int fd = 0;

		for (int off = 3; off >= 0; off--)
		{
			int p = 4 * g + off;
			int i = p + scale;
			int d = (i >= 0 && i < nd) ? digs[i] : 0;

			fd = fd * 10 + d;
		}
		*weight = g;
		*firstdigit = fd;
	}
}

int
gg_avg_rscale(uint64_t sum_lo, int64_t sum_hi, int sum_scale, int64_t count)
{
	i128 sum = mk128(sum_lo, sum_hi);
	u128 sum_mag = sum < 0 ? (u128) (-sum) : (u128) sum;
	u128 cnt_mag = count < 0 ? (u128) (-(i128) count) : (u128) count;
	int w1, fd1, w2, fd2, qweight, rscale;

	nbase_norm(sum_mag, sum_scale, &w1, &fd1);
	nbase_norm(cnt_mag, 0, &w2, &fd2);

	qweight = w1 - w2;
	if (fd1 <= fd2)
		qweight--;

	/* NUMERIC_MIN_SIG_DIGITS 16, DEC_DIGITS 4 (numeric.c:67–69) */
	rscale = 16 - qweight * 4;
	if (rscale < sum_scale)
		rscale = sum_scale;
	/* count dscale is 0; NUMERIC_MIN_DISPLAY_SCALE 0 */
	if (rscale < 0)
		rscale = 0;
	if (rscale > 1000)	/* NUMERIC_MAX_DISPLAY_SCALE */
		rscale = 1000;
	return rscale;
}

void
gg_numeric_avg_to_str(uint64_t sum_lo, int64_t sum_hi, int sum_scale,
		      int64_t count, char *buf)
{
	i128 sum = mk128(sum_lo, sum_hi);
	int rscale = gg_avg_rscale(sum_lo, sum_hi, sum_scale, count);
	int neg = (sum < 0) ^ (count < 0);
	u128 n = sum < 0 ? (u128) (-sum) : (u128) sum;
	u128 c = count < 0 ? (u128) (-(i128) count) : (u128) count;
	u128 q;

	if (count == 0)
	{
		/* SQL AVG over zero rows is NULL; callers never ask */
		strcpy(buf, "NULL");
		return;
	}
	/* scale numerator to rscale (rscale >= sum_scale by construction) */
	for (int i = 0; i < rscale - sum_scale; i++)
		n *= 10;
	/* round half away from zero (round_var semantics) */
	q = (2 * n + c) / (2 * c);
	{
		i128 signedq = neg ? -(i128) q : (i128) q;
		uint64_t qlo;
		int64_t qhi;

		split128(signedq, &qlo, &qhi);
		gg_numeric_to_str(qlo, qhi, rscale, buf);
	}
}

/* ---------------- Q3 ---------------- */

typedef struct topk_state
{
	gg_q3_row rows[64];
	int n;
	int64_t k;
} topk_state;

/* ORDER BY revenue DESC, o_orderdate ASC; orderkey ASC as the shared
 * refinement for fully tied keys (oracle.h header note). */
static int
q3_better(const gg_q3_row *a, const gg_q3_row *b)
{
	i128 ra = mk128(a->rev_lo, a->rev_hi);
	i128 rb = mk128(b->rev_lo, b->rev_hi);

	if (ra != rb)
		return ra > rb;
	if (a->orderdate != b->orderdate)
		return a->orderdate < b->orderdate;
	return a->orderkey < b->orderkey;
}

static void
topk_push(topk_state *t, const gg_q3_row *r)
{
	int i;

	if (t->n < t->k)
	{
		t->rows[t->n++] = *r;
		/* insertion sort up */
		for (i = t->n - 1; i > 0 && q3_better(&t->rows[i], &t->rows[i - 1]); i--)
		{
			gg_q3_row tmp = t->rows[i];

			t->rows[i] = t->rows[i - 1];
			t->rows[i - 1] = tmp;
		}
		return;
	}
	if (!q3_better(r, &t->rows[t->n - 1]))
		return;
	t->rows[t->n - 1] = *r;
	for (i = t->n - 1; i > 0 && q3_better(&t->rows[i], &t->rows[i - 1]); i--)
	{
		gg_q3_row tmp = t->rows[i];

		t->rows[i] = t->rows[i - 1];
		t->rows[i - 1] = tmp;
	}
}

static void
topk_merge(topk_state *dst, const topk_state *src)
{
	for (int i = 0; i < src->n; i++)
		topk_push(dst, &src->rows[i]);
}

/*
 * Core Q3 given dense per-orderkey state arrays.  ord_date[ok] holds the
 * orderdate of orders surviving the customer join + date filter, or
 * INT32_MIN for "no such order".  rev[ok] accumulates scale-4 revenue;
 * touched[ok] marks groups with at least one joined lineitem row.
 */
#define ORD_ABSENT INT32_MIN

typedef struct q3_core
{
	int32_t *ord_date;
	int32_t *ord_prio;
	int64_t *rev;		/* scale 4; per-group fits int64 */
	uint8_t *touched;
	int64_t max_ok;		/* arrays sized max_ok+1 */
} q3_core;

static int
q3_core_alloc(q3_core *c, int64_t max_ok)
{
	c->max_ok = max_ok;
	c->ord_date = malloc((size_t) (max_ok + 1) * 4);
	c->ord_prio = calloc((size_t) (max_ok + 1), 4);
	c->rev = calloc((size_t) (max_ok + 1), 8);
	c->touched = calloc((size_t) (max_ok + 1), 1);
	if (!c->ord_date || !c->ord_prio || !c->rev || !c->touched)
		return -1;
	for (int64_t i = 0; i <= max_ok; i++)
		c->ord_date[i] = ORD_ABSENT;
	return 0;
}

static void
q3_core_free(q3_core *c)
{
	free(c->ord_date);
	free(c->ord_prio);
	free(c->rev);
	free(c->touched);
}

static void
q3_finish(const q3_core *c, int64_t k, gg_q3_row *out_topk,
	  gg_q3_result *res)
{
	topk_state total;
	int64_t ngroups = 0;
	i128 revsum = 0;
	uint64_t checksum = 0;

	total.n = 0;
	total.k = k > 64 ? 64 : k;

#pragma omp parallel
	{
		topk_state local;
		int64_t my_ng = 0;
		i128 my_rev = 0;
		uint64_t my_ck = 0;

		local.n = 0;
		local.k = total.k;
#pragma omp for schedule(static)
		for (int64_t ok = 0; ok <= c->max_ok; ok++)
		{
			gg_q3_row r;

			if (!c->touched[ok])
				continue;
			r.orderkey = ok;
			split128((i128) c->rev[ok], &r.rev_lo, &r.rev_hi);
			r.orderdate = c->ord_date[ok];
			r.shippriority = c->ord_prio[ok];
			my_ng++;
			my_rev += c->rev[ok];
			my_ck += gg_group_hash(r.orderkey, r.rev_lo, r.rev_hi,
					       r.orderdate, r.shippriority);
			topk_push(&local, &r);
		}
#pragma omp critical
		{
			ngroups += my_ng;
			revsum += my_rev;
			checksum += my_ck;
			topk_merge(&total, &local);
		}
	}

	res->n_groups = ngroups;
	split128(revsum, &res->rev_sum_lo, &res->rev_sum_hi);
	res->group_checksum = checksum;
	res->n_out = total.n;
	for (int i = 0; i < total.n; i++)
		out_topk[i] = total.rows[i];
}

int
gg_oracle_q3_arrays(const int64_t *c_custkey, const uint8_t *c_mktseg,
		    int64_t ncust, uint8_t seg_code, const int64_t *o_orderkey,
		    const int64_t *o_custkey, const int32_t *o_orderdate,
		    const int32_t *o_shippriority, int64_t nord,
		    const int64_t *l_orderkey, const int32_t *l_shipdate,
		    const int64_t *l_price_c, const int64_t *l_disc_c,
		    int64_t nli, int32_t cutoff_date, int64_t k,
		    gg_q3_row *out_topk, gg_q3_result *res, int nthreads)
{
	int64_t max_ck = 0, max_ok = 0;
	uint8_t *cust_ok;
	q3_core core;
	int64_t njoin = 0;

	set_threads(nthreads);

	for (int64_t i = 0; i < ncust; i++)
		if (c_custkey[i] > max_ck)
			max_ck = c_custkey[i];
	for (int64_t i = 0; i < nord; i++)
		if (o_orderkey[i] > max_ok)
			max_ok = o_orderkey[i];
	/* dense-key guard: these are TPC-H synthetic/fixture keys */
	if (max_ck > 8 * ncust + 16 || max_ok > 8 * nord + 16)
		return -2;

	cust_ok = calloc((size_t) (max_ck + 1), 1);
	if (!cust_ok || q3_core_alloc(&core, max_ok) != 0)
		return -1;

#pragma omp parallel for schedule(static)
	for (int64_t i = 0; i < ncust; i++)
		if (c_mktseg[i] == seg_code)
			cust_ok[c_custkey[i]] = 1;

#pragma omp parallel for schedule(static)
	for (int64_t i = 0; i < nord; i++)
	{
		if (o_orderdate[i] >= cutoff_date)	/* qual: < cutoff */
			continue;
		if (o_custkey[i] > max_ck || !cust_ok[o_custkey[i]])
			continue;
		core.ord_date[o_orderkey[i]] = o_orderdate[i];
		core.ord_prio[o_orderkey[i]] = o_shippriority[i];
	}

#pragma omp parallel for schedule(static) reduction(+:njoin)
	for (int64_t i = 0; i < nli; i++)
	{
		int64_t ok = l_orderkey[i];

		if (l_shipdate[i] <= cutoff_date)	/* qual: > cutoff */
			continue;
		if (ok > max_ok || core.ord_date[ok] == ORD_ABSENT)
			continue;
		{
			int64_t rev4 = l_price_c[i] * (100 - l_disc_c[i]);

#pragma omp atomic
			core.rev[ok] += rev4;
			core.touched[ok] = 1;
			njoin++;
		}
	}

	q3_finish(&core, k, out_topk, res);
	res->n_join_rows = njoin;

	free(cust_ok);
	q3_core_free(&core);
	return 0;
}

int
gg_oracle_q3_synth(uint64_t seed, int64_t sf, int32_t cutoff_date, int64_t k,
		   gg_q3_row *out_topk, gg_q3_result *res, int nthreads)
{
	int64_t nord = gg_n_orders(sf);
	int64_t nli = gg_n_lineitem(sf);
	q3_core core;
	int64_t njoin = 0;

	set_threads(nthreads);
	if (q3_core_alloc(&core, nord) != 0)
		return -1;

	/* orders pass: customer "join" is a generator lookup (dense keys) */
#pragma omp parallel for schedule(static)
	for (int64_t okey = 1; okey <= nord; okey++)
	{
		int32_t odate = gg_o_orderdate(seed, okey);
		int64_t ck;

		if (odate >= cutoff_date)
			continue;
		ck = gg_o_custkey(seed, okey, sf);
		if (gg_c_mktsegment(seed, ck) != GG_MKTSEG_MACHINERY)
			continue;
		core.ord_date[okey] = odate;
		core.ord_prio[okey] = gg_o_shippriority(seed, okey);
	}

#pragma omp parallel for schedule(static) reduction(+:njoin)
	for (int64_t i = 0; i < nli; i++)
	{
		gg_lineitem_row r;

		gg_gen_lineitem(seed, i, &r);
		if (r.l_shipdate <= cutoff_date)
			continue;
		if (core.ord_date[r.l_orderkey] == ORD_ABSENT)
			continue;
		{
			int64_t rev4 = r.l_extendedprice_c * (100 - r.l_discount_c);

#pragma omp atomic
			core.rev[r.l_orderkey] += rev4;
			core.touched[r.l_orderkey] = 1;
			njoin++;
		}
	}

	q3_finish(&core, k, out_topk, res);
	res->n_join_rows = njoin;
	q3_core_free(&core);
	return 0;
}

/* ---------------- Q5 (mpph5 semantics; see oracle.h) ---------------- */

#define NATION_ABSENT 255

static void
q5_lineitem_update(int64_t price_c, int64_t disc_c, uint8_t onation,
		   uint8_t snation, const int32_t *nation_region,
		   int32_t regionkey, i128 rev[25], int64_t cnt[25])
{
	if (onation == NATION_ABSENT)
		return;
	if (snation != onation)	/* c_nationkey = s_nationkey */
		return;
	if (snation >= 25 || nation_region[snation] != regionkey)
		return;
	rev[snation] += (i128) (price_c * (100 - disc_c));
	cnt[snation] += 1;
}

static void
q5_out(const i128 rev[25], const int64_t cnt[25], gg_q5_group out[25])
{
	for (int n = 0; n < 25; n++)
	{
		out[n].count = cnt[n];
		split128(rev[n], &out[n].rev_lo, &out[n].rev_hi);
	}
}

int
gg_oracle_q5_synth(uint64_t seed, int64_t sf, int32_t regionkey,
		   int32_t date_lo, int32_t date_hi, gg_q5_group out[25],
		   int nthreads)
{
	int64_t nord = gg_n_orders(sf);
	int64_t nli = gg_n_lineitem(sf);
	uint8_t *onation;
	int32_t nation_region[25];
	i128 rev[25] = {0};
	int64_t cnt[25] = {0};

	for (int n = 0; n < 25; n++)
		nation_region[n] = gg_nation_region(n);
	set_threads(nthreads);
	onation = malloc((size_t) nord + 1);
	if (!onation)
		return -1;
	memset(onation, NATION_ABSENT, (size_t) nord + 1);

	/* orders ⋈ customer: FK always matches; date range filter */
#pragma omp parallel for schedule(static)
	for (int64_t okey = 1; okey <= nord; okey++)
	{
		int32_t odate = gg_o_orderdate(seed, okey);

		if (odate < date_lo || odate >= date_hi)
			continue;
		onation[okey] =
			gg_c_nationkey(seed, gg_o_custkey(seed, okey, sf));
	}

#pragma omp parallel
	{
		i128 lrev[25] = {0};
		int64_t lcnt[25] = {0};

#pragma omp for schedule(static)
		for (int64_t i = 0; i < nli; i++)
		{
			gg_lineitem_row r;
			int64_t sk;

			gg_gen_lineitem(seed, i, &r);
			if (onation[r.l_orderkey] == NATION_ABSENT)
				continue;
			sk = gg_l_suppkey(seed, i, sf);
			q5_lineitem_update(r.l_extendedprice_c, r.l_discount_c,
					   onation[r.l_orderkey],
					   gg_s_nationkey(seed, sk),
					   nation_region, regionkey, lrev,
					   lcnt);
		}
#pragma omp critical
		for (int n = 0; n < 25; n++)
		{
			rev[n] += lrev[n];
			cnt[n] += lcnt[n];
		}
	}

	free(onation);
	q5_out(rev, cnt, out);
	return 0;
}

int
gg_oracle_q5_arrays(const int64_t *c_custkey, const uint8_t *c_nation,
		    int64_t ncust, const int64_t *o_orderkey,
		    const int64_t *o_custkey, const int32_t *o_orderdate,
		    int64_t nord, const int64_t *l_orderkey,
		    const int64_t *l_suppkey, const int64_t *l_price_c,
		    const int64_t *l_disc_c, int64_t nli,
		    const int64_t *s_suppkey, const uint8_t *s_nation,
		    int64_t nsupp, const int32_t *nation_region,
		    int32_t regionkey, int32_t date_lo, int32_t date_hi,
		    gg_q5_group out[25], int nthreads)
{
	int64_t max_ck = 0, max_ok = 0, max_sk = 0;
	uint8_t *ck_nation, *onation, *sk_nation;
	i128 rev[25] = {0};
	int64_t cnt[25] = {0};

	set_threads(nthreads);
	for (int64_t i = 0; i < ncust; i++)
		if (c_custkey[i] > max_ck)
			max_ck = c_custkey[i];
	for (int64_t i = 0; i < nord; i++)
		if (o_orderkey[i] > max_ok)
			max_ok = o_orderkey[i];
	for (int64_t i = 0; i < nsupp; i++)
		if (s_suppkey[i] > max_sk)
			max_sk = s_suppkey[i];
	if (max_ck > 8 * ncust + 16 || max_ok > 8 * nord + 16 ||
	    max_sk > 8 * nsupp + 16)
		return -2;
	ck_nation = malloc((size_t) max_ck + 1);
	onation = malloc((size_t) max_ok + 1);
	sk_nation = malloc((size_t) max_sk + 1);
	if (!ck_nation || !onation || !sk_nation)
		return -1;
	memset(ck_nation, NATION_ABSENT, (size_t) max_ck + 1);
	memset(onation, NATION_ABSENT, (size_t) max_ok + 1);
	memset(sk_nation, NATION_ABSENT, (size_t) max_sk + 1);

#pragma omp parallel for schedule(static)
	for (int64_t i = 0; i < ncust; i++)
		ck_nation[c_custkey[i]] = c_nation[i];
#pragma omp parallel for schedule(static)
	for (int64_t i = 0; i < nsupp; i++)
		sk_nation[s_suppkey[i]] = s_nation[i];
#pragma omp parallel for schedule(static)
	for (int64_t i = 0; i < nord; i++)
	{
		if (o_orderdate[i] < date_lo || o_orderdate[i] >= date_hi)
			continue;
		if (o_custkey[i] > max_ck)
			continue;
		onation[o_orderkey[i]] = ck_nation[o_custkey[i]];
	}

#pragma omp parallel
	{
		i128 lrev[25] = {0};
		int64_t lcnt[25] = {0};

#pragma omp for schedule(static)
		for (int64_t i = 0; i < nli; i++)
		{
			int64_t ok = l_orderkey[i];
			int64_t sk = l_suppkey[i];

			if (ok > max_ok || sk > max_sk)
				continue;
			if (sk_nation[sk] == NATION_ABSENT)
				continue;
			q5_lineitem_update(l_price_c[i], l_disc_c[i],
					   onation[ok], sk_nation[sk],
					   nation_region, regionkey, lrev,
					   lcnt);
		}
#pragma omp critical
		for (int n = 0; n < 25; n++)
		{
			rev[n] += lrev[n];
			cnt[n] += lcnt[n];
		}
	}

	free(ck_nation);
	free(onation);
	free(sk_nation);
	q5_out(rev, cnt, out);
	return 0;
}

/* ---------------- config 1: sum(price) where shipdate < cutoff ---------------- */

int
gg_oracle_sumprice_arrays(const int32_t *shipdate, const int64_t *price_c,
			  int64_t nrows, int32_t cutoff, int64_t *out_sum_c,
			  int64_t *out_count, int nthreads)
{
	int64_t sum = 0, cnt = 0;

	set_threads(nthreads);
#pragma omp parallel for schedule(static) reduction(+:sum) reduction(+:cnt)
	for (int64_t i = 0; i < nrows; i++)
		if (shipdate[i] < cutoff)
		{
			sum += price_c[i];
			cnt++;
		}
	*out_sum_c = sum;
	*out_count = cnt;
	return 0;
}

int
gg_oracle_sumprice_synth(uint64_t seed, int64_t sf, int32_t cutoff,
			 int64_t *out_sum_c, int64_t *out_count, int nthreads)
{
	int64_t n = gg_n_lineitem(sf);
	int64_t sum = 0, cnt = 0;

	set_threads(nthreads);
#pragma omp parallel for schedule(static) reduction(+:sum) reduction(+:cnt)
	for (int64_t i = 0; i < n; i++)
	{
		gg_lineitem_row r;

		gg_gen_lineitem(seed, i, &r);
		if (r.l_shipdate < cutoff)
		{
			sum += r.l_extendedprice_c;
			cnt++;
		}
	}
	*out_sum_c = sum;
	*out_count = cnt;
	return 0;
}

/* ---------------- re-exports ---------------- */

uint32_t gg_oracle_hash_any(const unsigned char *k, int len)
{ return gg_hash_any(k, len); }
uint32_t gg_oracle_hash_uint32(uint32_t k) { return gg_hash_uint32(k); }
uint32_t gg_oracle_hashint4(int32_t v) { return gg_hashint4(v); }
uint32_t gg_oracle_hashint8(int64_t v) { return gg_hashint8(v); }
uint32_t gg_oracle_hashchar(char c) { return gg_hashchar(c); }
int32_t gg_oracle_segment_int8(int64_t key, int32_t nseg)
{ return gg_cdbhash_segment_int8(key, nseg); }
int32_t gg_oracle_segment_int4(int32_t key, int32_t nseg)
{ return gg_cdbhash_segment_int4(key, nseg); }
int32_t gg_oracle_jump_hash(uint64_t key, int32_t nseg)
{ return gg_jump_consistent_hash(key, nseg); }
int32_t gg_oracle_pgdate(int y, int m, int d)
{ return gg_pgdate(y, (unsigned) m, (unsigned) d); }

void
gg_oracle_gen_lineitem(uint64_t seed, int64_t row_lo, int64_t row_hi,
		       int64_t *orderkey, int64_t *qty_c, int64_t *price_c,
		       int64_t *disc_c, int64_t *tax_c, int32_t *shipdate,
		       uint8_t *rflag, uint8_t *lstatus)
{
#pragma omp parallel for schedule(static)
	for (int64_t i = row_lo; i < row_hi; i++)
	{
		gg_lineitem_row r;
		int64_t j = i - row_lo;

		gg_gen_lineitem(seed, i, &r);
		orderkey[j] = r.l_orderkey;
		qty_c[j] = r.l_quantity_c;
		price_c[j] = r.l_extendedprice_c;
		disc_c[j] = r.l_discount_c;
		tax_c[j] = r.l_tax_c;
		shipdate[j] = r.l_shipdate;
		rflag[j] = r.l_returnflag;
		lstatus[j] = r.l_linestatus;
	}
}

void
gg_oracle_gen_orders(uint64_t seed, int64_t sf, int64_t row_lo,
		     int64_t row_hi, int64_t *orderkey, int64_t *custkey,
		     int32_t *orderdate, int32_t *shippriority)
{
#pragma omp parallel for schedule(static)
	for (int64_t i = row_lo; i < row_hi; i++)
	{
		int64_t okey = i + 1;
		int64_t j = i - row_lo;

		orderkey[j] = okey;
		custkey[j] = gg_o_custkey(seed, okey, sf);
		orderdate[j] = gg_o_orderdate(seed, okey);
		shippriority[j] = gg_o_shippriority(seed, okey);
	}
}

void
gg_oracle_gen_customer(uint64_t seed, int64_t row_lo, int64_t row_hi,
		       int64_t *custkey, uint8_t *mktseg, uint8_t *nationkey)
{
#pragma omp parallel for schedule(static)
	for (int64_t i = row_lo; i < row_hi; i++)
	{
		int64_t ck = i + 1;
		int64_t j = i - row_lo;

		custkey[j] = ck;
		mktseg[j] = gg_c_mktsegment(seed, ck);
		if (nationkey)
			nationkey[j] = gg_c_nationkey(seed, ck);
	}
}

void
gg_oracle_gen_supplier(uint64_t seed, int64_t row_lo, int64_t row_hi,
		       int64_t *suppkey, uint8_t *nationkey)
{
#pragma omp parallel for schedule(static)
	for (int64_t i = row_lo; i < row_hi; i++)
	{
		int64_t sk = i + 1;
		int64_t j = i - row_lo;

		suppkey[j] = sk;
		nationkey[j] = gg_s_nationkey(seed, sk);
	}
}

void
gg_oracle_gen_l_suppkey(uint64_t seed, int64_t sf, int64_t row_lo,
			int64_t row_hi, int64_t *suppkey)
{
#pragma omp parallel for schedule(static)
	for (int64_t i = row_lo; i < row_hi; i++)
		suppkey[i - row_lo] = gg_l_suppkey(seed, i, sf);
}
