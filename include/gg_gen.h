/*
 * Deterministic TPC-H-shaped synthetic data generator — the single
 * definition used by BOTH the CPU oracle and the HIP device kernels
 * (there is no network for real datasets; BASELINE.md §"Workloads"
 * prescribes deterministic seeded dbgen-shaped synthetic inputs).
 *
 * Every column value is a pure function of (seed, table, row index) via
 * splitmix64, so any subrange of any table can be (re)generated on any
 * device without materializing anything else — the CPU oracle streams
 * rows through the same functions the GPU generator kernel uses.
 *
 * Shapes (per TPC-H spec sizes):
 *   customers(SF) = 150,000 × SF     c_custkey = i+1 (dense)
 *   orders(SF)    = 1,500,000 × SF   o_orderkey = i+1 (dense)
 *   lineitem(SF)  = 4 × orders(SF)   l_orderkey = i/4 + 1 (exactly 4
 *                                    lines/order; dbgen draws 1–7, avg 4 —
 *                                    documented deviation)
 *
 * Distributions follow dbgen's shapes (TPC-H spec 4.2.2/4.2.3):
 *   o_orderdate   uniform [1992-01-01, 1998-08-02]
 *   o_custkey     uniform 1..customers(SF)
 *   l_shipdate    = o_orderdate + 1..121
 *   l_receiptdate = l_shipdate + 1..30
 *   l_returnflag  'R' or 'A' (50/50) if receiptdate <= 1995-06-17 else 'N'
 *   l_linestatus  'O' if shipdate > 1995-06-17 else 'F'
 *   l_quantity    1..50 (stored as cents: ×100, numeric(15,2))
 *   l_extendedprice cents, 90,000..10,000,000 (~$900..$100k, dbgen range)
 *   l_discount    0.00..0.10 (stored ×100: 0..10)
 *   l_tax         0.00..0.08 (stored ×100: 0..8)
 *   c_mktsegment  uniform over 5 segments; code 2 = 'MACHINERY'
 *
 * Decimal columns are scaled int64 (cents, scale 2) exactly as the engine
 * stores numeric(15,2) — see DESIGN.md §data layout.
 */
#ifndef GG_GEN_H
#define GG_GEN_H

#include <stdint.h>
#include "gg_pgdate.h"

#if defined(__HIPCC__) || defined(__HIP_DEVICE_COMPILE__)
#define GG_GEN_HOSTDEV __host__ __device__ static inline
#else
#define GG_GEN_HOSTDEV static inline
#endif

#define GG_SEED_DEFAULT 42ull

/* table tags mixed into the stream so tables are independent */
#define GG_TAB_LINEITEM 0x4c49ull	/* 'LI' */
#define GG_TAB_ORDERS   0x4f52ull	/* 'OR' */
#define GG_TAB_CUSTOMER 0x4355ull	/* 'CU' */
#define GG_TAB_SUPPLIER 0x5355ull	/* 'SU' */

#define GG_MKTSEG_MACHINERY 2

/* dict codes for the char(1) columns: the actual ASCII byte is stored */
#define GG_RF_A 'A'
#define GG_RF_N 'N'
#define GG_RF_R 'R'
#define GG_LS_F 'F'
#define GG_LS_O 'O'

GG_GEN_HOSTDEV uint64_t gg_splitmix64(uint64_t x)
{
	x += 0x9e3779b97f4a7c15ull;
	x = (x ^ (x >> 30)) * 0xbf58476d1ce4e5b9ull;
	x = (x ^ (x >> 27)) * 0x94d049bb133111ebull;
	return x ^ (x >> 31);
}

/* one 64-bit stream value for (seed, table, row, column-slot) */
GG_GEN_HOSTDEV uint64_t gg_rnd(uint64_t seed, uint64_t tab, uint64_t row,
			       uint64_t slot)
{
	return gg_splitmix64(seed ^ (tab << 48) ^ (slot << 40) ^ row);
}

/*
 * Uniform value in [0, n) by multiply-shift on the hash's high word.
 * Deliberately avoids 64-bit modulo: hipcc (ROCm 7.2, gfx950) miscompiled
 * `u64 % const` inside the large generator kernel (the magic-multiply
 * quotient reconstruction dropped high-word bits → garbage in bits 29–31
 * of the remainder; isolated kernels were fine).  Multiply-shift uses
 * only 32×32→64 multiplies, is bias-free enough for benchmark data
 * (n ≤ 2^32, bias < 2^-32), and is identical on host and device.
 */
GG_GEN_HOSTDEV uint32_t gg_rnd_range(uint64_t seed, uint64_t tab,
				     uint64_t row, uint64_t slot, uint32_t n)
{
	uint32_t hi = (uint32_t) (gg_rnd(seed, tab, row, slot) >> 32);

	return (uint32_t) (((uint64_t) hi * n) >> 32);
}

/* ---- table sizes; sf100 = 100 etc. (sf as integer ×1; SF1 => sf=1) ---- */
GG_GEN_HOSTDEV int64_t gg_n_customers(int64_t sf) { return 150000 * sf; }
GG_GEN_HOSTDEV int64_t gg_n_orders(int64_t sf)    { return 1500000 * sf; }
GG_GEN_HOSTDEV int64_t gg_n_lineitem(int64_t sf)  { return 6000000 * sf; }
GG_GEN_HOSTDEV int64_t gg_n_suppliers(int64_t sf) { return 10000 * sf; }

/* ---- nation/region: the 25 TPC-H nations (n_nationkey → n_regionkey),
 * verified against the reference fixture
 * src/test/regress/data/nation.csv / region.csv ---- */
#define GG_NNATIONS 25
#define GG_NREGIONS 5
GG_GEN_HOSTDEV int32_t gg_nation_region(int32_t nationkey)
{
	/* ALGERIA..UNITED STATES in nationkey order */
	const uint8_t reg[GG_NNATIONS] = {
		0, 1, 1, 1, 4, 0, 3, 3, 2, 2, 4, 4, 2, 4, 0,
		0, 0, 1, 2, 3, 4, 2, 3, 3, 1
	};

	return (nationkey >= 0 && nationkey < GG_NNATIONS)
		? (int32_t) reg[nationkey] : -1;
}

/* date-range constants are generated, never hardcoded */
GG_GEN_HOSTDEV int32_t gg_orderdate_lo(void) { return gg_pgdate(1992, 1, 1); }
GG_GEN_HOSTDEV int32_t gg_orderdate_hi(void) { return gg_pgdate(1998, 8, 2); }

/* ---- orders ---- */
GG_GEN_HOSTDEV int32_t gg_o_orderdate(uint64_t seed, int64_t orderkey)
{
	const int32_t lo = gg_orderdate_lo();
	const uint32_t n = (uint32_t) (gg_orderdate_hi() - lo + 1);

	return lo + (int32_t) gg_rnd_range(seed, GG_TAB_ORDERS,
					   (uint64_t) orderkey, 0, n);
}

GG_GEN_HOSTDEV int64_t gg_o_custkey(uint64_t seed, int64_t orderkey,
				    int64_t sf)
{
	return 1 + (int64_t) gg_rnd_range(seed, GG_TAB_ORDERS,
					  (uint64_t) orderkey, 1,
					  (uint32_t) gg_n_customers(sf));
}

GG_GEN_HOSTDEV int32_t gg_o_shippriority(uint64_t seed, int64_t orderkey)
{
	(void) seed; (void) orderkey;
	return 0;		/* dbgen: constant 0 */
}

/* ---- customer ---- */
GG_GEN_HOSTDEV uint8_t gg_c_mktsegment(uint64_t seed, int64_t custkey)
{
	return (uint8_t) gg_rnd_range(seed, GG_TAB_CUSTOMER,
				      (uint64_t) custkey, 0, 5);
}

GG_GEN_HOSTDEV uint8_t gg_c_nationkey(uint64_t seed, int64_t custkey)
{
	return (uint8_t) gg_rnd_range(seed, GG_TAB_CUSTOMER,
				      (uint64_t) custkey, 1, GG_NNATIONS);
}

/* ---- supplier ---- */
GG_GEN_HOSTDEV uint8_t gg_s_nationkey(uint64_t seed, int64_t suppkey)
{
	return (uint8_t) gg_rnd_range(seed, GG_TAB_SUPPLIER,
				      (uint64_t) suppkey, 0, GG_NNATIONS);
}

/* lineitem's supplier (slot 7; dbgen spreads uniformly over suppliers) */
GG_GEN_HOSTDEV int64_t gg_l_suppkey(uint64_t seed, int64_t row, int64_t sf)
{
	return 1 + (int64_t) gg_rnd_range(seed, GG_TAB_LINEITEM,
					  (uint64_t) row, 7,
					  (uint32_t) gg_n_suppliers(sf));
}

/* ---- lineitem (row i, 0-based; orderkey = i/4 + 1) ---- */
typedef struct gg_lineitem_row
{
	int64_t l_orderkey;
	int64_t l_quantity_c;	/* cents (scale 2) */
	int64_t l_extendedprice_c;
	int64_t l_discount_c;	/* 0..10 */
	int64_t l_tax_c;	/* 0..8 */
	int32_t l_shipdate;	/* DateADT */
	uint8_t l_returnflag;	/* 'A'|'N'|'R' */
	uint8_t l_linestatus;	/* 'F'|'O' */
} gg_lineitem_row;

GG_GEN_HOSTDEV void gg_gen_lineitem(uint64_t seed, int64_t i,
				    gg_lineitem_row *r)
{
	const int64_t orderkey = (i >> 2) + 1;
	const uint64_t u = (uint64_t) i;
	const int32_t odate = gg_o_orderdate(seed, orderkey);
	const int32_t cur = gg_pgdate(1995, 6, 17);	/* dbgen CURRENTDATE */
	int32_t ship, receipt;

	/* range mapping via gg_rnd_range (no 64-bit modulo — see its
	 * comment; the generator freeze vectors in tests/golden pin the
	 * exact values) */
	const int32_t m_qty = (int32_t) gg_rnd_range(seed, GG_TAB_LINEITEM, u, 0, 50);
	const int32_t m_price = (int32_t) gg_rnd_range(seed, GG_TAB_LINEITEM, u, 1, 9910001u);
	const int32_t m_disc = (int32_t) gg_rnd_range(seed, GG_TAB_LINEITEM, u, 2, 11);
	const int32_t m_tax = (int32_t) gg_rnd_range(seed, GG_TAB_LINEITEM, u, 3, 9);
	const int32_t m_ship = (int32_t) gg_rnd_range(seed, GG_TAB_LINEITEM, u, 4, 121);
	const int32_t m_rcpt = (int32_t) gg_rnd_range(seed, GG_TAB_LINEITEM, u, 5, 30);

	r->l_orderkey = orderkey;
	r->l_quantity_c = (int64_t) ((1 + m_qty) * 100);
	r->l_extendedprice_c = (int64_t) (90000 + m_price);
	r->l_discount_c = (int64_t) m_disc;
	r->l_tax_c = (int64_t) m_tax;
	ship = odate + 1 + m_ship;
	receipt = ship + 1 + m_rcpt;
	r->l_shipdate = ship;
	r->l_linestatus = (ship > cur) ? GG_LS_O : GG_LS_F;
	if (receipt <= cur)
		r->l_returnflag = (gg_rnd(seed, GG_TAB_LINEITEM, u, 6) & 1)
			? GG_RF_R : GG_RF_A;
	else
		r->l_returnflag = GG_RF_N;
}

#endif /* GG_GEN_H */
