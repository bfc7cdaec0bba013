/*
 * Internal engine declarations — host side (engine_abi.cpp, comm.cpp)
 * and the kernel-launcher entry points implemented in kernels.hip.
 * Product code: never includes anything from oracle/.
 */
#ifndef GG_ENGINE_INTERNAL_H
#define GG_ENGINE_INTERNAL_H

#include <hip/hip_runtime.h>
#include <cstdint>
#include <map>
#include <memory>
#include <string>
#include <vector>

#include "../../include/engine_abi.h"
#include "../../include/gg_result.h"

namespace gg
{

/* error plumbing: set message, return status */
gg_status fail(gg_status st, const char *fmt, ...);

#define GG_HIP(call) \
	do { \
		hipError_t _e = (call); \
		if (_e != hipSuccess) \
			return ::gg::fail(GG_EGPU, "%s:%d HIP error %s in %s", \
					  __FILE__, __LINE__, \
					  hipGetErrorString(_e), #call); \
	} while (0)

/* device accumulator block for Q1 (matches kernels.hip layout) */
struct Q1DeviceAcc
{
	/* [group][field]: count, qty, base, dcol, disc_lo, disc_hi,
	 * charge_lo, charge_hi — 6 groups × 8 u64 (the two int128 sums
	 * carry into _hi on _lo wraparound) */
	unsigned long long v[6][8];
	unsigned long long err;	/* nonzero: unexpected flag/status byte */
};

struct SumPriceAcc
{
	unsigned long long sum_c;
	unsigned long long count;
};

/* open-addressing hash table (SoA), power-of-two slots, key 0 = empty.
 * bloom: blocked Bloom filter (one u64 word per key, 2 bits set) probed
 * before the slot array — the GPU analog of the reference's
 * bloom-tagged hash buckets (execHHashagg.c:481 BLOOMVAL). */
struct DeviceHashTable
{
	unsigned long long *keys = nullptr;	/* build keys (int64 > 0) */
	unsigned long long *payload = nullptr;	/* (date u32) | (prio u64<<32) */
	unsigned long long *rev = nullptr;	/* Q3 group revenue, scale 4 */
	uint64_t nslots = 0;			/* power of two */
	unsigned long long *bloom = nullptr;	/* pow2 words; may be null */
	uint64_t bloom_words = 0;
};

struct Table
{
	std::string name;
	int64_t nrows = 0;
	/* named device columns (all 8-byte or 4/1-byte arrays) */
	struct Col
	{
		std::string name;
		gg_coltype type;
		void *dev = nullptr;
		size_t bytes = 0;
		/* optional device NULL-flag array (byte per row) —
		 * gg_engine_table_set_nulls / nullable AO mounts */
		void *nulls = nullptr;
		/* for dictionary-encoded text columns: the sorted dict
		 * (host-side; categorical, small) */
		std::vector<uint8_t> dict_bytes;
		std::vector<int64_t> dict_offs;
	};
	std::vector<Col> cols;

	void *col(const char *n) const
	{
		for (auto &c : cols)
			if (c.name == n)
				return c.dev;
		return nullptr;
	}

	Col *find(const char *n)
	{
		for (auto &c : cols)
			if (c.name == n)
				return &c;
		return nullptr;
	}
};

struct KernelStatAcc
{
	std::string name;
	int64_t launches = 0;
	double total_ms = 0;
	int64_t rows_in = 0;
	int64_t rows_out = 0;
	int64_t hbm_bytes = 0;
};

/* internal pipeline kind for generic compiled plans */
#define GG_PIPE_PLAN_INTERNAL 99

struct Pipeline
{
	gg_pipeline_desc desc;
	/* resolved generic plan (plan.cpp PlanResolved) when desc.kind
	 * == GG_PIPE_PLAN_INTERNAL; shared_ptr carries the deleter */
	std::shared_ptr<void> plan;
	std::vector<KernelStatAcc> stats;

	/* named device scratch reused across executes (hash tables,
	 * exchange buffers): avoids per-execute hipMalloc/hipFree of
	 * hundreds of MB (ExecHashTableCreate is likewise once per
	 * rescan in the reference, nodeHash.c:270) */
	std::vector<std::pair<std::string, std::pair<void *, size_t>>> scratch;
	/* table sizings resolved on first execute (data is immutable
	 * per pipeline), skipping the count passes afterwards */
	uint64_t cust_slots = 0;
	uint64_t ord_slots = 0;
	uint64_t supp_slots = 0;
	/* max(key) per registered column (immutable per pipeline):
	 * computed once for the guarded dense-map sizing, not per step */
	std::vector<std::pair<std::pair<const void *, int64_t>,
			      unsigned long long>> maxk_cache;
	/* Q3 dense rev[] lifecycle: the 1.2 GB accumulator is fully
	 * memset only when freshly allocated or left dirty (fallback
	 * path); otherwise the previous execute's survivor sweep zeroed
	 * exactly the touched entries */
	bool q3_rev_inited = false;
	bool q3_rev_dirty = false;

	KernelStatAcc &stat(const char *name)
	{
		for (auto &s : stats)
			if (s.name == name)
				return s;
		stats.push_back({});
		stats.back().name = name;
		return stats.back();
	}

	void *sget(const char *name, size_t bytes)
	{
		for (auto &kv : scratch)
			if (kv.first == name)
			{
				if (kv.second.second >= bytes)
					return kv.second.first;
				(void) hipFree(kv.second.first);
				kv.second.first = nullptr;
				if (hipMalloc(&kv.second.first, bytes) != hipSuccess)
					return nullptr;
				kv.second.second = bytes;
				return kv.second.first;
			}
		void *p = nullptr;

		if (hipMalloc(&p, bytes ? bytes : 1) != hipSuccess)
			return nullptr;
		scratch.push_back({name, {p, bytes}});
		return p;
	}

	~Pipeline()
	{
		for (auto &kv : scratch)
			(void) hipFree(kv.second.first);
	}
};

struct Engine
{
	gg_engine_config cfg{};
	bool inited = false;
	hipStream_t stream = nullptr;
	std::vector<Table *> tables;
	std::vector<Pipeline *> pipelines;
	void *comm = nullptr;	/* ncclComm_t when comm_init'ed */
	/* per-column min/max cache (registered columns are immutable);
	 * keyed by device pointer */
	std::map<const void *, std::pair<long long, long long>> mm_cache;
	/* engine-level grow-only device scratch (standalone ABI calls
	 * like hash_groupby_i64: repeated calls reuse instead of
	 * re-mapping tens of GB per call); freed at shutdown */
	std::vector<std::pair<std::string, std::pair<void *, size_t>>> escratch;

	void *esget(const char *name, size_t bytes)
	{
		for (auto &kv : escratch)
			if (kv.first == name)
			{
				if (kv.second.second >= bytes)
					return kv.second.first;
				(void) hipFree(kv.second.first);
				kv.second.first = nullptr;
				if (hipMalloc(&kv.second.first, bytes)
				    != hipSuccess)
					return nullptr;
				kv.second.second = bytes;
				return kv.second.first;
			}
		void *p = nullptr;

		if (hipMalloc(&p, bytes ? bytes : 1) != hipSuccess)
			return nullptr;
		escratch.push_back({name, {p, bytes}});
		return p;
	}

	/* same, for pinned host memory (hipHostMalloc is ~1 GB/s of
	 * page-locking — reuse across calls) */
	std::vector<std::pair<std::string, std::pair<void *, size_t>>> ehscratch;

	void *ehget(const char *name, size_t bytes)
	{
		for (auto &kv : ehscratch)
			if (kv.first == name)
			{
				if (kv.second.second >= bytes)
					return kv.second.first;
				(void) hipHostFree(kv.second.first);
				kv.second.first = nullptr;
				if (hipHostMalloc(&kv.second.first, bytes)
				    != hipSuccess)
					return nullptr;
				kv.second.second = bytes;
				return kv.second.first;
			}
		void *p = nullptr;

		if (hipHostMalloc(&p, bytes ? bytes : 1) != hipSuccess)
			return nullptr;
		ehscratch.push_back({name, {p, bytes}});
		return p;
	}
};

Engine &engine();

/* timed launch helper: records HIP events around fn(stream).
 * Events come from a small per-thread pool (creating + destroying two
 * events per stat block measured ~50–100 µs/step of harness overhead
 * on the bench headline).  Stat blocks never nest (sequential per
 * pipeline step), so a depth-indexed pool of 4 suffices; deeper
 * nesting falls back to create/destroy. */
struct Timed
{
	static constexpr int POOL = 4;
	struct Pool
	{
		hipEvent_t ev[POOL][2] = {};
		int depth = 0;
	};
	static Pool &pool()
	{
		static thread_local Pool p;
		return p;
	}

	hipEvent_t a = nullptr, b = nullptr;
	hipStream_t s;
	bool pooled = false;

	explicit Timed(hipStream_t st) : s(st)
	{
		Pool &p = pool();

		if (p.depth < POOL)
		{
			auto &slot = p.ev[p.depth];

			if (!slot[0])
			{
				(void) hipEventCreate(&slot[0]);
				(void) hipEventCreate(&slot[1]);
			}
			a = slot[0];
			b = slot[1];
			pooled = true;
			p.depth++;
		}
		else
		{
			(void) hipEventCreate(&a);
			(void) hipEventCreate(&b);
		}
		(void) hipEventRecord(a, s);
	}
	/* returns elapsed ms (synchronizes the events) */
	double stop()
	{
		float ms = 0;
		(void) hipEventRecord(b, s);
		(void) hipEventSynchronize(b);
		(void) hipEventElapsedTime(&ms, a, b);
		if (!pooled)
		{
			(void) hipEventDestroy(a);
			(void) hipEventDestroy(b);
		}
		a = b = nullptr;
		return ms;
	}
	~Timed()
	{
		if (a && !pooled)
		{
			(void) hipEventDestroy(a);
			(void) hipEventDestroy(b);
		}
		if (pooled)
			pool().depth--;
	}
};

/* ---- kernel launchers (kernels.hip) — return hipError_t ---- */

hipError_t launch_gen_lineitem(hipStream_t s, uint64_t seed, int64_t sf,
			       int64_t row_lo, int64_t n, int nseg, int seg,
			       int64_t *orderkey, int64_t *qty, int64_t *price,
			       int64_t *disc, int64_t *tax, int32_t *shipdate,
			       uint8_t *rflag, uint8_t *lstatus,
			       int64_t *suppkey,
			       unsigned long long *out_count);
hipError_t launch_gen_orders(hipStream_t s, uint64_t seed, int64_t sf,
			     int64_t row_lo, int64_t n, int nseg, int seg,
			     int64_t *orderkey, int64_t *custkey,
			     int32_t *orderdate, int32_t *prio,
			     unsigned long long *out_count);
hipError_t launch_gen_customer(hipStream_t s, uint64_t seed, int64_t row_lo,
			       int64_t n, int nseg, int seg, int64_t *custkey,
			       uint8_t *mktseg, uint8_t *nationkey,
			       unsigned long long *out_count);
hipError_t launch_gen_supplier(hipStream_t s, uint64_t seed, int64_t row_lo,
			       int64_t n, int nseg, int seg, int64_t *suppkey,
			       uint8_t *nation, unsigned long long *out_count);
hipError_t launch_build_kv(hipStream_t s, const int64_t *keys,
			   const uint8_t *vals, int64_t n, DeviceHashTable t);
hipError_t launch_build_supp(hipStream_t s, const int64_t *suppkey,
			     const uint8_t *snation, int64_t n,
			     const uint8_t *region_of, uint8_t regionkey,
			     DeviceHashTable t, unsigned long long *out_count);
hipError_t launch_supp_filter_compact(hipStream_t s, const int64_t *suppkey,
				      const uint8_t *snation, int64_t n,
				      const uint8_t *region_of,
				      uint8_t regionkey, int64_t *out_sk,
				      int64_t *out_sn,
				      unsigned long long *out_count);
hipError_t launch_insert_supp(hipStream_t s, const int64_t *sk,
			      const int64_t *sn, int64_t n,
			      DeviceHashTable t);
hipError_t launch_build_orders_q5(hipStream_t s, const int64_t *okey,
				  const int64_t *ckey, const int32_t *odate,
				  int64_t n, int32_t date_lo, int32_t date_hi,
				  DeviceHashTable cust,
				  const uint8_t *cust_dense,
				  int64_t cust_dlen, DeviceHashTable ord,
				  unsigned long long *match_count);
hipError_t launch_count_date_range(hipStream_t s, const int32_t *odate,
				   int64_t n, int32_t date_lo,
				   int32_t date_hi, unsigned long long *out);
hipError_t launch_probe_cust_map_compact(hipStream_t s, const int64_t *ckey,
					 const int64_t *okey, int64_t n,
					 DeviceHashTable cust,
					 const uint8_t *cust_dense,
					 int64_t cust_dlen,
					 int64_t *out_okey, int64_t *out_nat,
					 unsigned long long *out_count);
hipError_t launch_probe_lineitem_q5(hipStream_t s, const int64_t *okey,
				    const int64_t *skey, const int64_t *price,
				    const int64_t *disc, int64_t n,
				    DeviceHashTable ord, DeviceHashTable supp,
				    const uint8_t *supp_dense,
				    int64_t supp_dense_len,
				    unsigned long long *acc,
				    unsigned long long *join_rows);
hipError_t launch_max_i64(hipStream_t s, const int64_t *v, int64_t n,
			  unsigned long long *out_max);
hipError_t launch_supp_dense_fill(hipStream_t s, const int64_t *suppkey,
				  const uint8_t *snation, int64_t n,
				  const uint8_t *region_of, uint8_t regionkey,
				  uint8_t *dense, int64_t dense_len);
hipError_t launch_supp_dense_fill_pairs(hipStream_t s, const int64_t *sk,
					const int64_t *sn, int64_t n,
					uint8_t *dense, int64_t dense_len);

hipError_t launch_q1(hipStream_t s, const int32_t *shipdate,
		     const uint8_t *rflag, const uint8_t *lstatus,
		     const int64_t *qty, const int64_t *price,
		     const int64_t *disc, const int64_t *tax, int64_t n,
		     int32_t cutoff, Q1DeviceAcc *acc);

hipError_t launch_count_shard(hipStream_t s, int64_t row_lo, int64_t n,
			      int nseg, int seg, int which_table,
			      uint64_t seed, int64_t sf,
			      unsigned long long *out_count);

hipError_t launch_sumprice(hipStream_t s, const int32_t *shipdate,
			   const int64_t *price, int64_t n, int32_t cutoff,
			   SumPriceAcc *acc);

hipError_t launch_build_set(hipStream_t s, const int64_t *keys,
			    const uint8_t *filter_col, uint8_t filter_val,
			    int64_t n, DeviceHashTable t);
hipError_t launch_build_orders(hipStream_t s, const int64_t *okey,
			       const int64_t *ckey, const int32_t *odate,
			       const int32_t *prio, int64_t n, int32_t cutoff,
			       DeviceHashTable cust, const unsigned long long *cust_bits,
			       int64_t cust_dlen, DeviceHashTable ord,
			       unsigned long long *match_count);
hipError_t launch_cust_dense_fill_seg(hipStream_t s, const int64_t *custkey,
				      const uint8_t *mktseg, int64_t n,
				      uint8_t segcode, unsigned long long *bits,
				      int64_t dense_len);
hipError_t launch_cust_dense_fill_nat(hipStream_t s, const int64_t *custkey,
				      const uint8_t *nation, int64_t n,
				      uint8_t *dense, int64_t dense_len);
hipError_t launch_probe_lineitem(hipStream_t s, const int64_t *okey,
				 const int32_t *shipdate,
				 const int64_t *price, const int64_t *disc,
				 int64_t n, int32_t cutoff,
				 DeviceHashTable ord,
				 unsigned long long *join_rows);
hipError_t launch_q3_stats(hipStream_t s, DeviceHashTable ord,
			   unsigned long long *out4 /* ngroups, revlo,
						     * revhi, checksum */);
hipError_t launch_q3_hist(hipStream_t s, DeviceHashTable ord,
			  const unsigned long long *stats5,
			  unsigned int *hist64k);
hipError_t launch_q3_threshold(hipStream_t s, const unsigned int *hist64k,
			       const unsigned long long *stats5, int64_t k,
			       unsigned long long *out_thr);
hipError_t launch_q3_collect(hipStream_t s, DeviceHashTable ord,
			     const unsigned long long *thr_ptr,
			     gg_q3_result_row *out,
			     unsigned long long *out_count, uint64_t cap);

hipError_t launch_count_filter_u8(hipStream_t s, const uint8_t *col,
				  uint8_t val, int64_t n,
				  unsigned long long *out);
hipError_t launch_count_orders_match(hipStream_t s, const int64_t *ckey,
				     const int32_t *odate, int64_t n,
				     int32_t cutoff, DeviceHashTable cust,
				     const unsigned long long *cust_bits,
				     int64_t cust_dlen,
				     unsigned long long *out);
hipError_t launch_orders_filter_compact(hipStream_t s, const int64_t *okey,
					const int64_t *ckey,
					const int32_t *odate,
					const int32_t *prio, int64_t n,
					int32_t date_lo, int32_t date_hi,
					int64_t *out_ckey,
					int64_t *out_okey, int64_t *out_pay,
					unsigned long long *out_count);
hipError_t launch_probe_cust_compact(hipStream_t s, const int64_t *ckey,
				     const int64_t *okey, const int64_t *pay,
				     int64_t n, DeviceHashTable cust,
				     const unsigned long long *cust_bits,
				     int64_t cust_dlen,
				     int64_t *out_okey, int64_t *out_pay,
				     unsigned long long *out_count);
hipError_t launch_insert_orders(hipStream_t s, const int64_t *okey,
				const int64_t *pay, int64_t n,
				DeviceHashTable ord);

/* Motion partition: count then scatter rows to per-dest buffers */
hipError_t launch_part_count(hipStream_t s, const int64_t *key, int64_t n,
			     int nseg, unsigned long long *counts);
hipError_t launch_part_scatter3(hipStream_t s, const int64_t *key, int64_t n,
				int nseg,
				const int64_t *a, const int64_t *b,
				const int64_t *c,
				unsigned long long *offsets,
				int64_t *oa, int64_t *ob, int64_t *oc);

hipError_t launch_fill_u64(hipStream_t s, unsigned long long *p,
			   uint64_t n, unsigned long long v);
hipError_t launch_narrow_i32_u8(hipStream_t s, const int32_t *in,
				int64_t n, uint8_t *out);
hipError_t launch_sj_build(hipStream_t s, const int64_t *keys,
			   const int64_t *vals, int64_t n,
			   unsigned long long *tkeys,
			   unsigned long long *tvals, uint64_t nslots);
hipError_t launch_sj_probe(hipStream_t s, const int64_t *keys,
			   const int64_t *idxs, int64_t n,
			   const unsigned long long *tkeys,
			   const unsigned long long *tvals, uint64_t nslots,
			   int64_t *out_idx, int64_t *out_val,
			   unsigned long long *out_count);
hipError_t launch_gb_part_scatter_idx(hipStream_t s, const int64_t *keys,
				      int64_t n, int64_t base, int shift,
				      unsigned long long *cursors,
				      int64_t *out_k, int64_t *out_v);
hipError_t launch_gb_prefix2_u64(hipStream_t s,
				 const unsigned long long *cnt, int n,
				 unsigned long long *work,
				 unsigned long long *pristine);
hipError_t launch_gb_part_count(hipStream_t s, const int64_t *keys,
				int64_t n, int shift,
				unsigned long long *counts);
hipError_t launch_gb_part_scatter(hipStream_t s, const int64_t *keys,
				  const int64_t *vals, int64_t n, int shift,
				  unsigned long long *cursors,
				  int64_t *out_k, int64_t *out_v);
hipError_t launch_groupby_build(hipStream_t s, const int64_t *keys,
				const int64_t *vals, int64_t n,
				unsigned long long *tkeys,
				unsigned long long *tsum,
				unsigned long long *tcnt, uint64_t nslots);
hipError_t launch_groupby_compact(hipStream_t s,
				  const unsigned long long *tkeys,
				  const unsigned long long *tsum,
				  const unsigned long long *tcnt,
				  uint64_t nslots, int64_t *out_keys,
				  int64_t *out_sums, int64_t *out_cnts,
				  unsigned long long *out_n, uint64_t cap);
/* dense-orderkey fast path (q3_dense.hip) */
hipError_t launch_dn_build_orders(hipStream_t s, const int64_t *okey,
				  const int64_t *ckey, const int32_t *odate,
				  const int32_t *prio, int64_t n,
				  int32_t cutoff, DeviceHashTable cust,
				  const unsigned long long *cust_bits,
				  int64_t cust_dlen, unsigned long long *pr,
				  int64_t dense_len, unsigned long long *bloom,
				  uint64_t bwords,
				  unsigned long long *match_count);
hipError_t launch_dn_insert_orders(hipStream_t s, const int64_t *okey,
				   const int64_t *rowpay, int64_t n,
				   unsigned long long *pr,
				   int64_t dense_len,
				   unsigned long long *bloom,
				   uint64_t bwords);
hipError_t launch_dn_probe_lineitem(hipStream_t s, const int64_t *okey,
				    const int32_t *shipdate,
				    const int64_t *price,
				    const int64_t *disc, int64_t n,
				    int32_t cutoff, unsigned long long *pr,
				    int64_t dense_len,
				    unsigned long long *bloom,
				    uint64_t bwords,
				    unsigned long long *join_rows,
				    unsigned long long *surv,
				    uint64_t region,
				    unsigned long long *counts,
				    unsigned long long *ovf, int grid);
hipError_t launch_td_insert(hipStream_t s, const uint8_t *pool,
			    const unsigned long long *offs,
			    const uint32_t *lens, const uint8_t *nulls,
			    int64_t n, unsigned long long *slots,
			    uint64_t nslots, uint32_t *rowslot,
			    unsigned long long *err);
hipError_t launch_td_map(hipStream_t s, const uint32_t *rowslot,
			 int64_t n, const int32_t *slot_to_id,
			 int32_t *codes);

/* ---- generic compiled plans (plan.hip / plan.cpp) ---- */

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
	const unsigned long long *bits;	/* dense membership bitmap, or */
	int64_t dlen;
	const unsigned long long *hkeys;	/* hash semi-set (key^msb, 0=empty) */
	uint64_t hslots;
};

struct PlanAggDev
{
	int kind;		/* gg_plan_aggkind */
	int nf;
	const void *col[3];
	const uint8_t *nulls[3];
	int width[3];
	int8_t mod[3];		/* gg_plan_fmod */
};

struct PlanDev
{
	int64_t n;
	int npreds, njoins, naggs, ngroup;
	PlanPredDev preds[GG_PLAN_MAX_PREDS];
	PlanJoinDev joins[GG_PLAN_MAX_JOINS];
	const void *gcol[2];
	const uint8_t *gnulls[2];
	int gwidth[2];
	unsigned long long *tkeys;	/* nslots; INT64_MIN pattern = empty */
	unsigned long long *tvals;	/* nslots * naggs * 2 (lo, hi) */
	uint64_t nslots;
	unsigned long long *err;
	PlanAggDev aggs[GG_PLAN_MAX_AGGS];
};

struct PlanBuildDev
{
	const void *key;
	const uint8_t *knulls;
	int kw;
	int64_t n;
	PlanPredDev preds[GG_PLAN_MAX_PREDS];
	int npreds;
	unsigned long long *bits;
	int64_t dlen;
	unsigned long long *hkeys;
	uint64_t hslots;
};

hipError_t launch_plan_build(hipStream_t s, const PlanBuildDev &b);
hipError_t launch_plan_scan_agg(hipStream_t s, const PlanDev &p);
hipError_t launch_plan_compact(hipStream_t s,
			       const unsigned long long *tkeys,
			       const unsigned long long *tvals,
			       uint64_t nslots, int naggs,
			       unsigned long long *out,
			       unsigned long long *out_n, uint64_t cap);

/* plan.cpp */
gg_status exec_plan(Pipeline *p, void *arena, size_t bytes,
		    size_t *written);

/* plan_rtc.cpp — per-plan kernel specialization via hipRTC.
 * bake/nbake: optional observed group codes (from a prior execute of
 * the same plan over the same immutable tables); when given, the
 * generated kernel indexes a direct LDS accumulator per baked code
 * (compare chain, no key probing) and routes unseen codes through
 * the global table. */
gg_status plan_rtc_compile(const PlanDev &D, bool has_gnull0,
			   bool has_gnull1, std::shared_ptr<void> *out,
			   const long long *bake = nullptr, int nbake = 0,
			   bool fast = false);
gg_status plan_rtc_launch(hipStream_t s, const std::shared_ptr<void> &h,
			  PlanDev P, int grid, int block);

/* engine_abi.cpp helpers shared with plan.cpp */
Table *engine_table(gg_table h);
gg_status engine_cached_max_i64(Engine &e, Pipeline *p, const int64_t *col,
				int64_t n, unsigned long long *ctr,
				unsigned long long *out);
/* signed min/max of a registered column (width 1/4/8), cached on the
 * engine (columns are immutable) */
gg_status engine_col_minmax(Engine &e, const void *dcol, int width,
			    int64_t n, long long *mn, long long *mx);
hipError_t launch_minmax_i64(hipStream_t s, const void *v, int width,
			     int64_t n, unsigned long long *out2);

#define GG_MT_MAX_ATTS 32

/* memtuple.hip — MemTuple codec */
struct MtBind
{
	int32_t natts;
	int32_t column_align;
	int32_t null_bitmap_extra;
	int32_t var_start;
	int32_t nvar;		/* number of varlena attrs */
	int32_t offset[GG_MT_MAX_ATTS];
	int32_t len[GG_MT_MAX_ATTS];	/* varoffset width (2) for varlena */
	int32_t len_aligned[GG_MT_MAX_ATTS];
	int32_t null_byte[GG_MT_MAX_ATTS];
	uint8_t null_mask[GG_MT_MAX_ATTS];
	int32_t phys[GG_MT_MAX_ATTS];
	int8_t is_var[GG_MT_MAX_ATTS];
	int8_t align_of[GG_MT_MAX_ATTS];	/* datum alignment (bytes) */
};
int mt_compute_binding(int natts, const int32_t *attlen,
		       const char *attalign, MtBind *out);
int mt_compute_binding_large(int natts, const int32_t *attlen,
			     const char *attalign, MtBind *out);
hipError_t launch_mt_encode(hipStream_t s, const MtBind *b,
			    const MtBind *bl, const void *const *cols,
			    const int64_t *const *var_offs,
			    const uint8_t *const *nulls, int64_t nrows,
			    const int64_t *offs, uint8_t *out);
hipError_t launch_mt_decode(hipStream_t s, const MtBind *b,
			    const MtBind *bl, const int64_t *offs,
			    int64_t nrows, const uint8_t *in,
			    int64_t in_len, void *const *cols,
			    unsigned long long *const *var_out_offs,
			    uint32_t *const *var_out_lens,
			    uint8_t *const *nulls, unsigned long long *err);

hipError_t launch_dn_build_orders_q5_u8(hipStream_t s, const int64_t *okey,
					 const int64_t *ckey,
					 const int32_t *odate, int64_t n,
					 int32_t date_lo, int32_t date_hi,
					 DeviceHashTable cust,
					 const uint8_t *cust_dense,
					 int64_t cust_dlen, uint8_t *pay8,
					 int64_t dense_len,
					 unsigned long long *match_count);
hipError_t launch_dn_q5_compact(hipStream_t s, const int64_t *okey,
				int64_t n, const uint8_t *pay8,
				int64_t dense_len, int64_t region,
				unsigned long long *out,
				unsigned long long *counts);
hipError_t launch_dn_q5_gather(hipStream_t s,
			       const unsigned long long *comp,
			       const unsigned long long *counts,
			       int64_t region, int64_t nregions,
			       const int64_t *skey, const int64_t *price,
			       const int64_t *disc,
			       const uint8_t *supp_dense, int64_t supp_dlen,
			       unsigned long long *acc,
			       unsigned long long *join_rows,
			       unsigned long long *overflow);
hipError_t launch_dn_insert_orders_q5_u8(hipStream_t s, const int64_t *okey,
					 const int64_t *rownat, int64_t n,
					 uint8_t *pay8, int64_t dense_len);
hipError_t launch_dn_probe_lineitem_q5_u8(hipStream_t s,
					  const int64_t *okey,
					  const int64_t *skey,
					  const int64_t *price,
					  const int64_t *disc, int64_t n,
					  const uint8_t *pay8,
					  int64_t dense_len,
					  const uint8_t *supp_dense,
					  int64_t supp_dlen,
					  unsigned long long *acc,
					  unsigned long long *join_rows);
int dn_probe_grid(int64_t n);
hipError_t launch_dn_q3_stats_surv(hipStream_t s,
				   const unsigned long long *surv,
				   const unsigned long long *counts,
				   uint64_t region, int64_t nregions,
				   const unsigned long long *pr,
				   unsigned long long *out5,
				   unsigned int *hist64k);
hipError_t launch_dn_q3_threshold2(hipStream_t s,
				   const unsigned int *hist64k, int64_t k,
				   unsigned long long *out_thr);
hipError_t launch_dn_q3_collect_surv(hipStream_t s,
				     const unsigned long long *surv,
				     const unsigned long long *counts,
				     uint64_t region, int64_t nregions,
				     const unsigned long long *pr,
				     const unsigned long long *thr_ptr,
				     gg_q3_result_row *out,
				     unsigned long long *out_count,
				     uint64_t cap);
hipError_t launch_dn_q3_clear_surv(hipStream_t s,
				   const unsigned long long *surv,
				   const unsigned long long *counts,
				   uint64_t region, int64_t nregions,
				   unsigned long long *pr);
hipError_t launch_dn_q3_stats(hipStream_t s, const unsigned long long *pr,
			      int64_t dense_len, unsigned long long *out5);
hipError_t launch_dn_q3_hist(hipStream_t s, const unsigned long long *pr,
			     int64_t dense_len,
			     const unsigned long long *stats5,
			     unsigned int *hist64k);
hipError_t launch_dn_q3_collect(hipStream_t s,
				const unsigned long long *pr,
				int64_t dense_len,
				const unsigned long long *thr_ptr,
				gg_q3_result_row *out,
				unsigned long long *out_count, uint64_t cap);

hipError_t launch_dsb_decode_text(hipStream_t s, const uint8_t *stream,
				  const uint8_t *spill,
				  const int64_t *offsets,
				  const int32_t *sizes,
				  const int32_t *rowcounts,
				  const int64_t *out_offsets,
				  const int64_t *pool_offsets,
				  int32_t nblocks, int version,
				  uint8_t *pool,
				  unsigned long long *out_offs,
				  uint32_t *out_lens, uint8_t *out_nulls,
				  unsigned long long *err,
				  unsigned long long *src_offs);
hipError_t launch_dsb_text_copy(hipStream_t s, const uint8_t *stream,
				const uint8_t *spill,
				const unsigned long long *src_offs,
				const unsigned long long *out_offs,
				const uint32_t *out_lens, int64_t nrows,
				uint8_t *pool);
hipError_t launch_dsb_decode2(hipStream_t s, const uint8_t *stream,
			      const uint8_t *spill,
			      const int64_t *offsets,
			      const int32_t *sizes,
			      const int32_t *rowcounts,
			      const int64_t *out_offsets, int32_t nblocks,
			      int version, int datumlen, void *out_vals,
			      uint8_t *out_nulls, int out_width,
			      unsigned long long *err);
hipError_t launch_dsb_decode(hipStream_t s, const uint8_t *stream,
			     const int64_t *offsets, const int32_t *sizes,
			     const int32_t *rowcounts,
			     const int64_t *out_offsets, int32_t nblocks,
			     int version, int datumlen, void *out_vals,
			     uint8_t *out_nulls, int out_width,
			     unsigned long long *err);
hipError_t launch_radix_sort_pass(hipStream_t s,
				  const unsigned long long *keys,
				  const unsigned long long *pay, int64_t n,
				  int shift, unsigned int *block_hist,
				  int nblocks, unsigned long long *out_keys,
				  unsigned long long *out_pay);
int radix_sort_nblocks(int64_t n);

}				/* namespace gg */

#endif
