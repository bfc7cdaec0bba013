/*
 * C-ABI engine implementation (see include/engine_abi.h for the
 * boundary contract and reference citations).  Host orchestration for
 * the HIP kernels in kernels.hip + RCCL exchanges in comm.cpp.
 *
 * PRODUCT code: self-contained (HIP runtime + RCCL only), no torch, no
 * oracle/.  Loaded via ctypes by greengage_amd/ and — in a live
 * Greengage deployment — by the segment shim.
 */
#include <algorithm>
#include <atomic>
#include <thread>
#include <cstdarg>
#include <cstdio>
#include <chrono>
#include <cstring>
#include <vector>

/* zlib (pg_compression.c:253 binds compress2/uncompress) */
#include <zlib.h>
/* zstd public API — libzstd.so.1 ships without dev headers here */
extern "C" size_t ZSTD_decompress(void *dst, size_t dstCap,
				  const void *src, size_t srcSize);
extern "C" unsigned ZSTD_isError(size_t code);

#include "engine_internal.h"
#include "../../include/gg_gen.h"

namespace gg
{

static thread_local char g_err[512];

gg_status fail(gg_status st, const char *fmt, ...)
{
	va_list ap;

	va_start(ap, fmt);
	vsnprintf(g_err, sizeof(g_err), fmt, ap);
	va_end(ap);
	return st;
}

Engine &engine()
{
	static Engine e;

	return e;
}

/* comm.cpp */
gg_status comm_make_id(void *out_id128);
gg_status comm_init(const void *id128);
gg_status comm_destroy();
bool comm_ready();
gg_status comm_allgather_u64(const void *dev_send, void *dev_recv,
			     size_t count);
gg_status comm_alltoallv_i64(const int64_t *send_base,
			     const unsigned long long *send_offs,
			     const unsigned long long *send_cnts, int64_t *recv_base,
			     const unsigned long long *recv_offs,
			     const unsigned long long *recv_cnts);

static uint64_t next_pow2(uint64_t v)
{
	uint64_t p = 1024;	/* floor tuned for hash-table slot counts */

	while (p < v)
		p <<= 1;
	return p;
}

/* exact power-of-two ceiling (no floor) — partition fan-outs */
static uint64_t pow2_ceil(uint64_t v)
{
	uint64_t p = 1;

	while (p < v)
		p <<= 1;
	return p;
}

static size_t coltype_size(gg_coltype t)
{
	switch (t)
	{
		case GG_COL_INT64:
		case GG_COL_DEC64_S2:
			return 8;
		case GG_COL_INT32:
			return 4;
		case GG_COL_CHAR1:
			return 1;
	}
	return 0;
}

#define GG_TRY(expr) \
	do { gg_status _s = (expr); if (_s != GG_OK) return _s; } while (0)

/* device scratch counter helpers */
static gg_status dev_counter(unsigned long long **p)
{
	GG_HIP(hipMalloc((void **) p, sizeof(unsigned long long)));
	GG_HIP(hipMemset(*p, 0, sizeof(unsigned long long)));
	return GG_OK;
}

static uint32_t ao_crc32c(const uint8_t *p, int64_t len);

static gg_status read_counter(unsigned long long *p, unsigned long long *out)
{
	GG_HIP(hipMemcpy(out, p, sizeof(*out), hipMemcpyDeviceToHost));
	return GG_OK;
}

/* max over an immutable registered column, cached per pipeline */
static gg_status
cached_max_i64(Engine &e, Pipeline *p, const int64_t *col, int64_t n,
	       unsigned long long *ctr, unsigned long long *out)
{
	for (auto &kv : p->maxk_cache)
		if (kv.first.first == (const void *) col &&
		    kv.first.second == n)
		{
			*out = kv.second;
			return GG_OK;
		}
	GG_HIP(hipMemsetAsync(ctr, 0, 8, e.stream));
	GG_HIP(launch_max_i64(e.stream, col, n, ctr));
	GG_HIP(hipStreamSynchronize(e.stream));
	GG_TRY(read_counter(ctr, out));
	p->maxk_cache.push_back({{(const void *) col, n}, *out});
	return GG_OK;
}

/* shared with plan.cpp */
gg_status
engine_cached_max_i64(Engine &e, Pipeline *p, const int64_t *col, int64_t n,
		      unsigned long long *ctr, unsigned long long *out)
{
	return cached_max_i64(e, p, col, n, ctr, out);
}

/* signed min/max of an immutable registered column, cached on the
 * engine; one scan per column ever */
gg_status
engine_col_minmax(Engine &e, const void *dcol, int width, int64_t n,
		  long long *mn, long long *mx)
{
	auto it = e.mm_cache.find(dcol);

	if (it != e.mm_cache.end())
	{
		*mn = it->second.first;
		*mx = it->second.second;
		return GG_OK;
	}
	{
		unsigned long long seed[2] = {~0ull, 0ull};
		unsigned long long *d = nullptr;

		GG_HIP(hipMalloc((void **) &d, 16));
		GG_HIP(hipMemcpyAsync(d, seed, 16, hipMemcpyHostToDevice,
				      e.stream));
		hipError_t he = launch_minmax_i64(e.stream, dcol, width, n, d);

		if (he != hipSuccess)
		{
			(void) hipFree(d);
			return fail(GG_EGPU, "minmax launch: %s",
				    hipGetErrorString(he));
		}
		GG_HIP(hipMemcpyAsync(seed, d, 16, hipMemcpyDeviceToHost,
				      e.stream));
		GG_HIP(hipStreamSynchronize(e.stream));
		(void) hipFree(d);
		*mn = (long long) (seed[0] ^ 0x8000000000000000ull);
		*mx = (long long) (seed[1] ^ 0x8000000000000000ull);
		e.mm_cache[dcol] = {*mn, *mx};
	}
	return GG_OK;
}

/* ---------------- int128 host helpers (combine/finalize) ---------------- */

typedef __int128 i128;
typedef unsigned __int128 u128;

static i128 mk128(uint64_t lo, int64_t hi)
{
	return ((i128) hi << 64) | (i128) lo;
}

static void split128(i128 v, uint64_t *lo, int64_t *hi)
{
	*lo = (uint64_t) (u128) v;
	*hi = (int64_t) (v >> 64);
}

/* ---------------- lifecycle ---------------- */

extern "C" const char *gg_engine_last_error(void)
{
	return g_err;
}

extern "C" const char *gg_engine_build_info(void)
{
	return "greengage_amd engine: HIP gfx950 (CDNA4) + RCCL; built "
		__DATE__ " " __TIME__;
}

extern "C" gg_status gg_engine_init(const gg_engine_config *cfg)
{
	Engine &e = engine();

	if (e.inited)
		return fail(GG_ESTATE, "engine already initialized");
	if (!cfg || cfg->n_segments < 1 || cfg->segment_id < 0 ||
	    cfg->segment_id >= cfg->n_segments)
		return fail(GG_EINVAL, "bad engine config");
	e.cfg = *cfg;
	GG_HIP(hipSetDevice(cfg->device));
	GG_HIP(hipStreamCreate(&e.stream));
	e.inited = true;
	return GG_OK;
}

extern "C" gg_status gg_engine_shutdown(void)
{
	Engine &e = engine();

	if (!e.inited)
		return GG_OK;
	comm_destroy();
	for (auto *t : e.tables)
	{
		if (t)
			for (auto &c : t->cols)
			{
				(void) hipFree(c.dev);
				if (c.nulls)
					(void) hipFree(c.nulls);
			}
		delete t;
	}
	e.tables.clear();
	e.mm_cache.clear();
	for (auto &kv : e.escratch)
		(void) hipFree(kv.second.first);
	e.escratch.clear();
	for (auto &kv : e.ehscratch)
		(void) hipHostFree(kv.second.first);
	e.ehscratch.clear();
	for (auto *p : e.pipelines)
		delete p;
	e.pipelines.clear();
	(void) hipStreamDestroy(e.stream);
	e.stream = nullptr;
	e.inited = false;
	return GG_OK;
}

/* ---------------- tables ---------------- */

static gg_status add_col(Table *t, const char *name, gg_coltype type,
			 int64_t nrows, const void *host_src,
			 void **out_dev)
{
	Table::Col c;

	c.name = name;
	c.type = type;
	c.bytes = (size_t) nrows * coltype_size(type);
	GG_HIP(hipMalloc(&c.dev, c.bytes ? c.bytes : 1));
	if (host_src)
		GG_HIP(hipMemcpy(c.dev, host_src, c.bytes,
				 hipMemcpyHostToDevice));
	t->cols.push_back(c);
	if (out_dev)
		*out_dev = c.dev;
	return GG_OK;
}

extern "C" gg_status
gg_engine_register_table(const char *name, const gg_column_desc *cols,
			 int ncols, int64_t nrows, gg_table *out)
{
	Engine &e = engine();

	if (!e.inited)
		return fail(GG_ESTATE, "engine not initialized");
	if (!name || !cols || ncols <= 0 || nrows < 0 || !out)
		return fail(GG_EINVAL, "bad register_table args");
	Table *t = new Table();

	t->name = name;
	t->nrows = nrows;
	for (int i = 0; i < ncols; i++)
	{
		if (cols[i].device_data)
		{
			Table::Col c;

			c.name = cols[i].name;
			c.type = cols[i].type;
			c.dev = const_cast<void *>(cols[i].device_data);
			c.bytes = 0;	/* not owned */
			t->cols.push_back(c);
		}
		else
		{
			gg_status s = add_col(t, cols[i].name, cols[i].type,
					      nrows, cols[i].host_data,
					      nullptr);
			if (s != GG_OK)
			{
				delete t;
				return s;
			}
		}
	}
	e.tables.push_back(t);
	*out = (gg_table) (e.tables.size() - 1);
	return GG_OK;
}

/* ---- zero-copy AO block parse (the fast path for the int decoders):
 * descriptors point INTO the original segfile bytes for uncompressed
 * Small/NonBulkDense/BulkDense blocks; only decompressed content and
 * reassembled LargeContent goes to a spill buffer.  Checksums verify
 * and codecs run OpenMP-parallel across blocks — the sequential
 * single-thread walk only reads headers. ---- */

/* simple dynamic parallel-for over host threads; grain = items per
 * grab (use 1 for few/large items — e.g. per-partition memcpys — or
 * the work degenerates to a couple of threads) */
template <typename F>
static void
par_for_grain(int64_t n, int64_t grain, F fn)
{
	unsigned nt = std::thread::hardware_concurrency();

	if (nt > 32)
		nt = 32;
	if (nt < 2 || n < grain || n < 2)
	{
		for (int64_t i = 0; i < n; i++)
			fn(i);
		return;
	}
	std::vector<std::thread> ths;
	std::atomic<int64_t> next(0);

	for (unsigned t = 0; t < nt; t++)
		ths.emplace_back([&]()
		{
			for (;;)
			{
				int64_t i = next.fetch_add(grain);

				if (i >= n)
					return;
				int64_t hi = i + grain < n ? i + grain : n;

				for (; i < hi; i++)
					fn(i);
			}
		});
	for (auto &th : ths)
		th.join();
}

/* legacy grain (many small items: AO block jobs) */
template <typename F>
static void
ao_parallel_for(int64_t n, F fn)
{
	par_for_grain(n, 64, fn);
}

struct AoDesc
{
	int64_t off;		/* >=0: stream offset; <0: spill (-off-1) */
	int32_t size;
	int32_t rowcount;
};

struct AoJob
{
	int64_t src;		/* stream offset of stored bytes */
	int32_t stored;		/* compressed (or raw) length */
	int32_t datalen;	/* uncompressed length */
	int64_t dst;		/* spill offset */
	int comptype;		/* 0 = plain copy (fragment), 1/2 codec */
};

static gg_status
ao_parse_blocks(const uint8_t *stream, int64_t stream_len, int checksums,
		int ao_version, int comptype, std::vector<AoDesc> &descs,
		std::vector<uint8_t> &spill)
{
	if (!stream || stream_len < 0)
		return fail(GG_EINVAL, "bad ao stream");
	if (ao_version < 2)
		return fail(GG_EINVAL, "ao_version %d unsupported",
			    ao_version);

	struct Blk
	{
		int64_t pos;
		int64_t overall;
		int kind;
	};
	std::vector<Blk> blks;
	std::vector<AoJob> jobs;
	int64_t pos = 0, expect_rownum = -1, spill_len = 0;
	int64_t large_remaining = 0, large_dst = 0;

	while (pos < stream_len)
	{
		uint32_t w0, w1;

		if (pos + 8 > stream_len)
			return fail(GG_EINVAL, "truncated AO header at %lld",
				    (long long) pos);
		std::memcpy(&w0, stream + pos, 4);
		std::memcpy(&w1, stream + pos + 4, 4);
		int kind = (int) ((w0 >> 28) & 7);
		int has_frn = (int) ((w0 >> 27) & 1);
		int32_t rowcount, datalen, complen = 0;

		if (kind == 1)
		{
			rowcount = (int32_t) ((w0 >> 10) & 0x3FFF);
			datalen = (int32_t) (((w0 & 0x3FF) << 11) |
					     ((w1 >> 21) & 0x7FF));
			complen = (int32_t) (w1 & 0x1FFFFF);
		}
		else if (kind == 2)
		{
			if (large_remaining > 0)
				return fail(GG_EINVAL,
					    "nested LargeContent at %lld",
					    (long long) pos);
			rowcount = (int32_t) (((w0 & 0x7FFFFF) << 2) |
					      ((w1 >> 30) & 3));
			datalen = 0;
			int32_t biglen = (int32_t) (w1 & 0x3FFFFFFF);

			large_remaining = biglen;
			large_dst = spill_len;
			descs.push_back({-(spill_len + 1), biglen,
					 rowcount});
			spill_len += biglen;
		}
		else if (kind == 3)
		{
			datalen = (int32_t) (w0 & 0x1FFFFF);
			rowcount = (int32_t) (w1 & 0x3FFFFFFF);
		}
		else if (kind == 4)
		{
			uint32_t e1;
			int64_t ext_at = pos + 8 + (checksums ? 8 : 0);

			if (ext_at + 8 > stream_len)
				return fail(GG_EINVAL,
					    "truncated BulkDense at %lld",
					    (long long) pos);
			datalen = (int32_t) (((w0 & 0x3FF) << 11) |
					     ((w1 >> 21) & 0x7FF));
			complen = (int32_t) (w1 & 0x1FFFFF);
			std::memcpy(&e1, stream + ext_at + 4, 4);
			rowcount = (int32_t) (e1 & 0x3FFFFFFF);
		}
		else
			return fail(GG_EINVAL,
				    "unsupported AoHeaderKind %d at %lld",
				    kind, (long long) pos);
		if (complen != 0 && comptype == 0)
			return fail(GG_EINVAL,
				    "compressed AO block at %lld but "
				    "comptype none", (long long) pos);
		if (large_remaining > 0 && kind != 1 && kind != 2)
			return fail(GG_EINVAL,
				    "expected SmallContent fragment at "
				    "%lld", (long long) pos);

		int32_t stored = complen ? complen : datalen;
		int64_t hdr_end = pos + 8 + (checksums ? 8 : 0) +
			(kind == 4 ? 8 : 0) + (has_frn ? 8 : 0);
		int64_t padded = ((int64_t) stored + 7) & ~(int64_t) 7;
		int64_t overall = hdr_end + padded;

		if (overall > stream_len || datalen < 0 || rowcount < 0)
			return fail(GG_EINVAL, "AO block at %lld overruns",
				    (long long) pos);
		blks.push_back({pos, overall, kind});
		if (has_frn)
		{
			int64_t frn;

			std::memcpy(&frn, stream + pos + 8 +
				    (checksums ? 8 : 0) +
				    (kind == 4 ? 8 : 0), 8);
			if (expect_rownum >= 0 && frn != expect_rownum)
				return fail(GG_EINVAL,
					    "firstRowNum discontinuity at "
					    "%lld", (long long) pos);
			expect_rownum = frn + rowcount;
		}
		bool fragment = (large_remaining > 0 && kind == 1);

		if (kind == 2)
			;	/* frame opened above */
		else if (fragment)
		{
			if (datalen > large_remaining)
				return fail(GG_EINVAL,
					    "LargeContent overrun at %lld",
					    (long long) pos);
			jobs.push_back({hdr_end, stored, datalen, large_dst,
					complen ? comptype : 0});
			large_dst += datalen;
			large_remaining -= datalen;
		}
		else if (complen == 0)
			descs.push_back({hdr_end, datalen, rowcount});
		else
		{
			jobs.push_back({hdr_end, stored, datalen, spill_len,
					comptype});
			descs.push_back({-(spill_len + 1), datalen,
					 rowcount});
			spill_len += datalen;
		}
		pos = overall;
	}
	if (large_remaining > 0)
		return fail(GG_EINVAL, "LargeContent truncated");

	/* parallel checksum verify (std::thread — hipcc must not see
	 * -fopenmp, it leaks into device codegen) */
	if (checksums)
	{
		std::atomic<int64_t> bad_hdr(-1), bad_blk(-1);

		ao_parallel_for((int64_t) blks.size(), [&](int64_t b)
		{
			const Blk &bl = blks[b];
			uint32_t stored_hdr, stored_blk;

			std::memcpy(&stored_blk, stream + bl.pos + 8, 4);
			std::memcpy(&stored_hdr, stream + bl.pos + 12, 4);
			if (ao_crc32c(stream + bl.pos, 12) != stored_hdr)
				bad_hdr.store(bl.pos);
			else
			{
				int64_t end = bl.kind == 2 ? bl.pos + 16
					: bl.overall;

				if (ao_crc32c(stream + bl.pos + 16,
					      end - bl.pos - 16) !=
				    stored_blk)
					bad_blk.store(bl.pos);
			}
		});
		if (bad_hdr.load() >= 0)
			return fail(GG_EINVAL, "AO header checksum "
				    "mismatch at %lld",
				    (long long) bad_hdr.load());
		if (bad_blk.load() >= 0)
			return fail(GG_EINVAL, "AO block checksum "
				    "mismatch at %lld",
				    (long long) bad_blk.load());
	}

	/* parallel decompress / fragment copy into the spill */
	spill.resize((size_t) spill_len);
	{
		std::atomic<int64_t> bad(-1);

		ao_parallel_for((int64_t) jobs.size(), [&](int64_t j)
		{
			const AoJob &jb = jobs[j];
			bool ok = true;

			if (jb.comptype == 0)
				std::memcpy(spill.data() + jb.dst,
					    stream + jb.src,
					    (size_t) jb.datalen);
			else if (jb.comptype == 1)
			{
				uLongf dl = (uLongf) jb.datalen;

				ok = uncompress(spill.data() + jb.dst, &dl,
						stream + jb.src,
						(uLong) jb.stored) == Z_OK
					&& dl == (uLongf) jb.datalen;
			}
			else
			{
				size_t dl = ZSTD_decompress(
					spill.data() + jb.dst,
					(size_t) jb.datalen,
					stream + jb.src,
					(size_t) jb.stored);

				ok = !ZSTD_isError(dl) &&
					dl == (size_t) jb.datalen;
			}
			if (!ok)
				bad.store(jb.src);
		});
		if (bad.load() >= 0)
			return fail(GG_EINVAL, "AO decompress failed at "
				    "%lld", (long long) bad.load());
	}
	return GG_OK;
}

/* upload + device datum-stream decode of parsed blocks; on success the
 * caller owns *d_vals (device) — *d_nulls is checked NOT NULL here */
static gg_status
ao_decode_to_device(Engine &e, const uint8_t *stream, int64_t stream_len,
		    const std::vector<AoDesc> &descs,
		    const std::vector<uint8_t> &spill, int dsb_version,
		    int datumlen, int out_width, void **out_d_vals,
		    uint8_t **out_d_nulls, int64_t *out_rows)
{
	size_t nb = descs.size();
	int64_t total_rows = 0;
	std::vector<int64_t> offs(nb), out_offs(nb);
	std::vector<int32_t> sizes(nb), rows(nb);

	for (size_t i = 0; i < nb; i++)
	{
		offs[i] = descs[i].off;
		sizes[i] = descs[i].size;
		rows[i] = descs[i].rowcount;
		out_offs[i] = total_rows;
		total_rows += descs[i].rowcount;
	}
	*out_rows = total_rows;
	*out_d_vals = nullptr;
	if (total_rows == 0)
		return GG_OK;

	gg_status st = GG_OK;
	uint8_t *d_stream = nullptr, *d_spill = nullptr, *d_nulls = nullptr;
	int64_t *d_offs = nullptr, *d_oo = nullptr;
	int32_t *d_sizes = nullptr, *d_rows = nullptr;
	void *d_vals = nullptr;
	unsigned long long *d_err = nullptr;

#define GG_HIP_AD(x) \
	{ hipError_t e_ = (x); \
	  if (st == GG_OK && e_ != hipSuccess) \
		st = fail(GG_EGPU, "ao_decode: %s", \
			  hipGetErrorString(e_)); }
	GG_HIP_AD(hipMalloc((void **) &d_stream,
			    stream_len ? (size_t) stream_len : 1));
	GG_HIP_AD(hipMalloc((void **) &d_spill,
			    spill.size() ? spill.size() : 1));
	GG_HIP_AD(hipMalloc((void **) &d_offs, nb * 8));
	GG_HIP_AD(hipMalloc((void **) &d_oo, nb * 8));
	GG_HIP_AD(hipMalloc((void **) &d_sizes, nb * 4));
	GG_HIP_AD(hipMalloc((void **) &d_rows, nb * 4));
	GG_HIP_AD(hipMalloc(&d_vals, (size_t) total_rows * out_width));
	GG_HIP_AD(hipMalloc((void **) &d_nulls, (size_t) total_rows));
	GG_HIP_AD(hipMalloc((void **) &d_err, 8));
	if (st == GG_OK)
	{
		if (stream_len)
			GG_HIP_AD(hipMemcpyAsync(d_stream, stream,
						 (size_t) stream_len,
						 hipMemcpyHostToDevice,
						 e.stream));
		if (spill.size())
			GG_HIP_AD(hipMemcpyAsync(d_spill, spill.data(),
						 spill.size(),
						 hipMemcpyHostToDevice,
						 e.stream));
		GG_HIP_AD(hipMemcpyAsync(d_offs, offs.data(), nb * 8,
					 hipMemcpyHostToDevice, e.stream));
		GG_HIP_AD(hipMemcpyAsync(d_oo, out_offs.data(), nb * 8,
					 hipMemcpyHostToDevice, e.stream));
		GG_HIP_AD(hipMemcpyAsync(d_sizes, sizes.data(), nb * 4,
					 hipMemcpyHostToDevice, e.stream));
		GG_HIP_AD(hipMemcpyAsync(d_rows, rows.data(), nb * 4,
					 hipMemcpyHostToDevice, e.stream));
		GG_HIP_AD(hipMemsetAsync(d_err, 0, 8, e.stream));
		GG_HIP_AD(launch_dsb_decode2(e.stream, d_stream, d_spill,
					     d_offs, d_sizes, d_rows, d_oo,
					     (int32_t) nb, dsb_version,
					     datumlen, d_vals, d_nulls,
					     out_width, d_err));
		GG_HIP_AD(hipStreamSynchronize(e.stream));
	}
	if (st == GG_OK)
	{
		unsigned long long herr = 0;

		GG_HIP_AD(hipMemcpy(&herr, d_err, 8,
				    hipMemcpyDeviceToHost));
		if (st == GG_OK && herr)
			st = fail(GG_EINVAL, "block decode error mask "
				  "0x%llx", herr);
	}
	if (st == GG_OK && !out_d_nulls)
	{
		/* caller wants a NOT NULL column (pipeline mounts) */
		std::vector<uint8_t> hn(total_rows);

		GG_HIP_AD(hipMemcpy(hn.data(), d_nulls,
				    (size_t) total_rows,
				    hipMemcpyDeviceToHost));
		for (int64_t r = 0; st == GG_OK && r < total_rows; r++)
			if (hn[r])
				st = fail(GG_EINVAL,
					  "NULL at row %lld in a NOT NULL "
					  "column", (long long) r);
	}
#undef GG_HIP_AD
	(void) hipFree(d_stream);
	(void) hipFree(d_spill);
	(void) hipFree(d_offs);
	(void) hipFree(d_oo);
	(void) hipFree(d_sizes);
	(void) hipFree(d_rows);
	(void) hipFree(d_err);
	if (st != GG_OK)
	{
		(void) hipFree(d_vals);
		(void) hipFree(d_nulls);
		return st;
	}
	*out_d_vals = d_vals;
	if (out_d_nulls)
		*out_d_nulls = d_nulls;
	else
		(void) hipFree(d_nulls);
	return GG_OK;
}

/*
 * Mount a REAL AO table: one AO segfile byte stream per column, the
 * way cdbbufferedread.c hands blocks up.  Each column goes through
 * the full storage layer (headers + CRC32C + zlib/zstd + datum-stream
 * decode) and lands DIRECTLY in a device-resident column — no host
 * round trip — then the table registers like any other.  Columns are
 * NOT NULL by default (a NULL anywhere errors); nullable mounts keep
 * the AO null bitmap as a device NULL-flag array on the column.
 */
extern "C" gg_status
gg_engine_register_table_ao(const char *name, const gg_ao_column *cols,
			    int ncols, gg_table *out)
{
	Engine &e = engine();

	if (!e.inited)
		return fail(GG_ESTATE, "engine not initialized");
	if (!name || !cols || ncols <= 0 || !out)
		return fail(GG_EINVAL, "bad register_table_ao args");

	Table *t = new Table();

	t->name = name;
	t->nrows = -1;

	gg_status st = GG_OK;

	for (int ci = 0; ci < ncols && st == GG_OK; ci++)
	{
		const gg_ao_column &ac = cols[ci];

		if (ac.text_dict)
		{
			/* TEXT column: decode varlena + GPU dictionary
			 * encode; codes become the u8 column, the sorted
			 * dictionary stays on the Table::Col for
			 * predicate-constant lookup */
			if (ac.type != GG_COL_CHAR1)
			{
				st = fail(GG_EINVAL, "text_dict column %s "
					  "must be char1", ac.name);
				break;
			}
			std::vector<uint64_t> toffs;
			std::vector<uint32_t> tlens;
			std::vector<uint8_t> tnulls, tpool;
			int64_t trows = 0, tpool_len = 0;

			{	/* size via a parse pass */
				std::vector<AoDesc> d2;
				std::vector<uint8_t> s2;

				st = ao_parse_blocks(ac.stream,
						     ac.stream_len,
						     ac.checksums,
						     ac.ao_version,
						     ac.comptype, d2, s2);
				if (st != GG_OK)
					break;
				for (auto &d3 : d2)
				{
					trows += d3.rowcount;
					tpool_len += d3.size;
				}
			}
			toffs.resize(trows + 1);
			tlens.resize(trows + 1);
			tnulls.resize(trows + 1);
			tpool.resize(tpool_len + 1);
			int64_t got_rows = 0, got_pool = 0;

			st = gg_engine_aocs_decode_ao_text(
				ac.stream, ac.stream_len, ac.checksums,
				ac.ao_version, ac.dsb_version, ac.comptype,
				toffs.data(), tlens.data(), tnulls.data(),
				trows + 1, tpool.data(), tpool_len + 1,
				&got_rows, &got_pool);
			if (st != GG_OK)
				break;

			std::vector<int32_t> codes(got_rows);
			std::vector<uint8_t> dbytes(got_pool + 16);
			std::vector<int64_t> doffs(257);
			int32_t nd = 0;

			st = gg_engine_text_dict_encode(
				tpool.data(), toffs.data(), tlens.data(),
				tnulls.data(), got_rows,
				255 /* u8 codes */ , codes.data(),
				dbytes.data(), (int64_t) dbytes.size(),
				doffs.data(), &nd);
			if (st != GG_OK)
				break;
			for (int64_t r = 0; r < got_rows; r++)
				if (codes[r] < 0)
				{
					st = fail(GG_EINVAL,
						  "column %s has NULL at "
						  "row %lld", ac.name,
						  (long long) r);
					break;
				}
			if (st != GG_OK)
				break;
			if (t->nrows < 0)
				t->nrows = got_rows;
			else if (t->nrows != got_rows)
			{
				st = fail(GG_EINVAL, "column %s has %lld "
					  "rows, table has %lld", ac.name,
					  (long long) got_rows,
					  (long long) t->nrows);
				break;
			}
			std::vector<uint8_t> u8codes(got_rows);

			for (int64_t r = 0; r < got_rows; r++)
				u8codes[r] = (uint8_t) codes[r];
			Table::Col c;

			c.name = ac.name;
			c.type = GG_COL_CHAR1;
			c.bytes = (size_t) got_rows;
			{
				hipError_t he = hipMalloc(&c.dev,
							  c.bytes ? c.bytes
							  : 1);

				if (he == hipSuccess && c.bytes)
					he = hipMemcpy(c.dev,
						       u8codes.data(),
						       c.bytes,
						       hipMemcpyHostToDevice);
				if (he != hipSuccess)
				{
					st = fail(GG_EGPU, "dict col: %s",
						  hipGetErrorString(he));
					break;
				}
			}
			c.dict_bytes.assign(dbytes.begin(),
					    dbytes.begin() + doffs[nd]);
			c.dict_offs.assign(doffs.begin(),
					   doffs.begin() + nd + 1);
			t->cols.push_back(c);
			continue;
		}

		int datumlen = (ac.type == GG_COL_INT64 ||
				ac.type == GG_COL_DEC64_S2) ? 8 : 4;
		std::vector<AoDesc> descs;
		std::vector<uint8_t> spill;

		st = ao_parse_blocks(ac.stream, ac.stream_len, ac.checksums,
				     ac.ao_version, ac.comptype, descs,
				     spill);
		if (st != GG_OK)
			break;

		void *d_vals = nullptr;
		uint8_t *d_nulls = nullptr;
		int64_t total_rows = 0;

		st = ao_decode_to_device(e, ac.stream, ac.stream_len,
					 descs, spill, ac.dsb_version,
					 datumlen, datumlen, &d_vals,
					 ac.nullable ? &d_nulls
					 : nullptr /* else NOT NULL required */,
					 &total_rows);
		if (st != GG_OK)
			break;
		if (t->nrows < 0)
			t->nrows = total_rows;
		else if (t->nrows != total_rows)
		{
			(void) hipFree(d_vals);
			(void) hipFree(d_nulls);
			st = fail(GG_EINVAL, "column %s has %lld rows, "
				  "table has %lld", ac.name,
				  (long long) total_rows,
				  (long long) t->nrows);
			break;
		}

		Table::Col c;

		c.name = ac.name;
		c.type = ac.type;
		c.nulls = d_nulls;
		if (ac.type == GG_COL_CHAR1 && total_rows)
		{
			uint8_t *d_u8 = nullptr;
			hipError_t he = hipMalloc((void **) &d_u8,
						  (size_t) total_rows);

			if (he == hipSuccess)
				he = launch_narrow_i32_u8(
					e.stream, (const int32_t *) d_vals,
					total_rows, d_u8);
			if (he == hipSuccess)
				he = hipStreamSynchronize(e.stream);
			(void) hipFree(d_vals);
			if (he != hipSuccess)
			{
				(void) hipFree(d_u8);
				(void) hipFree(d_nulls);
				st = fail(GG_EGPU, "narrow: %s",
					  hipGetErrorString(he));
				break;
			}
			c.dev = d_u8;
			c.bytes = (size_t) total_rows;
		}
		else
		{
			c.dev = d_vals;
			c.bytes = (size_t) total_rows * datumlen;
		}
		t->cols.push_back(c);
	}
	if (st != GG_OK)
	{
		for (auto &c : t->cols)
		{
			if (c.bytes)
				(void) hipFree(c.dev);
			if (c.nulls)
				(void) hipFree(c.nulls);
		}
		delete t;
		return st;
	}
	if (t->nrows < 0)
		t->nrows = 0;
	e.tables.push_back(t);
	*out = (gg_table) (e.tables.size() - 1);
	return GG_OK;
}

static Table *get_table(gg_table h)
{
	Engine &e = engine();

	if (h < 0 || (size_t) h >= e.tables.size())
		return nullptr;
	return e.tables[h];
}

/* shared with plan.cpp */
Table *engine_table(gg_table h)
{
	return get_table(h);
}

extern "C" gg_status
gg_engine_table_text_dict(gg_table h, const char *col, uint8_t *out_bytes,
			  int64_t cap, int64_t *out_offs,
			  int32_t max_entries, int32_t *out_n)
{
	Table *t = get_table(h);

	if (!t || !col || !out_bytes || !out_offs || !out_n)
		return fail(GG_EINVAL, "bad table_text_dict args");
	for (auto &c : t->cols)
		if (c.name == col)
		{
			if (c.dict_offs.empty())
				return fail(GG_EINVAL,
					    "column %s has no dictionary",
					    col);
			int32_t n = (int32_t) c.dict_offs.size() - 1;

			if (n > max_entries)
				return fail(GG_EINVAL,
					    "dict has %d entries > %d",
					    n, max_entries);
			if ((int64_t) c.dict_bytes.size() > cap)
				return fail(GG_EINVAL, "dict cap");
			std::memcpy(out_bytes, c.dict_bytes.data(),
				    c.dict_bytes.size());
			for (int32_t i = 0; i <= n; i++)
				out_offs[i] = c.dict_offs[i];
			*out_n = n;
			return GG_OK;
		}
	return fail(GG_EINVAL, "no column %s", col);
}

extern "C" gg_status gg_engine_table_nrows(gg_table h, int64_t *out)
{
	Table *t = get_table(h);

	if (!t)
		return fail(GG_EINVAL, "bad table handle %d", h);
	*out = t->nrows;
	return GG_OK;
}

extern "C" gg_status
gg_engine_fetch_column(gg_table h, const char *col_name, void *host_buf,
		       size_t buf_bytes)
{
	Table *t = get_table(h);

	if (!t)
		return fail(GG_EINVAL, "bad table handle %d", h);
	for (auto &c : t->cols)
		if (c.name == col_name)
		{
			size_t want = (size_t) t->nrows * coltype_size(c.type);

			if (buf_bytes != want)
				return fail(GG_EINVAL,
					    "fetch_column size %zu != %zu",
					    buf_bytes, want);
			if (want)
				GG_HIP(hipMemcpy(host_buf, c.dev, want,
						 hipMemcpyDeviceToHost));
			return GG_OK;
		}
	return fail(GG_EINVAL, "no column '%s' in table '%s'", col_name,
		    t->name.c_str());
}

extern "C" gg_status gg_engine_drop_table(gg_table h)
{
	Engine &e = engine();
	Table *t = get_table(h);

	if (!t)
		return fail(GG_EINVAL, "bad table handle %d", h);
	for (auto &c : t->cols)
	{
		/* the min/max cache is keyed by device pointer: a freed
		 * buffer's address can be handed out again by a later
		 * hipMalloc, so stale entries must not survive the free */
		e.mm_cache.erase(c.dev);
		if (c.bytes)
			(void) hipFree(c.dev);
		if (c.nulls)
			(void) hipFree(c.nulls);
	}
	delete t;
	e.tables[h] = nullptr;
	return GG_OK;
}

/*
 * Synthetic shard registration: generates this segment's shard of the
 * named TPC-H table directly in HBM (include/gg_gen.h; sharding by the
 * table's DISTRIBUTED BY key via bit-exact cdbhash —
 * input/bb_mpph.source:11–92 DDL, cdbhash.c:190/:549).
 */
extern "C" gg_status
gg_engine_register_synth(const char *table_name, uint64_t seed, int64_t sf,
			 gg_table *out)
{
	Engine &e = engine();

	if (!e.inited)
		return fail(GG_ESTATE, "engine not initialized");
	int nseg = e.cfg.n_segments;
	int seg = e.cfg.segment_id;
	std::string nm = table_name ? table_name : "";
	int which;
	int64_t total;

	if (nm == "lineitem")
	{
		which = 0;
		total = gg_n_lineitem(sf);
	}
	else if (nm == "orders")
	{
		which = 1;
		total = gg_n_orders(sf);
	}
	else if (nm == "customer")
	{
		which = 2;
		total = gg_n_customers(sf);
	}
	else if (nm == "supplier")
	{
		which = 3;	/* keys dense: shard by suppkey like cust */
		total = gg_n_suppliers(sf);
	}
	else if (nm == "nation")
	{
		/* tiny dim (25 rows), replicated on every segment — the
		 * reference broadcasts/replicates small dims */
		Table *t = new Table();
		int32_t natk[GG_NNATIONS], regk[GG_NNATIONS];

		for (int i = 0; i < GG_NNATIONS; i++)
		{
			natk[i] = i;
			regk[i] = gg_nation_region(i);
		}
		t->name = nm;
		t->nrows = GG_NNATIONS;
		{
			gg_status s2;

			if ((s2 = add_col(t, "nationkey", GG_COL_INT32,
					  GG_NNATIONS, natk, nullptr)) != GG_OK ||
			    (s2 = add_col(t, "regionkey", GG_COL_INT32,
					  GG_NNATIONS, regk, nullptr)) != GG_OK)
			{
				delete t;
				return s2;
			}
		}
		e.tables.push_back(t);
		*out = (gg_table) (e.tables.size() - 1);
		return GG_OK;
	}
	else
		return fail(GG_EINVAL, "unknown synth table '%s'",
			    table_name);

	int64_t shard_rows = total;
	unsigned long long *ctr = nullptr;

	GG_TRY(dev_counter(&ctr));
	if (nseg > 1)
	{
		unsigned long long c = 0;

		GG_HIP(launch_count_shard(e.stream, 0, total, nseg, seg,
					  which, seed, sf, ctr));
		GG_HIP(hipStreamSynchronize(e.stream));
		GG_TRY(read_counter(ctr, &c));
		shard_rows = (int64_t) c;
		GG_HIP(hipMemset(ctr, 0, 8));
	}

	Table *t = new Table();

	t->name = nm;
	t->nrows = shard_rows;

	gg_status s = GG_OK;

	if (nm == "lineitem")
	{
		void *ok, *q, *p, *d, *tx, *sd, *rf, *ls, *sk;

		if ((s = add_col(t, "orderkey", GG_COL_INT64, shard_rows, nullptr, &ok)) == GG_OK &&
		    (s = add_col(t, "qty", GG_COL_DEC64_S2, shard_rows, nullptr, &q)) == GG_OK &&
		    (s = add_col(t, "price", GG_COL_DEC64_S2, shard_rows, nullptr, &p)) == GG_OK &&
		    (s = add_col(t, "disc", GG_COL_DEC64_S2, shard_rows, nullptr, &d)) == GG_OK &&
		    (s = add_col(t, "tax", GG_COL_DEC64_S2, shard_rows, nullptr, &tx)) == GG_OK &&
		    (s = add_col(t, "shipdate", GG_COL_INT32, shard_rows, nullptr, &sd)) == GG_OK &&
		    (s = add_col(t, "rflag", GG_COL_CHAR1, shard_rows, nullptr, &rf)) == GG_OK &&
		    (s = add_col(t, "lstatus", GG_COL_CHAR1, shard_rows, nullptr, &ls)) == GG_OK &&
		    (s = add_col(t, "suppkey", GG_COL_INT64, shard_rows, nullptr, &sk)) == GG_OK)
		{
			hipError_t he = launch_gen_lineitem(
				e.stream, seed, sf, 0, total, nseg, seg,
				(int64_t *) ok, (int64_t *) q, (int64_t *) p,
				(int64_t *) d, (int64_t *) tx, (int32_t *) sd,
				(uint8_t *) rf, (uint8_t *) ls,
				(int64_t *) sk, ctr);
			if (he != hipSuccess)
				s = fail(GG_EGPU, "gen_lineitem: %s",
					 hipGetErrorString(he));
		}
	}
	else if (nm == "orders")
	{
		void *ok, *ck, *od, *pr;

		if ((s = add_col(t, "orderkey", GG_COL_INT64, shard_rows, nullptr, &ok)) == GG_OK &&
		    (s = add_col(t, "custkey", GG_COL_INT64, shard_rows, nullptr, &ck)) == GG_OK &&
		    (s = add_col(t, "orderdate", GG_COL_INT32, shard_rows, nullptr, &od)) == GG_OK &&
		    (s = add_col(t, "shippriority", GG_COL_INT32, shard_rows, nullptr, &pr)) == GG_OK)
		{
			hipError_t he = launch_gen_orders(
				e.stream, seed, sf, 0, total, nseg, seg,
				(int64_t *) ok, (int64_t *) ck,
				(int32_t *) od, (int32_t *) pr, ctr);
			if (he != hipSuccess)
				s = fail(GG_EGPU, "gen_orders: %s",
					 hipGetErrorString(he));
		}
	}
	else if (nm == "customer")
	{
		void *ck, *ms, *nk;

		if ((s = add_col(t, "custkey", GG_COL_INT64, shard_rows, nullptr, &ck)) == GG_OK &&
		    (s = add_col(t, "mktseg", GG_COL_CHAR1, shard_rows, nullptr, &ms)) == GG_OK &&
		    (s = add_col(t, "nationkey", GG_COL_CHAR1, shard_rows, nullptr, &nk)) == GG_OK)
		{
			hipError_t he = launch_gen_customer(
				e.stream, seed, 0, total, nseg, seg,
				(int64_t *) ck, (uint8_t *) ms,
				(uint8_t *) nk, ctr);
			if (he != hipSuccess)
				s = fail(GG_EGPU, "gen_customer: %s",
					 hipGetErrorString(he));
		}
	}
	else	/* supplier */
	{
		void *sk, *nk;

		if ((s = add_col(t, "suppkey", GG_COL_INT64, shard_rows, nullptr, &sk)) == GG_OK &&
		    (s = add_col(t, "nationkey", GG_COL_CHAR1, shard_rows, nullptr, &nk)) == GG_OK)
		{
			hipError_t he = launch_gen_supplier(
				e.stream, seed, 0, total, nseg, seg,
				(int64_t *) sk, (uint8_t *) nk, ctr);
			if (he != hipSuccess)
				s = fail(GG_EGPU, "gen_supplier: %s",
					 hipGetErrorString(he));
		}
	}
	GG_HIP(hipStreamSynchronize(e.stream));
	(void) hipFree(ctr);
	if (s != GG_OK)
	{
		for (auto &c : t->cols)
			(void) hipFree(c.dev);
		delete t;
		return s;
	}
	e.tables.push_back(t);
	*out = (gg_table) (e.tables.size() - 1);
	return GG_OK;
}

/* ---------------- pipelines ---------------- */

extern "C" gg_status
gg_engine_compile_pipeline(const gg_pipeline_desc *desc, gg_pipeline *out)
{
	Engine &e = engine();

	if (!e.inited)
		return fail(GG_ESTATE, "engine not initialized");
	if (!desc || !out)
		return fail(GG_EINVAL, "null pipeline args");
	switch (desc->kind)
	{
		case GG_PIPE_Q1:
		case GG_PIPE_SUMPRICE:
			if (!get_table(desc->lineitem))
				return fail(GG_EINVAL, "Q1: bad lineitem table");
			break;
		case GG_PIPE_Q3:
			if (!get_table(desc->lineitem) ||
			    !get_table(desc->orders) ||
			    !get_table(desc->customer))
				return fail(GG_EINVAL, "Q3: bad table handle");
			break;
		case GG_PIPE_Q5:
			if (!get_table(desc->lineitem) ||
			    !get_table(desc->orders) ||
			    !get_table(desc->customer) ||
			    !get_table(desc->supplier) ||
			    !get_table(desc->nation))
				return fail(GG_EINVAL, "Q5: bad table handle");
			break;
		default:
			return fail(GG_ENOTSUP,
				    "pipeline kind %d not supported (fall "
				    "back to standard_ExecutorRun)",
				    (int) desc->kind);
	}
	Pipeline *p = new Pipeline();

	p->desc = *desc;
	e.pipelines.push_back(p);
	*out = (gg_pipeline) (e.pipelines.size() - 1);
	return GG_OK;
}

extern "C" gg_status gg_engine_drop_pipeline(gg_pipeline h)
{
	Engine &e = engine();

	if (h < 0 || (size_t) h >= e.pipelines.size() || !e.pipelines[h])
		return fail(GG_EINVAL, "bad pipeline handle");
	delete e.pipelines[h];
	e.pipelines[h] = nullptr;
	return GG_OK;
}

/* ---------------- execution: SUMPRICE ---------------- */

static gg_status exec_sumprice(Pipeline *p, void *arena, size_t bytes,
			       size_t *written)
{
	Engine &e = engine();
	Table *li = get_table(p->desc.lineitem);

	if (bytes < sizeof(gg_sumprice_result))
		return fail(GG_EINVAL, "arena too small");
	const int32_t *sd = (const int32_t *) li->col("shipdate");
	const int64_t *pr = (const int64_t *) li->col("price");

	if (!sd || !pr)
		return fail(GG_EINVAL, "lineitem lacks shipdate/price");
	SumPriceAcc *acc = (SumPriceAcc *) p->sget("sp.acc", sizeof(*acc));

	if (!acc)
		return fail(GG_ENOMEM, "sumprice acc");
	GG_HIP(hipMemsetAsync(acc, 0, sizeof(*acc), e.stream));
	{
		Timed tm(e.stream);

		GG_HIP(launch_sumprice(e.stream, sd, pr, li->nrows,
				       p->desc.cutoff_date, acc));
		double ms = tm.stop();
		KernelStatAcc &st = p->stat("sumprice");

		st.launches++;
		st.total_ms += ms;
		st.rows_in += li->nrows;
		st.hbm_bytes += li->nrows * 12;
	}
	SumPriceAcc h;

	GG_HIP(hipMemcpy(&h, acc, sizeof(h), hipMemcpyDeviceToHost));

	u128 sum = h.sum_c;
	u128 cnt = h.count;

	if (e.cfg.n_segments > 1)
	{
		if (!comm_ready())
			return fail(GG_ESTATE, "multi-segment without comm");
		unsigned long long *dbuf;
		std::vector<unsigned long long> all(2 * e.cfg.n_segments);

		GG_HIP(hipMalloc((void **) &dbuf,
				 (2 + 2 * e.cfg.n_segments) * 8));
		GG_HIP(hipMemcpy(dbuf, &h, 16, hipMemcpyHostToDevice));
		GG_TRY(comm_allgather_u64(dbuf, dbuf + 2, 2));
		GG_HIP(hipMemcpy(all.data(), dbuf + 2, all.size() * 8,
				 hipMemcpyDeviceToHost));
		(void) hipFree(dbuf);
		sum = 0;
		cnt = 0;
		for (int r = 0; r < e.cfg.n_segments; r++)
		{
			sum += all[2 * r];
			cnt += all[2 * r + 1];
		}
	}
	gg_sumprice_result *res = (gg_sumprice_result *) arena;

	res->sum_c = (int64_t) sum;
	res->count = (int64_t) cnt;
	*written = sizeof(*res);
	return GG_OK;
}

/* ---------------- execution: Q1 ---------------- */

static gg_status exec_q1(Pipeline *p, void *arena, size_t bytes,
			 size_t *written)
{
	Engine &e = engine();
	Table *li = get_table(p->desc.lineitem);

	if (bytes < sizeof(gg_q1_result))
		return fail(GG_EINVAL, "arena too small");
	const int32_t *sd = (const int32_t *) li->col("shipdate");
	const uint8_t *rf = (const uint8_t *) li->col("rflag");
	const uint8_t *ls = (const uint8_t *) li->col("lstatus");
	const int64_t *q = (const int64_t *) li->col("qty");
	const int64_t *pr = (const int64_t *) li->col("price");
	const int64_t *d = (const int64_t *) li->col("disc");
	const int64_t *tx = (const int64_t *) li->col("tax");

	if (!sd || !rf || !ls || !q || !pr || !d || !tx)
		return fail(GG_EINVAL, "lineitem lacks a Q1 column");

	Q1DeviceAcc *acc = (Q1DeviceAcc *) p->sget("q1.acc", sizeof(*acc));

	if (!acc)
		return fail(GG_ENOMEM, "q1 acc");
	GG_HIP(hipMemsetAsync(acc, 0, sizeof(*acc), e.stream));
	{
		Timed tm(e.stream);

		GG_HIP(launch_q1(e.stream, sd, rf, ls, q, pr, d, tx,
				 li->nrows, p->desc.cutoff_date, acc));
		double ms = tm.stop();
		KernelStatAcc &st = p->stat("q1_agg");

		st.launches++;
		st.total_ms += ms;
		st.rows_in += li->nrows;
		st.hbm_bytes += li->nrows * 38;	/* SURVEY §8(d) */
	}

	/* combine partial states (2-stage agg: cdbgroup.c:1245 /
	 * nodeAgg.c:2130–2142 — exact partials + one combine) */
	static_assert(sizeof(Q1DeviceAcc) == 49 * 8, "acc layout");
	std::vector<unsigned long long> parts;
	int nseg = e.cfg.n_segments;
	bool exch = nseg > 1 || (getenv("GG_FORCE_EXCHANGE") != nullptr &&
				 comm_ready());

	if (exch)
	{
		if (!comm_ready())
			return fail(GG_ESTATE, "multi-segment without comm");
		unsigned long long *gbuf;

		GG_HIP(hipMalloc((void **) &gbuf, 49 * 8 * (size_t) (nseg + 1)));
		GG_HIP(hipMemcpy(gbuf, acc, 49 * 8, hipMemcpyDeviceToDevice));
		GG_TRY(comm_allgather_u64(gbuf, gbuf + 49, 49));
		parts.resize(49 * (size_t) nseg);
		GG_HIP(hipMemcpy(parts.data(), gbuf + 49, parts.size() * 8,
				 hipMemcpyDeviceToHost));
		(void) hipFree(gbuf);
	}
	else
	{
		parts.resize(49);
		GG_HIP(hipMemcpy(parts.data(), acc, 49 * 8,
				 hipMemcpyDeviceToHost));
	}

	gg_q1_result *res = (gg_q1_result *) arena;

	std::memset(res, 0, sizeof(*res));
	{
		static const char RF[6] = { 'A', 'A', 'N', 'N', 'R', 'R' };
		static const char LS[6] = { 'F', 'O', 'F', 'O', 'F', 'O' };
		int occupied = 0;

		for (int g = 0; g < 6; g++)
		{
			u128 cnt = 0, qty = 0, base = 0, dcol = 0;
			i128 disc = 0, charge = 0;

			for (int r = 0; r < nseg; r++)
			{
				const unsigned long long *v =
					&parts[49 * (size_t) r + 8 * (size_t) g];

				cnt += v[0];
				qty += v[1];
				base += v[2];
				dcol += v[3];
				disc += mk128(v[4], (int64_t) v[5]);
				charge += mk128(v[6], (int64_t) v[7]);
			}
			for (int r = 0; r < nseg; r++)
				if (parts[49 * (size_t) r + 48])
					return fail(GG_EINVAL,
						    "Q1: unexpected returnflag/"
						    "linestatus byte in input");
			gg_q1_result_group *o = &res->groups[g];

			o->count = (int64_t) cnt;
			o->sum_qty_c = (int64_t) qty;
			o->sum_base_c = (int64_t) base;
			o->sum_dcol_c = (int64_t) dcol;
			split128(disc, &o->disc_lo, &o->disc_hi);
			split128(charge, &o->charge_lo, &o->charge_hi);
			o->returnflag = RF[g];
			o->linestatus = LS[g];
			if (o->count)
				occupied++;
		}
		res->n_groups = occupied;
	}
	*written = sizeof(*res);
	return GG_OK;
}

/* ---------------- execution: Q3 ---------------- */

struct Q3TopkCmp
{
	bool operator()(const gg_q3_result_row &a,
			const gg_q3_result_row &b) const
	{
		u128 ra = ((u128) (uint64_t) a.rev_hi << 64) | a.rev_lo;
		u128 rb = ((u128) (uint64_t) b.rev_hi << 64) | b.rev_lo;

		if (ra != rb)
			return ra > rb;
		if (a.orderdate != b.orderdate)
			return a.orderdate < b.orderdate;
		return a.orderkey < b.orderkey;
	}
};

static gg_status exec_q3(Pipeline *p, void *arena, size_t bytes,
			 size_t *written)
{
	Engine &e = engine();
	Table *li = get_table(p->desc.lineitem);
	Table *od = get_table(p->desc.orders);
	Table *cu = get_table(p->desc.customer);
	int nseg = e.cfg.n_segments;
	int me = e.cfg.segment_id;
	int64_t k = p->desc.limit_k > 0 ? p->desc.limit_k : 10;
	/* the Motion-exchange path normally runs for nseg > 1; the env
	 * knob forces it at nseg == 1 (self-loopback over RCCL) so the
	 * whole redistribute plumbing is testable on one GPU */
	bool exch = nseg > 1 || getenv("GG_FORCE_EXCHANGE") != nullptr;

	if (bytes < sizeof(gg_q3_result_hdr) +
	    (size_t) k * sizeof(gg_q3_result_row))
		return fail(GG_EINVAL, "arena too small");
	if (k > 1000)
		return fail(GG_ENOTSUP, "LIMIT > 1000 not supported in v1");

	const int64_t *c_ck = (const int64_t *) cu->col("custkey");
	const uint8_t *c_ms = (const uint8_t *) cu->col("mktseg");
	const int64_t *o_ok = (const int64_t *) od->col("orderkey");
	const int64_t *o_ck = (const int64_t *) od->col("custkey");
	const int32_t *o_dt = (const int32_t *) od->col("orderdate");
	const int32_t *o_pr = (const int32_t *) od->col("shippriority");
	const int64_t *l_ok = (const int64_t *) li->col("orderkey");
	const int32_t *l_sd = (const int32_t *) li->col("shipdate");
	const int64_t *l_pc = (const int64_t *) li->col("price");
	const int64_t *l_dc = (const int64_t *) li->col("disc");

	if (!c_ck || !c_ms || !o_ok || !o_ck || !o_dt || !o_pr || !l_ok ||
	    !l_sd || !l_pc || !l_dc)
		return fail(GG_EINVAL, "Q3: missing column");

	int32_t cutoff = p->desc.cutoff_date;
	uint8_t segcode = p->desc.mktsegment;
	unsigned long long *ctr =
		(unsigned long long *) p->sget("ctr", 8);

	if (!ctr)
		return fail(GG_ENOMEM, "scratch");

	/* 1. customer build side (nodeHash.c:450 sizing / :905 insert).
	 * Dense custkeys (max <= 8x rows) collapse to a 1-bit-per-key
	 * membership bitmap (SF100: 15M keys = 1.9 MB — resident in
	 * every XCD's L2, unlike a 15 MB byte array); hash set fallback
	 * for sparse keys. */
	DeviceHashTable cust{};
	unsigned long long *cust_bits = nullptr;
	int64_t cust_dlen = 0;
	{
		unsigned long long maxk = 0;

		GG_TRY(cached_max_i64(e, p, c_ck, cu->nrows, ctr, &maxk));
		if (cu->nrows > 0 && maxk > 0 &&
		    maxk <= (unsigned long long) (8 * cu->nrows + 16))
			cust_dlen = (int64_t) maxk + 1;
	}
	/* record which build-side layout ran (nodeHash.c:450 sizing
	 * decision analog) so EXPLAIN-style stats show the taken path */
	p->stat(cust_dlen ? "path_cust_bitmap" : "path_cust_hash").launches++;
	{
		Timed tm(e.stream);

		if (cust_dlen)
		{
			size_t words = (size_t) (cust_dlen / 64 + 2);

			cust_bits = (unsigned long long *)
				p->sget("cust.bits", words * 8);
			if (!cust_bits)
				return fail(GG_ENOMEM, "cust bits");
			GG_HIP(hipMemsetAsync(cust_bits, 0, words * 8,
					      e.stream));
			GG_HIP(launch_cust_dense_fill_seg(e.stream, c_ck,
							  c_ms, cu->nrows,
							  segcode, cust_bits,
							  cust_dlen));
		}
		else
		{
			if (!p->cust_slots)
			{
				unsigned long long nfil = 0;

				GG_HIP(hipMemsetAsync(ctr, 0, 8, e.stream));
				GG_HIP(launch_count_filter_u8(e.stream, c_ms,
							      segcode,
							      cu->nrows, ctr));
				GG_HIP(hipStreamSynchronize(e.stream));
				GG_TRY(read_counter(ctr, &nfil));
				p->cust_slots = next_pow2(2 * (nfil + 1));
			}
			cust.nslots = p->cust_slots;
			cust.bloom_words = cust.nslots / 8 < 1024
				? 1024 : cust.nslots / 8;
			cust.keys = (unsigned long long *)
				p->sget("cust.keys", cust.nslots * 8);
			cust.bloom = (unsigned long long *)
				p->sget("cust.bloom", cust.bloom_words * 8);
			if (!cust.keys || !cust.bloom)
				return fail(GG_ENOMEM, "cust table");
			GG_HIP(hipMemsetAsync(cust.keys, 0, cust.nslots * 8,
					      e.stream));
			GG_HIP(hipMemsetAsync(cust.bloom, 0,
					      cust.bloom_words * 8, e.stream));
			GG_HIP(launch_build_set(e.stream, c_ck, c_ms, segcode,
						cu->nrows, cust));
		}
		double ms = tm.stop();
		KernelStatAcc &st = p->stat("build_customer");

		st.launches++;
		st.total_ms += ms;
		st.rows_in += cu->nrows;
		st.hbm_bytes += cu->nrows * 9;
	}

	/* 2. orders side.  Dense o_orderkey (the common case: TPC-H PK)
	 * collapses the orders hash table to direct-map pay/rev arrays
	 * (q3_dense.hip); hash table fallback otherwise. */
	DeviceHashTable ord{};
	unsigned long long *ordd_pr = nullptr;	/* interleaved (rev, pay) */
	unsigned long long *ordd_bloom = nullptr;
	uint64_t ordd_bwords = 0;
	int64_t ord_dlen = 0;
	unsigned long long nmatch = 0;

	{
		unsigned long long maxk = 0;

		GG_TRY(cached_max_i64(e, p, o_ok, od->nrows, ctr, &maxk));
		if (od->nrows > 0 && maxk > 0 &&
		    maxk <= (unsigned long long) (8 * od->nrows + 16))
			ord_dlen = (int64_t) maxk + 1;
	}
	p->stat(ord_dlen ? "path_orders_dense" : "path_orders_hash").launches++;
	if (ord_dlen)
	{
		/* exact membership bitmap over dense keys */
		ordd_bwords = (uint64_t) (ord_dlen / 64 + 2);
		/* (rev, pay) interleave as 16-B pairs: the top-k's cold
		 * random gathers then touch ONE cache line per survivor */
		ordd_pr = (unsigned long long *)
			p->sget("ordd.pr", (size_t) ord_dlen * 16);
		ordd_bloom = (unsigned long long *)
			p->sget("ordd.bloom", ordd_bwords * 8);
		if (!ordd_pr || !ordd_bloom)
			return fail(GG_ENOMEM, "ord dense");
		/* pay slots need NO init: they are read only behind the
		 * exact membership bitmap, which is rebuilt every pass,
		 * so any read slot was stored this pass.  rev slots are
		 * fully zeroed only on first use / after a fallback pass;
		 * the fused top-k otherwise zeroes exactly the entries
		 * the probe touched (clear over the survivor list). */
		if (!p->q3_rev_inited || p->q3_rev_dirty)
		{
			GG_HIP(hipMemsetAsync(ordd_pr, 0,
					      (size_t) ord_dlen * 16,
					      e.stream));
			p->q3_rev_inited = true;
		}
		p->q3_rev_dirty = true;	/* until the sparse clear runs */
		GG_HIP(hipMemsetAsync(ordd_bloom, 0, ordd_bwords * 8,
				      e.stream));
	}

	if (!exch)
	{
		Timed tm(e.stream);

		GG_HIP(hipMemsetAsync(ctr, 0, 8, e.stream));
		if (ord_dlen)
		{
			GG_HIP(launch_dn_build_orders(
				e.stream, o_ok, o_ck, o_dt, o_pr, od->nrows,
				cutoff, cust, cust_bits, cust_dlen, ordd_pr,
				ord_dlen, ordd_bloom, ordd_bwords, ctr));
			GG_HIP(hipStreamSynchronize(e.stream));
			GG_TRY(read_counter(ctr, &nmatch));
		}
		else
		{
			if (!p->ord_slots)
			{
				GG_HIP(launch_count_orders_match(
					e.stream, o_ck, o_dt, od->nrows,
					cutoff, cust, cust_bits, cust_dlen,
					ctr));
				GG_HIP(hipStreamSynchronize(e.stream));
				GG_TRY(read_counter(ctr, &nmatch));
				p->ord_slots = next_pow2(2 * (nmatch + 1));
				GG_HIP(hipMemsetAsync(ctr, 0, 8, e.stream));
			}
			ord.nslots = p->ord_slots;
			ord.bloom_words = ord.nslots / 8 < 1024
				? 1024 : ord.nslots / 8;
			ord.keys = (unsigned long long *)
				p->sget("ord.keys", ord.nslots * 8);
			ord.payload = (unsigned long long *)
				p->sget("ord.payload", ord.nslots * 8);
			ord.rev = (unsigned long long *)
				p->sget("ord.rev", ord.nslots * 8);
			ord.bloom = (unsigned long long *)
				p->sget("ord.bloom", ord.bloom_words * 8);
			if (!ord.keys || !ord.payload || !ord.rev || !ord.bloom)
				return fail(GG_ENOMEM, "ord table");
			GG_HIP(hipMemsetAsync(ord.keys, 0, ord.nslots * 8,
					      e.stream));
			GG_HIP(hipMemsetAsync(ord.rev, 0, ord.nslots * 8,
					      e.stream));
			GG_HIP(hipMemsetAsync(ord.bloom, 0,
					      ord.bloom_words * 8, e.stream));
			GG_HIP(launch_build_orders(e.stream, o_ok, o_ck, o_dt,
						   o_pr, od->nrows, cutoff,
						   cust, cust_bits, cust_dlen,
						   ord, ctr));
			GG_HIP(hipStreamSynchronize(e.stream));
			GG_TRY(read_counter(ctr, &nmatch));
		}
		double ms = tm.stop();
		KernelStatAcc &st = p->stat("build_orders");

		st.launches++;
		st.total_ms += ms;
		st.rows_in += od->nrows;
		st.rows_out += (int64_t) nmatch;
		st.hbm_bytes += od->nrows * 24;
	}
	else
	{
		/* Motion redistribute legs:
		 * leg 1: filtered orders → owner of o_custkey;
		 * leg 2: customer-joined orders → owner of o_orderkey. */
		if (!comm_ready())
			return fail(GG_ESTATE,
				    "exchange path requires gg_engine_comm_init");
		Timed tm(e.stream);

		int64_t *f_ck = (int64_t *) p->sget("f_ck", (od->nrows + 1) * 8);
		int64_t *f_ok = (int64_t *) p->sget("f_ok", (od->nrows + 1) * 8);
		int64_t *f_pay = (int64_t *) p->sget("f_pay", (od->nrows + 1) * 8);
		unsigned long long *dcnt =
			(unsigned long long *) p->sget("dcnt", (size_t) nseg * 8);
		unsigned long long *doffs =
			(unsigned long long *) p->sget("doffs", (size_t) nseg * 8);

		if (!f_ck || !f_ok || !f_pay || !dcnt || !doffs)
			return fail(GG_ENOMEM, "exchange scratch");

		unsigned long long nfil = 0;

		GG_HIP(hipMemsetAsync(ctr, 0, 8, e.stream));
		GG_HIP(launch_orders_filter_compact(e.stream, o_ok, o_ck, o_dt,
						    o_pr, od->nrows,
						    INT32_MIN, cutoff,
						    f_ck, f_ok, f_pay, ctr));
		GG_HIP(hipStreamSynchronize(e.stream));
		GG_TRY(read_counter(ctr, &nfil));

		/* partition by custkey (doSendTuple routing,
		 * nodeMotion.c:1600–1636, bit-exact cdbhash) */
		std::vector<unsigned long long> cnts(nseg), offs(nseg + 1, 0);

		GG_HIP(hipMemsetAsync(dcnt, 0, (size_t) nseg * 8, e.stream));
		GG_HIP(launch_part_count(e.stream, f_ck, (int64_t) nfil, nseg,
					 dcnt));
		GG_HIP(hipStreamSynchronize(e.stream));
		GG_HIP(hipMemcpy(cnts.data(), dcnt, (size_t) nseg * 8,
				 hipMemcpyDeviceToHost));
		for (int i = 0; i < nseg; i++)
			offs[i + 1] = offs[i] + cnts[i];

		int64_t *s_ck = (int64_t *) p->sget("s_ck", (nfil + 1) * 8);
		int64_t *s_ok = (int64_t *) p->sget("s_ok", (nfil + 1) * 8);
		int64_t *s_pay = (int64_t *) p->sget("s_pay", (nfil + 1) * 8);

		if (!s_ck || !s_ok || !s_pay)
			return fail(GG_ENOMEM, "send scratch");
		GG_HIP(hipMemcpy(doffs, offs.data(), (size_t) nseg * 8,
				 hipMemcpyHostToDevice));
		GG_HIP(launch_part_scatter3(e.stream, f_ck, (int64_t) nfil,
					    nseg, f_ck, f_ok, f_pay, doffs,
					    s_ck, s_ok, s_pay));
		GG_HIP(hipStreamSynchronize(e.stream));

		/* exchange the count vector, derive receive layout */
		std::vector<unsigned long long> allcnt((size_t) nseg * nseg);
		{
			unsigned long long *g = (unsigned long long *)
				p->sget("cntg", (size_t) (nseg + nseg * nseg) * 8);

			if (!g)
				return fail(GG_ENOMEM, "cnt allgather");
			GG_HIP(hipMemcpy(g, cnts.data(), (size_t) nseg * 8,
					 hipMemcpyHostToDevice));
			GG_TRY(comm_allgather_u64(g, g + nseg, nseg));
			GG_HIP(hipMemcpy(allcnt.data(), g + nseg,
					 allcnt.size() * 8,
					 hipMemcpyDeviceToHost));
		}
		std::vector<unsigned long long> rcnts(nseg), roffs(nseg + 1, 0);

		for (int s2 = 0; s2 < nseg; s2++)
			rcnts[s2] = allcnt[(size_t) s2 * nseg + me];
		for (int i = 0; i < nseg; i++)
			roffs[i + 1] = roffs[i] + rcnts[i];
		uint64_t rtotal = roffs[nseg];

		int64_t *r_ck = (int64_t *) p->sget("r_ck", (rtotal + 1) * 8);
		int64_t *r_ok = (int64_t *) p->sget("r_ok", (rtotal + 1) * 8);
		int64_t *r_pay = (int64_t *) p->sget("r_pay", (rtotal + 1) * 8);

		if (!r_ck || !r_ok || !r_pay)
			return fail(GG_ENOMEM, "recv scratch");
		GG_TRY(comm_alltoallv_i64(s_ck, offs.data(), cnts.data(),
					  r_ck, roffs.data(), rcnts.data()));
		GG_TRY(comm_alltoallv_i64(s_ok, offs.data(), cnts.data(),
					  r_ok, roffs.data(), rcnts.data()));
		GG_TRY(comm_alltoallv_i64(s_pay, offs.data(), cnts.data(),
					  r_pay, roffs.data(), rcnts.data()));

		/* probe local customer set → matched (okey, pay) */
		unsigned long long nm = 0;
		int64_t *m_ok = (int64_t *) p->sget("m_ok", (rtotal + 1) * 8);
		int64_t *m_pay = (int64_t *) p->sget("m_pay", (rtotal + 1) * 8);

		if (!m_ok || !m_pay)
			return fail(GG_ENOMEM, "match scratch");
		GG_HIP(hipMemsetAsync(ctr, 0, 8, e.stream));
		GG_HIP(launch_probe_cust_compact(e.stream, r_ck, r_ok, r_pay,
						 (int64_t) rtotal, cust,
						 cust_bits, cust_dlen, m_ok,
						 m_pay, ctr));
		GG_HIP(hipStreamSynchronize(e.stream));
		GG_TRY(read_counter(ctr, &nm));

		/* leg 2: matched orders → owner of o_orderkey */
		std::vector<unsigned long long> cnts2(nseg), offs2(nseg + 1, 0);

		GG_HIP(hipMemsetAsync(dcnt, 0, (size_t) nseg * 8, e.stream));
		GG_HIP(launch_part_count(e.stream, m_ok, (int64_t) nm, nseg,
					 dcnt));
		GG_HIP(hipStreamSynchronize(e.stream));
		GG_HIP(hipMemcpy(cnts2.data(), dcnt, (size_t) nseg * 8,
				 hipMemcpyDeviceToHost));
		for (int i = 0; i < nseg; i++)
			offs2[i + 1] = offs2[i] + cnts2[i];

		int64_t *s2_ok = (int64_t *) p->sget("s2_ok", (nm + 1) * 8);
		int64_t *s2_pay = (int64_t *) p->sget("s2_pay", (nm + 1) * 8);

		if (!s2_ok || !s2_pay)
			return fail(GG_ENOMEM, "send2 scratch");
		GG_HIP(hipMemcpy(doffs, offs2.data(), (size_t) nseg * 8,
				 hipMemcpyHostToDevice));
		GG_HIP(launch_part_scatter3(e.stream, m_ok, (int64_t) nm, nseg,
					    m_ok, m_pay, nullptr, doffs,
					    s2_ok, s2_pay, nullptr));
		GG_HIP(hipStreamSynchronize(e.stream));

		std::vector<unsigned long long> allcnt2((size_t) nseg * nseg);
		{
			unsigned long long *g = (unsigned long long *)
				p->sget("cntg", (size_t) (nseg + nseg * nseg) * 8);

			GG_HIP(hipMemcpy(g, cnts2.data(), (size_t) nseg * 8,
					 hipMemcpyHostToDevice));
			GG_TRY(comm_allgather_u64(g, g + nseg, nseg));
			GG_HIP(hipMemcpy(allcnt2.data(), g + nseg,
					 allcnt2.size() * 8,
					 hipMemcpyDeviceToHost));
		}
		std::vector<unsigned long long> rcnts2(nseg), roffs2(nseg + 1, 0);

		for (int s2 = 0; s2 < nseg; s2++)
			rcnts2[s2] = allcnt2[(size_t) s2 * nseg + me];
		for (int i = 0; i < nseg; i++)
			roffs2[i + 1] = roffs2[i] + rcnts2[i];
		uint64_t rtotal2 = roffs2[nseg];

		int64_t *r2_ok = (int64_t *) p->sget("r2_ok", (rtotal2 + 1) * 8);
		int64_t *r2_pay = (int64_t *) p->sget("r2_pay", (rtotal2 + 1) * 8);

		if (!r2_ok || !r2_pay)
			return fail(GG_ENOMEM, "recv2 scratch");
		GG_TRY(comm_alltoallv_i64(s2_ok, offs2.data(), cnts2.data(),
					  r2_ok, roffs2.data(), rcnts2.data()));
		GG_TRY(comm_alltoallv_i64(s2_pay, offs2.data(), cnts2.data(),
					  r2_pay, roffs2.data(),
					  rcnts2.data()));

		nmatch = rtotal2;
		if (ord_dlen)
		{
			GG_HIP(launch_dn_insert_orders(e.stream, r2_ok, r2_pay,
						       (int64_t) rtotal2,
						       ordd_pr, ord_dlen,
						       ordd_bloom,
						       ordd_bwords));
			GG_HIP(hipStreamSynchronize(e.stream));
		}
		else
		{
			if (!p->ord_slots)
				p->ord_slots = next_pow2(2 * (rtotal2 + 1));
			ord.nslots = p->ord_slots;
			ord.bloom_words = ord.nslots / 8 < 1024
				? 1024 : ord.nslots / 8;
			ord.keys = (unsigned long long *)
				p->sget("ord.keys", ord.nslots * 8);
			ord.payload = (unsigned long long *)
				p->sget("ord.payload", ord.nslots * 8);
			ord.rev = (unsigned long long *)
				p->sget("ord.rev", ord.nslots * 8);
			ord.bloom = (unsigned long long *)
				p->sget("ord.bloom", ord.bloom_words * 8);
			if (!ord.keys || !ord.payload || !ord.rev || !ord.bloom)
				return fail(GG_ENOMEM, "ord table");
			GG_HIP(hipMemsetAsync(ord.keys, 0, ord.nslots * 8,
					      e.stream));
			GG_HIP(hipMemsetAsync(ord.rev, 0, ord.nslots * 8,
					      e.stream));
			GG_HIP(hipMemsetAsync(ord.bloom, 0,
					      ord.bloom_words * 8, e.stream));
			GG_HIP(launch_insert_orders(e.stream, r2_ok, r2_pay,
						    (int64_t) rtotal2, ord));
			GG_HIP(hipStreamSynchronize(e.stream));
		}

		double ms = tm.stop();
		KernelStatAcc &st = p->stat("orders_exchange");

		st.launches++;
		st.total_ms += ms;
		st.rows_in += od->nrows;
		st.rows_out += (int64_t) nmatch;
	}

	/* 3. lineitem probe + group aggregation (probe fused with the
	 * group-by transition; group slot ≡ matched-order slot).  The
	 * dense probe also appends each group's index ONCE (0→nonzero
	 * rev transition) into a per-block survivor region — LDS
	 * counter, no contended global atomic — so the top-k never
	 * sweeps the dense array. */
	unsigned long long njoin = 0, sovf = 0;
	unsigned long long *dsurv = nullptr, *dscnt = nullptr,
		*dsovf = nullptr;
	uint64_t region = 0;
	int pgrid = 0;

	if (ord_dlen)
	{
		pgrid = dn_probe_grid(li->nrows);
		region = 2 * ((uint64_t) nmatch / (uint64_t) pgrid) + 256;
		dsurv = (unsigned long long *)
			p->sget("surv", (size_t) pgrid * region * 8);
		dscnt = (unsigned long long *)
			p->sget("surv.cnts", (size_t) pgrid * 8);
		dsovf = (unsigned long long *) p->sget("surv.ovf", 8);
		if (!dsurv || !dscnt || !dsovf)
			return fail(GG_ENOMEM, "survivor scratch");
		GG_HIP(hipMemsetAsync(dsovf, 0, 8, e.stream));
	}
	GG_HIP(hipMemsetAsync(ctr, 0, 8, e.stream));
	{
		Timed tm(e.stream);

		if (ord_dlen)
			GG_HIP(launch_dn_probe_lineitem(
				e.stream, l_ok, l_sd, l_pc, l_dc, li->nrows,
				cutoff, ordd_pr, ord_dlen,
				ordd_bloom, ordd_bwords, ctr, dsurv, region,
				dscnt, dsovf, pgrid));
		else
			GG_HIP(launch_probe_lineitem(e.stream, l_ok, l_sd,
						     l_pc, l_dc, li->nrows,
						     cutoff, ord, ctr));
		double ms = tm.stop();
		GG_TRY(read_counter(ctr, &njoin));
		if (ord_dlen)
			GG_TRY(read_counter(dsovf, &sovf));
		KernelStatAcc &st = p->stat("probe_lineitem");

		st.launches++;
		st.total_ms += ms;
		st.rows_in += li->nrows;
		st.rows_out += (int64_t) njoin;
		st.hbm_bytes += li->nrows * 28;	/* SURVEY §8(d) */
	}

	/* 4. top-k select.  Dense path: stats+hist, threshold, collect
	 * and sparse clear all walk the ~ngroups survivor list the probe
	 * recorded — no sweep of the GB-sized dense array, no host round
	 * trip between kernels.  Hash-table path: the original
	 * stats→hist→collect sweeps. */
	unsigned long long hstats[5] = {0, 0, 0, 0, 0};
	unsigned long long *stats5 =
		(unsigned long long *) p->sget("stats5", 5 * 8);

	if (!stats5)
		return fail(GG_ENOMEM, "stats scratch");
	{
		Timed tm(e.stream);
		std::vector<gg_q3_result_row> cand;
		bool need_old = !ord_dlen || sovf != 0;

		if (!need_old)
		{
			uint64_t cap = 4 * (uint64_t) k + 65536;
			unsigned int *dhist =
				(unsigned int *) p->sget("hist", 65536 * 4);
			unsigned long long *dthr =
				(unsigned long long *) p->sget("thr", 8);
			gg_q3_result_row *dout = (gg_q3_result_row *)
				p->sget("cand", cap * sizeof(gg_q3_result_row));
			unsigned long long ncand = 0;

			if (!dhist || !dthr || !dout)
				return fail(GG_ENOMEM, "topk scratch");
			GG_HIP(hipMemsetAsync(stats5, 0, 5 * 8, e.stream));
			GG_HIP(hipMemsetAsync(dhist, 0, 65536 * 4, e.stream));
			GG_HIP(hipMemsetAsync(ctr, 0, 8, e.stream));
			GG_HIP(launch_dn_q3_stats_surv(e.stream, dsurv,
						       dscnt, region, pgrid,
						       ordd_pr,
						       stats5, dhist));
			GG_HIP(launch_dn_q3_threshold2(e.stream, dhist, k,
						       dthr));
			GG_HIP(launch_dn_q3_collect_surv(
				e.stream, dsurv, dscnt, region, pgrid,
				ordd_pr, dthr, dout, ctr, cap));
			GG_HIP(hipStreamSynchronize(e.stream));
			GG_HIP(hipMemcpy(hstats, stats5, 40,
					 hipMemcpyDeviceToHost));
			GG_TRY(read_counter(ctr, &ncand));
			if (ncand > cap)
			{
				/* tie-heavy threshold bin: retry collect
				 * alone with an exact-size buffer */
				cap = hstats[0] + 1;
				dout = (gg_q3_result_row *) p->sget(
					"cand",
					cap * sizeof(gg_q3_result_row));
				if (!dout)
					return fail(GG_ENOMEM, "topk retry");
				GG_HIP(hipMemsetAsync(ctr, 0, 8, e.stream));
				GG_HIP(launch_dn_q3_collect_surv(
					e.stream, dsurv, dscnt, region,
					pgrid, ordd_pr, dthr,
					dout, ctr, cap));
				GG_HIP(hipStreamSynchronize(e.stream));
				GG_TRY(read_counter(ctr, &ncand));
				if (ncand > cap)
					return fail(GG_EINVAL,
						    "top-k overflow (%llu)",
						    ncand);
			}
			if (ncand)
			{
				cand.resize(ncand);
				GG_HIP(hipMemcpy(cand.data(), dout,
						 ncand * sizeof(gg_q3_result_row),
						 hipMemcpyDeviceToHost));
			}
			/* zero only the touched rev[] entries so the next
			 * execute skips the dense memset (runs async;
			 * later work queues behind it on the stream) */
			GG_HIP(launch_dn_q3_clear_surv(e.stream, dsurv,
						       dscnt, region, pgrid,
						       ordd_pr));
			p->q3_rev_dirty = false;
		}

		if (need_old)
		{
			GG_HIP(hipMemsetAsync(stats5, 0, 5 * 8, e.stream));
			if (ord_dlen)
				GG_HIP(launch_dn_q3_stats(e.stream, ordd_pr,
							  ord_dlen,
							  stats5));
			else
				GG_HIP(launch_q3_stats(e.stream, ord, stats5));
			GG_HIP(hipStreamSynchronize(e.stream));
			GG_HIP(hipMemcpy(hstats, stats5, 40,
					 hipMemcpyDeviceToHost));

			unsigned long long maxrev = hstats[4];

			cand.clear();
			if (maxrev > 0)
			{
				unsigned int *dhist = (unsigned int *)
					p->sget("hist", 65536 * 4);
				unsigned long long *dthr =
					(unsigned long long *) p->sget("thr", 8);
				uint64_t cap = hstats[0] + 1;	/* <= n_groups */
				gg_q3_result_row *dout = (gg_q3_result_row *)
					p->sget("cand",
						cap * sizeof(gg_q3_result_row));
				unsigned long long ncand = 0;

				if (!dhist || !dthr || !dout)
					return fail(GG_ENOMEM, "topk scratch");
				GG_HIP(hipMemsetAsync(dhist, 0, 65536 * 4,
						      e.stream));
				GG_HIP(hipMemsetAsync(ctr, 0, 8, e.stream));
				if (ord_dlen)
				{
					GG_HIP(launch_dn_q3_hist(
						e.stream, ordd_pr, ord_dlen,
						stats5, dhist));
					GG_HIP(launch_q3_threshold(
						e.stream, dhist, stats5, k,
						dthr));
					GG_HIP(launch_dn_q3_collect(
						e.stream, ordd_pr,
						ord_dlen, dthr, dout, ctr,
						cap));
				}
				else
				{
					GG_HIP(launch_q3_hist(e.stream, ord,
							      stats5, dhist));
					GG_HIP(launch_q3_threshold(
						e.stream, dhist, stats5, k,
						dthr));
					GG_HIP(launch_q3_collect(
						e.stream, ord, dthr, dout,
						ctr, cap));
				}
				GG_HIP(hipStreamSynchronize(e.stream));
				GG_TRY(read_counter(ctr, &ncand));
				if (ncand > cap)
					return fail(GG_EINVAL,
						    "top-k candidate overflow (%llu)",
						    ncand);
				cand.resize(ncand);
				if (ncand)
					GG_HIP(hipMemcpy(
						cand.data(), dout,
						ncand * sizeof(gg_q3_result_row),
						hipMemcpyDeviceToHost));
			}
		}
		std::sort(cand.begin(), cand.end(), Q3TopkCmp());
		if ((int64_t) cand.size() > k)
			cand.resize(k);

		double ms = tm.stop();
		KernelStatAcc &st = p->stat("q3_topk");

		st.launches++;
		st.total_ms += ms;
		st.rows_in += (int64_t) hstats[0];
		st.rows_out += (int64_t) cand.size();
		/* dominant traffic: one 8 B/slot sweep of the dense rev
		 * array (survivor finish pass is ~ngroups gathers) */
		st.hbm_bytes += ord_dlen ? ord_dlen * 8
			: (int64_t) ord.nslots * 16;

		/* 5. global combine (the Gather Motion to the QD:
		 * nodeMotion gather + final-stage combine) */
		u128 revsum = ((u128) hstats[1]) | ((u128) hstats[2] << 64);
		uint64_t ngroups = hstats[0];
		uint64_t checksum = hstats[3];
		uint64_t njoin_g = njoin;

		if (nseg > 1)
		{
			size_t per = 5 + (size_t) k * 4;
			std::vector<unsigned long long> lbuf(per, 0);

			lbuf[0] = ngroups;
			lbuf[1] = (uint64_t) revsum;
			lbuf[2] = (uint64_t) (revsum >> 64);
			lbuf[3] = njoin_g;
			lbuf[4] = checksum;
			for (size_t i = 0; i < cand.size(); i++)
				std::memcpy(&lbuf[5 + i * 4], &cand[i], 32);

			unsigned long long *g = (unsigned long long *)
				p->sget("gatherg", (per + per * (size_t) nseg) * 8);

			if (!g)
				return fail(GG_ENOMEM, "gather scratch");
			GG_HIP(hipMemcpy(g, lbuf.data(), per * 8,
					 hipMemcpyHostToDevice));
			GG_TRY(comm_allgather_u64(g, g + per, per));
			std::vector<unsigned long long> all(per * (size_t) nseg);

			GG_HIP(hipMemcpy(all.data(), g + per, all.size() * 8,
					 hipMemcpyDeviceToHost));

			ngroups = 0;
			revsum = 0;
			checksum = 0;
			njoin_g = 0;
			cand.clear();
			for (int r = 0; r < nseg; r++)
			{
				const unsigned long long *v =
					&all[per * (size_t) r];

				ngroups += v[0];
				revsum += ((u128) v[1]) | ((u128) v[2] << 64);
				njoin_g += v[3];
				checksum += v[4];
				for (int64_t i = 0; i < k; i++)
				{
					gg_q3_result_row row;

					std::memcpy(&row, &v[5 + i * 4], 32);
					if (row.rev_lo || row.rev_hi)
						cand.push_back(row);
				}
			}
			std::sort(cand.begin(), cand.end(), Q3TopkCmp());
			if ((int64_t) cand.size() > k)
				cand.resize(k);
		}

		gg_q3_result_hdr *hdr = (gg_q3_result_hdr *) arena;

		hdr->n_out = (int64_t) cand.size();
		hdr->n_groups = (int64_t) ngroups;
		hdr->rev_sum_lo = (uint64_t) revsum;
		hdr->rev_sum_hi = (int64_t) (revsum >> 64);
		hdr->group_checksum = checksum;
		hdr->n_join_rows = (int64_t) njoin_g;
		std::memcpy(hdr + 1, cand.data(),
			    cand.size() * sizeof(gg_q3_result_row));
		*written = sizeof(*hdr) +
			cand.size() * sizeof(gg_q3_result_row);
	}
	return GG_OK;
}

/* ---------------- execution: Q5 (mpph5) ---------------- */

/* n_name strings in nationkey order (reference fixture nation.csv;
 * needed only for the ORDER BY tiebreak and display) */
static const char *const GG_NATION_NAMES[GG_NNATIONS] = {
	"ALGERIA", "ARGENTINA", "BRAZIL", "CANADA", "EGYPT", "ETHIOPIA",
	"FRANCE", "GERMANY", "INDIA", "INDONESIA", "IRAN", "IRAQ", "JAPAN",
	"JORDAN", "KENYA", "MOROCCO", "MOZAMBIQUE", "PERU", "CHINA",
	"ROMANIA", "SAUDI ARABIA", "VIETNAM", "RUSSIA", "UNITED KINGDOM",
	"UNITED STATES"
};

struct Q5RowCmp
{
	bool operator()(const gg_q5_result_row &a,
			const gg_q5_result_row &b) const
	{
		u128 ra = ((u128) (uint64_t) a.rev_hi << 64) | a.rev_lo;
		u128 rb = ((u128) (uint64_t) b.rev_hi << 64) | b.rev_lo;

		if (ra != rb)
			return ra > rb;	/* ORDER BY revenue DESC */
		return std::strcmp(GG_NATION_NAMES[a.nationkey],
				   GG_NATION_NAMES[b.nationkey]) < 0;
	}
};

static gg_status exec_q5(Pipeline *p, void *arena, size_t bytes,
			 size_t *written)
{
	Engine &e = engine();
	Table *li = get_table(p->desc.lineitem);
	Table *od = get_table(p->desc.orders);
	Table *cu = get_table(p->desc.customer);
	Table *su = get_table(p->desc.supplier);
	Table *na = get_table(p->desc.nation);
	int nseg = e.cfg.n_segments;
	int me = e.cfg.segment_id;
	bool exch = nseg > 1 || getenv("GG_FORCE_EXCHANGE") != nullptr;

	if (bytes < sizeof(gg_q5_result_hdr) +
	    GG_NNATIONS * sizeof(gg_q5_result_row))
		return fail(GG_EINVAL, "arena too small");

	const int64_t *l_ok = (const int64_t *) li->col("orderkey");
	const int64_t *l_sk = (const int64_t *) li->col("suppkey");
	const int64_t *l_pc = (const int64_t *) li->col("price");
	const int64_t *l_dc = (const int64_t *) li->col("disc");
	const int64_t *o_ok = (const int64_t *) od->col("orderkey");
	const int64_t *o_ck = (const int64_t *) od->col("custkey");
	const int32_t *o_dt = (const int32_t *) od->col("orderdate");
	const int64_t *c_ck = (const int64_t *) cu->col("custkey");
	const uint8_t *c_nk = (const uint8_t *) cu->col("nationkey");
	const int64_t *s_sk = (const int64_t *) su->col("suppkey");
	const uint8_t *s_nk = (const uint8_t *) su->col("nationkey");
	const int32_t *n_nk = (const int32_t *) na->col("nationkey");
	const int32_t *n_rk = (const int32_t *) na->col("regionkey");

	if (!l_ok || !l_sk || !l_pc || !l_dc || !o_ok || !o_ck || !o_dt ||
	    !c_ck || !c_nk || !s_sk || !s_nk || !n_nk || !n_rk)
		return fail(GG_EINVAL, "Q5: missing column");

	int32_t date_lo = p->desc.cutoff_date;
	int32_t date_hi = p->desc.cutoff_hi;
	uint8_t regionkey = p->desc.regionkey;
	unsigned long long *ctr =
		(unsigned long long *) p->sget("ctr", 8);

	if (!ctr)
		return fail(GG_ENOMEM, "scratch");

	/* nation dim (25 rows) → host → device u8 region_of[25]; the
	 * region⋈nation semi-join resolved here, the broadcast-Motion
	 * analog for the tiny dims */
	uint8_t region_of_h[GG_NNATIONS];
	{
		std::vector<int32_t> nk(na->nrows), rk(na->nrows);

		GG_HIP(hipMemcpy(nk.data(), n_nk, na->nrows * 4,
				 hipMemcpyDeviceToHost));
		GG_HIP(hipMemcpy(rk.data(), n_rk, na->nrows * 4,
				 hipMemcpyDeviceToHost));
		std::memset(region_of_h, 0xff, sizeof(region_of_h));
		for (int64_t i = 0; i < na->nrows; i++)
			if (nk[i] >= 0 && nk[i] < GG_NNATIONS)
				region_of_h[nk[i]] = (uint8_t) rk[i];
	}
	uint8_t *region_of = (uint8_t *) p->sget("region_of", 32);

	if (!region_of)
		return fail(GG_ENOMEM, "region_of");
	GG_HIP(hipMemcpy(region_of, region_of_h, GG_NNATIONS,
			 hipMemcpyHostToDevice));

	/* 1. customer map: c_custkey → c_nationkey.  Dense custkeys use a
	 * u8 nation array (255 = absent) in L2/L3; hash map fallback. */
	DeviceHashTable cust{};
	uint8_t *cust_dense = nullptr;
	int64_t cust_dlen = 0;
	{
		unsigned long long maxk = 0;

		GG_TRY(cached_max_i64(e, p, c_ck, cu->nrows, ctr, &maxk));
		if (cu->nrows > 0 && maxk > 0 &&
		    maxk <= (unsigned long long) (8 * cu->nrows + 16))
			cust_dlen = (int64_t) maxk + 1;
	}
	p->stat(cust_dlen ? "path_cust_dense" : "path_cust_hash").launches++;
	{
		Timed tm(e.stream);

		if (cust_dlen)
		{
			cust_dense = (uint8_t *)
				p->sget("cust.dense", (size_t) cust_dlen);
			if (!cust_dense)
				return fail(GG_ENOMEM, "cust dense");
			GG_HIP(hipMemsetAsync(cust_dense, 0xff,
					      (size_t) cust_dlen, e.stream));
			GG_HIP(launch_cust_dense_fill_nat(e.stream, c_ck,
							  c_nk, cu->nrows,
							  cust_dense,
							  cust_dlen));
		}
		else
		{
			if (!p->cust_slots)
				p->cust_slots =
					next_pow2(2 * (uint64_t) (cu->nrows + 1));
			cust.nslots = p->cust_slots;
			cust.keys = (unsigned long long *)
				p->sget("cust.keys", cust.nslots * 8);
			cust.payload = (unsigned long long *)
				p->sget("cust.pay", cust.nslots * 8);
			if (!cust.keys || !cust.payload)
				return fail(GG_ENOMEM, "cust map");
			GG_HIP(hipMemsetAsync(cust.keys, 0, cust.nslots * 8,
					      e.stream));
			GG_HIP(launch_build_kv(e.stream, c_ck, c_nk,
					       cu->nrows, cust));
		}
		double ms = tm.stop();
		KernelStatAcc &st = p->stat("build_customer_map");

		st.launches++;
		st.total_ms += ms;
		st.rows_in += cu->nrows;
		st.hbm_bytes += cu->nrows * 9;
	}

	/* 2. supplier table: s_suppkey → s_nationkey, in-region only.
	 * Dense keys (max <= 8x rows) use a u8 direct-map array in L2 —
	 * one byte load replaces bloom+table probes; hash fallback kept
	 * for sparse keys. */
	DeviceHashTable supp{};
	uint8_t *supp_dense = nullptr;
	int64_t supp_dense_len = 0;
	{
		unsigned long long maxk = 0;

		GG_TRY(cached_max_i64(e, p, s_sk, su->nrows, ctr, &maxk));
		if (su->nrows > 0 && maxk > 0 &&
		    maxk <= (unsigned long long) (8 * su->nrows + 16))
			supp_dense_len = (int64_t) maxk + 1;
	}
	{
		Timed tm(e.stream);

		if (supp_dense_len && !exch)
		{
			supp_dense = (uint8_t *)
				p->sget("supp.dense", (size_t) supp_dense_len);
			if (!supp_dense)
				return fail(GG_ENOMEM, "supp dense");
			GG_HIP(hipMemsetAsync(supp_dense, 0xff,
					      (size_t) supp_dense_len,
					      e.stream));
			GG_HIP(launch_supp_dense_fill(e.stream, s_sk, s_nk,
						      su->nrows, region_of,
						      regionkey, supp_dense,
						      supp_dense_len));
			GG_HIP(hipStreamSynchronize(e.stream));
		}
		else if (!exch)
		{
			if (!p->supp_slots)
				p->supp_slots =
					next_pow2(2 * (uint64_t) (su->nrows + 1));
			supp.nslots = p->supp_slots;
			supp.bloom_words = supp.nslots / 8 < 1024
				? 1024 : supp.nslots / 8;
			supp.keys = (unsigned long long *)
				p->sget("supp.keys", supp.nslots * 8);
			supp.payload = (unsigned long long *)
				p->sget("supp.pay", supp.nslots * 8);
			supp.bloom = (unsigned long long *)
				p->sget("supp.bloom", supp.bloom_words * 8);
			if (!supp.keys || !supp.payload || !supp.bloom)
				return fail(GG_ENOMEM, "supp table");
			GG_HIP(hipMemsetAsync(supp.keys, 0, supp.nslots * 8,
					      e.stream));
			GG_HIP(hipMemsetAsync(supp.bloom, 0,
					      supp.bloom_words * 8, e.stream));
			GG_HIP(hipMemsetAsync(ctr, 0, 8, e.stream));
			GG_HIP(launch_build_supp(e.stream, s_sk, s_nk,
						 su->nrows, region_of,
						 regionkey, supp, ctr));
			GG_HIP(hipStreamSynchronize(e.stream));
		}
		else
		{
			/* broadcast Motion of the small dim: compact the
			 * local in-region shard, allgather (padded), insert
			 * every rank's rows locally */
			if (!comm_ready())
				return fail(GG_ESTATE,
					    "exchange path requires comm");
			unsigned long long nkept = 0;
			int64_t *csk = (int64_t *) p->sget("supp.csk",
							   (su->nrows + 1) * 8);
			int64_t *csn = (int64_t *) p->sget("supp.csn",
							   (su->nrows + 1) * 8);

			if (!csk || !csn)
				return fail(GG_ENOMEM, "supp compact");
			GG_HIP(hipMemsetAsync(ctr, 0, 8, e.stream));
			GG_HIP(launch_supp_filter_compact(e.stream, s_sk, s_nk,
							  su->nrows, region_of,
							  regionkey, csk, csn,
							  ctr));
			GG_HIP(hipStreamSynchronize(e.stream));
			GG_TRY(read_counter(ctr, &nkept));

			/* exchange counts */
			std::vector<unsigned long long> counts(nseg);
			{
				unsigned long long *g = (unsigned long long *)
					p->sget("supp.cnt", (1 + (size_t) nseg) * 8);

				if (!g)
					return fail(GG_ENOMEM, "supp cnt");
				GG_HIP(hipMemcpy(g, &nkept, 8,
						 hipMemcpyHostToDevice));
				GG_TRY(comm_allgather_u64(g, g + 1, 1));
				GG_HIP(hipMemcpy(counts.data(), g + 1,
						 (size_t) nseg * 8,
						 hipMemcpyDeviceToHost));
			}
			unsigned long long maxc = 0, total = 0;

			for (int r = 0; r < nseg; r++)
			{
				total += counts[r];
				if (counts[r] > maxc)
					maxc = counts[r];
			}
			/* padded allgather of (sk, sn) pairs */
			uint64_t per = 2 * (maxc + 1);
			int64_t *gs = (int64_t *)
				p->sget("supp.gather",
					(per + per * (size_t) nseg) * 8);

			if (!gs)
				return fail(GG_ENOMEM, "supp gather");
			GG_HIP(hipMemcpyAsync(gs, csk, (nkept + 1) * 8,
					      hipMemcpyDeviceToDevice, e.stream));
			GG_HIP(hipMemcpyAsync(gs + (maxc + 1), csn,
					      (nkept + 1) * 8,
					      hipMemcpyDeviceToDevice, e.stream));
			GG_HIP(hipStreamSynchronize(e.stream));
			GG_TRY(comm_allgather_u64(gs, gs + per, per));

			if (!p->supp_slots)
				p->supp_slots = next_pow2(2 * (total + 1));
			supp.nslots = p->supp_slots;
			supp.bloom_words = supp.nslots / 8 < 1024
				? 1024 : supp.nslots / 8;
			supp.keys = (unsigned long long *)
				p->sget("supp.keys", supp.nslots * 8);
			supp.payload = (unsigned long long *)
				p->sget("supp.pay", supp.nslots * 8);
			supp.bloom = (unsigned long long *)
				p->sget("supp.bloom", supp.bloom_words * 8);
			if (!supp.keys || !supp.payload || !supp.bloom)
				return fail(GG_ENOMEM, "supp table");
			GG_HIP(hipMemsetAsync(supp.keys, 0, supp.nslots * 8,
					      e.stream));
			GG_HIP(hipMemsetAsync(supp.bloom, 0,
					      supp.bloom_words * 8, e.stream));
			for (int r = 0; r < nseg; r++)
			{
				int64_t *base = gs + per + (size_t) r * per;

				if (counts[r])
					GG_HIP(launch_insert_supp(
						e.stream, base,
						base + (maxc + 1),
						(int64_t) counts[r], supp));
			}
			/* dense direct-map over the gathered pairs too,
			 * but only after checking the GLOBAL max key stays
			 * within the density bound (the pre-exchange check
			 * saw only the local shard) */
			supp_dense_len = 0;
			{
				unsigned long long gmax = 0;

				GG_HIP(hipMemsetAsync(ctr, 0, 8, e.stream));
				for (int r = 0; r < nseg; r++)
				{
					int64_t *base =
						gs + per + (size_t) r * per;

					if (counts[r])
						GG_HIP(launch_max_i64(
							e.stream, base,
							(int64_t) counts[r],
							ctr));
				}
				GG_HIP(hipStreamSynchronize(e.stream));
				GG_TRY(read_counter(ctr, &gmax));
				if (gmax > 0 && (int64_t) gmax <=
				    8 * su->nrows * (int64_t) nseg + 16)
					supp_dense_len = (int64_t) gmax + 1;
			}
			if (supp_dense_len)
			{
				supp_dense = (uint8_t *)
					p->sget("supp.dense",
						(size_t) supp_dense_len);
				if (!supp_dense)
					return fail(GG_ENOMEM, "supp dense");
				GG_HIP(hipMemsetAsync(supp_dense, 0xff,
						      (size_t) supp_dense_len,
						      e.stream));
				for (int r = 0; r < nseg; r++)
				{
					int64_t *base =
						gs + per + (size_t) r * per;

					if (counts[r])
						GG_HIP(launch_supp_dense_fill_pairs(
							e.stream, base,
							base + (maxc + 1),
							(int64_t) counts[r],
							supp_dense,
							supp_dense_len));
				}
			}
			GG_HIP(hipStreamSynchronize(e.stream));
		}
		double ms = tm.stop();
		KernelStatAcc &st = p->stat("build_supplier");

		st.launches++;
		st.total_ms += ms;
		st.rows_in += su->nrows;
		st.hbm_bytes += su->nrows * 9;	/* suppkey 8 + nation 1 */
	}

	/* 3. orders → okey → c_nationkey map.  Dense o_orderkey (with a
	 * dense supplier side) uses the direct-map arrays (q3_dense.hip);
	 * hash table fallback otherwise. */
	DeviceHashTable ord{};
	uint8_t *ordd_pay8 = nullptr;
	int64_t ord_dlen = 0;
	unsigned long long nmatch = 0;

	if (supp_dense)
	{
		unsigned long long maxk = 0;

		GG_TRY(cached_max_i64(e, p, o_ok, od->nrows, ctr, &maxk));
		if (od->nrows > 0 && maxk > 0 &&
		    maxk <= (unsigned long long) (8 * od->nrows + 16))
			ord_dlen = (int64_t) maxk + 1;
	}
	p->stat(ord_dlen ? "path_orders_dense" : "path_orders_hash").launches++;
	if (ord_dlen)
	{
		/* nation fits a byte; 255 = no matching order.  The u8 map
		 * is its own membership filter — no Bloom reads on probe. */
		ordd_pay8 = (uint8_t *)
			p->sget("ordd.pay8", (size_t) ord_dlen);
		if (!ordd_pay8)
			return fail(GG_ENOMEM, "ord dense");
		GG_HIP(hipMemsetAsync(ordd_pay8, 0xff, (size_t) ord_dlen,
				      e.stream));
	}

	if (!exch)
	{
		Timed tm(e.stream);

		if (ord_dlen)
		{
			GG_HIP(hipMemsetAsync(ctr, 0, 8, e.stream));
			GG_HIP(launch_dn_build_orders_q5_u8(
				e.stream, o_ok, o_ck, o_dt, od->nrows,
				date_lo, date_hi, cust, cust_dense, cust_dlen,
				ordd_pay8, ord_dlen, ctr));
			GG_HIP(hipStreamSynchronize(e.stream));
			GG_TRY(read_counter(ctr, &nmatch));
			double ms0 = tm.stop();
			KernelStatAcc &st0 = p->stat("build_orders_q5");

			st0.launches++;
			st0.total_ms += ms0;
			st0.rows_in += od->nrows;
			st0.rows_out += (int64_t) nmatch;
			st0.hbm_bytes += od->nrows * 20;  /* okey,ckey,odate */
			goto orders_done;
		}
		if (!p->ord_slots)
		{
			unsigned long long nfil = 0;

			GG_HIP(hipMemsetAsync(ctr, 0, 8, e.stream));
			GG_HIP(launch_count_date_range(e.stream, o_dt,
						       od->nrows, date_lo,
						       date_hi, ctr));
			GG_HIP(hipStreamSynchronize(e.stream));
			GG_TRY(read_counter(ctr, &nfil));
			p->ord_slots = next_pow2(2 * (nfil + 1));
		}
		ord.nslots = p->ord_slots;
		ord.bloom_words = ord.nslots / 8 < 1024 ? 1024 : ord.nslots / 8;
		ord.keys = (unsigned long long *)
			p->sget("ord.keys", ord.nslots * 8);
		ord.payload = (unsigned long long *)
			p->sget("ord.pay", ord.nslots * 8);
		ord.bloom = (unsigned long long *)
			p->sget("ord.bloom", ord.bloom_words * 8);
		if (!ord.keys || !ord.payload || !ord.bloom)
			return fail(GG_ENOMEM, "ord table");
		GG_HIP(hipMemsetAsync(ord.keys, 0, ord.nslots * 8, e.stream));
		GG_HIP(hipMemsetAsync(ord.bloom, 0, ord.bloom_words * 8,
				      e.stream));
		GG_HIP(hipMemsetAsync(ctr, 0, 8, e.stream));
		GG_HIP(launch_build_orders_q5(e.stream, o_ok, o_ck, o_dt,
					      od->nrows, date_lo, date_hi,
					      cust, cust_dense, cust_dlen,
					      ord, ctr));
		GG_HIP(hipStreamSynchronize(e.stream));
		GG_TRY(read_counter(ctr, &nmatch));
		{
			double ms = tm.stop();
			KernelStatAcc &st = p->stat("build_orders_q5");

			st.launches++;
			st.total_ms += ms;
			st.rows_in += od->nrows;
			st.rows_out += (int64_t) nmatch;
			st.hbm_bytes += od->nrows * 20;  /* okey,ckey,odate */
		}
orders_done:;
	}
	else
	{
		/* two Motion redistribute legs, as in Q3 but with the
		 * customer MAP payload (c_nationkey) */
		if (!comm_ready())
			return fail(GG_ESTATE, "exchange path requires comm");
		Timed tm(e.stream);
		int64_t *f_ck = (int64_t *) p->sget("f_ck", (od->nrows + 1) * 8);
		int64_t *f_ok = (int64_t *) p->sget("f_ok", (od->nrows + 1) * 8);
		int64_t *f_pay = (int64_t *) p->sget("f_pay", (od->nrows + 1) * 8);
		unsigned long long *dcnt =
			(unsigned long long *) p->sget("dcnt", (size_t) nseg * 8);
		unsigned long long *doffs =
			(unsigned long long *) p->sget("doffs", (size_t) nseg * 8);
		unsigned long long nfil = 0;

		if (!f_ck || !f_ok || !f_pay || !dcnt || !doffs)
			return fail(GG_ENOMEM, "exchange scratch");
		GG_HIP(hipMemsetAsync(ctr, 0, 8, e.stream));
		GG_HIP(launch_orders_filter_compact(e.stream, o_ok, o_ck, o_dt,
						    o_dt /* prio unused */,
						    od->nrows, date_lo,
						    date_hi, f_ck, f_ok,
						    f_pay, ctr));
		GG_HIP(hipStreamSynchronize(e.stream));
		GG_TRY(read_counter(ctr, &nfil));

		std::vector<unsigned long long> cnts(nseg), offs(nseg + 1, 0);

		GG_HIP(hipMemsetAsync(dcnt, 0, (size_t) nseg * 8, e.stream));
		GG_HIP(launch_part_count(e.stream, f_ck, (int64_t) nfil, nseg,
					 dcnt));
		GG_HIP(hipStreamSynchronize(e.stream));
		GG_HIP(hipMemcpy(cnts.data(), dcnt, (size_t) nseg * 8,
				 hipMemcpyDeviceToHost));
		for (int i = 0; i < nseg; i++)
			offs[i + 1] = offs[i] + cnts[i];

		int64_t *s_ckb = (int64_t *) p->sget("s_ck", (nfil + 1) * 8);
		int64_t *s_okb = (int64_t *) p->sget("s_ok", (nfil + 1) * 8);

		if (!s_ckb || !s_okb)
			return fail(GG_ENOMEM, "send scratch");
		GG_HIP(hipMemcpy(doffs, offs.data(), (size_t) nseg * 8,
				 hipMemcpyHostToDevice));
		GG_HIP(launch_part_scatter3(e.stream, f_ck, (int64_t) nfil,
					    nseg, f_ck, f_ok, nullptr, doffs,
					    s_ckb, s_okb, nullptr));
		GG_HIP(hipStreamSynchronize(e.stream));

		std::vector<unsigned long long> allcnt((size_t) nseg * nseg);
		{
			unsigned long long *g = (unsigned long long *)
				p->sget("cntg", (size_t) (nseg + nseg * nseg) * 8);

			if (!g)
				return fail(GG_ENOMEM, "cnt allgather");
			GG_HIP(hipMemcpy(g, cnts.data(), (size_t) nseg * 8,
					 hipMemcpyHostToDevice));
			GG_TRY(comm_allgather_u64(g, g + nseg, nseg));
			GG_HIP(hipMemcpy(allcnt.data(), g + nseg,
					 allcnt.size() * 8,
					 hipMemcpyDeviceToHost));
		}
		std::vector<unsigned long long> rcnts(nseg), roffs(nseg + 1, 0);

		for (int s2 = 0; s2 < nseg; s2++)
			rcnts[s2] = allcnt[(size_t) s2 * nseg + me];
		for (int i = 0; i < nseg; i++)
			roffs[i + 1] = roffs[i] + rcnts[i];
		uint64_t rtotal = roffs[nseg];

		int64_t *r_ck = (int64_t *) p->sget("r_ck", (rtotal + 1) * 8);
		int64_t *r_ok = (int64_t *) p->sget("r_ok", (rtotal + 1) * 8);

		if (!r_ck || !r_ok)
			return fail(GG_ENOMEM, "recv scratch");
		GG_TRY(comm_alltoallv_i64(s_ckb, offs.data(), cnts.data(),
					  r_ck, roffs.data(), rcnts.data()));
		GG_TRY(comm_alltoallv_i64(s_okb, offs.data(), cnts.data(),
					  r_ok, roffs.data(), rcnts.data()));

		unsigned long long nm = 0;
		int64_t *m_ok = (int64_t *) p->sget("m_ok", (rtotal + 1) * 8);
		int64_t *m_nat = (int64_t *) p->sget("m_pay", (rtotal + 1) * 8);

		if (!m_ok || !m_nat)
			return fail(GG_ENOMEM, "match scratch");
		GG_HIP(hipMemsetAsync(ctr, 0, 8, e.stream));
		GG_HIP(launch_probe_cust_map_compact(e.stream, r_ck, r_ok,
						     (int64_t) rtotal, cust,
						     cust_dense, cust_dlen,
						     m_ok, m_nat, ctr));
		GG_HIP(hipStreamSynchronize(e.stream));
		GG_TRY(read_counter(ctr, &nm));

		std::vector<unsigned long long> cnts2(nseg), offs2(nseg + 1, 0);

		GG_HIP(hipMemsetAsync(dcnt, 0, (size_t) nseg * 8, e.stream));
		GG_HIP(launch_part_count(e.stream, m_ok, (int64_t) nm, nseg,
					 dcnt));
		GG_HIP(hipStreamSynchronize(e.stream));
		GG_HIP(hipMemcpy(cnts2.data(), dcnt, (size_t) nseg * 8,
				 hipMemcpyDeviceToHost));
		for (int i = 0; i < nseg; i++)
			offs2[i + 1] = offs2[i] + cnts2[i];

		int64_t *s2_ok = (int64_t *) p->sget("s2_ok", (nm + 1) * 8);
		int64_t *s2_nat = (int64_t *) p->sget("s2_pay", (nm + 1) * 8);

		if (!s2_ok || !s2_nat)
			return fail(GG_ENOMEM, "send2 scratch");
		GG_HIP(hipMemcpy(doffs, offs2.data(), (size_t) nseg * 8,
				 hipMemcpyHostToDevice));
		GG_HIP(launch_part_scatter3(e.stream, m_ok, (int64_t) nm, nseg,
					    m_ok, m_nat, nullptr, doffs,
					    s2_ok, s2_nat, nullptr));
		GG_HIP(hipStreamSynchronize(e.stream));

		std::vector<unsigned long long> allcnt2((size_t) nseg * nseg);
		{
			unsigned long long *g = (unsigned long long *)
				p->sget("cntg", (size_t) (nseg + nseg * nseg) * 8);

			GG_HIP(hipMemcpy(g, cnts2.data(), (size_t) nseg * 8,
					 hipMemcpyHostToDevice));
			GG_TRY(comm_allgather_u64(g, g + nseg, nseg));
			GG_HIP(hipMemcpy(allcnt2.data(), g + nseg,
					 allcnt2.size() * 8,
					 hipMemcpyDeviceToHost));
		}
		std::vector<unsigned long long> rcnts2(nseg), roffs2(nseg + 1, 0);

		for (int s2 = 0; s2 < nseg; s2++)
			rcnts2[s2] = allcnt2[(size_t) s2 * nseg + me];
		for (int i = 0; i < nseg; i++)
			roffs2[i + 1] = roffs2[i] + rcnts2[i];
		uint64_t rtotal2 = roffs2[nseg];

		int64_t *r2_ok = (int64_t *) p->sget("r2_ok", (rtotal2 + 1) * 8);
		int64_t *r2_nat = (int64_t *) p->sget("r2_pay", (rtotal2 + 1) * 8);

		if (!r2_ok || !r2_nat)
			return fail(GG_ENOMEM, "recv2 scratch");
		GG_TRY(comm_alltoallv_i64(s2_ok, offs2.data(), cnts2.data(),
					  r2_ok, roffs2.data(), rcnts2.data()));
		GG_TRY(comm_alltoallv_i64(s2_nat, offs2.data(), cnts2.data(),
					  r2_nat, roffs2.data(),
					  rcnts2.data()));

		nmatch = rtotal2;
		if (ord_dlen)
		{
			GG_HIP(launch_dn_insert_orders_q5_u8(
				e.stream, r2_ok, r2_nat, (int64_t) rtotal2,
				ordd_pay8, ord_dlen));
			GG_HIP(hipStreamSynchronize(e.stream));
		}
		else
		{
		if (!p->ord_slots)
			p->ord_slots = next_pow2(2 * (rtotal2 + 1));
		ord.nslots = p->ord_slots;
		ord.bloom_words = ord.nslots / 8 < 1024 ? 1024 : ord.nslots / 8;
		ord.keys = (unsigned long long *)
			p->sget("ord.keys", ord.nslots * 8);
		ord.payload = (unsigned long long *)
			p->sget("ord.pay", ord.nslots * 8);
		ord.bloom = (unsigned long long *)
			p->sget("ord.bloom", ord.bloom_words * 8);
		if (!ord.keys || !ord.payload || !ord.bloom)
			return fail(GG_ENOMEM, "ord table");
		GG_HIP(hipMemsetAsync(ord.keys, 0, ord.nslots * 8, e.stream));
		GG_HIP(hipMemsetAsync(ord.bloom, 0, ord.bloom_words * 8,
				      e.stream));
		GG_HIP(launch_insert_orders(e.stream, r2_ok, r2_nat,
					    (int64_t) rtotal2, ord));
		GG_HIP(hipStreamSynchronize(e.stream));
		}
		double ms = tm.stop();
		KernelStatAcc &st = p->stat("orders_exchange");

		st.launches++;
		st.total_ms += ms;
		st.rows_in += od->nrows;
		st.rows_out += (int64_t) nmatch;
	}

	/* 4. lineitem probe + 25-nation aggregation */
	unsigned long long *acc = (unsigned long long *)
		p->sget("q5acc", GG_NNATIONS * 3 * 8);
	unsigned long long njoin = 0;

	if (!acc)
		return fail(GG_ENOMEM, "acc");
	GG_HIP(hipMemsetAsync(acc, 0, GG_NNATIONS * 3 * 8, e.stream));
	GG_HIP(hipMemsetAsync(ctr, 0, 8, e.stream));
	{
		Timed tm(e.stream);

		bool twopass_done = false;

		if (ord_dlen && getenv("GG_Q5_TWOPASS"))
		{
			/* EXPERIMENT (measured equal to the fused probe at
			 * SF100: 2.38 vs 2.32 ms — the compact pass's 4.8 GB
			 * stream costs what the skipped wide-column lines
			 * saved; kept behind GG_Q5_TWOPASS as evidence):
			 * compact order-matching rows into per-block
			 * regions (LDS counters, no global returning
			 * atomics), then gather the wide columns for the
			 * ~3% survivors.  Overflow falls back to the fused
			 * probe. */
			/* must equal the launcher's dn_grid(n) */
			int grid = (int) ((li->nrows + 255) / 256);

			if (grid > 2048)
				grid = 2048;
			if (grid < 1)
				grid = 1;
			int64_t per_block =
				(li->nrows + (int64_t) grid * 256 - 1) /
				((int64_t) grid * 256) * 256;
			int64_t region = per_block / 4 + 4096;
			unsigned long long *comp = (unsigned long long *)
				p->sget("q5.comp",
					(size_t) grid * region * 8);
			unsigned long long *ccnt = (unsigned long long *)
				p->sget("q5.ccnt", (size_t) grid * 8 + 8);

			if (!comp || !ccnt)
				return fail(GG_ENOMEM, "q5 compact");
			unsigned long long *ovf = ccnt + grid;

			GG_HIP(hipMemsetAsync(ovf, 0, 8, e.stream));
			GG_HIP(launch_dn_q5_compact(e.stream, l_ok,
						    li->nrows, ordd_pay8,
						    ord_dlen, region, comp,
						    ccnt));
			GG_HIP(launch_dn_q5_gather(e.stream, comp, ccnt,
						   region, grid, l_sk, l_pc,
						   l_dc, supp_dense,
						   supp_dense_len, acc, ctr,
						   ovf));
			unsigned long long hovf = 0;

			GG_HIP(hipStreamSynchronize(e.stream));
			GG_TRY(read_counter(ovf, &hovf));
			if (!hovf)
				twopass_done = true;
			else
			{	/* region overflow: redo with the fused
				 * one-pass probe */
				GG_HIP(hipMemsetAsync(acc, 0,
						      GG_NNATIONS * 3 * 8,
						      e.stream));
				GG_HIP(hipMemsetAsync(ctr, 0, 8, e.stream));
			}
		}
		if (ord_dlen)
		{
			if (!twopass_done)
				GG_HIP(launch_dn_probe_lineitem_q5_u8(
					e.stream, l_ok, l_sk, l_pc, l_dc,
					li->nrows, ordd_pay8, ord_dlen,
					supp_dense, supp_dense_len, acc,
					ctr));
		}
		else
			GG_HIP(launch_probe_lineitem_q5(
				e.stream, l_ok, l_sk, l_pc, l_dc, li->nrows,
				ord, supp, supp_dense, supp_dense_len, acc,
				ctr));
		double ms = tm.stop();
		GG_TRY(read_counter(ctr, &njoin));
		KernelStatAcc &st = p->stat("probe_lineitem_q5");

		st.launches++;
		st.total_ms += ms;
		st.rows_in += li->nrows;
		st.rows_out += (int64_t) njoin;
		st.hbm_bytes += li->nrows * 32;	/* okey,skey,price,disc */
	}

	/* 5. combine + finalize */
	std::vector<unsigned long long> hacc(GG_NNATIONS * 3);

	GG_HIP(hipMemcpy(hacc.data(), acc, hacc.size() * 8,
			 hipMemcpyDeviceToHost));

	u128 rev[GG_NNATIONS];
	uint64_t cnt[GG_NNATIONS];

	for (int n = 0; n < GG_NNATIONS; n++)
	{
		cnt[n] = hacc[n * 3];
		rev[n] = ((u128) hacc[n * 3 + 1]) |
			((u128) hacc[n * 3 + 2] << 64);
	}
	if (nseg > 1)
	{
		size_t per = GG_NNATIONS * 3;
		unsigned long long *g = (unsigned long long *)
			p->sget("q5gather", (per + per * (size_t) nseg) * 8);

		if (!g)
			return fail(GG_ENOMEM, "q5 gather");
		GG_HIP(hipMemcpy(g, acc, per * 8, hipMemcpyDeviceToDevice));
		GG_TRY(comm_allgather_u64(g, g + per, per));
		std::vector<unsigned long long> all(per * (size_t) nseg);

		GG_HIP(hipMemcpy(all.data(), g + per, all.size() * 8,
				 hipMemcpyDeviceToHost));
		for (int n = 0; n < GG_NNATIONS; n++)
		{
			cnt[n] = 0;
			rev[n] = 0;
		}
		for (int r = 0; r < nseg; r++)
			for (int n = 0; n < GG_NNATIONS; n++)
			{
				const unsigned long long *v =
					&all[per * (size_t) r + (size_t) n * 3];

				cnt[n] += v[0];
				rev[n] += ((u128) v[1]) | ((u128) v[2] << 64);
			}
	}

	std::vector<gg_q5_result_row> rows;

	for (int n = 0; n < GG_NNATIONS; n++)
	{
		if (!cnt[n])
			continue;
		gg_q5_result_row r;

		r.nationkey = n;
		r._pad = 0;
		r.count = (int64_t) cnt[n];
		r.rev_lo = (uint64_t) rev[n];
		r.rev_hi = (int64_t) (rev[n] >> 64);
		rows.push_back(r);
	}
	std::sort(rows.begin(), rows.end(), Q5RowCmp());

	gg_q5_result_hdr *hdr = (gg_q5_result_hdr *) arena;

	hdr->n_out = (int64_t) rows.size();
	std::memcpy(hdr + 1, rows.data(),
		    rows.size() * sizeof(gg_q5_result_row));
	*written = sizeof(*hdr) + rows.size() * sizeof(gg_q5_result_row);
	return GG_OK;
}

extern "C" gg_status
gg_engine_execute(gg_pipeline h, void *arena, size_t bytes, size_t *written)
{
	Engine &e = engine();

	if (!e.inited)
		return fail(GG_ESTATE, "engine not initialized");
	if (h < 0 || (size_t) h >= e.pipelines.size() || !e.pipelines[h])
		return fail(GG_EINVAL, "bad pipeline handle");
	if (!arena || !written)
		return fail(GG_EINVAL, "null arena");
	Pipeline *p = e.pipelines[h];

	if ((int) p->desc.kind == GG_PIPE_PLAN_INTERNAL)
		return exec_plan(p, arena, bytes, written);
	switch (p->desc.kind)
	{
		case GG_PIPE_Q1:
			return exec_q1(p, arena, bytes, written);
		case GG_PIPE_Q3:
			return exec_q3(p, arena, bytes, written);
		case GG_PIPE_SUMPRICE:
			return exec_sumprice(p, arena, bytes, written);
		case GG_PIPE_Q5:
			return exec_q5(p, arena, bytes, written);
	}
	return fail(GG_ENOTSUP, "unsupported pipeline");
}

/* ---------------- comm ---------------- */

extern "C" gg_status gg_engine_comm_id(void *out_id128)
{
	if (!out_id128)
		return fail(GG_EINVAL, "null id");
	return comm_make_id(out_id128);
}

extern "C" gg_status gg_engine_comm_init(const void *id128)
{
	if (!engine().inited)
		return fail(GG_ESTATE, "engine not initialized");
	if (!id128)
		return fail(GG_EINVAL, "null id");
	return comm_init(id128);
}

extern "C" gg_status gg_engine_comm_destroy(void)
{
	return comm_destroy();
}

/* ---------------- stats ---------------- */

extern "C" gg_status
gg_engine_stats(gg_pipeline h, gg_kernel_stat *out, int cap, int *out_n)
{
	Engine &e = engine();

	if (h < 0 || (size_t) h >= e.pipelines.size() || !e.pipelines[h])
		return fail(GG_EINVAL, "bad pipeline handle");
	Pipeline *p = e.pipelines[h];
	int n = 0;

	for (auto &s : p->stats)
	{
		if (n >= cap)
			break;
		std::snprintf(out[n].name, sizeof(out[n].name), "%s",
			      s.name.c_str());
		out[n].launches = s.launches;
		out[n].total_ms = s.total_ms;
		out[n].rows_in = s.rows_in;
		out[n].rows_out = s.rows_out;
		out[n].hbm_bytes_algorithmic = s.hbm_bytes;
		n++;
	}
	*out_n = n;
	return GG_OK;
}




/* ---------------- AOCS datum-stream decode (ABI surface) ---------------- */

extern "C" gg_status
gg_engine_aocs_decode(const uint8_t *stream, int64_t stream_len, int version,
		      int datumlen, void *out_vals, int out_width,
		      uint8_t *out_nulls, int64_t cap, int64_t *out_nrows)
{
	Engine &e = engine();

	if (!e.inited)
		return fail(GG_ESTATE, "engine not initialized");
	if (!stream || stream_len < 0 || (version < 0 || version > 2) ||
	    (datumlen != 4 && datumlen != 8) ||
	    (out_width != 4 && out_width != 8) || !out_vals || !out_nulls ||
	    !out_nrows)
		return fail(GG_EINVAL, "bad aocs_decode args");

	/* host parse of the frame stream into per-block descriptors */
	std::vector<int64_t> offs, out_offs;
	std::vector<int32_t> sizes, rows;
	int64_t pos = 0, total_rows = 0;

	while (pos < stream_len)
	{
		int32_t sz, rc;

		if (pos + 8 > stream_len)
			return fail(GG_EINVAL, "truncated frame header");
		std::memcpy(&sz, stream + pos, 4);
		std::memcpy(&rc, stream + pos + 4, 4);
		if (sz < 16 || pos + 8 + sz > stream_len || rc < 0)
			return fail(GG_EINVAL, "bad frame at %lld",
				    (long long) pos);
		offs.push_back(pos + 8);
		sizes.push_back(sz);
		rows.push_back(rc);
		out_offs.push_back(total_rows);
		total_rows += rc;
		/* frames are 8-aligned, mirroring AO storage block alignment */
		pos = (pos + 8 + sz + 7) & ~(int64_t) 7;
	}
	if (total_rows > cap)
		return fail(GG_EINVAL, "cap %lld < rows %lld",
			    (long long) cap, (long long) total_rows);
	*out_nrows = total_rows;
	if (total_rows == 0)
		return GG_OK;

	uint8_t *d_stream = nullptr;
	int64_t *d_offs = nullptr, *d_oo = nullptr;
	int32_t *d_sizes = nullptr, *d_rows = nullptr;
	void *d_vals = nullptr;
	uint8_t *d_nulls = nullptr;
	unsigned long long *d_err = nullptr;
	size_t nb = offs.size();
	gg_status st = GG_OK;

	GG_HIP(hipMalloc((void **) &d_stream, (size_t) stream_len));
	GG_HIP(hipMalloc((void **) &d_offs, nb * 8));
	GG_HIP(hipMalloc((void **) &d_oo, nb * 8));
	GG_HIP(hipMalloc((void **) &d_sizes, nb * 4));
	GG_HIP(hipMalloc((void **) &d_rows, nb * 4));
	GG_HIP(hipMalloc(&d_vals, (size_t) total_rows * out_width));
	GG_HIP(hipMalloc((void **) &d_nulls, (size_t) total_rows));
	GG_HIP(hipMalloc((void **) &d_err, 8));
	GG_HIP(hipMemcpy(d_stream, stream, (size_t) stream_len,
			 hipMemcpyHostToDevice));
	GG_HIP(hipMemcpy(d_offs, offs.data(), nb * 8, hipMemcpyHostToDevice));
	GG_HIP(hipMemcpy(d_oo, out_offs.data(), nb * 8,
			 hipMemcpyHostToDevice));
	GG_HIP(hipMemcpy(d_sizes, sizes.data(), nb * 4,
			 hipMemcpyHostToDevice));
	GG_HIP(hipMemcpy(d_rows, rows.data(), nb * 4, hipMemcpyHostToDevice));
	GG_HIP(hipMemset(d_err, 0, 8));
	{
		hipError_t he = launch_dsb_decode(
			e.stream, d_stream, d_offs, d_sizes, d_rows, d_oo,
			(int32_t) nb, version, datumlen, d_vals, d_nulls,
			out_width, d_err);

		if (he != hipSuccess)
			st = fail(GG_EGPU, "dsb_decode: %s",
				  hipGetErrorString(he));
	}
	if (st == GG_OK)
	{
		unsigned long long herr = 0;

		GG_HIP(hipStreamSynchronize(e.stream));
		GG_HIP(hipMemcpy(&herr, d_err, 8, hipMemcpyDeviceToHost));
		if (herr)
			st = fail(GG_EINVAL,
				  "block decode error mask 0x%llx", herr);
		else
		{
			GG_HIP(hipMemcpy(out_vals, d_vals,
					 (size_t) total_rows * out_width,
					 hipMemcpyDeviceToHost));
			GG_HIP(hipMemcpy(out_nulls, d_nulls,
					 (size_t) total_rows,
					 hipMemcpyDeviceToHost));
		}
	}
	(void) hipFree(d_stream);
	(void) hipFree(d_offs);
	(void) hipFree(d_oo);
	(void) hipFree(d_sizes);
	(void) hipFree(d_rows);
	(void) hipFree(d_vals);
	(void) hipFree(d_nulls);
	(void) hipFree(d_err);
	return st;
}

/* ------------- AO storage-block layer (headers + CRC32C) ------------- */

/*
 * CRC-32C (Castagnoli), restated from the reference's software
 * implementation (src/port/pg_crc32c_sb8.c; table generation
 * equivalent to the byte-at-a-time reflected form, poly 0x82F63B78).
 * The AO layer seeds with 0xFFFFFFFF and — "by historical accident"
 * (cdbappendonlystorageformat.c:40) — does NOT invert at the end.
 */
static uint32_t ao_crc32c_table[256];
static bool ao_crc32c_ready = false;

static void
ao_crc32c_init(void)
{
	for (uint32_t i = 0; i < 256; i++)
	{
		uint32_t c = i;

		for (int k = 0; k < 8; k++)
			c = (c & 1) ? (0x82F63B78u ^ (c >> 1)) : (c >> 1);
		ao_crc32c_table[i] = c;
	}
	ao_crc32c_ready = true;
}

static uint32_t
ao_crc32c(const uint8_t *p, int64_t len)
{
	uint32_t crc = 0xFFFFFFFFu;

	if (!ao_crc32c_ready)
		ao_crc32c_init();
	for (int64_t i = 0; i < len; i++)
		crc = ao_crc32c_table[(crc ^ p[i]) & 0xFF] ^ (crc >> 8);
	return crc;		/* no final inversion (see above) */
}

/*
 * Decode a stream of REAL Append-Only storage blocks — the reference's
 * on-disk segfile block format (cdbappendonlystorageformat.c;
 * bit layout cdbappendonlystorage_int.h:139–147, :317–325):
 *   word0: reserved(1) kind(3) hasFirstRowNum(1) executorBlockKind(3) …
 *   SmallContent(kind=1): rowCount 14 bits (w0>>10), dataLength split
 *     10+11 bits across w0/w1, compressedLength w1 & 0x1FFFFF;
 *   NonBulkDenseContent(kind=3): dataLength w0 & 0x1FFFFF,
 *     largeRowCount w1 & 0x3FFFFFFF.
 * Layout: [8B header][blockCrc][headerCrc][firstRowNum?][content,
 * rounded up to 8B for AORelationVersion_Aligned64bit=2
 * (AOStorage_RoundUp, cdbappendonlystorage.h)].  Header CRC covers
 * bytes [0,12), block CRC covers [16, overall) — both uninverted
 * CRC-32C (AddBlockHeaderChecksums, cdbappendonlystorageformat.c:125).
 *
 * The parsed blocks' datum-stream content is then decoded on the GPU
 * via gg_engine_aocs_decode.  Compressed blocks (compressedLength!=0)
 * and ao_version<2 are out of scope this round.
 */

extern "C" gg_status
gg_engine_aocs_decode_ao(const uint8_t *stream, int64_t stream_len,
			 int checksums, int ao_version, int dsb_version,
			 int comptype, int datumlen, void *out_vals,
			 int out_width, uint8_t *out_nulls, int64_t cap,
			 int64_t *out_nrows)
{
	if (!out_vals || !out_nulls || !out_nrows ||
	    (out_width != 4 && out_width != 8) ||
	    (datumlen != 4 && datumlen != 8))
		return fail(GG_EINVAL, "bad aocs_decode_ao args");

	/* zero-copy parse (host-only — corruption is detected here even
	 * without a GPU): uncompressed content decodes straight from the
	 * uploaded segfile bytes; only decompressed/reassembled content
	 * travels through the spill */
	std::vector<AoDesc> descs;
	std::vector<uint8_t> spill;
	gg_status st = ao_parse_blocks(stream, stream_len, checksums,
				       ao_version, comptype, descs, spill);

	if (st != GG_OK)
		return st;

	Engine &e = engine();

	if (!e.inited)
		return fail(GG_ESTATE, "engine not initialized");

	void *d_vals = nullptr;
	uint8_t *d_nulls = nullptr;
	int64_t total_rows = 0;

	st = ao_decode_to_device(e, stream, stream_len, descs, spill,
				 dsb_version, datumlen, out_width, &d_vals,
				 &d_nulls, &total_rows);
	if (st != GG_OK)
		return st;
	if (total_rows > cap)
	{
		(void) hipFree(d_vals);
		(void) hipFree(d_nulls);
		return fail(GG_EINVAL, "cap %lld < rows %lld",
			    (long long) cap, (long long) total_rows);
	}
	*out_nrows = total_rows;
	if (total_rows)
	{
		hipError_t he = hipMemcpy(out_vals, d_vals,
					  (size_t) total_rows * out_width,
					  hipMemcpyDeviceToHost);

		if (he == hipSuccess)
			he = hipMemcpy(out_nulls, d_nulls,
				       (size_t) total_rows,
				       hipMemcpyDeviceToHost);
		if (he != hipSuccess)
			st = fail(GG_EGPU, "decode_ao copyback: %s",
				  hipGetErrorString(he));
	}
	(void) hipFree(d_vals);
	(void) hipFree(d_nulls);
	return st;
}

/* same, for TEXT columns: AO layer (headers, checksums, codecs) on the
 * host via the zero-copy parallel parser, varlena datum-stream decode
 * on the GPU.  Output offsets reference `pool`. */
extern "C" gg_status
gg_engine_aocs_decode_ao_text(const uint8_t *stream, int64_t stream_len,
			      int checksums, int ao_version,
			      int dsb_version, int comptype,
			      uint64_t *out_offs, uint32_t *out_lens,
			      uint8_t *out_nulls, int64_t cap,
			      uint8_t *pool, int64_t pool_cap,
			      int64_t *out_nrows, int64_t *out_pool_len)
{
	if (!out_offs || !out_lens || !out_nulls || !pool || !out_nrows ||
	    !out_pool_len)
		return fail(GG_EINVAL, "bad aocs_decode_ao_text args");

	std::vector<AoDesc> descs;
	std::vector<uint8_t> spill;
	gg_status st = ao_parse_blocks(stream, stream_len, checksums,
				       ao_version, comptype, descs, spill);

	if (st != GG_OK)
		return st;

	Engine &e = engine();

	if (!e.inited)
		return fail(GG_ESTATE, "engine not initialized");

	size_t nb = descs.size();
	int64_t total_rows = 0, pool_len = 0;
	std::vector<int64_t> offs(nb), out_off(nb), pool_off(nb);
	std::vector<int32_t> sizes(nb), rows(nb);

	for (size_t i = 0; i < nb; i++)
	{
		offs[i] = descs[i].off;
		sizes[i] = descs[i].size;
		rows[i] = descs[i].rowcount;
		out_off[i] = total_rows;
		pool_off[i] = pool_len;
		total_rows += descs[i].rowcount;
		pool_len += descs[i].size;
	}
	if (total_rows > cap)
		return fail(GG_EINVAL, "cap %lld < rows %lld",
			    (long long) cap, (long long) total_rows);
	if (pool_len > pool_cap)
		return fail(GG_EINVAL, "pool_cap %lld < %lld",
			    (long long) pool_cap, (long long) pool_len);
	*out_nrows = total_rows;
	*out_pool_len = pool_len;
	if (total_rows == 0)
		return GG_OK;

	uint8_t *d_stream = nullptr, *d_spill = nullptr, *d_pool = nullptr,
		*d_nulls = nullptr;
	int64_t *d_offs = nullptr, *d_oo = nullptr, *d_po = nullptr;
	int32_t *d_sizes = nullptr, *d_rows = nullptr;
	unsigned long long *d_out_offs = nullptr;
	uint32_t *d_lens = nullptr;
	unsigned long long *d_err = nullptr;
	unsigned long long *d_srco = nullptr;

#define GG_HIP_AT(x) \
	{ hipError_t e_ = (x); \
	  if (st == GG_OK && e_ != hipSuccess) \
		st = fail(GG_EGPU, "decode_ao_text:%d: %s", __LINE__, \
			  hipGetErrorString(e_)); }
	GG_HIP_AT(hipMalloc((void **) &d_stream,
			    stream_len ? (size_t) stream_len : 1));
	GG_HIP_AT(hipMalloc((void **) &d_spill,
			    spill.size() ? spill.size() : 1));
	GG_HIP_AT(hipMalloc((void **) &d_pool, (size_t) pool_len));
	GG_HIP_AT(hipMalloc((void **) &d_offs, nb * 8));
	GG_HIP_AT(hipMalloc((void **) &d_oo, nb * 8));
	GG_HIP_AT(hipMalloc((void **) &d_po, nb * 8));
	GG_HIP_AT(hipMalloc((void **) &d_sizes, nb * 4));
	GG_HIP_AT(hipMalloc((void **) &d_rows, nb * 4));
	GG_HIP_AT(hipMalloc((void **) &d_out_offs,
			    (size_t) total_rows * 8));
	GG_HIP_AT(hipMalloc((void **) &d_lens, (size_t) total_rows * 4));
	GG_HIP_AT(hipMalloc((void **) &d_nulls, (size_t) total_rows));
	GG_HIP_AT(hipMalloc((void **) &d_err, 8));
	{ hipError_t e_ = hipMalloc((void **) &d_srco,
				    (size_t) total_rows * 8);
	  if (st == GG_OK && e_ != hipSuccess)
		st = fail(GG_EGPU, "decode_ao_text srco-alloc: %s",
			  hipGetErrorString(e_)); }
	if (st == GG_OK)
	{
		if (stream_len)
			GG_HIP_AT(hipMemcpyAsync(d_stream, stream,
						 (size_t) stream_len,
						 hipMemcpyHostToDevice,
						 e.stream));
		if (spill.size())
			GG_HIP_AT(hipMemcpyAsync(d_spill, spill.data(),
						 spill.size(),
						 hipMemcpyHostToDevice,
						 e.stream));
		GG_HIP_AT(hipMemcpyAsync(d_offs, offs.data(), nb * 8,
					 hipMemcpyHostToDevice, e.stream));
		GG_HIP_AT(hipMemcpyAsync(d_oo, out_off.data(), nb * 8,
					 hipMemcpyHostToDevice, e.stream));
		GG_HIP_AT(hipMemcpyAsync(d_po, pool_off.data(), nb * 8,
					 hipMemcpyHostToDevice, e.stream));
		GG_HIP_AT(hipMemcpyAsync(d_sizes, sizes.data(), nb * 4,
					 hipMemcpyHostToDevice, e.stream));
		GG_HIP_AT(hipMemcpyAsync(d_rows, rows.data(), nb * 4,
					 hipMemcpyHostToDevice, e.stream));
		GG_HIP_AT(hipMemsetAsync(d_err, 0, 8, e.stream));
		/* 0xFF = the ~0 "no payload" sentinel: rows of a block
		 * that errors out mid-parse stay skipped by the copy */
		{ hipError_t e_ = hipMemsetAsync(d_srco, 0xFF,
						 (size_t) total_rows * 8,
						 e.stream);
		  if (st == GG_OK && e_ != hipSuccess)
			st = fail(GG_EGPU, "decode_ao_text srco-set: %s",
				  hipGetErrorString(e_)); }
		{
			const char *sp_ = getenv("GG_TEXT_SPLIT");
			bool split_ = !(sp_ && sp_[0] == '0');

			GG_HIP_AT(launch_dsb_decode_text(
				e.stream, d_stream, d_spill, d_offs,
				d_sizes, d_rows, d_oo, d_po, (int32_t) nb,
				dsb_version, d_pool, d_out_offs, d_lens,
				d_nulls, d_err,
				split_ ? d_srco : nullptr));
			if (split_)
			{
				hipError_t e_ = launch_dsb_text_copy(
					e.stream, d_stream, d_spill,
					d_srco, d_out_offs, d_lens,
					total_rows, d_pool);

				if (st == GG_OK && e_ != hipSuccess)
					st = fail(GG_EGPU,
						  "decode_ao_text copy: %s",
						  hipGetErrorString(e_));
			}
		}
		GG_HIP_AT(hipStreamSynchronize(e.stream));
	}
	if (st == GG_OK)
	{
		unsigned long long herr = 0;

		GG_HIP_AT(hipMemcpy(&herr, d_err, 8,
				    hipMemcpyDeviceToHost));
		if (st == GG_OK && herr)
			st = fail(GG_EINVAL,
				  "text block decode error mask 0x%llx",
				  herr);
	}
	if (st == GG_OK)
	{
		GG_HIP_AT(hipMemcpy(out_offs, d_out_offs,
				    (size_t) total_rows * 8,
				    hipMemcpyDeviceToHost));
		GG_HIP_AT(hipMemcpy(out_lens, d_lens,
				    (size_t) total_rows * 4,
				    hipMemcpyDeviceToHost));
		GG_HIP_AT(hipMemcpy(out_nulls, d_nulls,
				    (size_t) total_rows,
				    hipMemcpyDeviceToHost));
		GG_HIP_AT(hipMemcpy(pool, d_pool, (size_t) pool_len,
				    hipMemcpyDeviceToHost));
	}
#undef GG_HIP_AT
	(void) hipFree(d_stream);
	(void) hipFree(d_spill);
	(void) hipFree(d_pool);
	(void) hipFree(d_offs);
	(void) hipFree(d_oo);
	(void) hipFree(d_po);
	(void) hipFree(d_sizes);
	(void) hipFree(d_rows);
	(void) hipFree(d_srco);
	(void) hipFree(d_out_offs);
	(void) hipFree(d_lens);
	(void) hipFree(d_nulls);
	(void) hipFree(d_err);
	return st;
}

/*
 * Decode a framed stream of TEXT (varlena) datum-stream blocks on the
 * GPU: per-row (pool offset, byte length, null flag) plus a byte pool.
 * Rows of one block land in a contiguous pool region; regions are
 * upper-bounded by the block content size, so pool_cap = stream_len is
 * always enough (gaps between regions are unreferenced).
 */
extern "C" gg_status
gg_engine_aocs_decode_text(const uint8_t *stream, int64_t stream_len,
			   int version, uint64_t *out_offs,
			   uint32_t *out_lens, uint8_t *out_nulls,
			   int64_t cap, uint8_t *pool, int64_t pool_cap,
			   int64_t *out_nrows, int64_t *out_pool_len)
{
	Engine &e = engine();

	if (!e.inited)
		return fail(GG_ESTATE, "engine not initialized");
	if (!stream || stream_len < 0 || version < 0 || version > 2 ||
	    !out_offs || !out_lens || !out_nulls || !pool || !out_nrows ||
	    !out_pool_len)
		return fail(GG_EINVAL, "bad aocs_decode_text args");

	std::vector<int64_t> offs, out_off, pool_off;
	std::vector<int32_t> sizes, rows;
	int64_t pos = 0, total_rows = 0, pool_len = 0;

	while (pos < stream_len)
	{
		int32_t sz, rc;

		if (pos + 8 > stream_len)
			return fail(GG_EINVAL, "truncated frame header");
		std::memcpy(&sz, stream + pos, 4);
		std::memcpy(&rc, stream + pos + 4, 4);
		if (sz < 16 || pos + 8 + sz > stream_len || rc < 0)
			return fail(GG_EINVAL, "bad frame at %lld",
				    (long long) pos);
		offs.push_back(pos + 8);
		sizes.push_back(sz);
		rows.push_back(rc);
		out_off.push_back(total_rows);
		pool_off.push_back(pool_len);
		total_rows += rc;
		pool_len += sz;	/* payload bytes <= content size */
		pos = (pos + 8 + sz + 7) & ~(int64_t) 7;
	}
	if (total_rows > cap)
		return fail(GG_EINVAL, "cap %lld < rows %lld",
			    (long long) cap, (long long) total_rows);
	if (pool_len > pool_cap)
		return fail(GG_EINVAL, "pool_cap %lld < %lld",
			    (long long) pool_cap, (long long) pool_len);
	*out_nrows = total_rows;
	*out_pool_len = pool_len;
	if (total_rows == 0)
		return GG_OK;

	gg_status st = GG_OK;
	uint8_t *d_stream = nullptr, *d_pool = nullptr, *d_nulls = nullptr;
	int64_t *d_offs = nullptr, *d_oo = nullptr, *d_po = nullptr;
	int32_t *d_sizes = nullptr, *d_rows = nullptr;
	unsigned long long *d_out_offs = nullptr;
	uint32_t *d_lens = nullptr;
	unsigned long long *d_err = nullptr;
	size_t nb = offs.size();

	GG_HIP(hipMalloc((void **) &d_stream, (size_t) stream_len));
	GG_HIP(hipMalloc((void **) &d_pool, (size_t) pool_len));
	GG_HIP(hipMalloc((void **) &d_offs, nb * 8));
	GG_HIP(hipMalloc((void **) &d_oo, nb * 8));
	GG_HIP(hipMalloc((void **) &d_po, nb * 8));
	GG_HIP(hipMalloc((void **) &d_sizes, nb * 4));
	GG_HIP(hipMalloc((void **) &d_rows, nb * 4));
	GG_HIP(hipMalloc((void **) &d_out_offs, (size_t) total_rows * 8));
	GG_HIP(hipMalloc((void **) &d_lens, (size_t) total_rows * 4));
	GG_HIP(hipMalloc((void **) &d_nulls, (size_t) total_rows));
	GG_HIP(hipMalloc((void **) &d_err, 8));
	unsigned long long *d_srco = nullptr;

	GG_HIP(hipMalloc((void **) &d_srco, (size_t) total_rows * 8));
	GG_HIP(hipMemcpy(d_stream, stream, (size_t) stream_len,
			 hipMemcpyHostToDevice));
	GG_HIP(hipMemcpy(d_offs, offs.data(), nb * 8,
			 hipMemcpyHostToDevice));
	GG_HIP(hipMemcpy(d_oo, out_off.data(), nb * 8,
			 hipMemcpyHostToDevice));
	GG_HIP(hipMemcpy(d_po, pool_off.data(), nb * 8,
			 hipMemcpyHostToDevice));
	GG_HIP(hipMemcpy(d_sizes, sizes.data(), nb * 4,
			 hipMemcpyHostToDevice));
	GG_HIP(hipMemcpy(d_rows, rows.data(), nb * 4,
			 hipMemcpyHostToDevice));
	GG_HIP(hipMemset(d_err, 0, 8));
	GG_HIP(hipMemsetAsync(d_srco, 0xFF, (size_t) total_rows * 8,
			      e.stream));
	{
		const char *sp_ = getenv("GG_TEXT_SPLIT");
		bool split_ = !(sp_ && sp_[0] == '0');
		hipError_t he = launch_dsb_decode_text(
			e.stream, d_stream, nullptr, d_offs, d_sizes,
			d_rows, d_oo, d_po, (int32_t) nb, version, d_pool,
			d_out_offs, d_lens, d_nulls, d_err,
			split_ ? d_srco : nullptr);

		if (split_ && he == hipSuccess)
			he = launch_dsb_text_copy(
				e.stream, d_stream, nullptr, d_srco,
				d_out_offs, d_lens, total_rows, d_pool);
		if (he != hipSuccess)
			st = fail(GG_EGPU, "dsb_decode_text: %s",
				  hipGetErrorString(he));
	}
	if (st == GG_OK)
	{
		unsigned long long herr = 0;

		GG_HIP(hipStreamSynchronize(e.stream));
		GG_HIP(hipMemcpy(&herr, d_err, 8, hipMemcpyDeviceToHost));
		if (herr)
			st = fail(GG_EINVAL,
				  "text block decode error mask 0x%llx",
				  herr);
		else
		{
			GG_HIP(hipMemcpy(out_offs, d_out_offs,
					 (size_t) total_rows * 8,
					 hipMemcpyDeviceToHost));
			GG_HIP(hipMemcpy(out_lens, d_lens,
					 (size_t) total_rows * 4,
					 hipMemcpyDeviceToHost));
			GG_HIP(hipMemcpy(out_nulls, d_nulls,
					 (size_t) total_rows,
					 hipMemcpyDeviceToHost));
			GG_HIP(hipMemcpy(pool, d_pool, (size_t) pool_len,
					 hipMemcpyDeviceToHost));
		}
	}
	(void) hipFree(d_stream);
	(void) hipFree(d_pool);
	(void) hipFree(d_offs);
	(void) hipFree(d_oo);
	(void) hipFree(d_po);
	(void) hipFree(d_sizes);
	(void) hipFree(d_rows);
	(void) hipFree(d_out_offs);
	(void) hipFree(d_lens);
	(void) hipFree(d_nulls);
	(void) hipFree(d_err);
	(void) hipFree(d_srco);
	return st;
}

/* --------- text dictionary encode (ABI surface, §8 a4) --------- */

/*
 * Dictionary-encode a categorical text column (the arrow-style shape
 * gg_engine_aocs_decode_text outputs) into per-row int32 codes + a
 * lexicographically sorted dictionary: codes are deterministic across
 * shards/runs, which the multi-GPU exchange requires.  NULL rows get
 * code -1.  max_dict bounds the cardinality (error beyond — this is
 * for categorical columns like c_mktsegment, not free text).
 */
extern "C" gg_status
gg_engine_text_dict_encode(const uint8_t *pool, const uint64_t *offs,
			   const uint32_t *lens, const uint8_t *nulls,
			   int64_t n, int32_t max_dict, int32_t *out_codes,
			   uint8_t *dict_bytes, int64_t dict_cap,
			   int64_t *dict_offs /* max_dict+1 */ ,
			   int32_t *out_ndict)
{
	Engine &e = engine();

	if (!e.inited)
		return fail(GG_ESTATE, "engine not initialized");
	if (!pool || !offs || !lens || n < 0 || max_dict < 1 ||
	    max_dict > (1 << 20) || !out_codes || !dict_bytes ||
	    !dict_offs || !out_ndict)
		return fail(GG_EINVAL, "bad text_dict args");
	*out_ndict = 0;
	if (n == 0)
	{
		dict_offs[0] = 0;
		return GG_OK;
	}

	int64_t pool_len = 0;

	for (int64_t i = 0; i < n; i++)
		if ((!nulls || !nulls[i]) &&
		    (int64_t) (offs[i] + lens[i]) > pool_len)
			pool_len = (int64_t) (offs[i] + lens[i]);

	uint64_t nslots = next_pow2(4 * (uint64_t) max_dict);
	gg_status st = GG_OK;
	uint8_t *d_pool = nullptr, *d_nulls = nullptr;
	unsigned long long *d_offs = nullptr, *d_slots = nullptr,
		*d_err = nullptr;
	uint32_t *d_lens = nullptr, *d_rowslot = nullptr;
	int32_t *d_map = nullptr, *d_codes = nullptr;

	GG_HIP(hipMalloc((void **) &d_pool, (size_t) (pool_len + 1)));
	GG_HIP(hipMalloc((void **) &d_offs, (size_t) n * 8));
	GG_HIP(hipMalloc((void **) &d_lens, (size_t) n * 4));
	GG_HIP(hipMalloc((void **) &d_rowslot, (size_t) n * 4));
	GG_HIP(hipMalloc((void **) &d_codes, (size_t) n * 4));
	GG_HIP(hipMalloc((void **) &d_slots, nslots * 8));
	GG_HIP(hipMalloc((void **) &d_map, nslots * 4));
	GG_HIP(hipMalloc((void **) &d_err, 8));
	if (pool_len)
		GG_HIP(hipMemcpy(d_pool, pool, (size_t) pool_len,
				 hipMemcpyHostToDevice));
	GG_HIP(hipMemcpy(d_offs, offs, (size_t) n * 8,
			 hipMemcpyHostToDevice));
	GG_HIP(hipMemcpy(d_lens, lens, (size_t) n * 4,
			 hipMemcpyHostToDevice));
	if (nulls)
	{
		GG_HIP(hipMalloc((void **) &d_nulls, (size_t) n));
		GG_HIP(hipMemcpy(d_nulls, nulls, (size_t) n,
				 hipMemcpyHostToDevice));
	}
	GG_HIP(hipMemsetAsync(d_slots, 0, nslots * 8, e.stream));
	GG_HIP(hipMemsetAsync(d_err, 0, 8, e.stream));
	{
		hipError_t he = launch_td_insert(
			e.stream, d_pool, d_offs, d_lens, d_nulls, n,
			d_slots, nslots, d_rowslot, d_err);

		if (he != hipSuccess)
			st = fail(GG_EGPU, "td_insert: %s",
				  hipGetErrorString(he));
	}
	std::vector<unsigned long long> hslots;

	if (st == GG_OK)
	{
		unsigned long long herr = 0;

		GG_HIP(hipStreamSynchronize(e.stream));
		GG_HIP(hipMemcpy(&herr, d_err, 8, hipMemcpyDeviceToHost));
		if (herr)
			st = fail(GG_EINVAL, "dictionary overflow "
				  "(> %d distinct values)", (int) max_dict);
	}
	if (st == GG_OK)
	{
		hslots.resize(nslots);
		GG_HIP(hipMemcpy(hslots.data(), d_slots, nslots * 8,
				 hipMemcpyDeviceToHost));

		/* unique strings -> lexicographic ids (bytewise compare,
		 * text_lt with C collation) */
		struct Ent
		{
			uint64_t slot;
			int64_t row;
		};
		std::vector<Ent> ents;

		for (uint64_t sl = 0; sl < nslots; sl++)
			if (hslots[sl])
				ents.push_back({sl,
						(int64_t) hslots[sl] - 1});
		if ((int32_t) ents.size() > max_dict)
			st = fail(GG_EINVAL, "dictionary overflow "
				  "(%zu > %d)", ents.size(),
				  (int) max_dict);
		if (st == GG_OK)
		{
			std::sort(ents.begin(), ents.end(),
				  [&](const Ent & a, const Ent & b)
				  {
				  const uint8_t *sa = pool + offs[a.row];
				  const uint8_t *sb = pool + offs[b.row];
				  uint32_t la = lens[a.row],
				  lb = lens[b.row];
				  uint32_t m = la < lb ? la : lb;

				  for (uint32_t z = 0; z < m; z++)
				  if (sa[z] != sb[z])
				  return sa[z] < sb[z];
				  return la < lb;
				  });
			std::vector<int32_t> slot_to_id(nslots, -1);
			int64_t dpos = 0;

			dict_offs[0] = 0;
			for (size_t id = 0; id < ents.size(); id++)
			{
				slot_to_id[ents[id].slot] = (int32_t) id;
				uint32_t l = lens[ents[id].row];

				if (dpos + l > dict_cap)
					return fail(GG_EINVAL,
						    "dict_cap too small");
				std::memcpy(dict_bytes + dpos,
					    pool + offs[ents[id].row], l);
				dpos += l;
				dict_offs[id + 1] = dpos;
			}
			*out_ndict = (int32_t) ents.size();
			GG_HIP(hipMemcpy(d_map, slot_to_id.data(),
					 nslots * 4,
					 hipMemcpyHostToDevice));
			{
				hipError_t he = launch_td_map(
					e.stream, d_rowslot, n, d_map,
					d_codes);

				if (he != hipSuccess)
					st = fail(GG_EGPU, "td_map: %s",
						  hipGetErrorString(he));
			}
			if (st == GG_OK)
			{
				GG_HIP(hipStreamSynchronize(e.stream));
				GG_HIP(hipMemcpy(out_codes, d_codes,
						 (size_t) n * 4,
						 hipMemcpyDeviceToHost));
			}
		}
	}
	(void) hipFree(d_pool);
	(void) hipFree(d_offs);
	(void) hipFree(d_lens);
	(void) hipFree(d_rowslot);
	(void) hipFree(d_codes);
	(void) hipFree(d_slots);
	(void) hipFree(d_map);
	(void) hipFree(d_err);
	(void) hipFree(d_nulls);
	return st;
}

/* -------------- MemTuple codec (ABI surface, §8(f)3) -------------- */

/*
 * Binding introspection (host-only — CPU-testable): writes
 * [column_align, null_bitmap_extra, var_start] then per attr
 * [offset, len, len_aligned, null_byte, null_mask] into out
 * (3 + 5*natts int32s).  Pinned against the reference's
 * create_memtuple_binding in tests.
 */
static gg_status
mt_binding_dump(int natts, const int32_t *attlen, const char *attalign,
		bool large, int32_t *out)
{
	if (!attlen || !attalign || !out)
		return fail(GG_EINVAL, "bad memtuple_binding args");

	MtBind b;
	int rc = large
		? mt_compute_binding_large(natts, attlen, attalign, &b)
		: mt_compute_binding(natts, attlen, attalign, &b);

	if (rc)
		return fail(GG_EINVAL, "unsupported memtuple schema (%d)",
			    rc);
	out[0] = b.column_align;
	out[1] = b.null_bitmap_extra;
	out[2] = b.var_start;
	for (int i = 0; i < natts; i++)
	{
		out[3 + i * 5 + 0] = b.offset[i];
		out[3 + i * 5 + 1] = b.len[i];
		out[3 + i * 5 + 2] = b.len_aligned[i];
		out[3 + i * 5 + 3] = b.null_byte[i];
		out[3 + i * 5 + 4] = b.null_mask[i];
	}
	return GG_OK;
}

extern "C" gg_status
gg_engine_memtuple_binding_large(int natts, const int32_t *attlen,
				 const char *attalign, int32_t *out)
{
	return mt_binding_dump(natts, attlen, attalign, true, out);
}

extern "C" gg_status
gg_engine_memtuple_binding(int natts, const int32_t *attlen,
			   const char *attalign, int32_t *out)
{
	return mt_binding_dump(natts, attlen, attalign, false, out);
}

/*
 * Bulk-encode host column arrays into a MemTuple byte stream (GPU does
 * the per-tuple layout).  cols[i] points at nrows elements of width
 * attlen[i]; nulls[i] is a byte-per-row flag array or NULL for a
 * NOT NULL column.  See memtuple.hip for the restated format rules.
 */
extern "C" gg_status
gg_engine_memtuple_encode(int natts, const int32_t *attlen,
			  const char *attalign, const void *const *cols,
			  const uint8_t *const *nulls, int64_t nrows,
			  uint8_t *out, int64_t cap, int64_t *out_len)
{
	Engine &e = engine();

	if (!e.inited)
		return fail(GG_ESTATE, "engine not initialized");
	if (!attlen || !attalign || !cols || nrows < 0 || !out || !out_len)
		return fail(GG_EINVAL, "bad memtuple_encode args");

	MtBind b, bl;
	int rc = mt_compute_binding(natts, attlen, attalign, &b);

	if (rc)
		return fail(GG_EINVAL, "unsupported memtuple schema (%d)",
			    rc);
	(void) mt_compute_binding_large(natts, attlen, attalign, &bl);

	/* per-row lengths + offsets on the host (it owns the null flags
	 * and varlena offsets); mirrors d_mt_len incl. the small->large
	 * binding switch over MEMTUPLE_LEN_FITSHORT */
	auto row_len = [&](const MtBind & bb, int64_t r) -> int64_t
	{
		int64_t len = bb.var_start;
		bool hasnull = false;

		for (int i = 0; i < natts; i++)
			if (nulls && nulls[i] && nulls[i][r])
			{
				hasnull = true;
				len -= bb.len_aligned[i];
			}
		if (hasnull)
			len += bb.null_bitmap_extra;
		for (int i = 0; i < natts; i++)
		{
			if (!bb.is_var[i] ||
			    (nulls && nulls[i] && nulls[i][r]))
				continue;
			const gg_text_col *tc =
				(const gg_text_col *) cols[i];
			int64_t paylen = tc->offs[r + 1] - tc->offs[r];

			if (paylen + 1 <= 0x7F)
				len += paylen + 1;
			else
			{
				int al = bb.align_of[i];

				len = (len + al - 1) & ~(int64_t) (al - 1);
				len += 4 + paylen;
			}
		}
		return (len + 7) & ~(int64_t) 7;
	};
	std::vector<int64_t> offs(nrows + 1, 0);

	for (int64_t r = 0; r < nrows; r++)
	{
		int64_t len = row_len(b, r);

		if (len > 0xFFF0)
			len = row_len(bl, r);
		offs[r + 1] = offs[r] + len;
	}
	int64_t total = offs[nrows];

	if (total > cap)
		return fail(GG_EINVAL, "cap %lld < stream %lld",
			    (long long) cap, (long long) total);
	*out_len = total;
	if (nrows == 0)
		return GG_OK;

	gg_status st = GG_OK;
	uint8_t *d_out = nullptr;
	int64_t *d_offs = nullptr;
	std::vector<void *> d_cols(natts, nullptr);
	std::vector<int64_t *> d_voffs(natts, nullptr);
	std::vector<uint8_t *> d_nulls(natts, nullptr);
	void **d_colp = nullptr;
	int64_t **d_voffp = nullptr;
	uint8_t **d_nullp = nullptr;

	GG_HIP(hipMalloc((void **) &d_out, (size_t) total));
	GG_HIP(hipMalloc((void **) &d_offs, (size_t) (nrows + 1) * 8));
	GG_HIP(hipMemcpy(d_offs, offs.data(), (size_t) (nrows + 1) * 8,
			 hipMemcpyHostToDevice));
	for (int i = 0; i < natts; i++)
	{
		if (b.is_var[i])
		{
			const gg_text_col *tc =
				(const gg_text_col *) cols[i];
			size_t nb = (size_t) tc->offs[nrows];

			GG_HIP(hipMalloc(&d_cols[i], nb ? nb : 1));
			if (nb)
				GG_HIP(hipMemcpy(d_cols[i], tc->bytes, nb,
						 hipMemcpyHostToDevice));
			GG_HIP(hipMalloc((void **) &d_voffs[i],
					 (size_t) (nrows + 1) * 8));
			GG_HIP(hipMemcpy(d_voffs[i], tc->offs,
					 (size_t) (nrows + 1) * 8,
					 hipMemcpyHostToDevice));
		}
		else
		{
			size_t nb = (size_t) nrows * attlen[i];

			GG_HIP(hipMalloc(&d_cols[i], nb));
			GG_HIP(hipMemcpy(d_cols[i], cols[i], nb,
					 hipMemcpyHostToDevice));
		}
		if (nulls && nulls[i])
		{
			GG_HIP(hipMalloc((void **) &d_nulls[i],
					 (size_t) nrows));
			GG_HIP(hipMemcpy(d_nulls[i], nulls[i],
					 (size_t) nrows,
					 hipMemcpyHostToDevice));
		}
	}
	GG_HIP(hipMalloc((void **) &d_colp, natts * sizeof(void *)));
	GG_HIP(hipMalloc((void **) &d_voffp, natts * sizeof(void *)));
	GG_HIP(hipMalloc((void **) &d_nullp, natts * sizeof(void *)));
	GG_HIP(hipMemcpy(d_colp, d_cols.data(), natts * sizeof(void *),
			 hipMemcpyHostToDevice));
	GG_HIP(hipMemcpy(d_voffp, d_voffs.data(), natts * sizeof(void *),
			 hipMemcpyHostToDevice));
	GG_HIP(hipMemcpy(d_nullp, d_nulls.data(), natts * sizeof(void *),
			 hipMemcpyHostToDevice));
	{
		hipError_t he = launch_mt_encode(
			e.stream, &b, &bl, (const void *const *) d_colp,
			(const int64_t *const *) d_voffp,
			(const uint8_t *const *) d_nullp, nrows, d_offs,
			d_out);

		if (he != hipSuccess)
			st = fail(GG_EGPU, "mt_encode: %s",
				  hipGetErrorString(he));
	}
	if (st == GG_OK)
	{
		GG_HIP(hipStreamSynchronize(e.stream));
		GG_HIP(hipMemcpy(out, d_out, (size_t) total,
				 hipMemcpyDeviceToHost));
	}
	(void) hipFree(d_out);
	(void) hipFree(d_offs);
	for (int i = 0; i < natts; i++)
	{
		(void) hipFree(d_cols[i]);
		(void) hipFree(d_voffs[i]);
		(void) hipFree(d_nulls[i]);
	}
	(void) hipFree(d_colp);
	(void) hipFree(d_voffp);
	(void) hipFree(d_nullp);
	return st;
}

extern "C" gg_status
gg_engine_memtuple_decode(int natts, const int32_t *attlen,
			  const char *attalign, const uint8_t *stream,
			  int64_t stream_len, void *const *cols,
			  uint8_t *const *nulls, int64_t cap_rows,
			  int64_t *out_nrows)
{
	Engine &e = engine();

	if (!e.inited)
		return fail(GG_ESTATE, "engine not initialized");
	if (!attlen || !attalign || !stream || stream_len < 0 || !cols ||
	    !out_nrows)
		return fail(GG_EINVAL, "bad memtuple_decode args");

	MtBind b, bl;
	int rc = mt_compute_binding(natts, attlen, attalign, &b);

	if (rc)
		return fail(GG_EINVAL, "unsupported memtuple schema (%d)",
			    rc);
	(void) mt_compute_binding_large(natts, attlen, attalign, &bl);

	/* walk tuple headers to find offsets (MEMTUP_LEN_MASK) */
	std::vector<int64_t> offs;
	int64_t pos = 0;

	while (pos < stream_len)
	{
		uint32_t hdr;

		if (pos + 8 > stream_len)
			return fail(GG_EINVAL, "truncated memtuple at %lld",
				    (long long) pos);
		std::memcpy(&hdr, stream + pos, 4);
		if (!(hdr & 0x80000000u))
			return fail(GG_EINVAL,
				    "lead bit clear at %lld (heaptuple?)",
				    (long long) pos);
		int64_t len = (int64_t) (hdr & 0x3FFFFFF8u);

		if (len < 8 || pos + len > stream_len)
			return fail(GG_EINVAL, "bad memtuple len %lld at "
				    "%lld", (long long) len,
				    (long long) pos);
		offs.push_back(pos);
		pos += len;
	}
	int64_t nrows = (int64_t) offs.size();

	if (nrows > cap_rows)
		return fail(GG_EINVAL, "cap_rows %lld < rows %lld",
			    (long long) cap_rows, (long long) nrows);
	*out_nrows = nrows;
	if (nrows == 0)
		return GG_OK;

	gg_status st = GG_OK;
	uint8_t *d_in = nullptr;
	int64_t *d_offs = nullptr;
	unsigned long long *d_err = nullptr;
	std::vector<void *> d_cols(natts, nullptr);
	std::vector<unsigned long long *> d_voffs(natts, nullptr);
	std::vector<uint32_t *> d_vlens(natts, nullptr);
	std::vector<uint8_t *> d_nulls(natts, nullptr);
	void **d_colp = nullptr;
	unsigned long long **d_voffp = nullptr;
	uint32_t **d_vlenp = nullptr;
	uint8_t **d_nullp = nullptr;

	GG_HIP(hipMalloc((void **) &d_in, (size_t) stream_len));
	GG_HIP(hipMalloc((void **) &d_offs, (size_t) nrows * 8));
	GG_HIP(hipMalloc((void **) &d_err, 8));
	GG_HIP(hipMemcpy(d_in, stream, (size_t) stream_len,
			 hipMemcpyHostToDevice));
	GG_HIP(hipMemcpy(d_offs, offs.data(), (size_t) nrows * 8,
			 hipMemcpyHostToDevice));
	GG_HIP(hipMemset(d_err, 0, 8));
	for (int i = 0; i < natts; i++)
	{
		if (b.is_var[i])
		{
			GG_HIP(hipMalloc((void **) &d_voffs[i],
					 (size_t) nrows * 8));
			GG_HIP(hipMalloc((void **) &d_vlens[i],
					 (size_t) nrows * 4));
		}
		else
			GG_HIP(hipMalloc(&d_cols[i],
					 (size_t) nrows * attlen[i]));
		GG_HIP(hipMalloc((void **) &d_nulls[i], (size_t) nrows));
	}
	GG_HIP(hipMalloc((void **) &d_colp, natts * sizeof(void *)));
	GG_HIP(hipMalloc((void **) &d_voffp, natts * sizeof(void *)));
	GG_HIP(hipMalloc((void **) &d_vlenp, natts * sizeof(void *)));
	GG_HIP(hipMalloc((void **) &d_nullp, natts * sizeof(void *)));
	GG_HIP(hipMemcpy(d_colp, d_cols.data(), natts * sizeof(void *),
			 hipMemcpyHostToDevice));
	GG_HIP(hipMemcpy(d_voffp, d_voffs.data(), natts * sizeof(void *),
			 hipMemcpyHostToDevice));
	GG_HIP(hipMemcpy(d_vlenp, d_vlens.data(), natts * sizeof(void *),
			 hipMemcpyHostToDevice));
	GG_HIP(hipMemcpy(d_nullp, d_nulls.data(), natts * sizeof(void *),
			 hipMemcpyHostToDevice));
	{
		hipError_t he = launch_mt_decode(
			e.stream, &b, &bl, d_offs, nrows, d_in, stream_len,
			(void *const *) d_colp,
			(unsigned long long *const *) d_voffp,
			(uint32_t *const *) d_vlenp,
			(uint8_t *const *) d_nullp, d_err);

		if (he != hipSuccess)
			st = fail(GG_EGPU, "mt_decode: %s",
				  hipGetErrorString(he));
	}
	if (st == GG_OK)
	{
		unsigned long long herr = 0;

		GG_HIP(hipStreamSynchronize(e.stream));
		GG_HIP(hipMemcpy(&herr, d_err, 8, hipMemcpyDeviceToHost));
		if (herr)
			st = fail(GG_EINVAL, "memtuple decode error 0x%llx",
				  herr);
		else
			for (int i = 0; i < natts; i++)
			{
				if (b.is_var[i])
				{
					gg_text_out *to =
						(gg_text_out *) cols[i];

					GG_HIP(hipMemcpy(to->offs,
							 d_voffs[i],
							 (size_t) nrows * 8,
							 hipMemcpyDeviceToHost));
					GG_HIP(hipMemcpy(to->lens,
							 d_vlens[i],
							 (size_t) nrows * 4,
							 hipMemcpyDeviceToHost));
				}
				else
					GG_HIP(hipMemcpy(cols[i], d_cols[i],
							 (size_t) nrows *
							 attlen[i],
							 hipMemcpyDeviceToHost));
				if (nulls && nulls[i])
					GG_HIP(hipMemcpy(nulls[i],
							 d_nulls[i],
							 (size_t) nrows,
							 hipMemcpyDeviceToHost));
			}
	}
	(void) hipFree(d_in);
	(void) hipFree(d_offs);
	(void) hipFree(d_err);
	for (int i = 0; i < natts; i++)
	{
		(void) hipFree(d_cols[i]);
		(void) hipFree(d_voffs[i]);
		(void) hipFree(d_vlens[i]);
		(void) hipFree(d_nulls[i]);
	}
	(void) hipFree(d_colp);
	(void) hipFree(d_voffp);
	(void) hipFree(d_vlenp);
	(void) hipFree(d_nullp);
	return st;
}

/* ---------- Motion wire chunk framing (ABI surface, §8(f)3) ---------- */

/*
 * The interconnect carries tuples as 4-byte-headed chunks
 * ([u16 payload size excl header][u16 TupleChunkType], native endian —
 * tupchunk.h:34-49,80-84): a MemTuple is sent TC_WHOLE padded to
 * TUPLE_CHUNK_ALIGN(4), or split into TC_PARTIAL_START/MID/END when it
 * exceeds the max chunk length (SerializeTuple, tupser.c:400-612;
 * splitting rule addByteStringToChunkList :230).  TC_END_OF_STREAM
 * closes a sender's stream.  These two host-side helpers convert
 * between a MemTuple stream (gg_engine_memtuple_encode's output) and
 * the exact byte stream a CPU segment's Motion sender/receiver uses.
 */
#define GG_TC_WHOLE 0
#define GG_TC_PARTIAL_START 1
#define GG_TC_PARTIAL_MID 2
#define GG_TC_PARTIAL_END 3
#define GG_TC_END_OF_STREAM 4

extern "C" gg_status
gg_engine_motion_chunkify(const uint8_t *tuples, int64_t tuples_len,
			  int32_t max_chunk, int append_eos, uint8_t *out,
			  int64_t cap, int64_t *out_len)
{
	if (!tuples || tuples_len < 0 || !out || !out_len ||
	    max_chunk < 16 || max_chunk > 0xFFFF)
		return fail(GG_EINVAL, "bad chunkify args");

	int64_t pos = 0, opos = 0;

	auto emit = [&](uint16_t type, const uint8_t *data, int32_t n,
			int32_t pad) -> bool
	{
		if (opos + 4 + n + pad > cap)
			return false;
		uint16_t sz = (uint16_t) (n + pad);

		std::memcpy(out + opos, &sz, 2);
		std::memcpy(out + opos + 2, &type, 2);
		if (n)
			std::memcpy(out + opos + 4, data, (size_t) n);
		if (pad)
			std::memset(out + opos + 4 + n, 0, (size_t) pad);
		opos += 4 + n + pad;
		return true;
	};

	while (pos < tuples_len)
	{
		uint32_t hdr;

		if (pos + 8 > tuples_len)
			return fail(GG_EINVAL, "truncated memtuple at %lld",
				    (long long) pos);
		std::memcpy(&hdr, tuples + pos, 4);
		if (!(hdr & 0x80000000u))
			return fail(GG_EINVAL, "lead bit clear at %lld",
				    (long long) pos);
		int64_t tlen = (int64_t) (hdr & 0x3FFFFFF8u);

		if (tlen < 8 || pos + tlen > tuples_len)
			return fail(GG_EINVAL, "bad memtuple len at %lld",
				    (long long) pos);
		/* memtuple sizes are 8-aligned, so the 4-align pad is 0 */
		int32_t payload_max = max_chunk - 4;

		if (tlen <= payload_max)
		{
			if (!emit(GG_TC_WHOLE, tuples + pos,
				  (int32_t) tlen, 0))
				return fail(GG_EINVAL, "chunk cap");
		}
		else
		{
			int64_t off = 0;
			int nchunk = 0;

			while (off < tlen)
			{
				int32_t n = (int32_t)
					((tlen - off < payload_max)
					 ? tlen - off : payload_max);
				uint16_t ty = nchunk == 0
					? GG_TC_PARTIAL_START
					: (off + n == tlen
					   ? GG_TC_PARTIAL_END
					   : GG_TC_PARTIAL_MID);

				if (!emit(ty, tuples + pos + off, n, 0))
					return fail(GG_EINVAL, "chunk cap");
				off += n;
				nchunk++;
			}
		}
		pos += tlen;
	}
	if (append_eos && !emit(GG_TC_END_OF_STREAM, nullptr, 0, 0))
		return fail(GG_EINVAL, "chunk cap");
	*out_len = opos;
	return GG_OK;
}

extern "C" gg_status
gg_engine_motion_dechunkify(const uint8_t *chunks, int64_t chunks_len,
			    uint8_t *out, int64_t cap, int64_t *out_len,
			    int *saw_eos)
{
	if (!chunks || chunks_len < 0 || !out || !out_len || !saw_eos)
		return fail(GG_EINVAL, "bad dechunkify args");

	int64_t pos = 0, opos = 0;
	bool in_partial = false;

	*saw_eos = 0;
	while (pos < chunks_len)
	{
		uint16_t sz, ty;

		if (pos + 4 > chunks_len)
			return fail(GG_EINVAL, "truncated chunk at %lld",
				    (long long) pos);
		std::memcpy(&sz, chunks + pos, 2);
		std::memcpy(&ty, chunks + pos + 2, 2);
		if (pos + 4 + sz > chunks_len)
			return fail(GG_EINVAL, "chunk overruns at %lld",
				    (long long) pos);
		switch (ty)
		{
			case GG_TC_WHOLE:
				if (in_partial)
					return fail(GG_EINVAL,
						    "WHOLE inside partial "
						    "at %lld",
						    (long long) pos);
				break;
			case GG_TC_PARTIAL_START:
				if (in_partial)
					return fail(GG_EINVAL,
						    "nested PARTIAL_START "
						    "at %lld",
						    (long long) pos);
				in_partial = true;
				break;
			case GG_TC_PARTIAL_MID:
			case GG_TC_PARTIAL_END:
				if (!in_partial)
					return fail(GG_EINVAL,
						    "PARTIAL_%s without "
						    "START at %lld",
						    ty == GG_TC_PARTIAL_END
						    ? "END" : "MID",
						    (long long) pos);
				if (ty == GG_TC_PARTIAL_END)
					in_partial = false;
				break;
			case GG_TC_END_OF_STREAM:
				*saw_eos = 1;
				pos += 4 + sz;
				continue;
			default:
				return fail(GG_EINVAL,
					    "bad chunk type %d at %lld",
					    (int) ty, (long long) pos);
		}
		if (opos + sz > cap)
			return fail(GG_EINVAL, "dechunk cap");
		std::memcpy(out + opos, chunks + pos + 4, sz);
		opos += sz;
		pos += 4 + sz;
	}
	if (in_partial)
		return fail(GG_EINVAL, "partial tuple unterminated");
	*out_len = opos;
	return GG_OK;
}

/* ---------------- general hash group-by (ABI surface) ---------------- */

/*
 * Spill-tier hash JOIN (the nodeHash.c:713 batching
 * side): both inputs are hash-range partitioned on the GPU in chunks
 * and staged in host memory (the reference writes batch files), then
 * each partition pair is built+probed on the GPU.  Build keys must be
 * unique (PK-side build, as in the engine's pipeline joins); output
 * is (probe_row_index, build_val) for every probe row with a match.
 */
/* pipelined hash-range partition stage shared by the spill tiers:
 * pinned double-buffered staging both directions, cursor prefix on
 * device (k_gb_prefix2_u64 — no mid-chunk host sync), one bulk D2H
 * per chunk, host distribution overlapped with the other buffer's
 * GPU work.  vals == nullptr selects the index side (out_v receives
 * GLOBAL row indices via the scatter_idx kernel).  All staging comes
 * from the engine pools. */
static gg_status
spill_stage_partitions(Engine &e, const int64_t *keys, const int64_t *vals,
		       int64_t n, int64_t budget_bytes, uint64_t P,
		       int shift,
		       std::vector<std::vector<int64_t>> &out_k,
		       std::vector<std::vector<int64_t>> &out_v)
{
	int64_t chunk = budget_bytes / (8 * 4 * 2);
	gg_status st = GG_OK;

	if (chunk < 1024)
		chunk = 1024;
	if (chunk > n)
		chunk = n;
	if (n <= 0)
		return GG_OK;

	struct GbsBuf
	{
		int64_t *dk = nullptr, *dv = nullptr;
		int64_t *sk = nullptr, *sv = nullptr;
		unsigned long long *dcnt = nullptr;	/* [P cnt][P work][P+1 pristine] */
		int64_t *hk = nullptr, *hv = nullptr;	/* pinned in */
		int64_t *ok = nullptr, *ov = nullptr;	/* pinned out */
		unsigned long long *hcur = nullptr;	/* pinned P+1 */
		hipStream_t stm = nullptr;
		hipEvent_t done = nullptr;
		bool busy = false;
	} B[2];
	bool alloc_ok = true;

	for (int b = 0; b < 2 && alloc_ok; b++)
	{
		GbsBuf &q = B[b];
		char nm[32];

#define GBS_POOL(fld, kind, sz) (std::snprintf(nm, sizeof(nm), "gbs.%s%d", #fld, b), (q.fld = (decltype(q.fld)) e.kind(nm, (sz))) != nullptr)
		alloc_ok =
			GBS_POOL(dk, esget, (size_t) chunk * 8) &&
			GBS_POOL(dv, esget, (size_t) chunk * 8) &&
			GBS_POOL(sk, esget, (size_t) chunk * 8) &&
			GBS_POOL(sv, esget, (size_t) chunk * 8) &&
			GBS_POOL(dcnt, esget, (3 * P + 1) * 8) &&
			GBS_POOL(hk, ehget, (size_t) chunk * 8) &&
			GBS_POOL(hv, ehget, (size_t) chunk * 8) &&
			GBS_POOL(ok, ehget, (size_t) chunk * 8) &&
			GBS_POOL(ov, ehget, (size_t) chunk * 8) &&
			GBS_POOL(hcur, ehget, (P + 1) * 8) &&
			hipStreamCreate(&q.stm) == hipSuccess &&
			hipEventCreate(&q.done) == hipSuccess;
#undef GBS_POOL
	}
	if (!alloc_ok)
		st = fail(GG_ENOMEM, "spill staging");

	/* error policy: record first failure, stop issuing, drain */
#define GG_HIP_GBS(x) \
	{ hipError_t e_ = (x); \
	  if (st == GG_OK && e_ != hipSuccess) \
		st = fail(GG_EGPU, "spill stage: %s", \
			  hipGetErrorString(e_)); }

	auto drain = [&](GbsBuf &q)
	{
		if (!q.busy)
			return;
		GG_HIP_GBS(hipEventSynchronize(q.done));
		q.busy = false;
		if (st != GG_OK)
			return;
		/* append staged rows to their partitions; resizes are
		 * serial, the copies partition-parallel */
		std::vector<size_t> old(P);

		for (uint64_t p2 = 0; p2 < P; p2++)
		{
			size_t c = (size_t) (q.hcur[p2 + 1] - q.hcur[p2]);

			old[p2] = out_k[p2].size();
			if (c)
			{
				size_t need = old[p2] + c;

				/* resize() grows to EXACTLY the request
				 * in libstdc++ — per-chunk exact growth
				 * re-copied every partition every chunk
				 * (hundreds of GB of hidden memcpy at
				 * 1B rows, measured 7.8s).  Reserve the
				 * uniform-hash expectation up front and
				 * grow geometrically past it. */
				if (out_k[p2].capacity() < need)
				{
					size_t want = std::max(
						need,
						std::max(out_k[p2]
							 .capacity() * 2,
							 (size_t) (n / P +
								   n / P / 8 +
								   1024)));

					out_k[p2].reserve(want);
					out_v[p2].reserve(want);
				}
				out_k[p2].resize(need);
				out_v[p2].resize(need);
			}
		}
		par_for_grain((int64_t) P, 1, [&](int64_t p2)
		{
			size_t c = (size_t) (q.hcur[p2 + 1] - q.hcur[p2]);

			if (!c)
				return;
			std::memcpy(out_k[p2].data() + old[p2],
				    q.ok + q.hcur[p2], c * 8);
			std::memcpy(out_v[p2].data() + old[p2],
				    q.ov + q.hcur[p2], c * 8);
		});
	};

	for (int64_t base = 0, it = 0; st == GG_OK && base < n;
	     base += chunk, it ^= 1)
	{
		GbsBuf &q = B[it];
		int64_t m = (n - base < chunk) ? n - base : chunk;

		drain(q);	/* reclaim this buffer's previous chunk */
		if (st != GG_OK)
			break;
		/* pageable -> pinned bounce (threaded memcpy;
		 * hipMemcpyAsync from pageable memory would serialize
		 * in the driver's staging path) */
		{
			int64_t nblk = (m + (1 << 17) - 1) >> 17;

			par_for_grain(nblk, 1, [&](int64_t blk)
			{
				int64_t o = blk << 17;
				int64_t c = (m - o < (1 << 17))
					? m - o : (1 << 17);

				std::memcpy(q.hk + o, keys + base + o,
					    c * 8);
				if (vals)
					std::memcpy(q.hv + o,
						    vals + base + o, c * 8);
			});
		}
		GG_HIP_GBS(hipMemcpyAsync(q.dk, q.hk, (size_t) m * 8,
					  hipMemcpyHostToDevice, q.stm));
		if (vals)
			GG_HIP_GBS(hipMemcpyAsync(q.dv, q.hv,
						  (size_t) m * 8,
						  hipMemcpyHostToDevice,
						  q.stm));
		GG_HIP_GBS(hipMemsetAsync(q.dcnt, 0, P * 8, q.stm));
		GG_HIP_GBS(launch_gb_part_count(q.stm, q.dk, m, shift,
						q.dcnt));
		GG_HIP_GBS(launch_gb_prefix2_u64(q.stm, q.dcnt, (int) P,
						 q.dcnt + P,
						 q.dcnt + 2 * P));
		if (vals)
		{
			GG_HIP_GBS(launch_gb_part_scatter(
				q.stm, q.dk, q.dv, m, shift, q.dcnt + P,
				q.sk, q.sv));
		}
		else
		{
			GG_HIP_GBS(launch_gb_part_scatter_idx(
				q.stm, q.dk, m, base, shift, q.dcnt + P,
				q.sk, q.sv));
		}
		GG_HIP_GBS(hipMemcpyAsync(q.ok, q.sk, (size_t) m * 8,
					  hipMemcpyDeviceToHost, q.stm));
		GG_HIP_GBS(hipMemcpyAsync(q.ov, q.sv, (size_t) m * 8,
					  hipMemcpyDeviceToHost, q.stm));
		GG_HIP_GBS(hipMemcpyAsync(q.hcur, q.dcnt + 2 * P,
					  (P + 1) * 8,
					  hipMemcpyDeviceToHost, q.stm));
		GG_HIP_GBS(hipEventRecord(q.done, q.stm));
		q.busy = true;
	}
	drain(B[0]);
	drain(B[1]);
#undef GG_HIP_GBS
	for (int b = 0; b < 2; b++)
	{
		GbsBuf &q = B[b];

		if (q.stm)
			(void) hipStreamSynchronize(q.stm);
		if (q.done)
			(void) hipEventDestroy(q.done);
		if (q.stm)
			(void) hipStreamDestroy(q.stm);
	}
	return st;
}

extern "C" gg_status
gg_engine_hash_join_i64_spill(const int64_t *build_keys,
			      const int64_t *build_vals, int64_t nb,
			      const int64_t *probe_keys, int64_t np,
			      int64_t budget_bytes, int64_t *out_probe_idx,
			      int64_t *out_vals, int64_t cap,
			      int64_t *out_nmatch, int32_t *out_npartitions)
{
	Engine &e = engine();

	if (!e.inited)
		return fail(GG_ESTATE, "engine not initialized");
	if (!build_keys || !build_vals || nb < 0 || !probe_keys || np < 0 ||
	    !out_probe_idx || !out_vals || !out_nmatch || !out_npartitions ||
	    budget_bytes < (1 << 20))
		return fail(GG_EINVAL, "bad join_spill args");
	*out_nmatch = 0;

	/* footprint ≈ build pairs + 2x table + probe pairs + outputs */
	int64_t need = nb * 8 * 6 + np * 8 * 4;
	uint64_t P = 1;
	int shift = 32;

	if (need > budget_bytes)
	{
		P = pow2_ceil((uint64_t) (need / budget_bytes) + 1);
		if (P > 4096)
			return fail(GG_EINVAL,
				    "join needs %llu partitions (>4096)",
				    (unsigned long long) P);
		for (uint64_t t = P; t > 1; t >>= 1)
			shift--;
	}
	*out_npartitions = (int32_t) P;

	/* stage both sides into host partitions (P==1: single "partition"
	 * passthrough still goes through the same code path), on the
	 * shared pipelined stage */
	std::vector<std::vector<int64_t>> bk(P), bv(P), pk(P), pi(P);
	gg_status st = spill_stage_partitions(e, build_keys, build_vals,
					      nb, budget_bytes, P, shift,
					      bk, bv);

	if (st == GG_OK)
		st = spill_stage_partitions(e, probe_keys, nullptr, np,
					    budget_bytes, P, shift, pk, pi);
	if (st != GG_OK)
		return st;

	/* per-partition build + probe */
	int64_t total = 0;

	for (uint64_t p2 = 0; p2 < P && st == GG_OK; p2++)
	{
		int64_t b_n = (int64_t) bk[p2].size();
		int64_t p_n = (int64_t) pk[p2].size();

		if (b_n == 0 || p_n == 0)
			continue;
		uint64_t nslots = next_pow2(2 * (uint64_t) b_n);
		/* engine pools: no per-partition hipMalloc/free; pinned
		 * bounce for the uploads (pageable hipMemcpy was the
		 * serial cost here, like the group-by reload) */
		int64_t *d_bk = (int64_t *) e.esget("sj.bk", (size_t) b_n * 8);
		int64_t *d_bv = (int64_t *) e.esget("sj.bv", (size_t) b_n * 8);
		int64_t *d_pk = (int64_t *) e.esget("sj.pk", (size_t) p_n * 8);
		int64_t *d_pi = (int64_t *) e.esget("sj.pi", (size_t) p_n * 8);
		int64_t *d_oi = (int64_t *) e.esget("sj.oi", (size_t) p_n * 8);
		int64_t *d_ov = (int64_t *) e.esget("sj.ov", (size_t) p_n * 8);
		unsigned long long *tk = (unsigned long long *)
			e.esget("sj.tk", nslots * 8);
		unsigned long long *tv = (unsigned long long *)
			e.esget("sj.tv", nslots * 8);
		unsigned long long *ctr = (unsigned long long *)
			e.esget("sj.ctr", 8);
		int64_t *hb = (int64_t *)
			e.ehget("sj.hb", (size_t) (b_n > p_n ? b_n : p_n) * 8);

		if (!d_bk || !d_bv || !d_pk || !d_pi || !d_oi || !d_ov ||
		    !tk || !tv || !ctr || !hb)
			st = fail(GG_ENOMEM, "join_spill buffers");
#define GG_HIP_BRK(x) \
		{ hipError_t e_ = (x); \
		  if (st == GG_OK && e_ != hipSuccess) \
			st = fail(GG_EGPU, "join_spill: %s", \
				  hipGetErrorString(e_)); }
		auto up = [&](int64_t *dst, const int64_t *src, int64_t nn)
		{
			if (st != GG_OK)
				return;
			int64_t nblk = (nn + (1 << 17) - 1) >> 17;

			par_for_grain(nblk, 1, [&](int64_t blk)
			{
				int64_t o = blk << 17;
				int64_t c = (nn - o < (1 << 17))
					? nn - o : (1 << 17);

				std::memcpy(hb + o, src + o, c * 8);
			});
			GG_HIP_BRK(hipMemcpyAsync(dst, hb, (size_t) nn * 8,
						  hipMemcpyHostToDevice,
						  e.stream));
			GG_HIP_BRK(hipStreamSynchronize(e.stream));
		};
		if (st == GG_OK)
		{
			up(d_bk, bk[p2].data(), b_n);
			up(d_bv, bv[p2].data(), b_n);
			up(d_pk, pk[p2].data(), p_n);
			up(d_pi, pi[p2].data(), p_n);
			GG_HIP_BRK(hipMemsetAsync(tk, 0, nslots * 8,
						  e.stream));
			GG_HIP_BRK(hipMemsetAsync(ctr, 0, 8, e.stream));
			GG_HIP_BRK(launch_sj_build(e.stream, d_bk, d_bv,
						   b_n, tk, tv, nslots));
			GG_HIP_BRK(launch_sj_probe(e.stream, d_pk, d_pi,
						   p_n, tk, tv, nslots,
						   d_oi, d_ov, ctr));
			GG_HIP_BRK(hipStreamSynchronize(e.stream));
		}

		unsigned long long nm = 0;

		if (st == GG_OK)
			GG_HIP_BRK(hipMemcpy(&nm, ctr, 8,
					     hipMemcpyDeviceToHost));
		if (st == GG_OK && total + (int64_t) nm > cap)
			st = fail(GG_EINVAL, "join cap %lld < %lld",
				  (long long) cap,
				  (long long) (total + (int64_t) nm));
		else if (st == GG_OK && nm)
		{
			GG_HIP_BRK(hipMemcpy(out_probe_idx + total, d_oi,
					     nm * 8,
					     hipMemcpyDeviceToHost));
			GG_HIP_BRK(hipMemcpy(out_vals + total, d_ov,
					     nm * 8,
					     hipMemcpyDeviceToHost));
			total += (int64_t) nm;
		}
#undef GG_HIP_BRK
		bk[p2].clear(); bk[p2].shrink_to_fit();
		bv[p2].clear(); bv[p2].shrink_to_fit();
		pk[p2].clear(); pk[p2].shrink_to_fit();
		pi[p2].clear(); pi[p2].shrink_to_fit();
	}
	if (st != GG_OK)
		return st;
	*out_nmatch = total;
	return GG_OK;
}

/*
 * Spill-tier hash group-by: when the input exceeds the
 * device budget, hash-range-partition it on the GPU in chunks, stage
 * the partitions in HOST memory (the spill arena — the reference
 * writes hash-range partitions to spill files, spill_hash_table
 * execHHashagg.c:1350), then reload and aggregate one partition at a
 * time (agg_hash_reload :1852).  Partitions are disjoint by key, so
 * the result is the concatenation of per-partition results.
 * Partition id = top hash bits; the group table uses the low bits, so
 * reloaded partitions still hash uniformly.
 */
extern "C" gg_status
gg_engine_hash_groupby_i64_spill(const int64_t *keys, const int64_t *vals,
				 int64_t n, int64_t budget_bytes,
				 int64_t *out_keys, int64_t *out_sums,
				 int64_t *out_counts, int64_t cap,
				 int64_t *out_ngroups,
				 int32_t *out_npartitions)
{
	Engine &e = engine();

	if (!e.inited)
		return fail(GG_ESTATE, "engine not initialized");
	if (!keys || !vals || n < 0 || !out_keys || !out_sums ||
	    !out_counts || !out_ngroups || !out_npartitions ||
	    budget_bytes < (1 << 20))
		return fail(GG_EINVAL, "bad groupby_spill args");

	/* in-memory fast path: input pairs + 2x table + compact buffers
	 * ≈ 7 x n x 8 bytes */
	if (n * 8 * 7 <= budget_bytes)
	{
		*out_npartitions = 1;
		return gg_engine_hash_groupby_i64(keys, vals, n, out_keys,
						  out_sums, out_counts, cap,
						  out_ngroups);
	}

	/* number of hash-range partitions: each must fit the budget */
	uint64_t P = pow2_ceil((uint64_t) ((n * 8 * 7) / budget_bytes) + 1);

	if (P > 4096)
		return fail(GG_EINVAL, "input needs %llu partitions (>4096)",
			    (unsigned long long) P);
	int shift = 32;

	for (uint64_t t = P; t > 1; t >>= 1)
		shift--;

	/* partition pass: the shared pipelined stage (see
	 * spill_stage_partitions above for the measured story) */
	std::vector<std::vector<int64_t>> part_k(P), part_v(P);
	const bool spdbg = getenv("GG_SPILL_DEBUG") != nullptr;
	auto spt0 = std::chrono::steady_clock::now();
	auto spmark = [&](const char *what)
	{
		if (!spdbg)
			return;
		auto t1 = std::chrono::steady_clock::now();

		fprintf(stderr, "[spill] %s: %.2fs\n", what,
			std::chrono::duration<double>(t1 - spt0).count());
		spt0 = t1;
	};
	gg_status st = spill_stage_partitions(e, keys, vals, n,
					      budget_bytes, P, shift,
					      part_k, part_v);

	spmark("partition pass");
	if (st != GG_OK)
		return st;

	/* reload + aggregate, pipelined like the partition pass: pinned
	 * double-buffered uploads; ONE shared hash table (partitions
	 * serialize on it via an inter-stream event — H2D of partition
	 * p+1 overlaps the kernels of p, which is where the time was);
	 * the compact kernel writes its (few) result rows directly to
	 * pinned host memory; each partition's run is host-sorted
	 * while the next runs on the GPU, and the sorted runs k-way
	 * merge at the end (the old single global std::sort of every
	 * group was serial tail time). */
	int64_t ng_total = 0;
	int64_t maxm = 0;

	for (uint64_t p2 = 0; p2 < P; p2++)
		maxm = std::max(maxm, (int64_t) part_k[p2].size());
	if (maxm == 0)
	{
		*out_ngroups = 0;
		*out_npartitions = (int32_t) P;
		return GG_OK;
	}
	{
		uint64_t rslots = next_pow2(2 * (uint64_t) maxm);
		unsigned long long *tk = nullptr, *ts = nullptr,
			*tc = nullptr;
		hipEvent_t tbl_free = nullptr;
		struct RlBuf
		{
			int64_t *dk = nullptr, *dv = nullptr;
			int64_t *hk = nullptr, *hv = nullptr;	/* pinned in */
			int64_t *pok = nullptr, *pos = nullptr,
				*poc = nullptr;			/* pinned out */
			unsigned long long *dctr = nullptr;
			unsigned long long *hng = nullptr;	/* pinned */
			hipStream_t stm = nullptr;
			hipEvent_t done = nullptr;
			bool busy = false;
		} R[2];
		std::vector<std::pair<int64_t, int64_t>> runs;	/* off,len */
		bool aok =
			(tk = (unsigned long long *)
			 e.esget("rl.tk", rslots * 8)) != nullptr &&
			(ts = (unsigned long long *)
			 e.esget("rl.ts", rslots * 8)) != nullptr &&
			(tc = (unsigned long long *)
			 e.esget("rl.tc", rslots * 8)) != nullptr &&
			hipEventCreate(&tbl_free) == hipSuccess;

		for (int b = 0; b < 2 && aok; b++)
		{
			RlBuf &q = R[b];
			char nm[32];

#define RL_POOL(fld, kind, sz) 			(std::snprintf(nm, sizeof(nm), "rl.%s%d", #fld, b), 			 (q.fld = (decltype(q.fld)) e.kind(nm, (sz))) != nullptr)
			aok =
				RL_POOL(dk, esget, (size_t) maxm * 8) &&
				RL_POOL(dv, esget, (size_t) maxm * 8) &&
				RL_POOL(dctr, esget, 8) &&
				RL_POOL(hk, ehget, (size_t) maxm * 8) &&
				RL_POOL(hv, ehget, (size_t) maxm * 8) &&
				RL_POOL(pok, ehget, (size_t) maxm * 8) &&
				RL_POOL(pos, ehget, (size_t) maxm * 8) &&
				RL_POOL(poc, ehget, (size_t) maxm * 8) &&
				RL_POOL(hng, ehget, 8) &&
				hipStreamCreate(&q.stm) == hipSuccess &&
				hipEventCreate(&q.done) == hipSuccess;
#undef RL_POOL
		}
		if (!aok && st == GG_OK)
			st = fail(GG_ENOMEM, "groupby_spill reload staging");

#define GG_HIP_GBR(x) \
	{ hipError_t e_ = (x); \
	  if (st == GG_OK && e_ != hipSuccess) \
		st = fail(GG_EGPU, "groupby_spill reload: %s", \
			  hipGetErrorString(e_)); }

		auto rl_finish = [&](RlBuf &q)
		{
			if (!q.busy)
				return;
			GG_HIP_GBR(hipEventSynchronize(q.done));
			q.busy = false;
			if (st != GG_OK)
				return;
			int64_t ng = (int64_t) *q.hng;

			if (ng_total + ng > cap)
			{
				st = fail(GG_EINVAL,
					  "groupby cap %lld exceeded",
					  (long long) cap);
				return;
			}
			/* sort this run by key while the other buffer's
			 * partition executes on the GPU */
			std::vector<size_t> idx((size_t) ng);

			for (int64_t i = 0; i < ng; i++)
				idx[i] = (size_t) i;
			std::sort(idx.begin(), idx.end(),
				  [&](size_t a2, size_t b2)
				  { return q.pok[a2] < q.pok[b2]; });
			for (int64_t i = 0; i < ng; i++)
			{
				out_keys[ng_total + i] = q.pok[idx[i]];
				out_sums[ng_total + i] = q.pos[idx[i]];
				out_counts[ng_total + i] = q.poc[idx[i]];
			}
			runs.push_back({ng_total, ng});
			ng_total += ng;
		};

		for (uint64_t p2 = 0, it = 0; st == GG_OK && p2 < P; p2++)
		{
			if (part_k[p2].empty())
				continue;
			RlBuf &q = R[it & 1];
			int64_t m = (int64_t) part_k[p2].size();

			it++;
			rl_finish(q);
			if (st != GG_OK)
				break;
			{
				int64_t nblk = (m + (1 << 17) - 1) >> 17;

				par_for_grain(nblk, 1, [&](int64_t blk)
				{
					int64_t o = blk << 17;
					int64_t c = (m - o < (1 << 17))
						? m - o : (1 << 17);

					std::memcpy(q.hk + o,
						    part_k[p2].data() + o,
						    c * 8);
					std::memcpy(q.hv + o,
						    part_v[p2].data() + o,
						    c * 8);
				});
			}
			part_k[p2].clear();
			part_k[p2].shrink_to_fit();
			part_v[p2].clear();
			part_v[p2].shrink_to_fit();
			GG_HIP_GBR(hipMemcpyAsync(q.dk, q.hk,
						  (size_t) m * 8,
						  hipMemcpyHostToDevice,
						  q.stm));
			GG_HIP_GBR(hipMemcpyAsync(q.dv, q.hv,
						  (size_t) m * 8,
						  hipMemcpyHostToDevice,
						  q.stm));
			/* the shared table frees when the PREVIOUS
			 * partition's compact is done */
			if (it > 1)
				GG_HIP_GBR(hipStreamWaitEvent(q.stm,
							      tbl_free, 0));
			GG_HIP_GBR(launch_fill_u64(q.stm, tk, rslots,
						   0x8000000000000000ull));
			GG_HIP_GBR(hipMemsetAsync(ts, 0, rslots * 8,
						  q.stm));
			GG_HIP_GBR(hipMemsetAsync(tc, 0, rslots * 8,
						  q.stm));
			GG_HIP_GBR(hipMemsetAsync(q.dctr, 0, 8, q.stm));
			GG_HIP_GBR(launch_groupby_build(q.stm, q.dk, q.dv,
							m, tk, ts, tc,
							rslots));
			GG_HIP_GBR(launch_groupby_compact(
				q.stm, tk, ts, tc, rslots, q.pok, q.pos,
				q.poc, q.dctr, (uint64_t) maxm));
			GG_HIP_GBR(hipEventRecord(tbl_free, q.stm));
			GG_HIP_GBR(hipMemcpyAsync(q.hng, q.dctr, 8,
						  hipMemcpyDeviceToHost,
						  q.stm));
			GG_HIP_GBR(hipEventRecord(q.done, q.stm));
			q.busy = true;
		}
		rl_finish(R[0]);
		rl_finish(R[1]);
		spmark("reload");
#undef GG_HIP_GBR
		for (int b = 0; b < 2; b++)
		{
			RlBuf &q = R[b];

			if (q.stm)
				(void) hipStreamSynchronize(q.stm);
			if (q.done)
				(void) hipEventDestroy(q.done);
			if (q.stm)
				(void) hipStreamDestroy(q.stm);
		}
		if (tbl_free)
			(void) hipEventDestroy(tbl_free);
		if (st != GG_OK)
			return st;

		/* k-way merge of the sorted runs into global key order
		 * (hash-range partitions interleave in keyspace) */
		if (runs.size() > 1)
		{
			std::vector<int64_t> mk(out_keys,
						out_keys + ng_total);
			std::vector<int64_t> ms(out_sums,
						out_sums + ng_total);
			std::vector<int64_t> mc(out_counts,
						out_counts + ng_total);
			/* min-heap of (key, run); pop -> emit -> push next */
			std::vector<std::pair<int64_t, size_t>> heap;
			std::vector<int64_t> pos(runs.size());

			auto cmp = [](const std::pair<int64_t, size_t> &a2,
				      const std::pair<int64_t, size_t> &b2)
			{ return a2.first > b2.first; };
			for (size_t r = 0; r < runs.size(); r++)
			{
				pos[r] = runs[r].first;
				if (runs[r].second)
					heap.push_back({mk[pos[r]], r});
			}
			std::make_heap(heap.begin(), heap.end(), cmp);
			for (int64_t o = 0; !heap.empty(); o++)
			{
				std::pop_heap(heap.begin(), heap.end(),
					      cmp);
				size_t r = heap.back().second;

				heap.pop_back();
				out_keys[o] = mk[pos[r]];
				out_sums[o] = ms[pos[r]];
				out_counts[o] = mc[pos[r]];
				pos[r]++;
				if (pos[r] <
				    runs[r].first + runs[r].second)
				{
					heap.push_back({mk[pos[r]], r});
					std::push_heap(heap.begin(),
						       heap.end(), cmp);
				}
			}
		}
	}
	spmark("merge");
	*out_ngroups = ng_total;
	*out_npartitions = (int32_t) P;
	return GG_OK;
}

extern "C" gg_status
gg_engine_hash_groupby_i64(const int64_t *keys, const int64_t *vals,
			   int64_t n, int64_t *out_keys, int64_t *out_sums,
			   int64_t *out_counts, int64_t cap,
			   int64_t *out_ngroups)
{
	Engine &e = engine();

	if (!e.inited)
		return fail(GG_ESTATE, "engine not initialized");
	if (!keys || !vals || n < 0 || !out_keys || !out_sums ||
	    !out_counts || !out_ngroups)
		return fail(GG_EINVAL, "bad groupby args");
	*out_ngroups = 0;
	if (n == 0)
		return GG_OK;

	uint64_t nslots = next_pow2(2 * (uint64_t) n);
	/* engine scratch pool: repeated calls at these sizes re-map
	 * ~6x n x 8 + 3 x nslots x 8 bytes of HBM per call otherwise
	 * (tens of GB of hipMalloc/hipFree at 1B rows — measured as
	 * the dominant in-memory cost, not the copies) */
	int64_t *dk = (int64_t *) e.esget("gb.dk", (size_t) n * 8);
	int64_t *dv = (int64_t *) e.esget("gb.dv", (size_t) n * 8);
	unsigned long long *tk = (unsigned long long *)
		e.esget("gb.tk", nslots * 8);
	unsigned long long *ts = (unsigned long long *)
		e.esget("gb.ts", nslots * 8);
	unsigned long long *tc = (unsigned long long *)
		e.esget("gb.tc", nslots * 8);
	int64_t *ok = (int64_t *) e.esget("gb.ok", (size_t) n * 8);
	int64_t *os = (int64_t *) e.esget("gb.os", (size_t) n * 8);
	int64_t *oc = (int64_t *) e.esget("gb.oc", (size_t) n * 8);
	unsigned long long *ctr = (unsigned long long *)
		e.esget("gb.ctr", 8);

	if (!dk || !dv || !tk || !ts || !tc || !ok || !os || !oc || !ctr)
		return fail(GG_ENOMEM, "groupby buffers");
	/* the empty sentinel (INT64_MIN) is not a memset byte pattern —
	 * fill with a kernel */
	GG_HIP(launch_fill_u64(e.stream, tk, nslots, 0x8000000000000000ull));
	GG_HIP(hipMemsetAsync(ts, 0, nslots * 8, e.stream));
	GG_HIP(hipMemsetAsync(tc, 0, nslots * 8, e.stream));
	GG_HIP(hipMemsetAsync(ctr, 0, 8, e.stream));
	/* chunked pinned upload overlapped with the build: the build
	 * kernel is per-row independent (CAS find-or-create + atomic
	 * transitions), so it runs on each chunk as it lands — a
	 * single pageable hipMemcpy of the whole input was measured
	 * bound at ~6 GB/s and serialized ahead of the build */
	{
		const int64_t ch = 1 << 24;	/* 16M rows = 128 MB */
		int64_t *bk_ = nullptr, *bv_ = nullptr;
		gg_status st2 = GG_OK;

		if (n > (1 << 22) &&
		    (bk_ = (int64_t *) e.ehget("gb.hbk", 2 * ch * 8))
		    != nullptr &&
		    (bv_ = (int64_t *) e.ehget("gb.hbv", 2 * ch * 8))
		    != nullptr)
		{
			hipEvent_t up[2] = {};
			hipStream_t cstream = nullptr;

			(void) hipEventCreate(&up[0]);
			(void) hipEventCreate(&up[1]);
			(void) hipStreamCreate(&cstream);
			for (int64_t base = 0, it = 0; base < n;
			     base += ch, it ^= 1)
			{
				int64_t m = (n - base < ch) ? n - base : ch;
				int64_t *pk_ = bk_ + it * ch;
				int64_t *pv_ = bv_ + it * ch;

				/* buffer free once its H2D completed */
				if (base >= 2 * ch)
					(void) hipEventSynchronize(up[it]);
				{
					int64_t nblk =
						(m + (1 << 17) - 1) >> 17;

					par_for_grain(nblk, 1,
						      [&](int64_t blk)
					{
						int64_t o = blk << 17;
						int64_t c =
							(m - o < (1 << 17))
							? m - o : (1 << 17);

						std::memcpy(pk_ + o,
							    keys + base + o,
							    c * 8);
						std::memcpy(pv_ + o,
							    vals + base + o,
							    c * 8);
					});
				}
				/* copies on their own stream: chunk
				 * i+1's upload overlaps chunk i's
				 * build kernel */
				hipError_t e1 = hipMemcpyAsync(
					dk + base, pk_, (size_t) m * 8,
					hipMemcpyHostToDevice, cstream);
				hipError_t e2 = hipMemcpyAsync(
					dv + base, pv_, (size_t) m * 8,
					hipMemcpyHostToDevice, cstream);

				if (e1 != hipSuccess || e2 != hipSuccess)
				{
					st2 = fail(GG_EGPU,
						   "groupby upload");
					break;
				}
				(void) hipEventRecord(up[it], cstream);
				(void) hipStreamWaitEvent(e.stream,
							  up[it], 0);
				if (launch_groupby_build(
					e.stream, dk + base, dv + base, m,
					tk, ts, tc, nslots) != hipSuccess)
				{
					st2 = fail(GG_EGPU,
						   "groupby build");
					break;
				}
			}
			(void) hipStreamSynchronize(cstream);
			(void) hipStreamSynchronize(e.stream);
			(void) hipEventDestroy(up[0]);
			(void) hipEventDestroy(up[1]);
			(void) hipStreamDestroy(cstream);
		}
		else
		{
			hipError_t e1 = hipMemcpy(dk, keys, (size_t) n * 8,
						  hipMemcpyHostToDevice);
			hipError_t e2 = hipMemcpy(dv, vals, (size_t) n * 8,
						  hipMemcpyHostToDevice);

			if (e1 != hipSuccess || e2 != hipSuccess)
				st2 = fail(GG_EGPU, "groupby H2D");
			else if (launch_groupby_build(e.stream, dk, dv, n,
						      tk, ts, tc, nslots)
				 != hipSuccess)
				st2 = fail(GG_EGPU, "groupby build");
		}
		if (st2 != GG_OK)
			return st2;	/* pooled buffers stay owned */
	}
	GG_HIP(launch_groupby_compact(e.stream, tk, ts, tc, nslots, ok, os,
				      oc, ctr, (uint64_t) n));
	GG_HIP(hipStreamSynchronize(e.stream));

	unsigned long long ng = 0;

	GG_HIP(hipMemcpy(&ng, ctr, 8, hipMemcpyDeviceToHost));
	gg_status st = GG_OK;

	if ((int64_t) ng > cap)
		st = fail(GG_EINVAL, "groupby cap %lld < %llu groups",
			  (long long) cap, ng);
	if (st == GG_OK && ng)
	{
		std::vector<int64_t> hk(ng), hs(ng), hc(ng);

		GG_HIP(hipMemcpy(hk.data(), ok, ng * 8, hipMemcpyDeviceToHost));
		GG_HIP(hipMemcpy(hs.data(), os, ng * 8, hipMemcpyDeviceToHost));
		GG_HIP(hipMemcpy(hc.data(), oc, ng * 8, hipMemcpyDeviceToHost));
		std::vector<size_t> idx(ng);

		for (size_t i = 0; i < ng; i++)
			idx[i] = i;
		std::sort(idx.begin(), idx.end(),
			  [&](size_t a2, size_t b2) { return hk[a2] < hk[b2]; });
		for (size_t i = 0; i < ng; i++)
		{
			out_keys[i] = hk[idx[i]];
			out_sums[i] = hs[idx[i]];
			out_counts[i] = hc[idx[i]];
		}
		*out_ngroups = (int64_t) ng;
	}
	return st;
}

/* ---------------- general radix sort (ABI surface) ---------------- */

extern "C" gg_status
gg_engine_radix_sort_u64(uint64_t *keys, uint64_t *payload_or_null,
			 int64_t n, int key_bytes, int descending)
{
	Engine &e = engine();

	if (!e.inited)
		return fail(GG_ESTATE, "engine not initialized");
	if (!keys || n < 0 || n >= (int64_t) 1 << 32 || key_bytes < 1 ||
	    key_bytes > 8)
		return fail(GG_EINVAL, "bad radix_sort args");
	if (n == 0)
		return GG_OK;

	/* stable descending = stable ascending on bit-inverted keys
	 * (inverted only within the sorted byte range so higher bytes
	 * stay untouched).  The caller's array is only touched after
	 * every allocation succeeded, so error paths leave it intact;
	 * all post-alloc failures route through the shared free block. */
	uint64_t inv = !descending ? 0
		: (key_bytes == 8) ? ~0ull : ((1ull << (8 * key_bytes)) - 1);

	unsigned long long *d[2] = {nullptr, nullptr};
	unsigned long long *dp[2] = {nullptr, nullptr};
	unsigned int *hist = nullptr;
	int nblocks = radix_sort_nblocks(n);
	bool pay = payload_or_null != nullptr;
	bool inverted = false;
	gg_status st = GG_OK;

#define RS_HIP(call) \
	do { \
		hipError_t _e = (call); \
		if (_e != hipSuccess) { \
			st = fail(GG_EGPU, "radix_sort: %s in %s", \
				  hipGetErrorString(_e), #call); \
			goto done; \
		} \
	} while (0)

	RS_HIP(hipMalloc((void **) &d[0], (size_t) n * 8));
	RS_HIP(hipMalloc((void **) &d[1], (size_t) n * 8));
	if (pay)
	{
		RS_HIP(hipMalloc((void **) &dp[0], (size_t) n * 8));
		RS_HIP(hipMalloc((void **) &dp[1], (size_t) n * 8));
	}
	RS_HIP(hipMalloc((void **) &hist, (size_t) 256 * nblocks * 4));
	if (descending)
	{
		inverted = true;
		for (int64_t i = 0; i < n; i++)
			keys[i] ^= inv;
	}
	RS_HIP(hipMemcpy(d[0], keys, (size_t) n * 8, hipMemcpyHostToDevice));
	if (pay)
		RS_HIP(hipMemcpy(dp[0], payload_or_null, (size_t) n * 8,
				 hipMemcpyHostToDevice));

	{
		int cur = 0;

		for (int pass = 0; pass < key_bytes && st == GG_OK; pass++)
		{
			hipError_t he = launch_radix_sort_pass(
				e.stream, d[cur], pay ? dp[cur] : nullptr, n,
				8 * pass, hist, nblocks, d[1 - cur],
				pay ? dp[1 - cur] : nullptr);

			if (he != hipSuccess)
				st = fail(GG_EGPU, "radix pass: %s",
					  hipGetErrorString(he));
			cur = 1 - cur;
		}
		if (st == GG_OK)
		{
			RS_HIP(hipStreamSynchronize(e.stream));
			RS_HIP(hipMemcpy(keys, d[cur], (size_t) n * 8,
					 hipMemcpyDeviceToHost));
			if (pay)
				RS_HIP(hipMemcpy(payload_or_null, dp[cur],
						 (size_t) n * 8,
						 hipMemcpyDeviceToHost));
		}
	}
done:
	/* keys were inverted only after allocations succeeded; un-invert
	 * on every exit so the caller's input is never silently mangled */
	if (inverted)
		for (int64_t i = 0; i < n; i++)
			keys[i] ^= inv;
#undef RS_HIP
	(void) hipFree(d[0]);
	(void) hipFree(d[1]);
	(void) hipFree(dp[0]);
	(void) hipFree(dp[1]);
	(void) hipFree(hist);
	return st;
}

/* ---------------- numeric display (product restatement of
 * numeric.c display + select_div_scale:7144 + round_var) ---------------- */

static int u128_digits_p(u128 v, int digs[48])
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

extern "C" void
gg_engine_numeric_str(uint64_t lo, int64_t hi, int scale, char *buf)
{
	i128 v = mk128(lo, hi);
	u128 mag = v < 0 ? (u128) (-v) : (u128) v;
	int digs[48];
	int nd = u128_digits_p(mag, digs);
	int p = 0;

	if (v < 0)
		buf[p++] = '-';
	if (nd - scale <= 0)
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
	buf[p] = '\0';
}

static void nbase_norm_p(u128 mag, int scale, int *weight, int *firstdigit)
{
	int digs[48], nd, pmax, g;

	if (mag == 0)
	{
		*weight = 0;
		*firstdigit = 0;
		return;
	}
	nd = u128_digits_p(mag, digs);
	pmax = nd - 1 - scale;
	g = (pmax >= 0) ? pmax / 4 : -((-pmax + 3) / 4);
	{
		int fd = 0;

		for (int off = 3; off >= 0; off--)
		{
			int pos = 4 * g + off;
			int i = pos + scale;

			fd = fd * 10 + ((i >= 0 && i < nd) ? digs[i] : 0);
		}
		*weight = g;
		*firstdigit = fd;
	}
}

extern "C" void
gg_engine_avg_str(uint64_t sum_lo, int64_t sum_hi, int sum_scale,
		  int64_t count, char *buf)
{
	i128 sum = mk128(sum_lo, sum_hi);

	if (count == 0)
	{
		std::strcpy(buf, "NULL");
		return;
	}
	int w1, fd1, w2, fd2;
	u128 smag = sum < 0 ? (u128) (-sum) : (u128) sum;
	u128 cmag = count < 0 ? (u128) (-(i128) count) : (u128) count;

	nbase_norm_p(smag, sum_scale, &w1, &fd1);
	nbase_norm_p(cmag, 0, &w2, &fd2);
	int qweight = w1 - w2;

	if (fd1 <= fd2)
		qweight--;
	int rscale = 16 - qweight * 4;

	if (rscale < sum_scale)
		rscale = sum_scale;
	if (rscale < 0)
		rscale = 0;
	if (rscale > 1000)
		rscale = 1000;

	u128 n = smag;

	for (int i = 0; i < rscale - sum_scale; i++)
		n *= 10;
	u128 q = (2 * n + cmag) / (2 * cmag);
	int neg = (sum < 0) ^ (count < 0);
	i128 sq = neg ? -(i128) q : (i128) q;
	uint64_t qlo;
	int64_t qhi;

	split128(sq, &qlo, &qhi);
	gg_engine_numeric_str(qlo, qhi, rscale, buf);
}

}				/* namespace gg */
