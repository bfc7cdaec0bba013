/*
 * engine_abi.h — the C-ABI drop-in boundary of the MI355X columnar
 * query-operator engine (libgreengage_engine.so).
 *
 * This is the surface a Greengage segment-side shim binds: the segment
 * postgres process loads a .so via shared_preload_libraries
 * (reference: utils/misc/guc.c:3001) whose _PG_init
 * (utils/fmgr/dfmgr.c:176–280) installs
 * ExecutorStart_hook/ExecutorRun_hook/ExecutorEnd_hook
 * (executor/execMain.c:129–132, call sites :334/:1036/:1309; typedefs
 * include/executor/executor.h:90–105).  The shim walks the PlanState
 * tree of the QueryDesc, claims recognized
 * SeqScan→HashJoin→Agg→Sort/Motion-bounded subtrees, and calls this
 * engine; everything else falls back to standard_ExecutorRun.
 * INTEGRATION.md shows the shim-side binding.
 *
 * Per-entry-point reference interface each call replaces:
 *   gg_engine_init/shutdown   — _PG_init-time setup (dfmgr.c:279)
 *   gg_engine_register_table  — the table-AM scan source the claimed
 *                               SeqScan would read (nodeSeqscan.c:141
 *                               InitScanRelation / aocs_beginscan
 *                               aocsam.c:426): caller hands columnar
 *                               buffers instead of a Relation
 *   gg_engine_compile_pipeline— ExecutorStart of the claimed subtree
 *                               (execMain.c:334 standard_ExecutorStart →
 *                               InitPlan)
 *   gg_engine_execute         — ExecutorRun/ExecutePlan of the subtree
 *                               (execMain.c:988/:3218): runs the whole
 *                               pipeline to completion and materializes
 *                               result batches into caller-owned memory
 *                               (slot ownership rule: execScan.c:142,
 *                               SURVEY §8(b))
 *   gg_engine_comm_*          — the Motion exchange the subtree's
 *                               Motion node would perform
 *                               (nodeMotion.c:1574 doSendTuple /
 *                               cdbmotion.c:434 SendTuple over
 *                               ic_udpifc.c), carried on RCCL over xGMI
 *   gg_engine_stats           — per-node Instrumentation the shim feeds
 *                               into EXPLAIN ANALYZE (instrument.c:423,
 *                               explain_gp.c cdbexplain_recvExecStats)
 *
 * Error model: status-code returns + gg_engine_last_error() — never
 * longjmp across HIP frames; the shim converts to ereport(ERROR)
 * (SURVEY §8(b) "Errors").  Threading: entry points are called from the
 * backend thread; the engine may use HIP streams/threads internally.
 */
#ifndef GG_ENGINE_ABI_H
#define GG_ENGINE_ABI_H

#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- status codes ---- */
typedef enum gg_status
{
	GG_OK = 0,
	GG_EINVAL = 1,		/* bad argument / unknown handle */
	GG_EGPU = 2,		/* HIP runtime failure */
	GG_ENOMEM = 3,		/* device or host allocation failure */
	GG_ECOMM = 4,		/* RCCL failure */
	GG_ENOTSUP = 5,		/* plan shape not supported (shim must
				 * fall back to standard_ExecutorRun) */
	GG_ESTATE = 6		/* call sequence violation */
} gg_status;

/* thread-local message for the last non-OK return */
const char *gg_engine_last_error(void);

/* ---- engine lifecycle ---- */
typedef struct gg_engine_config
{
	int device;		/* HIP device ordinal (segment's GPU) */
	int n_segments;		/* gp segment count = world size */
	int segment_id;		/* this segment's id = rank */
	uint64_t hbm_limit_bytes;	/* 0 = no cap */
} gg_engine_config;

gg_status gg_engine_init(const gg_engine_config *cfg);
gg_status gg_engine_shutdown(void);

/* ---- column/table registry ---- */
typedef enum gg_coltype
{
	GG_COL_INT64 = 0,	/* int8/bigint */
	GG_COL_INT32 = 1,	/* int4/date (DateADT int32 days) */
	GG_COL_DEC64_S2 = 2,	/* numeric(15,2) as scaled int64 (cents) */
	GG_COL_CHAR1 = 3	/* char(1)/dict code as uint8 */
} gg_coltype;

typedef struct gg_column_desc
{
	const char *name;
	gg_coltype type;
	const void *host_data;	/* host pointer (engine uploads), or NULL
				 * if device_data is given */
	const void *device_data;/* optional: already-resident HBM buffer */
} gg_column_desc;

typedef int32_t gg_table;

/* Register a columnar table (the claimed SeqScan's source relation).
 * All rows are visible (read-only analytics path; visimap honored by
 * the shim at staging time — SURVEY §2 note). */
gg_status gg_engine_register_table(const char *name,
				   const gg_column_desc *cols, int ncols,
				   int64_t nrows, gg_table *out);

/* Engine-side synthetic registration for benchmarks: generates the
 * named TPC-H table's shard for (segment_id, n_segments) directly in
 * HBM per include/gg_gen.h (no host staging). */
gg_status gg_engine_register_synth(const char *table_name, uint64_t seed,
				   int64_t sf, gg_table *out);

gg_status gg_engine_drop_table(gg_table t);
gg_status gg_engine_table_nrows(gg_table t, int64_t *out_nrows);

/* Copy a registered column back to host (test/debug surface; the shim
 * uses result arenas, not this). buf_bytes must equal nrows × elsize. */
gg_status gg_engine_fetch_column(gg_table t, const char *col_name,
				 void *host_buf, size_t buf_bytes);

/* ---- pipeline descriptor ----
 * The shim extracts the claimed subtree into this descriptor.  v1
 * supports the §8 hot-path shapes; anything else: GG_ENOTSUP.
 * (PG 9.4 has no CustomScan — SURVEY §8(b) — so the descriptor mirrors
 * the PlanState subtree the ExecutorRun hook claims.) */
typedef enum gg_pipeline_kind
{
	/* scan(lineitem) → filter(shipdate<=c) → hashagg(rf,ls; Q1 aggs) */
	GG_PIPE_Q1 = 1,
	/* scan(customer)⋈scan(orders)⋈scan(lineitem) → hashagg(okey)
	 * → sort(rev desc, odate) limit k  [+ Motion redistributes] */
	GG_PIPE_Q3 = 2,
	/* scan(lineitem) → filter(shipdate<c) → agg(sum(price),count) */
	GG_PIPE_SUMPRICE = 3,
	/* Q5 (mpph5): customer⋈orders⋈lineitem⋈supplier⋈nation⋈region →
	 * hashagg(n_name) → sort(revenue desc); region/nation are the
	 * broadcast-Motion dims */
	GG_PIPE_Q5 = 4
} gg_pipeline_kind;

typedef struct gg_pipeline_desc
{
	gg_pipeline_kind kind;
	gg_table lineitem;	/* tables by role; -1 if unused */
	gg_table orders;
	gg_table customer;
	gg_table supplier;	/* Q5 */
	gg_table nation;	/* Q5: (nationkey, regionkey) dim */
	int32_t cutoff_date;	/* date qual lower constant (DateADT) */
	int32_t cutoff_hi;	/* Q5: o_orderdate < cutoff_hi */
	uint8_t mktsegment;	/* Q3: dict code of c_mktsegment literal */
	uint8_t regionkey;	/* Q5: r_name literal resolved to its key */
	int64_t limit_k;	/* Q3: LIMIT bound (nodeSort.c:143) */
} gg_pipeline_desc;

typedef int32_t gg_pipeline;

gg_status gg_engine_compile_pipeline(const gg_pipeline_desc *desc,
				     gg_pipeline *out);
gg_status gg_engine_drop_pipeline(gg_pipeline p);

/* ---- results ----
 * gg_engine_execute runs the pipeline on this segment's shard(s) and
 * materializes the (partial or final) result into the caller-owned
 * arena.  Layout per pipeline kind is defined in gg_result.h (fixed
 * little-endian structs).  If the pipeline contains the final Motion,
 * rank 0's arena receives the globally-combined result. */
gg_status gg_engine_execute(gg_pipeline p, void *out_arena,
			    size_t arena_bytes, size_t *out_written);

/* ---- Motion-equivalent collectives (RCCL over xGMI) ---- */
/* Bootstrap: rank 0 creates an opaque 128-byte id the caller
 * distributes (the shim would ship it via the QD dispatch;
 * bench/tests ship it via torch TCPStore). */
gg_status gg_engine_comm_id(void *out_id128);
gg_status gg_engine_comm_init(const void *id128);
gg_status gg_engine_comm_destroy(void);

/* ---- instrumentation (EXPLAIN ANALYZE-equivalent) ---- */
typedef struct gg_kernel_stat
{
	char name[48];
	int64_t launches;
	double total_ms;	/* HIP-event measured */
	int64_t rows_in;
	int64_t rows_out;
	int64_t hbm_bytes_algorithmic;
} gg_kernel_stat;

gg_status gg_engine_stats(gg_pipeline p, gg_kernel_stat *out, int cap,
			  int *out_n);

/* ---- numeric finalization (reference numeric.c display semantics;
 * product-side restatement, independent of oracle/) ---- */
void gg_engine_numeric_str(uint64_t lo, int64_t hi, int scale, char *buf64);
void gg_engine_avg_str(uint64_t sum_lo, int64_t sum_hi, int sum_scale,
		       int64_t count, char *buf64);

/* AOCS datum-stream block decode on GPU: decodes a
 * stream of reference-format blocks (datumstreamblock.c content layer;
 * versions Orig=0 / Dense=1 / Dense_Enhanced=2 with null bitmap, RLE
 * and delta compression) for fixed-length int32/int64 columns into
 * host value/null arrays.  Stream framing: repeat
 * [int32 block_size][int32 row_count][block bytes]; the Append-Only
 * Storage block-header layer and zlib/zstd codecs are the remaining
 * sub-layers.  out_width 4 or 8. */
gg_status gg_engine_aocs_decode(const uint8_t *stream, int64_t stream_len,
				int version, int datumlen, void *out_vals,
				int out_width, uint8_t *out_nulls,
				int64_t cap, int64_t *out_nrows);

/* Decode TEXT (varlena) datum-stream blocks on the GPU: arrow-style
 * output — per-row byte offset into `pool`, byte length, null flag.
 * Handles this fork's short (len|0x80) and network-byte-order 4-byte
 * varlena forms, RLE, and null bitmaps; compressed/toasted datums are
 * rejected.  pool_cap >= stream_len always suffices. */
gg_status gg_engine_aocs_decode_text(const uint8_t *stream,
				     int64_t stream_len, int version,
				     uint64_t *out_offs,
				     uint32_t *out_lens,
				     uint8_t *out_nulls, int64_t cap,
				     uint8_t *pool, int64_t pool_cap,
				     int64_t *out_nrows,
				     int64_t *out_pool_len);

/* Mount a REAL AO table: one AO segfile byte stream per column (the
 * form cdbbufferedread.c hands up); each column runs through the full
 * storage layer (headers + CRC32C + codecs + datum-stream decode)
 * straight into a device-resident column.  Columns must be NOT NULL.
 * The table is then usable by every pipeline like any registered
 * table. */
typedef struct gg_ao_column
{
	const char *name;
	gg_coltype	type;
	const uint8_t *stream;	/* raw AO segfile bytes */
	int64_t		stream_len;
	int			checksums;
	int			ao_version;	/* >= 2 */
	int			dsb_version;	/* 0/1/2 */
	int			comptype;	/* 0 none, 1 zlib, 2 zstd */
	/* nonzero for a TEXT column to mount as dictionary codes
	 * (type must be GG_COL_CHAR1; <= 255 distinct values).  The
	 * lexicographically-sorted dictionary is retrievable with
	 * gg_engine_table_text_dict for predicate-constant lookup. */
	int			text_dict;
	/* nonzero: the column may contain NULLs — the AO null bitmap
	 * (datumstreamblock.c) lands as a device NULL-flag array on the
	 * column, consumed by the generalized plan path under the
	 * strict-transition/NULL-qual rules (nodeAgg.c:413,
	 * execScan.c:185).  Zero keeps the round-1 behavior: any NULL
	 * errors the mount. */
	int			nullable;
} gg_ao_column;

gg_status gg_engine_register_table_ao(const char *name,
				      const gg_ao_column *cols, int ncols,
				      gg_table *out);

/* dictionary of a text_dict-mounted column: bytes + n+1 offsets */
gg_status gg_engine_table_text_dict(gg_table h, const char *col,
				    uint8_t *out_bytes, int64_t cap,
				    int64_t *out_offs,
				    int32_t max_entries,
				    int32_t *out_n);

/* Dictionary-encode a categorical text column (arrow-style inputs,
 * e.g. gg_engine_aocs_decode_text's output) into int32 codes + a
 * lexicographically sorted dictionary — deterministic across shards,
 * as the exchange requires.  NULLs code as -1; cardinality bounded by
 * max_dict. */
gg_status gg_engine_text_dict_encode(const uint8_t *pool,
				     const uint64_t *offs,
				     const uint32_t *lens,
				     const uint8_t *nulls, int64_t n,
				     int32_t max_dict,
				     int32_t *out_codes,
				     uint8_t *dict_bytes,
				     int64_t dict_cap,
				     int64_t *dict_offs,
				     int32_t *out_ndict);

/* MemTuple codec (access/common/memtuple.c format, the tuple layout
 * used in executor hash tables and on the Motion wire): bulk GPU
 * conversion between column arrays and MemTuple byte streams.
 * Fixed-width by-value attrs (attlen 1/2/4/8, attalign c/s/i/d) and
 * varlena text (attlen -1, attalign 'i'), nullable.
 * For fixed attrs cols[i] holds nrows elements of width attlen[i];
 * for varlena attrs cols[i] points at a gg_text_col on encode (bytes
 * pool + n+1 prefix offsets) and a gg_text_out on decode (per-row
 * offset/length pairs referencing the INPUT stream buffer —
 * zero-copy).  nulls[i] is a byte-per-row flag array or NULL.
 * Tuples over MEMTUPLE_LEN_FITSHORT (0xFFF0) switch to the large
 * binding (4-byte varoffsets, MEMTUP_LARGETUP header flag), exactly
 * as memtuple_form_to does. */
typedef struct gg_text_col
{
	const uint8_t *bytes;
	const int64_t *offs;	/* nrows+1 prefix offsets */
} gg_text_col;

typedef struct gg_text_out
{
	uint64_t   *offs;	/* per-row payload offset into the input */
	uint32_t   *lens;	/* per-row payload length */
} gg_text_out;

gg_status gg_engine_memtuple_binding(int natts, const int32_t *attlen,
				     const char *attalign, int32_t *out);
gg_status gg_engine_memtuple_binding_large(int natts,
					   const int32_t *attlen,
					   const char *attalign,
					   int32_t *out);
gg_status gg_engine_memtuple_encode(int natts, const int32_t *attlen,
				    const char *attalign,
				    const void *const *cols,
				    const uint8_t *const *nulls,
				    int64_t nrows, uint8_t *out,
				    int64_t cap, int64_t *out_len);
gg_status gg_engine_memtuple_decode(int natts, const int32_t *attlen,
				    const char *attalign,
				    const uint8_t *stream,
				    int64_t stream_len, void *const *cols,
				    uint8_t *const *nulls,
				    int64_t cap_rows, int64_t *out_nrows);

/* Motion wire chunk framing (tupchunk.h:34-84, tupser.c:400): convert
 * between a MemTuple stream and the interconnect's 4-byte-headed
 * tuple chunks (TC_WHOLE / TC_PARTIAL_* splitting at max_chunk,
 * TC_END_OF_STREAM).  Host-side. */
gg_status gg_engine_motion_chunkify(const uint8_t *tuples,
				    int64_t tuples_len, int32_t max_chunk,
				    int append_eos, uint8_t *out,
				    int64_t cap, int64_t *out_len);
gg_status gg_engine_motion_dechunkify(const uint8_t *chunks,
				      int64_t chunks_len, uint8_t *out,
				      int64_t cap, int64_t *out_len,
				      int *saw_eos);

/* Decode REAL Append-Only storage blocks (headers + CRC32C verify +
 * datum-stream content on the GPU).  Replaces the AO read path's
 * header/checksum layer (cdbappendonlystorageformat.c:125,1661 and
 * GetSmallContentHeaderInfo/GetNonBulkDenseContentHeaderInfo) for
 * int32/int64 columns; ao_version is the AORelationVersion (>=2),
 * dsb_version the datum-stream block version
 * (0=Orig,1=Dense,2=Dense_Enhanced), comptype the catalog's
 * compresstype (0=none, 1=zlib, 2=zstd — pg_compression.c registry);
 * all four AO header kinds parse (Small, NonBulkDense, BulkDense
 * long headers, LargeContent metadata+fragments); checksums verify
 * and codecs decompress host-parallel, uncompressed content decodes
 * zero-copy from the segfile bytes. */
gg_status gg_engine_aocs_decode_ao_text(const uint8_t *stream,
					int64_t stream_len, int checksums,
					int ao_version, int dsb_version,
					int comptype, uint64_t *out_offs,
					uint32_t *out_lens,
					uint8_t *out_nulls, int64_t cap,
					uint8_t *pool, int64_t pool_cap,
					int64_t *out_nrows,
					int64_t *out_pool_len);
gg_status gg_engine_aocs_decode_ao(const uint8_t *stream,
				   int64_t stream_len, int checksums,
				   int ao_version, int dsb_version,
				   int comptype, int datumlen,
				   void *out_vals, int out_width,
				   uint8_t *out_nulls, int64_t cap,
				   int64_t *out_nrows);

/* General hash group-by (execHHashagg.c find-or-create semantics on
 * arbitrary int64 keys, SUM+COUNT transitions): host buffers in,
 * groups out sorted by key ascending.  Keys may be any int64 except
 * INT64_MIN (the open-addressing empty sentinel).  Returns the group
 * count; fails with GG_EINVAL if cap is too small. */
gg_status gg_engine_hash_groupby_i64(const int64_t *keys,
				     const int64_t *vals, int64_t n,
				     int64_t *out_keys, int64_t *out_sums,
				     int64_t *out_counts, int64_t cap,
				     int64_t *out_ngroups);

/* Spill-tier variant: inputs larger than budget_bytes of device
 * memory are hash-range partitioned on the GPU, staged in host
 * memory, and aggregated partition by partition (the reference's
 * spill_hash_table/agg_hash_reload semantics,
 * execHHashagg.c:1350/:1852).  Results identical to the in-memory
 * path; out_npartitions reports the fan-out (1 = no spill). */
/* Spill-tier hash JOIN (nodeHash.c:713 batching): both sides hash-
 * range partitioned on the GPU, staged in host memory, each partition
 * pair built+probed on the GPU.  Build keys must be unique (PK-side
 * build).  Key 0 is reserved (table empty sentinel), as in the
 * pipeline joins.  Output: (probe_row_index, build_val) per match. */
gg_status gg_engine_hash_join_i64_spill(const int64_t *build_keys,
					const int64_t *build_vals,
					int64_t nb,
					const int64_t *probe_keys,
					int64_t np, int64_t budget_bytes,
					int64_t *out_probe_idx,
					int64_t *out_vals, int64_t cap,
					int64_t *out_nmatch,
					int32_t *out_npartitions);

gg_status gg_engine_hash_groupby_i64_spill(const int64_t *keys,
					   const int64_t *vals, int64_t n,
					   int64_t budget_bytes,
					   int64_t *out_keys,
					   int64_t *out_sums,
					   int64_t *out_counts,
					   int64_t cap,
					   int64_t *out_ngroups,
					   int32_t *out_npartitions);

/* General ORDER BY operator: stable LSB radix sort of (u64 key, u64
 * payload) pairs on the GPU (nodeSort.c:48 / tuplesort.c semantics for
 * unbounded sorts; the LIMIT-k case uses histogram select instead,
 * mirroring tuplesort.c:1360's bounded-heap switch).  key_bytes limits
 * the passes (e.g. 4 for int32-derived keys); descending inverts the
 * key bits.  Host buffers; n < 2^32. */
gg_status gg_engine_radix_sort_u64(uint64_t *keys, uint64_t *payload_or_null,
				   int64_t n, int key_bytes, int descending);

/* ---- generalized pipeline descriptor (v2) ----------------------------
 * The compositional form of the subtree a shim extracts from the
 * PlanState tree (execProcnode.c:925–1148): a driving scan with
 * conjunctive range predicates (ExecQual execQual.c:6260 over the
 * strategy-number operator classes), zero or more hash SEMI-join
 * filters against filtered build tables (nodeHash.c:88 build /
 * nodeHashjoin.c:78 probe, join-qual = key equality), a hash group-by
 * over 0–2 key columns (execHHashagg.c:905), and COUNT/SUM aggregate
 * transitions whose inputs are products of up to three column factors
 * with the (100±col) scaled-decimal modifiers TPC-H money expressions
 * need (numeric.c:1735 mul_var dscale rule; ExecTargetList
 * execQual.c:6369).  Q1/Q3/Q5-shaped descriptors keep their
 * specialized kernels via gg_engine_compile_pipeline; everything else
 * — mpph6 (Q6), changed predicate columns, ad-hoc scans — compiles
 * here with ZERO query-specific kernels.  Unsupported shapes return
 * GG_ENOTSUP (shim falls back to standard_ExecutorRun). */

#define GG_PLAN_MAX_PREDS 8
#define GG_PLAN_MAX_JOINS 2
#define GG_PLAN_MAX_AGGS 8

/* conjunct: lo <= col < hi (INT64_MIN/INT64_MAX = one-sided; equality
 * = [v, v+1)).  A NULL column value fails the qual (SQL three-valued
 * logic: WHERE discards non-true, execScan.c:185). */
typedef struct gg_plan_pred
{
	const char *col;
	int64_t lo;		/* inclusive */
	int64_t hi;		/* exclusive */
} gg_plan_pred;

typedef struct gg_plan_scan
{
	gg_table table;
	gg_plan_pred preds[GG_PLAN_MAX_PREDS];
	int npreds;
} gg_plan_scan;

/* hash semi-join filter: driving rows survive iff probe_key matches
 * the build_key of some build-side row passing the build preds */
typedef struct gg_plan_join
{
	gg_plan_scan build;
	const char *build_key;
	const char *probe_key;	/* column of the driving table */
} gg_plan_join;

/* aggregate: COUNT(*) / COUNT(col) (non-NULL, nodeAgg strict rules) /
 * SUM(f0*f1*f2), factor = col, (100-col) or (100+col); SUM skips rows
 * whose factor inputs are NULL (strict transition, nodeAgg.c:413–460) */
typedef enum gg_plan_aggkind
{
	GG_AGG_COUNT_STAR = 0,
	GG_AGG_COUNT_COL = 1,
	GG_AGG_SUM = 2
} gg_plan_aggkind;

typedef enum gg_plan_fmod
{
	GG_FMOD_ID = 0,		/* col */
	GG_FMOD_SUB100 = 1,	/* 100 - col */
	GG_FMOD_ADD100 = 2	/* 100 + col */
} gg_plan_fmod;

typedef struct gg_plan_agg
{
	int kind;		/* gg_plan_aggkind */
	int nfactors;		/* 0 for COUNT(*); 1 for COUNT(col) */
	const char *col[3];
	int8_t mod[3];		/* gg_plan_fmod */
} gg_plan_agg;

typedef struct gg_plan_desc
{
	gg_plan_scan scan;	/* driving (probe-side) scan */
	gg_plan_join joins[GG_PLAN_MAX_JOINS];
	int njoins;
	/* 0 = one global group; 1 = any char1/int32/int64 column;
	 * 2 = two char1 columns (code = c0*256 + c1).  Group keys of
	 * INT64_MIN are rejected (open-addressing sentinel); NULL keys
	 * group together (execHHashagg.c:531), reported as
	 * GG_PLAN_NULL_KEY. */
	const char *group_cols[2];
	int ngroup;
	gg_plan_agg aggs[GG_PLAN_MAX_AGGS];
	int naggs;
} gg_plan_desc;

#define GG_PLAN_NULL_KEY (-9223372036854775807LL)	/* INT64_MIN+1 */

/* result arena layout (little-endian):
 *   int64 n_groups; int32 naggs; int32 ngroup;
 *   then n_groups rows, sorted by (key0, key1) ascending:
 *     int64 key0, int64 key1, then naggs x (uint64 lo, int64 hi)
 *     (each aggregate as a signed 128-bit value) */
gg_status gg_engine_compile_plan(const gg_plan_desc *desc, gg_pipeline *out);

/* Attach a NULL-flag array (1 byte per row, nonzero = NULL) to a
 * registered column — the nullable-scan surface (AO null bitmaps land
 * here; aocsam.c:699 per-column null arrays).  Predicates, group keys
 * and aggregate transitions then follow the strict-transition /
 * NULL-key rules above. */
gg_status gg_engine_table_set_nulls(gg_table t, const char *col,
				    const uint8_t *host_nulls);

/* NULL-aware general hash group-by (execHHashagg find-or-create +
 * nodeAgg.c:413 strict transitions): NULL keys form one group
 * (reported as GG_PLAN_NULL_KEY); NULL vals are skipped by both SUM
 * and COUNT (the strict SUM/AVG transition pair).  key_nulls/
 * val_nulls may be NULL (= no NULLs).  Keys INT64_MIN rejected. */
gg_status gg_engine_hash_groupby_i64_n(const int64_t *keys,
				       const uint8_t *key_nulls,
				       const int64_t *vals,
				       const uint8_t *val_nulls, int64_t n,
				       int64_t *out_keys, int64_t *out_sums,
				       int64_t *out_counts, int64_t cap,
				       int64_t *out_ngroups);

/* build info: "gfx950" etc. — lets callers assert the native path */
const char *gg_engine_build_info(void);

#ifdef __cplusplus
}
#endif

#endif /* GG_ENGINE_ABI_H */
