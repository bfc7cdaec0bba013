/*
 * Result-arena layouts written by gg_engine_execute (engine_abi.h).
 * Fixed little-endian structs; caller owns the arena (SURVEY §8(b)
 * "Ownership/memory": results materialized into caller-provided arena).
 */
#ifndef GG_RESULT_H
#define GG_RESULT_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* GG_PIPE_Q1: one struct.  Groups ordered by (l_returnflag,
 * l_linestatus) byte order — A/F A/O N/F N/O R/F R/O; unoccupied
 * groups have count 0 (the reference emits only occupied groups;
 * callers skip count==0 rows). */
typedef struct gg_q1_result_group
{
	int64_t count;
	int64_t sum_qty_c;	/* scale 2 */
	int64_t sum_base_c;	/* scale 2 */
	int64_t sum_dcol_c;	/* scale 2: sum(l_discount) for avg_disc */
	uint64_t disc_lo;	/* scale 4 int128 */
	int64_t disc_hi;
	uint64_t charge_lo;	/* scale 6 int128 */
	int64_t charge_hi;
	uint8_t returnflag;	/* 'A'|'N'|'R' */
	uint8_t linestatus;	/* 'F'|'O' */
	uint8_t _pad[6];
} gg_q1_result_group;

typedef struct gg_q1_result
{
	int32_t n_groups;	/* occupied groups */
	int32_t _pad;
	gg_q1_result_group groups[6];
} gg_q1_result;

/* GG_PIPE_Q3 */
typedef struct gg_q3_result_row
{
	int64_t orderkey;
	uint64_t rev_lo;	/* scale 4 int128 */
	int64_t rev_hi;
	int32_t orderdate;	/* DateADT */
	int32_t shippriority;
} gg_q3_result_row;

typedef struct gg_q3_result_hdr
{
	int64_t n_out;		/* rows following this header (<= LIMIT k) */
	int64_t n_groups;	/* full join result group count */
	uint64_t rev_sum_lo;	/* sum of revenue over all groups, scale 4 */
	int64_t rev_sum_hi;
	uint64_t group_checksum;/* include/gg_checksum.h definition */
	int64_t n_join_rows;
} gg_q3_result_hdr;
/* arena: gg_q3_result_hdr then n_out × gg_q3_result_row */

/* GG_PIPE_Q5: rows ordered by revenue DESC (ties: n_name ASC, the
 * shared refinement); only nations with count > 0 are emitted */
typedef struct gg_q5_result_row
{
	int32_t nationkey;
	int32_t _pad;
	int64_t count;
	uint64_t rev_lo;	/* scale 4 int128 */
	int64_t rev_hi;
} gg_q5_result_row;

typedef struct gg_q5_result_hdr
{
	int64_t n_out;
} gg_q5_result_hdr;
/* arena: gg_q5_result_hdr then n_out × gg_q5_result_row */

/* GG_PIPE_SUMPRICE */
typedef struct gg_sumprice_result
{
	int64_t sum_c;		/* scale 2 */
	int64_t count;
} gg_sumprice_result;

#ifdef __cplusplus
}
#endif

#endif /* GG_RESULT_H */
