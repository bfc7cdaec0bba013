/*
 * Order-independent group checksum shared by the engine (GPU reduction)
 * and the CPU oracle, used to pin the FULL Q3 join/group row set beyond
 * the top-k rows (the judge's parity rule is on row sets — SURVEY §0).
 * checksum = wrapping uint64 sum over groups of gg_group_hash(...).
 */
#ifndef GG_CHECKSUM_H
#define GG_CHECKSUM_H

#include <stdint.h>
#include "gg_gen.h"		/* gg_splitmix64 */

GG_GEN_HOSTDEV uint64_t gg_group_hash(int64_t orderkey, uint64_t rev_lo,
				      int64_t rev_hi, int32_t orderdate,
				      int32_t shippriority)
{
	uint64_t h = gg_splitmix64((uint64_t) orderkey);

	h ^= gg_splitmix64(rev_lo + (uint64_t) rev_hi * 0x9E3779B97F4A7C15ull);
	h ^= gg_splitmix64((uint64_t) (uint32_t) orderdate |
			   ((uint64_t) (uint32_t) shippriority << 32));
	return gg_splitmix64(h);
}

#endif /* GG_CHECKSUM_H */
