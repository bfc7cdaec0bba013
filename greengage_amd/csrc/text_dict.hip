/*
 * GPU dictionary encoding for text columns (predicates
 * like `c_mktsegment = 'MACHINERY'` over categorical text).  The
 * engine's pipelines take small integer codes at the boundary
 * (DESIGN.md data layout); this kernel set turns a decoded text
 * column (the arrow-style output of gg_engine_aocs_decode_text) into
 * codes + a DETERMINISTIC dictionary:
 *
 *   1. per row: hash the bytes (gg_hash_any — the reference's own
 *      hash, device-capable) and find-or-claim a slot in an
 *      open-addressing table (byte-compare on collision); record the
 *      row's slot.
 *   2. compact the claimed slots; the HOST sorts the unique strings
 *      lexicographically (bytewise, like text_lt with C collation)
 *      and builds slot -> final-id; a map kernel emits per-row codes.
 *
 * The dictionary order is independent of hashing/scheduling, so codes
 * are identical across shards and runs — the property the multi-GPU
 * exchange needs.
 */
#include <hip/hip_runtime.h>

#include "engine_internal.h"
#include "../../include/gg_pg_hash.h"

namespace gg
{

namespace
{

/* slot state: 0 empty, else 1 + row index of the claiming row */
__global__ void
k_td_insert(const uint8_t *__restrict__ pool,
	    const unsigned long long *__restrict__ offs,
	    const uint32_t *__restrict__ lens,
	    const uint8_t *__restrict__ nulls, int64_t n,
	    unsigned long long *__restrict__ slots, uint64_t nslots,
	    uint32_t *__restrict__ rowslot,
	    unsigned long long *__restrict__ err)
{
	const int64_t stride = (int64_t) gridDim.x * blockDim.x;

	for (int64_t i = (int64_t) blockIdx.x * blockDim.x + threadIdx.x;
	     i < n; i += stride)
	{
		if (nulls && nulls[i])
		{
			rowslot[i] = 0xFFFFFFFFu;
			continue;
		}
		const uint8_t *s = pool + offs[i];
		uint32_t len = lens[i];
		uint32_t h = gg_hash_any(s, (int) len);
		uint64_t pos = (uint64_t) h & (nslots - 1);
		int probes = 0;

		for (;;)
		{
			unsigned long long cur = slots[pos];

			if (cur == 0)
			{
				unsigned long long prev =
					atomicCAS(&slots[pos], 0ull,
						  (unsigned long long)
						  (i + 1));

				if (prev == 0)
				{
					rowslot[i] = (uint32_t) pos;
					break;
				}
				cur = prev;
			}
			{	/* byte-compare with the claiming row */
				int64_t j = (int64_t) cur - 1;
				const uint8_t *t = pool + offs[j];
				bool eq = lens[j] == len;

				for (uint32_t z = 0; eq && z < len; z++)
					eq = t[z] == s[z];
				if (eq)
				{
					rowslot[i] = (uint32_t) pos;
					break;
				}
			}
			pos = (pos + 1) & (nslots - 1);
			if (++probes > (int) nslots)
			{
				atomicOr(err, 1ull);	/* table full */
				rowslot[i] = 0xFFFFFFFFu;
				break;
			}
		}
	}
}

__global__ void
k_td_map(const uint32_t *__restrict__ rowslot, int64_t n,
	 const int32_t *__restrict__ slot_to_id,
	 int32_t *__restrict__ codes)
{
	const int64_t stride = (int64_t) gridDim.x * blockDim.x;

	for (int64_t i = (int64_t) blockIdx.x * blockDim.x + threadIdx.x;
	     i < n; i += stride)
		codes[i] = rowslot[i] == 0xFFFFFFFFu ? -1
			: slot_to_id[rowslot[i]];
}

}				/* anonymous namespace */

hipError_t
launch_td_insert(hipStream_t s, const uint8_t *pool,
		 const unsigned long long *offs, const uint32_t *lens,
		 const uint8_t *nulls, int64_t n, unsigned long long *slots,
		 uint64_t nslots, uint32_t *rowslot, unsigned long long *err)
{
	int blocks = (int) ((n + 255) / 256);

	if (blocks > 2048)
		blocks = 2048;
	if (blocks < 1)
		blocks = 1;
	hipLaunchKernelGGL(k_td_insert, dim3(blocks), dim3(256), 0, s, pool,
			   offs, lens, nulls, n, slots, nslots, rowslot,
			   err);
	return hipGetLastError();
}

hipError_t
launch_td_map(hipStream_t s, const uint32_t *rowslot, int64_t n,
	      const int32_t *slot_to_id, int32_t *codes)
{
	int blocks = (int) ((n + 255) / 256);

	if (blocks > 2048)
		blocks = 2048;
	if (blocks < 1)
		blocks = 1;
	hipLaunchKernelGGL(k_td_map, dim3(blocks), dim3(256), 0, s, rowslot,
			   n, slot_to_id, codes);
	return hipGetLastError();
}

}				/* namespace gg */
