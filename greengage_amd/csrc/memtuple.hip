/*
 * MemTuple codec — GPDB's compact tuple format, used in executor hash
 * tables and on the Motion wire (reference
 * access/common/memtuple.c, format comment :24–35).
 *
 * The engine's native exchange is columnar (SoA over RCCL) — strictly
 * better on a GPU — so this codec is the BOUNDARY piece: bulk
 * conversion between device column arrays and MemTuple byte streams,
 * for interoperating with CPU segments in a mixed gang.
 *
 * Restated layout rules (each cited to the reference):
 *  - binding: attrs are laid out physically by descending alignment
 *    pass (8-aligned, then 4, 2, 1 — create_col_bind,
 *    memtuple.c:200–416); offset starts at 8 when any 'd'-aligned attr
 *    exists else 4 (create_memtuple_binding :420–440,
 *    cur_offset init :206);
 *  - header: uint32 [lead bit 0x80000000 | len | HASNULL=1]
 *    (memtup.h MEMTUP_* macros); len is MEMTUP_ALIGN(…)=8-aligned;
 *  - null bitmap: one bit per PHYSICAL column (null_byte/null_mask,
 *    :262–264), stored at tuple+4; when it needs more than the 4
 *    header-pad bytes (col_align 8) or any bytes at all (col_align 4),
 *    attrs shift by null_bitmap_extra_size
 *    (compute_null_bitmap_extra_size :58–71, form_to :685–691);
 *  - null attrs occupy no space: each later attr moves back by the sum
 *    of len_aligned of preceding-physical null attrs
 *    (add_null_save_aligned :130, memtuple_get_attr_ptr :524 with
 *    null_saves_aligned, form_to :717);
 *  - tuples over 0xFFF0 bytes use the large binding (form_to :622) —
 *    identical to the small one for fixed-width attrs, which is all
 *    this round supports (varlena is the remaining (f)3 sub-row).
 *
 * Parity is pinned against memtuple.c itself compiled in place
 * (oracle/ref_build: ref_mt_create_binding/ref_mt_form/ref_mt_getattr).
 */
#include <hip/hip_runtime.h>
#include <cstring>

#include "engine_internal.h"

namespace gg
{

__host__ __device__ static inline uint32_t
mt_align(uint32_t off, int align)
{
	return (off + (uint32_t) align - 1) & ~((uint32_t) align - 1);
}

/*
 * Restatement of create_memtuple_binding + create_col_bind for
 * fixed-width attrs (attlen 1/2/4/8, attalign c/s/i/d, byval).
 * Returns 0 on success, nonzero for unsupported schemas.
 */
static int
mt_compute_one(int natts, const int32_t *attlen, const char *attalign,
	       bool islarge, MtBind *out)
{
	if (natts < 1 || natts > GG_MT_MAX_ATTS)
		return 1;
	out->nvar = 0;
	for (int i = 0; i < natts; i++)
	{
		char a = attalign[i];

		if (a != 'c' && a != 's' && a != 'i' && a != 'd')
			return 3;
		if (attlen[i] == -1)
		{		/* varlena (text/bytea/varchar) */
			out->nvar++;
			continue;
		}
		if (attlen[i] != 1 && attlen[i] != 2 && attlen[i] != 4 &&
		    attlen[i] != 8)
			return 2;	/* cstring (-2): not supported */
	}
	out->natts = natts;
	out->column_align = 4;
	for (int i = 0; i < natts; i++)
		if (attlen[i] > 0 && attalign[i] == 'd')
			out->column_align = 8;
	{
		/* compute_null_bitmap_extra_size (memtuple.c:58) */
		int nbytes = (natts + 7) >> 3;
		int avail = (out->column_align == 4) ? 0 : 4;

		out->null_bitmap_extra = nbytes <= avail ? 0 :
			(int32_t) mt_align((uint32_t) (nbytes - avail),
					   out->column_align);
	}
	/* four passes by alignment, as create_col_bind does; varlena
	 * takes a 2-byte varoffset slot in the 's' pass (small binding)
	 * or a 4-byte slot in the 'i' pass (large binding, tuples over
	 * MEMTUPLE_LEN_FITSHORT=0xFFF0 — create_col_bind islarge) */
	uint32_t cur = (out->column_align == 8) ? 8 : 4;
	int physical = 0;
	int prev = -1;
	static const char pass_align[4] = {'d', 'i', 's', 'c'};

	for (int pass = 0; pass < 4; pass++)
	{
		for (int i = 0; i < natts; i++)
		{
			bool isvar = attlen[i] == -1;
			char a;
			int slot;

			if (isvar)
			{
				if (pass != (islarge ? 1 : 2))
					continue;
				a = islarge ? 'i' : 's';
				slot = islarge ? 4 : 2;
			}
			else
			{
				if (attalign[i] != pass_align[pass])
					continue;
				a = attalign[i];
				slot = attlen[i];
			}
			int al = (a == 'd') ? 8 : (a == 'i') ? 4 :
				(a == 's') ? 2 : 1;

			cur = mt_align(cur, al);
			out->offset[i] = (int32_t) cur;
			out->len[i] = slot;
			out->is_var[i] = isvar ? 1 : 0;
			/* the DATUM alignment in the varlen section is the
			 * declared attalign (att_align_nominal(...,
			 * attr->attalign), form_to :788) — 'i' for text */
			out->align_of[i] =
				(attalign[i] == 'd') ? 8 :
				(attalign[i] == 'i') ? 4 :
				(attalign[i] == 's') ? 2 : 1;
			out->null_byte[i] = physical >> 3;
			out->null_mask[i] =
				(uint8_t) (1u << (physical & 7));
			out->phys[i] = physical;
			if (prev >= 0)	/* len_aligned of the PREVIOUS
					 * physical attr is its len aligned
					 * to THIS attr's alignment
					 * (add_null_save_aligned) */
				out->len_aligned[prev] = (int32_t)
					mt_align((uint32_t) out->len[prev],
						 al);
			prev = i;
			physical++;
			cur += (uint32_t) slot;
		}
	}
	if (prev >= 0)		/* last attr: no extra alignment ('c') */
		out->len_aligned[prev] = out->len[prev];
	out->var_start = (int32_t) cur;
	return 0;
}

int
mt_compute_binding(int natts, const int32_t *attlen, const char *attalign,
		   MtBind *out)
{
	return mt_compute_one(natts, attlen, attalign, false, out);
}

int
mt_compute_binding_large(int natts, const int32_t *attlen,
			 const char *attalign, MtBind *out)
{
	return mt_compute_one(natts, attlen, attalign, true, out);
}

namespace
{

__device__ inline uint32_t
d_mt_align8(uint32_t x)
{
	return (x + 7u) & ~7u;
}

/* tuple length for one row (compute_memtuple_size_using_bind :462);
 * var_offs[i] = row-offset array (n+1 prefix) for varlena attr i, or
 * null for fixed-width attrs.  Text payloads <= 126 bytes convert to
 * the short form (no alignment); longer keep the 4-byte header,
 * aligned to the attr alignment (:486-500). */
__device__ uint32_t
d_mt_len_one(const MtBind &b, const int64_t *const *var_offs,
	     const uint8_t *const *nulls, int64_t row, bool *hasnull_out)
{
	uint32_t len = (uint32_t) b.var_start;
	bool hasnull = false;

	for (int i = 0; i < b.natts; i++)
		if (nulls[i] && nulls[i][row])
		{
			hasnull = true;
			len -= (uint32_t) b.len_aligned[i];
		}
	if (hasnull)
		len += (uint32_t) b.null_bitmap_extra;
	for (int i = 0; i < b.natts; i++)
	{
		if (!b.is_var[i] || (nulls[i] && nulls[i][row]))
			continue;
		uint32_t paylen = (uint32_t)
			(var_offs[i][row + 1] - var_offs[i][row]);

		if (paylen + 1 <= 0x7F)
			len += paylen + 1;	/* short form */
		else
		{
			len = mt_align(len, (int) b.align_of[i]);
			len += 4 + paylen;
		}
	}
	*hasnull_out = hasnull;
	return d_mt_align8(len);
}

/* size with the small binding first; over FITSHORT the large binding
 * applies (compute_memtuple_size :509) */
__device__ uint32_t
d_mt_len(const MtBind &b, const MtBind &bl,
	 const int64_t *const *var_offs, const uint8_t *const *nulls,
	 int64_t row, bool *hasnull_out, bool *islarge_out)
{
	uint32_t len = d_mt_len_one(b, var_offs, nulls, row, hasnull_out);

	if (len <= 0xFFF0u)
	{
		*islarge_out = false;
		return len;
	}
	*islarge_out = true;
	return d_mt_len_one(bl, var_offs, nulls, row, hasnull_out);
}

__global__ void
k_mt_encode(MtBind bs, MtBind blg, const void *const *__restrict__ cols,
	    const int64_t *const *__restrict__ var_offs,
	    const uint8_t *const *__restrict__ nulls, int64_t nrows,
	    const int64_t *__restrict__ offs, uint8_t *__restrict__ out)
{
	const int64_t stride = (int64_t) gridDim.x * blockDim.x;

	for (int64_t r = (int64_t) blockIdx.x * blockDim.x + threadIdx.x;
	     r < nrows; r += stride)
	{
		bool hasnull, islarge;
		uint32_t len = d_mt_len(bs, blg, var_offs, nulls, r,
					&hasnull, &islarge);
		const MtBind &b = islarge ? blg : bs;
		uint8_t *tup = out + offs[r];

		for (uint32_t z = 0; z < len; z++)
			tup[z] = 0;
		/* header: lead bit | len | hasnull | LARGETUP (memtup.h) */
		{
			uint32_t hdr = 0x80000000u | len |
				(hasnull ? 1u : 0u) | (islarge ? 2u : 0u);

			memcpy(tup, &hdr, 4);
		}
		uint8_t *start = tup +
			(hasnull ? b.null_bitmap_extra : 0);
		uint8_t *nullp = tup + 4;

		if (hasnull)
			for (int i = 0; i < b.natts; i++)
				if (nulls[i] && nulls[i][r])
					nullp[b.null_byte[i]] |=
						b.null_mask[i];
		/* varlen section cursor, relative to `start`
		 * (form_to :679-691: varlen_start = mtup + var_start -
		 * null_save_len, then += extra when hasnull — i.e. the
		 * section starts right after the null-compacted fixed
		 * area, at start + (var_start - total_null_save)) */
		uint32_t nullsave_total = 0;

		if (hasnull)
			for (int j = 0; j < b.natts; j++)
				if (nulls[j] && nulls[j][r])
					nullsave_total += (uint32_t)
						b.len_aligned[j];
		uint32_t vcur = (uint32_t) b.var_start - nullsave_total;

		for (int i = 0; i < b.natts; i++)
		{
			if (nulls[i] && nulls[i][r])
				continue;
			/* shift back by len_aligned of preceding-physical
			 * null attrs (memtuple_get_attr_ptr) */
			uint32_t save = 0;

			if (hasnull)
				for (int j = 0; j < b.natts; j++)
					if (b.phys[j] < b.phys[i] &&
					    nulls[j] && nulls[j][r])
						save += (uint32_t)
							b.len_aligned[j];
			uint8_t *dst = start + b.offset[i] - save;

			if (b.is_var[i])
			{
				const uint8_t *bytes =
					(const uint8_t *) cols[i];
				int64_t o0 = var_offs[i][r];
				uint32_t paylen = (uint32_t)
					(var_offs[i][r + 1] - o0);
				uint32_t attr_len;

				if (paylen + 1 <= 0x7F)
				{	/* short form, unaligned */
					start[vcur] = (uint8_t)
						((paylen + 1) | 0x80);
					for (uint32_t z = 0; z < paylen; z++)
						start[vcur + 1 + z] =
							bytes[o0 + z];
					attr_len = paylen + 1;
				}
				else
				{	/* 4-byte network-order header,
					 * aligned */
					vcur = mt_align(vcur,
							(int) b.align_of[i]);
					uint32_t h = (paylen + 4) &
						0x3FFFFFFFu;

					start[vcur] = (uint8_t) (h >> 24);
					start[vcur + 1] =
						(uint8_t) (h >> 16);
					start[vcur + 2] =
						(uint8_t) (h >> 8);
					start[vcur + 3] = (uint8_t) h;
					for (uint32_t z = 0; z < paylen; z++)
						start[vcur + 4 + z] =
							bytes[o0 + z];
					attr_len = paylen + 4;
				}
				/* varoffset relative to start (form_to
				 * :794): 2 bytes small, 4 bytes large */
				if (b.len[i] == 2)
				{
					uint16_t voff = (uint16_t) vcur;

					memcpy(dst, &voff, 2);
				}
				else
				{
					uint32_t voff = vcur;

					memcpy(dst, &voff, 4);
				}
				vcur += attr_len;
				continue;
			}
			switch (b.len[i])
			{
				case 1:
					*dst = ((const uint8_t *)
						cols[i])[r];
					break;
				case 2:
					memcpy(dst, (const int16_t *)
					       cols[i] + r, 2);
					break;
				case 4:
					memcpy(dst, (const int32_t *)
					       cols[i] + r, 4);
					break;
				default:
					memcpy(dst, (const int64_t *)
					       cols[i] + r, 8);
			}
		}
	}
}

__global__ void
k_mt_decode(MtBind bs, MtBind blg, const int64_t *__restrict__ offs,
	    int64_t nrows, const uint8_t *__restrict__ in, int64_t in_len,
	    void *const *__restrict__ cols,
	    unsigned long long *const *__restrict__ var_out_offs,
	    uint32_t *const *__restrict__ var_out_lens,
	    uint8_t *const *__restrict__ nulls,
	    unsigned long long *__restrict__ err)
{
	const int64_t stride = (int64_t) gridDim.x * blockDim.x;

	for (int64_t r = (int64_t) blockIdx.x * blockDim.x + threadIdx.x;
	     r < nrows; r += stride)
	{
		const uint8_t *tup = in + offs[r];
		uint32_t hdr;

		memcpy(&hdr, tup, 4);
		if (!(hdr & 0x80000000u))
		{
			atomicOr(err, 1ull);	/* not a memtuple */
			continue;
		}
		bool hasnull = (hdr & 1u) != 0;
		const MtBind &b = (hdr & 2u) ? blg : bs;	/* LARGETUP */
		const uint8_t *start = tup +
			(hasnull ? b.null_bitmap_extra : 0);
		const uint8_t *nullp = tup + 4;

		for (int i = 0; i < b.natts; i++)
		{
			bool isnull = hasnull &&
				(nullp[b.null_byte[i]] & b.null_mask[i]);

			if (nulls[i])
				nulls[i][r] = isnull ? 1 : 0;
			if (isnull)
			{
				if (b.is_var[i])
				{
					var_out_offs[i][r] = 0;
					var_out_lens[i][r] = 0;
				}
				else
					switch (b.len[i])
					{
						case 1:
							((uint8_t *)
							 cols[i])[r] = 0;
							break;
						case 2:
							((int16_t *)
							 cols[i])[r] = 0;
							break;
						case 4:
							((int32_t *)
							 cols[i])[r] = 0;
							break;
						default:
							((int64_t *)
							 cols[i])[r] = 0;
					}
				continue;
			}
			uint32_t save = 0;

			if (hasnull)
				for (int j = 0; j < b.natts; j++)
					if (b.phys[j] < b.phys[i] &&
					    (nullp[b.null_byte[j]] &
					     b.null_mask[j]))
						save += (uint32_t)
							b.len_aligned[j];
			const uint8_t *src = start + b.offset[i] - save;

			if (b.is_var[i])
			{
				/* varoffset relative to start
				 * (memtuple_get_attr_data_ptr :533) */
				uint32_t voff;

				if (b.len[i] == 2)
				{
					uint16_t v16;

					memcpy(&v16, src, 2);
					voff = v16;
				}
				else
					memcpy(&voff, src, 4);
				const uint8_t *vp = start + voff;
				uint32_t paylen;
				int64_t payoff;

				if (*vp & 0x80)
				{
					if (*vp == 0x80)
					{
						atomicOr(err, 2ull);
						continue;
					}
					paylen = (uint32_t) (*vp & 0x7F) - 1;
					payoff = (vp + 1) - in;
				}
				else
				{
					uint32_t h = ((uint32_t) vp[0] << 24) |
						((uint32_t) vp[1] << 16) |
						((uint32_t) vp[2] << 8) |
						(uint32_t) vp[3];

					if ((vp[0] & 0xC0) == 0x40)
					{
						atomicOr(err, 4ull);
						continue;
					}
					paylen = (h & 0x3FFFFFFFu) - 4;
					payoff = (vp + 4) - in;
				}
				if (payoff + paylen > (int64_t) in_len)
				{
					atomicOr(err, 8ull);
					continue;
				}
				/* pool IS the input stream: report the
				 * payload's position in it (zero-copy) */
				var_out_offs[i][r] =
					(unsigned long long) payoff;
				var_out_lens[i][r] = paylen;
				continue;
			}
			switch (b.len[i])
			{
				case 1:
					((uint8_t *) cols[i])[r] = *src;
					break;
				case 2:
					memcpy((int16_t *) cols[i] + r, src,
					       2);
					break;
				case 4:
					memcpy((int32_t *) cols[i] + r, src,
					       4);
					break;
				default:
					memcpy((int64_t *) cols[i] + r, src,
					       8);
			}
		}
	}
}

int
mt_grid(int64_t n)
{
	int64_t blk = (n + 255) / 256;

	if (blk > 2048)
		blk = 2048;
	if (blk < 1)
		blk = 1;
	return (int) blk;
}

}				/* anonymous namespace */

hipError_t
launch_mt_encode(hipStream_t s, const MtBind *b, const MtBind *bl,
		 const void *const *cols, const int64_t *const *var_offs,
		 const uint8_t *const *nulls, int64_t nrows,
		 const int64_t *offs, uint8_t *out)
{
	hipLaunchKernelGGL(k_mt_encode, dim3(mt_grid(nrows)), dim3(256), 0,
			   s, *b, *bl, cols, var_offs, nulls, nrows, offs,
			   out);
	return hipGetLastError();
}

hipError_t
launch_mt_decode(hipStream_t s, const MtBind *b, const MtBind *bl,
		 const int64_t *offs, int64_t nrows, const uint8_t *in,
		 int64_t in_len, void *const *cols,
		 unsigned long long *const *var_out_offs,
		 uint32_t *const *var_out_lens, uint8_t *const *nulls,
		 unsigned long long *err)
{
	hipLaunchKernelGGL(k_mt_decode, dim3(mt_grid(nrows)), dim3(256), 0,
			   s, *b, *bl, offs, nrows, in, in_len, cols,
			   var_out_offs, var_out_lens, nulls, err);
	return hipGetLastError();
}

}				/* namespace gg */
