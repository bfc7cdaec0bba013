/*
 * Plain-C export wrapper around the REFERENCE's own datumstreamblock.c
 * (compiled in place — see Makefile): the genuine AOCS columnar block
 * encoder/decoder, used as the parity oracle for the engine's GPU
 * datumstream decoder.
 *
 * Streams are a simple framing of reference-encoded blocks:
 *   repeat: [int32 block_size][int32 row_count][block bytes...]
 * (the framing replaces the Append-Only Storage block headers, which
 * belong to cdbappendonlystorageformat.c — a separate layer; the datum
 * stream CONTENT bytes here are exactly what the reference writes).
 *
 * Fixed-length by-value types only (int32/int64 — the hot path's
 * column types); datumlen 4 or 8.
 */
#include "postgres.h"
#include "access/tupmacs.h"
#include "utils/datumstreamblock.h"

/* GUC definitions the compiled reference code references */
bool		Debug_appendonly_print_insert = false;
bool		Debug_appendonly_print_insert_tuple = false;
bool		Debug_appendonly_print_scan = false;
bool		Debug_appendonly_print_scan_tuple = false;
bool		Debug_datumstream_write_print_small_varlena_info = false;
bool		Debug_datumstream_write_print_large_varlena_info = false;
bool		Debug_datumstream_read_print_varlena_info = false;
bool		Debug_datumstream_write_use_small_initial_buffers = false;
bool		Debug_datumstream_block_read_check_integrity = true;
bool		Debug_datumstream_block_write_check_integrity = true;
bool		Debug_datumstream_read_check_large_varlena_integrity = false;
bool		Debug_datumstream_write_check_large_varlena_integrity = false;

MemoryContext CurrentMemoryContext = NULL;

int
stub_errmsg(const char *fmt,...)
{
	va_list		ap;

	fprintf(stderr, "[refmsg] ");
	va_start(ap, fmt);
	vfprintf(stderr, fmt, ap);
	va_end(ap);
	fprintf(stderr, "\n");
	return 0;
}

int
stub_errfinish(int level)
{
	if (level >= ERROR)
	{
		fprintf(stderr, "reference dsb code raised ERROR (level %d)\n",
				level);
		abort();
	}
	return 0;
}

/* varlena helpers — restated from the reference for the in-memory
 * (non-toasted, non-compressed) forms the wrapper produces:
 * varattrib_untoast_ptr_len (tuptoaster.c:179) and
 * value_type_could_short (tupmacs.h:202). */
void
varattrib_untoast_ptr_len(Datum d, char **datastart, int *len, void **tofree)
{
	char	   *p = (char *) DatumGetPointer(d);

	*tofree = NULL;
	if (VARATT_IS_EXTERNAL(p) || VARATT_IS_COMPRESSED(p))
	{
		fprintf(stderr, "varattrib_untoast_ptr_len: toasted datum "
			"unsupported in wrapper\n");
		abort();
	}
	if (VARATT_IS_SHORT(p))
	{
		*len = VARSIZE_SHORT(p) - VARHDRSZ_SHORT;
		*datastart = p + VARHDRSZ_SHORT;
	}
	else
	{
		*len = VARSIZE(p) - VARHDRSZ;
		*datastart = p + VARHDRSZ;
	}
}

bool
value_type_could_short(Pointer ptr, Oid typid)
{
	return !VARATT_IS_EXTERNAL(ptr) &&
		(VARATT_IS_SHORT(ptr) ||
		 (VARATT_CAN_MAKE_SHORT(ptr) &&
		  typid != 22 /* INT2VECTOROID */ &&
		  typid != 30 /* OIDVECTOROID */ &&
		  typid < 16384 /* FirstNormalObjectId */ ));
}

#define MAXDATUM_ORIG 0x3FFF	/* AOSmallContentHeader_MaxRowCount */
#define MAXDATUM_DENSE 0x3FFFFFFF	/* AONonBulkDense..MaxLargeRowCount */

static void
typeinfo_for(DatumStreamTypeInfo * ti, int datumlen)
{
	ti->datumlen = datumlen;
	ti->typid = (datumlen == 4) ? 23 : 20;	/* int4 / int8 oids */
	ti->align = (datumlen == 4) ? 'i' : 'd';
	ti->byval = true;
}

/*
 * Encode n values (vals[i] significant low datumlen bytes; nulls[i]
 * nonzero = NULL) into the framed stream.  version: 0 Orig, 1 Dense,
 * 2 Dense_Enhanced; rle/delta per DatumStreamVersion capabilities.
 * Returns 0, or -1 if out_cap too small.
 */
int
ref_dsb_encode(const int64 *vals, const uint8 *nulls, int64 n, int datumlen,
			   int version, int rle, int delta, int32 max_block_size,
			   uint8 *out, int64 out_cap, int64 *out_len,
			   int32 *out_nblocks)
{
	DatumStreamTypeInfo ti;
	DatumStreamBlockWrite dsw;
	int64		pos = 0;
	int32		nblocks = 0;
	int64		i = 0;

	memset(&dsw, 0, sizeof(dsw));
	typeinfo_for(&ti, datumlen);
	DatumStreamBlockWrite_Init(&dsw, &ti, (DatumStreamVersion) version,
							   rle != 0, delta != 0,
							   (version == 0) ? MAXDATUM_ORIG : MAXDATUM_ORIG,
							   (version == 0) ? MAXDATUM_ORIG : MAXDATUM_DENSE,
							   max_block_size,
							   NULL, NULL, NULL, NULL);
	DatumStreamBlockWrite_GetReady(&dsw);

	while (i < n)
	{
		Datum		d;
		bool		isnull = nulls && nulls[i];
		void	   *tofree = NULL;
		int			res;

		d = (datumlen == 4) ? Int32GetDatum((int32) vals[i])
			: Int64GetDatum(vals[i]);
		res = DatumStreamBlockWrite_Put(&dsw, isnull ? 0 : d, isnull,
										&tofree);
		if (res >= 0)
		{
			i++;
			continue;
		}
		/* block full: emit it */
		{
			int32		rowcount = DatumStreamBlockWrite_Nth(&dsw);
			int64		sz;

			if (pos + 8 + max_block_size > out_cap)
				return -1;
			sz = DatumStreamBlockWrite_Block(&dsw, out + pos + 8);
			memcpy(out + pos, &sz, 4);
			memcpy(out + pos + 4, &rowcount, 4);
			pos += 8 + sz;
			/* keep every block 8-aligned, as AO storage does
			 * (AOStorage_RoundUp8) — the reference's varlena walk
			 * is absolute-pointer-aligned */
			while (pos & 7)
				out[pos++] = 0;
			nblocks++;
			DatumStreamBlockWrite_GetReady(&dsw);
		}
	}
	if (DatumStreamBlockWrite_Nth(&dsw) > 0)
	{
		int32		rowcount = DatumStreamBlockWrite_Nth(&dsw);
		int64		sz;

		if (pos + 8 + max_block_size > out_cap)
			return -1;
		sz = DatumStreamBlockWrite_Block(&dsw, out + pos + 8);
		memcpy(out + pos, &sz, 4);
		memcpy(out + pos + 4, &rowcount, 4);
		pos += 8 + sz;
		while (pos & 7)
			out[pos++] = 0;
		nblocks++;
	}
	DatumStreamBlockWrite_Finish(&dsw);
	*out_len = pos;
	*out_nblocks = nblocks;
	return 0;
}

/*
 * Decode the framed stream back with the REFERENCE reader.
 * Returns 0, or -1 on capacity, -2 on row-count mismatch.
 */
int
ref_dsb_decode(const uint8 *stream, int64 stream_len, int datumlen,
			   int version, int rle, int64 *out_vals, uint8 *out_nulls,
			   int64 cap, int64 *out_n)
{
	DatumStreamTypeInfo ti;
	DatumStreamBlockRead dsr;
	int64		pos = 0;
	int64		nout = 0;

	memset(&dsr, 0, sizeof(dsr));
	typeinfo_for(&ti, datumlen);
	DatumStreamBlockRead_Init(&dsr, &ti, (DatumStreamVersion) version,
							  rle != 0, NULL, NULL, NULL, NULL);

	while (pos < stream_len)
	{
		int32		sz,
					rowcount;
		bool		adjusted = false;
		int32		adjustedRowCount = 0;
		int			have;

		memcpy(&sz, stream + pos, 4);
		memcpy(&rowcount, stream + pos + 4, 4);
		pos += 8;
		DatumStreamBlockRead_Reset(&dsr);	/* nth = -1 before GetReady */
		DatumStreamBlockRead_GetReady(&dsr, (uint8 *) stream + pos, sz,
									  1 /* firstRowNum */ , rowcount,
									  &adjusted, &adjustedRowCount);
		pos += sz;
		pos = (pos + 7) & ~(int64) 7;
		for (have = 0; have < rowcount; have++)
		{
			Datum		d = 0;
			bool		isnull = false;

			if (DatumStreamBlockRead_Advance(&dsr) == 0)
				return -2;
			DatumStreamBlockRead_Get(&dsr, &d, &isnull);
			if (nout >= cap)
				return -1;
			out_nulls[nout] = isnull ? 1 : 0;
			if (isnull)
				out_vals[nout] = 0;
			else
				out_vals[nout] = (datumlen == 4)
					? (int64) DatumGetInt32(d)
					: DatumGetInt64(d);
			nout++;
		}
	}
	DatumStreamBlockRead_Finish(&dsr);
	*out_n = nout;
	return 0;
}

/* ---- varlena (text) variants: typid 25, align 'i', byval false ---- */

#define REF_VARDATA_ANY(PTR) \
	(VARATT_IS_1B(PTR) ? ((char *) (PTR) + 1) : ((char *) (PTR) + 4))
#define REF_VARSIZE_ANY_EXHDR(PTR) \
	(VARATT_IS_1B(PTR) ? VARSIZE_1B(PTR) - 1 : VARSIZE_4B(PTR) - 4)

static void
typeinfo_text(DatumStreamTypeInfo * ti)
{
	ti->datumlen = -1;
	ti->typid = 25;		/* TEXTOID */
	ti->align = 'i';
	ti->byval = false;
}

/*
 * Encode n text values (bytes[offs[i]..offs[i+1]) is row i's payload;
 * nulls[i] nonzero = NULL) with the REFERENCE writer.  Same framing as
 * ref_dsb_encode.
 */
int
ref_dsb_encode_text(const uint8 *bytes, const int64 *offs,
		    const uint8 *nulls, int64 n, int version, int rle,
		    int32 max_block_size, uint8 *out, int64 out_cap,
		    int64 *out_len, int32 *out_nblocks)
{
	DatumStreamTypeInfo ti;
	DatumStreamBlockWrite dsw;
	int64		pos = 0;
	int32		nblocks = 0;
	int64		i = 0;
	static uint8 vbuf[1 << 20];

	memset(&dsw, 0, sizeof(dsw));
	typeinfo_text(&ti);
	DatumStreamBlockWrite_Init(&dsw, &ti, (DatumStreamVersion) version,
				   rle != 0, false /* no delta for text */ ,
				   MAXDATUM_ORIG,
				   (version == 0) ? MAXDATUM_ORIG
				   : MAXDATUM_DENSE,
				   max_block_size, NULL, NULL, NULL, NULL);
	DatumStreamBlockWrite_GetReady(&dsw);

	while (i < n)
	{
		Datum		d = 0;
		bool		isnull = nulls && nulls[i];
		void	   *tofree = NULL;
		int			res;
		int64		len = offs[i + 1] - offs[i];

		if (!isnull)
		{
			if (len + 4 > (int64) sizeof(vbuf))
				return -3;
			/* standard 4-byte-header varlena; the writer itself
			 * converts short candidates (datumstreamblock.c:1646) */
			SET_VARSIZE(vbuf, len + 4);
			memcpy(vbuf + 4, bytes + offs[i], len);
			d = PointerGetDatum(vbuf);
		}
		res = DatumStreamBlockWrite_Put(&dsw, d, isnull, &tofree);
		if (res >= 0)
		{
			i++;
			continue;
		}
		{
			int32		rowcount = DatumStreamBlockWrite_Nth(&dsw);
			int64		sz;

			if (rowcount == 0)
				return -4;	/* datum larger than a block */
			if (pos + 8 + max_block_size > out_cap)
				return -1;
			sz = DatumStreamBlockWrite_Block(&dsw, out + pos + 8);
			memcpy(out + pos, &sz, 4);
			memcpy(out + pos + 4, &rowcount, 4);
			pos += 8 + sz;
			/* keep every block 8-aligned, as AO storage does
			 * (AOStorage_RoundUp8) — the reference's varlena walk
			 * is absolute-pointer-aligned */
			while (pos & 7)
				out[pos++] = 0;
			nblocks++;
			DatumStreamBlockWrite_GetReady(&dsw);
		}
	}
	if (DatumStreamBlockWrite_Nth(&dsw) > 0)
	{
		int32		rowcount = DatumStreamBlockWrite_Nth(&dsw);
		int64		sz;

		if (pos + 8 + max_block_size > out_cap)
			return -1;
		sz = DatumStreamBlockWrite_Block(&dsw, out + pos + 8);
		memcpy(out + pos, &sz, 4);
		memcpy(out + pos + 4, &rowcount, 4);
		pos += 8 + sz;
		while (pos & 7)
			out[pos++] = 0;
		nblocks++;
	}
	DatumStreamBlockWrite_Finish(&dsw);
	*out_len = pos;
	*out_nblocks = nblocks;
	return 0;
}

/* Decode text back with the REFERENCE reader: writes payload bytes to
 * out_bytes, row offsets to out_offs[0..n], null flags to out_nulls. */
int
ref_dsb_decode_text(const uint8 *stream, int64 stream_len, int version,
		    int rle, uint8 *out_bytes, int64 bytes_cap,
		    int64 *out_offs, uint8 *out_nulls, int64 cap,
		    int64 *out_n)
{
	DatumStreamTypeInfo ti;
	DatumStreamBlockRead dsr;
	int64		pos = 0;
	int64		nout = 0;
	int64		bpos = 0;

	memset(&dsr, 0, sizeof(dsr));
	typeinfo_text(&ti);
	DatumStreamBlockRead_Init(&dsr, &ti, (DatumStreamVersion) version,
				  rle != 0, NULL, NULL, NULL, NULL);

	out_offs[0] = 0;
	while (pos < stream_len)
	{
		int32		sz,
					rowcount;
		bool		adjusted = false;
		int32		adjustedRowCount = 0;
		int			have;

		memcpy(&sz, stream + pos, 4);
		memcpy(&rowcount, stream + pos + 4, 4);
		pos += 8;
		DatumStreamBlockRead_Reset(&dsr);
		DatumStreamBlockRead_GetReady(&dsr, (uint8 *) stream + pos, sz,
					      1, rowcount, &adjusted,
					      &adjustedRowCount);
		pos += sz;
		pos = (pos + 7) & ~(int64) 7;
		for (have = 0; have < rowcount; have++)
		{
			Datum		d = 0;
			bool		isnull = false;

			if (DatumStreamBlockRead_Advance(&dsr) == 0)
				return -2;
			DatumStreamBlockRead_Get(&dsr, &d, &isnull);
			if (nout >= cap)
				return -1;
			out_nulls[nout] = isnull ? 1 : 0;
			if (!isnull)
			{
				char	   *vp = (char *) DatumGetPointer(d);
				int64		len = REF_VARSIZE_ANY_EXHDR(vp);

				if (bpos + len > bytes_cap)
					return -1;
				memcpy(out_bytes + bpos, REF_VARDATA_ANY(vp),
				       len);
				bpos += len;
			}
			nout++;
			out_offs[nout] = bpos;
		}
	}
	DatumStreamBlockRead_Finish(&dsr);
	*out_n = nout;
	return 0;
}

/* ------------------------------------------------------------------ */
/* Append-Only storage block layer (cdbappendonlystorageformat.c,      */
/* compiled in place): wrap datum-stream content into REAL AO blocks   */
/* with block/header CRC32C checksums.                                 */
/* ------------------------------------------------------------------ */
bool		Debug_appendonly_print_storage_headers = false;

#include "cdb/cdbappendonlystorage_int.h"
#include "cdb/cdbappendonlystorage.h"
#include "cdb/cdbappendonlystorageformat.h"

#define REF_AO_VERSION 2	/* AORelationVersion_Aligned64bit */
#define REF_AO_EXECKIND 1	/* AOCSBK_BLOCK */

static int32 ao_roundup8(int32 l) { return ((l + 7) / 8) * 8; }

/*
 * Convert a [int32 size][int32 rowcount][content] framed stream (from
 * ref_dsb_encode) into a stream of real AO storage blocks:
 *   [8B header][block crc32c][header crc32c][firstRowNum(8B, optional)]
 *   [content, zero-padded to 8B]
 * Small-content headers when rowcount fits 14 bits, NonBulkDense
 * otherwise — the same choice datumstream.c:944 makes.
 */
int
ref_ao_wrap_stream(const uint8 *framed, int64 framed_len, int checksums,
		   int has_firstrownum, uint8 *out, int64 out_cap,
		   int64 *out_len)
{
	int64		pos = 0,
				opos = 0;
	int64		firstRowNum = 1;

	while (pos < framed_len)
	{
		int32		sz,
					rowcount;
		int32		hdrlen,
					padded,
					overall;
		uint8	   *hdr;

		memcpy(&sz, framed + pos, 4);
		memcpy(&rowcount, framed + pos + 4, 4);
		pos += 8;
		hdrlen = (checksums ? 16 : 8) + (has_firstrownum ? 8 : 0);
		padded = ao_roundup8(sz);
		overall = hdrlen + padded;
		if (opos + overall > out_cap)
			return -1;
		hdr = out + opos;
		memset(hdr, 0, overall);
		memcpy(hdr + hdrlen, framed + pos, sz);
		if (rowcount <= 0x3FFF)
			AppendOnlyStorageFormat_MakeSmallContentHeader(
				hdr, checksums != 0, has_firstrownum != 0,
				REF_AO_VERSION, firstRowNum, REF_AO_EXECKIND,
				rowcount, sz, /* compressedLength */ 0);
		else
			AppendOnlyStorageFormat_MakeNonBulkDenseContentHeader(
				hdr, checksums != 0, has_firstrownum != 0,
				REF_AO_VERSION, firstRowNum, REF_AO_EXECKIND,
				rowcount, sz);
		firstRowNum += rowcount;
		pos += sz;
		pos = (pos + 7) & ~(int64) 7;
		opos += overall;
	}
	*out_len = opos;
	return 0;
}

/*
 * Same, with bulk compression: the reference's compressed-block write
 * rule (AppendOnlyStorageWrite_CompressAppend,
 * cdbappendonlystoragewrite.c:1134–1230): content is stored compressed
 * only when strictly smaller, dataLength = uncompressed size,
 * compressedLength = stored size (0 when kept uncompressed), pad to
 * RoundUp8 of what is stored.  Codecs are the ones the reference binds
 * (pg_compression.c:253 — zlib compress2; gpcontrib zstd_compression.c
 * :117 — ZSTD plain).  Compression rides the SmallContent header only
 * (NonBulkDense has no compressedLength field).
 */
#include <zlib.h>
/* libzstd.so.1 is present without dev headers; public stable API: */
extern size_t ZSTD_compress(void *dst, size_t dstCap, const void *src,
			    size_t srcSize, int level);
extern unsigned ZSTD_isError(size_t code);

int
ref_ao_wrap_stream_c(const uint8 *framed, int64 framed_len, int checksums,
		     int has_firstrownum, int comptype, int complevel,
		     uint8 *out, int64 out_cap, int64 *out_len)
{
	int64		pos = 0,
				opos = 0;
	int64		firstRowNum = 1;
	static uint8 cbuf[4 * 1024 * 1024];

	while (pos < framed_len)
	{
		int32		sz,
					rowcount;
		int32		hdrlen,
					stored,
					complen,
					padded,
					overall;
		uint8	   *hdr;
		const uint8 *content;

		memcpy(&sz, framed + pos, 4);
		memcpy(&rowcount, framed + pos + 4, 4);
		pos += 8;
		if (rowcount > 0x3FFF)
			return -2;	/* compression => SmallContent only */

		complen = 0;
		content = framed + pos;
		if (comptype == 1)
		{
			uLongf		dl = sizeof(cbuf);

			if (compress2(cbuf, &dl, framed + pos, sz,
				      complevel) != Z_OK)
				return -3;
			complen = (int32) dl;
		}
		else if (comptype == 2)
		{
			size_t		dl = ZSTD_compress(cbuf, sizeof(cbuf),
						   framed + pos, sz, complevel);

			if (ZSTD_isError(dl))
				return -3;
			complen = (int32) dl;
		}
		else if (comptype != 0)
			return -4;
		if (complen == 0 || complen >= sz)
		{
			stored = sz;
			complen = 0;
			content = framed + pos;
		}
		else
		{
			stored = complen;
			content = cbuf;
		}

		hdrlen = (checksums ? 16 : 8) + (has_firstrownum ? 8 : 0);
		padded = ao_roundup8(stored);
		overall = hdrlen + padded;
		if (opos + overall > out_cap)
			return -1;
		hdr = out + opos;
		memset(hdr, 0, overall);
		memcpy(hdr + hdrlen, content, stored);
		AppendOnlyStorageFormat_MakeSmallContentHeader(
			hdr, checksums != 0, has_firstrownum != 0,
			REF_AO_VERSION, firstRowNum, REF_AO_EXECKIND,
			rowcount, sz, complen);
		firstRowNum += rowcount;
		pos += sz;
		pos = (pos + 7) & ~(int64) 7;
		opos += overall;
	}
	*out_len = opos;
	return 0;
}

/* ------------------------------------------------------------------ */
/* MemTuple layer (access/common/memtuple.c compiled in place): GPDB's */
/* compact tuple format used in executor hash tables and on the Motion */
/* wire (format comment memtuple.c:24-35).  The wrapper exposes the    */
/* REAL binding/form/getattr so the engine's restated codec can be     */
/* pinned bit-exactly.                                                 */
/* ------------------------------------------------------------------ */
#include "access/memtup.h"

/* only reached for external toast pointers — never from the wrapper */
struct varlena *
heap_tuple_fetch_attr(struct varlena *attr)
{
	fprintf(stderr, "heap_tuple_fetch_attr: unsupported in wrapper\n");
	abort();
}

void *
ref_mt_create_binding(int natts, const int32 *attlen,
		      const uint8 *attbyval, const char *attalign,
		      const uint32 *atttypid)
{
	TupleDesc	desc = (TupleDesc) calloc(1, sizeof(*desc));
	int			i;

	desc->natts = natts;
	desc->tdhasoid = false;
	desc->attrs = (Form_pg_attribute *)
		calloc(natts, sizeof(Form_pg_attribute));
	for (i = 0; i < natts; i++)
	{
		Form_pg_attribute a = (Form_pg_attribute)
			calloc(1, sizeof(*a));

		a->atttypid = atttypid[i];
		a->attlen = (int16) attlen[i];
		a->attbyval = attbyval[i] != 0;
		a->attalign = attalign[i];
		a->attisdropped = false;
		a->attstorage = attlen[i] > 0 ? 'p' : 'x';
		a->atttypmod = -1;
		desc->attrs[i] = a;
	}
	return create_memtuple_binding(desc);
}

int32
ref_mt_get_colbind(void *pbind, int attnum_1based, int use_large,
		   int32 *offset, int32 *len, int32 *len_aligned,
		   int32 *flag, int32 *null_byte, int32 *null_mask)
{
	MemTupleBinding *b = (MemTupleBinding *) pbind;
	MemTupleBindingCols *cols = use_large ? &b->large_bind : &b->bind;
	MemTupleAttrBinding *ab;

	if (attnum_1based < 1 || attnum_1based > b->tupdesc->natts)
		return -1;
	ab = &cols->bindings[attnum_1based - 1];
	*offset = ab->offset;
	*len = ab->len;
	*len_aligned = ab->len_aligned;
	*flag = (int32) ab->flag;
	*null_byte = ab->null_byte;
	*null_mask = ab->null_mask;
	return 0;
}

int32
ref_mt_binding_info(void *pbind, int32 *column_align,
		    int32 *null_bitmap_extra, int32 *var_start)
{
	MemTupleBinding *b = (MemTupleBinding *) pbind;

	*column_align = b->column_align;
	*null_bitmap_extra = b->null_bitmap_extra_size;
	*var_start = (int32) b->bind.var_start;
	return 0;
}

/* form one tuple from int64 datums; returns total length, or -1 */
int32
ref_mt_form(void *pbind, const int64 *values, const uint8 *isnull,
	    uint8 *out, int32 cap)
{
	MemTupleBinding *b = (MemTupleBinding *) pbind;
	int			n = b->tupdesc->natts;
	Datum		dvals[64];
	bool		dnull[64];
	uint32		destlen = 0;
	int			i;
	MemTuple	res;

	if (n > 64)
		return -1;
	for (i = 0; i < n; i++)
	{
		dvals[i] = (Datum) values[i];
		dnull[i] = isnull[i] != 0;
	}
	/* first call computes required length */
	memtuple_form_to(b, dvals, dnull, NULL, &destlen, false);
	if ((int32) destlen > cap)
		return -2;
	memset(out, 0, destlen);
	res = memtuple_form_to(b, dvals, dnull, (MemTuple) out, &destlen,
			       false);
	if (res == NULL)
		return -3;
	return (int32) destlen;
}

/* form with mixed fixed/text attrs: for is_text[i] != 0, values[i] is
 * the ROW INDEX into (bytes, offs) — a 4-byte-header varlena is built
 * and passed; the writer converts to short form when it can. */
int32
ref_mt_form_var(void *pbind, const int64 *values, const uint8 *is_text,
		const uint8 *bytes, const int64 *offs, const uint8 *isnull,
		uint8 *out, int32 cap)
{
	MemTupleBinding *b = (MemTupleBinding *) pbind;
	int			n = b->tupdesc->natts;
	Datum		dvals[64];
	bool		dnull[64];
	uint32		destlen = 0;
	int			i;
	MemTuple	res;
	static uint8 scratch[64][1 << 16];

	if (n > 64)
		return -1;
	for (i = 0; i < n; i++)
	{
		dnull[i] = isnull[i] != 0;
		if (dnull[i])
		{
			dvals[i] = 0;
			continue;
		}
		if (is_text[i])
		{
			int64		row = values[i];
			int64		len = offs[row + 1] - offs[row];

			if (len + 4 > (int64) sizeof(scratch[0]))
				return -5;
			SET_VARSIZE(scratch[i], len + 4);
			memcpy(scratch[i] + 4, bytes + offs[row], len);
			dvals[i] = PointerGetDatum(scratch[i]);
		}
		else
			dvals[i] = (Datum) values[i];
	}
	memtuple_form_to(b, dvals, dnull, NULL, &destlen, false);
	if ((int32) destlen > cap)
		return -2;
	memset(out, 0, destlen);
	res = memtuple_form_to(b, dvals, dnull, (MemTuple) out, &destlen,
			       false);
	if (res == NULL)
		return -3;
	return (int32) destlen;
}

int32
ref_mt_getattr(void *pbind, uint8 *tup, int attnum_1based, int64 *val,
	       uint8 *out_isnull)
{
	MemTupleBinding *b = (MemTupleBinding *) pbind;
	bool		isnull = false;
	Datum		d;

	d = memtuple_getattr((MemTuple) tup, b, attnum_1based, &isnull);
	*val = (int64) d;
	*out_isnull = isnull ? 1 : 0;
	return 0;
}

/*
 * BulkDense variant: the header the reference writes for RLE dense
 * content with bulk compression (AoHeaderKind_BulkDenseContent = 4,
 * LONG header: [8B][crc][crc][8B ext w/ largeRowCount][firstRowNum]).
 * Every frame gets a BulkDense header; comptype 0 stores uncompressed.
 */
int
ref_ao_wrap_stream_bd(const uint8 *framed, int64 framed_len, int checksums,
		      int has_firstrownum, int comptype, int complevel,
		      uint8 *out, int64 out_cap, int64 *out_len)
{
	int64		pos = 0,
				opos = 0;
	int64		firstRowNum = 1;
	static uint8 cbuf[4 * 1024 * 1024];

	while (pos < framed_len)
	{
		int32		sz,
					rowcount;
		int32		hdrlen,
					stored,
					complen,
					padded,
					overall;
		uint8	   *hdr;
		const uint8 *content;

		memcpy(&sz, framed + pos, 4);
		memcpy(&rowcount, framed + pos + 4, 4);
		pos += 8;

		complen = 0;
		content = framed + pos;
		if (comptype == 1)
		{
			uLongf		dl = sizeof(cbuf);

			if (compress2(cbuf, &dl, framed + pos, sz,
				      complevel) != Z_OK)
				return -3;
			complen = (int32) dl;
		}
		else if (comptype == 2)
		{
			size_t		dl = ZSTD_compress(cbuf, sizeof(cbuf),
						   framed + pos, sz, complevel);

			if (ZSTD_isError(dl))
				return -3;
			complen = (int32) dl;
		}
		else if (comptype != 0)
			return -4;
		if (complen == 0 || complen >= sz)
		{
			stored = sz;
			complen = 0;
			content = framed + pos;
		}
		else
		{
			stored = complen;
			content = cbuf;
		}

		hdrlen = 8 + (checksums ? 8 : 0) + 8 /* ext */ +
			(has_firstrownum ? 8 : 0);
		padded = ao_roundup8(stored);
		overall = hdrlen + padded;
		if (opos + overall > out_cap)
			return -1;
		hdr = out + opos;
		memset(hdr, 0, overall);
		memcpy(hdr + hdrlen, content, stored);
		AppendOnlyStorageFormat_MakeBulkDenseContentHeader(
			hdr, checksums != 0, has_firstrownum != 0,
			REF_AO_VERSION, firstRowNum, REF_AO_EXECKIND,
			rowcount, sz, complen);
		firstRowNum += rowcount;
		pos += sz;
		pos = (pos + 7) & ~(int64) 7;
		opos += overall;
	}
	*out_len = opos;
	return 0;
}

/*
 * LargeContent form: each frame becomes a header-only LargeContent
 * metadata block (kind 2: largeRowCount + largeContentLength,
 * firstRowNum) followed by SmallContent fragment blocks with
 * rowCount=0 and no firstRowNum — exactly the
 * AppendOnlyStorageWrite_LargeContent layout
 * (cdbappendonlystoragewrite.c:1600-1720).  frag_size bounds each
 * fragment's uncompressed content; comptype compresses per fragment.
 */
int
ref_ao_wrap_stream_large(const uint8 *framed, int64 framed_len,
			 int checksums, int comptype, int complevel,
			 int32 frag_size, uint8 *out, int64 out_cap,
			 int64 *out_len)
{
	int64		pos = 0,
				opos = 0;
	int64		firstRowNum = 1;
	static uint8 cbuf[4 * 1024 * 1024];

	while (pos < framed_len)
	{
		int32		sz,
					rowcount;
		int32		hdrlen,
					remaining;
		const uint8 *content;
		uint8	   *hdr;

		memcpy(&sz, framed + pos, 4);
		memcpy(&rowcount, framed + pos + 4, 4);
		pos += 8;
		content = framed + pos;

		/* metadata block: header only (+FRN), no content */
		hdrlen = 8 + (checksums ? 8 : 0) + 8 /* firstRowNum */ ;
		if (opos + hdrlen > out_cap)
			return -1;
		hdr = out + opos;
		memset(hdr, 0, hdrlen);
		AppendOnlyStorageFormat_MakeLargeContentHeader(
			hdr, checksums != 0, true, REF_AO_VERSION, firstRowNum,
			REF_AO_EXECKIND, rowcount, sz);
		firstRowNum += rowcount;
		opos += hdrlen;

		/* fragments */
		remaining = sz;
		while (remaining > 0)
		{
			int32		flen = remaining < frag_size
				? remaining : frag_size;
			int32		complen = 0,
						stored,
						fh,
						padded;
			const uint8 *src = content + (sz - remaining);
			const uint8 *body = src;

			if (comptype == 1)
			{
				uLongf		dl = sizeof(cbuf);

				if (compress2(cbuf, &dl, src, flen,
					      complevel) != Z_OK)
					return -3;
				complen = (int32) dl;
			}
			else if (comptype == 2)
			{
				size_t		dl = ZSTD_compress(cbuf,
							   sizeof(cbuf), src, flen,
							   complevel);

				if (ZSTD_isError(dl))
					return -3;
				complen = (int32) dl;
			}
			else if (comptype != 0)
				return -4;
			if (complen == 0 || complen >= flen)
			{
				stored = flen;
				complen = 0;
				body = src;
			}
			else
			{
				stored = complen;
				body = cbuf;
			}
			fh = 8 + (checksums ? 8 : 0);	/* no firstRowNum */
			padded = ao_roundup8(stored);
			if (opos + fh + padded > out_cap)
				return -1;
			hdr = out + opos;
			memset(hdr, 0, fh + padded);
			memcpy(hdr + fh, body, stored);
			AppendOnlyStorageFormat_MakeSmallContentHeader(
				hdr, checksums != 0, false /* no FRN */ ,
				REF_AO_VERSION, 0, REF_AO_EXECKIND,
				0 /* rowCount */ , flen, complen);
			opos += fh + padded;
			remaining -= flen;
		}
		pos += sz;
		pos = (pos + 7) & ~(int64) 7;
	}
	*out_len = opos;
	return 0;
}

/* reference-side parse of one AO block (for cross-checking the
 * engine's restated parser in tests) */
int
ref_ao_probe_block(uint8 *block, int checksums, int32 *kind,
		   int32 *rowcount, int32 *datalen, int32 *content_off,
		   int32 *overall, int32 *cksum_ok)
{
	AoHeaderKind hk;
	int32		actualHeaderLen;
	AOHeaderCheckError err;

	err = AppendOnlyStorageFormat_GetHeaderInfo(block, checksums != 0,
						    &hk, &actualHeaderLen);
	if (err != AOHeaderCheckOk)
		return -1;
	*kind = (int32) hk;
	*cksum_ok = 1;
	if (checksums)
	{
		pg_crc32	stored,
					computed;

		if (!AppendOnlyStorageFormat_VerifyHeaderChecksum(
			block, &stored, &computed))
			*cksum_ok = 0;
	}
	{
		int32		offset = 0,
					uncompressedLen = 0,
					compressedLen = 0;
		int			execKind = 0,
					rc = 0;
		bool		hasFRN = false,
					isCompressed = false;
		int64		frn = 0;

		if (hk == AoHeaderKind_SmallContent)
			err = AppendOnlyStorageFormat_GetSmallContentHeaderInfo(
				block, actualHeaderLen, checksums != 0,
				1 << 30, overall, &offset, &uncompressedLen,
				&execKind, &hasFRN, REF_AO_VERSION, &frn, &rc,
				&isCompressed, &compressedLen);
		else
			err = AppendOnlyStorageFormat_GetNonBulkDenseContentHeaderInfo(
				block, actualHeaderLen, checksums != 0,
				1 << 30, overall, &offset, &uncompressedLen,
				&execKind, &hasFRN, REF_AO_VERSION, &frn, &rc);
		if (err != AOHeaderCheckOk)
			return -2;
		*rowcount = rc;
		*datalen = uncompressedLen;
		*content_off = offset;
	}
	if (checksums && *cksum_ok)
	{
		pg_crc32	stored,
					computed;

		if (!AppendOnlyStorageFormat_VerifyBlockChecksum(
			block, *overall, &stored, &computed))
			*cksum_ok = 0;
	}
	return 0;
}
