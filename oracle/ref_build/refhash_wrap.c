/*
 * Plain-C export wrapper around the REFERENCE's own hashfunc.c
 * (compiled in place from /root/reference — see Makefile).  Gives the
 * oracle's tests direct, fmgr-free entry points to the reference's
 * hash implementations so the restatement in oracle/pg_hash.h can be
 * pinned bit-for-bit.
 */
#include "postgres.h"
#include "access/hash.h"

uint32
ref_hash_any(const unsigned char *k, int keylen)
{
	return DatumGetUInt32(hash_any(k, keylen));
}

uint32
ref_hash_uint32(uint32 k)
{
	return DatumGetUInt32(hash_uint32(k));
}

uint32
ref_hashint4(int32 v)
{
	FunctionCallInfoData fcinfo;

	fcinfo.arg[0] = (Datum) (uintptr_t) (uint32) v;	/* Int32GetDatum */
	fcinfo.nargs = 1;
	return DatumGetUInt32(hashint4(&fcinfo));
}

uint32
ref_hashint8(int64 v)
{
	FunctionCallInfoData fcinfo;

	fcinfo.arg[0] = (Datum) (uint64) v;	/* Int64GetDatum, by value */
	fcinfo.nargs = 1;
	return DatumGetUInt32(hashint8(&fcinfo));
}

uint32
ref_hashchar(char c)
{
	FunctionCallInfoData fcinfo;

	fcinfo.arg[0] = (Datum) (uintptr_t) (unsigned char) c;
	fcinfo.nargs = 1;
	return DatumGetUInt32(hashchar(&fcinfo));
}
