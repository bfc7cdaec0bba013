/*
 * Minimal stand-in for the PostgreSQL/Greengage "postgres.h" environment,
 * just enough to compile the REFERENCE's own
 *   src/backend/access/hash/hashfunc.c
 * in place (see ../Makefile).  This is test infrastructure for the oracle:
 * it lets us link the reference's actual hash_any()/hashint4()/hashint8()
 * and pin our restatement (oracle/pg_hash.h) bit-for-bit against it.
 *
 * No reference source is copied here; only the ABI surface hashfunc.c
 * touches is re-declared (types + fmgr accessor macros, matching
 * src/include/postgres.h and src/include/fmgr.h of the reference).
 */
#ifndef ORACLE_STUB_POSTGRES_H
#define ORACLE_STUB_POSTGRES_H

#include <stdint.h>
#include <stddef.h>
#include <string.h>

typedef uintptr_t Datum;
typedef uint8_t uint8;
typedef uint16_t uint16;
typedef uint32_t uint32;
typedef uint64_t uint64;
typedef int8_t int8;
typedef int16_t int16;
typedef int32_t int32;
typedef int64_t int64;
typedef float float4;
typedef double float8;
typedef uint32 Oid;

/* little-endian x86-64 / aarch64 only — matches the build hosts here */
#undef WORDS_BIGENDIAN

#define Assert(x) ((void) 0)

/* ---- pass-by-value Datum conversions (64-bit build: float8 by value) ---- */
static inline Datum UInt32GetDatum(uint32 x) { return (Datum) x; }
static inline uint32 DatumGetUInt32(Datum d) { return (uint32) d; }

typedef union { float8 f; Datum d; } stub_f8_datum_u;
typedef union { struct { float4 f; uint32 pad; } s; Datum d; } stub_f4_datum_u;

/* ---- minimal fmgr: FunctionCallInfoData with an arg array ---- */
typedef struct FunctionCallInfoData
{
	Datum		arg[8];
	int		nargs;
} FunctionCallInfoData;
typedef FunctionCallInfoData *FunctionCallInfo;

#define PG_FUNCTION_ARGS FunctionCallInfo fcinfo

#define PG_GETARG_DATUM(n)   (fcinfo->arg[n])
#define PG_GETARG_CHAR(n)    ((char) fcinfo->arg[n])
#define PG_GETARG_INT16(n)   ((int16) fcinfo->arg[n])
#define PG_GETARG_INT32(n)   ((int32) fcinfo->arg[n])
#define PG_GETARG_INT64(n)   ((int64) fcinfo->arg[n])
#define PG_GETARG_OID(n)     ((Oid) fcinfo->arg[n])
#define PG_GETARG_POINTER(n) ((void *) fcinfo->arg[n])

static inline float4 stub_getarg_f4(FunctionCallInfo fcinfo, int n)
{ stub_f4_datum_u u; u.d = fcinfo->arg[n]; return u.s.f; }
static inline float8 stub_getarg_f8(FunctionCallInfo fcinfo, int n)
{ stub_f8_datum_u u; u.d = fcinfo->arg[n]; return u.f; }
#define PG_GETARG_FLOAT4(n)  stub_getarg_f4(fcinfo, n)
#define PG_GETARG_FLOAT8(n)  stub_getarg_f8(fcinfo, n)

#define PG_RETURN_UINT32(x)  return UInt32GetDatum(x)

/* ---- varlena (no TOAST in the stub: *_PP forms are plain pointers) ---- */
typedef struct varlena
{
	uint32		vl_len_;	/* 4-byte length header, untoasted */
	char		vl_dat[1];
} varlena;
typedef struct varlena text;

#define VARHDRSZ ((int32) sizeof(uint32))
#define PG_GETARG_TEXT_PP(n)    ((text *) PG_GETARG_POINTER(n))
#define PG_GETARG_VARLENA_PP(n) ((struct varlena *) PG_GETARG_POINTER(n))
#define VARDATA_ANY(v)        (((varlena *) (v))->vl_dat)
#define VARSIZE_ANY_EXHDR(v)  ((int32) (((varlena *) (v))->vl_len_) - VARHDRSZ)
#define PG_FREE_IF_COPY(ptr, n) ((void) 0)

/* ---- fixed-size array types used by hashoidvector/hashint2vector ---- */
typedef struct oidvector
{
	int32		dim1;
	Oid		values[1];
} oidvector;
typedef struct int2vector
{
	int32		dim1;
	int16		values[1];
} int2vector;

/* ---- Name ---- */
#define NAMEDATALEN 64
typedef struct nameData { char data[NAMEDATALEN]; } NameData;
typedef NameData *Name;
#define NameStr(name) ((name).data)
#define PG_GETARG_NAME(n) ((Name) PG_GETARG_POINTER(n))

#endif
