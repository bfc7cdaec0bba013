/*
 * Stand-in "postgres.h" environment for compiling the REFERENCE's own
 * src/backend/utils/datumstream/datumstreamblock.c in place (see
 * ../Makefile) — the genuine AOCS block encoder/decoder becomes the
 * parity oracle for the engine's GPU datumstream decoder.
 * Only the ABI surface that file touches is re-declared; semantics of
 * each macro follow the PostgreSQL 9.4 originals the reference builds
 * against.  Little-endian hosts only.
 */
#ifndef ORACLE_STUB_DSB_POSTGRES_H
#define ORACLE_STUB_DSB_POSTGRES_H

#include <stdint.h>
#include <stddef.h>
#include <string.h>
#include <stdio.h>
#include <stdarg.h>
#include <stdlib.h>
#include <stdlib.h>
#include <stdbool.h>

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
typedef char *Pointer;

#define InvalidOid ((Oid) 0)
#define Assert(x) do { if (!(x)) { fprintf(stderr, "Assert failed: %s (%s:%d)\n", #x, __FILE__, __LINE__); abort(); } } while (0)
#define AssertImply(a, b) Assert(!(a) || (b))
#define StaticAssertStmt(cond, msg) ((void) 0)
#define MemSet(start, val, len) memset(start, val, len)
#define lengthof(array) (sizeof(array) / sizeof((array)[0]))
#define pg_attribute_always_inline inline
#define PGDLLIMPORT
#define INT64_FORMAT "%ld"

/* ---- error reporting: print + abort on >= ERROR ---- */
#define DEBUG5 10
#define DEBUG4 11
#define DEBUG3 12
#define DEBUG2 13
#define DEBUG1 14
#define LOG 15
#define INFO 17
#define NOTICE 18
#define WARNING 19
#define ERROR 20
#define FATAL 21
#define PANIC 22

extern int stub_errmsg(const char *fmt, ...);
extern int stub_errfinish(int level);
#define elog(level, ...) \
	do { \
		if ((level) >= WARNING) \
			{ fprintf(stderr, "elog(%d): ", level); fprintf(stderr, __VA_ARGS__); fprintf(stderr, "\n"); } \
		if ((level) >= ERROR) abort(); \
	} while (0)
#define ereport(level, rest) \
	do { \
		(void) (rest); \
		(void) stub_errfinish(level); \
	} while (0)
#define errmsg stub_errmsg
#define errmsg_internal stub_errmsg
#define errdetail stub_errmsg
#define errdetail_internal stub_errmsg
#define errcontext stub_errmsg
#define errhint stub_errmsg
#define errcode(c) (c)
#define errOmitLocation(b) 0
#define ERRCODE_INTERNAL_ERROR 0
#define ERRCODE_GP_INTERNAL_ERROR 0
#define ERRCODE_APPENDONLY_INTERNAL_ERROR 0
#define ERRCODE_DATA_CORRUPTED 0

/* ---- memory: malloc-backed palloc; contexts are inert tokens ---- */
typedef struct MemoryContextData *MemoryContext;
extern MemoryContext CurrentMemoryContext;
static inline MemoryContext MemoryContextSwitchTo(MemoryContext ctx)
{
	MemoryContext old = CurrentMemoryContext;
	CurrentMemoryContext = ctx;
	return old;
}
static inline void *palloc(size_t sz) { return malloc(sz); }
static inline void *palloc0(size_t sz) { return calloc(1, sz); }
static inline void pfree(void *p) { free(p); }

/* ---- Datum conversions (64-bit, by-value; postgres.h originals) ---- */
#define DatumGetPointer(X) ((Pointer) (X))
#define PointerGetDatum(X) ((Datum) (X))
#define DatumGetUInt8(X) ((uint8) (X))
#define DatumGetUInt16(X) ((uint16) (X))
#define DatumGetUInt32(X) ((uint32) (X))
#define DatumGetInt32(X) ((int32) (X))
#define DatumGetInt64(X) ((int64) (X))
#define DatumGetChar(X) ((char) (X))
#define DatumGetCString(X) ((char *) (X))
#define Int32GetDatum(X) ((Datum) (uint32) (X))
#define Int16GetDatum(X) ((Datum) (uint16) (X))
#define DatumGetInt16(X) ((int16) (X))
#define CharGetDatum(X) ((Datum) (uint8) (X))
#define Int64GetDatum(X) ((Datum) (X))
#define UInt32GetDatum(X) ((Datum) (X))

/* ---- alignment (c.h originals) ---- */
#define TYPEALIGN(ALIGNVAL, LEN) \
	(((uintptr_t) (LEN) + ((ALIGNVAL) - 1)) & ~((uintptr_t) ((ALIGNVAL) - 1)))
#define SHORTALIGN(LEN) TYPEALIGN(2, (LEN))
#define INTALIGN(LEN) TYPEALIGN(4, (LEN))
#define LONGALIGN(LEN) TYPEALIGN(8, (LEN))
#define DOUBLEALIGN(LEN) TYPEALIGN(8, (LEN))
#define MAXIMUM_ALIGNOF 8
#define MAXALIGN(LEN) TYPEALIGN(MAXIMUM_ALIGNOF, (LEN))

/* ---- varlena — THIS FORK'S layout (reference postgres.h:160–300):
 * 4-byte headers stored in NETWORK byte order ("Greengage stored the
 * 4 byte varlena header in network byte order" — postgres.h:169),
 * flag bits in the physically first byte: 00=4B uncompressed,
 * 01=4B compressed, 10000000=toast pointer, 1xxxxxxx=short (length
 * includes the header byte, no shift). ---- */
#include <arpa/inet.h>
typedef struct varlena
{
	char		vl_len_[4];
	char		vl_dat[1];
} varlena;
#define VARHDRSZ ((int32) sizeof(int32))
#define VARHDRSZ_SHORT 1

typedef union
{
	struct
	{
		uint32		va_header;
		char		va_data[1];
	}			va_4byte;
	struct
	{
		uint32		va_header;
		uint32		va_rawsize;
		char		va_data[1];
	}			va_compressed;
} varattrib_4b;
typedef varattrib_4b varattrib_4b_stub;
typedef struct
{
	uint8		va_header;
	char		va_data[1];
} varattrib_1b_stub;
typedef varattrib_1b_stub varattrib_1b;

#define VARATT_IS_4B(PTR) \
	((((varattrib_1b_stub *) (PTR))->va_header & 0x80) == 0x00)
#define VARATT_IS_4B_U(PTR) \
	((((varattrib_1b_stub *) (PTR))->va_header & 0xC0) == 0x00)
#define VARATT_IS_4B_C(PTR) \
	((((varattrib_1b_stub *) (PTR))->va_header & 0xC0) == 0x40)
#define VARATT_IS_1B(PTR) \
	((((varattrib_1b_stub *) (PTR))->va_header & 0x80) == 0x80)
#define VARATT_IS_1B_E(PTR) \
	((((varattrib_1b_stub *) (PTR))->va_header) == 0x80)
#define VARATT_NOT_PAD_BYTE(PTR) (*((uint8 *) (PTR)) != 0)
#define VARATT_IS_COMPRESSED(PTR) VARATT_IS_4B_C(PTR)
#define VARATT_IS_EXTERNAL(PTR) VARATT_IS_1B_E(PTR)
#define VARATT_IS_SHORT(PTR) VARATT_IS_1B(PTR)
#define VARATT_IS_EXTENDED(PTR) (!VARATT_IS_4B_U(PTR))

#define VARSIZE_4B(PTR) \
	(ntohl(((varattrib_4b_stub *) (PTR))->va_4byte.va_header) & 0x3FFFFFFF)
#define VARSIZE_1B(PTR) ((((varattrib_1b_stub *) (PTR))->va_header) & 0x7F)
#define VARSIZE(PTR) VARSIZE_4B(PTR)
#define VARSIZE_SHORT(PTR) VARSIZE_1B(PTR)
#define VARSIZE_ANY(PTR) \
	(VARATT_IS_1B(PTR) ? VARSIZE_1B(PTR) : VARSIZE_4B(PTR))
#define VARDATA_4B(PTR) (((varattrib_4b_stub *) (PTR))->va_4byte.va_data)
#define VARDATA_1B(PTR) (((varattrib_1b_stub *) (PTR))->va_data)
#define VARDATA(PTR) VARDATA_4B(PTR)
#define VARDATA_SHORT(PTR) VARDATA_1B(PTR)
#define VARDATA_ANY(PTR) \
	(VARATT_IS_1B(PTR) ? VARDATA_1B(PTR) : VARDATA_4B(PTR))
#define SET_VARSIZE(PTR, len) \
	(((varattrib_4b_stub *) (PTR))->va_4byte.va_header = \
	 htonl(((uint32) (len)) & 0x3FFFFFFF))
#define SET_VARSIZE_SHORT(PTR, len) \
	(((varattrib_1b_stub *) (PTR))->va_header = ((uint8) (len)) | 0x80)
/* tuptoaster.h:22 — produces the final short HEADER BYTE */
#define VARSIZE_TO_SHORT(PTR) \
	((char) (VARSIZE(PTR) - VARHDRSZ + VARHDRSZ_SHORT) | 0x80)
#define VARSIZE_TO_SHORT_D(D) VARSIZE_TO_SHORT(DatumGetPointer(D))
#define VARATT_CONVERTED_SHORT_SIZE(PTR) \
	(VARSIZE(PTR) - VARHDRSZ + VARHDRSZ_SHORT)
#define VARATT_CAN_MAKE_SHORT(PTR) \
	(VARATT_IS_4B_U(PTR) && \
	 (VARSIZE(PTR) - VARHDRSZ + VARHDRSZ_SHORT) <= 0x7F)

/* ---- extras datumstreamblock.c references (real decls:
 * access/tupmacs.h:202 value_type_could_short,
 * access/tuptoaster.h:182 varattrib_untoast_ptr_len,
 * cdbappendonlystorage.h IsAligned) ---- */
#define IsAligned(ptr, alignment) \
	(((uintptr_t) (ptr)) % (alignment) == 0)
typedef struct varatt_external
{
	int32		va_rawsize;
	int32		va_extsize;
	Oid			va_valueid;
	Oid			va_toastrelid;
} varatt_external;
extern void varattrib_untoast_ptr_len(Datum d, char **datastart, int *len,
				      void **tofree);
extern bool value_type_could_short(Pointer ptr, Oid typid);


/* extras for cdbappendonlystorageformat.c */
#define elogif(p, level, ...) do { if (p) elog(level, __VA_ARGS__); } while (0)
#ifndef INT64CONST
#define INT64CONST(x) ((int64) (x##LL))
#endif
/* pg_appendonly.h:113 — version > AORelationVersion_Original(=1) */
#define IsAOBlockAndMemtupleAlignmentFixed(version) ((version) > 1)
static inline char *
psprintf(const char *fmt, ...)
{
	char	   *buf = (char *) malloc(1024);
	va_list		ap;

	va_start(ap, fmt);
	vsnprintf(buf, 1024, fmt, ap);
	va_end(ap);
	return buf;
}


/* external (toast-pointer) sizes: the wrapper never feeds external
 * datums; keep the macros compilable, trap if ever reached */
#define VARHDRSZ_EXTERNAL 4	/* "In GPDB, it's 4, due to padding" */
#define VARSIZE_EXTERNAL(PTR) (abort(), 0)
#define VARSIZE_ANY_EXHDR(PTR) \
	(VARATT_IS_1B_E(PTR) ? (abort(), 0) : \
	 (VARATT_IS_1B(PTR) ? VARSIZE_1B(PTR) - VARHDRSZ_SHORT : \
	  VARSIZE_4B(PTR) - VARHDRSZ))

#endif
