/*
 * Wrapper exposing the REFERENCE's own Motion tuple-chunk serializer
 * (cdb/motion/tupser.c SerializeTuple:400 + tupchunklist.c), compiled
 * in place from /root/reference — category (b) test infrastructure:
 * the byte-level pin for gg_engine_motion_chunkify/dechunkify
 * (Motion chunks were pinned by restatement, not
 * by reference bytes).
 *
 * We drive the chunked (out-of-line) path: b->pri = NULL makes
 * CandidateForSerializeDirect false, so SerializeTuple emits the
 * TupleChunkList framing (TC_WHOLE + TC_PARTIAL_* splitting at
 * Gp_max_tuple_chunk_size) that ic_udpifc puts on the wire.
 */
#include "postgres.h"
#include "access/memtup.h"
#include "access/heapam.h"
#include "cdb/cdbmotion.h"
#include "cdb/tupser.h"
#include "cdb/tupchunk.h"
#include <string.h>

/* globals the reference objects reference (cdbvars.c in the server) */
int			Gp_max_tuple_chunk_size;
MemoryContext CurrentMemoryContext = NULL;

/*
 * Serialize one MemTuple through the reference's chunked path.
 * Returns total chunk-stream bytes written (concatenated
 * [4-byte header][payload] chunks, exactly as the interconnect sends
 * them), or -1 on overflow/-2 on unexpected state.
 */
int
ref_tupser_chunks(const unsigned char *mt, int mt_len, int natts,
		  int max_chunk, unsigned char *out, int cap)
{
	struct tupleDesc td;
	SerTupInfo	si;
	TupleTableSlot slot;
	struct directTransportBuffer b;
	TupleChunkListData tcl;
	TupleChunkListItem it;
	int			off = 0;

	(void) mt_len;
	Gp_max_tuple_chunk_size = max_chunk;
	memset(&td, 0, sizeof(td));
	td.natts = natts;
	memset(&si, 0, sizeof(si));
	si.tupdesc = &td;
	memset(&slot, 0, sizeof(slot));
	slot.PRIVATE_tts_memtuple = (MemTuple) mt;
	b.pri = NULL;
	b.prilen = 0;
	memset(&tcl, 0, sizeof(tcl));

	SerializeTuple(&slot, &si, &b, &tcl, 0);

	for (it = tcl.p_first; it != NULL; it = it->p_next)
	{
		if (off + (int) it->chunk_length > cap)
			return -1;
		memcpy(out + off, it->chunk_data, it->chunk_length);
		off += it->chunk_length;
	}
	return off;
}

/* data symbols the reference objects reference eagerly at load */
MemoryContext TopMemoryContext = NULL;

#include <stdarg.h>
#include <stdio.h>
#include <stdlib.h>

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
		fprintf(stderr, "reference tupser code raised ERROR (%d)\n",
			level);
		abort();
	}
	return 0;
}

/* leftover macro-shaped references the stub headers declare */
void AssertArg(int c) { (void) c; }
void AssertMacro(int c) { (void) c; }
void AssertState(int c) { (void) c; }
void *MemSetAligned(void *p, int v, size_t n) { return memset(p, v, n); }
long Min(long a, long b) { return a < b ? a : b; }
Datum ObjectIdGetDatum(Oid o) { return (Datum) o; }
int PointerIsValid(const void *p) { return p != NULL; }

/* never reached on the chunk-serialization path; abort loudly if the
 * reference code ever wanders there */
#define GG_UNREACHED(name) \
	{ fprintf(stderr, "tupser stub: unexpected call to " #name "\n"); \
	  abort(); }
void *SearchSysCache1(int id, Datum k) GG_UNREACHED(SearchSysCache1)
void ReleaseSysCache(void *t) GG_UNREACHED(ReleaseSysCache)
char *format_type_be(Oid t) GG_UNREACHED(format_type_be)
void *build_tuple_node_list(int t) GG_UNREACHED(build_tuple_node_list)
char *serializeNode(void *n, int *sz, char **x) GG_UNREACHED(serializeNode)
void *deserializeNode(const char *b, int sz) GG_UNREACHED(deserializeNode)
void TRHandleTypeLists(void *a, void *b) GG_UNREACHED(TRHandleTypeLists)
void *AllocSetContextCreate(void *p, const char *n, size_t a, size_t b,
			    size_t c) GG_UNREACHED(AllocSetContextCreate)
bool value_type_could_short(Pointer p, Oid t)
	GG_UNREACHED(value_type_could_short)
HeapTuple heaptuple_form_to(TupleDesc d, Datum *v, bool *n, HeapTuple t,
			    uint32 *l) GG_UNREACHED(heaptuple_form_to)
struct varlena *heap_tuple_fetch_attr(struct varlena *v)
	GG_UNREACHED(heap_tuple_fetch_attr)
void _slot_getsomeattrs(TupleTableSlot *s, int n)
	GG_UNREACHED(_slot_getsomeattrs)
