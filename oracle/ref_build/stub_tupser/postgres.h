/* stub_tupser: extends the dsb stub environment with the types the
 * Motion serializer's include graph needs (category (b) test infra:
 * compiles the REFERENCE's tupser.c/tupchunklist.c in place). */
#ifndef GG_STUB_TUPSER_POSTGRES_H
#define GG_STUB_TUPSER_POSTGRES_H

#include "../stub_dsb/postgres.h"

typedef uint32 CommandId;
typedef uint32 TransactionId;
typedef uint32 SubTransactionId;
typedef uint32 LocalTransactionId;
#define FUNC_MAX_ARGS 100
typedef struct varlena bytea;
typedef struct varlena text;
typedef int16 AttrNumber;
#ifndef Size
typedef size_t Size;
#endif
typedef int pgsocket;
typedef struct List List;
typedef struct Node Node;
struct MotionConn { int32 sent_record_typmod; };
#define ERRCODE_UNDEFINED_OBJECT 0
#define ERRCODE_PROTOCOL_VIOLATION 0
#define ERRCODE_GP_INTERCONNECTION_ERROR 0
typedef uint8 bits8;
typedef uint16 bits16;
#include "nodes/nodes.h"
typedef struct MemoryContextCallback MemoryContextCallback;

static inline void MemoryContextReset(MemoryContext c) { (void) c; }

#endif
