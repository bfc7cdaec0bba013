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

static inline void MemoryContextReset(MemoryContext c) { (void) c; }

#endif
