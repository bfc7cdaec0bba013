#ifndef GG_STUB_TS_SYSCACHE_H
#define GG_STUB_TS_SYSCACHE_H
#include "postgres.h"
#include "access/htup.h"
enum SysCacheIdentifier { TYPEOID = 76 };
extern HeapTuple SearchSysCache1(int cacheId, Datum key1);
extern void ReleaseSysCache(HeapTuple tuple);
#endif
