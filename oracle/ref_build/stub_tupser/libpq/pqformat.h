#ifndef GG_STUB_PQFORMAT_H
#define GG_STUB_PQFORMAT_H
#include "postgres.h"
#include "lib/stringinfo.h"
extern void pq_begintypsend(StringInfo buf);
extern bytea *pq_endtypsend(StringInfo buf);
extern void pq_sendbytes(StringInfo buf, const char *data, int datalen);
extern int pq_getmsgint(StringInfo msg, int b);
extern int64 pq_getmsgint64(StringInfo msg);
extern const char *pq_getmsgbytes(StringInfo msg, int datalen);
extern void pq_copymsgbytes(StringInfo msg, char *buf, int datalen);
extern char *pq_getmsgstring(StringInfo msg);
extern void pq_getmsgend(StringInfo msg);
#endif
