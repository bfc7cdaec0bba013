/* Stub: type OIDs memtuple.c tests against (values are PostgreSQL's
 * public catalog numbers, pg_type.h). */
#ifndef ORACLE_STUB_DSB_PG_TYPE_H
#define ORACLE_STUB_DSB_PG_TYPE_H
#define BPCHAROID 1042
#endif
