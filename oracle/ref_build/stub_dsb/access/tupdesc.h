/* Stub of access/tupdesc.h for compiling memtuple.c in place: just the
 * fields memtuple.c dereferences (natts, attrs[], tdhasoid). */
#ifndef ORACLE_STUB_DSB_TUPDESC_H
#define ORACLE_STUB_DSB_TUPDESC_H
#include "postgres.h"
#include "catalog/pg_attribute.h"

typedef struct tupleDesc
{
	int			natts;
	Form_pg_attribute *attrs;
	bool		tdhasoid;
}		   *TupleDesc;

#endif
