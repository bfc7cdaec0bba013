#ifndef GG_STUB_TS_TUPDESC_H
#define GG_STUB_TS_TUPDESC_H
#include "postgres.h"
#include "catalog/pg_attribute.h"
typedef struct tupleDesc
{
	int			natts;
	Form_pg_attribute *attrs;
	bool		tdhasoid;
}		   *TupleDesc;
#endif
