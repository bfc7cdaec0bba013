/* stub (tupser variant): adds attcacheoff/attnum/attndims, which the
 * real htup_details.h and tuptable.h dereference */
#ifndef GG_STUB_TS_PG_ATTRIBUTE_H
#define GG_STUB_TS_PG_ATTRIBUTE_H
#include "postgres.h"
typedef struct FormData_pg_attribute
{
	Oid			atttypid;
	int16		attlen;
	int16		attnum;
	int32		attcacheoff;
	int32		atttypmod;
	int32		attndims;
	bool		attbyval;
	char		attalign;
	bool		attisdropped;
	bool		attnotnull;
	char		attstorage;
} FormData_pg_attribute;
typedef FormData_pg_attribute *Form_pg_attribute;
#endif
