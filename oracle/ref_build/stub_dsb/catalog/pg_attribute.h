/* Stub: datumstreamblock.h includes this; the .c uses nothing from it. */
#ifndef ORACLE_STUB_DSB_PG_ATTRIBUTE_H
#define ORACLE_STUB_DSB_PG_ATTRIBUTE_H
#include "postgres.h"
typedef struct FormData_pg_attribute
{
	Oid			atttypid;
	int16		attlen;
	bool		attbyval;
	char		attalign;
	bool		attisdropped;
	char		attstorage;
	int32		atttypmod;
} FormData_pg_attribute;
typedef FormData_pg_attribute *Form_pg_attribute;
#endif
