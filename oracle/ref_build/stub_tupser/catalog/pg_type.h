/* stub: the pg_type fields + OIDs tupser.c touches */
#ifndef GG_STUB_TS_PG_TYPE_H
#define GG_STUB_TS_PG_TYPE_H
#include "postgres.h"
typedef struct FormData_pg_type
{
	char		typtype;
	bool		typisdefined;
	bool		typbyval;
	int16		typlen;
} FormData_pg_type;
typedef FormData_pg_type *Form_pg_type;
#define TYPTYPE_BASE 'b'
#define BPCHAROID 1042
#define RECORDOID 2249
#define INT4OID 23
#define INT8OID 20
#define TEXTOID 25
#endif
