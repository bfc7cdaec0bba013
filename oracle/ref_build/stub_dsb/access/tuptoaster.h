/* Stub of access/tuptoaster.h: only the short-varlena conversion size. */
#ifndef ORACLE_STUB_DSB_TUPTOASTER_H
#define ORACLE_STUB_DSB_TUPTOASTER_H
#include "postgres.h"
#define VARATT_CONVERTED_SHORT_SIZE(PTR) \
	(VARSIZE(PTR) - VARHDRSZ + VARHDRSZ_SHORT)
#endif
