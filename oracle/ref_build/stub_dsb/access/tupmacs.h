/* Stub of access/tupmacs.h: alignment helpers datumstreamblock.c uses. */
#ifndef ORACLE_STUB_DSB_TUPMACS_H
#define ORACLE_STUB_DSB_TUPMACS_H
#include "postgres.h"

#define att_align_nominal(cur_offset, attalign) \
	( \
	 ((attalign) == 'i') ? INTALIGN(cur_offset) : \
	 (((attalign) == 'c') ? (uintptr_t) (cur_offset) : \
	  (((attalign) == 'd') ? DOUBLEALIGN(cur_offset) : \
	   SHORTALIGN(cur_offset))) \
	)
#define att_align_datum(cur_offset, attalign, attlen, attptr) \
	( \
	 ((attlen) == -1 && VARATT_IS_SHORT(attptr)) ? \
	 (uintptr_t) (cur_offset) : \
	 att_align_nominal(cur_offset, attalign) \
	)
#endif
