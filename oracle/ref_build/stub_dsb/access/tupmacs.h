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

/* PostgreSQL's standard attribute fetch/store/advance macros
 * (public tupmacs.h API), restated for the fixed-width and varlena
 * cases memtuple.c exercises. */
#define fetch_att(T, attbyval, attlen) \
	( \
	 (attbyval) ? \
	 ( \
	  (attlen) == (int) sizeof(Datum) ? *((Datum *) (T)) : \
	  ( \
	   (attlen) == (int) sizeof(int32) ? Int32GetDatum(*((int32 *) (T))) : \
	   ( \
	    (attlen) == (int) sizeof(int16) ? Int16GetDatum(*((int16 *) (T))) : \
	    CharGetDatum(*((char *) (T))) \
	   ) \
	  ) \
	 ) \
	 : \
	 PointerGetDatum(T) \
	)
#define fetchatt(A, T) fetch_att(T, (A)->attbyval, (A)->attlen)

#define att_addlength_datum(cur_offset, attlen, attdatum) \
	att_addlength_pointer(cur_offset, attlen, DatumGetPointer(attdatum))

#define att_addlength_pointer(cur_offset, attlen, attptr) \
	( \
	 ((attlen) > 0) ? ((uintptr_t) (cur_offset) + (attlen)) : \
	 (((attlen) == -1) ? ((uintptr_t) (cur_offset) + VARSIZE_ANY(attptr)) : \
	  ((uintptr_t) (cur_offset) + strlen((char *) (attptr)) + 1)) \
	)

#define store_att_byval(T, newdatum, attlen) \
	do { \
		switch (attlen) \
		{ \
			case sizeof(char): \
				*(char *) (T) = DatumGetChar(newdatum); \
				break; \
			case sizeof(int16): \
				*(int16 *) (T) = DatumGetInt16(newdatum); \
				break; \
			case sizeof(int32): \
				*(int32 *) (T) = DatumGetInt32(newdatum); \
				break; \
			case sizeof(Datum): \
				*(Datum *) (T) = (newdatum); \
				break; \
			default: \
				abort(); \
		} \
	} while (0)

/* value_type_could_short: declared in stub postgres.h, defined once in
 * dsb_wrap.c */

#endif