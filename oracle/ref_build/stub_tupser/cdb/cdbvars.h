#ifndef GG_STUB_TS_CDBVARS_H
#define GG_STUB_TS_CDBVARS_H
#include "../stub_dsb/cdb/cdbvars.h"
#define TUPLE_CHUNK_ALIGN 4	/* 64-bit build (cdbvars.h:32) */
#endif
