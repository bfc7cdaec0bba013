#ifndef GG_STUB_HEAPAM_H
#define GG_STUB_HEAPAM_H
#include "postgres.h"
#include "access/htup.h"
#include "executor/tuptable.h"
#endif
