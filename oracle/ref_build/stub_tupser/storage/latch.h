#ifndef GG_STUB_STORAGE_LATCH_H
#define GG_STUB_STORAGE_LATCH_H
#include "postgres.h"
#endif
