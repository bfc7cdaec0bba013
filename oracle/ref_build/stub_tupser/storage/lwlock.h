#ifndef GG_STUB_STORAGE_LWLOCK_H
#define GG_STUB_STORAGE_LWLOCK_H
#include "postgres.h"
#endif
