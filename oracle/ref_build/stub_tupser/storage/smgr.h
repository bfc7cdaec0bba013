#ifndef GG_STUB_STORAGE_SMGR_H
#define GG_STUB_STORAGE_SMGR_H
#include "postgres.h"
#endif
