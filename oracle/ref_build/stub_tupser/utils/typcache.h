#ifndef GG_STUB_UTILS_TYPCACHE_H
#define GG_STUB_UTILS_TYPCACHE_H
#include "postgres.h"
#endif
