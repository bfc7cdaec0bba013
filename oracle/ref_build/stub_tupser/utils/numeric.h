#ifndef GG_STUB_UTILS_NUMERIC_H
#define GG_STUB_UTILS_NUMERIC_H
#include "postgres.h"
#endif
