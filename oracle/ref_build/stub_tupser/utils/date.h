#ifndef GG_STUB_UTILS_DATE_H
#define GG_STUB_UTILS_DATE_H
#include "postgres.h"
#endif
