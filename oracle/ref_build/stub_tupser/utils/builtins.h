#ifndef GG_STUB_UTILS_BUILTINS_H
#define GG_STUB_UTILS_BUILTINS_H
#include "postgres.h"
#endif
