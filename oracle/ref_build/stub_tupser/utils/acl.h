#ifndef GG_STUB_UTILS_ACL_H
#define GG_STUB_UTILS_ACL_H
#include "postgres.h"
#endif
