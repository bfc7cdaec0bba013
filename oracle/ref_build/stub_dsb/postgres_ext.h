/* Stub of postgres_ext.h: types already provided by the stub postgres.h. */
#ifndef ORACLE_STUB_DSB_POSTGRES_EXT_H
#define ORACLE_STUB_DSB_POSTGRES_EXT_H
#include "postgres.h"
#endif
