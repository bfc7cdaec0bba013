/* Stub of c.h: route to the stub postgres.h environment. */
#ifndef ORACLE_STUB_DSB_C_H
#define ORACLE_STUB_DSB_C_H
#include "postgres.h"
#endif
