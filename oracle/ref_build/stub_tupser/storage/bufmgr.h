#ifndef GG_STUB_BUFMGR_H
#define GG_STUB_BUFMGR_H
#include "storage/buf.h"
#endif
