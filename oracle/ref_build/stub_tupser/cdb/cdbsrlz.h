#ifndef GG_STUB_CDBSRLZ_H
#define GG_STUB_CDBSRLZ_H
#endif
