#ifndef GG_STUB_BUF_H
#define GG_STUB_BUF_H
typedef int Buffer;
#define InvalidBuffer 0
#endif
