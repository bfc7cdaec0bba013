#ifndef GG_STUB_HSEARCH_H
#define GG_STUB_HSEARCH_H
typedef struct HTAB HTAB;
#endif
