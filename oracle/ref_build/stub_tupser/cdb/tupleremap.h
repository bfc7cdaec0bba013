#ifndef GG_STUB_TUPLEREMAP_H
#define GG_STUB_TUPLEREMAP_H
typedef struct TupleRemapper TupleRemapper;
#endif
