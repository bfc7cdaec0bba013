/* Stub: memtuple.c includes this but uses nothing from it. */
#ifndef ORACLE_STUB_DSB_TRANSAM_H
#define ORACLE_STUB_DSB_TRANSAM_H
#endif
