/* stub: only what tupser.c/tupchunklist.c need from cdbmotion.h
 * (the real header drags in the whole interconnect include graph) */
#ifndef GG_STUB_CDBMOTION_H
#define GG_STUB_CDBMOTION_H
#include "postgres.h"
#include "cdb/tupchunklist.h"

struct directTransportBuffer
{
	unsigned char *pri;
	int prilen;
};
extern int Gp_max_tuple_chunk_size;
typedef enum SendReturnCode { SEND_COMPLETE, STOP_SENDING } SendReturnCode;
typedef struct MotionLayerState MotionLayerState;
typedef struct ChunkTransportState ChunkTransportState;
#endif
