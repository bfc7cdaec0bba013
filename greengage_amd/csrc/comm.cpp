/*
 * RCCL-over-xGMI collectives — the engine's Motion-equivalent transport
 * (replaces the reference's reliable-UDP interconnect, cdb/motion/
 * ic_udpifc.c; SURVEY §5 "Distributed communication backend").
 * One process per GPU (1 segment = 1 GPU); redistribute = alltoallv of
 * per-destination compacted device buffers, combine/gather = allgather.
 */
#include <rccl/rccl.h>
#include <cstring>

#include "engine_internal.h"

namespace gg
{

static_assert(sizeof(ncclUniqueId) <= 128, "comm id fits 128 bytes");

#define GG_NCCL(call) \
	do { \
		ncclResult_t _r = (call); \
		if (_r != ncclSuccess) \
			return ::gg::fail(GG_ECOMM, "%s:%d RCCL error %s in %s", \
					  __FILE__, __LINE__, \
					  ncclGetErrorString(_r), #call); \
	} while (0)

static ncclComm_t comm_handle()
{
	return (ncclComm_t) engine().comm;
}

gg_status comm_make_id(void *out_id128)
{
	ncclUniqueId id;

	GG_NCCL(ncclGetUniqueId(&id));
	std::memset(out_id128, 0, 128);
	std::memcpy(out_id128, &id, sizeof(id));
	return GG_OK;
}

gg_status comm_init(const void *id128)
{
	Engine &e = engine();
	ncclUniqueId id;
	ncclComm_t c = nullptr;

	if (e.comm)
		return fail(GG_ESTATE, "comm already initialized");
	std::memcpy(&id, id128, sizeof(id));
	GG_NCCL(ncclCommInitRank(&c, e.cfg.n_segments, id, e.cfg.segment_id));
	e.comm = (void *) c;
	return GG_OK;
}

gg_status comm_destroy()
{
	Engine &e = engine();

	if (e.comm)
	{
		ncclCommDestroy(comm_handle());
		e.comm = nullptr;
	}
	return GG_OK;
}

bool comm_ready()
{
	return engine().comm != nullptr;
}

/* allgather count u64 words per rank; dev buffers */
gg_status comm_allgather_u64(const void *dev_send, void *dev_recv,
			     size_t count)
{
	Engine &e = engine();

	GG_NCCL(ncclAllGather(dev_send, dev_recv, count, ncclUint64,
			      comm_handle(), e.stream));
	GG_HIP(hipStreamSynchronize(e.stream));
	return GG_OK;
}

/*
 * alltoallv of int64 rows: send[p] = send_base + send_offs[p], length
 * send_cnts[p]; likewise receive into recv_base at recv_offs[p].
 * Counts/offsets are HOST arrays (elements, not bytes).
 */
gg_status comm_alltoallv_i64(const int64_t *send_base,
			     const unsigned long long *send_offs,
			     const unsigned long long *send_cnts, int64_t *recv_base,
			     const unsigned long long *recv_offs,
			     const unsigned long long *recv_cnts)
{
	Engine &e = engine();
	int n = e.cfg.n_segments;

	/* an error between GroupStart and GroupEnd must still close the
	 * group, or the communicator is left group-started and the next
	 * collective on it misbehaves */
	GG_NCCL(ncclGroupStart());
	ncclResult_t r = ncclSuccess;

	for (int p = 0; p < n && r == ncclSuccess; p++)
	{
		if (send_cnts[p])
			r = ncclSend(send_base + send_offs[p], send_cnts[p],
				     ncclInt64, p, comm_handle(), e.stream);
		if (recv_cnts[p] && r == ncclSuccess)
			r = ncclRecv(recv_base + recv_offs[p], recv_cnts[p],
				     ncclInt64, p, comm_handle(), e.stream);
	}
	ncclResult_t rend = ncclGroupEnd();

	if (r != ncclSuccess || rend != ncclSuccess)
		return fail(GG_ECOMM, "alltoallv RCCL error %s",
			    ncclGetErrorString(r != ncclSuccess ? r : rend));
	GG_HIP(hipStreamSynchronize(e.stream));
	return GG_OK;
}

}				/* namespace gg */
