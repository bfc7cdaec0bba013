/*
 * RCCL-over-xGMI collectives — the engine's Motion-equivalent transport
 * (replaces the reference's reliable-UDP interconnect, cdb/motion/
 * ic_udpifc.c; SURVEY §5 "Distributed communication backend").
 * One process per GPU (1 segment = 1 GPU); redistribute = alltoallv of
 * per-destination compacted device buffers, combine/gather = allgather.
 */
#include <rccl/rccl.h>
#include <cstring>
#include <cstdlib>
#include <cstdio>
#include <atomic>
#include <fcntl.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <unistd.h>
#include <sched.h>

#include <ctime>

#include "engine_internal.h"

namespace gg
{

#define GG_TRY(expr) \
	do { gg_status _s = (expr); if (_s != GG_OK) return _s; } while (0)

static_assert(sizeof(ncclUniqueId) <= 128, "comm id fits 128 bytes");

/*
 * Shared-memory transport (opt-in, GG_COMM_SHM=1 at id creation):
 * RCCL refuses two communicator ranks on the SAME device
 * (ncclCommInitRank → invalid usage, see profiles/r02_rccl_samedev.md),
 * so the engine's world>1 exchange logic could never execute on a
 * 1-GPU box.  This backend carries the same collectives through a
 * POSIX shm segment (device↔host staged) so multi-rank Q1/Q3/Q5 runs
 * on one GPU end-to-end; it is selected ONLY by a "GGSHM:" comm id,
 * which gg_engine_comm_id only mints when GG_COMM_SHM is set — an
 * 8-GPU production run always takes the RCCL path.
 */
struct ShmCtl
{
	std::atomic<uint64_t> magic;	/* 0 until rank 0 finished init */
	std::atomic<uint64_t> arrive;
	std::atomic<uint64_t> epoch;
};

struct ShmComm
{
	uint8_t *base = nullptr;
	size_t bytes = 0;
	size_t slot = 0;	/* per-rank data slot size */
	int world = 0;
	int rank = 0;
	char name[96];
};

static ShmComm *g_shm = nullptr;

static constexpr size_t SHM_CTL = 4096;

static gg_status shm_barrier(ShmComm *c)
{
	ShmCtl *ctl = (ShmCtl *) c->base;
	uint64_t gen = ctl->epoch.load(std::memory_order_acquire);

	if (ctl->arrive.fetch_add(1) + 1 == (uint64_t) c->world)
	{
		ctl->arrive.store(0, std::memory_order_relaxed);
		ctl->epoch.store(gen + 1, std::memory_order_release);
	}
	else
	{
		for (uint64_t spin = 0;
		     ctl->epoch.load(std::memory_order_acquire) == gen;
		     spin++)
		{
			if (spin > (uint64_t) 600 * 1000 * 1000)
				return fail(GG_ECOMM,
					    "shm barrier timeout (world %d)",
					    c->world);
			sched_yield();
		}
	}
	return GG_OK;
}

static gg_status shm_comm_make_id(void *out_id128)
{
	unsigned long long r1 = 0;
	FILE *f = fopen("/dev/urandom", "rb");

	if (f)
	{
		if (fread(&r1, 1, 8, f) != 8)
			r1 = 0;
		fclose(f);
	}
	r1 ^= ((unsigned long long) getpid() << 32) ^
		(unsigned long long) time(nullptr);
	std::memset(out_id128, 0, 128);
	std::snprintf((char *) out_id128, 128, "GGSHM:/gg-shm-%016llx", r1);
	return GG_OK;
}

static gg_status shm_comm_init(const char *id)
{
	Engine &e = engine();
	const char *name = id + 6;	/* skip "GGSHM:" */
	long mb = 256;
	const char *mbs = getenv("GG_COMM_SHM_MB");

	if (mbs && atol(mbs) > 0)
		mb = atol(mbs);
	ShmComm *c = new ShmComm();

	c->world = e.cfg.n_segments;
	c->rank = e.cfg.segment_id;
	c->slot = (size_t) mb * 1024 * 1024;
	c->bytes = SHM_CTL + c->slot * (size_t) c->world;
	std::snprintf(c->name, sizeof(c->name), "%s", name);

	int fd = shm_open(name, O_CREAT | O_RDWR, 0600);

	if (fd < 0)
	{
		delete c;
		return fail(GG_ECOMM, "shm_open(%s) failed", name);
	}
	if (c->rank == 0 && ftruncate(fd, (off_t) c->bytes) != 0)
	{
		close(fd);
		delete c;
		return fail(GG_ECOMM, "shm ftruncate(%zu) failed", c->bytes);
	}
	/* peers wait for rank 0's ftruncate */
	for (uint64_t spin = 0;; spin++)
	{
		struct stat st;

		if (fstat(fd, &st) == 0 && (size_t) st.st_size >= c->bytes)
			break;
		if (spin > 600ull * 1000 * 1000)
		{
			close(fd);
			delete c;
			return fail(GG_ECOMM, "shm size wait timeout");
		}
		sched_yield();
	}
	c->base = (uint8_t *) mmap(nullptr, c->bytes,
				   PROT_READ | PROT_WRITE, MAP_SHARED, fd, 0);
	close(fd);
	if (c->base == MAP_FAILED)
	{
		delete c;
		return fail(GG_ECOMM, "shm mmap failed");
	}
	ShmCtl *ctl = (ShmCtl *) c->base;

	if (c->rank == 0)
	{
		ctl->arrive.store(0);
		ctl->epoch.store(0);
		ctl->magic.store(0x47474d31ull, std::memory_order_release);
	}
	else
		for (uint64_t spin = 0;
		     ctl->magic.load(std::memory_order_acquire) !=
		     0x47474d31ull; spin++)
		{
			if (spin > 600ull * 1000 * 1000)
			{
				munmap(c->base, c->bytes);
				delete c;
				return fail(GG_ECOMM, "shm magic wait timeout");
			}
			sched_yield();
		}
	g_shm = c;
	e.comm = (void *) c;
	{
		gg_status st = shm_barrier(c);

		if (st != GG_OK)
			return st;
	}
	return GG_OK;
}

static uint8_t *shm_slot(ShmComm *c, int rank)
{
	return c->base + SHM_CTL + c->slot * (size_t) rank;
}

static gg_status shm_allgather_u64(ShmComm *c, const void *dev_send,
				   void *dev_recv, size_t count)
{
	if (count * 8 > c->slot)
		return fail(GG_ECOMM, "shm allgather %zu B > slot "
			    "(raise GG_COMM_SHM_MB)", count * 8);
	GG_HIP(hipMemcpy(shm_slot(c, c->rank), dev_send, count * 8,
			 hipMemcpyDeviceToHost));
	GG_TRY(shm_barrier(c));
	for (int r = 0; r < c->world; r++)
		GG_HIP(hipMemcpy((uint8_t *) dev_recv + (size_t) r * count * 8,
				 shm_slot(c, r), count * 8,
				 hipMemcpyHostToDevice));
	GG_TRY(shm_barrier(c));
	return GG_OK;
}

static gg_status shm_alltoallv_i64(ShmComm *c, const int64_t *send_base,
				   const unsigned long long *send_offs,
				   const unsigned long long *send_cnts,
				   int64_t *recv_base,
				   const unsigned long long *recv_offs,
				   const unsigned long long *recv_cnts)
{
	/* own slot layout: u64 hdr[world] = send counts, then the
	 * per-destination segments concatenated in destination order */
	int w = c->world;
	uint8_t *mine = shm_slot(c, c->rank);
	unsigned long long *hdr = (unsigned long long *) mine;
	size_t need = (size_t) w * 8;

	for (int p = 0; p < w; p++)
		need += (size_t) send_cnts[p] * 8;
	if (need > c->slot)
		return fail(GG_ECOMM, "shm alltoallv %zu B > slot "
			    "(raise GG_COMM_SHM_MB)", need);
	{
		size_t off = (size_t) w * 8;

		for (int p = 0; p < w; p++)
		{
			hdr[p] = send_cnts[p];
			if (send_cnts[p])
				GG_HIP(hipMemcpy(mine + off,
						 send_base + send_offs[p],
						 (size_t) send_cnts[p] * 8,
						 hipMemcpyDeviceToHost));
			off += (size_t) send_cnts[p] * 8;
		}
	}
	GG_TRY(shm_barrier(c));
	for (int s = 0; s < w; s++)
	{
		uint8_t *src = shm_slot(c, s);
		unsigned long long *shdr = (unsigned long long *) src;
		size_t off = (size_t) w * 8;

		for (int q = 0; q < c->rank; q++)
			off += (size_t) shdr[q] * 8;
		if (shdr[c->rank] != recv_cnts[s])
			return fail(GG_ECOMM,
				    "shm alltoallv count mismatch from %d: "
				    "%llu vs %llu", s, shdr[c->rank],
				    recv_cnts[s]);
		if (recv_cnts[s])
			GG_HIP(hipMemcpy(recv_base + recv_offs[s], src + off,
					 (size_t) recv_cnts[s] * 8,
					 hipMemcpyHostToDevice));
	}
	GG_TRY(shm_barrier(c));
	return GG_OK;
}

static gg_status shm_comm_destroy(ShmComm *c)
{
	if (c->rank == 0)
		shm_unlink(c->name);
	if (c->base)
		munmap(c->base, c->bytes);
	delete c;
	g_shm = nullptr;
	return GG_OK;
}

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

	if (getenv("GG_COMM_SHM"))
		return shm_comm_make_id(out_id128);
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
	if (std::memcmp(id128, "GGSHM:", 6) == 0)
		return shm_comm_init((const char *) id128);
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
		if (g_shm && e.comm == (void *) g_shm)
			shm_comm_destroy(g_shm);
		else
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

	if (g_shm && e.comm == (void *) g_shm)
	{
		GG_HIP(hipStreamSynchronize(e.stream));
		return shm_allgather_u64(g_shm, dev_send, dev_recv, count);
	}
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

	if (g_shm && e.comm == (void *) g_shm)
	{
		GG_HIP(hipStreamSynchronize(e.stream));
		return shm_alltoallv_i64(g_shm, send_base, send_offs,
					 send_cnts, recv_base, recv_offs,
					 recv_cnts);
	}
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
