/*
 * Restatement of the reference's 32-bit hashing and distribution-hash
 * semantics, shared by the CPU oracle (plain C) and the HIP kernels
 * (compiled as device code via GG_HOSTDEV).
 *
 * Semantics sources (reference file:line, studied, not copied):
 *   - hash_any / hash_uint32: Bob Jenkins 2006 lookup-style hash as adopted
 *     by PostgreSQL, src/backend/access/hash/hashfunc.c:302 (hash_any) and
 *     :527 (hash_uint32); mix()/final() macros hashfunc.c:241/:276.
 *     Little-endian path only (build hosts + MI355X are little-endian).
 *   - hashint4: hashfunc.c:46 (= hash_uint32 of the value).
 *   - hashint8: hashfunc.c:52 (lo ^= (v>=0 ? hi : ~hi), then hash_uint32) —
 *     makes int2/int4/int8 hash-join compatible.
 *   - hashchar: hashfunc.c:34 (= hash_uint32 of the char value).
 *   - cdbhash per-attribute combine: src/backend/cdb/cdbhash.c:190–218 —
 *     running hash rotated left 1 bit each attribute, XOR the attribute's
 *     hash if non-NULL (NULL contributes only the rotate).
 *   - cdbhashinit: cdbhash.c:172 — running hash starts at 0 (non-legacy).
 *   - cdbhashreduce: cdbhash.c:254 — non-legacy opclasses reduce with
 *     jump_consistent_hash (cdbhash.c:549), the algorithm of Lamping &
 *     Veach, "A Fast, Minimal Memory, Consistent Hash Algorithm"
 *     (arXiv:1406.2294), restated from the paper; the 32-bit running hash
 *     is zero-extended to the uint64 key.
 *
 * Pinned bit-for-bit against the reference's own hashfunc.c compiled in
 * place (oracle/ref_build/) by tests/test_hash_cpu.py via the committed
 * golden vectors tests/golden/hash_vectors.json.
 *
 * This header is PRODUCT code (used by the HIP kernels and the C-ABI
 * engine); the oracle includes it too so both sides hash identically —
 * the reference pin above is what keeps it honest.
 */
#ifndef GG_PG_HASH_H
#define GG_PG_HASH_H

#include <stdint.h>

#if defined(__HIPCC__) || defined(__HIP_DEVICE_COMPILE__)
#define GG_HOSTDEV __host__ __device__ static inline
#else
#define GG_HOSTDEV static inline
#endif

#define GG_ROT32(x, k) (((x) << (k)) | ((x) >> (32 - (k))))

/* Jenkins mix(): reversible mix of three 32-bit lanes (hashfunc.c:241). */
#define GG_JMIX(a, b, c) \
	do { \
		a -= c; a ^= GG_ROT32(c, 4);  c += b; \
		b -= a; b ^= GG_ROT32(a, 6);  a += c; \
		c -= b; c ^= GG_ROT32(b, 8);  b += a; \
		a -= c; a ^= GG_ROT32(c, 16); c += b; \
		b -= a; b ^= GG_ROT32(a, 19); a += c; \
		c -= b; c ^= GG_ROT32(b, 4);  b += a; \
	} while (0)

/* Jenkins final(): avalanche three lanes into c (hashfunc.c:276). */
#define GG_JFINAL(a, b, c) \
	do { \
		c ^= b; c -= GG_ROT32(b, 14); \
		a ^= c; a -= GG_ROT32(c, 11); \
		b ^= a; b -= GG_ROT32(a, 25); \
		c ^= b; c -= GG_ROT32(b, 16); \
		a ^= c; a -= GG_ROT32(c, 4);  \
		b ^= a; b -= GG_ROT32(a, 14); \
		c ^= b; c -= GG_ROT32(b, 24); \
	} while (0)

/* hash_uint32 (hashfunc.c:527): one final() over the seeded lanes. */
GG_HOSTDEV uint32_t gg_hash_uint32(uint32_t k)
{
	uint32_t a, b, c;

	a = b = c = 0x9e3779b9u + (uint32_t) sizeof(uint32_t) + 3923095u;
	a += k;
	GG_JFINAL(a, b, c);
	return c;
}

/*
 * hash_any (hashfunc.c:302), little-endian byte-at-a-time path.
 * The reference has a word-aligned fast path and an unaligned path that
 * compute the same value on little-endian machines; this restatement uses
 * one path (equality with both is what the golden vectors check, using
 * aligned and misaligned inputs).
 */
GG_HOSTDEV uint32_t gg_hash_any(const unsigned char *k, int keylen)
{
	uint32_t a, b, c, len;

	len = (uint32_t) keylen;
	a = b = c = 0x9e3779b9u + len + 3923095u;

	while (len >= 12)
	{
		a += (uint32_t) k[0] + ((uint32_t) k[1] << 8) +
			((uint32_t) k[2] << 16) + ((uint32_t) k[3] << 24);
		b += (uint32_t) k[4] + ((uint32_t) k[5] << 8) +
			((uint32_t) k[6] << 16) + ((uint32_t) k[7] << 24);
		c += (uint32_t) k[8] + ((uint32_t) k[9] << 8) +
			((uint32_t) k[10] << 16) + ((uint32_t) k[11] << 24);
		GG_JMIX(a, b, c);
		k += 12;
		len -= 12;
	}

	/* last 11 bytes; c's low byte holds only the length */
	switch (len)
	{
		case 11: c += ((uint32_t) k[10] << 24); /* fall through */
		case 10: c += ((uint32_t) k[9] << 16);  /* fall through */
		case 9:  c += ((uint32_t) k[8] << 8);   /* fall through */
		case 8:  b += ((uint32_t) k[7] << 24);  /* fall through */
		case 7:  b += ((uint32_t) k[6] << 16);  /* fall through */
		case 6:  b += ((uint32_t) k[5] << 8);   /* fall through */
		case 5:  b += (uint32_t) k[4];          /* fall through */
		case 4:  a += ((uint32_t) k[3] << 24);  /* fall through */
		case 3:  a += ((uint32_t) k[2] << 16);  /* fall through */
		case 2:  a += ((uint32_t) k[1] << 8);   /* fall through */
		case 1:  a += (uint32_t) k[0];          /* case 0: nothing */
	}

	GG_JFINAL(a, b, c);
	return c;
}

/* hashint4 (hashfunc.c:46) */
GG_HOSTDEV uint32_t gg_hashint4(int32_t v)
{
	return gg_hash_uint32((uint32_t) v);
}

/* hashint8 (hashfunc.c:52): fold high half into low, sign-dependent. */
GG_HOSTDEV uint32_t gg_hashint8(int64_t v)
{
	uint32_t lohalf = (uint32_t) v;
	uint32_t hihalf = (uint32_t) ((uint64_t) v >> 32);

	lohalf ^= (v >= 0) ? hihalf : ~hihalf;
	return gg_hash_uint32(lohalf);
}

/* hashchar (hashfunc.c:34): used for "char"/bool columns. */
GG_HOSTDEV uint32_t gg_hashchar(char ch)
{
	return gg_hash_uint32((uint32_t) (int32_t) ch);
}

/*
 * cdbhash attribute combine (cdbhash.c:190–218, non-legacy):
 * rotate the running hash left one bit, XOR the attribute hash when the
 * attribute is not NULL.  Running hash starts at 0 (cdbhashinit,
 * cdbhash.c:172).
 */
GG_HOSTDEV uint32_t gg_cdbhash_combine(uint32_t running, uint32_t attr_hash,
				       int isnull)
{
	uint32_t h = (running << 1) | (running >> 31);

	if (!isnull)
		h ^= attr_hash;
	return h;
}

/*
 * Jump consistent hash (Lamping & Veach, arXiv:1406.2294; used by
 * cdbhashreduce, cdbhash.c:549).  key is the 32-bit cdbhash zero-extended.
 */
GG_HOSTDEV int32_t gg_jump_consistent_hash(uint64_t key, int32_t num_segments)
{
	int64_t b = -1;
	int64_t j = 0;

	while (j < num_segments)
	{
		b = j;
		key = key * 2862933555777941757ULL + 1;
		j = (int64_t) ((double) (b + 1) *
			       ((double) (1LL << 31) / (double) ((key >> 33) + 1)));
	}
	return (int32_t) b;
}

/* Segment of a single-int64-key row (e.g. l_orderkey DISTRIBUTED BY). */
GG_HOSTDEV int32_t gg_cdbhash_segment_int8(int64_t key, int32_t num_segments)
{
	uint32_t h = gg_cdbhash_combine(0, gg_hashint8(key), 0);

	return gg_jump_consistent_hash((uint64_t) h, num_segments);
}

/* Segment of a single-int32-key row (e.g. c_custkey integer). */
GG_HOSTDEV int32_t gg_cdbhash_segment_int4(int32_t key, int32_t num_segments)
{
	uint32_t h = gg_cdbhash_combine(0, gg_hashint4(key), 0);

	return gg_jump_consistent_hash((uint64_t) h, num_segments);
}

#endif /* GG_PG_HASH_H */
