/*
 * PostgreSQL DateADT encoding helpers.
 *
 * A date column value is an int32 count of days since 2000-01-01
 * (reference: src/include/utils/date.h:22 "typedef int32 DateADT";
 * src/include/datatype/timestamp.h:171 POSTGRES_EPOCH_JDATE 2451545 =
 * 2000-01-01).  The engine stores date columns in exactly this encoding.
 *
 * Conversion uses the standard days-from-civil algorithm (public domain,
 * Howard Hinnant's date algorithms), checked against Python's
 * datetime.date in tests/test_oracle_cpu.py.
 */
#ifndef GG_PGDATE_H
#define GG_PGDATE_H

#include <stdint.h>

#if defined(__HIPCC__) || defined(__HIP_DEVICE_COMPILE__)
#define GG_PGDATE_HOSTDEV __host__ __device__ static inline
#else
#define GG_PGDATE_HOSTDEV static inline
#endif

/* days from 1970-01-01 (unix epoch) to civil y-m-d */
GG_PGDATE_HOSTDEV int64_t gg_days_from_civil(int y, unsigned m, unsigned d)
{
	y -= m <= 2;
	{
		const int era = (y >= 0 ? y : y - 399) / 400;
		const unsigned yoe = (unsigned) (y - era * 400);
		const unsigned doy = (153 * (m + (m > 2 ? -3 : 9)) + 2) / 5 + d - 1;
		const unsigned doe = yoe * 365 + yoe / 4 - yoe / 100 + doy;

		return (int64_t) era * 146097 + (int64_t) doe - 719468;
	}
}

/* PG DateADT (days since 2000-01-01) for civil y-m-d */
GG_PGDATE_HOSTDEV int32_t gg_pgdate(int y, unsigned m, unsigned d)
{
	/* 2000-01-01 is unix day 10957 */
	return (int32_t) (gg_days_from_civil(y, m, d) - 10957);
}

#endif /* GG_PGDATE_H */
