/* Host-only harness: reproduce the plan_rtc codegen for a Q1-shaped
 * grouped plan and dump the generated source (GG_PLAN_RTC_DUMP).
 * hipRTC compiles without a GPU; module load fails afterwards on a
 * CPU-only box, which is fine — the dump is what we want.  Test/dev
 * tooling only; never part of the product path. */
#include <cstdio>
#include <cstdlib>
#include <memory>

#include "../greengage_amd/csrc/engine_internal.h"

using namespace gg;

int
main(int argc, char **argv)
{
	setenv("GG_PLAN_RTC_DUMP", argc > 1 ? argv[1] : "/tmp/rtc_dump.cu",
	       0);

	PlanDev D{};
	char dummy[256];	/* distinct fake pointers per column so
				 * the load-dedup behaves as with real
				 * tables */

	D.n = 600000000;
	D.npreds = 1;
	D.preds[0].col = dummy + 0;
	D.preds[0].width = 4;
	D.preds[0].lo = -2000000000;
	D.preds[0].hi = 10000;
	D.njoins = 0;
	D.ngroup = 2;
	D.gcol[0] = dummy + 8;
	D.gcol[1] = dummy + 16;
	D.gwidth[0] = D.gwidth[1] = 1;
	D.naggs = 6;
	/* count, sum(qty), sum(price), sum(disc), sum(price*(100-disc)),
	 * sum(price*(100-disc)*(100+tax)) */
	int kinds[6] = {0, 2, 2, 2, 2, 2};
	int nfs[6] = {0, 1, 1, 1, 2, 3};
	int8_t mods[6][3] = {{0}, {0}, {0}, {0}, {0, 1}, {0, 1, 2}};

	for (int a = 0; a < 6; a++)
	{
		D.aggs[a].kind = kinds[a];
		D.aggs[a].nf = nfs[a];
		/* factor columns qty/price/disc/tax as four distinct
		 * pointers, shared across aggs like the real Q1 plan */
		const void *cols_q1[6][3] = {
			{}, {dummy + 24}, {dummy + 32}, {dummy + 40},
			{dummy + 32, dummy + 40},
			{dummy + 32, dummy + 40, dummy + 48}};
		for (int f = 0; f < nfs[a]; f++)
		{
			D.aggs[a].col[f] = cols_q1[a][f];
			D.aggs[a].width[f] = 8;
			D.aggs[a].mod[f] = mods[a][f];
		}
	}

	long long codes[6] = {16710, 20038, 20050, 33350, 36934, 33347};
	std::shared_ptr<void> out;
	gg_status s1 = plan_rtc_compile(D, false, false, &out);
	gg_status s2 = plan_rtc_compile(D, false, false, &out, codes, 6,
					true);

	fprintf(stderr, "generic rc=%d baked-fast rc=%d (module-load "
		"failure is expected without a GPU)\n", s1, s2);
	return 0;
}
