/* SPDX-License-Identifier: GPL-2.0 OR MIT
 *
 * bridge_tests.c — userspace unit + race tests of the rocp2p bridge,
 * driven through the fake IB core / fake KFD the way real OFED + amdkfd
 * would drive it.  Covers the behaviors the reference could only test on
 * real hardware (reference test surface: /root/reference/tests/
 * amdp2ptest.c) plus the races it could not test at all.
 */
#define __ROCNR_SHIM__ 1
#include "rocnr_shim_all.h"
#include "fake_kfd.h"
#include "fake_ibcore.h"
#include "rocp2p_sg.h"

#include <assert.h>
#include <time.h>
#include <stdlib.h>
#include <unistd.h>

#define CHECK(cond) do { \
	if (!(cond)) { \
		fprintf(stderr, "FAIL %s:%d: %s\n", __FILE__, __LINE__, #cond); \
		exit(1); \
	} \
} while (0)

#define MiB (1ULL << 20)
#define GiB (1ULL << 30)

static void check_balances(struct device **devs, int ndev)
{
	int i;

	CHECK(rocnr_shim_pid_balance() == 0);
	CHECK(rocnr_shim_alloc_balance() == 0);
	CHECK(rocnr_shim_sg_balance() == 0);
	CHECK(rocnr_shim_module_refcount() == 0);
	CHECK(fake_kfd_live_pins() == 0);
	CHECK(fake_kfd_bad_put_calls() == 0);
	for (i = 0; i < ndev; i++)
		CHECK(atomic64_read(&devs[i]->live_maps) == 0);
}

/* ---- rocnr_coalesce unit tests (array-backed iterator) ---- */
struct arr_iter {
	struct rocnr_seg_iter it;
	const struct rocnr_seg *segs;
	size_t n, i;
};

static int arr_next(struct rocnr_seg_iter *it, struct rocnr_seg *seg)
{
	struct arr_iter *a = (struct arr_iter *)it;

	if (a->i >= a->n)
		return 0;
	*seg = a->segs[a->i++];
	return 1;
}

static struct arr_iter mk_iter(const struct rocnr_seg *segs, size_t n)
{
	struct arr_iter a = { { arr_next, 0 }, segs, n, 0 };
	return a;
}

struct collect_ctx {
	struct rocnr_seg out[64];
	size_t n;
};

static int collect(void *ctx, const struct rocnr_seg *seg)
{
	struct collect_ctx *c = ctx;

	if (c->n >= 64)
		return 1;
	c->out[c->n++] = *seg;
	return 0;
}

static void test_coalesce_unit(void)
{
	struct collect_ctx c;
	struct arr_iter it;

	{	/* empty */
		it = mk_iter(NULL, 0);
		CHECK(rocnr_coalesce_count(&it.it, 0) == 0);
	}
	{	/* adjacent merge + gap */
		const struct rocnr_seg s[] = {
			{ 0x1000, 0x1000 }, { 0x2000, 0x1000 },
			{ 0x4000, 0x1000 }, { 0x5000, 0x1000 },
		};
		it = mk_iter(s, 4);
		c.n = 0;
		CHECK(rocnr_coalesce(&it.it, 0, collect, &c) == 2);
		CHECK(c.out[0].addr == 0x1000 && c.out[0].len == 0x2000);
		CHECK(c.out[1].addr == 0x4000 && c.out[1].len == 0x2000);
	}
	{	/* zero-length entries skipped */
		const struct rocnr_seg s[] = {
			{ 0x1000, 0 }, { 0x1000, 0x1000 }, { 0x2000, 0 },
			{ 0x2000, 0x1000 },
		};
		it = mk_iter(s, 4);
		CHECK(rocnr_coalesce_count(&it.it, 0) == 1);
	}
	{	/* max_seg split: 10 pages merged then split at 3 pages */
		struct rocnr_seg s[10];
		int i;

		for (i = 0; i < 10; i++) {
			s[i].addr = 0x100000 + (rocnr_u64)i * 0x1000;
			s[i].len = 0x1000;
		}
		it = mk_iter(s, 10);
		c.n = 0;
		CHECK(rocnr_coalesce(&it.it, 0x3000, collect, &c) == 4);
		CHECK(c.out[0].len == 0x3000 && c.out[3].len == 0x1000);
	}
	{	/* emit abort propagates */
		struct rocnr_seg s[80];
		int i;

		for (i = 0; i < 80; i++) {
			s[i].addr = 0x2000000 + (rocnr_u64)i * 0x2000;
			s[i].len = 0x1000;	/* all disjoint */
		}
		it = mk_iter(s, 80);
		c.n = 0;
		CHECK(rocnr_coalesce(&it.it, 0, collect, &c) == (size_t)-1);
	}
	printf("ok: coalesce unit\n");
}

/* ---- bridge flows ---- */
static struct device dev1 = { .name = "hca0", .iova_offset = 0x10000000000ULL,
			      .fail_after = -1 };
static struct device dev2 = { .name = "hca1", .iova_offset = 0x20000000000ULL,
			      .fail_after = -1 };
static struct device *devs[] = { &dev1, &dev2 };

static void test_not_gpu(void)
{
	struct fake_ib_mr *mr = NULL;

	CHECK(fake_ib_reg_mr(0x1234000, 4096, &dev1, &mr) == -ENODEV);
	check_balances(devs, 2);
	printf("ok: not-gpu fallback\n");
}

static void test_lifecycle(void)
{
	uint64_t va = fake_kfd_alloc(64 * MiB, 0);
	struct fake_ib_mr *mr = NULL;
	uint64_t total = 0;
	struct scatterlist *sg;
	int i;

	CHECK(fake_ib_reg_mr(va, 64 * MiB, &dev1, &mr) == 0);
	CHECK(mr->page_size == 2 * MiB);
	/* 32 bus-contiguous 2 MiB chunks must coalesce to ONE segment */
	CHECK(mr->nmap == 1);
	for_each_sg(mr->sgt.sgl, sg, mr->sgt.nents, i)
		total += sg_dma_len(sg);
	CHECK(total == 64 * MiB);
	/* simulated IOMMU applied */
	CHECK(sg_dma_address(mr->sgt.sgl) >= dev1.iova_offset);

	CHECK(fake_ib_dereg_mr(mr) == 0);
	free(mr);
	fake_kfd_free(va);
	check_balances(devs, 2);
	printf("ok: lifecycle + full coalesce\n");
}

static void test_fragmented(void)
{
	/* hole after every 4th chunk: 32 chunks -> 8 runs */
	uint64_t va = fake_kfd_alloc(64 * MiB, 4);
	struct fake_ib_mr *mr = NULL;

	CHECK(fake_ib_reg_mr(va, 64 * MiB, &dev1, &mr) == 0);
	CHECK(mr->nmap == 8);
	CHECK(fake_ib_dereg_mr(mr) == 0);
	free(mr);
	fake_kfd_free(va);
	check_balances(devs, 2);
	printf("ok: fragmented pin coalesces per run\n");
}

static void test_max_seg(void)
{
	uint64_t va = fake_kfd_alloc(64 * MiB, 0);
	struct device small = { .name = "small", .max_seg = 8 * MiB,
				.fail_after = -1 };
	struct fake_ib_mr *mr = NULL;

	CHECK(fake_ib_reg_mr(va, 64 * MiB, &small, &mr) == 0);
	CHECK(mr->nmap == 8);	/* 64 MiB / 8 MiB max_seg */
	CHECK(fake_ib_dereg_mr(mr) == 0);
	free(mr);
	fake_kfd_free(va);
	CHECK(atomic64_read(&small.live_maps) == 0);
	check_balances(devs, 2);
	printf("ok: device max_seg honored\n");
}

static void test_range_mismatch(void)
{
	uint64_t va = fake_kfd_alloc(8 * MiB, 0);
	const struct peer_memory_client *cl = fake_ib_client();
	void *ctx = NULL;
	char name[IB_PEER_MEMORY_NAME_MAX];
	struct sg_table sgt = { 0 };

	CHECK(cl->acquire(va, 8 * MiB, NULL, name, &ctx) == 1);
	CHECK(cl->get_pages(va + 4096, 8 * MiB, 1, 0, &sgt, ctx, 0) ==
	      -EINVAL);
	CHECK(cl->get_pages(va, 4 * MiB, 1, 0, &sgt, ctx, 0) == -EINVAL);
	cl->release(ctx);
	fake_kfd_free(va);
	check_balances(devs, 2);
	printf("ok: get_pages range mismatch rejected\n");
}

static void test_map_failure_unwinds(void)
{
	uint64_t va = fake_kfd_alloc(64 * MiB, 4);	/* 8 runs */
	struct device flaky = { .name = "flaky", .fail_after = 3,
				.iova_offset = 0 };
	struct fake_ib_mr *mr = NULL;

	CHECK(fake_ib_reg_mr(va, 64 * MiB, &flaky, &mr) != 0);
	CHECK(atomic64_read(&flaky.live_maps) == 0);
	fake_kfd_free(va);
	check_balances(devs, 2);
	printf("ok: dma_map failure unwinds cleanly\n");
}

static void test_two_devices(void)
{
	uint64_t va = fake_kfd_alloc(32 * MiB, 0);
	struct fake_ib_mr *mr = NULL;
	struct sg_table sgt2 = { 0 };
	int nmap2 = 0;

	CHECK(fake_ib_reg_mr(va, 32 * MiB, &dev1, &mr) == 0);
	CHECK(fake_ib_mr_map_also(mr, &dev2, &sgt2, &nmap2) == 0);
	CHECK(nmap2 == 1);
	CHECK(sg_dma_address(sgt2.sgl) >= dev2.iova_offset);
	CHECK(sg_dma_address(sgt2.sgl) != sg_dma_address(mr->sgt.sgl));
	CHECK(fake_ib_mr_unmap_also(mr, &dev2, &sgt2) == 0);
	CHECK(fake_ib_dereg_mr(mr) == 0);
	free(mr);
	fake_kfd_free(va);
	check_balances(devs, 2);
	printf("ok: two concurrent device mappings\n");
}

static void test_release_cleans_leftovers(void)
{
	uint64_t va = fake_kfd_alloc(16 * MiB, 0);
	struct fake_ib_mr *mr = NULL;
	struct sg_table sgt2 = { 0 };
	int nmap2 = 0;

	CHECK(fake_ib_reg_mr(va, 16 * MiB, &dev1, &mr) == 0);
	CHECK(fake_ib_mr_map_also(mr, &dev2, &sgt2, &nmap2) == 0);
	/* dereg WITHOUT unmapping dev2: release must clean it up */
	CHECK(fake_ib_dereg_mr(mr) == 0);
	CHECK(atomic64_read(&dev2.live_maps) == 0);
	free(mr);
	fake_kfd_free(va);
	check_balances(devs, 2);
	printf("ok: release cleans leftover mappings\n");
}

static void test_invalidate(void)
{
	uint64_t va = fake_kfd_alloc(16 * MiB, 0);
	struct fake_ib_mr *mr = NULL;
	long inv0 = fake_ib_invalidate_count();

	CHECK(fake_ib_reg_mr(va, 16 * MiB, &dev1, &mr) == 0);
	fake_kfd_free(va);	/* GPU frees under a live MR */
	CHECK(fake_ib_invalidate_count() == inv0 + 1);
	CHECK(mr->invalidated == 1);
	CHECK(fake_kfd_bad_put_calls() == 0);
	/* app's dereg after invalidation must be a safe no-op */
	CHECK(fake_ib_dereg_mr(mr) == 0);
	free(mr);
	check_balances(devs, 2);
	printf("ok: async invalidation tears down MR exactly once\n");
}

struct race_arg {
	uint64_t va;
	unsigned int delay_ns;
};

static void *race_free_thread(void *argp)
{
	struct race_arg *a = argp;
	struct timespec ts = { 0, a->delay_ns };

	nanosleep(&ts, NULL);
	fake_kfd_free(a->va);
	return NULL;
}

static void test_invalidate_race(void)
{
	int ITERS = 3000;
	const char *env_iters = getenv("ROCNR_RACE_ITERS");

	if (env_iters && atoi(env_iters) > 0)
		ITERS = atoi(env_iters);
	unsigned int seed = 12345;
	int i;

	for (i = 0; i < ITERS; i++) {
		struct fake_ib_mr *mr = NULL;
		struct race_arg a;
		pthread_t th;
		struct timespec ts = { 0, 0 };

		a.va = fake_kfd_alloc(4 * MiB, 0);
		CHECK(fake_ib_reg_mr(a.va, 4 * MiB, &dev1, &mr) == 0);
		a.delay_ns = rand_r(&seed) % 20000;
		pthread_create(&th, NULL, race_free_thread, &a);
		ts.tv_nsec = rand_r(&seed) % 20000;
		nanosleep(&ts, NULL);
		fake_ib_dereg_mr(mr);
		pthread_join(th, NULL);
		fake_ib_dereg_mr(mr);	/* idempotent */
		free(mr);
		CHECK(fake_kfd_bad_put_calls() == 0);
	}
	check_balances(devs, 2);
	printf("ok: %d dereg-vs-invalidate races, no double-put, no leak\n",
	       ITERS);
}

static void test_page_size_fallback(void)
{
	/* KFD page-size query failing mid-registration must not fail the
	 * MR: the bridge warns and reports the 2 MiB VRAM granule
	 * (bridge policy, mirroring the reference's 4096 fallback at
	 * amdp2p.c:334-340 but with the MI355X granule). */
	uint64_t va = fake_kfd_alloc(8 * MiB, 0);
	struct fake_ib_mr *mr = NULL;

	fake_kfd_fail_page_size(2);	/* get_pages + get_page_size */
	CHECK(fake_ib_reg_mr(va, 8 * MiB, &dev1, &mr) == 0);
	CHECK(mr->page_size == 2 * MiB);
	CHECK(fake_ib_dereg_mr(mr) == 0);
	free(mr);
	fake_kfd_free(va);
	fake_kfd_fail_page_size(0);
	check_balances(devs, 2);
	printf("ok: page-size query failure falls back to 2 MiB\n");
}

#if ROCNR_AMD_RDMA_HAS_DMA_DEV
static void test_null_dev_fallback(void)
{
	/* a modern ROCK KFD may refuse get_pages with
	 * dma_dev=NULL.  The bridge must fall back to deferring the pin to
	 * dma_map (where the HCA device is known), use KFD's device-mapped
	 * table verbatim (no dma_map_resource), and still handle dereg and
	 * invalidation. */
	uint64_t va, va2;
	struct fake_ib_mr *mr = NULL;
	struct sg_table sgt2 = { 0 };
	int nmap2 = 0;
	long inv0;

	fake_kfd_reject_null_dev(1);

	/* lifecycle via dereg */
	va = fake_kfd_alloc(32 * MiB, 0);
	CHECK(fake_ib_reg_mr(va, 32 * MiB, &dev1, &mr) == 0);
	CHECK(mr->nmap == 1);
	/* KFD applied the device mapping internally */
	CHECK(sg_dma_address(mr->sgt.sgl) >= dev1.iova_offset);
	/* and the bridge did NOT dma_map_resource on top */
	CHECK(atomic64_read(&dev1.live_maps) == 0);
	/* a second HCA shares the table (reference-style, warns) */
	CHECK(fake_ib_mr_map_also(mr, &dev2, &sgt2, &nmap2) == 0);
	CHECK(nmap2 == 1);
	CHECK(sg_dma_address(sgt2.sgl) == sg_dma_address(mr->sgt.sgl));
	CHECK(fake_ib_mr_unmap_also(mr, &dev2, &sgt2) == 0);
	CHECK(fake_ib_dereg_mr(mr) == 0);
	free(mr);
	fake_kfd_free(va);

	/* lifecycle via invalidation */
	va2 = fake_kfd_alloc(8 * MiB, 0);
	mr = NULL;
	inv0 = fake_ib_invalidate_count();
	CHECK(fake_ib_reg_mr(va2, 8 * MiB, &dev1, &mr) == 0);
	fake_kfd_free(va2);
	CHECK(fake_ib_invalidate_count() == inv0 + 1);
	CHECK(fake_ib_dereg_mr(mr) == 0);
	free(mr);

	fake_kfd_reject_null_dev(0);
	check_balances(devs, 2);
	printf("ok: dma_dev=NULL drift fallback (deferred pin at dma_map)\n");
}
#else
static void test_null_dev_fallback(void)
{
	printf("ok: dma_dev=NULL drift fallback (n/a: legacy amd_rdma ABI)\n");
}
#endif

static void test_revoke_during_pin(void)
{
	/* KFD fires the free callback synchronously from
	 * inside get_pages.  With the lock-free PINNING install this must
	 * neither deadlock nor touch the reclaimed pin; the registration
	 * fails cleanly. */
	uint64_t va = fake_kfd_alloc(8 * MiB, 0);
	struct fake_ib_mr *mr = NULL;
	long cb0 = fake_kfd_callbacks_fired();

	fake_kfd_revoke_in_get_pages(1);
	CHECK(fake_ib_reg_mr(va, 8 * MiB, &dev1, &mr) == -ENODEV);
	CHECK(fake_kfd_callbacks_fired() == cb0 + 1);
	CHECK(fake_kfd_bad_put_calls() == 0);
	fake_kfd_revoke_in_get_pages(0);
	fake_kfd_free(va);
	check_balances(devs, 2);
	printf("ok: synchronous revoke inside get_pages (no deadlock, clean fail)\n");
}

static void test_huge_pin(void)
{
	/* 64 GiB pin (288 GB HBM sizing): 32768 chunks, hole every 1024 */
	uint64_t va = fake_kfd_alloc(64 * GiB, 1024);
	struct fake_ib_mr *mr = NULL;

	CHECK(fake_ib_reg_mr(va, 64 * GiB, &dev1, &mr) == 0);
	/* 32 runs of 2 GiB, each under the 4 GiB default max_seg */
	CHECK(mr->nmap == 32);
	CHECK(fake_ib_dereg_mr(mr) == 0);
	free(mr);
	fake_kfd_free(va);
	check_balances(devs, 2);
	printf("ok: 64 GiB pin -> 32 sg entries\n");
}

extern bool rocp2p_peer_ex;

static void test_ex_registration(void)
{
	/* Modern surface: default registration is a
	 * peer_memory_client_ex with PEER_MEM_INVALIDATE_UNMAPS; an
	 * ex-aware core then SKIPS dma_unmap on the invalidate path and
	 * the bridge's release() must reclaim the per-device mappings. */
	uint64_t va;
	struct fake_ib_mr *mr = NULL;

#if ROCNR_PEER_MEM_HAS_EX
	CHECK(fake_ib_client_is_ex() == 1);
	CHECK(fake_ib_client_flags() == PEER_MEM_INVALIDATE_UNMAPS);
#else
	CHECK(fake_ib_client_is_ex() == 0);
#endif
	va = fake_kfd_alloc(16 * MiB, 0);
	CHECK(fake_ib_reg_mr(va, 16 * MiB, &dev1, &mr) == 0);
	fake_kfd_free(va);	/* invalidate; ex core skips dma_unmap */
	CHECK(mr->invalidated == 1);
	CHECK(atomic64_read(&dev1.live_maps) == 0);	/* release cleaned */
	CHECK(fake_ib_dereg_mr(mr) == 0);
	free(mr);
	check_balances(devs, 2);
	printf("ok: ex registration + INVALIDATE_UNMAPS (core skips dma_unmap)\n");
}

static void test_plain_generation(void)
{
	/* peer_ex=0: the bridge registers the 2016-style plain client;
	 * the core sees no ex marker and uses the full teardown. */
	uint64_t va;
	struct fake_ib_mr *mr = NULL;

	rocnr_shim_module_exit();
	rocp2p_peer_ex = false;
	CHECK(rocnr_shim_module_init() == 0);
	CHECK(fake_ib_client_is_ex() == 0);
	CHECK(fake_ib_client_flags() == 0);
	va = fake_kfd_alloc(8 * MiB, 0);
	CHECK(fake_ib_reg_mr(va, 8 * MiB, &dev1, &mr) == 0);
	fake_kfd_free(va);
	CHECK(mr->invalidated == 1);
	CHECK(fake_ib_dereg_mr(mr) == 0);
	free(mr);
	check_balances(devs, 2);
	rocnr_shim_module_exit();
	rocp2p_peer_ex = true;
	CHECK(rocnr_shim_module_init() == 0);
	printf("ok: plain-generation registration (peer_ex=0)\n");
}

static void test_core_owned_invalidation_race(void)
{
	/* Newer rdma-core ordering: invalidate() returns immediately and
	 * the core tears the MR down LATER from its own thread — so KFD
	 * reclaims the pin long before put_pages/release arrive, while a
	 * concurrent ibv_dereg_mr may race the deferred teardown. */
	int ITERS = 1000;
	const char *env_iters = getenv("ROCNR_RACE_ITERS");
	unsigned int seed = 777;
	int i;

	if (env_iters && atoi(env_iters) > 0)
		ITERS = atoi(env_iters) / 3 + 1;
	fake_ib_set_async_invalidate(1);
	for (i = 0; i < ITERS; i++) {
		struct fake_ib_mr *mr = NULL;
		struct race_arg a;
		pthread_t th;
		struct timespec ts = { 0, 0 };

		a.va = fake_kfd_alloc(4 * MiB, 0);
		CHECK(fake_ib_reg_mr(a.va, 4 * MiB, &dev1, &mr) == 0);
		a.delay_ns = rand_r(&seed) % 20000;
		pthread_create(&th, NULL, race_free_thread, &a);
		ts.tv_nsec = rand_r(&seed) % 20000;
		nanosleep(&ts, NULL);
		fake_ib_dereg_mr(mr);
		pthread_join(th, NULL);
		fake_ib_quiesce();
		fake_ib_dereg_mr(mr);	/* idempotent */
		free(mr);
		CHECK(fake_kfd_bad_put_calls() == 0);
	}
	fake_ib_set_async_invalidate(0);
	check_balances(devs, 2);
	printf("ok: %d core-owned (deferred) invalidation races, clean\n",
	       ITERS);
}

static void test_bar_aperture_probe(void)
{
	/* init already probed an empty PCI table (warned).  Re-run module
	 * init against a small-BAR and a large-BAR topology and make sure
	 * both load fine (the check warns, never fails the load). */
	static struct pci_dev devs[] = {
		{ .vendor = 0x1002, .device = 0x75a3,
		  .class = 0x030000, .bar_len = { 256ULL << 20, 2 << 20 } },
		{ .vendor = 0x8086, .device = 0x1234,
		  .class = 0x020000, .bar_len = { 1 << 20, 0 } },
	};

	rocnr_shim_module_exit();
	rocnr_shim_set_pci_devices(devs, 2);
	CHECK(rocnr_shim_module_init() == 0);	/* small BAR: warns, loads */
	rocnr_shim_module_exit();

	devs[0].bar_len[0] = 288ULL << 30;	/* full-VRAM BAR */
	CHECK(rocnr_shim_module_init() == 0);
	rocnr_shim_set_pci_devices(0, 0);
	printf("ok: BAR aperture probe (small warns, large ok, load never fails)\n");
}

int main(void)
{
	test_coalesce_unit();

	CHECK(rocnr_shim_module_init() == 0);
	CHECK(fake_ib_client() != NULL);

	test_not_gpu();
	test_lifecycle();
	test_fragmented();
	test_max_seg();
	test_range_mismatch();
	test_map_failure_unwinds();
	test_two_devices();
	test_release_cleans_leftovers();
	test_invalidate();
	test_invalidate_race();
	test_page_size_fallback();
	test_null_dev_fallback();
	test_revoke_during_pin();
	test_ex_registration();
	test_plain_generation();
	test_core_owned_invalidation_race();
	test_huge_pin();
	test_bar_aperture_probe();

	rocnr_shim_module_exit();
	CHECK(fake_ib_client() == NULL);

	printf("ALL BRIDGE TESTS PASSED\n");
	return 0;
}
