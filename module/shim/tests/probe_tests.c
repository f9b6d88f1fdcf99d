/* SPDX-License-Identifier: GPL-2.0 OR MIT
 *
 * probe_tests.c — userspace unit + race tests of the rocp2p_probe char
 * device, driven through its file_operations against the fake KFD.
 * Covers every ioctl, the leak-proof close, duplicate-pin release
 * semantics, the fixed mmap (all sg entries, per-entry lengths, correct
 * containment), and the revoke races the reference could not test.
 */
#define __ROCNR_SHIM__ 1
#include "rocnr_shim_all.h"
#include "fake_kfd.h"
#include "rocp2p_probe_abi.h"

#include <assert.h>
#include <time.h>
#include <stdlib.h>

#define CHECK(cond) do { \
	if (!(cond)) { \
		fprintf(stderr, "FAIL %s:%d: %s\n", __FILE__, __LINE__, #cond); \
		exit(1); \
	} \
} while (0)

#define MiB (1ULL << 20)

static const struct file_operations *fops;

static struct file *dev_open(void)
{
	struct file *f = calloc(1, sizeof(*f));

	CHECK(fops->open(0, f) == 0);
	return f;
}

static void dev_close(struct file *f)
{
	CHECK(fops->release(0, f) == 0);
	free(f);
}

static long xioctl(struct file *f, unsigned int cmd, void *arg)
{
	return fops->unlocked_ioctl(f, cmd, (unsigned long)arg);
}

static void test_is_gpu_and_page_size(void)
{
	struct file *f = dev_open();
	uint64_t va = fake_kfd_alloc(8 * MiB, 0);
	struct rocp2p_probe_is_gpu ig = { .addr = va };
	struct rocp2p_probe_page_size ps = { .addr = va, .length = 8 * MiB };

	CHECK(xioctl(f, ROCP2P_PROBE_IS_GPU_ADDRESS, &ig) == 0);
	CHECK(ig.is_gpu == 1);
	ig.addr = 0x1234;
	CHECK(xioctl(f, ROCP2P_PROBE_IS_GPU_ADDRESS, &ig) == 0);
	CHECK(ig.is_gpu == 0);

	CHECK(xioctl(f, ROCP2P_PROBE_GET_PAGE_SIZE, &ps) == 0);
	CHECK(ps.page_size == FAKE_KFD_VRAM_PAGE);

	dev_close(f);
	fake_kfd_free(va);
	printf("ok: is_gpu_address + get_page_size ioctls\n");
}

static void test_pin_info_unpin(void)
{
	struct file *f = dev_open();
	uint64_t va = fake_kfd_alloc(64 * MiB, 4);	/* 32 chunks, holes */
	struct rocp2p_probe_pin pin = { .addr = va, .length = 64 * MiB };
	struct rocp2p_probe_info info = { .addr = va, .length = 64 * MiB };
	struct rocp2p_probe_unpin unpin = { .addr = va, .length = 64 * MiB };

	CHECK(xioctl(f, ROCP2P_PROBE_GET_PAGES, &pin) == 0);
	CHECK(fake_kfd_live_pins() == 1);

	CHECK(xioctl(f, ROCP2P_PROBE_GET_INFO, &info) == 0);
	CHECK(info.nents == 32);	/* raw KFD granularity, no coalesce */
	CHECK(info.total_bytes == 64 * MiB);
	CHECK(info.max_seg_bytes == FAKE_KFD_VRAM_PAGE);

	CHECK(xioctl(f, ROCP2P_PROBE_PUT_PAGES, &unpin) == 0);
	CHECK(unpin.released == 1);
	CHECK(fake_kfd_live_pins() == 0);
	/* second put: nothing left */
	CHECK(xioctl(f, ROCP2P_PROBE_PUT_PAGES, &unpin) == -ENOENT);

	dev_close(f);
	fake_kfd_free(va);
	printf("ok: pin -> info -> unpin\n");
}

static void test_duplicate_pins_released_together(void)
{
	struct file *f = dev_open();
	uint64_t va = fake_kfd_alloc(4 * MiB, 0);
	struct rocp2p_probe_pin pin = { .addr = va, .length = 4 * MiB };
	struct rocp2p_probe_unpin unpin = { .addr = va, .length = 4 * MiB };

	CHECK(xioctl(f, ROCP2P_PROBE_GET_PAGES, &pin) == 0);
	CHECK(xioctl(f, ROCP2P_PROBE_GET_PAGES, &pin) == 0);
	CHECK(xioctl(f, ROCP2P_PROBE_GET_PAGES, &pin) == 0);
	CHECK(fake_kfd_live_pins() == 3);
	CHECK(xioctl(f, ROCP2P_PROBE_PUT_PAGES, &unpin) == 0);
	CHECK(unpin.released == 3);
	CHECK(fake_kfd_live_pins() == 0);
	dev_close(f);
	fake_kfd_free(va);
	printf("ok: duplicate pins all released by one PUT\n");
}

static void test_close_unpins_everything(void)
{
	struct file *f = dev_open();
	uint64_t va1 = fake_kfd_alloc(4 * MiB, 0);
	uint64_t va2 = fake_kfd_alloc(2 * MiB, 0);
	struct rocp2p_probe_pin p1 = { .addr = va1, .length = 4 * MiB };
	struct rocp2p_probe_pin p2 = { .addr = va2, .length = 2 * MiB };

	CHECK(xioctl(f, ROCP2P_PROBE_GET_PAGES, &p1) == 0);
	CHECK(xioctl(f, ROCP2P_PROBE_GET_PAGES, &p2) == 0);
	CHECK(fake_kfd_live_pins() == 2);
	dev_close(f);	/* leak-proof close */
	CHECK(fake_kfd_live_pins() == 0);
	fake_kfd_free(va1);
	fake_kfd_free(va2);
	printf("ok: close unpins everything\n");
}

static void test_pin_errors(void)
{
	struct file *f = dev_open();
	struct rocp2p_probe_pin bad = { .addr = 0xdead000, .length = MiB };

	CHECK(xioctl(f, ROCP2P_PROBE_GET_PAGES, &bad) == -EINVAL);
	bad.length = 0;
	CHECK(xioctl(f, ROCP2P_PROBE_GET_PAGES, &bad) == -EINVAL);
	CHECK(xioctl(f, 0xDEAD, &bad) == -ENOTTY);
	dev_close(f);
	CHECK(rocnr_shim_alloc_balance() == 0);
	printf("ok: error paths leak nothing\n");
}

static void test_revoke_unlinks(void)
{
	struct file *f = dev_open();
	uint64_t va = fake_kfd_alloc(4 * MiB, 0);
	struct rocp2p_probe_pin pin = { .addr = va, .length = 4 * MiB };
	struct rocp2p_probe_unpin unpin = { .addr = va, .length = 4 * MiB };
	struct rocp2p_probe_info info = { .addr = va, .length = 4 * MiB };

	CHECK(xioctl(f, ROCP2P_PROBE_GET_PAGES, &pin) == 0);
	fake_kfd_free(va);	/* async revoke */
	CHECK(fake_kfd_live_pins() == 0);
	/* the pin is gone from the fd's list too */
	CHECK(xioctl(f, ROCP2P_PROBE_GET_INFO, &info) == -ENOENT);
	CHECK(xioctl(f, ROCP2P_PROBE_PUT_PAGES, &unpin) == -ENOENT);
	CHECK(fake_kfd_bad_put_calls() == 0);
	dev_close(f);
	printf("ok: async revoke unlinks the pin safely\n");
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
	return 0;
}

static void test_revoke_vs_put_race(void)
{
	int ITERS = 2000;
	const char *env_iters = getenv("ROCNR_RACE_ITERS");

	if (env_iters && atoi(env_iters) > 0)
		ITERS = atoi(env_iters);
	unsigned int seed = 777;
	int i;

	for (i = 0; i < ITERS; i++) {
		struct file *f = dev_open();
		struct race_arg a;
		pthread_t th;
		struct timespec ts = { 0, 0 };
		struct rocp2p_probe_pin pin;
		struct rocp2p_probe_unpin unpin;

		a.va = fake_kfd_alloc(2 * MiB, 0);
		a.delay_ns = rand_r(&seed) % 15000;
		pin.addr = a.va;
		pin.length = 2 * MiB;
		CHECK(xioctl(f, ROCP2P_PROBE_GET_PAGES, &pin) == 0);
		pthread_create(&th, 0, race_free_thread, &a);
		ts.tv_nsec = rand_r(&seed) % 15000;
		nanosleep(&ts, NULL);
		unpin.addr = a.va;
		unpin.length = 2 * MiB;
		xioctl(f, ROCP2P_PROBE_PUT_PAGES, &unpin); /* may be -ENOENT */
		pthread_join(th, 0);
		dev_close(f);
		CHECK(fake_kfd_bad_put_calls() == 0);
		CHECK(fake_kfd_live_pins() == 0);
	}
	CHECK(rocnr_shim_alloc_balance() == 0);
	CHECK(rocnr_shim_pid_balance() == 0);
	printf("ok: %d unpin-vs-revoke races, no bad put, no leak\n", ITERS);
}

static void test_mmap_maps_every_segment(void)
{
	struct file *f = dev_open();
	/* 8 MiB with a hole every 2 chunks -> 2 runs of 2 chunks */
	uint64_t va = fake_kfd_alloc(8 * MiB, 2);
	struct rocp2p_probe_pin pin = { .addr = va, .length = 8 * MiB };
	struct vm_area_struct vma;
	long i;
	unsigned long total = 0;

	CHECK(xioctl(f, ROCP2P_PROBE_GET_PAGES, &pin) == 0);

	rocnr_shim_maps_reset();
	memset(&vma, 0, sizeof(vma));
	vma.vm_start = 0x40000000;
	vma.vm_end = 0x40000000 + 8 * MiB;
	vma.vm_pgoff = va >> PAGE_SHIFT;
	CHECK(fops->mmap(f, &vma) == 0);
	/* every KFD segment in range must be mapped (4 chunks) with its
	 * own length — the reference mapped only the first sg entry */
	CHECK(rocnr_shim_maps_count() == 4);
	for (i = 0; i < rocnr_shim_maps_count(); i++) {
		const struct rocnr_shim_map *m = rocnr_shim_maps_get(i);

		CHECK(m->size == FAKE_KFD_VRAM_PAGE);
		total += m->size;
		if (i)
			CHECK(m->vaddr ==
			      rocnr_shim_maps_get(i - 1)->vaddr +
				      rocnr_shim_maps_get(i - 1)->size);
	}
	CHECK(total == 8 * MiB);

	/* sub-range mmap: second half only */
	rocnr_shim_maps_reset();
	memset(&vma, 0, sizeof(vma));
	vma.vm_start = 0x50000000;
	vma.vm_end = 0x50000000 + 4 * MiB;
	vma.vm_pgoff = (va + 4 * MiB) >> PAGE_SHIFT;
	CHECK(fops->mmap(f, &vma) == 0);
	CHECK(rocnr_shim_maps_count() == 2);

	/* vma exceeding the pin must be rejected (reference accepted it) */
	rocnr_shim_maps_reset();
	memset(&vma, 0, sizeof(vma));
	vma.vm_start = 0x60000000;
	vma.vm_end = 0x60000000 + 16 * MiB;
	vma.vm_pgoff = va >> PAGE_SHIFT;
	CHECK(fops->mmap(f, &vma) == -ENOENT);
	CHECK(rocnr_shim_maps_count() == 0);

	dev_close(f);
	fake_kfd_free(va);
	printf("ok: mmap maps all segments, honors sub-ranges, rejects overrun\n");
}

static void test_mmap_rejects_subpage_offset(void)
{
	/* a pin whose BAR address lands at a sub-page offset
	 * must be refused (-EINVAL), not silently truncated by the pfn
	 * shift (which would map the wrong bytes).  Skew the fake BAR by
	 * 0x100 so every sg_dma_address is sub-page misaligned. */
	struct file *f = dev_open();
	uint64_t va;
	struct rocp2p_probe_pin pin;
	struct vm_area_struct vma;

	fake_kfd_bus_skew(0x100);
	va = fake_kfd_alloc(8 * MiB, 0);
	pin.addr = va;
	pin.length = 4 * MiB;
	CHECK(xioctl(f, ROCP2P_PROBE_GET_PAGES, &pin) == 0);
	rocnr_shim_maps_reset();
	memset(&vma, 0, sizeof(vma));
	vma.vm_start = 0x70000000;
	vma.vm_end = 0x70000000 + 1 * MiB;
	vma.vm_pgoff = va >> PAGE_SHIFT;
	CHECK(fops->mmap(f, &vma) == -EINVAL);
	CHECK(rocnr_shim_maps_count() == 0);
	dev_close(f);
	fake_kfd_free(va);
	fake_kfd_bus_skew(0);
	printf("ok: mmap rejects sub-page BAR offsets (-EINVAL, no truncation)\n");
}

int main(void)
{
	CHECK(rocnr_shim_module_init() == 0);
	fops = rocnr_shim_misc_dev()->fops;
	CHECK(fops && fops->open && fops->unlocked_ioctl && fops->mmap);

	test_is_gpu_and_page_size();
	test_pin_info_unpin();
	test_duplicate_pins_released_together();
	test_close_unpins_everything();
	test_pin_errors();
	test_revoke_unlinks();
	test_revoke_vs_put_race();
	test_mmap_maps_every_segment();
	test_mmap_rejects_subpage_offset();

	rocnr_shim_module_exit();
	CHECK(rocnr_shim_misc_dev() == NULL);
	CHECK(rocnr_shim_alloc_balance() == 0);
	CHECK(rocnr_shim_pid_balance() == 0);
	CHECK(rocnr_shim_sg_balance() == 0);

	printf("ALL PROBE TESTS PASSED\n");
	return 0;
}
