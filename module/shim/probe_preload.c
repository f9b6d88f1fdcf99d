/* SPDX-License-Identifier: GPL-2.0 OR MIT
 *
 * probe_preload.c — LD_PRELOAD loopback for the probe CLI (VERDICT r1
 * item 8): lets the UNMODIFIED rocp2p_probe_cli binary execute its full
 * open → ioctl → mmap → verify → unpin flow in CI with no kernel
 * module and no GPU.  The library carries the real rocp2p_probe module
 * code compiled as userspace (same shim the unit suites use) plus the
 * fake KFD, and interposes open/ioctl/mmap:
 *
 *   open("/dev/rocp2p_probe")  -> probe fops->open on an in-process file
 *   ioctl(fd, ...)             -> fops->unlocked_ioctl (copy_to/from_user
 *                                 are in-process memcpy in the shim)
 *   mmap(..., fd, gpu_va)      -> fops->mmap into a recorder, then the
 *                                 recorded pfn ranges are materialized
 *                                 from the fake KFD's BACKED bus memory
 *                                 (snapshot semantics: a CPU-readback
 *                                 window, like the real BAR mapping)
 *
 * Env:
 *   ROCNR_PRELOAD_ALLOC_MIB=N   create one backed fake-GPU allocation
 *   ROCNR_PRELOAD_SEED=S        pattern its backing with
 *                               rocp2p_pattern_word(S, i) (default 0x42)
 *   ROCNR_PRELOAD_ADDR_FILE=P   write "0x<va> <bytes>" to P so the test
 *                               can hand the address to the CLI
 */
#define _GNU_SOURCE
#include <dlfcn.h>
#include <fcntl.h>
#include <stdarg.h>
#include <sys/mman.h>
#include <unistd.h>

#include "rocnr_shim_all.h"
#include "fake_kfd.h"
#include "../../rocnrdma_amd/ops/csrc/p2p_pattern.h"

static int (*real_open)(const char *, int, ...);
static void *(*real_mmap)(void *, size_t, int, int, int, off_t);
static int (*real_ioctl)(int, unsigned long, ...);

static const struct file_operations *pfops;
static struct file probe_file;
static int probe_fd = -1;

__attribute__((constructor)) static void preload_init(void)
{
	const char *mib, *seed_s, *out;

	real_open = dlsym(RTLD_NEXT, "open");
	real_mmap = dlsym(RTLD_NEXT, "mmap");
	real_ioctl = dlsym(RTLD_NEXT, "ioctl");

	if (rocnr_shim_module_init() != 0) {
		fprintf(stderr, "probe_preload: module init failed\n");
		return;
	}
	pfops = rocnr_shim_misc_dev()->fops;

	mib = getenv("ROCNR_PRELOAD_ALLOC_MIB");
	if (mib) {
		uint64_t bytes = (uint64_t)atoll(mib) << 20;
		uint64_t seed = 0x42, i;
		uint64_t va = fake_kfd_alloc_backed(bytes);
		uint64_t *w = fake_kfd_bus_to_ptr(
			(uint64_t)fake_kfd_backing_bus(va));

		seed_s = getenv("ROCNR_PRELOAD_SEED");
		if (seed_s)
			seed = strtoull(seed_s, 0, 0);
		for (i = 0; i < bytes / 8; i++)
			w[i] = rocp2p_pattern_word(seed, i);
		out = getenv("ROCNR_PRELOAD_ADDR_FILE");
		if (out) {
			FILE *f = fopen(out, "w");

			if (f) {
				fprintf(f, "0x%llx %llu\n",
					(unsigned long long)va,
					(unsigned long long)bytes);
				fclose(f);
			}
		}
	}
}

int open(const char *path, int flags, ...)
{
	mode_t mode = 0;

	if (flags & O_CREAT) {
		va_list ap;

		va_start(ap, flags);
		mode = va_arg(ap, mode_t);
		va_end(ap);
	}
	if (pfops && !strcmp(path, "/dev/rocp2p_probe")) {
		int fd = real_open("/dev/null", O_RDWR);

		if (fd < 0)
			return -1;
		memset(&probe_file, 0, sizeof(probe_file));
		if (pfops->open(0, &probe_file) != 0) {
			close(fd);
			errno = EIO;
			return -1;
		}
		probe_fd = fd;
		return fd;
	}
	return real_open(path, flags, mode);
}

int open64(const char *path, int flags, ...) __attribute__((alias("open")));

int ioctl(int fd, unsigned long req, ...)
{
	void *argp;
	va_list ap;

	va_start(ap, req);
	argp = va_arg(ap, void *);
	va_end(ap);
	if (fd == probe_fd && pfops) {
		long r = pfops->unlocked_ioctl(&probe_file, (unsigned int)req,
					       (unsigned long)argp);

		if (r < 0) {
			errno = (int)-r;
			return -1;
		}
		return (int)r;
	}
	return real_ioctl(fd, req, argp);
}

void *mmap(void *addr, size_t len, int prot, int flags, int fd, off_t off)
{
	if (fd == probe_fd && pfops) {
		struct vm_area_struct vma;
		uint8_t *base;
		long i, n;
		int r;

		base = real_mmap(NULL, len, PROT_READ | PROT_WRITE,
				 MAP_PRIVATE | MAP_ANONYMOUS, -1, 0);
		if (base == MAP_FAILED)
			return MAP_FAILED;
		memset(&vma, 0, sizeof(vma));
		vma.vm_start = (unsigned long)base;
		vma.vm_end = (unsigned long)base + len;
		vma.vm_pgoff = (unsigned long)(off >> PAGE_SHIFT);
		rocnr_shim_maps_reset();
		r = pfops->mmap(&probe_file, &vma);
		if (r < 0) {
			munmap(base, len);
			errno = -r;
			return MAP_FAILED;
		}
		/* materialize the recorded BAR ranges (CPU snapshot) */
		n = rocnr_shim_maps_count();
		for (i = 0; i < n; i++) {
			const struct rocnr_shim_map *m = rocnr_shim_maps_get(i);
			void *src = fake_kfd_bus_to_ptr((uint64_t)m->pfn
							<< PAGE_SHIFT);

			if (src)
				memcpy((void *)m->vaddr, src, m->size);
		}
		if (!(prot & PROT_WRITE))
			mprotect(base, len, prot);
		return base;
	}
	return real_mmap(addr, len, prot, flags, fd, off);
}

void *mmap64(void *addr, size_t len, int prot, int flags, int fd,
	     off_t off) __attribute__((alias("mmap")));
