// SPDX-License-Identifier: MIT
/*
 * rocp2p_probe_cli — userspace client for /dev/rocp2p_probe.
 *
 * The reference defined the probe ioctl ABI but never shipped a client
 * (SURVEY.md §4 "the expected userspace ioctl client is absent"); this
 * is that client, against the fixed ABI (module/include/
 * rocp2p_probe_abi.h).
 *
 * Build (HIP mode, allocates GPU memory itself):
 *   hipcc -DWITH_HIP -O2 tools/rocp2p_probe_cli.c -Imodule/include \
 *         -o build/rocp2p_probe_cli
 * Build (plain mode, operates on caller-supplied addresses):
 *   gcc -O2 tools/rocp2p_probe_cli.c -Imodule/include -o rocp2p_probe_cli
 *
 * Commands:
 *   is-gpu <hexaddr>           address classification
 *   pagesize <hexaddr> <len>   GPU page size of a range
 *   pin <hexaddr> <len>        pin (holds until unpin/exit)
 *   info <hexaddr> <len>       sg shape of a pinned range
 *   selftest <MiB>             (HIP builds) hipMalloc + full flow:
 *                              pin, double-pin, info, page size, mmap
 *                              CPU readback vs GPU-written pattern,
 *                              unpin-all
 */
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <stdint.h>
#include <errno.h>
#include <fcntl.h>
#include <unistd.h>
#include <sys/ioctl.h>
#include <sys/mman.h>

#include "rocp2p_probe_abi.h"
#include "../rocnrdma_amd/ops/csrc/p2p_pattern.h"

#ifdef WITH_HIP
#include <hip/hip_runtime.h>
#define HIP_CHECK(x)                                                    \
	do {                                                            \
		hipError_t e_ = (x);                                    \
		if (e_ != hipSuccess) {                                 \
			fprintf(stderr, "%s failed: %s\n", #x,          \
				hipGetErrorString(e_));                 \
			exit(1);                                        \
		}                                                       \
	} while (0)
#endif

static int dev_fd;

static void need(int ok, const char *what)
{
	if (!ok) {
		fprintf(stderr, "FAIL: %s (errno %d: %s)\n", what, errno,
			strerror(errno));
		exit(1);
	}
}

static int cmd_is_gpu(uint64_t addr)
{
	struct rocp2p_probe_is_gpu p = { .addr = addr };

	need(ioctl(dev_fd, ROCP2P_PROBE_IS_GPU_ADDRESS, &p) == 0,
	     "IS_GPU_ADDRESS ioctl");
	printf("0x%llx: %s\n", (unsigned long long)addr,
	       p.is_gpu ? "GPU address" : "not a GPU address");
	return p.is_gpu ? 0 : 1;
}

static int cmd_pagesize(uint64_t addr, uint64_t len)
{
	struct rocp2p_probe_page_size p = { .addr = addr, .length = len };

	need(ioctl(dev_fd, ROCP2P_PROBE_GET_PAGE_SIZE, &p) == 0,
	     "GET_PAGE_SIZE ioctl");
	printf("page_size: %llu\n", (unsigned long long)p.page_size);
	return 0;
}

static int cmd_pin(uint64_t addr, uint64_t len)
{
	struct rocp2p_probe_pin p = { .addr = addr, .length = len };

	need(ioctl(dev_fd, ROCP2P_PROBE_GET_PAGES, &p) == 0,
	     "GET_PAGES ioctl");
	printf("pinned 0x%llx +%llu\n", (unsigned long long)addr,
	       (unsigned long long)len);
	return 0;
}

static int cmd_info(uint64_t addr, uint64_t len)
{
	struct rocp2p_probe_info p = { .addr = addr, .length = len };

	need(ioctl(dev_fd, ROCP2P_PROBE_GET_INFO, &p) == 0,
	     "GET_INFO ioctl");
	printf("nents %llu total %llu first_dma 0x%llx max_seg %llu\n",
	       (unsigned long long)p.nents,
	       (unsigned long long)p.total_bytes,
	       (unsigned long long)p.first_dma_addr,
	       (unsigned long long)p.max_seg_bytes);
	return 0;
}

/* Full flow on a CALLER-SUPPLIED range (no GPU needed): pin, duplicate
 * pin, info, page size, mmap CPU window, optional splitmix64 pattern
 * verification (the preload loopback patterns the backing with the
 * same generator — rocnrdma_amd/ops/csrc/p2p_pattern.h), unpin-all.
 * This is the flow CI executes through the UNMODIFIED binary via
 * module/shim/probe_preload.c (LD_PRELOAD). */
static int cmd_selftest_extern(uint64_t addr, uint64_t len, uint64_t seed,
			       int have_seed)
{
	struct rocp2p_probe_pin pin = { .addr = addr, .length = len };
	struct rocp2p_probe_unpin unpin = { .addr = addr, .length = len };
	struct rocp2p_probe_info info = { .addr = addr, .length = len };
	struct rocp2p_probe_page_size ps = { .addr = addr, .length = len };
	uint64_t *map, i, words = len / 8, bad = 0;

	need(cmd_is_gpu(addr) == 0, "address classified");
	need(ioctl(dev_fd, ROCP2P_PROBE_GET_PAGE_SIZE, &ps) == 0,
	     "GET_PAGE_SIZE");
	printf("page_size %llu\n", (unsigned long long)ps.page_size);

	need(ioctl(dev_fd, ROCP2P_PROBE_GET_PAGES, &pin) == 0, "pin #1");
	need(ioctl(dev_fd, ROCP2P_PROBE_GET_PAGES, &pin) == 0,
	     "pin #2 (duplicate range)");

	need(ioctl(dev_fd, ROCP2P_PROBE_GET_INFO, &info) == 0, "GET_INFO");
	printf("nents %llu total %llu max_seg %llu\n",
	       (unsigned long long)info.nents,
	       (unsigned long long)info.total_bytes,
	       (unsigned long long)info.max_seg_bytes);
	need(info.total_bytes == len, "sg covers the pin");

	map = (uint64_t *)mmap(NULL, len, PROT_READ, MAP_SHARED, dev_fd,
			       (off_t)addr);
	need(map != MAP_FAILED, "mmap CPU window");
	if (have_seed) {
		for (i = 0; i < words; i++)
			bad += (map[i] != rocp2p_pattern_word(seed, i));
		need(bad == 0, "mmap readback matches pattern");
		printf("mmap readback: %llu words OK\n",
		       (unsigned long long)words);
	}
	munmap(map, len);

	need(ioctl(dev_fd, ROCP2P_PROBE_PUT_PAGES, &unpin) == 0, "unpin");
	need(unpin.released == 2, "both duplicate pins released");
	printf("SELFTEST-EXTERN PASSED\n");
	return 0;
}

#ifdef WITH_HIP
static int cmd_selftest(uint64_t mib)
{
	uint64_t len = mib << 20;
	void *gpu = NULL;
	struct rocp2p_probe_pin pin;
	struct rocp2p_probe_unpin unpin;
	struct rocp2p_probe_info info;
	struct rocp2p_probe_page_size ps;
	uint32_t *host, *map;
	uint64_t i, words = len / 4;

	HIP_CHECK(hipMalloc(&gpu, len));
	/* GPU-side pattern */
	host = (uint32_t *)malloc(len);
	for (i = 0; i < words; i++)
		host[i] = (uint32_t)(0x9E3779B9u * (i + 1));
	HIP_CHECK(hipMemcpy(gpu, host, len, hipMemcpyHostToDevice));

	need(cmd_is_gpu((uint64_t)(uintptr_t)gpu) == 0, "address classified");

	ps.addr = (uint64_t)(uintptr_t)gpu;
	ps.length = len;
	need(ioctl(dev_fd, ROCP2P_PROBE_GET_PAGE_SIZE, &ps) == 0,
	     "GET_PAGE_SIZE");
	printf("page_size %llu\n", (unsigned long long)ps.page_size);

	pin.addr = (uint64_t)(uintptr_t)gpu;
	pin.length = len;
	need(ioctl(dev_fd, ROCP2P_PROBE_GET_PAGES, &pin) == 0, "pin #1");
	need(ioctl(dev_fd, ROCP2P_PROBE_GET_PAGES, &pin) == 0,
	     "pin #2 (duplicate range)");

	info.addr = pin.addr;
	info.length = pin.length;
	need(ioctl(dev_fd, ROCP2P_PROBE_GET_INFO, &info) == 0, "GET_INFO");
	printf("nents %llu total %llu max_seg %llu\n",
	       (unsigned long long)info.nents,
	       (unsigned long long)info.total_bytes,
	       (unsigned long long)info.max_seg_bytes);
	need(info.total_bytes == len, "sg covers the pin");

	/* CPU window over the pinned VRAM (BAR readback) */
	map = (uint32_t *)mmap(NULL, len, PROT_READ, MAP_SHARED, dev_fd,
			       (off_t)pin.addr);
	if (map == MAP_FAILED) {
		printf("mmap readback unavailable (%s) — skipping "
		       "data check\n", strerror(errno));
	} else {
		uint64_t bad = 0;

		for (i = 0; i < words; i++)
			bad += (map[i] != host[i]);
		need(bad == 0, "mmap readback matches GPU pattern");
		printf("mmap readback: %llu words OK\n",
		       (unsigned long long)words);
		munmap(map, len);
	}

	unpin.addr = pin.addr;
	unpin.length = pin.length;
	need(ioctl(dev_fd, ROCP2P_PROBE_PUT_PAGES, &unpin) == 0, "unpin");
	need(unpin.released == 2, "both duplicate pins released");

	HIP_CHECK(hipFree(gpu));
	free(host);
	printf("SELFTEST PASSED\n");
	return 0;
}
#endif

int main(int argc, char **argv)
{
	if (argc < 2) {
		fprintf(stderr,
			"usage: %s is-gpu|pagesize|pin|info|selftest ...\n",
			argv[0]);
		return 2;
	}
	dev_fd = open(ROCP2P_PROBE_DEVICE_PATH, O_RDWR);
	need(dev_fd >= 0, "open " ROCP2P_PROBE_DEVICE_PATH
	     " (is rocp2p_probe.ko loaded?)");

	if (!strcmp(argv[1], "is-gpu") && argc == 3)
		return cmd_is_gpu(strtoull(argv[2], 0, 16));
	if (!strcmp(argv[1], "pagesize") && argc == 4)
		return cmd_pagesize(strtoull(argv[2], 0, 16),
				    strtoull(argv[3], 0, 0));
	if (!strcmp(argv[1], "pin") && argc == 4)
		return cmd_pin(strtoull(argv[2], 0, 16),
			       strtoull(argv[3], 0, 0));
	if (!strcmp(argv[1], "info") && argc == 4)
		return cmd_info(strtoull(argv[2], 0, 16),
				strtoull(argv[3], 0, 0));
	if (!strcmp(argv[1], "selftest-extern") && (argc == 4 || argc == 5))
		return cmd_selftest_extern(strtoull(argv[2], 0, 16),
					   strtoull(argv[3], 0, 0),
					   argc == 5 ?
						strtoull(argv[4], 0, 0) : 0,
					   argc == 5);
#ifdef WITH_HIP
	if (!strcmp(argv[1], "selftest"))
		return cmd_selftest(argc > 2 ? strtoull(argv[2], 0, 0) : 64);
#endif
	fprintf(stderr, "unknown/invalid command\n");
	return 2;
}
