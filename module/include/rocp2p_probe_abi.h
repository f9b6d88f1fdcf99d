/* SPDX-License-Identifier: GPL-2.0 OR MIT
 *
 * rocp2p_probe_abi.h — ioctl ABI of the rocp2p_probe char device, shared
 * between the kernel module (module/probe/) and userspace clients
 * (tools/rocp2p_probe_cli.c, rocnrdma_amd/probe/).
 *
 * Re-design of the reference ABI (reference:
 * /root/reference/include/amdp2ptest.h:27-72) with its defects fixed:
 *  - every ioctl that copies results back to userspace is _IOWR (the
 *    reference declared IS_GPU_ADDRESS and PUT_PAGES _IOW although the
 *    handlers copy_to_user — amdp2ptest.h:68-72, amdp2ptest.c:156,248);
 *  - ioctl 4 no longer names a nonexistent struct type
 *    (amdp2ptest.h:71-72 typo);
 *  - codes pass the struct, not a pointer-to-struct, so _IOC_SIZE is
 *    meaningful and the kernel can bound-check;
 *  - adds GET_INFO (nents/page_size readback of a pinned range) so
 *    userspace can verify sg coalescing without mmap.
 */
#ifndef ROCP2P_PROBE_ABI_H_
#define ROCP2P_PROBE_ABI_H_

#ifdef __KERNEL__
#include <linux/types.h>
#include <linux/ioctl.h>
#else
#include <stdint.h>
#include <sys/ioctl.h>
#endif

#define ROCP2P_PROBE_DEVICE_NAME "rocp2p_probe"
#define ROCP2P_PROBE_DEVICE_PATH "/dev/rocp2p_probe"

#define ROCP2P_PROBE_IOCTL_MAGIC 'R'

struct rocp2p_probe_page_size {
	__extension__ uint64_t addr;	/* in: GPU VA */
	uint64_t length;		/* in */
	uint64_t page_size;		/* out */
};

struct rocp2p_probe_pin {
	uint64_t addr;			/* in: GPU VA */
	uint64_t length;		/* in */
};

struct rocp2p_probe_unpin {
	uint64_t addr;			/* in */
	uint64_t length;		/* in */
	uint64_t released;		/* out: #registrations released */
};

struct rocp2p_probe_is_gpu {
	uint64_t addr;			/* in */
	uint64_t is_gpu;		/* out: 0/1 */
};

struct rocp2p_probe_info {
	uint64_t addr;			/* in */
	uint64_t length;		/* in */
	uint64_t nents;			/* out: sg entries of the pin */
	uint64_t total_bytes;		/* out: Σ sg lengths */
	uint64_t first_dma_addr;	/* out: bus addr of entry 0 */
	uint64_t max_seg_bytes;		/* out: largest sg entry */
};

#define ROCP2P_PROBE_GET_PAGE_SIZE \
	_IOWR(ROCP2P_PROBE_IOCTL_MAGIC, 1, struct rocp2p_probe_page_size)
#define ROCP2P_PROBE_GET_PAGES \
	_IOW(ROCP2P_PROBE_IOCTL_MAGIC, 2, struct rocp2p_probe_pin)
#define ROCP2P_PROBE_PUT_PAGES \
	_IOWR(ROCP2P_PROBE_IOCTL_MAGIC, 3, struct rocp2p_probe_unpin)
#define ROCP2P_PROBE_IS_GPU_ADDRESS \
	_IOWR(ROCP2P_PROBE_IOCTL_MAGIC, 4, struct rocp2p_probe_is_gpu)
#define ROCP2P_PROBE_GET_INFO \
	_IOWR(ROCP2P_PROBE_IOCTL_MAGIC, 5, struct rocp2p_probe_info)

#endif /* ROCP2P_PROBE_ABI_H_ */
