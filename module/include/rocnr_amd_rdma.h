/* SPDX-License-Identifier: GPL-2.0 OR MIT
 *
 * rocnr_amd_rdma.h — vendored amdkfd RDMA (P2P pinning) interface.
 *
 * The reference (reference: /root/reference/amdp2p.c:45,
 * /root/reference/Makefile:23-26) compiles against "amd_rdma.h" from the
 * ROCK kernel tree, which is not vendored there.  Shapes are re-derived
 * from the reference's call sites:
 *   is_gpu_address(addr, pid)                       amdp2p.c:127
 *   get_pages(addr, size, pid, &info, cb, priv)     amdp2p.c:200-205
 *   put_pages(&info)                                amdp2p.c:305
 *   get_page_size(va, size, pid, &page_size)        amdp2p.c:328-332
 *   amd_p2p_info { va, size, pages:sg_table* }      amdp2p.c:78, 258-261
 *   amdkfd_query_rdma_interface(&vtable)            amdp2p.c:381
 *
 * MI355X-era drift: the modern ROCK-Kernel-Driver get_pages takes the DMA
 * target device so KFD can produce device-mapped (IOMMU-aware) bus
 * addresses.  Switch:
 *
 *   ROCNR_AMD_RDMA_HAS_DMA_DEV  (default 1)
 *       1: get_pages(addr, length, pid, dma_dev, ...)   [modern ROCK]
 *       0: get_pages(addr, length, pid, ...)            [2016 KFD]
 *
 * When building against a real ROCK tree (AMD_RDMA_DIR set), the module
 * Makefile defines ROCNR_USE_SYSTEM_AMD_RDMA and this header includes the
 * system copy instead.
 */
#ifndef ROCNR_AMD_RDMA_H_
#define ROCNR_AMD_RDMA_H_

#ifndef ROCNR_AMD_RDMA_HAS_DMA_DEV
#define ROCNR_AMD_RDMA_HAS_DMA_DEV 1
#endif

#ifdef ROCNR_USE_SYSTEM_AMD_RDMA
/* Real ROCK tree (or the third_party/ reconstructions).  The
 * ROCNR_AMD_RDMA_HAS_DMA_DEV switch must be set to match the tree's
 * get_pages signature; the bridge's call site compiles against it, so
 * a wrong setting is a hard compile error (arg-count mismatch). */
#include <drm/amd_rdma.h>
#else

#include <linux/types.h>
#include <linux/scatterlist.h>

struct pid;
struct device;

/* One pinned GPU range.  `pages` holds the range as a scatter table whose
 * dma_address/dma_length entries are bus addresses in the GPU's PCIe BAR
 * aperture (2 MB VRAM page granularity on MI355X; a 64 GB pin of
 * physically contiguous VRAM may coalesce to a handful of entries). */
struct amd_p2p_info {
	uint64_t	 va;
	uint64_t	 size;
	struct pid	*pid;
	struct sg_table	*pages;
	void		*priv;	/* owned by KFD */
};

struct amd_rdma_interface {
	int (*get_pages)(uint64_t address, uint64_t length, struct pid *pid,
#if ROCNR_AMD_RDMA_HAS_DMA_DEV
			 struct device *dma_dev,
#endif
			 struct amd_p2p_info **amd_p2p_data,
			 void (*free_callback)(void *client_priv),
			 void *client_priv);
	int (*put_pages)(struct amd_p2p_info **amd_p2p_data);
	int (*is_gpu_address)(uint64_t address, struct pid *pid);
	int (*get_page_size)(uint64_t address, uint64_t length,
			     struct pid *pid, unsigned long *page_size);
};

int amdkfd_query_rdma_interface(const struct amd_rdma_interface **rdma);

#endif /* !ROCNR_USE_SYSTEM_AMD_RDMA */
#endif /* ROCNR_AMD_RDMA_H_ */
