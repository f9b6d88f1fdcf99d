/* SPDX-License-Identifier: GPL-2.0 OR MIT
 *
 * fake_ibcore.h — userspace stand-in for the OFED PeerDirect dispatcher
 * (the L3 layer of the reference stack).  Drives a
 * registered peer_memory_client exactly the way ib_umem_get's peer path
 * does — acquire → get_pages → get_page_size → dma_map on registration,
 * dma_unmap → put_pages → release on deregistration, and synchronous MR
 * teardown when the client calls the invalidate entry point — so bridge
 * logic is exercised against the real callback ordering with no OFED.
 */
#ifndef ROCNR_FAKE_IBCORE_H_
#define ROCNR_FAKE_IBCORE_H_

#include "rocnr_shim_all.h"
#include "rocnr_peer_mem.h"

#ifdef __cplusplus
extern "C" {
#endif

struct fake_ib_mr {
	void *client_context;
	struct sg_table sgt;	/* filled by dma_map */
	int nmap;
	unsigned long page_size;
	struct device *dev;
	int invalidated;	/* torn down via the invalidate path */
	int dead;		/* fully released */
	struct mutex lock;
};

/* ibv_reg_mr equivalent.  Returns 0 and *out on success; -ENODEV if no
 * client claimed the address (CPU fallback in real OFED); other negative
 * errno from the client. */
int fake_ib_reg_mr(unsigned long addr, size_t size, struct device *dev,
		   struct fake_ib_mr **out);

/* Map the same registration for a second device (multi-HCA case). */
int fake_ib_mr_map_also(struct fake_ib_mr *mr, struct device *dev2,
			struct sg_table *sgt_out, int *nmap_out);
int fake_ib_mr_unmap_also(struct fake_ib_mr *mr, struct device *dev2,
			  struct sg_table *sgt);

/* ibv_dereg_mr equivalent; idempotent vs. invalidation. */
int fake_ib_dereg_mr(struct fake_ib_mr *mr);

long fake_ib_invalidate_count(void);
void fake_ib_reset_stats(void);

/* The registered client (NULL when none). */
const struct peer_memory_client *fake_ib_client(void);

/* Modern-core (MOFED 5.x generation) surface: extended-registration
 * detection and capability flags, readable by tests. */
int fake_ib_client_is_ex(void);
u32 fake_ib_client_flags(void);

/* Core-owned invalidation mode: the invalidate entry returns
 * immediately and the MR teardown (dma_unmap -> put_pages -> release)
 * runs later on a core-owned thread — the ordering of the newer
 * rdma-core peer-mem flow.  fake_ib_quiesce() waits for all deferred
 * teardowns (call before freeing the fake_ib_mr). */
void fake_ib_set_async_invalidate(int on);
void fake_ib_quiesce(void);

#ifdef __cplusplus
}
#endif

#endif /* ROCNR_FAKE_IBCORE_H_ */
