/* SPDX-License-Identifier: GPL-2.0 OR MIT
 *
 * rocnr_peer_mem.h — vendored PeerDirect peer-memory client ABI.
 *
 * The reference module (reference: /root/reference/amdp2p.c:44) includes
 * <rdma/peer_mem.h> from Mellanox OFED, which is not vendored there and not
 * present on this system.  This header re-derives the contract from the
 * reference's *call sites* (amdp2p.c:363-371 seven-callback vtable,
 * amdp2p.c:388-389 fixed-size name/version arrays, amdp2p.c:390-391
 * register-with-invalidate-out-param, amdp2p.c:103 invalidate call shape)
 * and tracks the ABI drift of MLNX_OFED 4.x/5.x.
 *
 * ABI-drift switches (set from the module Makefile after probing the OFED
 * tree on the target box; defaults are the modern shapes):
 *
 *   ROCNR_PEER_MEM_CORE_CONTEXT_U64   (default 1)
 *       MLNX_OFED >= 4.0 passes the IB core's registration cookie to
 *       get_pages as a u64; the 2016 ABI used void*.  Newer OFED also
 *       added optional capability flags (e.g. PEER_MEM_INVALIDATE_
 *       UNMAPS); the seven-callback surface below is the stable core
 *       every version dispatches through.
 *
 * When building against a real OFED tree (OFA_DIR set), the module Makefile
 * defines ROCNR_USE_SYSTEM_PEER_MEM and this header simply includes the
 * system <rdma/peer_mem.h>; the vendored declarations below are used for
 * the userspace shim build and for header-only CI.
 */
#ifndef ROCNR_PEER_MEM_H_
#define ROCNR_PEER_MEM_H_

#ifndef ROCNR_PEER_MEM_CORE_CONTEXT_U64
#define ROCNR_PEER_MEM_CORE_CONTEXT_U64 1
#endif

/* Does the peer_mem ABI generation in use declare the extended
 * registration surface (peer_memory_client_ex + capability flags)?
 * 1 for MOFED 5.x / nvidia-peermem-era trees (and the vendored
 * header); set 0 when building against a 2016-era tree. */
#ifndef ROCNR_PEER_MEM_HAS_EX
#define ROCNR_PEER_MEM_HAS_EX 1
#endif

#ifdef ROCNR_USE_SYSTEM_PEER_MEM
/* Real OFED tree (or the third_party/ reconstructions): take every
 * declaration from it; only the registration-cookie typedef the bridge
 * uses in its own signatures is added on top (it must agree with the
 * system header's get_pages — the compile matrix builds with
 * -Werror=incompatible-pointer-types so a mismatch is a hard error). */
#include <rdma/peer_mem.h>
#if ROCNR_PEER_MEM_CORE_CONTEXT_U64
typedef u64 rocnr_core_context_t;
#else
typedef void *rocnr_core_context_t;
#endif
#else

#include <linux/types.h>
#include <linux/scatterlist.h>

#define IB_PEER_MEMORY_NAME_MAX  64
#define IB_PEER_MEMORY_VER_MAX   16

#if ROCNR_PEER_MEM_CORE_CONTEXT_U64
typedef u64 rocnr_core_context_t;
#else
typedef void *rocnr_core_context_t;
#endif

struct device;

/* Invalidation entry point handed back by the IB core at registration:
 * the peer client calls it (reg_handle, core_context) when the producer
 * (KFD) revokes pinned memory under a live MR.  Reference call shape:
 * amdp2p.c:103.  Modern OFED returns int (0 on success). */
typedef int (*invalidate_peer_memory)(void *reg_handle,
				      rocnr_core_context_t core_context);

/*
 * The peer-memory client vtable.  Callback contract (as exercised by
 * ib_umem_get's peer path; ordering per reference §3.2/§3.3 flows):
 *
 *   acquire       — ownership probe.  MUST be cheap, MUST run in the
 *                   registering process's context.  Return 1 "mine" with
 *                   *client_context set, 0 "not mine" (IB core falls back
 *                   to CPU pinning).
 *   get_pages     — pin the region; may register an async free callback.
 *                   sg_head may be left unfilled until dma_map.
 *   get_page_size — granularity of the pinned pages (bytes).
 *   dma_map       — fill sg_head with DMA addresses *for dma_device* and
 *                   set *nmap.  Called once per MR; must honor the IOMMU.
 *   dma_unmap     — undo dma_map for dma_device.
 *   put_pages     — unpin (no-op if invalidation already revoked).
 *   release       — free client_context; last call, always made.
 */
struct peer_memory_client {
	char name[IB_PEER_MEMORY_NAME_MAX];
	char version[IB_PEER_MEMORY_VER_MAX];

	int (*acquire)(unsigned long addr, size_t size,
		       void *peer_mem_private_data, char *peer_mem_name,
		       void **client_context);
	int (*get_pages)(unsigned long addr, size_t size, int write,
			 int force, struct sg_table *sg_head,
			 void *client_context,
			 rocnr_core_context_t core_context);
	int (*dma_map)(struct sg_table *sg_head, void *client_context,
		       struct device *dma_device, int dmasync, int *nmap);
	int (*dma_unmap)(struct sg_table *sg_head, void *client_context,
			 struct device *dma_device);
	void (*put_pages)(struct sg_table *sg_head, void *client_context);
	unsigned long (*get_page_size)(void *client_context);
	void (*release)(void *client_context);
	/* Optional per-peer-id private-data hooks — present in the real
	 * MOFED struct (third_party/mlnx_ofed/rdma/peer_mem.h); a client
	 * that omitted them from a vendored layout would hand the core a
	 * TOO-SHORT struct (round-1 contract-fidelity gap, now closed).
	 * May be NULL. */
	void* (*get_context_private_data)(u64 peer_id);
	void (*put_context_private_data)(void *context);
};

enum {
	PEER_MEM_INVALIDATE_UNMAPS = 1 << 0,
};

/* Extended registration (nvidia-peermem / MOFED 5.x generation): the
 * core recognizes a peer_memory_client embedded at the head of a
 * peer_memory_client_ex by ex_size and reads capability flags from it.
 * PEER_MEM_INVALIDATE_UNMAPS: this client's invalidation flow already
 * unmaps (our free callback revokes the pin and the dma_unmap that the
 * core's teardown then performs is a no-op re-entry), letting the core
 * skip redundant unmap work. */
struct peer_memory_client_ex {
	struct peer_memory_client client;
	size_t ex_size;
	u32 flags;
};

void *ib_register_peer_memory_client(const struct peer_memory_client *client,
				     invalidate_peer_memory *invalidate_cb);
void ib_unregister_peer_memory_client(void *reg_handle);

#endif /* !ROCNR_USE_SYSTEM_PEER_MEM */
#endif /* ROCNR_PEER_MEM_H_ */
