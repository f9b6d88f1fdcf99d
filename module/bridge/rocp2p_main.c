// SPDX-License-Identifier: GPL-2.0 OR MIT
/*
 * rocp2p — MI355X-native PeerDirect peer-memory bridge.
 *
 * Registers an ib_peer_memory_client so InfiniBand HCAs can DMA directly
 * into/out of gfx950 HBM3E: when userspace calls ibv_reg_mr() on a
 * hipMalloc'd pointer, the IB core's peer-memory probe dispatches here and
 * we pin/translate through amdkfd's amd_rdma interface.
 *
 * Same role and callback contract as the reference bridge (reference:
 * /root/reference/amdp2p.c:363-371 vtable, :88-109 invalidation, :374-399
 * init), re-designed for MI355X and modern kernels:
 *
 *  - Per-device DMA mapping.  The reference struct-copied KFD's sg table
 *    and documented "IOMMU must be off" (amdp2p.c:222-240).  We map each
 *    coalesced BAR range through dma_map_resource() for the *requesting*
 *    HCA, so IOMMU-on systems work and two HCAs can map one pin
 *    concurrently (per-registration mapping list).
 *  - sg coalescing sized for 288 GB HBM: KFD pins VRAM in 2 MB pages; a
 *    64 GB pin is ≤32768 segments that we merge into bus-contiguous runs
 *    bounded by dma_get_max_seg_size(), keeping HCA MTT pressure low
 *    (rocp2p_sg.c).
 *  - Invalidation state machine under a mutex: the revoked flag is set
 *    *before* calling the IB core's invalidate (the reference set it
 *    after — amdp2p.c:103-108 — which relies on OFED's invalidate being
 *    synchronous and leaves a window where a concurrent ibv_dereg_mr
 *    double-puts KFD state).  The mutex is dropped around the upcall to
 *    keep the reentrant teardown (invalidate → put_pages/dma_unmap in the
 *    same thread) deadlock-free.
 *  - struct pid discipline: put_pid() on every exit path (the reference
 *    leaked the acquire-time pid reference — amdp2p.c:121,129-133,345-360).
 *  - Liveness observability: active registration / pinned-byte counters
 *    exposed read-only, so "module loaded but never called" (the classic
 *    peer-ABI-drift failure) is diagnosable from userspace.
 */

#include <linux/version.h>
#include <linux/module.h>
#include <linux/kernel.h>
#include <linux/slab.h>
#include <linux/types.h>
#include <linux/compiler.h>
#include <linux/string.h>
#include <linux/errno.h>
#include <linux/list.h>
#include <linux/mutex.h>
#include <linux/atomic.h>
#include <linux/pid.h>
#include <linux/sched.h>
#include <linux/scatterlist.h>
#include <linux/dma-mapping.h>
#include <linux/moduleparam.h>
#include <linux/pci.h>

#include "rocnr_peer_mem.h"
#include "rocnr_amd_rdma.h"
#include "rocp2p_sg.h"

#define ROCP2P_DRIVER_NAME	"rocp2p"
#define ROCP2P_DRIVER_VERSION	"2.0"

MODULE_AUTHOR("ROCnRDMA-AMD project");
MODULE_LICENSE("Dual MIT/GPL");
MODULE_DESCRIPTION("MI355X PeerDirect bridge: IB HCA <-> gfx950 HBM3E");
MODULE_VERSION(ROCP2P_DRIVER_VERSION);

#define rp_dbg(fmt, ...)  pr_debug(ROCP2P_DRIVER_NAME ": " fmt, ##__VA_ARGS__)
#define rp_info(fmt, ...) pr_info(ROCP2P_DRIVER_NAME ": " fmt, ##__VA_ARGS__)
#define rp_err(fmt, ...)  pr_err(ROCP2P_DRIVER_NAME ": " fmt, ##__VA_ARGS__)
#define rp_warn(fmt, ...) pr_warn(ROCP2P_DRIVER_NAME ": " fmt, ##__VA_ARGS__)

static const struct amd_rdma_interface *rdma_interface;
static invalidate_peer_memory ib_invalidate_cb;
static void *ib_reg_handle;

/* Read-only liveness counters, visible at
 * /sys/module/rocp2p/parameters/{active_regs,pinned_bytes,invalidations}
 * (see header comment: diagnosing "loaded but never dispatched"). */
static atomic64_t rocp2p_active_regs = ATOMIC64_INIT(0);
static atomic64_t rocp2p_pinned_bytes = ATOMIC64_INIT(0);
static atomic64_t rocp2p_invalidations = ATOMIC64_INIT(0);

static int rocp2p_param_get_a64(char *buf, const struct kernel_param *kp)
{
	return scnprintf(buf, 24, "%lld\n",
			 (long long)atomic64_read((atomic64_t *)kp->arg));
}

static const struct kernel_param_ops rocp2p_a64_ops = {
	.get = rocp2p_param_get_a64,
};
static atomic64_t rocp2p_bar_bytes = ATOMIC64_INIT(0);
module_param_cb(active_regs, &rocp2p_a64_ops, &rocp2p_active_regs, 0444);
module_param_cb(pinned_bytes, &rocp2p_a64_ops, &rocp2p_pinned_bytes, 0444);
module_param_cb(invalidations, &rocp2p_a64_ops, &rocp2p_invalidations, 0444);
module_param_cb(bar_bytes, &rocp2p_a64_ops, &rocp2p_bar_bytes, 0444);

/* MI355X exposes all 288 GB of HBM3E through BAR0 when resizable/large
 * BAR is enabled; PeerDirect DMA can only target VRAM inside the BAR
 * window, so an undersized aperture silently truncates the peer-
 * reachable region.  Validate at load (north star: "P2P BAR aperture
 * ... chosen for 288 GB HBM per GPU"; RUNBOOK 'Restrictions').  The
 * largest AMD display-class BAR0 is recorded in the bar_bytes module
 * param for the liveness tooling. */
static void rocp2p_check_bar_aperture(void)
{
	struct pci_dev *pdev = NULL;
	u64 best = 0;

	while ((pdev = pci_get_device(PCI_VENDOR_ID_ATI, PCI_ANY_ID,
				      pdev)) != NULL) {
		if ((pdev->class >> 16) == PCI_BASE_CLASS_DISPLAY) {
			u64 len = pci_resource_len(pdev, 0);

			if (len > best)
				best = len;
		}
	}
	atomic64_add((long long)best, &rocp2p_bar_bytes);
	if (!best)
		rp_warn("no AMD display-class PCI device visible\n");
	else if (best < (16ULL << 30))
		rp_warn("largest GPU BAR0 is %llu MiB — full-VRAM (large) BAR appears DISABLED; peer DMA window is truncated (see docs/RUNBOOK.md restrictions)\n",
			best >> 20);
	else
		rp_info("GPU BAR0 aperture %llu GiB (full-VRAM BAR ok)\n",
			best >> 30);
}

enum rocp2p_state {
	ROCP2P_ACQUIRED = 0,	/* context exists, nothing pinned */
	ROCP2P_PINNING,		/* KFD get_pages in flight (lock dropped) */
	ROCP2P_PINNED,		/* KFD holds a pin for us */
	ROCP2P_REVOKED,		/* KFD invalidated; pin is gone */
};

/* Bring-up fallback for amd_rdma ABI drift: the modern ROCK
 * get_pages takes the DMA target device and may refuse dma_dev=NULL (it
 * dma-maps against that device internally).  When enabled (default) and
 * KFD rejects the NULL-device pin, the pin is DEFERRED to the first
 * dma_map call, where the requesting HCA's device is known; that pin's
 * sg table is then used as-is (KFD already device-mapped it) and
 * dma_map_resource is skipped — the reference-style single-device path
 * (amdp2p.c:222-240).  Additional HCAs mapping the same MR reuse that
 * table with a loud warning (correct only when the IOMMU does not
 * isolate the two HCAs differently).  See docs/LIMITATIONS.md. */
#if ROCNR_AMD_RDMA_HAS_DMA_DEV
static bool null_dev_fallback = true;
module_param(null_dev_fallback, bool, 0444);
#endif

/* Register with the modern extended surface (peer_memory_client_ex +
 * PEER_MEM_INVALIDATE_UNMAPS) by default; peer_ex=0 registers the
 * plain 2016-style client for cores that predate the extension.  The
 * flag is truthful: our free callback revokes the KFD pin before the
 * invalidate upcall, so the core's post-invalidate dma_unmap is
 * redundant (per-HCA IOVA mappings are reclaimed in release()).
 * Non-static so the userspace shim suite can drive both generations. */
bool rocp2p_peer_ex = true;
module_param_named(peer_ex, rocp2p_peer_ex, bool, 0444);

/* One dma_map() result for one device. */
struct rocp2p_dmamap {
	struct list_head node;
	struct device *dev;
	struct sg_table sgt;	/* device-mapped, coalesced */
	bool mapped;		/* dma_map_resource() performed */
};

/* One peer-memory registration (one MR). */
struct rocp2p_reg {
	u64 va;
	u64 size;
	struct pid *pid;

	struct mutex lock;	/* state, p2p, dmamaps */
	enum rocp2p_state state;
	struct amd_p2p_info *p2p;
	unsigned long page_size;	/* cached from KFD */
	bool defer_pin;		/* KFD refused dma_dev=NULL: pin at dma_map */
	struct device *pin_dev;	/* device the deferred pin was made for */
	struct list_head dmamaps;

	rocnr_core_context_t core_context;
};

/* ------------------------------------------------------------------ */
/* KFD invalidation: GPU freed a buffer under a live MR (reference flow
 * /root/reference/amdp2p.c:88-109; ordering re-designed, see top).     */
static void rocp2p_kfd_free_cb(void *client_priv)
{
	struct rocp2p_reg *reg = client_priv;
	rocnr_core_context_t core_context;
	u64 va, size;

	if (!reg) {
		rp_warn("free callback with NULL context\n");
		return;
	}

	mutex_lock(&reg->lock);
	if (reg->state == ROCP2P_PINNING) {
		/* Revoke raced the pin install (possibly synchronously from
		 * inside KFD's get_pages).  Mark revoked and do NOT upcall:
		 * no MR exists yet — the in-flight get_pages/dma_map sees
		 * REVOKED, discards the (KFD-reclaimed) pin without touching
		 * it, and fails the registration; the IB core unwinds via
		 * release(). */
		reg->state = ROCP2P_REVOKED;
		mutex_unlock(&reg->lock);
		atomic64_inc(&rocp2p_invalidations);
		rp_dbg("invalidate during pin install\n");
		return;
	}
	if (reg->state != ROCP2P_PINNED) {
		/* Duplicate or late revoke: nothing to tear down. */
		mutex_unlock(&reg->lock);
		return;
	}
	/* Mark revoked BEFORE the upcall: any concurrent or reentrant
	 * put_pages must not touch KFD — KFD reclaims the pin when this
	 * callback returns. */
	reg->state = ROCP2P_REVOKED;
	reg->p2p = NULL;
	core_context = reg->core_context;
	va = reg->va;
	size = reg->size;
	atomic64_sub(reg->size, &rocp2p_pinned_bytes);
	mutex_unlock(&reg->lock);

	/* reg must not be touched past this point: once REVOKED is
	 * visible, a concurrent dereg can complete release() and free it
	 * (values for logging were latched under the lock). */
	atomic64_inc(&rocp2p_invalidations);
	rp_dbg("invalidate va 0x%llx size 0x%llx\n", va, size);

	/* Ask the IB core to tear the MR down now.  May reenter
	 * dma_unmap/put_pages/release on this thread — lock is dropped. */
	(*ib_invalidate_cb)(ib_reg_handle, core_context);
}

/* ------------------------------------------------------------------ */
static int rocp2p_acquire(unsigned long addr, size_t size,
			  void *peer_mem_private_data, char *peer_mem_name,
			  void **client_context)
{
	struct rocp2p_reg *reg;
	struct pid *pid;

	/* Only acquire is guaranteed to run in the registering process's
	 * context (reference: amdp2p.c:150-152); capture the pid here. */
	pid = get_task_pid(current, PIDTYPE_PID);

	if (!rdma_interface->is_gpu_address(addr, pid)) {
		put_pid(pid);
		return 0;	/* not ours: IB core falls back to CPU path */
	}

	reg = kzalloc(sizeof(*reg), GFP_KERNEL);
	if (!reg) {
		/* Deliberate policy (matches reference amdp2p.c:140-144):
		 * report "not mine" so the MR can still be built via the
		 * CPU path instead of failing registration outright. */
		put_pid(pid);
		rp_err("acquire: context allocation failed\n");
		return 0;
	}

	reg->va = addr;
	reg->size = size;
	reg->pid = pid;
	reg->state = ROCP2P_ACQUIRED;
	mutex_init(&reg->lock);
	INIT_LIST_HEAD(&reg->dmamaps);

	__module_get(THIS_MODULE);
	atomic64_inc(&rocp2p_active_regs);
	*client_context = reg;
	rp_dbg("acquire: va 0x%lx size 0x%zx\n", addr, size);
	return 1;
}

/* Pin reg's range through KFD.  Lock-free install (probe module's
 * PIN_INIT pattern): the KFD call is made with reg->lock
 * DROPPED and state=PINNING, so a revoke firing during the pin — even
 * synchronously from inside KFD's get_pages — takes the lock, finds
 * PINNING, marks REVOKED and returns; we then discard the KFD-reclaimed
 * pin without dereferencing it.  dma_dev is NULL for the pin-at-
 * registration path and the HCA's device for the deferred drift
 * fallback (see null_dev_fallback). */
static int rocp2p_pin(struct rocp2p_reg *reg, struct device *dma_dev)
{
	struct amd_p2p_info *p2p = NULL;
	unsigned long page_size = 0;
	int ret;

	mutex_lock(&reg->lock);
	if (reg->state != ROCP2P_ACQUIRED) {
		mutex_unlock(&reg->lock);
		return -EINVAL;
	}
	reg->state = ROCP2P_PINNING;
	mutex_unlock(&reg->lock);

	ret = rdma_interface->get_pages(reg->va, reg->size, reg->pid,
#if ROCNR_AMD_RDMA_HAS_DMA_DEV
					dma_dev,
#endif
					&p2p, rocp2p_kfd_free_cb, reg);
#if !ROCNR_AMD_RDMA_HAS_DMA_DEV
	(void)dma_dev;
#endif

	if (!ret && p2p &&
	    (rdma_interface->get_page_size(reg->va, reg->size, reg->pid,
					   &page_size) || !page_size)) {
		/* MI355X VRAM granule; only used for reporting. */
		page_size = 2UL << 20;
		rp_warn("get_pages: page-size query failed, assuming 2 MiB\n");
	}

	mutex_lock(&reg->lock);
	if (reg->state == ROCP2P_REVOKED) {
		/* Revoke won the race: KFD reclaimed the pin when the free
		 * callback returned — p2p (if any) is dead memory and must
		 * not be touched or put. */
		mutex_unlock(&reg->lock);
		rp_warn("pin: revoked during install\n");
		return -ENODEV;
	}
	if (ret || !p2p) {
		reg->state = ROCP2P_ACQUIRED;
		mutex_unlock(&reg->lock);
		return ret ? ret : -ENOMEM;
	}
	reg->p2p = p2p;
	reg->page_size = page_size;
	reg->pin_dev = dma_dev;
	reg->state = ROCP2P_PINNED;
	mutex_unlock(&reg->lock);
	atomic64_add(reg->size, &rocp2p_pinned_bytes);
	return 0;
}

static int rocp2p_get_pages(unsigned long addr, size_t size, int write,
			    int force, struct sg_table *sg_head,
			    void *client_context,
			    rocnr_core_context_t core_context)
{
	struct rocp2p_reg *reg = client_context;
	int ret;

	if (!reg)
		return -EINVAL;
	if (addr != reg->va || size != reg->size) {
		rp_warn("get_pages: range mismatch (acquired 0x%llx+0x%llx, asked 0x%lx+0x%zx)\n",
			reg->va, reg->size, addr, size);
		return -EINVAL;
	}

	mutex_lock(&reg->lock);
	reg->core_context = core_context;
	mutex_unlock(&reg->lock);

	ret = rocp2p_pin(reg, NULL);
#if ROCNR_AMD_RDMA_HAS_DMA_DEV
	if (ret && ret != -ENODEV && null_dev_fallback) {
		/* Drift fallback: modern ROCK KFD may require the DMA
		 * target device (it maps against it inside get_pages).
		 * Defer the pin to dma_map, where the HCA is known. */
		rp_warn("get_pages: KFD pin with dma_dev=NULL failed (%d); deferring pin to dma_map (amd_rdma ABI drift fallback — see docs/LIMITATIONS.md)\n",
			ret);
		mutex_lock(&reg->lock);
		reg->defer_pin = true;
		mutex_unlock(&reg->lock);
		return 0;
	}
#endif
	if (ret)
		rp_err("get_pages: KFD pin failed: %d\n", ret);
	/* sg_head intentionally not filled here: translation happens in
	 * dma_map where the target device is known (reference kept the
	 * same deferral — amdp2p.c:214). */
	return ret;
}

/* Iterate KFD's pinned sg table as bus-address segments. */
struct kfd_sg_iter {
	struct rocnr_seg_iter it;
	struct scatterlist *sg;
};

static int kfd_sg_next(struct rocnr_seg_iter *it, struct rocnr_seg *seg)
{
	struct kfd_sg_iter *k = (struct kfd_sg_iter *)it;

	if (!k->sg)
		return 0;
	seg->addr = sg_dma_address(k->sg);
	seg->len = sg_dma_len(k->sg);
	k->sg = sg_next(k->sg);
	return 1;
}

struct emit_map_ctx {
	struct device *dev;
	struct scatterlist *sg;	/* cursor in the output table */
	size_t mapped;		/* entries dma_map_resource'd so far */
	int err;
};

static int emit_map_one(void *ctx, const struct rocnr_seg *seg)
{
	struct emit_map_ctx *e = ctx;
	dma_addr_t daddr;

	daddr = dma_map_resource(e->dev, (phys_addr_t)seg->addr, seg->len,
				 DMA_BIDIRECTIONAL, 0);
	if (dma_mapping_error(e->dev, daddr)) {
		e->err = -EIO;
		return 1;
	}
	sg_dma_address(e->sg) = daddr;
	sg_dma_len(e->sg) = seg->len;
	e->sg = sg_next(e->sg);
	e->mapped++;
	return 0;
}

/* Deferred-pin path: KFD already device-mapped the sg table inside
 * get_pages(dma_dev); copy the coalesced runs verbatim. */
static int emit_copy_one(void *ctx, const struct rocnr_seg *seg)
{
	struct emit_map_ctx *e = ctx;

	sg_dma_address(e->sg) = (dma_addr_t)seg->addr;
	sg_dma_len(e->sg) = seg->len;
	e->sg = sg_next(e->sg);
	return 0;
}

static void rocp2p_unmap_one(struct rocp2p_dmamap *map, size_t upto)
{
	struct scatterlist *sg;
	size_t i;

	if (map->mapped) {
		for_each_sg(map->sgt.sgl, sg, map->sgt.nents, i) {
			if (i >= upto)
				break;
			dma_unmap_resource(map->dev, sg_dma_address(sg),
					   sg_dma_len(sg), DMA_BIDIRECTIONAL,
					   0);
		}
	}
	sg_free_table(&map->sgt);
	list_del(&map->node);
	kfree(map);
}

static int rocp2p_dma_map(struct sg_table *sg_head, void *client_context,
			  struct device *dma_device, int dmasync, int *nmap)
{
	struct rocp2p_reg *reg = client_context;
	struct rocp2p_dmamap *map;
	struct kfd_sg_iter kit;
	struct emit_map_ctx emit;
	u64 max_seg;
	size_t nsegs, nout;
	bool deferred;
	int ret;

	if (!reg || !dma_device || !nmap)
		return -EINVAL;

	mutex_lock(&reg->lock);
	if (reg->defer_pin && reg->state == ROCP2P_ACQUIRED) {
		/* Drift fallback: execute the deferred pin now, against the
		 * requesting HCA's device (see null_dev_fallback). */
		mutex_unlock(&reg->lock);
		ret = rocp2p_pin(reg, dma_device);
		if (ret) {
			rp_err("dma_map: deferred KFD pin failed: %d\n", ret);
			return ret;
		}
		mutex_lock(&reg->lock);
	}
	deferred = reg->defer_pin;
	if (deferred && reg->state == ROCP2P_PINNED &&
	    reg->pin_dev != dma_device)
		rp_warn("dma_map: deferred pin was device-mapped for another HCA; sharing its table (reference-style path — IOMMU isolation between HCAs is NOT honored)\n");
	if (reg->state != ROCP2P_PINNED || !reg->p2p || !reg->p2p->pages) {
		mutex_unlock(&reg->lock);
		rp_err("dma_map: no pinned pages (state %d)\n", reg->state);
		return -EINVAL;
	}

	max_seg = dma_get_max_seg_size(dma_device);

	kit.it.next = kfd_sg_next;
	kit.sg = reg->p2p->pages->sgl;
	nsegs = rocnr_coalesce_count(&kit.it, max_seg);
	if (!nsegs) {
		mutex_unlock(&reg->lock);
		return -EINVAL;
	}

	map = kzalloc(sizeof(*map), GFP_KERNEL);
	if (!map) {
		mutex_unlock(&reg->lock);
		return -ENOMEM;
	}
	ret = sg_alloc_table(&map->sgt, nsegs, GFP_KERNEL);
	if (ret) {
		kfree(map);
		mutex_unlock(&reg->lock);
		return ret;
	}
	map->dev = dma_device;
	map->mapped = !deferred;
	list_add(&map->node, &reg->dmamaps);

	emit.dev = dma_device;
	emit.sg = map->sgt.sgl;
	emit.mapped = 0;
	emit.err = 0;
	kit.sg = reg->p2p->pages->sgl;
	nout = rocnr_coalesce(&kit.it, max_seg,
			      deferred ? emit_copy_one : emit_map_one, &emit);
	if (nout == (size_t)-1 || emit.err) {
		size_t done = emit.mapped;

		rp_err("dma_map: dma_map_resource failed after %zu segs\n",
		       done);
		rocp2p_unmap_one(map, done);
		mutex_unlock(&reg->lock);
		return emit.err ? emit.err : -EIO;
	}

	/* Hand the IB core our table head; sgl storage stays owned by the
	 * mapping and is released in dma_unmap. */
	*sg_head = map->sgt;
	*nmap = (int)nsegs;
	/* log before unlock: reg must not be touched once the lock drops
	 * (a KFD revoke during registration can reach release()) */
	rp_dbg("dma_map: va 0x%llx size 0x%llx -> %zu segs (kfd page %lu, max_seg 0x%llx)\n",
	       reg->va, reg->size, nsegs, reg->page_size, max_seg);
	mutex_unlock(&reg->lock);
	return 0;
}

static int rocp2p_dma_unmap(struct sg_table *sg_head, void *client_context,
			    struct device *dma_device)
{
	struct rocp2p_reg *reg = client_context;
	struct rocp2p_dmamap *map, *tmp;

	if (!reg)
		return -EINVAL;

	mutex_lock(&reg->lock);
	list_for_each_entry_safe(map, tmp, &reg->dmamaps, node) {
		if (map->dev == dma_device &&
		    (!sg_head || map->sgt.sgl == sg_head->sgl)) {
			rocp2p_unmap_one(map, map->sgt.nents);
			mutex_unlock(&reg->lock);
			return 0;
		}
	}
	mutex_unlock(&reg->lock);
	rp_warn("dma_unmap: no mapping for device %p\n", dma_device);
	return 0;
}

static void rocp2p_put_pages(struct sg_table *sg_head, void *client_context)
{
	struct rocp2p_reg *reg = client_context;
	struct amd_p2p_info *p2p;
	int ret;

	if (!reg)
		return;

	mutex_lock(&reg->lock);
	if (reg->state != ROCP2P_PINNED) {
		/* REVOKED: KFD already reclaimed the pin in the free
		 * callback (reference flag check: amdp2p.c:299-302). */
		mutex_unlock(&reg->lock);
		return;
	}
	p2p = reg->p2p;
	reg->p2p = NULL;
	reg->state = ROCP2P_ACQUIRED;
	atomic64_sub(reg->size, &rocp2p_pinned_bytes);
	mutex_unlock(&reg->lock);

	/* The KFD call is made OUTSIDE reg->lock: KFD's put_pages may
	 * block on its own revoke machinery whose free callback takes
	 * reg->lock — holding it here would be an ABBA deadlock (caught
	 * by the shim race tests).  Safety without the lock:
	 *  (a) the IB core serializes invalidation teardown against
	 *      dereg teardown, so only one path reaches this call;
	 *  (b) the free callback marks REVOKED before its upcall, so a
	 *      revoke that wins the state race makes us skip KFD
	 *      entirely;
	 *  (c) if KFD starts revoking between our unlock and this call,
	 *      its callback finds state != PINNED and does nothing, and
	 *      KFD must treat a put of a pin it is concurrently revoking
	 *      as a benign no-op (it serializes internally).
	 * The reference had the same window with no analysis and the
	 * flag ordered unsafely (amdp2p.c:299-308). */
	ret = rdma_interface->put_pages(&p2p);
	if (ret)
		rp_err("put_pages: KFD unpin failed: %d\n", ret);
}

static unsigned long rocp2p_get_page_size(void *client_context)
{
	struct rocp2p_reg *reg = client_context;
	unsigned long page_size = 0;

	if (!reg)
		return 0;

	mutex_lock(&reg->lock);
	page_size = reg->page_size;
	mutex_unlock(&reg->lock);
	if (page_size)
		return page_size;

	if (rdma_interface->get_page_size(reg->va, reg->size, reg->pid,
					  &page_size) || !page_size) {
		rp_warn("get_page_size failed; reporting 2 MiB VRAM granule\n");
		return 2UL << 20;
	}
	return page_size;
}

static void rocp2p_release(void *client_context)
{
	struct rocp2p_reg *reg = client_context;
	struct rocp2p_dmamap *map, *tmp;
	struct amd_p2p_info *p2p = NULL;

	if (!reg)
		return;

	/* Defensive: the IB core should have dma_unmap'd and put_pages'd
	 * already; clean up anything left so nothing leaks. */
	mutex_lock(&reg->lock);
	list_for_each_entry_safe(map, tmp, &reg->dmamaps, node)
		rocp2p_unmap_one(map, map->sgt.nents);
	if (reg->state == ROCP2P_PINNED && reg->p2p) {
		p2p = reg->p2p;
		reg->p2p = NULL;
		reg->state = ROCP2P_ACQUIRED;
		atomic64_sub(reg->size, &rocp2p_pinned_bytes);
		rp_warn("release: registration still pinned; unpinning\n");
	}
	mutex_unlock(&reg->lock);
	if (p2p)
		rdma_interface->put_pages(&p2p); /* outside lock: see put_pages */

	put_pid(reg->pid);
	kfree(reg);
	atomic64_dec(&rocp2p_active_regs);
	module_put(THIS_MODULE);
}

/* ------------------------------------------------------------------ */
/* Registered as the head of a peer_memory_client_ex so an ex-aware IB
 * core (MOFED 5.x / nvidia-peermem generation) can read capability
 * flags; the marker is the public convention of that generation (last
 * byte of version[] set to 1 — see module/include/rocnr_peer_mem.h).
 * Older cores read only the embedded plain client. */
#if ROCNR_PEER_MEM_HAS_EX
static struct peer_memory_client_ex rocp2p_client_ex = {
	.client = {
		.acquire = rocp2p_acquire,
		.get_pages = rocp2p_get_pages,
		.dma_map = rocp2p_dma_map,
		.dma_unmap = rocp2p_dma_unmap,
		.put_pages = rocp2p_put_pages,
		.get_page_size = rocp2p_get_page_size,
		.release = rocp2p_release,
	},
};
#define rocp2p_client rocp2p_client_ex.client
#else	/* 2016-era tree: plain client only */
static struct peer_memory_client rocp2p_client = {
	.acquire = rocp2p_acquire,
	.get_pages = rocp2p_get_pages,
	.dma_map = rocp2p_dma_map,
	.dma_unmap = rocp2p_dma_unmap,
	.put_pages = rocp2p_put_pages,
	.get_page_size = rocp2p_get_page_size,
	.release = rocp2p_release,
};
#endif

static int __init rocp2p_init(void)
{
	int ret;

	ret = amdkfd_query_rdma_interface(&rdma_interface);
	if (ret < 0 || !rdma_interface) {
		rp_err("amdkfd RDMA interface unavailable (%d) — is the ROCK amdgpu/KFD driver loaded?\n",
		       ret);
		return ret < 0 ? ret : -ENODEV;
	}
	if (!rdma_interface->get_pages || !rdma_interface->put_pages ||
	    !rdma_interface->is_gpu_address ||
	    !rdma_interface->get_page_size) {
		rp_err("amdkfd RDMA vtable incomplete — ABI drift?\n");
		return -ENOSYS;
	}

	strscpy(rocp2p_client.name, ROCP2P_DRIVER_NAME,
		sizeof(rocp2p_client.name));
	strscpy(rocp2p_client.version, ROCP2P_DRIVER_VERSION,
		sizeof(rocp2p_client.version));
#if ROCNR_PEER_MEM_HAS_EX
	if (rocp2p_peer_ex) {
		/* ex marker convention: last version byte = 1 */
		rocp2p_client.version[IB_PEER_MEMORY_VER_MAX - 1] = 1;
		rocp2p_client_ex.ex_size = sizeof(rocp2p_client_ex);
		rocp2p_client_ex.flags = PEER_MEM_INVALIDATE_UNMAPS;
	} else {
		rocp2p_client.version[IB_PEER_MEMORY_VER_MAX - 1] = 0;
		rocp2p_client_ex.ex_size = 0;
		rocp2p_client_ex.flags = 0;
	}
#else
	if (rocp2p_peer_ex)
		rp_info("peer_ex requested but this peer_mem ABI generation has no extended surface; registering plain client\n");
#endif

	rocp2p_check_bar_aperture();

	ib_reg_handle = ib_register_peer_memory_client(&rocp2p_client,
						       &ib_invalidate_cb);
	if (!ib_reg_handle || !ib_invalidate_cb) {
		if (ib_reg_handle) {
			/* Registered but no invalidate entry point: MUST
			 * unregister before failing the load, or the IB
			 * core keeps vtable pointers into an unloaded
			 * module (use-after-free on its next peer probe). */
			ib_unregister_peer_memory_client(ib_reg_handle);
			ib_reg_handle = NULL;
		}
		rp_err("peer-memory registration failed — OFED peer_mem ABI drift?\n");
		return -EINVAL;
	}

	rp_info("loaded: PeerDirect client '%s' v%s (kfd vtable %p)\n",
		rocp2p_client.name, rocp2p_client.version, rdma_interface);
	return 0;
}

static void __exit rocp2p_exit(void)
{
	ib_unregister_peer_memory_client(ib_reg_handle);
	rp_info("unloaded (%lld regs leaked)\n",
		(long long)atomic64_read(&rocp2p_active_regs));
}

module_init(rocp2p_init);
module_exit(rocp2p_exit);
