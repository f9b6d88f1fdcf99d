// SPDX-License-Identifier: GPL-2.0 OR MIT
/*
 * rocp2p_probe — raw amd_rdma interface probe char device.
 *
 * Layer-isolation test surface: exercises the KFD pin/translate path
 * one layer BELOW the PeerDirect bridge, so bridge bugs and KFD bugs
 * can be told apart on real hardware (same role as the reference's
 * amdp2ptest — reference: /root/reference/tests/amdp2ptest.c — with its
 * defects fixed):
 *
 *  - pin-node leak on get_pages failure (amdp2ptest.c:243-246): fixed
 *    by a single error path;
 *  - free_callback unlinking a node while list walkers iterate
 *    (TOCTOU, amdp2ptest.c:77-89): fixed with a dying flag — the
 *    callback marks + unlinks under the lock, and put/release skip
 *    dying nodes (KFD owns their reclamation);
 *  - mmap mapping only the FIRST sg entry, passing the full vma size
 *    to every remap call, and testing node-inside-vma instead of
 *    vma-inside-node (amdp2ptest.c:361-391): rewritten to walk every
 *    entry with per-entry lengths and a correct containment check;
 *  - ioctl ABI direction bits / typo'd type (amdp2ptest.h:62-72):
 *    fixed in module/include/rocp2p_probe_abi.h;
 *  - adds GET_INFO so userspace can check sg-table shape (coalescing,
 *    2 MiB VRAM granularity) without needing mmap.
 *
 * Duplicate registrations of one range stay supported: PUT_PAGES
 * releases every matching (va, size) pin and reports the count
 * (documented behavior of the reference, amdp2ptest.c:296-299).
 */

#include <linux/version.h>
#include <linux/module.h>
#include <linux/kernel.h>
#include <linux/slab.h>
#include <linux/types.h>
#include <linux/compiler.h>
#include <linux/string.h>
#include <linux/errno.h>
#include <linux/uaccess.h>
#include <linux/fs.h>
#include <linux/miscdevice.h>
#include <linux/list.h>
#include <linux/mutex.h>
#include <linux/pid.h>
#include <linux/sched.h>
#include <linux/scatterlist.h>
#include <linux/mm.h>
#include <linux/io.h>

#include "rocnr_amd_rdma.h"
#include "rocp2p_probe_abi.h"

/* vm_flags_set() appeared in 6.3; earlier kernels mutate vm_flags
 * directly (the shim provides its own definition). */
#if !defined(__ROCNR_SHIM__) && LINUX_VERSION_CODE < KERNEL_VERSION(6, 3, 0)
static inline void vm_flags_set(struct vm_area_struct *vma,
				vm_flags_t flags)
{
	vma->vm_flags |= flags;
}
#endif

MODULE_AUTHOR("ROCnRDMA-AMD project");
MODULE_LICENSE("Dual MIT/GPL");
MODULE_DESCRIPTION("raw amd_rdma probe device for MI355X GPU-direct RDMA");
MODULE_VERSION("2.0");

#define pp_info(fmt, ...) \
	pr_info(ROCP2P_PROBE_DEVICE_NAME ": " fmt, ##__VA_ARGS__)
#define pp_err(fmt, ...) \
	pr_err(ROCP2P_PROBE_DEVICE_NAME ": " fmt, ##__VA_ARGS__)
#define pp_dbg(fmt, ...) pr_debug(fmt, ##__VA_ARGS__)

static const struct amd_rdma_interface *rdma_interface;

/* Device-node mode.  Default 0600 (root only): the probe lets its user
 * pin unbounded GPU memory and mmap BAR pages — a diagnostic surface,
 * not a service.  Deployments that want group access relax it
 * explicitly (insmod rocp2p_probe.ko devmode=0660 + a udev group rule);
 * see docs/RUNBOOK.md.  (was 0666.) */
static ushort devmode = 0600;
module_param(devmode, ushort, 0444);

struct probe_ctx {
	struct list_head pins;
	struct mutex lock;
	struct pid *pid;
};

enum probe_pin_state {
	PIN_INIT = 0,	/* pinned, not yet linked (get_pages in progress) */
	PIN_LIVE,	/* linked in ctx->pins */
	PIN_PUT,	/* unlinked by put/release; that path frees the node */
	PIN_REVOKED,	/* revoked by the KFD callback */
};

struct probe_pin {
	struct list_head node;
	struct probe_ctx *ctx;
	struct amd_p2p_info *info;
	u64 va;
	u64 size;
	enum probe_pin_state state;
};

/* KFD revoked a pinned allocation (GPU buffer freed while pinned).
 * Unlink under the lock; the pin's KFD resources die when this
 * callback returns, so nothing else may touch info afterwards. */
static void probe_free_callback(void *client_priv)
{
	struct probe_pin *pin = client_priv;
	struct probe_ctx *ctx;
	int mine = 0;

	if (!pin)
		return;
	ctx = pin->ctx;
	mutex_lock(&ctx->lock);
	switch (pin->state) {
	case PIN_LIVE:
		/* we win the race: unlink and own the node */
		pin->state = PIN_REVOKED;
		list_del(&pin->node);
		mine = 1;
		break;
	case PIN_INIT:
		/* revoke before the registering ioctl linked the node:
		 * flag it; the ioctl path frees it */
		pin->state = PIN_REVOKED;
		break;
	case PIN_PUT:
		/* a concurrent put/release detached it and will free the
		 * node after its KFD put_pages returns (which KFD
		 * serializes against this callback) — touch nothing */
		break;
	case PIN_REVOKED:
		break;
	}
	mutex_unlock(&ctx->lock);
	if (mine) {
		pp_dbg("revoked pin va 0x%llx size 0x%llx\n", pin->va,
		       pin->size);
		kfree(pin);
	}
}

static int probe_open(struct inode *inode, struct file *filp)
{
	struct probe_ctx *ctx;

	ctx = kzalloc(sizeof(*ctx), GFP_KERNEL);
	if (!ctx)
		return -ENOMEM;
	INIT_LIST_HEAD(&ctx->pins);
	mutex_init(&ctx->lock);
	ctx->pid = get_task_pid(current, PIDTYPE_PID);
	filp->private_data = ctx;
	return 0;
}

/* Detach one pin under the lock; the caller calls KFD put_pages
 * OUTSIDE the lock (same deadlock discipline as the bridge) and then
 * frees the node — safe because KFD's put_pages returns only after any
 * in-flight revoke callback completed, and that callback sees PIN_PUT
 * and leaves the node alone. */
static void probe_detach_locked(struct probe_pin *pin)
{
	list_del(&pin->node);
	pin->state = PIN_PUT;
}

static int probe_release(struct inode *inode, struct file *filp)
{
	struct probe_ctx *ctx = filp->private_data;
	struct amd_p2p_info *info;
	int ret;

	/* leak-proof close: unpin everything still registered */
	for (;;) {
		struct probe_pin *pin;

		mutex_lock(&ctx->lock);
		if (list_empty(&ctx->pins)) {
			mutex_unlock(&ctx->lock);
			break;
		}
		pin = list_first_entry(&ctx->pins, struct probe_pin, node);
		probe_detach_locked(pin);
		info = pin->info;
		mutex_unlock(&ctx->lock);
		ret = rdma_interface->put_pages(&info);
		if (ret)
			pp_err("release: put_pages failed: %d\n", ret);
		kfree(pin);
	}
	put_pid(ctx->pid);
	kfree(ctx);
	return 0;
}

/* ---- ioctl handlers ---- */

static long ioctl_is_gpu_address(struct probe_ctx *ctx, unsigned long arg)
{
	struct rocp2p_probe_is_gpu p;

	if (copy_from_user(&p, (void __user *)arg, sizeof(p)))
		return -EFAULT;
	p.is_gpu = rdma_interface->is_gpu_address(p.addr, ctx->pid) ? 1 : 0;
	if (copy_to_user((void __user *)arg, &p, sizeof(p)))
		return -EFAULT;
	return 0;
}

static long ioctl_get_page_size(struct probe_ctx *ctx, unsigned long arg)
{
	struct rocp2p_probe_page_size p;
	unsigned long page_size = 0;
	int ret;

	if (copy_from_user(&p, (void __user *)arg, sizeof(p)))
		return -EFAULT;
	ret = rdma_interface->get_page_size(p.addr, p.length, ctx->pid,
					    &page_size);
	if (ret)
		return ret;
	p.page_size = page_size;
	if (copy_to_user((void __user *)arg, &p, sizeof(p)))
		return -EFAULT;
	return 0;
}

static long ioctl_get_pages(struct probe_ctx *ctx, unsigned long arg)
{
	struct rocp2p_probe_pin p;
	struct probe_pin *pin;
	int ret;

	if (copy_from_user(&p, (void __user *)arg, sizeof(p)))
		return -EFAULT;
	if (!p.length)
		return -EINVAL;

	pin = kzalloc(sizeof(*pin), GFP_KERNEL);
	if (!pin)
		return -ENOMEM;
	pin->ctx = ctx;
	pin->va = p.addr;
	pin->size = p.length;

	ret = rdma_interface->get_pages(p.addr, p.length, ctx->pid,
#if ROCNR_AMD_RDMA_HAS_DMA_DEV
					NULL,
#endif
					&pin->info, probe_free_callback, pin);
	if (ret || !pin->info) {
		kfree(pin);	/* reference leaked here: amdp2ptest.c:243 */
		return ret ? ret : -ENOMEM;
	}

	mutex_lock(&ctx->lock);
	if (pin->state == PIN_REVOKED) {
		/* revoked before we could link it: KFD already reclaimed
		 * the pin; nothing registered */
		mutex_unlock(&ctx->lock);
		kfree(pin);
		return -ENOENT;
	}
	pin->state = PIN_LIVE;
	list_add_tail(&pin->node, &ctx->pins);
	mutex_unlock(&ctx->lock);
	return 0;
}

static long ioctl_put_pages(struct probe_ctx *ctx, unsigned long arg)
{
	struct rocp2p_probe_unpin p;
	struct probe_pin *pin, *tmp;
	u64 released = 0;
	int ret;

	if (copy_from_user(&p, (void __user *)arg, sizeof(p)))
		return -EFAULT;

	/* release EVERY matching registration (duplicate pins of one
	 * range are a supported test case) */
	for (;;) {
		struct amd_p2p_info *info = NULL;
		struct probe_pin *match = NULL;

		mutex_lock(&ctx->lock);
		list_for_each_entry_safe(pin, tmp, &ctx->pins, node) {
			if (pin->va == p.addr && pin->size == p.length) {
				probe_detach_locked(pin);
				match = pin;
				info = pin->info;
				break;
			}
		}
		mutex_unlock(&ctx->lock);
		if (!match)
			break;
		ret = rdma_interface->put_pages(&info);
		if (ret)
			pp_err("put_pages failed: %d\n", ret);
		else
			released++;
		kfree(match);
	}

	p.released = released;
	if (copy_to_user((void __user *)arg, &p, sizeof(p)))
		return -EFAULT;
	return released ? 0 : -ENOENT;
}

static long ioctl_get_info(struct probe_ctx *ctx, unsigned long arg)
{
	struct rocp2p_probe_info p;
	struct probe_pin *pin;
	struct scatterlist *sg;
	int found = 0, i;

	if (copy_from_user(&p, (void __user *)arg, sizeof(p)))
		return -EFAULT;

	mutex_lock(&ctx->lock);
	list_for_each_entry(pin, &ctx->pins, node) {
		if (pin->va == p.addr && pin->size == p.length &&
		    pin->info && pin->info->pages) {
			p.nents = pin->info->pages->nents;
			p.total_bytes = 0;
			p.max_seg_bytes = 0;
			p.first_dma_addr =
				sg_dma_address(pin->info->pages->sgl);
			for_each_sg(pin->info->pages->sgl, sg,
				    pin->info->pages->nents, i) {
				p.total_bytes += sg_dma_len(sg);
				if (sg_dma_len(sg) > p.max_seg_bytes)
					p.max_seg_bytes = sg_dma_len(sg);
			}
			found = 1;
			break;
		}
	}
	mutex_unlock(&ctx->lock);

	if (!found)
		return -ENOENT;
	if (copy_to_user((void __user *)arg, &p, sizeof(p)))
		return -EFAULT;
	return 0;
}

static long probe_unlocked_ioctl(struct file *filp, unsigned int cmd,
				 unsigned long arg)
{
	struct probe_ctx *ctx = filp->private_data;

	switch (cmd) {
	case ROCP2P_PROBE_IS_GPU_ADDRESS:
		return ioctl_is_gpu_address(ctx, arg);
	case ROCP2P_PROBE_GET_PAGE_SIZE:
		return ioctl_get_page_size(ctx, arg);
	case ROCP2P_PROBE_GET_PAGES:
		return ioctl_get_pages(ctx, arg);
	case ROCP2P_PROBE_PUT_PAGES:
		return ioctl_put_pages(ctx, arg);
	case ROCP2P_PROBE_GET_INFO:
		return ioctl_get_info(ctx, arg);
	default:
		return -ENOTTY;
	}
}

/* Map pinned GPU pages into CPU user space.  vm_pgoff carries the GPU
 * VA (page-aligned).  CPU-visible BAR access assumes bus address ==
 * CPU physical BAR address (probe-only diagnostic; IOMMU off — the
 * bridge's data path has no such assumption).
 *
 * Reference defects fixed here: every sg entry is mapped (not just the
 * first), each with its own clamped length, and the containment test
 * requires the requested vma range to lie INSIDE the pinned range. */
static int probe_mmap(struct file *filp, struct vm_area_struct *vma)
{
	struct probe_ctx *ctx = filp->private_data;
	struct probe_pin *pin;
	struct scatterlist *sg;
	u64 want_va = (u64)vma->vm_pgoff << PAGE_SHIFT;
	u64 want_len = vma->vm_end - vma->vm_start;
	u64 seg_start, seg_end, ov_start, ov_end;
	unsigned long vaddr;
	int i, ret = -ENOENT;

	mutex_lock(&ctx->lock);
	list_for_each_entry(pin, &ctx->pins, node) {
		if (!pin->info || !pin->info->pages)
			continue;
		/* vma inside pin (NOT pin inside vma) */
		if (want_va < pin->va ||
		    want_va + want_len > pin->va + pin->size)
			continue;

		vm_flags_set(vma, VM_IO | VM_PFNMAP | VM_DONTEXPAND |
				  VM_DONTDUMP);
		vma->vm_page_prot = pgprot_noncached(vma->vm_page_prot);

		ret = 0;
		seg_start = pin->va;	/* GPU VA the sg walk has reached */
		vaddr = vma->vm_start;
		for_each_sg(pin->info->pages->sgl, sg,
			    pin->info->pages->nents, i) {
			seg_end = seg_start + sg_dma_len(sg);
			ov_start = want_va > seg_start ? want_va : seg_start;
			ov_end = (want_va + want_len) < seg_end ?
					 (want_va + want_len) : seg_end;
			if (ov_start < ov_end) {
				u64 pa = sg_dma_address(sg) +
					 (ov_start - seg_start);
				if (pa & ~PAGE_MASK) {
					/* Sub-page offset would be silently
					 * truncated by the pfn shift and map
					 * the wrong bytes: refuse (pin the
					 * range page-aligned instead). */
					ret = -EINVAL;
					break;
				}
				ret = io_remap_pfn_range(
					vma, vaddr,
					(unsigned long)(pa >> PAGE_SHIFT),
					(unsigned long)(ov_end - ov_start),
					vma->vm_page_prot);
				if (ret)
					break;
				vaddr += ov_end - ov_start;
			}
			seg_start = seg_end;
			if (seg_start >= want_va + want_len)
				break;
		}
		break;
	}
	mutex_unlock(&ctx->lock);
	return ret;
}

static const struct file_operations probe_fops = {
	.owner = THIS_MODULE,
	.open = probe_open,
	.release = probe_release,
	.unlocked_ioctl = probe_unlocked_ioctl,
	.mmap = probe_mmap,
};

static struct miscdevice probe_dev = {
	.minor = MISC_DYNAMIC_MINOR,
	.name = ROCP2P_PROBE_DEVICE_NAME,
	.fops = &probe_fops,
	.mode = 0600,	/* overridden by the devmode param at init */
};

static int __init probe_init(void)
{
	int ret;

	ret = amdkfd_query_rdma_interface(&rdma_interface);
	if (ret < 0 || !rdma_interface) {
		pp_err("amdkfd RDMA interface unavailable (%d)\n", ret);
		return ret < 0 ? ret : -ENODEV;
	}
	/* liveness log, as the reference did (amdp2ptest.c:441-445) */
	pp_info("kfd vtable: get_pages %p put_pages %p is_gpu %p pagesz %p\n",
		rdma_interface->get_pages, rdma_interface->put_pages,
		rdma_interface->is_gpu_address,
		rdma_interface->get_page_size);
	probe_dev.mode = devmode;
	ret = misc_register(&probe_dev);
	if (ret)
		pp_err("misc_register failed: %d\n", ret);
	return ret;
}

static void __exit probe_exit(void)
{
	misc_deregister(&probe_dev);
}

module_init(probe_init);
module_exit(probe_exit);
