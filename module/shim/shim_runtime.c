/* SPDX-License-Identifier: GPL-2.0 OR MIT
 *
 * shim_runtime.c — stateful pieces of the userspace kernel shim:
 * allocation/pid/sg/dma-balance bookkeeping used by the leak tests.
 */
#include "rocnr_shim_all.h"
#include <stdarg.h>

static volatile long shim_allocs;
static volatile long shim_pids;
static volatile long shim_sgs;
static int shim_verbose;

struct module rocnr_shim_this_module = { .refcnt = ATOMIC_INIT(0) };

int rocnr_shim_printk(const char *level, const char *fmt, ...)
{
	va_list ap;
	int n = 0;

	if (!shim_verbose && !strcmp(level, "dbg"))
		return 0;
	fprintf(stderr, "[%s] ", level);
	va_start(ap, fmt);
	n = vfprintf(stderr, fmt, ap);
	va_end(ap);
	return n;
}

void rocnr_shim_set_verbose(int v) { shim_verbose = v; }

void *rocnr_shim_kzalloc(size_t sz)
{
	void *p = calloc(1, sz);

	if (p)
		__atomic_fetch_add(&shim_allocs, 1, __ATOMIC_SEQ_CST);
	return p;
}

void *rocnr_shim_kmalloc(size_t sz)
{
	void *p = malloc(sz);

	if (p)
		__atomic_fetch_add(&shim_allocs, 1, __ATOMIC_SEQ_CST);
	return p;
}

void rocnr_shim_kfree(void *p)
{
	if (!p)
		return;
	__atomic_fetch_sub(&shim_allocs, 1, __ATOMIC_SEQ_CST);
	free(p);
}

long rocnr_shim_alloc_balance(void) { return shim_allocs; }

/* ---- pid ---- */
static struct task_struct shim_task;

struct task_struct *rocnr_shim_current(void) { return &shim_task; }

struct pid *get_task_pid(struct task_struct *t, enum pid_type type)
{
	struct pid *p = malloc(sizeof(*p));

	(void)t;
	(void)type;
	p->refs.counter = 1;
	p->nr = 4242;
	__atomic_fetch_add(&shim_pids, 1, __ATOMIC_SEQ_CST);
	return p;
}

void put_pid(struct pid *pid)
{
	if (!pid)
		return;
	if (atomic_dec_and_test(&pid->refs)) {
		__atomic_fetch_sub(&shim_pids, 1, __ATOMIC_SEQ_CST);
		free(pid);
	}
}

long rocnr_shim_pid_balance(void) { return shim_pids; }

/* ---- scatterlist ---- */
int sg_alloc_table(struct sg_table *t, unsigned int nents, gfp_t gfp)
{
	(void)gfp;
	if (!nents)
		return -EINVAL;
	t->sgl = calloc(nents, sizeof(struct scatterlist));
	if (!t->sgl)
		return -ENOMEM;
	t->nents = nents;
	t->orig_nents = nents;
	t->sgl[nents - 1].is_last = 1;
	__atomic_fetch_add(&shim_sgs, 1, __ATOMIC_SEQ_CST);
	return 0;
}

void sg_free_table(struct sg_table *t)
{
	if (!t->sgl)
		return;
	free(t->sgl);
	t->sgl = NULL;
	t->nents = 0;
	__atomic_fetch_sub(&shim_sgs, 1, __ATOMIC_SEQ_CST);
}

long rocnr_shim_sg_balance(void) { return shim_sgs; }

/* ---- dma ---- */
dma_addr_t dma_map_resource(struct device *dev, phys_addr_t phys, size_t size,
			    enum dma_data_direction dir, unsigned long attrs)
{
	(void)dir;
	(void)attrs;
	(void)size;
	atomic64_inc(&dev->map_calls);
	if (dev->fail_after >= 0 &&
	    atomic64_read(&dev->map_calls) > dev->fail_after)
		return DMA_MAPPING_ERROR;
	atomic64_inc(&dev->live_maps);
	return phys + dev->iova_offset;
}

void dma_unmap_resource(struct device *dev, dma_addr_t addr, size_t size,
			enum dma_data_direction dir, unsigned long attrs)
{
	(void)addr;
	(void)size;
	(void)dir;
	(void)attrs;
	atomic64_dec(&dev->live_maps);
}

/* ---- pci ---- */
static struct pci_dev *shim_pci_devs;
static int shim_pci_n;

void rocnr_shim_set_pci_devices(struct pci_dev *devs, int n)
{
	shim_pci_devs = devs;
	shim_pci_n = n;
}

struct pci_dev *pci_get_device(unsigned int vendor, unsigned int device,
			       struct pci_dev *from)
{
	int i = 0;

	if (from)
		i = (int)(from - shim_pci_devs) + 1;
	for (; i < shim_pci_n; i++) {
		if (shim_pci_devs[i].vendor == vendor &&
		    (device == (unsigned int)PCI_ANY_ID ||
		     shim_pci_devs[i].device == device))
			return &shim_pci_devs[i];
	}
	return 0;
}

/* ---- misc device ---- */
static struct miscdevice *shim_misc;

int misc_register(struct miscdevice *dev)
{
	if (shim_misc)
		return -EBUSY;
	shim_misc = dev;
	return 0;
}

void misc_deregister(struct miscdevice *dev)
{
	if (shim_misc == dev)
		shim_misc = NULL;
}

const struct miscdevice *rocnr_shim_misc_dev(void)
{
	return shim_misc;
}

/* ---- io_remap_pfn_range recorder ---- */
#define SHIM_MAX_MAPS 512
static struct rocnr_shim_map shim_maps[SHIM_MAX_MAPS];
static long shim_nmaps;

int io_remap_pfn_range(struct vm_area_struct *vma, unsigned long vaddr,
		       unsigned long pfn, unsigned long size,
		       unsigned long prot)
{
	(void)vma;
	(void)prot;
	if (shim_nmaps >= SHIM_MAX_MAPS)
		return -ENOMEM;
	shim_maps[shim_nmaps].vaddr = vaddr;
	shim_maps[shim_nmaps].pfn = pfn;
	shim_maps[shim_nmaps].size = size;
	shim_nmaps++;
	return 0;
}

void rocnr_shim_maps_reset(void) { shim_nmaps = 0; }
long rocnr_shim_maps_count(void) { return shim_nmaps; }
const struct rocnr_shim_map *rocnr_shim_maps_get(long i)
{
	return (i >= 0 && i < shim_nmaps) ? &shim_maps[i] : 0;
}
