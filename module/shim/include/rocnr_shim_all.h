/* SPDX-License-Identifier: GPL-2.0 OR MIT
 *
 * rocnr_shim_all.h — userspace kernel shim.
 *
 * The kernel-module sources under module/bridge and module/probe compile
 * unmodified against this header set (module/shim/include/linux/<h>.h are
 * one-line forwards here) and run as ordinary user code, so lifetime and
 * invalidation logic is unit- and race-tested on a box with no kernel
 * headers, no HCA and no GPU (this replaces the reference's only test
 * apparatus, a kernel-mode driver needing real hardware — reference:
 * /root/reference/tests/amdp2ptest.c).
 *
 * Only the API surface the modules actually use is provided.  Semantics
 * mirror the kernel closely enough for logic tests: pid refcounts are real
 * (leak-checked), mutexes are pthread mutexes, dma_map_resource applies a
 * per-device IOVA offset (simulated IOMMU) and is pair-checked against
 * dma_unmap_resource.
 */
#ifndef ROCNR_SHIM_ALL_H_
#define ROCNR_SHIM_ALL_H_

#ifndef __ROCNR_SHIM__
#error "shim headers require -D__ROCNR_SHIM__"
#endif

#include <stdint.h>
#include <stddef.h>
#include <stdbool.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <errno.h>
#include <pthread.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- basic types ---- */
typedef uint8_t  u8;
typedef uint16_t u16;
typedef uint32_t u32;
typedef unsigned short ushort;
/* unsigned long long (not uint64_t) so %llx format strings match the
 * kernel's u64 without warnings on LP64 userspace */
typedef unsigned long long u64;
typedef long long s64;
typedef u64 phys_addr_t;
typedef u64 dma_addr_t;
typedef unsigned int gfp_t;
#define GFP_KERNEL 0
#define GFP_ATOMIC 1

#define PAGE_SHIFT 12
#ifndef PAGE_SIZE
#define PAGE_SIZE 4096UL
#endif
#define PAGE_MASK (~(PAGE_SIZE - 1))

/* ---- printk ---- */
int rocnr_shim_printk(const char *level, const char *fmt, ...)
	__attribute__((format(printf, 2, 3)));
#define pr_debug(...) rocnr_shim_printk("dbg", __VA_ARGS__)
#define pr_info(...)  rocnr_shim_printk("info", __VA_ARGS__)
#define pr_warn(...)  rocnr_shim_printk("warn", __VA_ARGS__)
#define pr_err(...)   rocnr_shim_printk("err", __VA_ARGS__)

/* ---- compiler.h ---- */
#define READ_ONCE(x) (*(const volatile __typeof__(x) *)&(x))
#define WRITE_ONCE(x, val) (*((volatile __typeof__(x) *)&(x)) = (val))
#define likely(x) (x)
#define unlikely(x) (x)
#define __user
#define __init
#define __exit
#define __force

/* ---- slab ---- */
void *rocnr_shim_kzalloc(size_t sz);
void *rocnr_shim_kmalloc(size_t sz);
void rocnr_shim_kfree(void *p);
long rocnr_shim_alloc_balance(void);
#define kzalloc(sz, gfp) rocnr_shim_kzalloc(sz)
#define kmalloc(sz, gfp) rocnr_shim_kmalloc(sz)
#define kcalloc(n, sz, gfp) rocnr_shim_kzalloc((n) * (sz))
#define kfree(p) rocnr_shim_kfree(p)

/* ---- string ---- */
static inline size_t strscpy(char *dst, const char *src, size_t n)
{
	size_t l;

	if (!n)
		return (size_t)-7 /* -E2BIG */;
	l = strlen(src);
	if (l >= n)
		l = n - 1;
	memcpy(dst, src, l);
	dst[l] = 0;
	return l;
}
#define scnprintf snprintf

/* ---- mutex ---- */
struct mutex {
	pthread_mutex_t m;
};
static inline void mutex_init(struct mutex *mx)
{
	pthread_mutexattr_t a;

	pthread_mutexattr_init(&a);
	pthread_mutexattr_settype(&a, PTHREAD_MUTEX_ERRORCHECK);
	pthread_mutex_init(&mx->m, &a);
	pthread_mutexattr_destroy(&a);
}
static inline void mutex_lock(struct mutex *mx)
{
	int r = pthread_mutex_lock(&mx->m);

	if (r) {
		fprintf(stderr, "shim: mutex_lock error %d (deadlock?)\n", r);
		abort();
	}
}
static inline void mutex_unlock(struct mutex *mx)
{
	pthread_mutex_unlock(&mx->m);
}
static inline void mutex_destroy(struct mutex *mx)
{
	pthread_mutex_destroy(&mx->m);
}

/* ---- atomic ---- */
typedef struct {
	volatile long long counter;
} atomic64_t;
typedef struct {
	volatile int counter;
} atomic_t;
#define ATOMIC64_INIT(v) { (v) }
#define ATOMIC_INIT(v) { (v) }
static inline long long atomic64_read(const atomic64_t *a)
{
	return __atomic_load_n(&a->counter, __ATOMIC_SEQ_CST);
}
static inline void atomic64_add(long long v, atomic64_t *a)
{
	__atomic_fetch_add(&a->counter, v, __ATOMIC_SEQ_CST);
}
static inline void atomic64_sub(long long v, atomic64_t *a)
{
	__atomic_fetch_sub(&a->counter, v, __ATOMIC_SEQ_CST);
}
static inline void atomic64_inc(atomic64_t *a) { atomic64_add(1, a); }
static inline void atomic64_dec(atomic64_t *a) { atomic64_sub(1, a); }
static inline int atomic_read(const atomic_t *a)
{
	return __atomic_load_n(&a->counter, __ATOMIC_SEQ_CST);
}
static inline void atomic_inc(atomic_t *a)
{
	__atomic_fetch_add(&a->counter, 1, __ATOMIC_SEQ_CST);
}
static inline int atomic_dec_and_test(atomic_t *a)
{
	return __atomic_sub_fetch(&a->counter, 1, __ATOMIC_SEQ_CST) == 0;
}

/* ---- list ---- */
struct list_head {
	struct list_head *next, *prev;
};
#define LIST_HEAD_INIT(name) { &(name), &(name) }
#define LIST_HEAD(name) struct list_head name = LIST_HEAD_INIT(name)
static inline void INIT_LIST_HEAD(struct list_head *h)
{
	h->next = h;
	h->prev = h;
}
static inline void list_add(struct list_head *n, struct list_head *h)
{
	n->next = h->next;
	n->prev = h;
	h->next->prev = n;
	h->next = n;
}
static inline void list_add_tail(struct list_head *n, struct list_head *h)
{
	n->next = h;
	n->prev = h->prev;
	h->prev->next = n;
	h->prev = n;
}
static inline void list_del(struct list_head *e)
{
	e->prev->next = e->next;
	e->next->prev = e->prev;
	e->next = (struct list_head *)0x1;
	e->prev = (struct list_head *)0x2;
}
static inline int list_empty(const struct list_head *h)
{
	return h->next == h;
}
#define container_of(ptr, type, member) \
	((type *)((char *)(ptr) - offsetof(type, member)))
#define list_entry(ptr, type, member) container_of(ptr, type, member)
#define list_first_entry(h, type, member) list_entry((h)->next, type, member)
#define list_for_each_entry(pos, head, member) \
	for (pos = list_entry((head)->next, __typeof__(*pos), member); \
	     &pos->member != (head); \
	     pos = list_entry(pos->member.next, __typeof__(*pos), member))
#define list_for_each_entry_safe(pos, n, head, member) \
	for (pos = list_entry((head)->next, __typeof__(*pos), member), \
	     n = list_entry(pos->member.next, __typeof__(*pos), member); \
	     &pos->member != (head); \
	     pos = n, n = list_entry(n->member.next, __typeof__(*pos), member))

/* ---- pid / sched ---- */
struct pid {
	atomic_t refs;
	int nr;
};
enum pid_type { PIDTYPE_PID = 0 };
struct task_struct {
	int dummy;
};
struct task_struct *rocnr_shim_current(void);
#define current rocnr_shim_current()
struct pid *get_task_pid(struct task_struct *t, enum pid_type type);
void put_pid(struct pid *pid);
long rocnr_shim_pid_balance(void);

/* ---- module ---- */
struct module {
	atomic_t refcnt;
};
extern struct module rocnr_shim_this_module;
#define THIS_MODULE (&rocnr_shim_this_module)
static inline void __module_get(struct module *m) { atomic_inc(&m->refcnt); }
static inline void module_put(struct module *m)
{
	(void)atomic_dec_and_test(&m->refcnt);
}
static inline int rocnr_shim_module_refcount(void)
{
	return atomic_read(&rocnr_shim_this_module.refcnt);
}
#define MODULE_AUTHOR(x)
#define MODULE_LICENSE(x)
#define MODULE_DESCRIPTION(x)
#define MODULE_VERSION(x)
#define module_init(fn) int rocnr_shim_module_init(void) { return fn(); }
#define module_exit(fn) void rocnr_shim_module_exit(void) { fn(); }
int rocnr_shim_module_init(void);
void rocnr_shim_module_exit(void);

/* ---- moduleparam ---- */
struct kernel_param {
	void *arg;
};
struct kernel_param_ops {
	int (*get)(char *buffer, const struct kernel_param *kp);
	int (*set)(const char *val, const struct kernel_param *kp);
};
#define module_param_cb(name, ops, arg, perm) \
	static const void *__rocnr_param_##name[] \
		__attribute__((unused)) = { (const void *)(ops), (void *)(arg) }
#define module_param(name, type, perm)
#define module_param_named(n, v, type, perm)
#define MODULE_PARM_DESC(var, desc)

/* ---- scatterlist ---- */
struct scatterlist {
	dma_addr_t dma_address;
	unsigned int dma_length;
	unsigned int length;
	int is_last;
};
struct sg_table {
	struct scatterlist *sgl;
	unsigned int nents;
	unsigned int orig_nents;
};
#define sg_dma_address(sg) ((sg)->dma_address)
#define sg_dma_len(sg) ((sg)->dma_length)
static inline struct scatterlist *sg_next(struct scatterlist *sg)
{
	return sg->is_last ? NULL : sg + 1;
}
int sg_alloc_table(struct sg_table *t, unsigned int nents, gfp_t gfp);
void sg_free_table(struct sg_table *t);
long rocnr_shim_sg_balance(void);
#define for_each_sg(sgl, sg, nents, i) \
	for ((i) = 0, (sg) = (sgl); (size_t)(i) < (size_t)(nents) && (sg); \
	     (i)++, (sg) = sg_next(sg))

/* ---- device / dma-mapping ---- */
struct device {
	const char *name;
	u64 iova_offset;	/* simulated IOMMU translation */
	u64 max_seg;		/* 0 => default 4 GiB */
	long fail_after;	/* map calls until injected failure; -1 off */
	atomic64_t live_maps;
	atomic64_t map_calls;
};
enum dma_data_direction {
	DMA_BIDIRECTIONAL = 0,
	DMA_TO_DEVICE = 1,
	DMA_FROM_DEVICE = 2,
};
#define DMA_MAPPING_ERROR ((dma_addr_t)-1)
dma_addr_t dma_map_resource(struct device *dev, phys_addr_t phys, size_t size,
			    enum dma_data_direction dir, unsigned long attrs);
void dma_unmap_resource(struct device *dev, dma_addr_t addr, size_t size,
			enum dma_data_direction dir, unsigned long attrs);
static inline int dma_mapping_error(struct device *dev, dma_addr_t a)
{
	(void)dev;
	return a == DMA_MAPPING_ERROR;
}
static inline u64 dma_get_max_seg_size(struct device *dev)
{
	return dev->max_seg ? dev->max_seg : 0x100000000ULL;
}

/* ---- pci (BAR aperture probe) ---- */
#define PCI_VENDOR_ID_ATI 0x1002
#define PCI_ANY_ID (~0)
#define PCI_BASE_CLASS_DISPLAY 0x03
struct pci_dev {
	unsigned int vendor;
	unsigned int device;
	unsigned int class;	/* class<<16 | subclass<<8 | progif */
	u64 bar_len[2];
};
struct pci_dev *pci_get_device(unsigned int vendor, unsigned int device,
			       struct pci_dev *from);
static inline u64 pci_resource_len(struct pci_dev *d, int bar)
{
	return (bar >= 0 && bar < 2) ? d->bar_len[bar] : 0;
}
static inline void pci_dev_put(struct pci_dev *d) { (void)d; }
/* test control: install a fake device table (NULL, 0 to clear) */
void rocnr_shim_set_pci_devices(struct pci_dev *devs, int n);

/* ---- char device / uaccess / mmap (for the probe module) ---- */
struct inode {
	int i_dummy;
};
struct file {
	void *private_data;
};
struct vm_area_struct {
	unsigned long vm_start;
	unsigned long vm_end;
	unsigned long vm_pgoff;
	unsigned long vm_flags;
	unsigned long vm_page_prot;
};
#define VM_READ		0x1UL
#define VM_WRITE	0x2UL
#define VM_SHARED	0x8UL
#define VM_IO		0x4000UL
#define VM_PFNMAP	0x400UL
#define VM_DONTEXPAND	0x40000UL
#define VM_DONTDUMP	0x4000000UL
static inline void vm_flags_set(struct vm_area_struct *vma,
				unsigned long flags)
{
	vma->vm_flags |= flags;
}
#define pgprot_noncached(p) (p)

struct file_operations {
	struct module *owner;
	int (*open)(struct inode *, struct file *);
	int (*release)(struct inode *, struct file *);
	long (*unlocked_ioctl)(struct file *, unsigned int, unsigned long);
	int (*mmap)(struct file *, struct vm_area_struct *);
};

struct miscdevice {
	int minor;
	const char *name;
	const struct file_operations *fops;
	unsigned short mode;
};
#define MISC_DYNAMIC_MINOR 255
int misc_register(struct miscdevice *dev);
void misc_deregister(struct miscdevice *dev);
const struct miscdevice *rocnr_shim_misc_dev(void);

static inline unsigned long copy_from_user(void *to, const void *from,
					   unsigned long n)
{
	memcpy(to, from, n);
	return 0;
}
static inline unsigned long copy_to_user(void *to, const void *from,
					 unsigned long n)
{
	memcpy(to, from, n);
	return 0;
}

/* io_remap_pfn_range recorder: tests assert the exact mappings the
 * probe's mmap produced. */
struct rocnr_shim_map {
	unsigned long vaddr;
	unsigned long pfn;
	unsigned long size;
};
int io_remap_pfn_range(struct vm_area_struct *vma, unsigned long vaddr,
		       unsigned long pfn, unsigned long size,
		       unsigned long prot);
#define remap_pfn_range io_remap_pfn_range
void rocnr_shim_maps_reset(void);
long rocnr_shim_maps_count(void);
const struct rocnr_shim_map *rocnr_shim_maps_get(long i);

#ifdef __cplusplus
}
#endif

#endif /* ROCNR_SHIM_ALL_H_ */
