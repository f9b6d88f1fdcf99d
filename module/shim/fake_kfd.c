/* SPDX-License-Identifier: GPL-2.0 OR MIT
 *
 * fake_kfd.c — see fake_kfd.h.
 */
#define __ROCNR_SHIM__ 1
#include "fake_kfd.h"
#include "rocnr_amd_rdma.h"

struct fk_alloc {
	struct fk_alloc *next;
	uint64_t va;
	uint64_t size;
	unsigned int frag_every;
	uint64_t bus_base;
	uint8_t *backing;	/* real memory behind the bus range (or NULL) */
	uint64_t backing_len;
};

struct fk_pin {
	struct fk_pin *next;
	struct amd_p2p_info info;
	struct fk_alloc *owner;
	void (*free_cb)(void *);
	void *client_priv;
	int dying;		/* revoke started */
	int cb_in_flight;
};

static pthread_mutex_t fk_lock = PTHREAD_MUTEX_INITIALIZER;
static pthread_cond_t fk_cb_done = PTHREAD_COND_INITIALIZER;

static struct fk_alloc *fk_allocs;
static struct fk_pin *fk_pins;
static uint64_t fk_next_va = 0x700000000000ULL;
static uint64_t fk_next_bus = 0xd000000000ULL;
static long fk_stat_get, fk_stat_put, fk_stat_bad_put, fk_stat_cb;
static int fk_fail_page_size;
static int fk_reject_null_dev;
static int fk_revoke_in_get_pages;
static unsigned int fk_bus_skew;

static void fk_destroy_pin_locked(struct fk_pin *p);
static void fk_tomb_add_locked(const void *info);
static struct fk_alloc *fk_find_locked(uint64_t addr);

uint64_t fake_kfd_alloc(uint64_t size, unsigned int frag_every)
{
	struct fk_alloc *a = calloc(1, sizeof(*a));
	uint64_t nchunks;

	pthread_mutex_lock(&fk_lock);
	a->va = fk_next_va;
	a->size = size;
	a->frag_every = frag_every;
	a->bus_base = fk_next_bus + fk_bus_skew;
	nchunks = (size + FAKE_KFD_VRAM_PAGE - 1) / FAKE_KFD_VRAM_PAGE;
	fk_next_va += (size + (1ULL << 30)) & ~((1ULL << 21) - 1);
	/* leave a hole after the allocation in bus space too */
	fk_next_bus += (nchunks + 16) * FAKE_KFD_VRAM_PAGE +
		       (frag_every ? (nchunks / frag_every + 2) *
					     FAKE_KFD_VRAM_PAGE : 0);
	a->next = fk_allocs;
	fk_allocs = a;
	pthread_mutex_unlock(&fk_lock);
	return a->va;
}

uint64_t fake_kfd_alloc_backed(uint64_t size)
{
	struct fk_alloc *a = calloc(1, sizeof(*a));
	uint64_t nchunks = (size + FAKE_KFD_VRAM_PAGE - 1) / FAKE_KFD_VRAM_PAGE;

	a->backing_len = nchunks * FAKE_KFD_VRAM_PAGE;
	a->backing = aligned_alloc(4096, a->backing_len);
	if (!a->backing) {
		free(a);
		return 0;
	}
	pthread_mutex_lock(&fk_lock);
	a->va = fk_next_va;
	a->size = size;
	a->frag_every = 0;
	/* the bus range IS the backing memory: bus_to_ptr is identity */
	a->bus_base = (uint64_t)(uintptr_t)a->backing;
	fk_next_va += (size + (1ULL << 30)) & ~((1ULL << 21) - 1);
	a->next = fk_allocs;
	fk_allocs = a;
	pthread_mutex_unlock(&fk_lock);
	return a->va;
}

void *fake_kfd_bus_to_ptr(uint64_t bus)
{
	struct fk_alloc *a;
	void *p = NULL;

	pthread_mutex_lock(&fk_lock);
	for (a = fk_allocs; a; a = a->next) {
		if (a->backing && bus >= a->bus_base &&
		    bus < a->bus_base + a->backing_len) {
			p = a->backing + (bus - a->bus_base);
			break;
		}
	}
	pthread_mutex_unlock(&fk_lock);
	return p;
}

uint64_t fake_kfd_backing_bus(uint64_t va)
{
	struct fk_alloc *a;
	uint64_t bus = 0;

	pthread_mutex_lock(&fk_lock);
	a = fk_find_locked(va);
	if (a)
		bus = a->bus_base;
	pthread_mutex_unlock(&fk_lock);
	return bus;
}

static struct fk_alloc *fk_find_locked(uint64_t addr)
{
	struct fk_alloc *a;

	for (a = fk_allocs; a; a = a->next)
		if (addr >= a->va && addr < a->va + a->size)
			return a;
	return NULL;
}

/* Bus address of chunk i of allocation a (holes per frag_every). */
static uint64_t fk_chunk_bus(const struct fk_alloc *a, uint64_t i)
{
	uint64_t holes = a->frag_every ? i / a->frag_every : 0;

	return a->bus_base + i * FAKE_KFD_VRAM_PAGE +
	       holes * FAKE_KFD_VRAM_PAGE;
}

static int fk_is_gpu_address(uint64_t address, struct pid *pid)
{
	int hit;

	(void)pid;
	pthread_mutex_lock(&fk_lock);
	hit = fk_find_locked(address) != NULL;
	pthread_mutex_unlock(&fk_lock);
	return hit;
}

static int fk_get_pages(uint64_t address, uint64_t length, struct pid *pid,
#if ROCNR_AMD_RDMA_HAS_DMA_DEV
			struct device *dma_dev,
#endif
			struct amd_p2p_info **amd_p2p_data,
			void (*free_callback)(void *), void *client_priv)
{
	struct fk_alloc *a;
	struct fk_pin *p;
	uint64_t first, last, nchunks, i, off, end;
	struct scatterlist *sg;

	(void)pid;
	pthread_mutex_lock(&fk_lock);
#if ROCNR_AMD_RDMA_HAS_DMA_DEV
	if (fk_reject_null_dev && !dma_dev) {
		pthread_mutex_unlock(&fk_lock);
		return -EINVAL;	/* drift: KFD requires the DMA device */
	}
#endif
	a = fk_find_locked(address);
	if (!a || address + length > a->va + a->size || !length) {
		pthread_mutex_unlock(&fk_lock);
		return -EINVAL;
	}

	p = calloc(1, sizeof(*p));
	first = (address - a->va) / FAKE_KFD_VRAM_PAGE;
	last = (address - a->va + length - 1) / FAKE_KFD_VRAM_PAGE;
	nchunks = last - first + 1;

	p->info.va = address;
	p->info.size = length;
	p->info.pid = pid;
	p->info.priv = p;
	p->info.pages = calloc(1, sizeof(struct sg_table));
	sg_alloc_table(p->info.pages, (unsigned int)nchunks, GFP_KERNEL);
	for (i = 0, sg = p->info.pages->sgl; i < nchunks; i++, sg++) {
		off = (first + i) * FAKE_KFD_VRAM_PAGE;
		end = off + FAKE_KFD_VRAM_PAGE;
		if (off < address - a->va)
			off = address - a->va;
		if (end > address - a->va + length)
			end = address - a->va + length;
		sg->dma_address = fk_chunk_bus(a, first + i) +
				  (off - (first + i) * FAKE_KFD_VRAM_PAGE);
#if ROCNR_AMD_RDMA_HAS_DMA_DEV
		/* Drift mode with a real device: KFD device-maps the pin
		 * internally (the addresses are iovas for dma_dev). */
		if (fk_reject_null_dev && dma_dev)
			sg->dma_address += dma_dev->iova_offset;
#endif
		sg->dma_length = (unsigned int)(end - off);
		sg->length = sg->dma_length;
	}
	p->owner = a;
	p->free_cb = free_callback;
	p->client_priv = client_priv;
	p->next = fk_pins;
	fk_pins = p;
	fk_stat_get++;
	*amd_p2p_data = &p->info;

	if (fk_revoke_in_get_pages > 0) {
		/* Buffer freed while the pin was being installed: fire the
		 * free callback synchronously BEFORE get_pages returns, then
		 * reclaim — the caller gets a pointer to a dead pin and must
		 * not touch it (its own state machine has seen the revoke). */
		fk_revoke_in_get_pages--;
		p->dying = 1;
		pthread_mutex_unlock(&fk_lock);
		if (free_callback) {
			fk_stat_cb++;
			free_callback(client_priv);
		}
		pthread_mutex_lock(&fk_lock);
		fk_tomb_add_locked(&p->info);
		fk_destroy_pin_locked(p);
		pthread_cond_broadcast(&fk_cb_done);
		pthread_mutex_unlock(&fk_lock);
		return 0;
	}

	pthread_mutex_unlock(&fk_lock);
	return 0;
}

static void fk_destroy_pin_locked(struct fk_pin *p)
{
	struct fk_pin **pp;

	for (pp = &fk_pins; *pp; pp = &(*pp)->next) {
		if (*pp == p) {
			*pp = p->next;
			break;
		}
	}
	sg_free_table(p->info.pages);
	free(p->info.pages);
	free(p);
}

/* Tombstones of recently revoked pins: a concurrent put_pages that lost
 * the race against a revoke is benign ("KFD serializes internally"); a
 * put of a pointer KFD never revoked nor tracks is a caller bug. */
#define FK_TOMBSTONES 64
static const void *fk_tombs[FK_TOMBSTONES];
static unsigned int fk_tomb_idx;

static void fk_tomb_add_locked(const void *info)
{
	fk_tombs[fk_tomb_idx++ % FK_TOMBSTONES] = info;
}

static int fk_tomb_has_locked(const void *info)
{
	unsigned int i;

	for (i = 0; i < FK_TOMBSTONES; i++)
		if (fk_tombs[i] == info)
			return 1;
	return 0;
}

static int fk_put_pages(struct amd_p2p_info **amd_p2p_data)
{
	struct amd_p2p_info *info;
	struct fk_pin *it, *p = NULL;

	if (!amd_p2p_data || !*amd_p2p_data)
		return -EINVAL;
	/* Pointer-identity lookup only — the info may already have been
	 * reclaimed by a concurrent revoke, so it must NOT be
	 * dereferenced before the registry confirms it is live. */
	info = *amd_p2p_data;

	pthread_mutex_lock(&fk_lock);
rescan:
	p = NULL;
	for (it = fk_pins; it; it = it->next)
		if (&it->info == info) {
			p = it;
			break;
		}
	if (!p) {
		if (fk_tomb_has_locked(info)) {
			/* benign: revoke finished first */
			pthread_mutex_unlock(&fk_lock);
			*amd_p2p_data = NULL;
			return 0;
		}
		fk_stat_bad_put++;
		pthread_mutex_unlock(&fk_lock);
		return -EINVAL;
	}
	if (p->dying) {
		/* Revoke in progress.  The hard contract a real KFD must
		 * honor: put_pages returns only after any in-flight free
		 * callback for the pin has completed, because the caller
		 * frees its callback context right after this returns
		 * (violating it shows up as a use-after-free in the
		 * bridge's free callback — caught by the ASan build).
		 * Wait it out, then rescan: the pin dies with the revoke
		 * and the tombstone makes the rescan return benign. */
		pthread_cond_wait(&fk_cb_done, &fk_lock);
		/* p may be freed once the lock was dropped — do not touch
		 * it again; rescan by identity (spurious wakeups land
		 * back here, completion lands on the tombstone). */
		goto rescan;
	}
	fk_stat_put++;
	fk_tomb_add_locked(info);
	fk_destroy_pin_locked(p);
	pthread_mutex_unlock(&fk_lock);
	*amd_p2p_data = NULL;
	return 0;
}

static int fk_get_page_size(uint64_t address, uint64_t length,
			    struct pid *pid, unsigned long *page_size)
{
	int ok;

	(void)pid;
	pthread_mutex_lock(&fk_lock);
	if (fk_fail_page_size > 0) {
		fk_fail_page_size--;
		pthread_mutex_unlock(&fk_lock);
		return -EIO;
	}
	ok = fk_find_locked(address) != NULL && length > 0;
	pthread_mutex_unlock(&fk_lock);
	if (!ok)
		return -EINVAL;
	*page_size = (unsigned long)FAKE_KFD_VRAM_PAGE;
	return 0;
}

void fake_kfd_fail_page_size(int n)
{
	pthread_mutex_lock(&fk_lock);
	fk_fail_page_size = n;
	pthread_mutex_unlock(&fk_lock);
}

void fake_kfd_bus_skew(unsigned int bytes)
{
	pthread_mutex_lock(&fk_lock);
	fk_bus_skew = bytes;
	pthread_mutex_unlock(&fk_lock);
}

void fake_kfd_reject_null_dev(int on)
{
	pthread_mutex_lock(&fk_lock);
	fk_reject_null_dev = on;
	pthread_mutex_unlock(&fk_lock);
}

void fake_kfd_revoke_in_get_pages(int n)
{
	pthread_mutex_lock(&fk_lock);
	fk_revoke_in_get_pages = n;
	pthread_mutex_unlock(&fk_lock);
}

void fake_kfd_free(uint64_t va)
{
	struct fk_alloc *a, **ap;
	struct fk_pin *p;
	void (*cb)(void *);
	void *priv;

	for (;;) {
		cb = NULL;
		priv = NULL;
		pthread_mutex_lock(&fk_lock);
		a = fk_find_locked(va);
		if (!a) {
			pthread_mutex_unlock(&fk_lock);
			return;
		}
		for (p = fk_pins; p; p = p->next) {
			if (p->owner == a && !p->dying) {
				p->dying = 1;
				p->cb_in_flight = 1;
				cb = p->free_cb;
				priv = p->client_priv;
				break;
			}
		}
		if (!p) {
			/* No live pins left: drop the allocation. */
			for (ap = &fk_allocs; *ap; ap = &(*ap)->next) {
				if (*ap == a) {
					*ap = a->next;
					break;
				}
			}
			pthread_mutex_unlock(&fk_lock);
			free(a->backing);
			free(a);
			return;
		}
		pthread_mutex_unlock(&fk_lock);

		if (cb) {
			fk_stat_cb++;
			cb(priv);	/* bridge's invalidation entry */
		}

		/* Resources stay valid until the callback returns (the
		 * contract the reference documents at amdp2p.c:105-107);
		 * reclaim now. */
		pthread_mutex_lock(&fk_lock);
		p->cb_in_flight = 0;
		fk_tomb_add_locked(&p->info);
		fk_destroy_pin_locked(p);
		pthread_cond_broadcast(&fk_cb_done);
		pthread_mutex_unlock(&fk_lock);
	}
}

void fake_kfd_reset(void)
{
	struct fk_alloc *a;
	struct fk_pin *p;

	pthread_mutex_lock(&fk_lock);
	while ((p = fk_pins))
		fk_destroy_pin_locked(p);
	while ((a = fk_allocs)) {
		fk_allocs = a->next;
		free(a->backing);
		free(a);
	}
	fk_stat_get = fk_stat_put = fk_stat_bad_put = fk_stat_cb = 0;
	pthread_mutex_unlock(&fk_lock);
}

long fake_kfd_live_pins(void)
{
	struct fk_pin *p;
	long n = 0;

	pthread_mutex_lock(&fk_lock);
	for (p = fk_pins; p; p = p->next)
		n++;
	pthread_mutex_unlock(&fk_lock);
	return n;
}

long fake_kfd_get_pages_calls(void) { return fk_stat_get; }
long fake_kfd_put_pages_calls(void) { return fk_stat_put; }
long fake_kfd_bad_put_calls(void) { return fk_stat_bad_put; }
long fake_kfd_callbacks_fired(void) { return fk_stat_cb; }

static const struct amd_rdma_interface fk_vtable = {
	.get_pages = fk_get_pages,
	.put_pages = fk_put_pages,
	.is_gpu_address = fk_is_gpu_address,
	.get_page_size = fk_get_page_size,
};

int amdkfd_query_rdma_interface(const struct amd_rdma_interface **rdma)
{
	*rdma = &fk_vtable;
	return 0;
}
