/* SPDX-License-Identifier: GPL-2.0 OR MIT
 *
 * fake_ibcore.c — see fake_ibcore.h.
 */
#define __ROCNR_SHIM__ 1
#include "fake_ibcore.h"

static const struct peer_memory_client *fc_client;
static long fc_invalidate_calls;
static int fc_is_ex;
static u32 fc_flags;
static int fc_async_invalidate;
static long fc_pending_teardowns;
static pthread_mutex_t fc_lock = PTHREAD_MUTEX_INITIALIZER;
static pthread_cond_t fc_idle = PTHREAD_COND_INITIALIZER;

const struct peer_memory_client *fake_ib_client(void) { return fc_client; }
long fake_ib_invalidate_count(void) { return fc_invalidate_calls; }
void fake_ib_reset_stats(void) { fc_invalidate_calls = 0; }
int fake_ib_client_is_ex(void) { return fc_is_ex; }
u32 fake_ib_client_flags(void) { return fc_flags; }

void fake_ib_set_async_invalidate(int on)
{
	pthread_mutex_lock(&fc_lock);
	fc_async_invalidate = on;
	pthread_mutex_unlock(&fc_lock);
}

void fake_ib_quiesce(void)
{
	pthread_mutex_lock(&fc_lock);
	while (fc_pending_teardowns > 0)
		pthread_cond_wait(&fc_idle, &fc_lock);
	pthread_mutex_unlock(&fc_lock);
}

/* One-shot MR teardown (dma_unmap → put_pages → release), the order the
 * IB core uses on both ibv_dereg_mr and invalidation (reference flow:
 * SURVEY.md §3.3/§3.4). */
static int mr_teardown(struct fake_ib_mr *mr, int from_invalidate)
{
	mutex_lock(&mr->lock);
	if (mr->dead) {
		mutex_unlock(&mr->lock);
		return 0;
	}
	mr->dead = 1;
	if (from_invalidate)
		mr->invalidated = 1;
	mutex_unlock(&mr->lock);

	/* Ex-generation core honoring PEER_MEM_INVALIDATE_UNMAPS: the
	 * client declared its invalidation flow leaves nothing to unmap,
	 * so the core skips dma_unmap on the invalidate path (the client
	 * must reclaim per-device mappings in release()). */
#if ROCNR_PEER_MEM_HAS_EX
	if (!(from_invalidate && fc_is_ex &&
	      (fc_flags & PEER_MEM_INVALIDATE_UNMAPS)))
#endif
		fc_client->dma_unmap(&mr->sgt, mr->client_context, mr->dev);
	fc_client->put_pages(&mr->sgt, mr->client_context);
	fc_client->release(mr->client_context);
	return 0;
}

static void *teardown_thread(void *arg)
{
	struct fake_ib_mr *mr = arg;
	struct timespec ts = { 0, (long)(rand() % 20000) };

	nanosleep(&ts, NULL);
	mr_teardown(mr, 1);
	pthread_mutex_lock(&fc_lock);
	fc_pending_teardowns--;
	if (!fc_pending_teardowns)
		pthread_cond_broadcast(&fc_idle);
	pthread_mutex_unlock(&fc_lock);
	return NULL;
}

static int fc_invalidate(void *reg_handle, rocnr_core_context_t core_context)
{
	struct fake_ib_mr *mr = (struct fake_ib_mr *)(uintptr_t)core_context;

	(void)reg_handle;
	__atomic_fetch_add(&fc_invalidate_calls, 1, __ATOMIC_SEQ_CST);
	pthread_mutex_lock(&fc_lock);
	if (fc_async_invalidate) {
		/* Core-owned invalidation: return to the client NOW (its
		 * producer reclaims the pin when the free callback
		 * returns) and tear the MR down later from a core
		 * thread — put_pages/release arrive AFTER the pin is
		 * gone, which the client's state machine must absorb. */
		pthread_t th;

		fc_pending_teardowns++;
		pthread_mutex_unlock(&fc_lock);
		if (pthread_create(&th, NULL, teardown_thread, mr) == 0) {
			pthread_detach(th);
			return 0;
		}
		pthread_mutex_lock(&fc_lock);
		fc_pending_teardowns--;
		pthread_mutex_unlock(&fc_lock);
		return mr_teardown(mr, 1);
	}
	pthread_mutex_unlock(&fc_lock);
	return mr_teardown(mr, 1);
}

static char fc_reg_handle_storage;

void *ib_register_peer_memory_client(const struct peer_memory_client *client,
				     invalidate_peer_memory *invalidate_cb)
{
	if (fc_client || !client || !client->acquire || !client->get_pages ||
	    !client->dma_map || !client->dma_unmap || !client->put_pages ||
	    !client->get_page_size || !client->release)
		return NULL;
	fc_is_ex = 0;
	fc_flags = 0;
#if ROCNR_PEER_MEM_HAS_EX
	/* Extended-registration detection, the public convention of the
	 * ex generation: version[] last byte == 1 marks the client as
	 * the head of a peer_memory_client_ex. */
	if (client->version[IB_PEER_MEMORY_VER_MAX - 1] == 1) {
		const struct peer_memory_client_ex *ex =
			(const struct peer_memory_client_ex *)client;

		if (ex->ex_size >= sizeof(*ex)) {
			fc_is_ex = 1;
			fc_flags = ex->flags;
		}
	}
#endif
	fc_client = client;
	*invalidate_cb = fc_invalidate;
	return &fc_reg_handle_storage;
}

void ib_unregister_peer_memory_client(void *reg_handle)
{
	if (reg_handle == &fc_reg_handle_storage)
		fc_client = NULL;
}

int fake_ib_reg_mr(unsigned long addr, size_t size, struct device *dev,
		   struct fake_ib_mr **out)
{
	struct fake_ib_mr *mr;
	char name_buf[IB_PEER_MEMORY_NAME_MAX];
	void *cctx = NULL;
	int ret;

	if (!fc_client)
		return -ENODEV;
	if (!fc_client->acquire(addr, size, NULL, name_buf, &cctx))
		return -ENODEV;	/* not peer memory: CPU fallback */

	mr = calloc(1, sizeof(*mr));
	mutex_init(&mr->lock);
	mr->client_context = cctx;
	mr->dev = dev;

	ret = fc_client->get_pages(addr, size, 1, 0, &mr->sgt, cctx,
				   (rocnr_core_context_t)(uintptr_t)mr);
	if (ret) {
		fc_client->release(cctx);
		free(mr);
		return ret;
	}

	mr->page_size = fc_client->get_page_size(cctx);

	ret = fc_client->dma_map(&mr->sgt, cctx, dev, 0, &mr->nmap);
	if (ret) {
		fc_client->put_pages(&mr->sgt, cctx);
		fc_client->release(cctx);
		free(mr);
		return ret;
	}

	*out = mr;
	return 0;
}

int fake_ib_mr_map_also(struct fake_ib_mr *mr, struct device *dev2,
			struct sg_table *sgt_out, int *nmap_out)
{
	if (!fc_client || mr->dead)
		return -EINVAL;
	return fc_client->dma_map(sgt_out, mr->client_context, dev2, 0,
				  nmap_out);
}

int fake_ib_mr_unmap_also(struct fake_ib_mr *mr, struct device *dev2,
			  struct sg_table *sgt)
{
	if (!fc_client)
		return -EINVAL;
	return fc_client->dma_unmap(sgt, mr->client_context, dev2);
}

int fake_ib_dereg_mr(struct fake_ib_mr *mr)
{
	if (!fc_client || !mr)
		return -EINVAL;
	return mr_teardown(mr, 0);
}
