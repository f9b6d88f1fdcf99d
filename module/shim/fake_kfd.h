/* SPDX-License-Identifier: GPL-2.0 OR MIT
 *
 * fake_kfd.h — userspace stand-in for the amdkfd amd_rdma interface
 * (contract: module/include/rocnr_amd_rdma.h).  Lets the bridge/probe
 * modules run real pin/unpin/invalidate flows, including the async revoke
 * race, with no GPU.
 *
 * Contract implemented (documented in the reference at
 * /root/reference/amdp2p.c:105-107 and re-stated here):
 *  - get_pages pins [va, va+size) of a fake GPU allocation and returns an
 *    amd_p2p_info whose sg table holds synthetic bus addresses at 2 MiB
 *    granularity (optionally fragmented, to exercise coalescing);
 *  - fake_kfd_free() revokes an allocation: each live pin's free_callback
 *    is invoked, then the pin's resources are reclaimed;
 *  - put_pages serializes against in-flight free callbacks for the same
 *    pin (waits for them), as a real KFD must for the peer contract to be
 *    sound.
 */
#ifndef ROCNR_FAKE_KFD_H_
#define ROCNR_FAKE_KFD_H_

#include "rocnr_shim_all.h"

#ifdef __cplusplus
extern "C" {
#endif

#define FAKE_KFD_VRAM_PAGE (2ULL << 20)	/* MI355X VRAM granule */

/* Create a fake GPU allocation.  frag_every > 0 inserts a bus-address
 * hole after every frag_every-th 2 MiB chunk (so a pin coalesces into
 * ceil(nchunks / frag_every) runs); 0 = fully bus-contiguous.  Returns
 * the GPU VA. */
uint64_t fake_kfd_alloc(uint64_t size, unsigned int frag_every);

/* Free an allocation: fires free callbacks on all live pins (the async
 * invalidation path), then reclaims them.  Safe from any thread. */
void fake_kfd_free(uint64_t va);

/* BACKED allocation: like fake_kfd_alloc (contiguous, no frag holes)
 * but the bus range is real host memory, so mmap-style consumers (the
 * probe CLI preload loopback) can materialize CPU windows with real
 * data.  Returns the fake GPU VA. */
uint64_t fake_kfd_alloc_backed(uint64_t size);
/* Bus address -> backing pointer for backed allocations (NULL for
 * synthetic bus ranges). */
void *fake_kfd_bus_to_ptr(uint64_t bus);
/* First bus address of a backed allocation's range (0 if va unknown). */
uint64_t fake_kfd_backing_bus(uint64_t va);

void fake_kfd_reset(void);

/* Introspection for tests. */
long fake_kfd_live_pins(void);
long fake_kfd_get_pages_calls(void);
long fake_kfd_put_pages_calls(void);
long fake_kfd_bad_put_calls(void);	/* put of unknown pin = bridge bug */
long fake_kfd_callbacks_fired(void);

/* Fault injection: make the next N get_page_size calls fail (tests the
 * bridge's documented 2 MiB fallback + warning path). */
void fake_kfd_fail_page_size(int n);

/* Skew the bus base of subsequent fake allocations by `bytes` (may be a
 * non-page multiple): models BAR placements whose sub-page alignment
 * differs from the GPU VA's, which the probe's mmap must refuse to map
 * rather than truncate.  0 restores natural (2 MiB) alignment. */
void fake_kfd_bus_skew(unsigned int bytes);

/* Drift simulation: when on, get_pages with dma_dev==NULL
 * fails with -EINVAL — modeling a ROCK KFD that dma-maps against the
 * passed device inside get_pages and refuses a NULL one.  When a device
 * IS passed, the sg table's addresses get that device's iova_offset
 * applied (the "KFD mapped it internally" shape).  Only meaningful with
 * ROCNR_AMD_RDMA_HAS_DMA_DEV=1 (the default shim build). */
void fake_kfd_reject_null_dev(int on);

/* Fault injection: the next N successful get_pages calls fire the pin's
 * free_callback SYNCHRONOUSLY before returning (buffer freed while the
 * pin was being installed), then reclaim the pin — the returned
 * amd_p2p_info is already dead and must not be dereferenced.  Tests the
 * bridge's lock-free PINNING install (a callback that re-takes the
 * registration lock from inside get_pages must not deadlock). */
void fake_kfd_revoke_in_get_pages(int n);

#ifdef __cplusplus
}
#endif

#endif /* ROCNR_FAKE_KFD_H_ */
