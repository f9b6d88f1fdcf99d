/* SPDX-License-Identifier: GPL-2.0 OR MIT
 *
 * peer_glue.h — minimal extern-C surface that lets the fake-verbs
 * layer dispatch ibv_reg_mr(GPU VA) through the REAL rocp2p bridge
 * (userspace shim build) the way the IB core's peer-memory probe does:
 *
 *   ibv_reg_mr(va)                        [harness / fake_verbs.cpp]
 *     -> rocnr_glue_reg_mr                [this glue]
 *       -> fake IB core fake_ib_reg_mr    [module/shim/fake_ibcore.c]
 *         -> bridge acquire/get_pages/dma_map   [module/bridge/]
 *           -> fake KFD pin (BACKED bus memory) [module/shim/fake_kfd.c]
 *
 * The returned segment list holds device-mapped bus addresses whose
 * bytes are real host memory (fake_kfd backed allocations), so the
 * fake NIC's data plane can move real payload through the registered
 * MR — the reference's complete L5->L0 registration + DMA flow
 *  in one process, with the real bridge code on the
 * hot path.  Plain C types only: the consumer is C++ (hipcc).
 */
#ifndef ROCNR_PEER_GLUE_H_
#define ROCNR_PEER_GLUE_H_

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* Bring up the stack (bridge module init + registration with the fake
 * IB core).  Idempotent; returns 0 on success. */
int rocnr_glue_init(void);

/* Allocate a fake GPU ("VRAM") region with real backing memory.
 * Returns the GPU VA (0 on failure). */
uint64_t rocnr_glue_alloc(uint64_t bytes);
/* CPU pointer to the backing of a glue allocation (the "VRAM" bytes);
 * NULL if va unknown. */
void *rocnr_glue_vram_ptr(uint64_t va);
void rocnr_glue_free(uint64_t va);

/* Is this VA claimable by the peer stack? (mirrors the IB core's
 * probe decision) */
int rocnr_glue_is_gpu(uint64_t va);

struct rocnr_glue_seg {
	uint64_t bus;
	uint64_t len;
};

/* Register [va, va+size) through the full peer path.  On success
 * returns 0, sets *handle_out and fills up to *nsegs_inout entries of
 * segs with the bridge's device-mapped sg table.  Negative errno on
 * failure (including -ENODEV when no peer client claims the VA). */
int rocnr_glue_reg_mr(uint64_t va, size_t size, void **handle_out,
		      struct rocnr_glue_seg *segs, int *nsegs_inout);
/* Deregister (ibv_dereg_mr shape): dma_unmap + put_pages + release
 * through the real bridge. */
int rocnr_glue_dereg_mr(void *handle);

/* Has this MR been torn down by the invalidation path (producer freed
 * the memory under it)?  The verbs layer checks this before touching
 * the MR's (now reclaimed) pages — modeling the remote-access errors
 * a real HCA raises after invalidation. */
int rocnr_glue_mr_dead(void *handle);

/* Revoke the most recent glue allocation (GPU-frees it under any live
 * MRs): drives free_callback -> bridge -> IB-core invalidate.  For the
 * harness revoke selftest. */
void rocnr_glue_revoke_last(void);

/* Bus address -> CPU pointer for the fake NIC's data plane (NULL if
 * the bus range is not backed). */
void *rocnr_glue_bus_ptr(uint64_t bus);

/* Liveness counters for tests (pins still held below the bridge). */
long rocnr_glue_live_pins(void);

#ifdef __cplusplus
}
#endif

#endif /* ROCNR_PEER_GLUE_H_ */
