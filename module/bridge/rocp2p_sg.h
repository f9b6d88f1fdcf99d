/* SPDX-License-Identifier: GPL-2.0 OR MIT
 *
 * rocp2p_sg.h — bus-address segment coalescing for huge HBM3E pins.
 *
 * MI355X exposes 288 GB of HBM through a full-VRAM BAR and KFD pins VRAM
 * at 2 MB granularity; a 64 GB pin arrives as up to 32768 segments that
 * are mostly bus-contiguous.  The reference struct-copied KFD's table
 * verbatim (reference: /root/reference/amdp2p.c:258-261), handing the HCA
 * one sg entry per GPU page.  We instead coalesce adjacent bus ranges —
 * bounded by the target device's max segment size — so the MR's
 * translation table stays small and MTT cache pressure on the HCA stays
 * low.  Pure logic, no kernel dependencies: unit-tested in userspace.
 */
#ifndef ROCP2P_SG_H_
#define ROCP2P_SG_H_

#ifdef __ROCNR_SHIM__
#include <stdint.h>
#include <stddef.h>
typedef uint64_t rocnr_u64;
#else
#include <linux/types.h>
typedef u64 rocnr_u64;
#endif

struct rocnr_seg {
	rocnr_u64 addr;		/* bus address */
	rocnr_u64 len;		/* bytes */
};

/* Iterator-based input so callers can feed a scatterlist without
 * materializing an array: next() fills *seg, returns 0 at end. */
struct rocnr_seg_iter {
	int (*next)(struct rocnr_seg_iter *it, struct rocnr_seg *seg);
	void *priv;
};

/* Count output segments after coalescing, splitting any run longer than
 * max_seg (0 = unbounded).  Deterministic: same answer as rocnr_coalesce
 * would produce. */
size_t rocnr_coalesce_count(struct rocnr_seg_iter *it, rocnr_u64 max_seg);

/* Emit coalesced segments through emit(ctx, seg); returns the number
 * emitted, or (size_t)-1 if emit returned nonzero (abort). */
size_t rocnr_coalesce(struct rocnr_seg_iter *it, rocnr_u64 max_seg,
		      int (*emit)(void *ctx, const struct rocnr_seg *seg),
		      void *ctx);

#endif /* ROCP2P_SG_H_ */
