/* SPDX-License-Identifier: GPL-2.0 OR MIT
 *
 * Segment coalescing (see rocp2p_sg.h).  Streaming one-pass merge:
 * adjacent when prev.addr + prev.len == cur.addr; runs split at max_seg.
 */
#include "rocp2p_sg.h"

struct coalesce_state {
	struct rocnr_seg cur;
	int have;
};

/* Split `seg` into max_seg-bounded pieces, emitting each; returns count
 * or (size_t)-1 on emit abort. */
static size_t emit_split(const struct rocnr_seg *seg, rocnr_u64 max_seg,
			 int (*emit)(void *ctx, const struct rocnr_seg *s),
			 void *ctx)
{
	struct rocnr_seg piece;
	rocnr_u64 off = 0;
	size_t n = 0;

	if (!seg->len)
		return 0;
	if (!max_seg)
		max_seg = (rocnr_u64)-1;
	while (off < seg->len) {
		piece.addr = seg->addr + off;
		piece.len = seg->len - off;
		if (piece.len > max_seg)
			piece.len = max_seg;
		if (emit && emit(ctx, &piece))
			return (size_t)-1;
		off += piece.len;
		n++;
	}
	return n;
}

static size_t do_coalesce(struct rocnr_seg_iter *it, rocnr_u64 max_seg,
			  int (*emit)(void *ctx, const struct rocnr_seg *s),
			  void *ctx)
{
	struct coalesce_state st = { .have = 0 };
	struct rocnr_seg in;
	size_t total = 0, n;

	while (it->next(it, &in)) {
		if (!in.len)
			continue;
		if (st.have && st.cur.addr + st.cur.len == in.addr &&
		    st.cur.len + in.len >= st.cur.len /* overflow guard */) {
			st.cur.len += in.len;
			continue;
		}
		if (st.have) {
			n = emit_split(&st.cur, max_seg, emit, ctx);
			if (n == (size_t)-1)
				return n;
			total += n;
		}
		st.cur = in;
		st.have = 1;
	}
	if (st.have) {
		n = emit_split(&st.cur, max_seg, emit, ctx);
		if (n == (size_t)-1)
			return n;
		total += n;
	}
	return total;
}

size_t rocnr_coalesce_count(struct rocnr_seg_iter *it, rocnr_u64 max_seg)
{
	return do_coalesce(it, max_seg, 0, 0);
}

size_t rocnr_coalesce(struct rocnr_seg_iter *it, rocnr_u64 max_seg,
		      int (*emit)(void *ctx, const struct rocnr_seg *seg),
		      void *ctx)
{
	return do_coalesce(it, max_seg, emit, ctx);
}
