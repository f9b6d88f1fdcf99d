/* SPDX-License-Identifier: GPL-2.0 OR MIT
 *
 * peer_glue.c — see peer_glue.h.
 */
#define __ROCNR_SHIM__ 1
#include "peer_glue.h"

#include "rocnr_shim_all.h"
#include "fake_ibcore.h"
#include "fake_kfd.h"

static int glue_ready;
/* One fake "HCA" device for the whole process; iova_offset 0 so the
 * bridge's dma_map_resource output addresses stay identity-mappable
 * to the backed bus memory. */
static struct device glue_hca = { .name = "fakehca0", .iova_offset = 0,
				  .fail_after = -1 };

int rocnr_glue_init(void)
{
	int ret;

	if (glue_ready)
		return 0;
	ret = rocnr_shim_module_init();	/* bridge init + ib registration */
	if (ret)
		return ret;
	if (!fake_ib_client())
		return -ENODEV;
	glue_ready = 1;
	return 0;
}

static uint64_t glue_last_va;

uint64_t rocnr_glue_alloc(uint64_t bytes)
{
	uint64_t va = fake_kfd_alloc_backed(bytes);

	if (va)
		glue_last_va = va;
	return va;
}

void rocnr_glue_revoke_last(void)
{
	if (glue_last_va)
		fake_kfd_free(glue_last_va);
	glue_last_va = 0;
}

void *rocnr_glue_vram_ptr(uint64_t va)
{
	uint64_t bus = fake_kfd_backing_bus(va);

	return bus ? fake_kfd_bus_to_ptr(bus) : 0;
}

void rocnr_glue_free(uint64_t va)
{
	fake_kfd_free(va);
}

int rocnr_glue_is_gpu(uint64_t va)
{
	return fake_kfd_backing_bus(va) != 0;
}

int rocnr_glue_reg_mr(uint64_t va, size_t size, void **handle_out,
		      struct rocnr_glue_seg *segs, int *nsegs_inout)
{
	struct fake_ib_mr *mr = NULL;
	struct scatterlist *sg;
	int ret, i, cap;

	if (!glue_ready || !handle_out || !nsegs_inout)
		return -EINVAL;
	ret = fake_ib_reg_mr((unsigned long)va, size, &glue_hca, &mr);
	if (ret)
		return ret;
	cap = *nsegs_inout;
	if (mr->nmap > cap) {
		fake_ib_dereg_mr(mr);
		free(mr);
		return -E2BIG;
	}
	for_each_sg(mr->sgt.sgl, sg, mr->sgt.nents, i) {
		segs[i].bus = sg_dma_address(sg);
		segs[i].len = sg_dma_len(sg);
	}
	*nsegs_inout = mr->nmap;
	*handle_out = mr;
	return 0;
}

int rocnr_glue_mr_dead(void *handle)
{
	struct fake_ib_mr *mr = handle;

	return mr ? mr->dead : 1;
}

int rocnr_glue_dereg_mr(void *handle)
{
	struct fake_ib_mr *mr = handle;
	int ret;

	if (!mr)
		return -EINVAL;
	ret = fake_ib_dereg_mr(mr);
	free(mr);
	return ret;
}

void *rocnr_glue_bus_ptr(uint64_t bus)
{
	return fake_kfd_bus_to_ptr(bus);
}

long rocnr_glue_live_pins(void)
{
	return fake_kfd_live_pins();
}
