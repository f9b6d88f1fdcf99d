// SPDX-License-Identifier: MIT
// C ABI of the gfx950 payload kernels (implementation: p2p_kernels.hip).
#pragma once
#include <hip/hip_runtime.h>
#include <cstdint>

extern "C" {

// One-time: upload CRC32 slice-by-8 tables + GF(2) shift matrices to
// device constant memory.  Idempotent, cheap.
hipError_t rocp2p_crc32_init();

// buf[i] = splitmix64(seed + (i+1)*GOLDEN) for each 8-byte word.
// nbytes % 8 == 0.  HBM-streaming (16 B/lane vector stores).
hipError_t rocp2p_fill(void* buf, uint64_t nbytes, uint64_t seed,
                       hipStream_t stream);

// Recompute the fill pattern and count mismatching words into
// *d_mismatch (device u64, caller zeroes it).
hipError_t rocp2p_verify(const void* buf, uint64_t nbytes, uint64_t seed,
                         unsigned long long* d_mismatch, hipStream_t stream);

// d_out[p] = zlib-compatible CRC32 of 4 KiB page p.  buf 4 KiB-aligned
// length (npages * 4096 bytes).  One wave per page; per-lane 64 B
// segment CRCs combined via GF(2) shift matrices + wave XOR reduce;
// lookup tables staged in LDS.
hipError_t rocp2p_crc32_pages(const void* buf, uint64_t npages,
                              uint32_t* d_out, hipStream_t stream);

// Streaming device copy (bandwidth ceiling probe), 16 B/lane.
hipError_t rocp2p_copy(void* dst, const void* src, uint64_t nbytes,
                       hipStream_t stream);
// Nontemporal variant (streamed-once data; see microarch nt rows).
hipError_t rocp2p_copy_nt(void* dst, const void* src, uint64_t nbytes,
                          hipStream_t stream);

// GPU-driven batched message engine (the NIC-WQE analog: one launch
// retires a whole queue of posted messages).  src_addrs/dst_offs are
// device-visible u64 arrays of n entries; every message is msg_bytes
// (16-byte multiple).  gather: host-pinned sources -> HBM region.
// scatter: HBM region -> host-pinned destinations.
hipError_t rocp2p_gather(void* dst_base, const uint64_t* d_dst_offs,
                         const uint64_t* d_src_addrs, uint64_t msg_bytes,
                         uint32_t n, hipStream_t stream);
hipError_t rocp2p_scatter(const void* src_base, const uint64_t* d_src_offs,
                          const uint64_t* d_dst_addrs, uint64_t msg_bytes,
                          uint32_t n, hipStream_t stream);

}  // extern "C"
