// SPDX-License-Identifier: MIT
// Torch bindings for the gfx950 payload kernels (p2p_kernels.hip).
// Thin: tensor checks + stream plumbing only; all device logic lives in
// the .hip TU.  Fails loudly — there is deliberately NO CPU fallback
// here, so a GPU test that silently skipped the native path is
// impossible (CPU references for tests live in rocnrdma_amd/utils).
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "p2p_kernels.h"

namespace {

void check_buf(const torch::Tensor& t, const char* who) {
  TORCH_CHECK(t.is_cuda(), who, ": tensor must be on GPU");
  TORCH_CHECK(t.is_contiguous(), who, ": tensor must be contiguous");
}

uint64_t nbytes_of(const torch::Tensor& t) {
  return (uint64_t)t.numel() * t.element_size();
}

#define HIP_OK(expr)                                              \
  do {                                                            \
    hipError_t _e = (expr);                                       \
    TORCH_CHECK(_e == hipSuccess, #expr, " failed: ",             \
                hipGetErrorString(_e));                           \
  } while (0)

void fill_(torch::Tensor buf, int64_t seed) {
  check_buf(buf, "fill_");
  auto stream = at::cuda::getCurrentCUDAStream();
  HIP_OK(rocp2p_fill(buf.data_ptr(), nbytes_of(buf), (uint64_t)seed,
                     stream.stream()));
}

int64_t verify(torch::Tensor buf, int64_t seed) {
  check_buf(buf, "verify");
  auto mis = torch::zeros({1}, torch::dtype(torch::kInt64).device(buf.device()));
  auto stream = at::cuda::getCurrentCUDAStream();
  HIP_OK(rocp2p_verify(buf.data_ptr(), nbytes_of(buf), (uint64_t)seed,
                       (unsigned long long*)mis.data_ptr<int64_t>(),
                       stream.stream()));
  return mis.item<int64_t>();
}

torch::Tensor crc32_pages(torch::Tensor buf) {
  check_buf(buf, "crc32_pages");
  uint64_t nbytes = nbytes_of(buf);
  TORCH_CHECK(nbytes % 4096 == 0, "crc32_pages: length must be 4 KiB pages");
  int64_t npages = (int64_t)(nbytes / 4096);
  auto out =
      torch::empty({npages}, torch::dtype(torch::kInt32).device(buf.device()));
  auto stream = at::cuda::getCurrentCUDAStream();
  HIP_OK(rocp2p_crc32_pages(buf.data_ptr(), (uint64_t)npages,
                            (uint32_t*)out.data_ptr<int32_t>(),
                            stream.stream()));
  return out;
}

void check_u64_dev(const torch::Tensor& t, const char* who) {
  TORCH_CHECK(t.is_cuda() && t.is_contiguous() &&
                  t.scalar_type() == torch::kInt64,
              who, ": descriptor tensor must be contiguous int64 on GPU");
}

void gather_(torch::Tensor region, torch::Tensor dst_offs,
             torch::Tensor src_addrs, int64_t msg_bytes) {
  check_buf(region, "gather_");
  check_u64_dev(dst_offs, "gather_");
  check_u64_dev(src_addrs, "gather_");
  TORCH_CHECK(dst_offs.numel() == src_addrs.numel(), "gather_: n mismatch");
  auto stream = at::cuda::getCurrentCUDAStream();
  HIP_OK(rocp2p_gather(region.data_ptr(),
                       (const uint64_t*)dst_offs.data_ptr<int64_t>(),
                       (const uint64_t*)src_addrs.data_ptr<int64_t>(),
                       (uint64_t)msg_bytes, (uint32_t)dst_offs.numel(),
                       stream.stream()));
}

void scatter_(torch::Tensor region, torch::Tensor src_offs,
              torch::Tensor dst_addrs, int64_t msg_bytes) {
  check_buf(region, "scatter_");
  check_u64_dev(src_offs, "scatter_");
  check_u64_dev(dst_addrs, "scatter_");
  TORCH_CHECK(src_offs.numel() == dst_addrs.numel(), "scatter_: n mismatch");
  auto stream = at::cuda::getCurrentCUDAStream();
  HIP_OK(rocp2p_scatter(region.data_ptr(),
                        (const uint64_t*)src_offs.data_ptr<int64_t>(),
                        (const uint64_t*)dst_addrs.data_ptr<int64_t>(),
                        (uint64_t)msg_bytes, (uint32_t)src_offs.numel(),
                        stream.stream()));
}

void copy_(torch::Tensor dst, torch::Tensor src) {
  check_buf(dst, "copy_");
  check_buf(src, "copy_");
  TORCH_CHECK(nbytes_of(dst) == nbytes_of(src), "copy_: size mismatch");
  auto stream = at::cuda::getCurrentCUDAStream();
  HIP_OK(rocp2p_copy(dst.data_ptr(), src.data_ptr(), nbytes_of(dst),
                     stream.stream()));
}

void copy_nt_(torch::Tensor dst, torch::Tensor src) {
  check_buf(dst, "copy_nt_");
  check_buf(src, "copy_nt_");
  TORCH_CHECK(nbytes_of(dst) == nbytes_of(src), "copy_nt_: size mismatch");
  auto stream = at::cuda::getCurrentCUDAStream();
  HIP_OK(rocp2p_copy_nt(dst.data_ptr(), src.data_ptr(), nbytes_of(dst),
                        stream.stream()));
}

// Export an HBM range as a dmabuf fd — the GPU half of the verbs
// backend's kernel-module-free MR mode (ibv_reg_dmabuf_mr); lets a
// GPU-only box validate that machinery without an HCA.  Caller owns
// the fd (os.close it).
int64_t dmabuf_fd(torch::Tensor buf) {
  check_buf(buf, "dmabuf_fd");
  int fd = -1;
  hipError_t e = hipMemGetHandleForAddressRange(
      &fd, buf.data_ptr(), nbytes_of(buf), hipMemRangeHandleTypeDmaBufFd,
      0);
  TORCH_CHECK(e == hipSuccess, "hipMemGetHandleForAddressRange: ",
              hipGetErrorString(e));
  return fd;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "MI355X payload kernels: splitmix64 fill / verify, per-page "
            "CRC32, streaming copy";
  m.def("fill_", &fill_, "in-place splitmix64 pattern fill");
  m.def("verify", &verify, "count words deviating from the pattern");
  m.def("crc32_pages", &crc32_pages, "zlib CRC32 of each 4 KiB page");
  m.def("copy_", &copy_, "streaming device copy dst <- src");
  m.def("copy_nt_", &copy_nt_, "nontemporal streaming device copy");
  m.def("gather_", &gather_,
        "batched message engine: host-pinned srcs -> HBM region offsets");
  m.def("scatter_", &scatter_,
        "batched message engine: HBM region offsets -> host-pinned dsts");
  m.def("dmabuf_fd", &dmabuf_fd,
        "export an HBM tensor range as a dmabuf fd (caller closes)");
}
