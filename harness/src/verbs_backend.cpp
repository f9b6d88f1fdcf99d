// SPDX-License-Identifier: MIT
// IB-verbs backend — the REAL PeerDirect data path: ibv_reg_mr on a
// hipMalloc pointer (dispatched to the rocp2p bridge by the IB core's
// peer-memory probe) and one-sided RDMA WRITE/READ, perftest-style.
//
// Modes:
//  - loopback (default): an RC QP pair inside one process (BASELINE
//    configs 1-3);
//  - client/server: two endpoints exchange QP/MR parameters over the
//    TCP OOB bootstrap (rocp2p_oob.h — the C++ twin of
//    rocnrdma_amd/transport/oob.py) and the client performs one-sided
//    ops against the server's registered region, with REMOTE
//    verification over the control socket (ib_write_bw server/client
//    shape; works across two hosts on HCA-equipped machines).
//
// MR modes: peer (ibv_reg_mr on hipMalloc VA — requires rocp2p.ko, the
// product under test), dmabuf (hipMemGetHandleForAddressRange +
// ibv_reg_dmabuf_mr — module-free cross-check), host.
//
// This pool ships no rdma-core, so everything is compile-gated on
// <infiniband/verbs.h>; the fake-verbs CI layer (harness/fakeverbs/)
// compiles and executes THIS file's loopback and client/server logic
// in-process, including the OOB exchange and remote QP bring-up
// (tests/test_native_harness.py).  First run on a real HCA is a
// bring-up step (docs/RUNBOOK.md).
#include <dlfcn.h>

#include <cstring>
#include <stdexcept>
#include <string>

#include "rocp2p_transport.h"

namespace rocp2p {

bool verbs_runtime_available() {
#ifdef ROCNR_FAKE_VERBS
  return true;  // the fake layer is linked in
#else
  void* h = dlopen("libibverbs.so.1", RTLD_NOW | RTLD_LOCAL);
  if (!h) return false;
  using get_list_t = void** (*)(int*);
  using free_list_t = void (*)(void**);
  auto get_list = (get_list_t)dlsym(h, "ibv_get_device_list");
  auto free_list = (free_list_t)dlsym(h, "ibv_free_device_list");
  bool ok = false;
  if (get_list) {
    int n = 0;
    void** devs = get_list(&n);
    ok = devs && n > 0;
    if (devs && free_list) free_list(devs);
  }
  dlclose(h);
  return ok;
#endif
}

}  // namespace rocp2p

#if defined(__has_include)
#if __has_include(<infiniband/verbs.h>)
#define ROCP2P_HAVE_VERBS 1
#endif
#endif

#ifdef ROCP2P_HAVE_VERBS
#include <infiniband/verbs.h>
#include <hip/hip_runtime.h>
#include <sys/mman.h>
#include <unistd.h>

#include <algorithm>
#include <cstdio>
#include <vector>

#include "rocp2p_oob.h"
#include "../../rocnrdma_amd/ops/csrc/p2p_kernels.h"
#include "../../rocnrdma_amd/ops/csrc/p2p_pattern.h"
#ifdef ROCNR_FAKEVERBS_PEER
#include "../../module/shim/peer_glue.h"
#endif

namespace rocp2p {

#define VB_THROW(cond, msg)                                              \
  do {                                                                   \
    if (!(cond)) throw std::runtime_error(std::string("verbs: ") + msg); \
  } while (0)

namespace {

// ---- small shared pieces -------------------------------------------

struct PeerInfo {
  uint32_t qpn = 0;
  uint16_t lid = 0;
  uint8_t gid[16] = {0};
  int mtu = IBV_MTU_4096;
};

std::string gid_hex(const uint8_t* g) {
  char buf[33];
  for (int i = 0; i < 16; i++) sprintf(buf + 2 * i, "%02x", g[i]);
  return std::string(buf, 32);
}

void gid_unhex(const std::string& s, uint8_t* g) {
  VB_THROW(s.size() == 32, "bad gid encoding");
  for (int i = 0; i < 16; i++)
    g[i] = (uint8_t)strtoul(s.substr(2 * i, 2).c_str(), nullptr, 16);
}

ibv_context* open_first_device() {
  int ndev = 0;
  ibv_device** devs = ibv_get_device_list(&ndev);
  VB_THROW(devs && ndev > 0, "no IB devices");
  ibv_context* ctx = ibv_open_device(devs[0]);
  ibv_free_device_list(devs);
  VB_THROW(ctx, "ibv_open_device failed");
  return ctx;
}

ibv_qp* make_rc_qp(ibv_pd* pd, ibv_cq* cq, size_t inflight) {
  ibv_qp_init_attr a;
  memset(&a, 0, sizeof(a));
  a.send_cq = cq;
  a.recv_cq = cq;
  a.qp_type = IBV_QPT_RC;
  a.cap.max_send_wr = (uint32_t)(2 * inflight + 8);
  a.cap.max_recv_wr = 16;
  a.cap.max_send_sge = 1;
  a.cap.max_recv_sge = 1;
  ibv_qp* qp = ibv_create_qp(pd, &a);
  VB_THROW(qp, "ibv_create_qp failed");
  return qp;
}

PeerInfo local_info(ibv_context* ctx, ibv_qp* qp, int gid_index) {
  ibv_port_attr pattr;
  VB_THROW(ibv_query_port(ctx, 1, &pattr) == 0, "ibv_query_port");
  PeerInfo pi;
  pi.qpn = qp->qp_num;
  pi.lid = pattr.lid;
  pi.mtu = (int)pattr.active_mtu;
  if (pattr.link_layer == IBV_LINK_LAYER_ETHERNET) {
    union ibv_gid gid;
    VB_THROW(ibv_query_gid(ctx, 1, gid_index, &gid) == 0, "ibv_query_gid");
    memcpy(pi.gid, gid.raw, 16);
  }
  return pi;
}

// Bring an RC QP to RTS against `peer` (mlx5-grade mask sets — the
// fake-verbs layer validates them strictly).
void qp_to_rts(ibv_context* ctx, ibv_qp* qp, const PeerInfo& peer,
               int gid_index) {
  ibv_port_attr pattr;
  VB_THROW(ibv_query_port(ctx, 1, &pattr) == 0, "ibv_query_port");

  ibv_qp_attr at;
  memset(&at, 0, sizeof(at));
  at.qp_state = IBV_QPS_INIT;
  at.pkey_index = 0;
  at.port_num = 1;
  at.qp_access_flags = IBV_ACCESS_LOCAL_WRITE | IBV_ACCESS_REMOTE_WRITE |
                       IBV_ACCESS_REMOTE_READ;
  VB_THROW(ibv_modify_qp(qp, &at,
                         IBV_QP_STATE | IBV_QP_PKEY_INDEX | IBV_QP_PORT |
                             IBV_QP_ACCESS_FLAGS) == 0,
           "modify->INIT");

  memset(&at, 0, sizeof(at));
  at.qp_state = IBV_QPS_RTR;
  at.path_mtu = (enum ibv_mtu)peer.mtu;
  at.dest_qp_num = peer.qpn;
  at.rq_psn = 0;
  at.max_dest_rd_atomic = 4;
  at.min_rnr_timer = 12;
  at.ah_attr.port_num = 1;
  if (pattr.link_layer == IBV_LINK_LAYER_ETHERNET) {
    at.ah_attr.is_global = 1;
    memcpy(at.ah_attr.grh.dgid.raw, peer.gid, 16);
    at.ah_attr.grh.sgid_index = (uint8_t)gid_index;
    at.ah_attr.grh.hop_limit = 64;
  } else {
    at.ah_attr.is_global = 0;
    at.ah_attr.dlid = peer.lid;
  }
  VB_THROW(ibv_modify_qp(qp, &at,
                         IBV_QP_STATE | IBV_QP_AV | IBV_QP_PATH_MTU |
                             IBV_QP_DEST_QPN | IBV_QP_RQ_PSN |
                             IBV_QP_MAX_DEST_RD_ATOMIC |
                             IBV_QP_MIN_RNR_TIMER) == 0,
           "modify->RTR");

  memset(&at, 0, sizeof(at));
  at.qp_state = IBV_QPS_RTS;
  at.sq_psn = 0;
  at.timeout = 14;
  at.retry_cnt = 7;
  at.rnr_retry = 7;
  at.max_rd_atomic = 4;
  VB_THROW(ibv_modify_qp(qp, &at,
                         IBV_QP_STATE | IBV_QP_SQ_PSN | IBV_QP_TIMEOUT |
                             IBV_QP_RETRY_CNT | IBV_QP_RNR_RETRY |
                             IBV_QP_MAX_QP_RD_ATOMIC) == 0,
           "modify->RTS");
}

// A registered region: host malloc, GPU peer (ibv_reg_mr on hipMalloc
// VA — the bridge path) or GPU dmabuf.
struct Region {
  uint8_t* host = nullptr;
  uint8_t* gpu = nullptr;
  uint8_t* ptr = nullptr;
  ibv_mr* mr = nullptr;
  size_t bytes = 0;
  bool host_is_map = false;  // host came from mmap (memfd dmabuf stand-in)
  uint64_t peer_va = 0;      // full-stack build: glue-owned fake VRAM
  std::string mode;

  void create(ibv_pd* pd, size_t nbytes, std::string mr_mode,
              int device_index) {
    bytes = nbytes;
    mode = mr_mode;
    if (mode == "auto") mode = hip_available() ? "peer" : "host";
    int acc = IBV_ACCESS_LOCAL_WRITE | IBV_ACCESS_REMOTE_WRITE |
              IBV_ACCESS_REMOTE_READ;
    if (mode == "host") {
      host = (uint8_t*)aligned_alloc(4096, nbytes);
      VB_THROW(host, "region alloc failed");
      memset(host, 0, nbytes);
      ptr = host;
      mr = ibv_reg_mr(pd, ptr, nbytes, acc);
      VB_THROW(mr, "ibv_reg_mr(region) failed");
      return;
    }
    if (mode == "dmabuf") {
      int fd = -1;
      if (hip_available()) {
        // VRAM dmabuf: the exporter's pages ARE the HBM region (BAR
        // window); module-free cross-check of the peer-MR path.
        VB_THROW(hipSetDevice(device_index) == hipSuccess, "hipSetDevice");
        VB_THROW(hipMalloc((void**)&gpu, nbytes) == hipSuccess,
                 "hipMalloc region");
        ptr = gpu;
        VB_THROW(hipMemGetHandleForAddressRange(
                     &fd, gpu, nbytes, hipMemRangeHandleTypeDmaBufFd, 0) ==
                     hipSuccess,
                 "hipMemGetHandleForAddressRange(dmabuf) failed");
      } else {
        // CPU CI: a memfd plays the exporter so the ibv_reg_dmabuf_mr
        // code path (fd handoff, iova addressing, MR lifetime) still
        // executes end-to-end.
        fd = (int)memfd_create("rocp2p_dmabuf", 0);
        VB_THROW(fd >= 0, "memfd_create failed");
        VB_THROW(ftruncate(fd, (off_t)nbytes) == 0, "ftruncate failed");
        host = (uint8_t*)mmap(nullptr, nbytes, PROT_READ | PROT_WRITE,
                              MAP_SHARED, fd, 0);
        VB_THROW(host != MAP_FAILED, "mmap(memfd) failed");
        host_is_map = true;
        ptr = host;
      }
      mr = ibv_reg_dmabuf_mr(pd, 0, nbytes, (uint64_t)ptr, fd, acc);
      close(fd);  // the MR holds its own reference to the dmabuf
      VB_THROW(mr, "ibv_reg_dmabuf_mr failed");
      return;
    }
    // peer mode: dispatched to the rocp2p bridge by the IB core
#ifdef ROCNR_FAKEVERBS_PEER
    if (!hip_available()) {
      // full-stack CI build: "VRAM" is a backed fake-KFD allocation;
      // ibv_reg_mr(va) below dispatches through the REAL bridge
      VB_THROW(rocnr_glue_init() == 0, "peer glue init");
      peer_va = rocnr_glue_alloc(nbytes);
      VB_THROW(peer_va, "glue VRAM alloc failed");
      ptr = (uint8_t*)peer_va;
      host = (uint8_t*)rocnr_glue_vram_ptr(peer_va);  // CPU view of VRAM
      VB_THROW(host, "glue VRAM ptr");
      mr = ibv_reg_mr(pd, ptr, nbytes, acc);
      VB_THROW(mr, "ibv_reg_mr(GPU VA) failed — peer client not dispatched?");
      return;
    }
#endif
    VB_THROW(hipSetDevice(device_index) == hipSuccess, "hipSetDevice");
    VB_THROW(hipMalloc((void**)&gpu, nbytes) == hipSuccess,
             "hipMalloc region");
    ptr = gpu;
    mr = ibv_reg_mr(pd, ptr, nbytes, acc);
    VB_THROW(mr, "ibv_reg_mr(GPU VA) failed — rocp2p bridge loaded?");
  }

  uint64_t verify(uint64_t seed) {
    if (gpu) {
      unsigned long long* d_bad = nullptr;
      VB_THROW(hipMalloc((void**)&d_bad, 8) == hipSuccess, "hipMalloc");
      (void)hipMemset(d_bad, 0, 8);
      VB_THROW(rocp2p_verify(gpu, bytes, seed, d_bad, 0) == hipSuccess,
               "verify kernel");
      unsigned long long h_bad = 0;
      (void)hipMemcpy(&h_bad, d_bad, 8, hipMemcpyDeviceToHost);
      (void)hipFree(d_bad);
      return h_bad;
    }
    const uint64_t* r = reinterpret_cast<const uint64_t*>(host);
    uint64_t bad = 0;
    for (size_t w = 0; w < bytes / 8; w++)
      bad += (r[w] != rocp2p_pattern_word(seed, w));
    return bad;
  }

  void fill(uint64_t seed) {
    if (gpu) {
      VB_THROW(rocp2p_fill(gpu, bytes, seed, 0) == hipSuccess,
               "fill kernel");
      (void)hipDeviceSynchronize();
      return;
    }
    uint64_t* r = reinterpret_cast<uint64_t*>(host);
    for (size_t w = 0; w < bytes / 8; w++)
      r[w] = rocp2p_pattern_word(seed, w);
  }

  void destroy() {
    if (mr) ibv_dereg_mr(mr);
    if (host) {
      if (peer_va) { /* backing owned by the glue */ }
      else if (host_is_map) munmap(host, bytes);
      else free(host);
    }
#ifdef ROCNR_FAKEVERBS_PEER
    if (peer_va) rocnr_glue_free(peer_va);
#endif
    if (gpu) (void)hipFree(gpu);
    mr = nullptr;
    host = gpu = ptr = nullptr;
    host_is_map = false;
    peer_va = 0;
  }
};

int env_gid_index() {
  const char* e = getenv("ROCP2P_GID_INDEX");
  return e ? atoi(e) : 1;
}

}  // namespace

// ---- loopback + client transports ----------------------------------

class VerbsTransport final : public Transport {
 public:
  // mode: loopback (oob == nullptr) or client (remote target via oob)
  VerbsTransport(const TransportConfig& cfg, OobSocket* oob)
      : Transport(cfg), oob_(oob) {
    if (cfg.region_bytes % cfg.msg_bytes)
      throw std::runtime_error("region must be a multiple of msg size");
    inflight_ = cfg.inflight ? cfg.inflight : 64;
    if (inflight_ > msgs_per_region()) inflight_ = msgs_per_region();
    chain_ = cfg.chain ? cfg.chain : std::min<size_t>(16, inflight_);
    if (chain_ > inflight_) chain_ = inflight_;
    wrs_.resize(chain_);
    sges_.resize(chain_);

    ctx_ = open_first_device();
    pd_ = ibv_alloc_pd(ctx_);
    VB_THROW(pd_, "ibv_alloc_pd failed");
    cq_ = ibv_create_cq(ctx_, 2 * (int)inflight_ + 16, nullptr, nullptr, 0);
    VB_THROW(cq_, "ibv_create_cq failed");

    // staging: host-pinned (hip) so the same buffers serve GPU paths
    bool use_hip = hip_available();
    if (use_hip) {
      VB_THROW(hipSetDevice(cfg.device_index) == hipSuccess, "hipSetDevice");
      VB_THROW(hipHostMalloc((void**)&staging_,
                             inflight_ * cfg.msg_bytes, 0) == hipSuccess,
               "hipHostMalloc staging");
      staging_hip_ = true;
    } else {
      staging_ = (uint8_t*)aligned_alloc(4096, inflight_ * cfg.msg_bytes);
      VB_THROW(staging_, "staging alloc failed");
    }
    staging_mr_ = ibv_reg_mr(pd_, staging_, inflight_ * cfg.msg_bytes,
                             IBV_ACCESS_LOCAL_WRITE);
    VB_THROW(staging_mr_, "ibv_reg_mr(staging) failed");

    qp_ = make_rc_qp(pd_, cq_, inflight_);

    if (!oob_) {
      // loopback: local region + a second QP as the passive end
      region_.create(pd_, cfg.region_bytes, cfg.verbs_mr, cfg.device_index);
      remote_addr_ = (uint64_t)region_.ptr;
      remote_rkey_ = region_.mr->rkey;
      qp_peer_ = make_rc_qp(pd_, cq_, inflight_);
      PeerInfo a = local_info(ctx_, qp_, env_gid_index());
      PeerInfo b = local_info(ctx_, qp_peer_, env_gid_index());
      qp_to_rts(ctx_, qp_, b, env_gid_index());
      qp_to_rts(ctx_, qp_peer_, a, env_gid_index());
      return;
    }

    // client: exchange with the remote target
    PeerInfo mine = local_info(ctx_, qp_, env_gid_index());
    oob_->send_kv({{"qpn", std::to_string(mine.qpn)},
                   {"lid", std::to_string(mine.lid)},
                   {"gid", gid_hex(mine.gid)},
                   {"mtu", std::to_string(mine.mtu)},
                   {"region", std::to_string(cfg.region_bytes)}});
    KvMap srv = oob_->recv_kv();
    PeerInfo peer;
    peer.qpn = (uint32_t)strtoul(srv.at("qpn").c_str(), nullptr, 10);
    peer.lid = (uint16_t)strtoul(srv.at("lid").c_str(), nullptr, 10);
    gid_unhex(srv.at("gid"), peer.gid);
    peer.mtu = atoi(srv.at("mtu").c_str());
    remote_addr_ = strtoull(srv.at("raddr").c_str(), nullptr, 10);
    remote_rkey_ = (uint32_t)strtoul(srv.at("rkey").c_str(), nullptr, 10);
    uint64_t rbytes = strtoull(srv.at("rbytes").c_str(), nullptr, 10);
    VB_THROW(rbytes >= cfg.region_bytes, "target region too small");
    qp_to_rts(ctx_, qp_, peer, env_gid_index());
  }

  ~VerbsTransport() override {
    if (qp_) ibv_destroy_qp(qp_);
    if (qp_peer_) ibv_destroy_qp(qp_peer_);
    region_.destroy();
    if (staging_mr_) ibv_dereg_mr(staging_mr_);
    if (cq_) ibv_destroy_cq(cq_);
    if (pd_) ibv_dealloc_pd(pd_);
    if (ctx_) ibv_close_device(ctx_);
    if (staging_) {
      if (staging_hip_) (void)hipHostFree(staging_);
      else free(staging_);
    }
    delete oob_;
  }

  const char* name() const override {
    return oob_ ? "verbs-client" : "verbs";
  }

  // WR chaining + selective signaling (the mlx5 doorbell-batching
  // shape): one ibv_post_send call rings ONE doorbell for a chain of
  // up to chain_ WRs, and only the chain's last WR is SIGNALED — its
  // wr_id carries the chain length so drain() retires the whole chain
  // per completion.  Posting one SIGNALED WR per call (round-1 shape)
  // would cap a real HCA's 4 KiB message rate at the per-doorbell +
  // per-CQE cost; see docs/PERF.md "posting rate".
  void post_many(uint64_t start, uint64_t n) override {
    uint64_t i = start;
    while (i < start + n) {
      if (outstanding_ >= inflight_) drain(1);
      uint64_t room = (uint64_t)(inflight_ - outstanding_);
      uint64_t chain = std::min({start + n - i, room, (uint64_t)chain_});
      for (uint64_t k = 0; k < chain; k++) {
        uint64_t m = i + k;
        sges_[k].addr =
            (uint64_t)(staging_ + (m % inflight_) * cfg_.msg_bytes);
        sges_[k].length = (uint32_t)cfg_.msg_bytes;
        sges_[k].lkey = staging_mr_->lkey;
        memset(&wrs_[k], 0, sizeof(ibv_send_wr));
        wrs_[k].wr_id = 0;  // unsignaled: retired by the chain tail
        wrs_[k].next = k + 1 < chain ? &wrs_[k + 1] : nullptr;
        wrs_[k].sg_list = &sges_[k];
        wrs_[k].num_sge = 1;
        wrs_[k].opcode = cfg_.dir == Direction::Write ? IBV_WR_RDMA_WRITE
                                                      : IBV_WR_RDMA_READ;
        wrs_[k].send_flags = 0;
        wrs_[k].wr.rdma.remote_addr =
            remote_addr_ + (m % msgs_per_region()) * cfg_.msg_bytes;
        wrs_[k].wr.rdma.rkey = remote_rkey_;
      }
      wrs_[chain - 1].send_flags = IBV_SEND_SIGNALED;
      wrs_[chain - 1].wr_id = chain;
      ibv_send_wr* bad = nullptr;
      VB_THROW(ibv_post_send(qp_, &wrs_[0], &bad) == 0, "ibv_post_send");
      outstanding_ += chain;
      i += chain;
    }
  }

  void flush() override { drain(outstanding_); }

  uint64_t integrity_check(uint64_t seed) override {
    const size_t words_per_msg = cfg_.msg_bytes / 8;
    uint64_t bad = 0;
    if (cfg_.dir == Direction::Write) {
      for (size_t base = 0; base < msgs_per_region(); base += inflight_) {
        size_t burst = std::min(inflight_, msgs_per_region() - base);
        for (size_t m = base; m < base + burst; m++) {
          uint64_t* slot = reinterpret_cast<uint64_t*>(
              staging_ + (m % inflight_) * cfg_.msg_bytes);
          for (size_t w = 0; w < words_per_msg; w++)
            slot[w] = rocp2p_pattern_word(seed, m * words_per_msg + w);
        }
        post_many(base, burst);
        flush();
      }
      if (oob_) {
        // REMOTE verification: the target audits its own region
        oob_->send_kv({{"op", "verify"}, {"seed", std::to_string(seed)}});
        return strtoull(oob_->recv_kv().at("bad").c_str(), nullptr, 10);
      }
      return region_.verify(seed);
    }
    // read direction
    if (oob_) {
      oob_->send_kv({{"op", "fill"}, {"seed", std::to_string(seed)}});
      oob_->recv_kv();
    } else {
      region_.fill(seed);
    }
    for (size_t base = 0; base < msgs_per_region(); base += inflight_) {
      size_t burst = std::min(inflight_, msgs_per_region() - base);
      post_many(base, burst);
      flush();
      for (size_t m = base; m < base + burst; m++) {
        const uint64_t* slot = reinterpret_cast<const uint64_t*>(
            staging_ + (m % inflight_) * cfg_.msg_bytes);
        for (size_t w = 0; w < words_per_msg; w++)
          bad += (slot[w] != rocp2p_pattern_word(seed, m * words_per_msg + w));
      }
    }
    return bad;
  }

 private:
  // at_least counts WRs; each completion retires wr_id WRs (its chain).
  void drain(size_t at_least) {
    ibv_wc wc[16];
    size_t done = 0;
    while (done < at_least && outstanding_ > 0) {
      int n = ibv_poll_cq(cq_, 16, wc);
      VB_THROW(n >= 0, "ibv_poll_cq");
      for (int i = 0; i < n; i++) {
        VB_THROW(wc[i].status == IBV_WC_SUCCESS,
                 std::string("completion error: ") +
                     ibv_wc_status_str(wc[i].status));
        size_t retired = (size_t)wc[i].wr_id;
        VB_THROW(retired >= 1 && retired <= outstanding_,
                 "completion retires more WRs than outstanding");
        done += retired;
        outstanding_ -= retired;
      }
    }
  }

  OobSocket* oob_ = nullptr;
  ibv_context* ctx_ = nullptr;
  ibv_pd* pd_ = nullptr;
  ibv_cq* cq_ = nullptr;
  ibv_qp* qp_ = nullptr;
  ibv_qp* qp_peer_ = nullptr;  // loopback passive end
  ibv_mr* staging_mr_ = nullptr;
  uint8_t* staging_ = nullptr;
  bool staging_hip_ = false;
  Region region_;  // loopback only
  uint64_t remote_addr_ = 0;
  uint32_t remote_rkey_ = 0;
  size_t outstanding_ = 0;	/* WRs in the SQ (signaled or not) */
  size_t chain_ = 1;		/* WRs per doorbell */
  std::vector<ibv_send_wr> wrs_;
  std::vector<ibv_sge> sges_;
};

std::unique_ptr<Transport> make_verbs_transport(const TransportConfig& cfg) {
  return std::make_unique<VerbsTransport>(cfg, nullptr);
}

std::unique_ptr<Transport> make_verbs_client(const TransportConfig& cfg,
                                             const std::string& host,
                                             int port) {
  return std::make_unique<VerbsTransport>(cfg, oob_connect(host, port));
}

// Passive target: register the region, exchange QP/MR params, answer
// verify/fill ops until "bye".  announce(port) fires once listening.
int run_verbs_target(const TransportConfig& cfg, int port,
                     void (*announce)(int)) {
  ibv_context* ctx = open_first_device();
  ibv_pd* pd = ibv_alloc_pd(ctx);
  VB_THROW(pd, "ibv_alloc_pd failed");
  ibv_cq* cq = ibv_create_cq(ctx, 64, nullptr, nullptr, 0);
  VB_THROW(cq, "ibv_create_cq failed");
  Region region;
  region.create(pd, cfg.region_bytes, cfg.verbs_mr, cfg.device_index);
  ibv_qp* qp = make_rc_qp(pd, cq, 64);

  OobServer server(port);
  if (announce) announce(server.port());
  OobSocket* sock = server.accept_one();
  KvMap cli = sock->recv_kv();
  PeerInfo peer;
  peer.qpn = (uint32_t)strtoul(cli.at("qpn").c_str(), nullptr, 10);
  peer.lid = (uint16_t)strtoul(cli.at("lid").c_str(), nullptr, 10);
  gid_unhex(cli.at("gid"), peer.gid);
  peer.mtu = atoi(cli.at("mtu").c_str());

  PeerInfo mine = local_info(ctx, qp, env_gid_index());
  sock->send_kv({{"qpn", std::to_string(mine.qpn)},
                 {"lid", std::to_string(mine.lid)},
                 {"gid", gid_hex(mine.gid)},
                 {"mtu", std::to_string(mine.mtu)},
                 {"raddr", std::to_string((uint64_t)region.ptr)},
                 {"rkey", std::to_string(region.mr->rkey)},
                 {"rbytes", std::to_string(region.bytes)}});
  qp_to_rts(ctx, qp, peer, env_gid_index());

  // control loop: the data plane never touches this thread
  try {
    for (;;) {
      KvMap msg = sock->recv_kv();
      auto op = msg.count("op") ? msg.at("op") : "bye";
      if (op == "verify") {
        uint64_t seed = strtoull(msg.at("seed").c_str(), nullptr, 10);
        sock->send_kv({{"bad", std::to_string(region.verify(seed))}});
      } else if (op == "fill") {
        uint64_t seed = strtoull(msg.at("seed").c_str(), nullptr, 10);
        region.fill(seed);
        sock->send_kv({{"ok", "1"}});
      } else {
        break;
      }
    }
  } catch (const std::exception&) {
    // client went away: normal shutdown
  }
  delete sock;
  ibv_destroy_qp(qp);
  region.destroy();
  ibv_destroy_cq(cq);
  ibv_dealloc_pd(pd);
  ibv_close_device(ctx);
  return 0;
}

}  // namespace rocp2p

#else  // !ROCP2P_HAVE_VERBS

namespace rocp2p {

std::unique_ptr<Transport> make_verbs_transport(const TransportConfig&) {
  throw std::runtime_error(
      "verbs backend compiled out: <infiniband/verbs.h> (rdma-core) was "
      "not present at build time. Rebuild on an HCA-equipped host.");
}

std::unique_ptr<Transport> make_verbs_client(const TransportConfig&,
                                             const std::string&, int) {
  throw std::runtime_error("verbs backend compiled out (see above)");
}

int run_verbs_target(const TransportConfig&, int, void (*)(int)) {
  throw std::runtime_error("verbs backend compiled out (see above)");
}

}  // namespace rocp2p

#endif
