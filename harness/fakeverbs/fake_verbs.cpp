// SPDX-License-Identifier: MIT
// In-process fake libibverbs (see infiniband/verbs.h in this dir).
// Strict where it matters: the QP state machine requires the exact
// attr-mask sets an mlx5 RC QP requires, MR access flags are enforced
// on remote ops, and data movement happens at post_send time with
// bounds checks — so the real verbs backend's connect/post/poll logic
// is genuinely validated without an HCA.
//
// Env knobs:
//   FAKE_VERBS_LINK=ib|eth   link layer reported by query_port
//                            (default ib) — covers both AH paths.
//   FAKE_VERBS_SHM=<name>    cross-process fabric: processes sharing
//                            the name publish their MRs in a shm
//                            registry and run a NIC-role engine thread
//                            that applies one-sided ops addressed to
//                            their rkeys — two REAL processes move
//                            real bytes (RDMA semantics: the remote
//                            application thread is never involved).
#include "infiniband/verbs.h"

#include <fcntl.h>
#include <pthread.h>
#include <sys/mman.h>
#include <unistd.h>

#include <atomic>
#include <cstdlib>
#include <cstring>
#include <cstdio>
#include <deque>
#include <vector>

#ifdef ROCNR_FAKEVERBS_PEER
// Full-stack build: ibv_reg_mr on a fake-GPU VA dispatches through the
// REAL rocp2p bridge (userspace shim) the way the IB core's peer-
// memory probe does — see module/shim/peer_glue.h for the call stack.
#include "../../module/shim/peer_glue.h"
#endif

namespace {

struct FakeCtx;

struct FakeMr {
  ibv_mr mr;
  int access;
  bool live;
  // dmabuf MRs: CPU window mapped from the fd (real BAR pages when the
  // fd came from hipMemGetHandleForAddressRange on VRAM; a memfd in
  // CPU-only CI).  mr.addr holds the iova; data-plane addresses are
  // translated iova -> host_map.
  uint8_t* host_map = nullptr;
  size_t map_len = 0;
#ifdef ROCNR_FAKEVERBS_PEER
  // peer MRs: handle + device-mapped sg table from the real bridge
  void* peer_handle = nullptr;
  std::vector<rocnr_glue_seg> segs;
#endif
};

struct FakeCq {
  std::deque<ibv_wc> completions;
};

struct FakeQp {
  ibv_qp qp;
  FakeCq* send_cq;
  int access_flags = 0;
  uint32_t dest_qp_num = 0;
  bool has_av = false;
};

struct FakeCtx {
  std::vector<FakeMr*> mrs;
  uint32_t next_key = 0x1000;
  uint32_t next_qpn = 0x40;
};

FakeCtx g_ctx;
char g_device_storage;
ibv_device* g_device_list[2] = {
    reinterpret_cast<ibv_device*>(&g_device_storage), nullptr};

// ---- cross-process fabric (FAKE_VERBS_SHM) --------------------------
// Layout in one shm segment: an MR registry (rkey -> owner node) and
// one op ring per node.  A posting process that does not own the
// target rkey writes a descriptor (+ payload for WRITE) into the
// OWNER's ring and spins for completion; each process's engine thread
// services its own ring by resolving the rkey against its local MRs
// (so peer- and dmabuf-backed regions work as remote targets too).
namespace fabric {

constexpr uint32_t kMagic = 0xFAB51C05;
constexpr int kMaxMrs = 256;
constexpr int kMaxNodes = 4;
constexpr int kSlots = 4;
constexpr size_t kSlotPayload = 1u << 20;

enum SlotState : uint32_t { FREE = 0, CLAIMED, POSTED, DONE, ERR };

struct MrEntry {
  std::atomic<uint32_t> live;
  uint32_t rkey;
  uint32_t node;
  int access;
  uint64_t iova;
  uint64_t len;
};

struct Slot {
  std::atomic<uint32_t> state;
  uint32_t op;  // IBV_WR_RDMA_WRITE / IBV_WR_RDMA_READ
  uint32_t rkey;
  uint32_t len;
  uint64_t remote_addr;
  uint8_t payload[kSlotPayload];
};

struct Ring {
  Slot slots[kSlots];
};

struct Shared {
  std::atomic<uint32_t> magic;
  std::atomic<uint32_t> next_node;
  std::atomic<uint32_t> next_key;  // fabric-global so rkeys never clash
  MrEntry mrs[kMaxMrs];
  Ring rings[kMaxNodes];
};

Shared* g_sh = nullptr;
int g_node = -1;
pthread_t g_engine;
std::atomic<bool> g_engine_stop{false};

uint8_t* local_ptr_for(uint32_t rkey, uint64_t addr, uint64_t len,
                       int* access_out);

void* engine_main(void*) {
  Ring& ring = g_sh->rings[g_node];
  int idle_rounds = 0;
  while (!g_engine_stop.load(std::memory_order_relaxed)) {
    bool idle = true;
    for (auto& s : ring.slots) {
      if (s.state.load(std::memory_order_acquire) != POSTED) continue;
      idle = false;
      int access = 0;
      uint8_t* p = local_ptr_for(s.rkey, s.remote_addr, s.len, &access);
      uint32_t next = DONE;
      if (!p) {
        next = ERR;
      } else if (s.op == IBV_WR_RDMA_WRITE) {
        if (!(access & IBV_ACCESS_REMOTE_WRITE)) next = ERR;
        else memcpy(p, s.payload, s.len);
      } else {
        if (!(access & IBV_ACCESS_REMOTE_READ)) next = ERR;
        else memcpy(s.payload, p, s.len);
      }
      s.state.store(next, std::memory_order_release);
    }
    if (idle) {
      // hybrid poll: stay hot while traffic flows, back off when idle
      if (++idle_rounds < 5000) sched_yield();
      else usleep(50);
    } else {
      idle_rounds = 0;
    }
  }
  return nullptr;
}

bool init_from_env() {
  const char* name = getenv("FAKE_VERBS_SHM");
  if (!name || g_sh) return g_sh != nullptr;
  int fd = shm_open(name, O_RDWR | O_CREAT, 0600);
  if (fd < 0) return false;
  if (ftruncate(fd, sizeof(Shared)) != 0) {
    close(fd);
    return false;
  }
  void* m = mmap(nullptr, sizeof(Shared), PROT_READ | PROT_WRITE,
                 MAP_SHARED, fd, 0);
  close(fd);
  if (m == MAP_FAILED) return false;
  g_sh = reinterpret_cast<Shared*>(m);
  uint32_t expect = 0;
  if (g_sh->magic.compare_exchange_strong(expect, 1)) {
    // first process initializes
    g_sh->next_node.store(0);
    g_sh->next_key.store(0x1000);
    for (auto& e : g_sh->mrs) e.live.store(0);
    for (auto& r : g_sh->rings)
      for (auto& s : r.slots) s.state.store(FREE);
    g_sh->magic.store(kMagic, std::memory_order_release);
  } else {
    while (g_sh->magic.load(std::memory_order_acquire) != kMagic)
      usleep(100);
  }
  g_node = (int)g_sh->next_node.fetch_add(1);
  if (g_node >= kMaxNodes) {
    fprintf(stderr, "fake_verbs: fabric node limit\n");
    g_sh = nullptr;
    return false;
  }
  g_engine_stop.store(false);
  pthread_create(&g_engine, nullptr, engine_main, nullptr);
  return true;
}

void publish(uint32_t rkey, uint64_t iova, uint64_t len, int access) {
  if (!g_sh) return;
  for (auto& e : g_sh->mrs) {
    uint32_t expect = 0;
    if (e.live.compare_exchange_strong(expect, 1)) {
      e.rkey = rkey;
      e.node = (uint32_t)g_node;
      e.iova = iova;
      e.len = len;
      e.access = access;
      e.live.store(2, std::memory_order_release);
      return;
    }
  }
  fprintf(stderr, "fake_verbs: fabric MR registry full\n");
}

void unpublish(uint32_t rkey) {
  if (!g_sh) return;
  for (auto& e : g_sh->mrs)
    if (e.live.load(std::memory_order_acquire) == 2 && e.rkey == rkey &&
        e.node == (uint32_t)g_node)
      e.live.store(0, std::memory_order_release);
}

const MrEntry* lookup(uint32_t rkey) {
  if (!g_sh) return nullptr;
  for (auto& e : g_sh->mrs)
    if (e.live.load(std::memory_order_acquire) == 2 && e.rkey == rkey)
      return &e;
  return nullptr;
}

// One remote one-sided op (chunks > slot payload are split by caller).
int remote_op(const MrEntry* mr, uint32_t op, uint64_t remote_addr,
              uint8_t* local, uint32_t len) {
  Ring& ring = g_sh->rings[mr->node];
  for (;;) {
    for (auto& s : ring.slots) {
      uint32_t expect = FREE;
      if (!s.state.compare_exchange_strong(expect, CLAIMED)) continue;
      s.op = op;
      s.rkey = mr->rkey;
      s.len = len;
      s.remote_addr = remote_addr;
      if (op == IBV_WR_RDMA_WRITE) memcpy(s.payload, local, len);
      s.state.store(POSTED, std::memory_order_release);
      // wait for the owner's engine: hot spin first, then 50 us polls
      // (bounded ~10 s total)
      for (long spins = 0; spins < 300000; spins++) {
        uint32_t st = s.state.load(std::memory_order_acquire);
        if (st == DONE || st == ERR) {
          int rc = (st == DONE) ? 0 : -1;
          if (!rc && op == IBV_WR_RDMA_READ)
            memcpy(local, s.payload, len);
          s.state.store(FREE, std::memory_order_release);
          return rc;
        }
        if (spins < 100000) sched_yield();
        else usleep(50);
      }
      s.state.store(FREE, std::memory_order_release);
      fprintf(stderr, "fake_verbs: fabric op timed out\n");
      return -1;
    }
    usleep(50);
  }
}

}  // namespace fabric

#define FAIL(msg)                                          \
  do {                                                     \
    fprintf(stderr, "fake_verbs: %s\n", msg);              \
    return -1;                                             \
  } while (0)

// g_ctx.mrs is shared with the fabric engine thread.
pthread_mutex_t g_mr_mu = PTHREAD_MUTEX_INITIALIZER;

FakeMr* find_mr_by_key(uint32_t key, bool remote) {
  pthread_mutex_lock(&g_mr_mu);
  for (auto* m : g_ctx.mrs)
    if (m->live && (remote ? m->mr.rkey : m->mr.lkey) == key) {
      pthread_mutex_unlock(&g_mr_mu);
      return m;
    }
  pthread_mutex_unlock(&g_mr_mu);
  return nullptr;
}

bool range_ok(const FakeMr* m, uint64_t addr, uint64_t len) {
  uint64_t base = (uint64_t)m->mr.addr;
  return addr >= base && addr + len <= base + m->mr.length;
}

// rkeys/lkeys: fabric-global when the cross-process fabric is up (so
// two processes never mint the same key), else process-local.
uint32_t next_key() {
  if (fabric::g_sh) return fabric::g_sh->next_key.fetch_add(1);
  return g_ctx.next_key++;
}

// Data-plane pointer for an in-MR address: identity for host MRs,
// iova->CPU-window translation for dmabuf MRs, sg-walk through the
// bridge's device-mapped table for peer MRs.
uint8_t* mr_data_ptr(const FakeMr* m, uint64_t addr) {
  if (m->host_map) return m->host_map + (addr - (uint64_t)m->mr.addr);
#ifdef ROCNR_FAKEVERBS_PEER
  if (!m->segs.empty()) {
    uint64_t off = addr - (uint64_t)m->mr.addr;
    for (const auto& s : m->segs) {
      if (off < s.len)
        return (uint8_t*)rocnr_glue_bus_ptr(s.bus + off);
      off -= s.len;
    }
    return nullptr;
  }
#endif
  return (uint8_t*)addr;
}

}  // namespace

namespace {
namespace fabric {
// Engine-side rkey resolution against this process's MRs (any MR kind:
// host, dmabuf window, bridge-registered peer).
uint8_t* local_ptr_for(uint32_t rkey, uint64_t addr, uint64_t len,
                       int* access_out) {
  FakeMr* m = find_mr_by_key(rkey, true);
  if (!m || !range_ok(m, addr, len)) return nullptr;
  *access_out = m->access;
  return mr_data_ptr(m, addr);
}
}  // namespace fabric
}  // namespace

extern "C" {

struct ibv_device** ibv_get_device_list(int* num) {
  if (num) *num = 1;
  return g_device_list;
}

void ibv_free_device_list(struct ibv_device**) {}

struct ibv_context* ibv_open_device(struct ibv_device* d) {
  if (d != g_device_list[0]) return nullptr;
  fabric::init_from_env();  // no-op unless FAKE_VERBS_SHM is set
  return reinterpret_cast<ibv_context*>(&g_ctx);
}

int ibv_close_device(struct ibv_context*) { return 0; }

struct ibv_pd* ibv_alloc_pd(struct ibv_context* c) {
  return reinterpret_cast<ibv_pd*>(c);
}

int ibv_dealloc_pd(struct ibv_pd*) { return 0; }

struct ibv_mr* ibv_reg_mr(struct ibv_pd* pd, void* addr, size_t length,
                          int access) {
  if (!pd || !addr || !length) return nullptr;
  auto* m = new FakeMr();
#ifdef ROCNR_FAKEVERBS_PEER
  // The IB core's peer-memory probe: a VA that normal pinning cannot
  // claim is offered to registered peer clients — here the REAL
  // rocp2p bridge, which pins through the (fake) KFD and returns the
  // device-mapped sg table the HCA would DMA against.
  if (rocnr_glue_init() == 0 && rocnr_glue_is_gpu((uint64_t)addr)) {
    rocnr_glue_seg segs[512];
    int n = 512;
    void* h = nullptr;
    int r = rocnr_glue_reg_mr((uint64_t)addr, length, &h, segs, &n);
    if (r != 0) {
      fprintf(stderr, "fake_verbs: peer-memory registration failed (%d)\n",
              r);
      delete m;
      return nullptr;
    }
    m->peer_handle = h;
    m->segs.assign(segs, segs + n);
  }
#endif
  m->mr.pd = pd;
  m->mr.addr = addr;
  m->mr.length = length;
  m->mr.lkey = next_key();
  m->mr.rkey = next_key();
  m->access = access;
  m->live = true;
  pthread_mutex_lock(&g_mr_mu);
  g_ctx.mrs.push_back(m);
  pthread_mutex_unlock(&g_mr_mu);
  fabric::publish(m->mr.rkey, (uint64_t)addr, length, access);
  return &m->mr;
}

struct ibv_mr* ibv_reg_dmabuf_mr(struct ibv_pd* pd, uint64_t offset,
                                 size_t length, uint64_t iova, int fd,
                                 int access) {
  // Map the dmabuf into this process the way a real HCA would DMA it:
  // through the exporter's backing pages.  For a VRAM buffer exported
  // by hipMemGetHandleForAddressRange this is the PCIe BAR window
  // (amdgpu implements dma-buf mmap), so posts against the MR move
  // real bytes over the bus; in CPU-only CI the fd is a memfd.
  if (fd < 0) return nullptr;
  void* map = mmap(nullptr, length + offset, PROT_READ | PROT_WRITE,
                   MAP_SHARED, fd, 0);
  if (map == MAP_FAILED) {
    fprintf(stderr,
            "fake_verbs: dmabuf fd %d not mmappable (%m) — registration "
            "refused\n", fd);
    return nullptr;
  }
  auto* m = new FakeMr();
  m->mr.pd = pd;
  m->mr.addr = (void*)iova;
  m->mr.length = length;
  m->mr.lkey = next_key();
  m->mr.rkey = next_key();
  m->access = access;
  m->live = true;
  m->host_map = (uint8_t*)map + offset;
  m->map_len = length + offset;
  pthread_mutex_lock(&g_mr_mu);
  g_ctx.mrs.push_back(m);
  pthread_mutex_unlock(&g_mr_mu);
  fabric::publish(m->mr.rkey, iova, length, access);
  return &m->mr;
}

int ibv_dereg_mr(struct ibv_mr* mr) {
  fabric::unpublish(mr->rkey);
  FakeMr* found = nullptr;
  pthread_mutex_lock(&g_mr_mu);
  for (auto* m : g_ctx.mrs)
    if (&m->mr == mr && m->live) {
      m->live = false;
      found = m;
      break;
    }
  pthread_mutex_unlock(&g_mr_mu);
  if (!found) return -1;
  if (found->host_map)
    munmap(found->host_map - (found->map_len - found->mr.length),
           found->map_len);
  found->host_map = nullptr;
#ifdef ROCNR_FAKEVERBS_PEER
  if (found->peer_handle) {
    // ibv_dereg_mr path: dma_unmap -> put_pages -> release through
    // the real bridge (fake IB core teardown ordering)
    rocnr_glue_dereg_mr(found->peer_handle);
    found->peer_handle = nullptr;
    found->segs.clear();
  }
#endif
  return 0;
}

struct ibv_cq* ibv_create_cq(struct ibv_context*, int cqe, void*, void*,
                             int) {
  if (cqe <= 0) return nullptr;
  return reinterpret_cast<ibv_cq*>(new FakeCq());
}

int ibv_destroy_cq(struct ibv_cq* cq) {
  delete reinterpret_cast<FakeCq*>(cq);
  return 0;
}

struct ibv_qp* ibv_create_qp(struct ibv_pd*, struct ibv_qp_init_attr* a) {
  if (!a || !a->send_cq || a->qp_type != IBV_QPT_RC) return nullptr;
  if (!a->cap.max_send_wr || !a->cap.max_send_sge) return nullptr;
  auto* q = new FakeQp();
  q->qp.qp_num = g_ctx.next_qpn++;
  q->qp.state = IBV_QPS_RESET;
  q->qp.qp_type = a->qp_type;
  q->qp.send_cq = a->send_cq;
  q->send_cq = reinterpret_cast<FakeCq*>(a->send_cq);
  return &q->qp;
}

int ibv_destroy_qp(struct ibv_qp* qp) {
  delete reinterpret_cast<FakeQp*>(qp);
  return 0;
}

int ibv_query_port(struct ibv_context*, uint8_t port,
                   struct ibv_port_attr* pa) {
  if (port != 1 || !pa) return -1;
  memset(pa, 0, sizeof(*pa));
  pa->active_mtu = IBV_MTU_4096;
  pa->max_mtu = IBV_MTU_4096;
  const char* link = getenv("FAKE_VERBS_LINK");
  if (link && !strcmp(link, "eth")) {
    pa->link_layer = IBV_LINK_LAYER_ETHERNET;
    pa->lid = 0;
  } else {
    pa->link_layer = IBV_LINK_LAYER_INFINIBAND;
    pa->lid = 7;
  }
  pa->gid_tbl_len = 4;
  return 0;
}

int ibv_query_gid(struct ibv_context*, uint8_t port, int index,
                  union ibv_gid* gid) {
  if (port != 1 || index < 0 || index >= 4 || !gid) return -1;
  memset(gid, 0, sizeof(*gid));
  gid->raw[15] = (uint8_t)(index + 1);
  return 0;
}

// RC QP state machine with mlx5-grade required-mask checking.
int ibv_modify_qp(struct ibv_qp* qp, struct ibv_qp_attr* a, int mask) {
  auto* q = reinterpret_cast<FakeQp*>(qp);
  if (!(mask & IBV_QP_STATE)) FAIL("modify_qp without IBV_QP_STATE");
  switch (a->qp_state) {
    case IBV_QPS_INIT: {
      if (q->qp.state != IBV_QPS_RESET && q->qp.state != IBV_QPS_INIT)
        FAIL("INIT from wrong state");
      const int need = IBV_QP_STATE | IBV_QP_PKEY_INDEX | IBV_QP_PORT |
                       IBV_QP_ACCESS_FLAGS;
      if ((mask & need) != need) FAIL("RESET->INIT mask incomplete");
      if (a->port_num != 1) FAIL("bad port");
      q->access_flags = (int)a->qp_access_flags;
      q->qp.state = IBV_QPS_INIT;
      return 0;
    }
    case IBV_QPS_RTR: {
      if (q->qp.state != IBV_QPS_INIT) FAIL("RTR from wrong state");
      const int need = IBV_QP_STATE | IBV_QP_AV | IBV_QP_PATH_MTU |
                       IBV_QP_DEST_QPN | IBV_QP_RQ_PSN |
                       IBV_QP_MAX_DEST_RD_ATOMIC | IBV_QP_MIN_RNR_TIMER;
      if ((mask & need) != need) FAIL("INIT->RTR mask incomplete");
      if (a->path_mtu < IBV_MTU_256 || a->path_mtu > IBV_MTU_4096)
        FAIL("bad mtu");
      ibv_port_attr pa;
      ibv_query_port(nullptr, 1, &pa);
      if (pa.link_layer == IBV_LINK_LAYER_ETHERNET) {
        if (!a->ah_attr.is_global) FAIL("RoCE needs GRH (is_global)");
      } else {
        if (!a->ah_attr.dlid) FAIL("IB needs dlid");
      }
      q->dest_qp_num = a->dest_qp_num;
      q->has_av = true;
      q->qp.state = IBV_QPS_RTR;
      return 0;
    }
    case IBV_QPS_RTS: {
      if (q->qp.state != IBV_QPS_RTR) FAIL("RTS from wrong state");
      const int need = IBV_QP_STATE | IBV_QP_SQ_PSN | IBV_QP_TIMEOUT |
                       IBV_QP_RETRY_CNT | IBV_QP_RNR_RETRY |
                       IBV_QP_MAX_QP_RD_ATOMIC;
      if ((mask & need) != need) FAIL("RTR->RTS mask incomplete");
      q->qp.state = IBV_QPS_RTS;
      return 0;
    }
    default:
      FAIL("unsupported target state");
  }
}

int ibv_post_send(struct ibv_qp* qp, struct ibv_send_wr* wr,
                  struct ibv_send_wr** bad) {
  auto* q = reinterpret_cast<FakeQp*>(qp);
  for (; wr; wr = wr->next) {
    if (bad) *bad = wr;
    if (q->qp.state != IBV_QPS_RTS) FAIL("post_send: QP not in RTS");
    if (wr->num_sge != 1) FAIL("post_send: expected 1 sge");
    FakeMr* local = find_mr_by_key(wr->sg_list[0].lkey, false);
    if (!local) FAIL("post_send: bad lkey");
    if (!range_ok(local, wr->sg_list[0].addr, wr->sg_list[0].length))
      FAIL("post_send: local sge out of MR bounds");
    if (wr->opcode != IBV_WR_RDMA_WRITE && wr->opcode != IBV_WR_RDMA_READ)
      FAIL("post_send: unsupported opcode");
    uint8_t* lptr = mr_data_ptr(local, wr->sg_list[0].addr);
    if (wr->opcode == IBV_WR_RDMA_READ &&
        !(local->access & IBV_ACCESS_LOCAL_WRITE))
      FAIL("post_send: local MR lacks LOCAL_WRITE");
    FakeMr* remote = find_mr_by_key(wr->wr.rdma.rkey, true);
    if (remote) {
#ifdef ROCNR_FAKEVERBS_PEER
      if (remote->peer_handle && rocnr_glue_mr_dead(remote->peer_handle))
        FAIL("post_send: peer MR invalidated (producer freed the memory "
             "under it) — remote access error");
#endif
      if (!range_ok(remote, wr->wr.rdma.remote_addr, wr->sg_list[0].length))
        FAIL("post_send: remote range out of MR bounds");
      uint8_t* rptr = mr_data_ptr(remote, wr->wr.rdma.remote_addr);
      if (wr->opcode == IBV_WR_RDMA_WRITE) {
        if (!(remote->access & IBV_ACCESS_REMOTE_WRITE))
          FAIL("post_send: remote MR lacks REMOTE_WRITE");
        memcpy(rptr, lptr, wr->sg_list[0].length);
      } else {
        if (!(remote->access & IBV_ACCESS_REMOTE_READ))
          FAIL("post_send: remote MR lacks REMOTE_READ");
        memcpy(lptr, rptr, wr->sg_list[0].length);
      }
    } else {
      // Cross-process fabric: the rkey may belong to another node —
      // the op is shipped to the owner's NIC-role engine (the remote
      // application thread is never involved, RDMA-style).
      const fabric::MrEntry* fe = fabric::lookup(wr->wr.rdma.rkey);
      if (!fe) FAIL("post_send: bad rkey");
      uint64_t raddr = wr->wr.rdma.remote_addr;
      uint32_t len = wr->sg_list[0].length;
      if (raddr < fe->iova || raddr + len > fe->iova + fe->len)
        FAIL("post_send: remote range out of fabric MR bounds");
      for (uint32_t off = 0; off < len;) {
        uint32_t chunk = len - off;
        if (chunk > fabric::kSlotPayload)
          chunk = (uint32_t)fabric::kSlotPayload;
        if (fabric::remote_op(fe, wr->opcode, raddr + off, lptr + off,
                              chunk) != 0)
          FAIL("post_send: fabric remote op failed");
        off += chunk;
      }
    }
    if (wr->send_flags & IBV_SEND_SIGNALED) {
      ibv_wc wc;
      memset(&wc, 0, sizeof(wc));
      wc.wr_id = wr->wr_id;
      wc.status = IBV_WC_SUCCESS;
      wc.byte_len = wr->sg_list[0].length;
      wc.qp_num = q->qp.qp_num;
      q->send_cq->completions.push_back(wc);
    }
  }
  if (bad) *bad = nullptr;
  return 0;
}

int ibv_poll_cq(struct ibv_cq* cq, int n, struct ibv_wc* wc) {
  auto* c = reinterpret_cast<FakeCq*>(cq);
  int got = 0;
  while (got < n && !c->completions.empty()) {
    wc[got++] = c->completions.front();
    c->completions.pop_front();
  }
  return got;
}

const char* ibv_wc_status_str(enum ibv_wc_status s) {
  return s == IBV_WC_SUCCESS ? "success" : "error";
}

}  // extern "C"
