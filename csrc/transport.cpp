// Transport implementations: RCCL-over-xGMI (GPU), c10d/gloo (CPU), local.
// See transport.hpp for the design rationale vs the reference's raw-MPI
// single transport (helmholtz-analytics/mpi4torch csrc/extension.cpp).

#include "transport.hpp"

#include <torch/csrc/distributed/c10d/Backend.hpp>
#include <torch/csrc/distributed/c10d/ProcessGroup.hpp>
#include <torch/csrc/distributed/c10d/GroupRegistry.hpp>
#include <torch/csrc/distributed/c10d/Types.hpp>

#include <ATen/hip/HIPContext.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>
#include <ATen/hip/impl/HIPCachingAllocatorMasqueradingAsCUDA.h>
#include <ATen/hip/impl/HIPGuardImplMasqueradingAsCUDA.h>

#include <hip/hip_runtime_api.h>
#include <rccl/rccl.h>

#include <atomic>
#include <cstdlib>
#include <cstring>
#include <deque>
#include <map>
#include <mutex>
#include <unordered_map>

namespace m4a {

static void config_from_env(Config& c) {
  const char* dbg = std::getenv("MPI4TORCH_AMD_DEBUG");
  c.debug_collectives = dbg && dbg[0] == '1';
  const char* ffp = std::getenv("MPI4TORCH_AMD_FORCE_FULL_PATH");
  c.force_full_path = ffp && ffp[0] == '1';
  if (const char* to = std::getenv("MPI4TORCH_AMD_TIMEOUT_S")) {
    c.op_timeout_ms = (int64_t)(std::atof(to) * 1000.0);
  }
  if (const char* pc = std::getenv("MPI4TORCH_AMD_PIPELINE_MB")) {
    c.pipeline_chunk_bytes = (int64_t)(std::atof(pc) * 1048576.0);
  }
  const char* fh = std::getenv("MPI4TORCH_AMD_FORCE_HIERARCHICAL");
  c.force_hierarchical = fh && fh[0] == '1';
}

Config& config() {
  static Config cfg = []() {
    Config c;
    config_from_env(c);
    return c;
  }();
  return cfg;
}

void reload_config_from_env() { config_from_env(config()); }

#define M4A_HIP_CHECK(expr)                                              \
  do {                                                                   \
    hipError_t _e = (expr);                                              \
    TORCH_CHECK(_e == hipSuccess, "mpi4torch_amd HIP error: ",           \
                hipGetErrorString(_e), " at " __FILE__ ":",              \
                __LINE__);                                               \
  } while (0)

#define M4A_NCCL_CHECK(expr)                                             \
  do {                                                                   \
    ncclResult_t _r = (expr);                                            \
    TORCH_CHECK(_r == ncclSuccess, "mpi4torch_amd RCCL error: ",         \
                ncclGetErrorString(_r), " at " __FILE__ ":", __LINE__);  \
  } while (0)

namespace {

// ---------------------------------------------------------------------------
// Request table (nonblocking p2p). Mirrors the role of MPI_Request in the
// reference (csrc/extension.cpp:1090-1107) with stream-ordered semantics:
// a GPU request completes by hipEvent, a CPU request by c10d::Work.
// ---------------------------------------------------------------------------

struct Request {
  bool gpu = false;
  bool self_pending = false;  // self-p2p not yet matched
  // Deferred RCCL p2p: the nccl call has not been issued yet; wait_request
  // triggers owner->flush_p2p(), which batches every pending send/recv of
  // this transport into one ncclGroupStart/End (so matched pairs can
  // rendezvous) and fills in `event`.
  bool deferred = false;
  std::weak_ptr<Transport> owner;
  hipEvent_t event = nullptr;
  int device = -1;
  c10::intrusive_ptr<c10d::Work> work;
  std::vector<c10::intrusive_ptr<c10d::Work>> works;  // multi-op requests
  at::Tensor buffer;  // keeps the comm buffer alive until wait
  std::vector<at::Tensor> buffers;  // multi-op requests
  // MPI4TORCH_AMD_DEBUG=1 p2p handshake: the sender ships [tag, dtype,
  // numel] over the host channel; the receiver compares at Wait and raises
  // on mismatch — FIFO-crossed transfers become errors, not silent swaps.
  c10::intrusive_ptr<c10d::Work> debug_work;
  at::Tensor debug_meta;
  std::vector<int64_t> debug_expect;  // recv side: [tag, dtype, numel]
  int debug_peer = -1;
};

struct RequestTable {
  std::mutex mu;
  std::unordered_map<uint64_t, Request> reqs;
  std::atomic<uint64_t> next{1};

  uint64_t add(Request r) {
    uint64_t id = next.fetch_add(1);
    std::lock_guard<std::mutex> g(mu);
    reqs.emplace(id, std::move(r));
    return id;
  }
  Request take(uint64_t id) {
    std::lock_guard<std::mutex> g(mu);
    auto it = reqs.find(id);
    TORCH_CHECK(it != reqs.end(),
                "mpi4torch_amd: unknown or already-waited request id ", id);
    Request r = std::move(it->second);
    reqs.erase(it);
    return r;
  }
  Request* peek(uint64_t id) {
    auto it = reqs.find(id);
    return it == reqs.end() ? nullptr : &it->second;
  }
};

RequestTable& requests() {
  static RequestTable t;
  return t;
}

// ---------------------------------------------------------------------------
// hipEvent pool (per device).
// ---------------------------------------------------------------------------

class EventPool {
 public:
  static EventPool& forDevice(int dev) {
    static std::mutex mu;
    static std::map<int, std::unique_ptr<EventPool>> pools;
    std::lock_guard<std::mutex> g(mu);
    auto& p = pools[dev];
    if (!p) p = std::unique_ptr<EventPool>(new EventPool(dev));
    return *p;
  }
  hipEvent_t acquire() {
    {
      std::lock_guard<std::mutex> g(mu_);
      if (!free_.empty()) {
        hipEvent_t e = free_.back();
        free_.pop_back();
        return e;
      }
    }
    // HIP events belong to the device current at creation time.
    c10::hip::HIPGuardMasqueradingAsCUDA guard(dev_);
    hipEvent_t e;
    M4A_HIP_CHECK(hipEventCreateWithFlags(&e, hipEventDisableTiming));
    return e;
  }
  void release(hipEvent_t e) {
    std::lock_guard<std::mutex> g(mu_);
    free_.push_back(e);
  }

 private:
  explicit EventPool(int dev) : dev_(dev) {}
  int dev_;
  std::mutex mu_;
  std::vector<hipEvent_t> free_;
};

// ---------------------------------------------------------------------------
// Self-p2p matcher: Isend/Irecv where peer == own rank. RCCL point-to-point
// to self outside a single group call is not usable from separate Isend /
// Irecv calls, so self traffic is matched here and lowered to (stream-
// ordered) copies. MPI gets this for free via its matching engine.
// ---------------------------------------------------------------------------

struct SelfMatcher {
  struct Pending {
    uint64_t req_id;
    at::Tensor buf;
    int tag;
  };
  std::mutex mu;
  // Keyed by channel only: matching is FIFO per (peer, channel) on EVERY
  // transport (RCCL has no tags), so self traffic follows the same
  // contract. Tags are validated at match time under MPI4TORCH_AMD_DEBUG=1.
  std::map<int, std::deque<Pending>> sends, recvs;

  static void check_tags(int stag, int rtag) {
    if (!config().debug_collectives) return;
    TORCH_CHECK(stag == rtag,
                "mpi4torch_amd[debug]: self send/recv matched FIFO but with "
                "different tags (send tag ", stag, ", recv tag ", rtag,
                "). Matching is FIFO per (peer, channel); tags do not "
                "reorder it — reorder your posts instead.");
  }

  static void complete_pair(const at::Tensor& src, at::Tensor& dst,
                            uint64_t sreq, uint64_t rreq) {
    dst.copy_(src, /*non_blocking=*/true);  // current-stream-ordered on GPU
    auto& tab = requests();
    std::lock_guard<std::mutex> g(tab.mu);
    for (uint64_t id : {sreq, rreq}) {
      Request* r = tab.peek(id);
      if (!r) continue;
      r->self_pending = false;
      if (src.is_cuda()) {
        r->gpu = true;
        r->device = (int)src.get_device();
        auto cur = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA(r->device);
        r->event = EventPool::forDevice(r->device).acquire();
        M4A_HIP_CHECK(hipEventRecord(r->event, cur.stream()));
      }
    }
  }

  uint64_t isend(const at::Tensor& buf, int tag, Channel ch) {
    Request r;
    r.buffer = buf;
    r.self_pending = true;
    uint64_t id = requests().add(std::move(r));
    std::unique_lock<std::mutex> g(mu);
    auto& rq = recvs[(int)ch];
    if (!rq.empty()) {
      Pending p = rq.front();
      rq.pop_front();
      g.unlock();
      check_tags(tag, p.tag);
      complete_pair(buf, p.buf, id, p.req_id);
    } else {
      sends[(int)ch].push_back({id, buf, tag});
    }
    return id;
  }
  uint64_t irecv(at::Tensor& buf, int tag, Channel ch) {
    Request r;
    r.buffer = buf;
    r.self_pending = true;
    uint64_t id = requests().add(std::move(r));
    std::unique_lock<std::mutex> g(mu);
    auto& sq = sends[(int)ch];
    if (!sq.empty()) {
      Pending p = sq.front();
      sq.pop_front();
      g.unlock();
      check_tags(p.tag, tag);
      complete_pair(p.buf, buf, p.req_id, id);
    } else {
      recvs[(int)ch].push_back({id, buf, tag});
    }
    return id;
  }
};

SelfMatcher& self_matcher() {
  static SelfMatcher m;
  return m;
}

// ---------------------------------------------------------------------------
// dtype / op maps
// ---------------------------------------------------------------------------

// Reduce-capable RCCL dtype map. Extends the reference's torch2mpitype
// (csrc/extension.cpp:106-129, which lacked bf16/half/fp8) to the full
// MI355X-relevant set.
ncclDataType_t nccl_reduce_dtype(at::ScalarType t) {
  switch (t) {
    case at::kByte: return ncclUint8;
    case at::kBool: return ncclUint8;
    case at::kChar: return ncclInt8;
    case at::kInt: return ncclInt32;
    case at::kLong: return ncclInt64;
    case at::kHalf: return ncclFloat16;
    case at::kFloat: return ncclFloat32;
    case at::kDouble: return ncclFloat64;
    case at::kBFloat16: return ncclBfloat16;
    case at::kFloat8_e4m3fn: return ncclFloat8e4m3;
    case at::kFloat8_e5m2: return ncclFloat8e5m2;
    default:
      TORCH_CHECK(false, "mpi4torch_amd: dtype ", t,
                  " not supported for RCCL reductions");
  }
}

ncclRedOp_t nccl_red_op(RedOp op) {
  switch (op) {
    case kSum: return ncclSum;
    case kProd: return ncclProd;
    case kMin: return ncclMin;
    case kMax: return ncclMax;
    default:
      TORCH_CHECK(false, "mpi4torch_amd internal: op ", red_op_name(op),
                  " must be lowered before reaching the RCCL transport");
  }
}

c10d::ReduceOp c10d_red_op(RedOp op) {
  switch (op) {
    case kSum: return c10d::ReduceOp(c10d::ReduceOp::RedOpType::SUM);
    case kProd: return c10d::ReduceOp(c10d::ReduceOp::RedOpType::PRODUCT);
    case kMin: return c10d::ReduceOp(c10d::ReduceOp::RedOpType::MIN);
    case kMax: return c10d::ReduceOp(c10d::ReduceOp::RedOpType::MAX);
    case kBAnd: return c10d::ReduceOp(c10d::ReduceOp::RedOpType::BAND);
    case kBOr: return c10d::ReduceOp(c10d::ReduceOp::RedOpType::BOR);
    case kBXor: return c10d::ReduceOp(c10d::ReduceOp::RedOpType::BXOR);
    default:
      TORCH_CHECK(false, "mpi4torch_amd internal: op ", red_op_name(op),
                  " must be lowered before reaching the c10d transport");
  }
}

void wait_work(const c10::intrusive_ptr<c10d::Work>& w) {
  const int64_t ms = config().op_timeout_ms;
  if (ms > 0) {
    TORCH_CHECK(w->wait(std::chrono::milliseconds(ms)),
                "mpi4torch_amd: collective timed out after ", ms,
                " ms (MPI4TORCH_AMD_TIMEOUT_S) — a peer likely diverged; "
                "run with MPI4TORCH_AMD_DEBUG=1 to locate the desync");
  } else {
    w->wait();
  }
}

c10::intrusive_ptr<c10d::Backend> gloo_backend(const std::string& group_name) {
  auto pg = c10d::resolve_process_group(group_name);
  TORCH_CHECK(pg, "mpi4torch_amd: process group '", group_name,
              "' not found in c10d registry");
  return pg->getBackend(c10::DeviceType::CPU);
}

// Host-channel gloo tag layout (disjoint from p2p data tags):
//   0            exchange() collective-internal block traffic
//   1 + ch       C10dTransport p2p data, FIFO per (peer, channel)
//   16 + ch      MPI4TORCH_AMD_DEBUG p2p metadata handshake
int chan_tag(Channel ch) { return 1 + (int)ch; }
int meta_tag(Channel ch) { return 16 + (int)ch; }

// MPI4TORCH_AMD_DEBUG=1 p2p handshake: piggyback [tag, dtype, numel] over
// the host (gloo) channel alongside every p2p transfer. The meta stream is
// FIFO per (peer, channel) exactly like the data stream, so if user posts
// cross (e.g. tags (a,b) sent against recvs (b,a)), the receiver's Wait
// raises with both ranks' views instead of silently delivering swapped
// payloads. Completion is checked at wait_request — a sync point anyway —
// so debug mode adds no new blocking before that.
void attach_debug_handshake(Request& r,
                            const c10::intrusive_ptr<c10d::Backend>& backend,
                            bool is_send, const at::Tensor& buf, int peer,
                            int tag, Channel ch) {
  if (!config().debug_collectives || !backend) return;
  auto meta = at::zeros({3}, at::TensorOptions().dtype(at::kLong));
  std::vector<at::Tensor> ts{meta};
  if (is_send) {
    auto* m = meta.data_ptr<int64_t>();
    m[0] = tag;
    m[1] = (int64_t)buf.scalar_type();
    m[2] = buf.numel();
    r.debug_work = backend->send(ts, peer, meta_tag(ch));
  } else {
    r.debug_expect = {(int64_t)tag, (int64_t)buf.scalar_type(), buf.numel()};
    r.debug_work = backend->recv(ts, peer, meta_tag(ch));
  }
  r.debug_meta = meta;
  r.debug_peer = peer;
}

void check_debug_handshake(Request& r) {
  if (!r.debug_work) return;
  wait_work(r.debug_work);
  if (r.debug_expect.empty()) return;  // send side: delivery only
  const auto* m = r.debug_meta.data_ptr<int64_t>();
  TORCH_CHECK(
      m[0] == r.debug_expect[0] && m[1] == r.debug_expect[1] &&
          m[2] == r.debug_expect[2],
      "mpi4torch_amd[debug]: p2p transfer mismatch with rank ", r.debug_peer,
      " — matching is FIFO per (peer, channel) and the pair that matched "
      "disagrees: sender posted (tag=", m[0], ", dtype=",
      (at::ScalarType)m[1], ", numel=", m[2], ") but this rank's recv "
      "expected (tag=", r.debug_expect[0], ", dtype=",
      (at::ScalarType)r.debug_expect[1], ", numel=", r.debug_expect[2],
      "). Tags do not reorder matching; make both ranks post their "
      "transfers per peer+channel in the same order.");
}

// ---------------------------------------------------------------------------
// LocalTransport — world of one, no runtime.
// ---------------------------------------------------------------------------

struct LocalTransport final : Transport {
  int rank() const override { return 0; }
  int size() const override { return 1; }
  bool is_gpu() const override { return false; }

  void allreduce(const at::Tensor& in, at::Tensor& out, RedOp) override {
    out.copy_(in, true);
  }
  void broadcast(at::Tensor&, int) override {}
  void reduce(at::Tensor&, RedOp, int) override {}
  void allgather_equal(const at::Tensor& in, at::Tensor& out) override {
    out.view_as(in).copy_(in, true);
  }
  void reduce_scatter_equal(const at::Tensor& in, at::Tensor& out,
                            RedOp) override {
    out.copy_(in.view_as(out), true);
  }
  void exchange(const std::vector<at::Tensor>& sendbufs,
                const std::vector<int>& speers,
                std::vector<at::Tensor>& recvbufs,
                const std::vector<int>& rpeers) override {
    TORCH_CHECK(sendbufs.size() == recvbufs.size(),
                "local exchange requires matched self pairs");
    for (size_t i = 0; i < sendbufs.size(); ++i) {
      TORCH_CHECK(speers[i] == 0 && rpeers[i] == 0);
      recvbufs[i].view({-1}).copy_(sendbufs[i].reshape({-1}), true);
    }
  }
  uint64_t iexchange(const std::vector<at::Tensor>& sendbufs,
                     const std::vector<int>& speers,
                     std::vector<at::Tensor>& recvbufs,
                     const std::vector<int>& rpeers) override {
    exchange(sendbufs, speers, recvbufs, rpeers);
    Request r;
    r.buffers = recvbufs;
    if (!recvbufs.empty() && recvbufs[0].is_cuda()) {
      r.gpu = true;
      r.device = (int)recvbufs[0].get_device();
      auto cur = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA(r.device);
      r.event = EventPool::forDevice(r.device).acquire();
      M4A_HIP_CHECK(hipEventRecord(r.event, cur.stream()));
    }
    return requests().add(std::move(r));
  }
  uint64_t isend(const at::Tensor& buf, int peer, int tag,
                 Channel ch) override {
    TORCH_CHECK(peer == 0, "world_size is 1; cannot send to rank ", peer);
    return self_matcher().isend(buf, tag, ch);
  }
  uint64_t irecv(at::Tensor& buf, int peer, int tag, Channel ch) override {
    TORCH_CHECK(peer == 0, "world_size is 1; cannot receive from rank ", peer);
    return self_matcher().irecv(buf, tag, ch);
  }
  uint64_t ireduce_scatter(const at::Tensor& in, at::Tensor& out,
                           RedOp op) override {
    return iallreduce(in.view_as(out), out, op);
  }
  uint64_t iallgather(const at::Tensor& in, at::Tensor& out) override {
    auto v = out.view_as(in);  // size 1: out and in have equal numel
    return iallreduce(in, v, kSum);
  }
  uint64_t iallreduce(const at::Tensor& in, at::Tensor& out,
                      RedOp) override {
    out.copy_(in, true);
    Request r;
    r.buffer = out;
    if (out.is_cuda()) {
      r.gpu = true;
      r.device = (int)out.get_device();
      auto cur = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA(r.device);
      r.event = EventPool::forDevice(r.device).acquire();
      M4A_HIP_CHECK(hipEventRecord(r.event, cur.stream()));
    }
    return requests().add(std::move(r));
  }
};

// ---------------------------------------------------------------------------
// C10dTransport — CPU tensors over a gloo backend. Replaces the reference's
// host-staging MPI path (csrc/extension.cpp:61-104) as the CPU story.
// ---------------------------------------------------------------------------

struct C10dTransport final : Transport {
  explicit C10dTransport(const std::string& group_name)
      : name_(group_name), backend_(gloo_backend(group_name)) {}

  int rank() const override { return backend_->getRank(); }
  int size() const override { return backend_->getSize(); }
  bool is_gpu() const override { return false; }

  void allreduce(const at::Tensor& in, at::Tensor& out, RedOp op) override {
    std::lock_guard<std::mutex> g(mu_);
    out.copy_(in);
    std::vector<at::Tensor> ts{out};
    c10d::AllreduceOptions opts;
    opts.reduceOp = c10d_red_op(op);
    wait_work(backend_->allreduce(ts, opts));
  }
  void broadcast(at::Tensor& t, int root) override {
    std::lock_guard<std::mutex> g(mu_);
    std::vector<at::Tensor> ts{t};
    c10d::BroadcastOptions opts;
    opts.rootRank = root;
    wait_work(backend_->broadcast(ts, opts));
  }
  void reduce(at::Tensor& t, RedOp op, int root) override {
    std::lock_guard<std::mutex> g(mu_);
    std::vector<at::Tensor> ts{t};
    c10d::ReduceOptions opts;
    opts.reduceOp = c10d_red_op(op);
    opts.rootRank = root;
    wait_work(backend_->reduce(ts, opts));
  }
  void allgather_equal(const at::Tensor& in, at::Tensor& out) override {
    std::lock_guard<std::mutex> g(mu_);
    auto chunks = out.view({size(), -1}).unbind(0);
    std::vector<at::Tensor> outs;
    for (auto& c : chunks) outs.push_back(c.view_as(in));
    std::vector<std::vector<at::Tensor>> outputs{outs};
    std::vector<at::Tensor> inputs{const_cast<at::Tensor&>(in)};
    wait_work(backend_->allgather(outputs, inputs));
  }
  void reduce_scatter_equal(const at::Tensor& in, at::Tensor& out,
                            RedOp op) override {
    // gloo has no reduce_scatter: allreduce a copy, slice our block.
    auto tmp = in.clone();
    {
      std::lock_guard<std::mutex> g(mu_);
      std::vector<at::Tensor> ts{tmp};
      c10d::AllreduceOptions opts;
      opts.reduceOp = c10d_red_op(op);
      wait_work(backend_->allreduce(ts, opts));
    }
    out.copy_(tmp.view({size(), -1})[rank()].view_as(out));
  }
  void exchange(const std::vector<at::Tensor>& sendbufs,
                const std::vector<int>& speers,
                std::vector<at::Tensor>& recvbufs,
                const std::vector<int>& rpeers) override {
    std::lock_guard<std::mutex> g(mu_);
    const int me = rank();
    std::vector<c10::intrusive_ptr<c10d::Work>> works;
    std::vector<const at::Tensor*> self_sends;
    std::vector<at::Tensor*> self_recvs;
    // Post receives first, then sends; gloo matches by (peer, tag) with
    // tag 0 reserved for collective-internal block traffic.
    for (size_t j = 0; j < recvbufs.size(); ++j) {
      if (rpeers[j] == me) {
        self_recvs.push_back(&recvbufs[j]);
        continue;
      }
      if (recvbufs[j].numel() == 0) continue;
      std::vector<at::Tensor> ts{recvbufs[j]};
      works.push_back(backend_->recv(ts, rpeers[j], /*tag=*/0));
    }
    for (size_t i = 0; i < sendbufs.size(); ++i) {
      if (speers[i] == me) {
        self_sends.push_back(&sendbufs[i]);
        continue;
      }
      if (sendbufs[i].numel() == 0) continue;
      std::vector<at::Tensor> ts{const_cast<at::Tensor&>(sendbufs[i])};
      works.push_back(backend_->send(ts, speers[i], /*tag=*/0));
    }
    TORCH_CHECK(self_sends.size() == self_recvs.size(),
                "exchange: unmatched self blocks");
    for (size_t i = 0; i < self_sends.size(); ++i) {
      // blocks may differ in logical shape (full tensor vs flat staging);
      // they always match in element count
      self_recvs[i]->view({-1}).copy_(self_sends[i]->reshape({-1}));
    }
    for (auto& w : works) wait_work(w);
  }
  uint64_t iexchange(const std::vector<at::Tensor>& sendbufs,
                     const std::vector<int>& speers,
                     std::vector<at::Tensor>& recvbufs,
                     const std::vector<int>& rpeers) override {
    std::lock_guard<std::mutex> g(mu_);
    const int me = rank();
    Request r;
    std::vector<const at::Tensor*> self_sends;
    std::vector<at::Tensor*> self_recvs;
    for (size_t j = 0; j < recvbufs.size(); ++j) {
      if (rpeers[j] == me) {
        self_recvs.push_back(&recvbufs[j]);
        continue;
      }
      if (recvbufs[j].numel() == 0) continue;
      std::vector<at::Tensor> ts{recvbufs[j]};
      r.works.push_back(backend_->recv(ts, rpeers[j], /*tag=*/0));
      r.buffers.push_back(recvbufs[j]);
    }
    for (size_t i = 0; i < sendbufs.size(); ++i) {
      if (speers[i] == me) {
        self_sends.push_back(&sendbufs[i]);
        continue;
      }
      if (sendbufs[i].numel() == 0) continue;
      std::vector<at::Tensor> ts{const_cast<at::Tensor&>(sendbufs[i])};
      r.works.push_back(backend_->send(ts, speers[i], /*tag=*/0));
      r.buffers.push_back(sendbufs[i]);
    }
    TORCH_CHECK(self_sends.size() == self_recvs.size(),
                "iexchange: unmatched self blocks");
    for (size_t i = 0; i < self_sends.size(); ++i) {
      self_recvs[i]->view({-1}).copy_(self_sends[i]->reshape({-1}));
    }
    return requests().add(std::move(r));
  }
  // p2p matching is FIFO per (peer, channel) — the fixed per-channel gloo
  // tag reproduces the RCCL transport's tag-free semantics exactly, so the
  // CPU SPMD suite exercises the same contract the GPU runs (user tags are
  // metadata, validated by the debug handshake).
  uint64_t isend(const at::Tensor& buf, int peer, int tag,
                 Channel ch) override {
    if (peer == rank()) return self_matcher().isend(buf, tag, ch);
    std::lock_guard<std::mutex> g(mu_);
    std::vector<at::Tensor> ts{const_cast<at::Tensor&>(buf)};
    Request r;
    r.buffer = buf;
    attach_debug_handshake(r, backend_, /*is_send=*/true, buf, peer, tag, ch);
    r.work = backend_->send(ts, peer, chan_tag(ch));
    return requests().add(std::move(r));
  }
  uint64_t irecv(at::Tensor& buf, int peer, int tag, Channel ch) override {
    if (peer == rank()) return self_matcher().irecv(buf, tag, ch);
    std::lock_guard<std::mutex> g(mu_);
    std::vector<at::Tensor> ts{buf};
    Request r;
    r.buffer = buf;
    attach_debug_handshake(r, backend_, /*is_send=*/false, buf, peer, tag, ch);
    r.work = backend_->recv(ts, peer, chan_tag(ch));
    return requests().add(std::move(r));
  }
  uint64_t iallreduce(const at::Tensor& in, at::Tensor& out,
                      RedOp op) override {
    std::lock_guard<std::mutex> g(mu_);
    out.copy_(in);
    std::vector<at::Tensor> ts{out};
    c10d::AllreduceOptions opts;
    opts.reduceOp = c10d_red_op(op);
    Request r;
    r.buffer = out;
    r.work = backend_->allreduce(ts, opts);
    return requests().add(std::move(r));
  }
  uint64_t ireduce_scatter(const at::Tensor& in, at::Tensor& out,
                           RedOp op) override {
    // gloo has no reduce_scatter: synchronous emulation, pre-completed
    // request (CPU correctness path; overlap is a GPU concern)
    reduce_scatter_equal(in, out, op);
    Request r;
    r.buffer = out;
    return requests().add(std::move(r));
  }
  uint64_t iallgather(const at::Tensor& in, at::Tensor& out) override {
    std::lock_guard<std::mutex> g(mu_);
    auto chunks = out.view({size(), -1}).unbind(0);
    std::vector<at::Tensor> outs;
    for (auto& c : chunks) outs.push_back(c.view_as(in));
    std::vector<std::vector<at::Tensor>> outputs{outs};
    std::vector<at::Tensor> inputs{const_cast<at::Tensor&>(in)};
    Request r;
    r.buffer = out;
    r.work = backend_->allgather(outputs, inputs);
    return requests().add(std::move(r));
  }

 private:
  std::string name_;
  c10::intrusive_ptr<c10d::Backend> backend_;
  std::mutex mu_;
};

// ---------------------------------------------------------------------------
// RcclTransport — the MI355X path. One process per GPU; three RCCL
// communicators (collectives / forward p2p / backward p2p) on dedicated
// high-priority HIP streams from the torch stream pool, joined to the
// caller's compute stream by events (never a host sync).
// ---------------------------------------------------------------------------

struct RcclTransport final : Transport {
  RcclTransport(const std::string& group_name, int device)
      : device_(device),
        streams_{c10::hip::getStreamFromPoolMasqueradingAsCUDA(true, device),
                 c10::hip::getStreamFromPoolMasqueradingAsCUDA(true, device),
                 c10::hip::getStreamFromPoolMasqueradingAsCUDA(true, device)} {
    auto backend = gloo_backend(group_name);
    host_backend_ = backend;  // kept for the debug p2p handshake
    rank_ = backend->getRank();
    size_ = backend->getSize();
    // ncclUniqueId exchange over the gloo backend: the MI355X equivalent of
    // the reference's MPI_Init_thread rendezvous (csrc/extension.cpp:1373).
    ncclUniqueId ids[3];
    auto idt = at::empty({(int64_t)(3 * sizeof(ncclUniqueId))},
                         at::TensorOptions().dtype(at::kByte));
    if (rank_ == 0) {
      for (auto& id : ids) M4A_NCCL_CHECK(ncclGetUniqueId(&id));
      std::memcpy(idt.data_ptr(), ids, sizeof(ids));
    }
    {
      std::vector<at::Tensor> ts{idt};
      c10d::BroadcastOptions opts;
      opts.rootRank = 0;
      backend->broadcast(ts, opts)->wait();
    }
    std::memcpy(ids, idt.data_ptr(), sizeof(ids));
    M4A_HIP_CHECK(hipSetDevice(device_));
    for (int i = 0; i < 3; ++i) {
      M4A_NCCL_CHECK(ncclCommInitRank(&comms_[i], size_, ids[i], rank_));
    }
  }

  ~RcclTransport() override {
    for (auto& c : comms_) {
      if (c) ncclCommDestroy(c);
    }
  }

  int rank() const override { return rank_; }
  int size() const override { return size_; }
  bool is_gpu() const override { return true; }

  void allreduce(const at::Tensor& in, at::Tensor& out, RedOp op) override {
    std::lock_guard<std::mutex> g(mu_);
    Hop hop(*this, Channel::Coll, {in, out});
    M4A_NCCL_CHECK(ncclAllReduce(in.data_ptr(), out.data_ptr(), in.numel(),
                                 nccl_reduce_dtype(in.scalar_type()),
                                 nccl_red_op(op), comm(Channel::Coll),
                                 hop.stream()));
  }
  void broadcast(at::Tensor& t, int root) override {
    std::lock_guard<std::mutex> g(mu_);
    Hop hop(*this, Channel::Coll, {t});
    // byte-typed: broadcast moves bytes, dtype-agnostic
    M4A_NCCL_CHECK(ncclBroadcast(t.data_ptr(), t.data_ptr(), nbytes(t),
                                 ncclUint8, root, comm(Channel::Coll),
                                 hop.stream()));
  }
  void reduce(at::Tensor& t, RedOp op, int root) override {
    std::lock_guard<std::mutex> g(mu_);
    Hop hop(*this, Channel::Coll, {t});
    M4A_NCCL_CHECK(ncclReduce(t.data_ptr(), t.data_ptr(), t.numel(),
                              nccl_reduce_dtype(t.scalar_type()),
                              nccl_red_op(op), root, comm(Channel::Coll),
                              hop.stream()));
  }
  void allgather_equal(const at::Tensor& in, at::Tensor& out) override {
    std::lock_guard<std::mutex> g(mu_);
    Hop hop(*this, Channel::Coll, {in, out});
    M4A_NCCL_CHECK(ncclAllGather(in.data_ptr(), out.data_ptr(), nbytes(in),
                                 ncclUint8, comm(Channel::Coll),
                                 hop.stream()));
  }
  void reduce_scatter_equal(const at::Tensor& in, at::Tensor& out,
                            RedOp op) override {
    std::lock_guard<std::mutex> g(mu_);
    Hop hop(*this, Channel::Coll, {in, out});
    M4A_NCCL_CHECK(ncclReduceScatter(in.data_ptr(), out.data_ptr(),
                                     out.numel(),
                                     nccl_reduce_dtype(out.scalar_type()),
                                     nccl_red_op(op), comm(Channel::Coll),
                                     hop.stream()));
  }
  void exchange(const std::vector<at::Tensor>& sendbufs,
                const std::vector<int>& speers,
                std::vector<at::Tensor>& recvbufs,
                const std::vector<int>& rpeers) override {
    std::lock_guard<std::mutex> g(mu_);
    std::vector<at::Tensor> all(sendbufs);
    all.insert(all.end(), recvbufs.begin(), recvbufs.end());
    Hop hop(*this, Channel::Coll, all);
    const int me = rank_;
    std::vector<const at::Tensor*> self_sends;
    std::vector<at::Tensor*> self_recvs;
    // One grouped launch: RCCL fuses these into a single kernel per peer
    // set — this IS the v-collective (no MPI_Gatherv/Scatterv analog
    // needed).
    M4A_NCCL_CHECK(ncclGroupStart());
    for (size_t i = 0; i < sendbufs.size(); ++i) {
      if (speers[i] == me) {
        self_sends.push_back(&sendbufs[i]);
        continue;
      }
      if (nbytes(sendbufs[i]) == 0) continue;
      M4A_NCCL_CHECK(ncclSend(sendbufs[i].data_ptr(), nbytes(sendbufs[i]),
                              ncclUint8, speers[i], comm(Channel::Coll),
                              hop.stream()));
    }
    for (size_t j = 0; j < recvbufs.size(); ++j) {
      if (rpeers[j] == me) {
        self_recvs.push_back(&recvbufs[j]);
        continue;
      }
      if (nbytes(recvbufs[j]) == 0) continue;
      M4A_NCCL_CHECK(ncclRecv(recvbufs[j].data_ptr(), nbytes(recvbufs[j]),
                              ncclUint8, rpeers[j], comm(Channel::Coll),
                              hop.stream()));
    }
    M4A_NCCL_CHECK(ncclGroupEnd());
    TORCH_CHECK(self_sends.size() == self_recvs.size(),
                "exchange: unmatched self blocks");
    for (size_t i = 0; i < self_sends.size(); ++i) {
      // device copy on the collective stream, inside the hop bracket
      M4A_HIP_CHECK(hipMemcpyAsync(
          self_recvs[i]->data_ptr(), self_sends[i]->data_ptr(),
          nbytes(*self_sends[i]), hipMemcpyDeviceToDevice, hop.stream()));
    }
  }
  // Non-blocking p2p is DEFERRED: the nccl call is not issued here. Every
  // pending send/recv of this transport is launched by flush_p2p() inside
  // ONE ncclGroupStart/End (triggered by the first Wait on any of them) so
  // matched pairs land in a single fused kernel and can rendezvous — the
  // c10d batch_isend_irecv pattern. Eagerly issuing ncclSend/ncclRecv
  // serially on one stream deadlocks a ring/bidirectional exchange on >=2
  // GPUs once payloads exceed RCCL's internal FIFO buffering: every rank's
  // send kernel spins for a peer recv that is queued *behind* a blocked
  // send on that peer's own p2p stream.
  uint64_t isend(const at::Tensor& buf, int peer, int tag,
                 Channel ch) override {
    if (peer == rank_) return self_matcher().isend(buf, tag, ch);
    std::lock_guard<std::mutex> g(mu_);
    check_not_capturing("Isend");
    return defer_p2p(/*is_send=*/true, buf, peer, tag, ch);
  }
  uint64_t irecv(at::Tensor& buf, int peer, int tag, Channel ch) override {
    if (peer == rank_) return self_matcher().irecv(buf, tag, ch);
    std::lock_guard<std::mutex> g(mu_);
    check_not_capturing("Irecv");
    return defer_p2p(/*is_send=*/false, buf, peer, tag, ch);
  }

  void flush_p2p() override {
    std::lock_guard<std::mutex> g(mu_);
    flush_pending_locked();
  }
  uint64_t iexchange(const std::vector<at::Tensor>& sendbufs,
                     const std::vector<int>& speers,
                     std::vector<at::Tensor>& recvbufs,
                     const std::vector<int>& rpeers) override {
    std::lock_guard<std::mutex> g(mu_);
    check_not_capturing("iexchange");
    std::vector<at::Tensor> all(sendbufs);
    all.insert(all.end(), recvbufs.begin(), recvbufs.end());
    enter_side(Channel::Coll, all);
    const int me = rank_;
    std::vector<const at::Tensor*> self_sends;
    std::vector<at::Tensor*> self_recvs;
    M4A_NCCL_CHECK(ncclGroupStart());
    for (size_t i = 0; i < sendbufs.size(); ++i) {
      if (speers[i] == me) {
        self_sends.push_back(&sendbufs[i]);
        continue;
      }
      if (nbytes(sendbufs[i]) == 0) continue;
      M4A_NCCL_CHECK(ncclSend(sendbufs[i].data_ptr(), nbytes(sendbufs[i]),
                              ncclUint8, speers[i], comm(Channel::Coll),
                              stream(Channel::Coll)));
    }
    for (size_t j = 0; j < recvbufs.size(); ++j) {
      if (rpeers[j] == me) {
        self_recvs.push_back(&recvbufs[j]);
        continue;
      }
      if (nbytes(recvbufs[j]) == 0) continue;
      M4A_NCCL_CHECK(ncclRecv(recvbufs[j].data_ptr(), nbytes(recvbufs[j]),
                              ncclUint8, rpeers[j], comm(Channel::Coll),
                              stream(Channel::Coll)));
    }
    M4A_NCCL_CHECK(ncclGroupEnd());
    TORCH_CHECK(self_sends.size() == self_recvs.size(),
                "iexchange: unmatched self blocks");
    for (size_t i = 0; i < self_sends.size(); ++i) {
      M4A_HIP_CHECK(hipMemcpyAsync(
          self_recvs[i]->data_ptr(), self_sends[i]->data_ptr(),
          nbytes(*self_sends[i]), hipMemcpyDeviceToDevice,
          stream(Channel::Coll)));
    }
    uint64_t id = make_gpu_request(Channel::Coll, at::Tensor());
    // keep every buffer alive until the wait
    {
      auto& tab = requests();
      std::lock_guard<std::mutex> tg(tab.mu);
      if (Request* rq = tab.peek(id)) rq->buffers = all;
    }
    return id;
  }
  uint64_t iallgather(const at::Tensor& in, at::Tensor& out) override {
    std::lock_guard<std::mutex> g(mu_);
    check_not_capturing("Iallgather");
    enter_side(Channel::Coll, {in, out});
    M4A_NCCL_CHECK(ncclAllGather(in.data_ptr(), out.data_ptr(), nbytes(in),
                                 ncclUint8, comm(Channel::Coll),
                                 stream(Channel::Coll)));
    return make_gpu_request(Channel::Coll, out);
  }
  uint64_t ireduce_scatter(const at::Tensor& in, at::Tensor& out,
                           RedOp op) override {
    std::lock_guard<std::mutex> g(mu_);
    check_not_capturing("Ireducescatter");
    enter_side(Channel::Coll, {in, out});
    M4A_NCCL_CHECK(ncclReduceScatter(in.data_ptr(), out.data_ptr(),
                                     out.numel(),
                                     nccl_reduce_dtype(out.scalar_type()),
                                     nccl_red_op(op), comm(Channel::Coll),
                                     stream(Channel::Coll)));
    return make_gpu_request(Channel::Coll, out);
  }

  // Runtime probe: does this RCCL build reduce fp8 natively? One tiny
  // allreduce per dtype, issued symmetrically on all ranks (the op layer
  // reaches this on every rank at the same program point). Cached.
  bool fp8_reduce_supported(at::ScalarType t) override {
    const int idx = (t == at::kFloat8_e4m3fn) ? 0 : 1;
    std::lock_guard<std::mutex> g(mu_);
    if (fp8_state_[idx] != 0) return fp8_state_[idx] > 0;
    void* buf = nullptr;
    M4A_HIP_CHECK(hipMalloc(&buf, 64));
    M4A_HIP_CHECK(hipMemsetAsync(buf, 0, 64, stream(Channel::Coll)));
    ncclResult_t rc = ncclAllReduce(
        buf, static_cast<char*>(buf) + 32, 32, nccl_reduce_dtype(t), ncclSum,
        comm(Channel::Coll), stream(Channel::Coll));
    if (rc == ncclSuccess) {
      M4A_HIP_CHECK(hipStreamSynchronize(stream(Channel::Coll)));
    }
    M4A_HIP_CHECK(hipFree(buf));
    fp8_state_[idx] = (rc == ncclSuccess) ? 1 : -1;
    return fp8_state_[idx] > 0;
  }

  uint64_t iallreduce(const at::Tensor& in, at::Tensor& out,
                      RedOp op) override {
    // On the collective stream WITHOUT the tail hop: the caller's stream
    // only waits when the returned request is waited — this is what lets
    // gradient-bucket allreduce overlap the rest of backward.
    std::lock_guard<std::mutex> g(mu_);
    check_not_capturing("Iallreduce");
    enter_side(Channel::Coll, {in, out});
    M4A_NCCL_CHECK(ncclAllReduce(in.data_ptr(), out.data_ptr(), in.numel(),
                                 nccl_reduce_dtype(in.scalar_type()),
                                 nccl_red_op(op), comm(Channel::Coll),
                                 stream(Channel::Coll)));
    return make_gpu_request(Channel::Coll, out);
  }

 private:
  static size_t nbytes(const at::Tensor& t) {
    return (size_t)t.numel() * t.element_size();
  }
  ncclComm_t comm(Channel ch) { return comms_[(int)ch]; }
  hipStream_t stream(Channel ch) { return streams_[(int)ch].stream(); }

  static bool stream_capturing(hipStream_t s) {
    hipStreamCaptureStatus st = hipStreamCaptureStatusNone;
    (void)hipStreamIsCapturing(s, &st);
    return st != hipStreamCaptureStatusNone;
  }

  // Failure detection (the reference has none beyond ierr
  // checks): surface asynchronous RCCL errors (peer crash, xGMI fault,
  // aborted communicator) as exceptions at the next collective instead of
  // hanging the job.
  void check_not_capturing(const char* what) {
    auto cur = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA(device_);
    TORCH_CHECK(!stream_capturing(cur.stream()),
                "mpi4torch_amd: ", what,
                " inside hipGraph capture is not supported (its Wait may "
                "fall outside the graph); capture blocking collectives only");
  }

  void check_async_errors() {
    for (int i = 0; i < 3; ++i) {
      if (!comms_[i]) continue;
      ncclResult_t async_err = ncclSuccess;
      if (ncclCommGetAsyncError(comms_[i], &async_err) == ncclSuccess) {
        TORCH_CHECK(async_err == ncclSuccess || async_err == ncclInProgress,
                    "mpi4torch_amd: RCCL communicator (channel ", i,
                    ") is in an error state: ",
                    ncclGetErrorString(async_err),
                    " — a peer likely failed; restart the job");
      }
    }
  }

  // Make the side stream wait on the caller's current stream, and record
  // every touched tensor with the caching allocator against the side
  // stream so its memory is not reused while the collective is in flight.
  void enter_side(Channel ch, const std::vector<at::Tensor>& tensors) {
    check_async_errors();
    auto cur = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA(device_);
    auto& pool = EventPool::forDevice(device_);
    hipEvent_t e = pool.acquire();
    M4A_HIP_CHECK(hipEventRecord(e, cur.stream()));
    M4A_HIP_CHECK(hipStreamWaitEvent(stream(ch), e, 0));
    pool.release(e);
    for (const auto& t : tensors) {
      c10::hip::HIPCachingAllocatorMasqueradingAsCUDA::
          recordStreamMasqueradingAsCUDA(t.storage().data_ptr(),
                                         streams_[(int)ch]);
    }
  }
  // Make the caller's current stream wait on the side stream.
  void exit_side(Channel ch) {
    auto cur = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA(device_);
    auto& pool = EventPool::forDevice(device_);
    hipEvent_t e = pool.acquire();
    M4A_HIP_CHECK(hipEventRecord(e, stream(ch)));
    M4A_HIP_CHECK(hipStreamWaitEvent(cur.stream(), e, 0));
    pool.release(e);
  }
  uint64_t make_gpu_request(Channel ch, const at::Tensor& buf) {
    Request r;
    r.gpu = true;
    r.device = device_;
    r.event = EventPool::forDevice(device_).acquire();
    M4A_HIP_CHECK(hipEventRecord(r.event, stream(ch)));
    r.buffer = buf;
    return requests().add(std::move(r));
  }

  struct PendingP2P {
    bool is_send;
    at::Tensor buf;
    int peer;
    Channel ch;
    uint64_t req_id;
    hipEvent_t ready;  // caller-stream position at enqueue time
  };

  // mu_ held. Record where the caller's stream is (the buffer's producing
  // ops), park the op, and hand back a deferred request.
  uint64_t defer_p2p(bool is_send, const at::Tensor& buf, int peer, int tag,
                     Channel ch) {
    check_async_errors();
    auto cur = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA(device_);
    auto& pool = EventPool::forDevice(device_);
    hipEvent_t ready = pool.acquire();
    M4A_HIP_CHECK(hipEventRecord(ready, cur.stream()));
    c10::hip::HIPCachingAllocatorMasqueradingAsCUDA::
        recordStreamMasqueradingAsCUDA(buf.storage().data_ptr(),
                                       streams_[(int)ch]);
    Request r;
    r.gpu = true;
    r.device = device_;
    r.deferred = true;
    r.buffer = buf;
    r.owner = weak_from_this();
    attach_debug_handshake(r, host_backend_, is_send, buf, peer, tag, ch);
    uint64_t id = requests().add(std::move(r));
    pending_.push_back({is_send, buf, peer, ch, id, ready});
    return id;
  }

  // mu_ held. Launch every pending p2p op in one grouped call (one fused
  // RCCL kernel per channel), then stamp each request's completion event.
  void flush_pending_locked() {
    if (pending_.empty()) return;
    check_async_errors();
    auto& pool = EventPool::forDevice(device_);
    for (auto& p : pending_) {
      M4A_HIP_CHECK(hipStreamWaitEvent(stream(p.ch), p.ready, 0));
      pool.release(p.ready);
    }
    M4A_NCCL_CHECK(ncclGroupStart());
    for (auto& p : pending_) {
      if (nbytes(p.buf) == 0) continue;
      if (p.is_send) {
        M4A_NCCL_CHECK(ncclSend(p.buf.data_ptr(), nbytes(p.buf), ncclUint8,
                                p.peer, comm(p.ch), stream(p.ch)));
      } else {
        M4A_NCCL_CHECK(ncclRecv(p.buf.data_ptr(), nbytes(p.buf), ncclUint8,
                                p.peer, comm(p.ch), stream(p.ch)));
      }
    }
    M4A_NCCL_CHECK(ncclGroupEnd());
    auto& tab = requests();
    std::lock_guard<std::mutex> tg(tab.mu);
    for (auto& p : pending_) {
      Request* r = tab.peek(p.req_id);
      if (!r) continue;
      r->deferred = false;
      r->event = pool.acquire();
      M4A_HIP_CHECK(hipEventRecord(r->event, stream(p.ch)));
    }
    pending_.clear();
  }

  // Stream bracket for one collective. Under hipGraph capture the op runs
  // directly on the capturing stream (events/recordStream are skipped: the
  // graph serializes ordering and graph memory pools own lifetimes) so
  // collectives are capturable like ProcessGroupNCCL's.
  struct Hop {
    Hop(RcclTransport& t, Channel ch, const std::vector<at::Tensor>& ts)
        : t_(t), ch_(ch) {
      auto cur = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA(t_.device_);
      capturing_ = stream_capturing(cur.stream());
      if (capturing_) {
        use_stream_ = cur.stream();
      } else {
        t_.enter_side(ch_, ts);
        use_stream_ = t_.stream(ch_);
      }
    }
    ~Hop() {
      if (!capturing_) t_.exit_side(ch_);
    }
    hipStream_t stream() const { return use_stream_; }
    RcclTransport& t_;
    Channel ch_;
    hipStream_t use_stream_;
    bool capturing_ = false;
  };

  int device_;
  int rank_ = 0, size_ = 1;
  std::mutex mu_;
  ncclComm_t comms_[3] = {nullptr, nullptr, nullptr};
  int fp8_state_[2] = {0, 0};  // 0 unknown, 1 native, -1 cast fallback
  c10::hip::HIPStreamMasqueradingAsCUDA streams_[3];
  c10::intrusive_ptr<c10d::Backend> host_backend_;
  std::vector<PendingP2P> pending_;
};

} // namespace

void wait_request(uint64_t id) {
  // Deferred RCCL p2p: launch the owning transport's whole pending batch
  // (one grouped call) before completing this request.
  {
    std::shared_ptr<Transport> owner;
    {
      auto& tab = requests();
      std::lock_guard<std::mutex> g(tab.mu);
      Request* r = tab.peek(id);
      TORCH_CHECK(r, "mpi4torch_amd: unknown or already-waited request id ",
                  id);
      if (r->deferred) owner = r->owner.lock();
    }
    if (owner) owner->flush_p2p();
  }
  Request r = requests().take(id);
  TORCH_CHECK(!r.self_pending,
              "mpi4torch_amd: Wait() on a self send/recv whose matching "
              "operation was never posted");
  TORCH_CHECK(!r.deferred,
              "mpi4torch_amd internal: deferred p2p request not flushed");
  // Recv side validates the handshake BEFORE the data wait: the sender's
  // metadata arrives no later than its payload, so a mismatched pair is
  // diagnosed even when the data transfer itself would hang.
  if (!r.debug_expect.empty()) check_debug_handshake(r);
  if (r.gpu) {
    auto cur = c10::hip::getCurrentHIPStreamMasqueradingAsCUDA(r.device);
    M4A_HIP_CHECK(hipStreamWaitEvent(cur.stream(), r.event, 0));
    EventPool::forDevice(r.device).release(r.event);
  } else if (r.work) {
    wait_work(r.work);
  }
  for (auto& w : r.works) wait_work(w);
  // Send side checks after the data wait (data delivered => the peer posted
  // its recv => its meta recv is posted too; no added blocking).
  if (r.debug_expect.empty()) check_debug_handshake(r);
}

std::shared_ptr<Transport> make_local_transport() {
  return std::make_shared<LocalTransport>();
}

std::shared_ptr<Transport> make_c10d_transport(const std::string& group_name) {
  return std::make_shared<C10dTransport>(group_name);
}

std::shared_ptr<Transport> make_rccl_transport(const std::string& group_name,
                                               int device) {
  return std::make_shared<RcclTransport>(group_name, device);
}

std::vector<int64_t> host_allgather_int64(const std::string& group_name,
                                          int64_t value) {
  auto backend = gloo_backend(group_name);
  const int n = backend->getSize();
  auto in = at::full({1}, value, at::TensorOptions().dtype(at::kLong));
  auto out = at::empty({n}, at::TensorOptions().dtype(at::kLong));
  std::vector<at::Tensor> outs;
  outs.reserve(n);
  for (int i = 0; i < n; ++i) outs.push_back(out[i].view({1}));
  std::vector<std::vector<at::Tensor>> outputs{outs};
  std::vector<at::Tensor> inputs{in};
  backend->allgather(outputs, inputs)->wait();
  std::vector<int64_t> res(n);
  std::memcpy(res.data(), out.data_ptr(), n * sizeof(int64_t));
  return res;
}

std::vector<int64_t> host_allgather_int64_vec(
    const std::string& group_name, const std::vector<int64_t>& v) {
  auto backend = gloo_backend(group_name);
  const int n = backend->getSize();
  const int64_t len = (int64_t)v.size();
  auto in = at::empty({len}, at::TensorOptions().dtype(at::kLong));
  std::memcpy(in.data_ptr(), v.data(), len * sizeof(int64_t));
  auto out = at::empty({n * len}, at::TensorOptions().dtype(at::kLong));
  std::vector<at::Tensor> outs;
  outs.reserve(n);
  for (int i = 0; i < n; ++i) outs.push_back(out.narrow(0, i * len, len));
  std::vector<std::vector<at::Tensor>> outputs{outs};
  std::vector<at::Tensor> inputs{in};
  backend->allgather(outputs, inputs)->wait();
  std::vector<int64_t> res(n * len);
  std::memcpy(res.data(), out.data_ptr(), n * len * sizeof(int64_t));
  return res;
}

std::vector<int64_t> host_broadcast_int64(const std::string& group_name,
                                          const std::vector<int64_t>& values,
                                          int root, int64_t fixed_len) {
  auto backend = gloo_backend(group_name);
  int64_t len = fixed_len;
  if (len < 0) {
    auto lt = at::full({1}, (int64_t)values.size(),
                       at::TensorOptions().dtype(at::kLong));
    std::vector<at::Tensor> ts{lt};
    c10d::BroadcastOptions opts;
    opts.rootRank = root;
    backend->broadcast(ts, opts)->wait();
    len = lt.item<int64_t>();
  }
  auto t = at::zeros({len}, at::TensorOptions().dtype(at::kLong));
  if (backend->getRank() == root) {
    TORCH_CHECK((int64_t)values.size() == len);
    std::memcpy(t.data_ptr(), values.data(), len * sizeof(int64_t));
  }
  std::vector<at::Tensor> ts{t};
  c10d::BroadcastOptions opts;
  opts.rootRank = root;
  backend->broadcast(ts, opts)->wait();
  std::vector<int64_t> res(len);
  std::memcpy(res.data(), t.data_ptr(), len * sizeof(int64_t));
  return res;
}

} // namespace m4a
