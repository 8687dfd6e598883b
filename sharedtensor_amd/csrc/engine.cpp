#include "engine.h"

#include <arpa/inet.h>
#include <fcntl.h>
#include <netdb.h>
#include <netinet/tcp.h>
#include <sys/socket.h>
#include <unistd.h>

#include <chrono>
#include <cmath>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <functional>
#include <stdexcept>

#include "codec_cpu.h"
#include "rccl_transport.h"

namespace shamd {

using Clock = std::chrono::steady_clock;

#define HIP_TRY(expr)                                                        \
  do {                                                                       \
    hipError_t _e = (expr);                                                  \
    if (_e != hipSuccess)                                                    \
      throw std::runtime_error(std::string("HIP error: ") +                  \
                               hipGetErrorString(_e) + " @" __FILE__ ":" +   \
                               std::to_string(__LINE__));                    \
  } while (0)

// Capture a stream-op sequence into an executable hipGraph (one replay per
// round instead of several launches).  ThreadLocal mode: other link threads'
// streams keep running during capture.
template <typename F>
static hipGraphExec_t capture_seq(hipStream_t s, F&& body) {
  hipGraph_t g = nullptr;
  HIP_TRY(hipStreamBeginCapture(s, hipStreamCaptureModeThreadLocal));
  body();
  HIP_TRY(hipStreamEndCapture(s, &g));
  hipGraphExec_t e = nullptr;
  HIP_TRY(hipGraphInstantiate(&e, g, nullptr, nullptr, 0));
  (void)hipGraphDestroy(g);
  return e;
}

// ------------------------------------------------------------ I/O helpers
// Loop partial reads/writes and retry EINTR (reference read_or_die/
// write_or_die, sharedtensor.c:53-87) — but surface failure instead of
// exiting the process.
static bool io_read(int fd, void* buf, size_t count) {
  auto* p = static_cast<uint8_t*>(buf);
  while (count) {
    ssize_t r = ::read(fd, p, count);
    if (r == 0) return false;  // peer closed
    if (r < 0) {
      if (errno == EINTR) continue;
      return false;
    }
    count -= static_cast<size_t>(r);
    p += r;
  }
  return true;
}

static bool io_write(int fd, const void* buf, size_t count) {
  auto* p = static_cast<const uint8_t*>(buf);
  while (count) {
    ssize_t r = ::write(fd, p, count);
    if (r < 0) {
      if (errno == EINTR) continue;
      return false;
    }
    count -= static_cast<size_t>(r);
    p += r;
  }
  return true;
}

static void set_sockopts(int fd) {
  int yes = 1;
  setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &yes, sizeof(yes));
  // on the up-connection this is what later allows binding our listener to
  // the same local (ip, port) — the reference's self-addressing trick
  // (sharedtensor.c:264,292-316).  SO_REUSEPORT additionally lets a REJOIN
  // bind its outgoing socket to the address our listener already occupies
  // (REUSEADDR alone does not permit bind beside a LISTEN socket on Linux).
  setsockopt(fd, SOL_SOCKET, SO_REUSEADDR, &yes, sizeof(yes));
  setsockopt(fd, SOL_SOCKET, SO_REUSEPORT, &yes, sizeof(yes));
  int buf = 8 << 20;
  setsockopt(fd, SOL_SOCKET, SO_SNDBUF, &buf, sizeof(buf));
  setsockopt(fd, SOL_SOCKET, SO_RCVBUF, &buf, sizeof(buf));
}

static bool resolve_ipv4(const std::string& host, int port, sockaddr_in* out) {
  std::memset(out, 0, sizeof(*out));
  out->sin_family = AF_INET;
  out->sin_port = htons(static_cast<uint16_t>(port));
  addrinfo hints{};
  hints.ai_family = AF_INET;
  hints.ai_socktype = SOCK_STREAM;
  addrinfo* res = nullptr;
  if (getaddrinfo(host.c_str(), nullptr, &hints, &res) != 0 || !res) return false;
  out->sin_addr = reinterpret_cast<sockaddr_in*>(res->ai_addr)->sin_addr;
  freeaddrinfo(res);
  return true;
}

static std::string addr_str(const sockaddr_in& a) {
  char b[64];
  inet_ntop(AF_INET, &a.sin_addr, b, sizeof(b));
  return std::string(b) + ":" + std::to_string(ntohs(a.sin_port));
}

static uint64_t compute_hostid() {
  // same-host detection for the RCCL/xGMI upgrade
  uint64_t h = 1469598103934665603ull;
  auto mix = [&h](const char* s, size_t n) {
    for (size_t i = 0; i < n; ++i) {
      h ^= static_cast<uint8_t>(s[i]);
      h *= 1099511628211ull;
    }
  };
  FILE* f = std::fopen("/etc/machine-id", "r");
  char buf[128] = {0};
  if (f) {
    size_t n = std::fread(buf, 1, sizeof(buf), f);
    std::fclose(f);
    mix(buf, n);
  }
  char hn[256] = {0};
  gethostname(hn, sizeof(hn) - 1);
  mix(hn, std::strlen(hn));
  return h;
}

// ----------------------------------------------------------------- Engine

int64_t Engine::scales_area(const Config& cfg) {
  return align8(4 * static_cast<int64_t>(cfg.sizes.size()));
}

int64_t Engine::msg_bytes(const Config& cfg) {
  int64_t p = 0;
  for (int64_t s : cfg.sizes) p += payload_bytes(cfg.codec, s);
  return scales_area(cfg) + p;
}

Engine::Engine(Config cfg) : cfg_(std::move(cfg)) {
  if (cfg_.sizes.empty()) throw std::runtime_error("sizes must be non-empty");
  T_ = static_cast<int>(cfg_.sizes.size());
  offs_.resize(T_ + 1);
  poffs_.resize(T_ + 1);
  offs_[0] = poffs_[0] = 0;
  for (int t = 0; t < T_; ++t) {
    offs_[t + 1] = offs_[t] + cfg_.sizes[t];
    poffs_[t + 1] = poffs_[t] + pad64(cfg_.sizes[t]);
  }
  n_ = offs_[T_];
  pe_ = poffs_[T_];
  SA_ = scales_area(cfg_);
  P_ = msg_bytes(cfg_) - SA_;
  if (cfg_.delta_bf16 && cfg_.device < 0)
    throw std::runtime_error("delta_bf16 requires a GPU engine");
  hostid_ = compute_hostid();
  for (int i = 0; i < 3; ++i) links_[i].idx = i;
}

Engine::~Engine() {
  try {
    close();
  } catch (...) {
  }
}

void Engine::set_values(uintptr_t p) { values_ = reinterpret_cast<float*>(p); }

void Engine::set_link_buffers(int link, uintptr_t delta, uintptr_t send_buf,
                              uintptr_t recv_buf, uintptr_t send_pin,
                              uintptr_t recv_pin) {
  Link& lk = links_[link];
  lk.delta = reinterpret_cast<void*>(delta);
  lk.send_buf = reinterpret_cast<uint8_t*>(send_buf);
  lk.recv_buf = reinterpret_cast<uint8_t*>(recv_buf);
  lk.send_pin = reinterpret_cast<uint8_t*>(send_pin);
  lk.recv_pin = reinterpret_cast<uint8_t*>(recv_pin);
  lk.provisioned = lk.delta != nullptr;
}

void Engine::set_error(const std::string& e) {
  std::lock_guard<std::mutex> g(err_m_);
  last_error_ = e;
}

std::string Engine::last_error() {
  std::lock_guard<std::mutex> g(err_m_);
  return last_error_;
}

void Engine::push_scale(bool sent, float s) {
  std::lock_guard<std::mutex> g(ring_m_);
  auto& r = sent ? ring_sent_ : ring_recv_;
  r.push_back(s);
  if (r.size() > 2048) r.pop_front();
}

std::vector<float> Engine::recent_scales_sent() {
  std::lock_guard<std::mutex> g(ring_m_);
  return {ring_sent_.begin(), ring_sent_.end()};
}

std::vector<float> Engine::recent_scales_recv() {
  std::lock_guard<std::mutex> g(ring_m_);
  return {ring_recv_.begin(), ring_recv_.end()};
}

// ------------------------------------------------------------------- GPU

void Engine::init_gpu() {
  if (!gpu()) return;
  HIP_TRY(hipSetDevice(cfg_.device));
  HIP_TRY(hipMalloc(&dtb_.offs, sizeof(int64_t) * (T_ + 1) * 2));
  dtb_.poffs = dtb_.offs + (T_ + 1);
  HIP_TRY(hipMemcpy(dtb_.offs, offs_.data(), sizeof(int64_t) * (T_ + 1),
                    hipMemcpyHostToDevice));
  HIP_TRY(hipMemcpy(dtb_.poffs, poffs_.data(), sizeof(int64_t) * (T_ + 1),
                    hipMemcpyHostToDevice));
  dtb_.T = T_;
  dtb_.n = n_;
  dtb_.pe = pe_;
  for (auto& lk : links_) {
    if (!lk.provisioned) continue;
    HIP_TRY(hipMalloc(&lk.reduce_buf, 8 * T_));
    HIP_TRY(hipStreamCreateWithFlags(&lk.s_send, hipStreamNonBlocking));
    HIP_TRY(hipStreamCreateWithFlags(&lk.s_recv, hipStreamNonBlocking));
  }
}

static void destroy_link_graphs(Link& lk) {
  for (hipGraphExec_t* e : {&lk.g_scale_lagged, &lk.g_scale_exact, &lk.g_quant,
                            &lk.g_apply}) {
    if (*e) (void)hipGraphExecDestroy(*e), *e = nullptr;
  }
}

void Engine::free_gpu() {
  if (!gpu()) return;
  for (auto& lk : links_) {
    destroy_link_graphs(lk);
    if (lk.s_send) (void)hipStreamDestroy(lk.s_send), lk.s_send = nullptr;
    if (lk.s_recv) (void)hipStreamDestroy(lk.s_recv), lk.s_recv = nullptr;
    if (lk.reduce_buf) (void)hipFree(lk.reduce_buf), lk.reduce_buf = nullptr;
  }
  if (dtb_.offs) (void)hipFree(dtb_.offs), dtb_.offs = nullptr;
}

// ------------------------------------------------------------------ start

void Engine::start() {
  if (started_) throw std::runtime_error("engine already started");
  if (!values_) throw std::runtime_error("values buffer not set");
  init_gpu();
  join_tree();
  started_ = true;
  listen_thread_ = std::thread([this] { listen_loop(); });
}

bool Engine::try_connect(const sockaddr_in& addr, int& out_fd,
                         const sockaddr_in* bind_local) {
  int fd = ::socket(AF_INET, SOCK_STREAM, 0);
  if (fd < 0) return false;
  set_sockopts(fd);
  if (bind_local &&
      ::bind(fd, reinterpret_cast<const sockaddr*>(bind_local),
             sizeof(*bind_local)) < 0) {
    ::close(fd);
    return false;
  }
  if (::connect(fd, reinterpret_cast<const sockaddr*>(&addr), sizeof(addr)) < 0) {
    ::close(fd);
    return false;
  }
  out_fd = fd;
  return true;
}

void Engine::join_tree() {
  if (!resolve_ipv4(cfg_.host, cfg_.port, &root_addr_))
    throw std::runtime_error("cannot resolve host " + cfg_.host);

  auto deadline = Clock::now() + std::chrono::duration_cast<Clock::duration>(
                                     std::chrono::duration<double>(cfg_.join_timeout_s));

  sockaddr_in target = root_addr_;
  bool explicit_mode = !cfg_.explicit_parent.empty();
  if (explicit_mode) {
    auto pos = cfg_.explicit_parent.rfind(':');
    if (pos == std::string::npos)
      throw std::runtime_error("explicit_parent must be ip:port");
    if (!resolve_ipv4(cfg_.explicit_parent.substr(0, pos),
                      std::stoi(cfg_.explicit_parent.substr(pos + 1)), &target))
      throw std::runtime_error("cannot resolve explicit_parent");
  }

  int hops_since_root = 0;
  while (true) {
    if (Clock::now() > deadline) throw std::runtime_error("join timed out");
    int fd = -1;
    if (!try_connect(target, fd)) {
      if (!explicit_mode && hops_since_root == 0) {
        // nobody is listening at the rendezvous: we are the master
        // (sharedtensor.c:271-277,318-322).  If another process won the race
        // to bind, retry the walk instead of dying.
        try {
          become_master();
          return;
        } catch (const std::exception&) {
          std::this_thread::sleep_for(std::chrono::milliseconds(100));
          continue;
        }
      }
      // a mid-walk node (or the explicit parent) is not up yet: back off
      std::this_thread::sleep_for(std::chrono::milliseconds(200));
      if (!explicit_mode) {
        target = root_addr_;
        hops_since_root = 0;
      }
      continue;
    }
    Hello h{};
    h.magic = MAGIC;
    h.version = PROTO_VERSION;
    h.flags = static_cast<uint16_t>(
        (gpu() ? HELLO_HAS_GPU : 0) |
        (gpu() && cfg_.use_rccl && rccl_failures_.load() < 2 ? HELLO_WANT_RCCL
                                                             : 0));
    h.n = static_cast<uint64_t>(n_);
    h.ntensors = static_cast<uint32_t>(T_);
    h.codec = static_cast<uint32_t>(cfg_.codec);
    h.hostid = hostid_;
    h.device = cfg_.device;
    uint8_t reply = 0;
    if (!io_write(fd, &h, sizeof(h)) || !io_read(fd, &reply, 1)) {
      ::close(fd);
      std::this_thread::sleep_for(std::chrono::milliseconds(100));
      if (!explicit_mode) { target = root_addr_; hops_since_root = 0; }
      continue;
    }
    if (reply == 'N') {
      // walk down the tree (sharedtensor.c:298-300)
      uint8_t buf[6];
      if (!io_read(fd, buf, 6)) {
        ::close(fd);
        target = root_addr_;
        hops_since_root = 0;
        continue;
      }
      ::close(fd);
      std::memcpy(&target.sin_addr.s_addr, buf, 4);
      std::memcpy(&target.sin_port, buf + 4, 2);
      hops_since_root++;
      continue;
    }
    if (reply != 'Y') {
      ::close(fd);
      throw std::runtime_error("protocol error during join (reply byte " +
                               std::to_string(int(reply)) + ")");
    }
    try {
      handshake_as_child(fd);
      return;
    } catch (const std::exception& e) {
      // transient handshake failure (e.g. RCCL init): close, back off and
      // retry the walk — after 2 RCCL failures the hello stops asking
      ::close(fd);
      if (listen_fd_ >= 0) {  // listener was bound in the failed attempt
        ::close(listen_fd_);
        listen_fd_ = -1;
      }
      Link& up0 = links_[LK_UP];
      up0.fd = -1;
      if (up0.rccl_link) {  // half-made upgrade: tear down before retrying
        rccl_destroy(static_cast<RcclLink*>(up0.rccl_link));
        up0.rccl_link = nullptr;
        up0.rccl = false;
      }
      // a partial snapshot may have been applied; reset to the pristine
      // pre-join state (all-zero) so the retry cannot double-count
      zero_buf(values_, n_);
      for (auto& l0 : links_)
        if (l0.provisioned) zero_delta(l0.delta);
      set_error(std::string("join handshake failed, retrying: ") + e.what());
      std::this_thread::sleep_for(std::chrono::milliseconds(300));
      target = explicit_mode ? target : root_addr_;
      hops_since_root = 0;
      continue;
    }
  }
}

void Engine::become_master() {
  bind_listen(root_addr_, /*shared=*/false);
  is_master_ = true;
  std::fprintf(stderr,
               "[sharedtensor_amd] master tensor at %s (n=%lld, %d tensor%s)\n",
               addr_str(root_addr_).c_str(), static_cast<long long>(n_), T_,
               T_ == 1 ? "" : "s");
}

void Engine::handshake_as_child(int fd, bool rejoin) {
  AcceptHello ah{};
  if (!io_read(fd, &ah, sizeof(ah)))
    throw std::runtime_error("parent hung up during handshake");
  if (ah.version != PROTO_VERSION || ah.n != static_cast<uint64_t>(n_) ||
      ah.ntensors != static_cast<uint32_t>(T_))
    throw std::runtime_error("handshake mismatch: parent has different tensor shape/version");
  if (ah.codec != static_cast<uint32_t>(cfg_.codec))
    throw std::runtime_error("handshake mismatch: parent uses a different codec");

  Link& up = links_[LK_UP];
  if (!up.provisioned)
    throw std::runtime_error("up link not provisioned but joining as child");
  up.fd = fd;
  socklen_t alen = sizeof(up.peer);
  getpeername(fd, reinterpret_cast<sockaddr*>(&up.peer), &alen);
  up.peer_desc = addr_str(up.peer);

  if (!rejoin) {
    // bind our listener to the local address of the up socket so our
    // parent's view of us (getpeername) is also our listen address — the
    // reference's self-addressing trick (sharedtensor.c:292-316)
    sockaddr_in self{};
    socklen_t slen = sizeof(self);
    getsockname(fd, reinterpret_cast<sockaddr*>(&self), &slen);
    if (cfg_.listen_port > 0)
      self.sin_port = htons(static_cast<uint16_t>(cfg_.listen_port));
    bind_listen(self);
  }

  if (ah.flags & ACC_RCCL) {
    uint8_t ids[2 * RCCL_ID_BYTES];
    if (!io_read(fd, ids, sizeof(ids)))
      throw std::runtime_error("failed to read RCCL ids from parent");
    try {
      rccl_upgrade(up, ids, /*is_parent=*/false);
    } catch (...) {
      rccl_failures_++;  // retried joins fall back to TCP after 2 failures
      throw;
    }
  }
  if (rejoin && (ah.flags & ACC_SNAPSHOT) && rejoin_preserve_) {
    // Subtree-preserving reconciliation (cfg.preserve_subtree): the
    // children stayed ATTACHED and keep gossiping.  They must end up
    // applying exactly corr = S + R - V_old on top of whatever they had:
    //   phase 1 (exclusive): R := up.delta (scratch fp32); up.delta := 0;
    //     corr := -values; values := 0.  Child residuals untouched — their
    //     U_c (what the child still lacks vs OUR values) stays exact.
    //   phase 2 (lock-free): snapshot chunks add into values AND corr,
    //     never into child residuals; concurrent user updates u land in
    //     values + every provisioned residual (including the children's)
    //     and never touch corr, so corr stays exactly S - V_old.
    //   phase 3 (exclusive): values += R; up.delta += R; corr += R;
    //     child.delta += corr for every ACTIVE child.
    //   End: values = S + R + u; each live child's residual = U_c + u +
    //     (S + R - V_old) == values - child_view, which drains as ordinary
    //     gossip — the child never disconnects and never re-snapshots.
    // On a phase-2 failure, phase 3 still runs with the partial S: the
    // state is then self-consistent (a truncated snapshot) and the retry
    // re-captures from it.
    float* corr = gpu() ? sub_corr_dev_ : sub_corr_host_.data();
    float* tmpR = nullptr;
    std::vector<float> tmpR_host;
    bool captured = false;
    {
      std::unique_lock<std::shared_mutex> ug(user_m_);
      if (gpu()) {
        HIP_TRY(hipSetDevice(cfg_.device));
        HIP_TRY(hipDeviceSynchronize());
        if (hipMalloc(&tmpR, n_ * 4) == hipSuccess) {
          HIP_TRY(hipMemsetAsync(tmpR, 0, n_ * 4, up.s_recv));
          hip_add_delta_scatter(up.delta, cfg_.delta_bf16, n_, tmpR, nullptr,
                                nullptr, up.s_recv);
          HIP_TRY(hipMemsetAsync(corr, 0, n_ * 4, up.s_recv));
          hip_add_scatter(values_, n_, -1.0f, corr, nullptr, nullptr,
                          nullptr, false, up.s_recv);
          HIP_TRY(hipStreamSynchronize(up.s_recv));
          captured = true;
        } else {
          (void)hipGetLastError();
          tmpR = nullptr;
        }
      } else {
        tmpR_host.resize(static_cast<size_t>(n_));
        cpu_pfor(n_, [&](int64_t lo, int64_t hi) {
          for (int64_t i = lo; i < hi; ++i) {
            tmpR_host[i] = atomic_load_f32(fdelta(up.delta) + i);
            corr[i] = -atomic_load_f32(values_ + i);
          }
        });
        captured = true;
      }
      if (captured) {
        zero_delta(up.delta);
        zero_buf(values_, n_);
      }
    }
    auto phase3 = [&] {
      std::unique_lock<std::shared_mutex> ug(user_m_);
      if (gpu()) {
        HIP_TRY(hipDeviceSynchronize());
        hip_add_scatter(tmpR, n_, 1.0f, values_, up.delta, nullptr, nullptr,
                        cfg_.delta_bf16, up.s_recv);
        hip_add_scatter(tmpR, n_, 1.0f, corr, nullptr, nullptr, nullptr,
                        false, up.s_recv);
        for (int i : {LK_LEFT, LK_RIGHT}) {
          Link& c = links_[i];
          if (c.state.load() == L_ACTIVE)
            hip_add_scatter(corr, n_, 1.0f, nullptr, c.delta, nullptr,
                            nullptr, cfg_.delta_bf16, up.s_recv);
        }
        HIP_TRY(hipStreamSynchronize(up.s_recv));
        (void)hipFree(tmpR);
        tmpR = nullptr;
      } else {
        float* cl = links_[LK_LEFT].state.load() == L_ACTIVE
                        ? fdelta(links_[LK_LEFT].delta) : nullptr;
        float* cr = links_[LK_RIGHT].state.load() == L_ACTIVE
                        ? fdelta(links_[LK_RIGHT].delta) : nullptr;
        cpu_pfor(n_, [&](int64_t lo, int64_t hi) {
          for (int64_t i = lo; i < hi; ++i) {
            float r = tmpR_host[i];
            if (r != 0.0f) {
              atomic_add_f32(values_ + i, r);
              atomic_add_f32(fdelta(up.delta) + i, r);
              corr[i] += r;
            }
            float cv = corr[i];
            if (cv != 0.0f) {
              if (cl) atomic_add_f32(cl + i, cv);
              if (cr) atomic_add_f32(cr + i, cv);
            }
          }
        });
      }
      notify_all_dirty();  // children have fresh residual to drain
    };
    if (captured) {
      try {
        recv_snapshot(fd, corr);
      } catch (...) {
        phase3();  // consistent truncated-snapshot state for the retry
        throw;
      }
      phase3();
    } else {
      // no scratch for R: single exclusive section, R stays in up.delta
      std::unique_lock<std::shared_mutex> ug(user_m_);
      HIP_TRY(hipDeviceSynchronize());
      HIP_TRY(hipMemsetAsync(corr, 0, n_ * 4, up.s_recv));
      hip_add_scatter(values_, n_, -1.0f, corr, nullptr, nullptr, nullptr,
                      false, up.s_recv);
      HIP_TRY(hipStreamSynchronize(up.s_recv));
      zero_buf(values_, n_);
      recv_snapshot(fd, corr);
      hip_add_delta_scatter(up.delta, cfg_.delta_bf16, n_, values_, nullptr,
                            nullptr, up.s_recv);
      // corr += R directly from the (fp32-or-bf16) residual
      hip_add_delta_scatter(up.delta, cfg_.delta_bf16, n_, corr, nullptr,
                            nullptr, up.s_recv);
      for (int i : {LK_LEFT, LK_RIGHT}) {
        Link& c = links_[i];
        if (c.state.load() == L_ACTIVE)
          hip_add_scatter(corr, n_, 1.0f, nullptr, c.delta, nullptr, nullptr,
                          cfg_.delta_bf16, up.s_recv);
      }
      HIP_TRY(hipStreamSynchronize(up.s_recv));
      notify_all_dirty();
    }
  } else if (rejoin && (ah.flags & ACC_SNAPSHOT)) {
    // Reconciliation on rejoin: V := S + R, where S is the new parent's
    // snapshot and R = our unsent up-residual at the instant reconciliation
    // begins.  A concurrent user update u (add_from/fused_sgd) must land
    // exactly once: R is captured into a scratch
    // fp32 buffer and {values, every provisioned delta} reset under a brief
    // exclusive user-op lock; the snapshot then streams WITHOUT the lock
    // (concurrent u adds consistently into values + all deltas); finally R
    // is re-added into values + all deltas under the lock again.  End
    // state: values = S + R + u, and every unconnected slot == values (the
    // invariant the reference seeds at sharedtensor.c:379-381).  No update
    // is ever lost: u landed after link-down is inside R via up.delta.
    void* fwd0 = links_[LK_LEFT].provisioned ? links_[LK_LEFT].delta : nullptr;
    void* fwd1 = links_[LK_RIGHT].provisioned ? links_[LK_RIGHT].delta : nullptr;
    float* tmp_dev = nullptr;
    std::vector<float> tmp_host;
    bool captured = false;
    {
      std::unique_lock<std::shared_mutex> ug(user_m_);
      if (gpu()) {
        HIP_TRY(hipSetDevice(cfg_.device));
        HIP_TRY(hipDeviceSynchronize());  // drain in-flight user kernels
        if (hipMalloc(&tmp_dev, n_ * 4) == hipSuccess) {
          HIP_TRY(hipMemsetAsync(tmp_dev, 0, n_ * 4, up.s_recv));
          // tmp (fp32) := up.delta (fp32 or bf16 residual)
          hip_add_delta_scatter(up.delta, cfg_.delta_bf16, n_, tmp_dev,
                                nullptr, nullptr, up.s_recv);
          HIP_TRY(hipStreamSynchronize(up.s_recv));
          captured = true;
        } else {
          (void)hipGetLastError();  // clear; fall back to full-lock mode
          tmp_dev = nullptr;
        }
      } else {
        tmp_host.resize(static_cast<size_t>(n_));
        cpu_pfor(n_, [&](int64_t lo, int64_t hi) {
          for (int64_t i = lo; i < hi; ++i)
            tmp_host[i] = atomic_load_f32(fdelta(up.delta) + i);
        });
        captured = true;
      }
      if (captured) {
        zero_delta(up.delta);
        if (fwd0) zero_delta(fwd0);
        if (fwd1) zero_delta(fwd1);
        zero_buf(values_, n_);
      }
    }
    auto readd = [&] {
      std::unique_lock<std::shared_mutex> ug(user_m_);
      if (gpu()) {
        HIP_TRY(hipDeviceSynchronize());
        hip_add_scatter(tmp_dev, n_, 1.0f, values_, up.delta, fwd0, fwd1,
                        cfg_.delta_bf16, up.s_recv);
        HIP_TRY(hipStreamSynchronize(up.s_recv));
        (void)hipFree(tmp_dev);
        tmp_dev = nullptr;
      } else {
        cpu_pfor(n_, [&](int64_t lo, int64_t hi) {
          for (int64_t i = lo; i < hi; ++i) {
            float v = tmp_host[i];
            if (v == 0.0f) continue;
            atomic_add_f32(values_ + i, v);
            atomic_add_f32(fdelta(up.delta) + i, v);
            if (fwd0) atomic_add_f32(fdelta(fwd0) + i, v);
            if (fwd1) atomic_add_f32(fdelta(fwd1) + i, v);
          }
        });
      }
    };
    if (captured) {
      try {
        recv_snapshot(fd);  // += S into values + child slots (no user lock)
      } catch (...) {
        readd();  // restore a consistent state before the retry resets it
        throw;
      }
      readd();
    } else {
      // scratch allocation failed (100 GB-scale tensor near HBM capacity):
      // run the whole reconciliation under the exclusive lock instead —
      // user ops block for the snapshot duration, but stay exactly-once
      std::unique_lock<std::shared_mutex> ug(user_m_);
      HIP_TRY(hipDeviceSynchronize());
      if (fwd0) zero_delta(fwd0);
      if (fwd1) zero_delta(fwd1);
      zero_buf(values_, n_);
      recv_snapshot(fd);
      hip_add_delta_scatter(up.delta, cfg_.delta_bf16, n_, values_, fwd0,
                            fwd1, up.s_recv);
      HIP_TRY(hipStreamSynchronize(up.s_recv));
    }
  } else if (ah.flags & ACC_SNAPSHOT) {
    recv_snapshot(fd);
  }

  up.state.store(L_ACTIVE);
  spawn_link_threads(up);
}

void Engine::zero_buf(float* p, int64_t n) {
  if (gpu()) {
    HIP_TRY(hipMemset(p, 0, n * 4));
  } else {
    std::memset(p, 0, n * 4);
  }
}

void Engine::zero_delta(void* p) {
  int64_t bytes = n_ * (cfg_.delta_bf16 ? 2 : 4);
  if (gpu()) {
    HIP_TRY(hipMemset(p, 0, bytes));
  } else {
    std::memset(p, 0, bytes);
  }
}

void* Engine::doff(void* delta, int64_t off) const {
  return static_cast<char*>(delta) + off * (cfg_.delta_bf16 ? 2 : 4);
}

float* Engine::fdelta(void* p) const {
  // CPU engines always keep fp32 residuals (validated in the constructor)
  return static_cast<float*>(p);
}

void Engine::drop_children() {
  std::lock_guard<std::mutex> g(slots_m_);
  for (int i : {LK_LEFT, LK_RIGHT}) {
    Link& lk = links_[i];
    // settle an in-flight join first: break its I/O and wait for the join
    // thread to land the slot in ACTIVE or FREE before demoting
    if (lk.state.load() == L_JOINING && lk.fd >= 0)
      ::shutdown(lk.fd, SHUT_RDWR);
    if (lk.t_join.joinable()) lk.t_join.join();
    if (lk.state.load() == L_ACTIVE && lk.fd >= 0) {
      PacketHeader bye{};
      bye.type = PKT_CLOSE;
      bye.ntensors = static_cast<uint32_t>(T_);
      std::lock_guard<std::mutex> wg(lk.wm);
      io_write(lk.fd, &bye, 8);
    }
    link_down(lk, "demoted for rejoin", true);
    if (lk.t_send.joinable()) lk.t_send.join();
    if (lk.t_recv.joinable()) lk.t_recv.join();
    if (lk.t_ctrl.joinable()) lk.t_ctrl.join();
    if (lk.fd >= 0) ::close(lk.fd), lk.fd = -1;
    if (lk.rccl_link) {
      rccl_destroy(static_cast<RcclLink*>(lk.rccl_link));
      lk.rccl_link = nullptr;
      lk.rccl = false;
    }
    lk.abort.store(false);
    lk.error.clear();
    if (gpu()) destroy_link_graphs(lk);  // transport may change on reuse
    if (lk.provisioned) zero_delta(lk.delta);
    lk.state.store(L_FREE);
  }
}

bool Engine::failover_master() {
  // the root is gone: race to take over the rendezvous address; loser
  // restores its own listener and rejoins the winner
  sockaddr_in old_addr = listen_addr_;
  if (listen_fd_ >= 0) ::shutdown(listen_fd_, SHUT_RDWR);
  if (listen_thread_.joinable()) listen_thread_.join();
  if (listen_fd_ >= 0) ::close(listen_fd_), listen_fd_ = -1;
  try {
    bind_listen(root_addr_, /*shared=*/false);
    is_master_ = true;
  } catch (const std::exception&) {
    try {
      bind_listen(old_addr);  // reclaim our previous address
      listen_thread_ = std::thread([this] { listen_loop(); });
    } catch (const std::exception& e2) {
      set_error(std::string("failover: lost listener: ") + e2.what());
    }
    return false;
  }
  listen_thread_ = std::thread([this] { listen_loop(); });
  std::fprintf(stderr, "[sharedtensor_amd] failover: now master at %s\n",
               addr_str(root_addr_).c_str());
  return true;
}

// Rejoin after losing the up link (the reference's acknowledged missing
// feature, README.md:33).  Demote-then-rejoin: children are released (they
// rejoin through the root themselves, which keeps the topology acyclic),
// then this node re-enters the tree carrying its unsent residual.
void Engine::reconnect_loop() try {
  Link& up = links_[LK_UP];
  if (up.t_send.joinable()) up.t_send.join();
  if (up.t_recv.joinable()) up.t_recv.join();
  if (up.t_ctrl.joinable()) up.t_ctrl.join();
  if (up.fd >= 0) ::close(up.fd), up.fd = -1;
  if (up.rccl_link) {
    rccl_destroy(static_cast<RcclLink*>(up.rccl_link));
    up.rccl_link = nullptr;
    up.rccl = false;
  }
  if (gpu()) destroy_link_graphs(up);
  up.abort.store(false);
  up.state.store(L_FREE);

  // Subtree-preserving mode: keep live children attached through the
  // rejoin instead of demoting them (they would otherwise re-join through
  // the root one by one, each with a fresh snapshot).  Needs an extra
  // fp32[n] scratch for the correction accumulator; if that cannot be
  // had, fall back to the demote path.
  rejoin_preserve_ = false;
  if (cfg_.preserve_subtree && cfg_.snapshot_join &&
      (links_[LK_LEFT].state.load() == L_ACTIVE ||
       links_[LK_RIGHT].state.load() == L_ACTIVE)) {
    if (gpu()) {
      HIP_TRY(hipSetDevice(cfg_.device));
      if (hipMalloc(&sub_corr_dev_, n_ * 4) == hipSuccess) {
        rejoin_preserve_ = true;
      } else {
        (void)hipGetLastError();
        sub_corr_dev_ = nullptr;
      }
    } else {
      sub_corr_host_.assign(static_cast<size_t>(n_), 0.0f);
      rejoin_preserve_ = true;
    }
  }
  if (!rejoin_preserve_) drop_children();

  bool explicit_mode = !cfg_.explicit_parent.empty();
  sockaddr_in target = root_addr_;
  if (explicit_mode) {
    auto pos = cfg_.explicit_parent.rfind(':');
    resolve_ipv4(cfg_.explicit_parent.substr(0, pos),
                 std::stoi(cfg_.explicit_parent.substr(pos + 1)), &target);
  }
  auto deadline = Clock::now() + std::chrono::duration_cast<Clock::duration>(
      std::chrono::duration<double>(cfg_.join_timeout_s));
  int hops = 0;
  while (!closing_) {
    if (Clock::now() > deadline) {
      set_error("rejoin timed out");
      break;
    }
    int fd = -1;
    sockaddr_in local = listen_addr_;
    if (!try_connect(target, fd, &local)) {
      if (!explicit_mode && hops == 0 && !closing_) {
        if (failover_master()) {
          // Restore the unconnected-slot invariant (slot delta == values,
          // sharedtensor.c:379-381 semantics): future children must receive
          // the full inherited state.  slot := values (zero + add) under
          // the exclusive user-op lock — a plain add would double-count any
          // update that landed in the slot since drop_children zeroed it
          // (ADVICE round 1, medium).  Slots that stayed ACTIVE (subtree-
          // preserving mode) keep their live residual untouched: this
          // node's values never changed, so their deltas are still exact.
          void* fwd0 = (links_[LK_LEFT].provisioned &&
                        links_[LK_LEFT].state.load() != L_ACTIVE)
                           ? links_[LK_LEFT].delta : nullptr;
          void* fwd1 = (links_[LK_RIGHT].provisioned &&
                        links_[LK_RIGHT].state.load() != L_ACTIVE)
                           ? links_[LK_RIGHT].delta : nullptr;
          if (fwd0 || fwd1) {
            std::unique_lock<std::shared_mutex> ug(user_m_);
            if (gpu()) {
              HIP_TRY(hipSetDevice(cfg_.device));
              HIP_TRY(hipDeviceSynchronize());  // drain in-flight user kernels
            }
            if (fwd0) zero_delta(fwd0);
            if (fwd1) zero_delta(fwd1);
            if (gpu()) {
              hip_add_scatter(values_, n_, 1.0f, nullptr, fwd0, fwd1, nullptr,
                              cfg_.delta_bf16, nullptr);
              HIP_TRY(hipStreamSynchronize(nullptr));
            } else {
              cpu_pfor(n_, [&](int64_t lo, int64_t hi) {
                for (int64_t i = lo; i < hi; ++i) {
                  float v = atomic_load_f32(values_ + i);
                  if (v == 0.0f) continue;
                  if (fwd0) atomic_add_f32(fdelta(fwd0) + i, v);
                  if (fwd1) atomic_add_f32(fdelta(fwd1) + i, v);
                }
              });
            }
          }
          break;
        }
      }
      std::this_thread::sleep_for(std::chrono::milliseconds(250));
      target = explicit_mode ? target : root_addr_;
      hops = 0;
      continue;
    }
    Hello h{};
    h.magic = MAGIC;
    h.version = PROTO_VERSION;
    h.flags = static_cast<uint16_t>(
        (gpu() ? HELLO_HAS_GPU : 0) |
        (gpu() && cfg_.use_rccl && rccl_failures_.load() < 2 ? HELLO_WANT_RCCL
                                                             : 0));
    h.n = static_cast<uint64_t>(n_);
    h.ntensors = static_cast<uint32_t>(T_);
    h.codec = static_cast<uint32_t>(cfg_.codec);
    h.hostid = hostid_;
    h.device = cfg_.device;
    uint8_t reply = 0;
    if (!io_write(fd, &h, sizeof(h)) || !io_read(fd, &reply, 1)) {
      ::close(fd);
      std::this_thread::sleep_for(std::chrono::milliseconds(100));
      target = explicit_mode ? target : root_addr_;
      hops = 0;
      continue;
    }
    if (reply == 'N') {
      uint8_t buf[6];
      if (io_read(fd, buf, 6)) {
        std::memcpy(&target.sin_addr.s_addr, buf, 4);
        std::memcpy(&target.sin_port, buf + 4, 2);
        hops++;
      } else {
        target = explicit_mode ? target : root_addr_;
        hops = 0;
      }
      ::close(fd);
      continue;
    }
    if (reply != 'Y') {
      ::close(fd);
      std::this_thread::sleep_for(std::chrono::milliseconds(250));
      continue;
    }
    try {
      handshake_as_child(fd, /*rejoin=*/true);
      reconnects_++;
      set_error("");
      break;
    } catch (const std::exception& e) {
      ::close(fd);
      up.fd = -1;
      if (up.rccl_link) {
        rccl_destroy(static_cast<RcclLink*>(up.rccl_link));
        up.rccl_link = nullptr;
        up.rccl = false;
      }
      set_error(std::string("rejoin handshake failed: ") + e.what());
      std::this_thread::sleep_for(std::chrono::milliseconds(250));
      target = explicit_mode ? target : root_addr_;
      hops = 0;
    }
  }
  if (sub_corr_dev_) (void)hipFree(sub_corr_dev_), sub_corr_dev_ = nullptr;
  sub_corr_host_.clear();
  sub_corr_host_.shrink_to_fit();
  rejoin_preserve_ = false;
  reconnecting_.store(false);
} catch (const std::exception& e) {
  if (sub_corr_dev_) (void)hipFree(sub_corr_dev_), sub_corr_dev_ = nullptr;
  sub_corr_host_.clear();
  sub_corr_host_.shrink_to_fit();
  rejoin_preserve_ = false;
  set_error(std::string("reconnect failed: ") + e.what());
  reconnecting_.store(false);
}

void Engine::bind_listen(const sockaddr_in& addr, bool shared) {
  listen_fd_ = ::socket(AF_INET, SOCK_STREAM, 0);
  if (listen_fd_ < 0) throw std::runtime_error("socket() failed");
  int yes = 1;
  setsockopt(listen_fd_, SOL_SOCKET, SO_REUSEADDR, &yes, sizeof(yes));
  // child listeners are REUSEPORT so a rejoin can bind its outgoing socket
  // beside them; the MASTER/rendezvous bind stays exclusive — bind() is the
  // failover arbiter and REUSEPORT would allow split-brain
  if (shared) setsockopt(listen_fd_, SOL_SOCKET, SO_REUSEPORT, &yes, sizeof(yes));
  if (::bind(listen_fd_, reinterpret_cast<const sockaddr*>(&addr), sizeof(addr)) < 0) {
    std::string err = std::strerror(errno);
    ::close(listen_fd_);
    listen_fd_ = -1;
    throw std::runtime_error("bind(" + addr_str(addr) + ") failed: " + err);
  }
  if (::listen(listen_fd_, 16) < 0) {
    ::close(listen_fd_);
    listen_fd_ = -1;
    throw std::runtime_error("listen() failed");
  }
  sockaddr_in got{};
  socklen_t glen = sizeof(got);
  getsockname(listen_fd_, reinterpret_cast<sockaddr*>(&got), &glen);
  listen_port_ = ntohs(got.sin_port);
  listen_addr_ = got;
}

// ----------------------------------------------------------------- listen

void Engine::listen_loop() {
  int redirect_rr = 0;  // alternate redirects (sharedtensor.c:230)
  while (!closing_) {
    sockaddr_in peer{};
    socklen_t plen = sizeof(peer);
    int fd = ::accept(listen_fd_, reinterpret_cast<sockaddr*>(&peer), &plen);
    if (closing_) {
      if (fd >= 0) ::close(fd);
      break;
    }
    if (fd < 0) {
      if (errno == EINTR) continue;
      break;  // listen socket shut down
    }
    set_sockopts(fd);
    timeval tv{5, 0};
    setsockopt(fd, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof(tv));
    Hello h{};
    if (!io_read(fd, &h, sizeof(h)) || h.magic != MAGIC ||
        h.version != PROTO_VERSION) {
      ::close(fd);
      continue;
    }
    if (h.n != static_cast<uint64_t>(n_) ||
        h.ntensors != static_cast<uint32_t>(T_) ||
        h.codec != static_cast<uint32_t>(cfg_.codec)) {
      set_error("rejected joiner with mismatched tensor shape/codec");
      ::close(fd);
      continue;
    }
    timeval tv0{0, 0};
    setsockopt(fd, SOL_SOCKET, SO_RCVTIMEO, &tv0, sizeof(tv0));

    if (reconnecting_.load()) {
      // mid-rejoin our state is being reconciled; drop the joiner (it
      // retries its walk) rather than snapshot half-rebuilt values
      ::close(fd);
      continue;
    }
    // pick a free child slot — slots_m_ is held ONLY for the claim, so a
    // multi-GB snapshot on one slot never serializes other joins/redirects
    int slot = -1;
    {
      std::lock_guard<std::mutex> slot_guard(slots_m_);
      for (int i : {LK_LEFT, LK_RIGHT}) {
        Link& lk = links_[i];
        if (!lk.provisioned) continue;
        int st = lk.state.load();
        if (st == L_FREE) { slot = i; }
        else if (st == L_DEAD) {
          // reconnection support (reference TODO, README.md:33): reclaim the
          // slot once its old threads have exited
          if (lk.t_send.joinable()) lk.t_send.join();
          if (lk.t_recv.joinable()) lk.t_recv.join();
          if (lk.t_ctrl.joinable()) lk.t_ctrl.join();
          if (lk.fd >= 0) ::close(lk.fd);
          if (lk.rccl_link) {
            rccl_destroy(static_cast<RcclLink*>(lk.rccl_link));
            lk.rccl_link = nullptr;
            lk.rccl = false;
          }
          if (gpu()) destroy_link_graphs(lk);
          lk.abort.store(false);
          lk.fd = -1;
          lk.error.clear();
          slot = i;
        }
        if (slot >= 0) {
          // claim: reap the previous join thread (it has finished — the
          // slot was FREE/DEAD) and mark the slot as being handshaken
          if (lk.t_join.joinable()) lk.t_join.join();
          lk.fd = fd;
          lk.peer = peer;
          lk.peer_desc = addr_str(peer);
          lk.state.store(L_JOINING);
          break;
        }
      }
    }
    if (slot < 0) {
      // no capacity: redirect down the tree (sharedtensor.c:224-234)
      std::vector<sockaddr_in> cands;
      for (int i : {LK_LEFT, LK_RIGHT})
        if (links_[i].state.load() == L_ACTIVE) cands.push_back(links_[i].peer);
      sockaddr_in tgt;
      if (!cands.empty()) {
        tgt = cands[(redirect_rr++) % cands.size()];
      } else if (links_[LK_UP].state.load() == L_ACTIVE) {
        tgt = links_[LK_UP].peer;  // leaf with no child slots: bounce upward
      } else {
        sockaddr_in self{};
        socklen_t sl = sizeof(self);
        getsockname(listen_fd_, reinterpret_cast<sockaddr*>(&self), &sl);
        tgt = self;
      }
      uint8_t msg[7];
      msg[0] = 'N';
      std::memcpy(msg + 1, &tgt.sin_addr.s_addr, 4);
      std::memcpy(msg + 5, &tgt.sin_port, 2);
      io_write(fd, msg, 7);
      ::close(fd);
      continue;
    }
    Link& lk = links_[slot];
    lk.t_join = std::thread([this, fd, h, peer, slot] {
      accept_child(fd, h, peer, slot);
    });
  }
}

// slot delta := values, exclusive vs user mutations and packet applies.
// Re-establishes the unconnected-slot invariant (slot == values,
// sharedtensor.c:379-381 semantics) regardless of slot history: residue of
// a dead child (undelivered deltas minus its snapshot debit), an aborted
// snapshot, or a failover demotion would otherwise be gossiped as garbage
// to the next joiner claiming the slot.
void Engine::rebuild_slot_invariant(Link& lk) {
  std::unique_lock<std::shared_mutex> ug(user_m_);
  if (gpu()) {
    HIP_TRY(hipSetDevice(cfg_.device));
    HIP_TRY(hipDeviceSynchronize());  // drain in-flight user kernels
  }
  zero_delta(lk.delta);
  if (gpu()) {
    hip_add_scatter(values_, n_, 1.0f, nullptr, lk.delta, nullptr, nullptr,
                    cfg_.delta_bf16, nullptr);
    HIP_TRY(hipStreamSynchronize(nullptr));
  } else {
    cpu_pfor(n_, [&](int64_t lo, int64_t hi) {
      for (int64_t i = lo; i < hi; ++i) {
        float v = atomic_load_f32(values_ + i);
        if (v != 0.0f) atomic_add_f32(fdelta(lk.delta) + i, v);
      }
    });
  }
}

void Engine::accept_child(int fd, const Hello& h, const sockaddr_in& peer,
                          int slot) {
  Link& lk = links_[slot];
  rebuild_slot_invariant(lk);
  uint8_t yes = 'Y';
  bool upgrade = rccl_wanted(h);
  AcceptHello ah{};
  ah.version = PROTO_VERSION;
  ah.flags = static_cast<uint16_t>((cfg_.snapshot_join ? ACC_SNAPSHOT : 0) |
                                   (upgrade ? ACC_RCCL : 0));
  ah.codec = static_cast<uint32_t>(cfg_.codec);
  ah.n = static_cast<uint64_t>(n_);
  ah.ntensors = static_cast<uint32_t>(T_);
  if (!io_write(fd, &yes, 1) || !io_write(fd, &ah, sizeof(ah))) {
    ::close(fd);
    lk.fd = -1;
    lk.state.store(L_FREE);
    return;
  }
  if (upgrade) {
    try {
      uint8_t ids[2 * RCCL_ID_BYTES];
      rccl_make_ids(ids);
      if (!io_write(fd, ids, sizeof(ids)))
        throw std::runtime_error("failed to send RCCL ids");
      rccl_upgrade(lk, ids, /*is_parent=*/true);
    } catch (const std::exception& e) {
      // after repeated failures both sides stop negotiating RCCL and the
      // joiner's retry lands on the plain TCP data plane
      rccl_failures_++;
      set_error(std::string("rccl upgrade failed (") +
                std::to_string(rccl_failures_.load()) + "): " + e.what());
      ::close(fd);
      lk.fd = -1;
      lk.state.store(L_FREE);
      return;
    }
  }
  if (cfg_.snapshot_join) {
    try {
      send_snapshot(lk);
    } catch (const std::exception& e) {
      set_error(std::string("snapshot send failed: ") + e.what());
      ::close(fd);
      lk.fd = -1;
      if (lk.rccl_link) {  // don't leak the freshly-made comm pair
        rccl_destroy(static_cast<RcclLink*>(lk.rccl_link));
        lk.rccl_link = nullptr;
        lk.rccl = false;
      }
      // the aborted snapshot's debit leaves lk.delta != values; harmless:
      // rebuild_slot_invariant repairs it when the slot is next claimed
      lk.state.store(L_FREE);
      return;
    }
  }
  lk.state.store(L_ACTIVE);
  spawn_link_threads(lk);
}

void Engine::spawn_link_threads(Link& lk) {
  lk.t_send = std::thread([this, &lk] { send_loop(lk); });
  lk.t_recv = std::thread([this, &lk] { recv_loop(lk); });
  if (lk.rccl) lk.t_ctrl = std::thread([this, &lk] { ctrl_loop(lk); });
}

bool Engine::rccl_wanted(const Hello& h) const {
  // Test-only override: let two processes sharing ONE device drive the full
  // ncclSend/ncclRecv data plane (enqueue ordering, stream polling, abort,
  // teardown) on a single leased GPU.  Same-device pairs are declined by
  // default — on one device the TCP loopback path is both correct and has
  // no xGMI to win back.
  // read per call, NOT latched in a static: tests toggle this env var
  // within one process lifetime
  const char* fsd = std::getenv("SHTENS_RCCL_FORCE_SAME_DEVICE");
  const bool force_same_dev = fsd && fsd[0] == '1';
  return gpu() && cfg_.use_rccl && rccl_failures_.load() < 2 &&
         (h.flags & HELLO_WANT_RCCL) && h.hostid == hostid_ && h.device >= 0 &&
         (h.device != cfg_.device || force_same_dev);
}

void Engine::rccl_upgrade(Link& lk, const uint8_t* ids, bool is_parent) {
  // Test-only fault injection: fail the upgrade deterministically (before
  // any comm is made) so the negotiation-failure -> TCP-fallback path can
  // be executed on a single GPU.  A REAL same-device duplicate comm is
  // rejected by RCCL ("Duplicate GPU detected") and aborting the half-made
  // comm can hang, so the fallback is tested via injection instead.
  if (std::getenv("SHTENS_TEST_RCCL_FAIL"))
    throw std::runtime_error("injected rccl failure (SHTENS_TEST_RCCL_FAIL)");
  double to = cfg_.join_timeout_s < 20.0 ? cfg_.join_timeout_s : 20.0;
  lk.rccl_link = rccl_link_create(cfg_.device, ids, is_parent, to);
  lk.rccl = true;
}

// TCP control reader for RCCL-upgraded links: keepalives + close + death
// detection while the payload flows over xGMI.
void Engine::ctrl_loop(Link& lk) {
  while (!closing_ && lk.state.load() == L_ACTIVE) {
    PacketHeader hdr{};
    if (!io_read(lk.fd, &hdr, 8)) {
      link_down(lk, "peer disconnected (ctrl)", true);
      break;
    }
    if (hdr.type == PKT_CLOSE) {
      link_down(lk, "peer closed", true);
      break;
    }
    // PING or anything else: ignore
  }
}

// --------------------------------------------------------------- snapshot
// v2 fast join: instead of the reference's bootstrap-by-accumulated-delta
// (sharedtensor.c:379-388, O(dynamic range) gossip rounds), the parent
// streams its current values once and debits exactly the bytes sent from the
// link's delta buffer — preserving the error-feedback invariant under
// concurrent updates.

void Engine::send_snapshot(Link& lk) {
  // test-only join-storm knob: per-chunk throttle (and smaller chunks) to
  // make a snapshot artificially slow; never set in production paths.
  // Read per call (not a latched static): tests toggle it mid-process.
  const char* tde = std::getenv("SHTENS_TEST_SNAPSHOT_DELAY_MS");
  const double test_delay_s = tde ? std::atof(tde) / 1e3 : 0.0;
  if (gpu()) {
    // Pipelined GPU path: the fused capture+debit kernel writes the
    // authoritative sent-bytes into half of send_buf, D2H lands it in the
    // matching half of send_pin, and the TCP write of chunk k overlaps the
    // capture+D2H of chunk k+1 (double buffering; for small tensors one
    // chunk covers everything and this degenerates to the simple path).
    HIP_TRY(hipSetDevice(cfg_.device));
    // device staging (send_buf, SA_+P_ bytes) holds both halves
    int64_t half_bytes = std::min<int64_t>((SA_ + P_) / 2, 256 << 20);
    if (test_delay_s > 0.0)
      half_bytes = std::min<int64_t>(half_bytes, 1 << 14);
    // chunk offsets must stay EVEN (bf16 residual debits use packed-pair
    // atomics on the (even, odd) uint32 — an odd element offset faults) and
    // prefer 64-alignment; SA_+P_ >= 16 for any codec so sub >= 2 fits
    int64_t sub = std::max<int64_t>(half_bytes / 4, 2);
    sub = sub >= 64 ? (sub & ~int64_t(63)) : (sub & ~int64_t(1));
    hipEvent_t ev[2];
    HIP_TRY(hipEventCreateWithFlags(&ev[0], hipEventDisableTiming));
    HIP_TRY(hipEventCreateWithFlags(&ev[1], hipEventDisableTiming));
    int64_t prev_off = -1, prev_ce = 0;
    bool ok = true;
    std::string err;
    int64_t nstages = (n_ + sub - 1) / sub;
    for (int64_t k = 0; k < nstages && ok; ++k) {
      int64_t off = k * sub;
      int64_t ce = std::min(sub, n_ - off);
      int b = static_cast<int>(k & 1);
      auto* dev = reinterpret_cast<float*>(lk.send_buf) + b * sub;
      uint8_t* pin = lk.send_pin + b * sub * 4;
      try {
        hip_snapshot_capture(values_ + off, doff(lk.delta, off),
                             cfg_.delta_bf16, dev, ce, lk.s_send);
        HIP_TRY(hipMemcpyAsync(pin, dev, ce * 4, hipMemcpyDeviceToHost,
                               lk.s_send));
        HIP_TRY(hipEventRecord(ev[b], lk.s_send));
      } catch (const std::exception& e) {
        ok = false;
        err = e.what();
        break;
      }
      if (prev_off >= 0) {  // write chunk k-1 while chunk k copies
        uint8_t* ppin = lk.send_pin + ((k - 1) & 1) * sub * 4;
        HIP_TRY(hipEventSynchronize(ev[(k - 1) & 1]));
        if (!io_write(lk.fd, ppin, prev_ce * 4)) {
          ok = false;
          err = "tcp write failed";
          break;
        }
        lk.bytes_sent += prev_ce * 4;
        if (test_delay_s > 0.0)
          std::this_thread::sleep_for(
              std::chrono::duration<double>(test_delay_s));
      }
      prev_off = off;
      prev_ce = ce;
    }
    if (ok && prev_off >= 0) {
      uint8_t* ppin = lk.send_pin + ((nstages - 1) & 1) * sub * 4;
      HIP_TRY(hipEventSynchronize(ev[(nstages - 1) & 1]));
      if (!io_write(lk.fd, ppin, prev_ce * 4)) {
        ok = false;
        err = "tcp write failed";
      } else {
        lk.bytes_sent += prev_ce * 4;
      }
    }
    (void)hipStreamSynchronize(lk.s_send);  // drain before buffer reuse
    (void)hipEventDestroy(ev[0]);
    (void)hipEventDestroy(ev[1]);
    if (!ok) throw std::runtime_error(err);
    return;
  }
  const int64_t chunk_elems =
      std::max<int64_t>((test_delay_s > 0.0 ? (1 << 14) : (1 << 26)) / 4, 1);
  std::vector<uint8_t> tmp(static_cast<size_t>(chunk_elems) * 4);
  for (int64_t off = 0; off < n_; off += chunk_elems) {
    int64_t ce = std::min(chunk_elems, n_ - off);
    cpu_pfor(ce, [&](int64_t lo, int64_t hi) {
      for (int64_t i = lo; i < hi; ++i)
        reinterpret_cast<float*>(tmp.data())[i] =
            atomic_load_f32(values_ + off + i);
    });
    if (!io_write(lk.fd, tmp.data(), ce * 4))
      throw std::runtime_error("tcp write failed");
    const float* snap = reinterpret_cast<const float*>(tmp.data());
    cpu_pfor(ce, [&](int64_t lo, int64_t hi) {
      for (int64_t i = lo; i < hi; ++i)
        if (snap[i] != 0.0f)
          atomic_add_f32(fdelta(lk.delta) + off + i, -snap[i]);
    });
    lk.bytes_sent += ce * 4;
    if (test_delay_s > 0.0)
      std::this_thread::sleep_for(std::chrono::duration<double>(test_delay_s));
  }
}

void Engine::recv_snapshot(int fd, float* corr) {
  Link& up = links_[LK_UP];
  const int64_t chunk_bytes = gpu() ? (SA_ + P_) : (1 << 26);
  // even (prefer 64-aligned) chunk offsets: bf16 packed-pair atomics in the
  // forward-scatter need pair-aligned bases (see send_snapshot)
  int64_t chunk_elems = std::max<int64_t>(chunk_bytes / 4, 2);
  chunk_elems = chunk_elems >= 64 ? (chunk_elems & ~int64_t(63))
                                  : (chunk_elems & ~int64_t(1));
  std::vector<uint8_t> tmp;
  if (!gpu()) tmp.resize(static_cast<size_t>(chunk_elems) * 4);
  void* fwd[2] = {
      links_[LK_LEFT].provisioned ? links_[LK_LEFT].delta : nullptr,
      links_[LK_RIGHT].provisioned ? links_[LK_RIGHT].delta : nullptr};
  if (corr) fwd[0] = fwd[1] = nullptr;  // preserving: no raw-S forwarding
  for (int64_t off = 0; off < n_; off += chunk_elems) {
    int64_t ce = std::min(chunk_elems, n_ - off);
    if (gpu()) {
      HIP_TRY(hipSetDevice(cfg_.device));
      if (!io_read(fd, up.recv_pin, ce * 4))
        throw std::runtime_error("snapshot read failed");
      HIP_TRY(hipMemcpyAsync(up.recv_buf, up.recv_pin, ce * 4,
                             hipMemcpyHostToDevice, up.s_recv));
      // values += snap; future children's deltas += snap (join-state
      // forwarding, the GPU analog of sharedtensor.c:379-381)
      hip_add_scatter(reinterpret_cast<float*>(up.recv_buf), ce, 1.0f,
                      values_ + off, fwd[0] ? doff(fwd[0], off) : nullptr,
                      fwd[1] ? doff(fwd[1], off) : nullptr, nullptr,
                      cfg_.delta_bf16, up.s_recv);
      if (corr)  // correction accumulator is plain fp32
        hip_add_scatter(reinterpret_cast<float*>(up.recv_buf), ce, 1.0f,
                        corr + off, nullptr, nullptr, nullptr, false,
                        up.s_recv);
      HIP_TRY(hipStreamSynchronize(up.s_recv));
    } else {
      if (!io_read(fd, tmp.data(), ce * 4))
        throw std::runtime_error("snapshot read failed");
      const float* snap = reinterpret_cast<const float*>(tmp.data());
      cpu_pfor(ce, [&](int64_t lo, int64_t hi) {
        for (int64_t i = lo; i < hi; ++i) {
          float v = snap[i];
          if (v == 0.0f) continue;
          atomic_add_f32(values_ + off + i, v);
          if (corr) corr[off + i] += v;  // corr is reconnect-thread-private
          if (fwd[0]) atomic_add_f32(fdelta(fwd[0]) + off + i, v);
          if (fwd[1]) atomic_add_f32(fdelta(fwd[1]) + off + i, v);
        }
      });
    }
    up.bytes_recv += ce * 4;
  }
}

// -------------------------------------------------------------- send side

void Engine::compute_scales(Link& lk, float* scales_host, bool lagged_valid) {
  if (gpu()) {
    HIP_TRY(hipSetDevice(cfg_.device));
    float* scales_dev = reinterpret_cast<float*>(lk.send_buf);
    bool lagged = cfg_.lagged_scale && lagged_valid;
    if (lk.scales_host.size() != static_cast<size_t>(T_))
      lk.scales_host.resize(T_);
    auto seq = [&] {
      if (lagged) {
        // steady state: the stats were accumulated by the previous round's
        // quantize kernel — no reduce pass over the residual needed
        hip_finalize_scales(cfg_.codec, dtb_, lk.reduce_buf, scales_dev,
                            /*stride=*/1, lk.s_send);
        HIP_TRY(hipMemsetAsync(lk.reduce_buf, 0, 8 * T_, lk.s_send));
      } else {
        hip_reduce_scales(cfg_.codec, lk.delta, cfg_.delta_bf16, dtb_,
                          lk.reduce_buf, scales_dev, cfg_.rms_sample_stride,
                          lk.s_send);
        if (cfg_.lagged_scale)
          HIP_TRY(hipMemsetAsync(lk.reduce_buf, 0, 8 * T_, lk.s_send));
      }
      HIP_TRY(hipMemcpyAsync(lk.scales_host.data(), lk.send_buf, 4 * T_,
                             hipMemcpyDeviceToHost, lk.s_send));
    };
    if (cfg_.use_graphs) {
      hipGraphExec_t& e = lagged ? lk.g_scale_lagged : lk.g_scale_exact;
      if (!e) e = capture_seq(lk.s_send, seq);
      HIP_TRY(hipGraphLaunch(e, lk.s_send));
    } else {
      seq();
    }
    HIP_TRY(hipStreamSynchronize(lk.s_send));
    std::memcpy(scales_host, lk.scales_host.data(), 4 * T_);
  } else {
    for (int t = 0; t < T_; ++t)
      scales_host[t] = cpu_compute_scale(cfg_.codec, fdelta(lk.delta) + offs_[t],
                                         cfg_.sizes[t], cfg_.rms_sample_stride);
    std::memcpy(lk.send_pin + 8, scales_host, 4 * T_);
  }
}

bool Engine::send_packet(Link& lk, const float* scales_host) {
  PacketHeader hdr{};
  hdr.type = PKT_DATA;
  hdr.codec = static_cast<uint8_t>(cfg_.codec);
  hdr.ntensors = static_cast<uint32_t>(T_);
  std::memcpy(lk.send_pin, &hdr, 8);
  if (gpu()) {
    // the payload D2H was part of the quantize phase (graph); just drain
    HIP_TRY(hipStreamSynchronize(lk.s_send));
  }
  std::lock_guard<std::mutex> g(lk.wm);
  if (!io_write(lk.fd, lk.send_pin, 8 + SA_ + P_)) return false;
  return true;
}

void Engine::send_loop(Link& lk) {
  if (gpu()) (void)hipSetDevice(cfg_.device);
  std::vector<float> scales(T_);
  auto last_send = Clock::now();
  auto next_allowed = Clock::now();
  const double keepalive = cfg_.keepalive_s;
  bool lagged_valid = false;
  while (!closing_ && lk.state.load() == L_ACTIVE) {
    try {
      compute_scales(lk, scales.data(), lagged_valid);
    } catch (const std::exception& e) {
      link_down(lk, std::string("scale reduction failed: ") + e.what(), false);
      break;
    }
    float maxs = 0.f;
    for (float s : scales) maxs = std::max(maxs, s);
    if (maxs == 0.0f) {
      // idle: cheap PING instead of the reference's full zero packet
      // (sharedtensor.c:161-177); wake instantly when a delta lands
      auto now = Clock::now();
      if (std::chrono::duration<double>(now - last_send).count() >= keepalive) {
        PacketHeader ping{};
        ping.type = PKT_PING;
        ping.ntensors = static_cast<uint32_t>(T_);
        std::lock_guard<std::mutex> g(lk.wm);
        if (!io_write(lk.fd, &ping, 8)) {
          link_down(lk, "keepalive write failed", true);
          break;
        }
        last_send = now;
      }
      lagged_valid = false;  // residual went quiet; restats on wake
      std::unique_lock<std::mutex> l(lk.m);
      lk.cv.wait_for(l, std::chrono::duration<double>(keepalive), [&] {
        return lk.dirty || closing_ || lk.state.load() != L_ACTIVE;
      });
      lk.dirty = false;
      continue;
    }
    // pacing: bandwidth cap (reference TODO, README.md:31) and/or a
    // minimum per-round interval (HBM-bandwidth courtesy towards compute)
    if (cfg_.bw_limit > 0 || cfg_.min_round_interval_s > 0) {
      auto now = Clock::now();
      if (now < next_allowed)
        std::this_thread::sleep_for(next_allowed - now);
      double gap = cfg_.min_round_interval_s;
      if (cfg_.bw_limit > 0)
        gap = std::max(gap, (8.0 + SA_ + P_) / cfg_.bw_limit);
      next_allowed = Clock::now() + std::chrono::duration_cast<Clock::duration>(
          std::chrono::duration<double>(gap));
    }
    try {
      if (gpu()) {
        auto qseq = [&] {
          hip_quantize(cfg_.codec, lk.delta, cfg_.delta_bf16, dtb_,
                       reinterpret_cast<float*>(lk.send_buf), lk.send_buf + SA_,
                       lk.s_send, cfg_.lagged_scale ? lk.reduce_buf : nullptr);
          if (!lk.rccl)  // TCP: stage the whole message to pinned host
            HIP_TRY(hipMemcpyAsync(lk.send_pin + 8, lk.send_buf, SA_ + P_,
                                   hipMemcpyDeviceToHost, lk.s_send));
        };
        if (cfg_.use_graphs) {
          if (!lk.g_quant) lk.g_quant = capture_seq(lk.s_send, qseq);
          HIP_TRY(hipGraphLaunch(lk.g_quant, lk.s_send));
        } else {
          qseq();
        }
        lagged_valid = true;
      } else {
        for (int t = 0; t < T_; ++t)
          cpu_quantize(cfg_.codec, fdelta(lk.delta) + offs_[t], cfg_.sizes[t], scales[t],
                       lk.send_pin + 8 + SA_ +
                           (poffs_[t] / 64) * (payload_bytes(cfg_.codec, 64)));
      }
    } catch (const std::exception& e) {
      link_down(lk, std::string("quantize failed: ") + e.what(), false);
      break;
    }
    if (lk.rccl) {
      // xGMI data plane: the packed message goes device-to-device
      if (!rccl_send(static_cast<RcclLink*>(lk.rccl_link), lk.send_buf,
                     SA_ + P_, lk.s_send, lk.abort)) {
        link_down(lk, "rccl send failed", lk.abort.load());
        break;
      }
    } else if (!send_packet(lk, scales.data())) {
      link_down(lk, "data write failed", true);
      break;
    }
    last_send = Clock::now();
    lk.rounds_sent++;
    lk.bytes_sent += 8 + SA_ + P_;
    lk.last_scale_sent.store(maxs);
    push_scale(true, maxs);
  }
}

// -------------------------------------------------------------- recv side

void Engine::apply_packet(Link& lk, const float* scales_host) {
  // Shared user-op lock: packet application forwards into other links'
  // delta buffers, which an exclusive invariant repair (rejoin/failover/
  // aborted-snapshot) must be able to quiesce along with user mutations.
  std::shared_lock<std::shared_mutex> ug(user_m_);
  // destinations: local replica + gossip-forward into the other links'
  // delta buffers, excluding the source (sharedtensor.c:124-127)
  void* fwd[2] = {nullptr, nullptr};
  int nf = 0;
  for (int i = 0; i < 3; ++i)
    if (i != lk.idx && links_[i].provisioned) fwd[nf++] = links_[i].delta;
  if (gpu()) {
    HIP_TRY(hipSetDevice(cfg_.device));
    auto aseq = [&] {
      if (!lk.rccl)  // TCP staging; RCCL already delivered into recv_buf
        HIP_TRY(hipMemcpyAsync(lk.recv_buf, lk.recv_pin + 8, SA_ + P_,
                               hipMemcpyHostToDevice, lk.s_recv));
      hip_apply(cfg_.codec, lk.recv_buf + SA_, dtb_,
                reinterpret_cast<float*>(lk.recv_buf), values_, fwd[0], fwd[1],
                cfg_.delta_bf16, lk.s_recv);
    };
    if (cfg_.use_graphs) {
      if (!lk.g_apply) lk.g_apply = capture_seq(lk.s_recv, aseq);
      HIP_TRY(hipGraphLaunch(lk.g_apply, lk.s_recv));
    } else {
      aseq();
    }
    HIP_TRY(hipStreamSynchronize(lk.s_recv));
  } else {
    for (int t = 0; t < T_; ++t) {
      float* dsts[3] = {values_ + offs_[t],
                        fwd[0] ? fdelta(fwd[0]) + offs_[t] : nullptr,
                        fwd[1] ? fdelta(fwd[1]) + offs_[t] : nullptr};
      int nd = 1 + (fwd[0] ? 1 : 0) + (fwd[1] ? 1 : 0);
      cpu_apply(cfg_.codec,
                lk.recv_pin + 8 + SA_ +
                    (poffs_[t] / 64) * (payload_bytes(cfg_.codec, 64)),
                cfg_.sizes[t], scales_host[t], dsts, nd);
    }
  }
}

void Engine::recv_loop(Link& lk) {
  if (gpu()) (void)hipSetDevice(cfg_.device);
  std::vector<float> scales(T_);
  while (!closing_ && lk.state.load() == L_ACTIVE) {
    if (lk.rccl) {
      if (!rccl_recv(static_cast<RcclLink*>(lk.rccl_link), lk.recv_buf,
                     SA_ + P_, lk.s_recv, lk.abort)) {
        link_down(lk, "rccl recv ended", true);
        break;
      }
      try {
        HIP_TRY(hipMemcpy(scales.data(), lk.recv_buf, 4 * T_,
                          hipMemcpyDeviceToHost));
        float maxs = 0.f;
        for (float s : scales) maxs = std::max(maxs, s);
        lk.rounds_recv++;
        lk.bytes_recv += SA_ + P_;
        lk.last_scale_recv.store(maxs);
        push_scale(false, maxs);
        if (maxs != 0.0f) apply_packet(lk, scales.data());
      } catch (const std::exception& e) {
        link_down(lk, std::string("apply failed: ") + e.what(), false);
        break;
      }
      for (int i = 0; i < 3; ++i) {
        if (i == lk.idx) continue;
        Link& o = links_[i];
        if (o.state.load() == L_ACTIVE) {
          std::lock_guard<std::mutex> g(o.m);
          o.dirty = true;
          o.cv.notify_all();
        }
      }
      continue;
    }
    PacketHeader hdr{};
    if (!io_read(lk.fd, &hdr, 8)) {
      link_down(lk, "peer disconnected", true);
      break;
    }
    if (hdr.type == PKT_CLOSE) {
      link_down(lk, "peer closed", true);
      break;
    }
    if (hdr.type == PKT_PING) continue;
    if (hdr.type != PKT_DATA ||
        hdr.ntensors != static_cast<uint32_t>(T_) ||
        hdr.codec != static_cast<uint8_t>(cfg_.codec)) {
      link_down(lk, "protocol error in data stream", false);
      break;
    }
    if (!io_read(lk.fd, lk.recv_pin + 8, SA_ + P_)) {
      link_down(lk, "payload read failed", true);
      break;
    }
    std::memcpy(scales.data(), lk.recv_pin + 8, 4 * T_);
    float maxs = 0.f;
    for (float s : scales) maxs = std::max(maxs, s);
    lk.rounds_recv++;
    lk.bytes_recv += 8 + SA_ + P_;
    lk.last_scale_recv.store(maxs);
    push_scale(false, maxs);
    if (maxs == 0.0f) continue;
    try {
      apply_packet(lk, scales.data());
    } catch (const std::exception& e) {
      link_down(lk, std::string("apply failed: ") + e.what(), false);
      break;
    }
    // wake the other links: they now have fresh residual to forward
    for (int i = 0; i < 3; ++i) {
      if (i == lk.idx) continue;
      Link& o = links_[i];
      if (o.state.load() == L_ACTIVE) {
        std::lock_guard<std::mutex> g(o.m);
        o.dirty = true;
        o.cv.notify_all();
      }
    }
  }
}

void Engine::link_down(Link& lk, const std::string& why, bool remote) {
  int expected = L_ACTIVE;
  if (!lk.state.compare_exchange_strong(expected, L_DEAD)) return;
  lk.error = why;
  if (!remote)
    set_error("link " + std::to_string(lk.idx) + " (" + lk.peer_desc +
              ") down: " + why);
  lk.abort.store(true);
  if (lk.rccl_link) rccl_abort(static_cast<RcclLink*>(lk.rccl_link));
  if (lk.fd >= 0) ::shutdown(lk.fd, SHUT_RDWR);
  lk.cv.notify_all();
  if (lk.idx == LK_UP && cfg_.reconnect && !closing_) {
    bool expected = false;
    if (reconnecting_.compare_exchange_strong(expected, true)) {
      if (reconnect_thread_.joinable()) reconnect_thread_.join();
      reconnect_thread_ = std::thread([this] { reconnect_loop(); });
    }
  }
}

// --------------------------------------------------------------- user API

void Engine::notify_all_dirty() {
  for (auto& lk : links_) {
    if (lk.state.load() != L_ACTIVE) continue;
    std::lock_guard<std::mutex> g(lk.m);
    lk.dirty = true;
    lk.cv.notify_all();
  }
}

void Engine::notify_dirty() { notify_all_dirty(); }

void Engine::add_from(uintptr_t src, int64_t n, uintptr_t stream) {
  if (n != n_) throw std::runtime_error("add_from: size mismatch");
  std::shared_lock<std::shared_mutex> ug(user_m_);
  const float* s = reinterpret_cast<const float*>(src);
  void* d[3];
  for (int i = 0; i < 3; ++i)
    d[i] = links_[i].provisioned ? links_[i].delta : nullptr;
  if (gpu()) {
    hip_add_scatter(s, n_, 1.0f, values_, d[0], d[1], d[2], cfg_.delta_bf16,
                    reinterpret_cast<hipStream_t>(stream));
  } else {
    float* dsts[4];
    int nd = 0;
    dsts[nd++] = values_;
    for (int i = 0; i < 3; ++i)
      if (d[i]) dsts[nd++] = fdelta(d[i]);
    cpu_add_scatter(s, n_, dsts, nd);
  }
  notify_all_dirty();
}

void Engine::copy_to(uintptr_t dst, int64_t n, uintptr_t stream) {
  if (n != n_) throw std::runtime_error("copy_to: size mismatch");
  // shared lock: users never observe the transiently-zeroed values of a
  // rejoin reconciliation (stale-but-consistent is the async contract)
  std::shared_lock<std::shared_mutex> ug(user_m_);
  if (gpu()) {
    HIP_TRY(hipMemcpyAsync(reinterpret_cast<void*>(dst), values_, n_ * 4,
                           hipMemcpyDeviceToDevice,
                           reinterpret_cast<hipStream_t>(stream)));
  } else {
    float* d = reinterpret_cast<float*>(dst);
    cpu_pfor(n_, [&](int64_t lo, int64_t hi) {
      for (int64_t i = lo; i < hi; ++i) d[i] = atomic_load_f32(values_ + i);
    });
  }
}

void Engine::fused_sgd(uintptr_t mom, uintptr_t grad, double lr,
                       double momentum, uintptr_t stream) {
  std::shared_lock<std::shared_mutex> ug(user_m_);
  void* d[3];
  for (int i = 0; i < 3; ++i)
    d[i] = links_[i].provisioned ? links_[i].delta : nullptr;
  if (gpu()) {
    hip_fused_sgd(reinterpret_cast<float*>(mom),
                  reinterpret_cast<const float*>(grad), static_cast<float>(lr),
                  static_cast<float>(momentum), n_, values_, d[0], d[1], d[2],
                  cfg_.delta_bf16, reinterpret_cast<hipStream_t>(stream));
  } else {
    float* m = reinterpret_cast<float*>(mom);
    const float* g = reinterpret_cast<const float*>(grad);
    cpu_pfor(n_, [&](int64_t lo, int64_t hi) {
      for (int64_t i = lo; i < hi; ++i) {
        float mm = static_cast<float>(momentum) * m[i] + g[i];
        m[i] = mm;
        float u = static_cast<float>(-lr) * mm;
        if (u == 0.0f) continue;
        atomic_add_f32(values_ + i, u);
        for (int k = 0; k < 3; ++k)
          if (d[k]) atomic_add_f32(fdelta(d[k]) + i, u);
      }
    });
  }
  notify_all_dirty();
}

void Engine::fused_sgd_bf16(uintptr_t mom, uintptr_t grad_bf16,
                            uintptr_t shadow_bf16, double lr, double momentum,
                            uintptr_t stream) {
  if (!gpu())
    throw std::runtime_error("fused_sgd_bf16 is a GPU-only path");
  std::shared_lock<std::shared_mutex> ug(user_m_);
  void* d[3];
  for (int i = 0; i < 3; ++i)
    d[i] = links_[i].provisioned ? links_[i].delta : nullptr;
  hip_fused_sgd_bf16(reinterpret_cast<float*>(mom),
                     reinterpret_cast<const uint16_t*>(grad_bf16),
                     reinterpret_cast<uint16_t*>(shadow_bf16),
                     static_cast<float>(lr), static_cast<float>(momentum), n_,
                     values_, d[0], d[1], d[2], cfg_.delta_bf16,
                     reinterpret_cast<hipStream_t>(stream));
  notify_all_dirty();
}

void Engine::fused_adamw(uintptr_t mom, uintptr_t vel, uintptr_t grad,
                         bool grad_bf16, uintptr_t shadow, double lr,
                         double beta1, double beta2, double eps, double wd,
                         int64_t step, uintptr_t stream) {
  if (step < 1) throw std::runtime_error("fused_adamw: step must be >= 1");
  if (grad_bf16 && !gpu())
    throw std::runtime_error("bf16 grads are a GPU-only path");
  std::shared_lock<std::shared_mutex> ug(user_m_);
  void* d[3];
  for (int i = 0; i < 3; ++i)
    d[i] = links_[i].provisioned ? links_[i].delta : nullptr;
  const float b1 = static_cast<float>(beta1), b2 = static_cast<float>(beta2);
  const float inv_bc1 =
      1.0f / (1.0f - std::pow(b1, static_cast<float>(step)));
  const float inv_bc2 =
      1.0f / (1.0f - std::pow(b2, static_cast<float>(step)));
  if (gpu()) {
    hip_fused_adamw(reinterpret_cast<float*>(mom),
                    reinterpret_cast<float*>(vel),
                    reinterpret_cast<const void*>(grad), grad_bf16,
                    reinterpret_cast<uint16_t*>(shadow),
                    static_cast<float>(lr), b1, b2, static_cast<float>(eps),
                    static_cast<float>(wd), inv_bc1, inv_bc2, n_, values_,
                    d[0], d[1], d[2], cfg_.delta_bf16,
                    reinterpret_cast<hipStream_t>(stream));
  } else {
    float* m = reinterpret_cast<float*>(mom);
    float* v = reinterpret_cast<float*>(vel);
    const float* g = reinterpret_cast<const float*>(grad);
    const float lrf = static_cast<float>(lr), epsf = static_cast<float>(eps);
    const float wdf = static_cast<float>(wd);
    cpu_pfor(n_, [&](int64_t lo, int64_t hi) {
      for (int64_t i = lo; i < hi; ++i) {
        float gi = g[i];
        float mi = b1 * m[i] + (1.f - b1) * gi;
        m[i] = mi;
        float vi = b2 * v[i] + (1.f - b2) * gi * gi;
        v[i] = vi;
        float w = atomic_load_f32(values_ + i);
        float u = -lrf * (mi * inv_bc1 / (std::sqrt(vi * inv_bc2) + epsf) +
                          wdf * w);
        if (u == 0.0f) continue;
        atomic_add_f32(values_ + i, u);
        for (int k = 0; k < 3; ++k)
          if (d[k]) atomic_add_f32(fdelta(d[k]) + i, u);
      }
    });
  }
  notify_all_dirty();
}

std::vector<LinkStatsSnap> Engine::link_stats() {
  std::vector<LinkStatsSnap> out;
  for (auto& lk : links_) {
    LinkStatsSnap s{};
    s.rounds_sent = lk.rounds_sent.load();
    s.rounds_recv = lk.rounds_recv.load();
    s.bytes_sent = lk.bytes_sent.load();
    s.bytes_recv = lk.bytes_recv.load();
    s.last_scale_sent = lk.last_scale_sent.load();
    s.last_scale_recv = lk.last_scale_recv.load();
    s.active = lk.state.load() == L_ACTIVE;
    s.dead = lk.state.load() == L_DEAD;
    s.peer = lk.peer_desc;
    s.rccl = lk.rccl;
    out.push_back(s);
  }
  return out;
}

void Engine::close() {
  // Idempotent AND safe under concurrency: the first caller performs the
  // teardown; later callers (destructor racing an explicit close) block on
  // close_m_ until it is done and then return.
  std::lock_guard<std::mutex> cg(close_m_);
  if (closing_.exchange(true)) return;
  // tell peers we are leaving (the reference cannot do this and exit(-1)s,
  // sharedtensor.c:421-430)
  for (auto& lk : links_) {
    if (lk.state.load() == L_ACTIVE && lk.fd >= 0) {
      PacketHeader bye{};
      bye.type = PKT_CLOSE;
      bye.ntensors = static_cast<uint32_t>(T_);
      std::lock_guard<std::mutex> g(lk.wm);
      io_write(lk.fd, &bye, 8);
    }
  }
  for (auto& lk : links_) {
    lk.abort.store(true);
    if (lk.rccl_link) rccl_abort(static_cast<RcclLink*>(lk.rccl_link));
    if (lk.fd >= 0) ::shutdown(lk.fd, SHUT_RDWR);
    lk.cv.notify_all();
  }
  if (reconnect_thread_.joinable()) reconnect_thread_.join();
  if (listen_fd_ >= 0) ::shutdown(listen_fd_, SHUT_RDWR);
  if (listen_thread_.joinable()) listen_thread_.join();
  for (auto& lk : links_) {
    if (lk.t_join.joinable()) lk.t_join.join();
    if (lk.t_send.joinable()) lk.t_send.join();
    if (lk.t_recv.joinable()) lk.t_recv.join();
    if (lk.t_ctrl.joinable()) lk.t_ctrl.join();
    if (lk.fd >= 0) ::close(lk.fd), lk.fd = -1;
    if (lk.rccl_link) {
      rccl_destroy(static_cast<RcclLink*>(lk.rccl_link));
      lk.rccl_link = nullptr;
    }
    lk.state.store(L_DEAD);
  }
  if (listen_fd_ >= 0) ::close(listen_fd_), listen_fd_ = -1;
  free_gpu();
}

}  // namespace shamd
