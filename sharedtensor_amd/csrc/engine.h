// The shared-tensor engine: replica + per-link delta accounting, compressed
// gossip over a self-organizing binary tree.
//
// Re-engineered from the capability set of /root/reference/src/sharedtensor.c:
//   * L3 replica/delta state (:24-44,334-344)  -> HBM-resident buffers,
//     lossless relaxed-atomic accumulation instead of benign-lossy races
//   * L2 sync engine (:106-189)                -> CDNA4 kernels on dedicated
//     HIP streams per link (CPU codec for host tensors)
//   * L1 topology (:192-332)                   -> same Y/N-redirect join walk,
//     v2 handshake (codec negotiation, snapshot state transfer, RCCL upgrade)
//   * L0 robust I/O (:47-104)                  -> errors surface as C++
//     exceptions / dead-link state, never exit(-1)
#pragma once

#include <netinet/in.h>

#include "common.h"
#include "hip_api.h"

namespace shamd {

enum LinkState { L_FREE = 0, L_ACTIVE = 1, L_DEAD = 2, L_JOINING = 3 };
enum LinkIdx { LK_UP = 0, LK_LEFT = 1, LK_RIGHT = 2 };

struct Link {
  int idx = 0;
  std::atomic<int> state{L_FREE};
  int fd = -1;
  sockaddr_in peer{};  // peer's listen address (ref trick: == getpeername,
                       // because a joiner binds its listener to the local
                       // addr of its up-connection, sharedtensor.c:292-316)
  bool provisioned = false;  // buffers exist for this slot
  void* delta = nullptr;             // residual, fp32[n] or bf16[n] (GPU)
  uint8_t* send_buf = nullptr;       // [scales 4T | pad | payload P]
  uint8_t* recv_buf = nullptr;
  uint8_t* send_pin = nullptr;       // pinned host: [hdr 8 | scales | payload]
  uint8_t* recv_pin = nullptr;
  void* reduce_buf = nullptr;        // device scratch, 8*T bytes
  hipStream_t s_send = nullptr, s_recv = nullptr;
  std::thread t_send, t_recv;
  std::mutex wm;  // serializes writes on fd (send loop vs CLOSE)
  std::mutex m;
  std::condition_variable cv;
  bool dirty = false;
  std::atomic<uint64_t> rounds_sent{0}, rounds_recv{0};
  std::atomic<uint64_t> bytes_sent{0}, bytes_recv{0};
  std::atomic<float> last_scale_sent{0.f}, last_scale_recv{0.f};
  std::string peer_desc;
  std::string error;
  bool rccl = false;           // data plane upgraded to RCCL over xGMI
  void* rccl_link = nullptr;   // opaque RcclLink*
  std::thread t_ctrl;          // TCP control reader when data plane is RCCL
  std::thread t_join;          // per-join handshake+snapshot thread: joins
                               // overlap and redirects keep flowing while a
                               // multi-GB snapshot streams (round-1 weak #4)
  std::atomic<bool> abort{false};
  // captured per-round sequences (GPU): one hipGraphLaunch replaces the
  // 2-4 launches of the scale / quantize+stage / apply phases
  hipGraphExec_t g_scale_lagged = nullptr;  // finalize+memset+scales D2H
  hipGraphExec_t g_scale_exact = nullptr;   // reduce+finalize(+memset)+D2H
  hipGraphExec_t g_quant = nullptr;         // quantize (+payload D2H for TCP)
  hipGraphExec_t g_apply = nullptr;         // (H2D for TCP) + apply
  std::vector<float> scales_host;           // persistent D2H target
};

class Engine {
 public:
  explicit Engine(Config cfg);
  ~Engine();

  // Buffer registration (python/torch owns all large allocations; pinned
  // pointers may be 0 for CPU engines).
  void set_values(uintptr_t p);
  void set_link_buffers(int link, uintptr_t delta, uintptr_t send_buf,
                        uintptr_t recv_buf, uintptr_t send_pin,
                        uintptr_t recv_pin);

  void start();  // join the tree (blocking, incl. snapshot); throws on error
  void add_from(uintptr_t src, int64_t n, uintptr_t stream);
  void copy_to(uintptr_t dst, int64_t n, uintptr_t stream);
  void fused_sgd(uintptr_t mom, uintptr_t grad, double lr, double momentum,
                 uintptr_t stream);
  void fused_sgd_bf16(uintptr_t mom, uintptr_t grad_bf16, uintptr_t shadow_bf16,
                      double lr, double momentum, uintptr_t stream);
  // shadow==0: fp32 grads, no shadow refresh (grad_bf16 must be false)
  void fused_adamw(uintptr_t mom, uintptr_t vel, uintptr_t grad, bool grad_bf16,
                   uintptr_t shadow, double lr, double beta1, double beta2,
                   double eps, double wd, int64_t step, uintptr_t stream);
  void notify_dirty();  // wake senders after out-of-band delta writes
  void close();

  bool is_master() const { return is_master_; }
  int listen_port() const { return listen_port_; }
  std::vector<LinkStatsSnap> link_stats();
  std::vector<float> recent_scales_sent();
  std::vector<float> recent_scales_recv();
  std::string last_error();
  uint64_t reconnect_count() const { return reconnects_.load(); }

  // exposed sizes (python uses these to size buffers; static helpers)
  static int64_t msg_bytes(const Config& cfg);    // SA + P
  static int64_t scales_area(const Config& cfg);  // align8(4*T)

 private:
  Config cfg_;
  int64_t n_ = 0, pe_ = 0, P_ = 0, SA_ = 0;
  int T_ = 1;
  std::vector<int64_t> offs_, poffs_;
  float* values_ = nullptr;
  Link links_[3];
  int listen_fd_ = -1;
  int listen_port_ = 0;
  sockaddr_in listen_addr_{};
  std::mutex slots_m_;  // child-slot assignment vs reconnect demotion
  // User mutation ops (add_from / fused_sgd* / fused_adamw / copy_to) and
  // packet applies hold this shared; rejoin reconciliation, failover
  // invariant-restore and slot rebuilds hold it exclusive so a concurrent
  // update cannot land once in values and twice via a captured residual
  // (ADVICE round 1, medium x2).  Note on fairness: glibc rwlocks prefer
  // readers, so a steady storm of shared holders could delay the exclusive
  // taker — acceptable here because shared sections are short (one kernel
  // enqueue / one packet apply) and exclusive takers only run on rare
  // failure-recovery paths where the up-link is already down (no inbound
  // gossip) and user steps leave gaps between ops.
  std::shared_mutex user_m_;
  std::mutex close_m_;  // serializes concurrent close() calls
  std::thread listen_thread_;
  std::thread reconnect_thread_;
  std::atomic<bool> reconnecting_{false};
  // subtree-preserving rejoin (cfg.preserve_subtree): children stay
  // attached; the reconciliation accumulates a correction delta
  // corr = S + R - V_old (exact: concurrent updates never touch corr) that
  // is added to every live child's residual and drains as ordinary gossip
  bool rejoin_preserve_ = false;   // decided per rejoin attempt
  float* sub_corr_dev_ = nullptr;  // GPU scratch fp32[n] (corr)
  std::vector<float> sub_corr_host_;
  std::atomic<uint64_t> reconnects_{0};
  std::atomic<int> rccl_failures_{0};  // >=2: stop offering/requesting RCCL
  std::atomic<bool> closing_{false};
  bool started_ = false;
  bool is_master_ = false;
  uint64_t hostid_ = 0;
  sockaddr_in root_addr_{};
  std::mutex err_m_;
  std::string last_error_;
  // staleness telemetry: ring of recent per-round max scales
  std::mutex ring_m_;
  std::deque<float> ring_sent_, ring_recv_;
  // device-side table
  DevTable dtb_;
  bool gpu() const { return cfg_.device >= 0; }

  void init_gpu();
  void free_gpu();
  void join_tree();
  bool try_connect(const sockaddr_in& addr, int& out_fd,
                   const sockaddr_in* bind_local = nullptr);
  void handshake_as_child(int fd, bool rejoin = false);
  void reconnect_loop();
  void drop_children();
  bool failover_master();
  void zero_buf(float* p, int64_t n);
  void zero_delta(void* p);
  void* doff(void* delta, int64_t off) const;  // typed element offset
  float* fdelta(void* p) const;  // CPU engines: deltas are always fp32
  void become_master();
  void bind_listen(const sockaddr_in& addr, bool shared = true);
  void listen_loop();
  void accept_child(int fd, const Hello& h, const sockaddr_in& peer, int slot);
  void rebuild_slot_invariant(Link& lk);  // slot delta := values
  void spawn_link_threads(Link& lk);
  void send_loop(Link& lk);
  void recv_loop(Link& lk);
  void ctrl_loop(Link& lk);
  bool rccl_wanted(const Hello& h) const;
  void rccl_upgrade(Link& lk, const uint8_t* ids, bool is_parent);
  bool send_packet(Link& lk, const float* scales_host);  // post-quantize I/O
  void send_snapshot(Link& lk);
  // corr == nullptr: snapshot chunks add into values + the child-slot
  // residuals (join-state forwarding).  corr != nullptr (subtree-preserving
  // rejoin): chunks add into values + corr ONLY — live children receive a
  // single exact correction later instead of the raw snapshot.
  void recv_snapshot(int fd, float* corr = nullptr);
  void apply_packet(Link& lk, const float* scales_host);
  void link_down(Link& lk, const std::string& why, bool remote);
  void set_error(const std::string& e);
  void compute_scales(Link& lk, float* scales_host, bool lagged_valid);
  void notify_all_dirty();
  void push_scale(bool sent, float s);
};

}  // namespace shamd
