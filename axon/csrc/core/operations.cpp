#include "operations.h"

#include <algorithm>
#include <atomic>
#include <condition_variable>
#include <deque>
#include <future>
#include <thread>
#include <unordered_map>

#include "controller.h"
#include "cpu_backend.h"
#include "message.h"
#include "parameter_manager.h"
#include "timeline.h"

namespace axon {

namespace {

constexpr const char* kShutdownBarrier = "__axon.shutdown.barrier__";

struct Global {
  std::atomic<bool> initialized{false};
  std::atomic<bool> running{false};
  std::atomic<bool> shutdown_requested{false};
  int rank = 0, size = 1, local_rank = 0;
  std::string control_path;
  double cycle_ms = 1.0;

  std::thread bg;
  std::mutex mu;
  std::condition_variable cv;
  std::deque<TensorEntry> queue;

  // Background-thread-only state.
  std::unique_ptr<ControlPlane> cp;
  std::unique_ptr<Controller> ctl;
  std::unique_ptr<CpuBackend> cpu;
  std::unique_ptr<GpuBackend> gpu;
  std::unique_ptr<ParameterManager> pm;
  std::unordered_map<std::string, TensorEntry> pending;
  bool joined = false;
  TensorEntry join_entry;
  int32_t join_device = -1;
  bool shutdown_barrier_sent = false;

  Timeline tl;
  std::mutex err_mu;
  std::string fatal;

  // Process-set registry: written by the user thread (collective
  // registration), read by the background thread each cycle.
  std::mutex ps_mu;
  std::vector<ProcessSetInfo> psets;
  int ps_version = 1;  // bumped on register AND remove (controller rescans)

  // End-to-end autotune signal: optimizer steps completed on this rank
  // (written by the user thread via pm_step_mark, drained by the bg loop).
  std::atomic<int64_t> step_marks{0};
};

Global g;

void set_fatal(const std::string& msg) {
  std::lock_guard<std::mutex> lk(g.err_mu);
  if (g.fatal.empty()) g.fatal = msg;
}

Request make_request(const TensorEntry& e) {
  Request q;
  q.rank = g.rank;
  q.op = e.op;
  q.red = e.red;
  q.name = e.name;
  if (e.in.defined()) {
    q.dtype = dtype_id(e.in.scalar_type());
    q.shape = e.in.sizes().vec();
  }
  q.wire_dtype =
      e.wire_dtype == at::ScalarType::Undefined ? 0 : dtype_id(e.wire_dtype);
  q.device = e.device;
  q.root_rank = e.root_rank;
  q.process_set = e.process_set;
  q.splits = e.splits;
  return q;
}

TensorEntry make_zero_entry(const Response& r, size_t i) {
  // A joined rank participates in peers' collectives with zero tensors
  // (SURVEY.md §3.5 hvd.join semantics). Allreduce-shaped only.
  TensorEntry e;
  e.name = r.names[i];
  e.op = r.op;
  e.red = r.red;
  auto opts = at::TensorOptions().dtype(id_dtype(r.dtype));
  if (r.device >= 0)
    opts = opts.device(at::Device(at::kCUDA, static_cast<int8_t>(g.join_device)));
  e.in = at::zeros({r.numels.at(i)}, opts);
  e.out = e.in;
  e.device = r.device >= 0 ? g.join_device : -1;
  e.process_set = r.process_set;
  return e;
}

void complete(TensorEntry& e, const Status& st) {
  g.tl.end(e.name);
  if (e.callback) e.callback(st);
}

void execute_response(const Response& r) {
  if (r.op == OpType::JOIN) {
    if (g.joined) {
      if (g.join_entry.out.defined())
        g.join_entry.out.fill_(static_cast<int64_t>(r.last_joined));
      complete(g.join_entry, Status::OK());
      g.join_entry = TensorEntry{};
      g.joined = false;
    }
    return;
  }

  ProcessSetInfo psi = ps_get(r.process_set);
  if (!psi.valid) {
    if (!r.error.empty()) {
      // Error response for a removed set: deliver it to whichever of the
      // named tensors this rank actually submitted; others never will be.
      for (const auto& name : r.names) {
        auto it = g.pending.find(name);
        if (it != g.pending.end()) {
          complete(it->second, Status::Error(r.error));
          g.pending.erase(it);
        }
      }
      return;
    }
    AXON_LOG(LogLevel::ERROR, "axon: response for unknown process set " +
                                  std::to_string(r.process_set));
    return;
  }
  const bool member = psi.member(g.rank);
  // Non-members are uninvolved — except rank 0, which is the control-plane
  // hub: for CPU-data ops of sets it doesn't belong to it acts as a pure
  // relay, and for GPU ops it relays the RCCL unique-id rendezvous. The
  // global response order is identical on every rank, so the relay executes
  // at the same logical point as the members.
  if (!member && g.rank != 0) return;

  std::vector<TensorEntry> entries;
  if (member) {
    entries.reserve(r.names.size());
    for (size_t i = 0; i < r.names.size(); ++i) {
      auto it = g.pending.find(r.names[i]);
      if (it != g.pending.end()) {
        entries.push_back(std::move(it->second));
        g.pending.erase(it);
      } else if (g.joined && r.op == OpType::ALLREDUCE) {
        entries.push_back(make_zero_entry(r, i));
      } else {
        AXON_LOG(LogLevel::ERROR,
                 "axon: response for unknown tensor '" + r.names[i] + "'");
        return;
      }
    }
  }

  if (!r.error.empty()) {
    for (auto& e : entries) complete(e, Status::Error(r.error));
    return;  // errors are delivered locally on every rank — nothing to relay
  }

  if (r.op == OpType::BARRIER) {
    bool is_shutdown = !r.names.empty() && r.names[0] == kShutdownBarrier;
    for (auto& e : entries) complete(e, Status::OK());
    if (is_shutdown && g.rank == 0) g.ctl->coordinator_shutdown();
    return;
  }

  for (auto& e : entries) g.tl.begin(e.name, r.device >= 0 ? "QUEUE_GPU" : "CPU_OP");

  if (r.device < 0) {
    Status st = g.cpu->execute(r, entries);  // member or rank-0 relay
    for (auto& e : entries) complete(e, st);
  } else {
    if (!g.gpu)
      g.gpu.reset(create_rccl_backend(g.rank, g.size, g.cp.get(), &g.tl));
    g.gpu->execute(r, std::move(entries));  // async; callbacks via poller
    // (rank-0 non-member: execute() only relays the set's unique-id
    // rendezvous on its first op, then returns with nothing enqueued)
  }
}

void abort_pending(const std::string& why) {
  for (auto& kv : g.pending) complete(kv.second, Status::Aborted(why));
  g.pending.clear();
  std::deque<TensorEntry> leftover;
  {
    std::lock_guard<std::mutex> lk(g.mu);
    leftover.swap(g.queue);
  }
  for (auto& e : leftover) complete(e, Status::Aborted(why));
  if (g.joined) {
    complete(g.join_entry, Status::Aborted(why));
    g.joined = false;
  }
}

void bg_loop(std::promise<void> ready) {
  try {
    double timeout = env_double("AXON_START_TIMEOUT", 120.0);
    g.cp = std::make_unique<ControlPlane>();
    if (g.rank == 0)
      g.cp->init_coordinator(g.control_path, g.size, timeout);
    else
      g.cp->init_worker(g.control_path, g.rank, timeout);
    g.ctl = std::make_unique<Controller>(g.rank, g.size, g.cp.get());
    g.cpu = std::make_unique<CpuBackend>(g.rank, g.size, g.cp.get());
    if (g.rank == 0)
      g.pm = std::make_unique<ParameterManager>(g.ctl->fusion_threshold(),
                                                g.cycle_ms);
    g.tl.init(g.rank);
    ready.set_value();
  } catch (...) {
    ready.set_exception(std::current_exception());
    return;
  }

  auto cycle = std::chrono::duration<double, std::milli>(g.cycle_ms);
  // Bounded shutdown: a rank exiting early (sys.exit, exception) must not
  // hang in atexit waiting for peers that never call shutdown(). After the
  // timeout the loop force-exits; closing our sockets fail-fasts the peers
  // (SURVEY.md §5.3b).
  const double shutdown_timeout = env_double("AXON_SHUTDOWN_TIMEOUT", 30.0);
  std::chrono::steady_clock::time_point shutdown_since{};
  bool shutdown = false;
  try {
    while (!shutdown) {
      std::vector<TensorEntry> fresh;
      {
        std::unique_lock<std::mutex> lk(g.mu);
        if (g.queue.empty() && !g.shutdown_requested.load())
          g.cv.wait_for(lk, cycle);
        while (!g.queue.empty()) {
          fresh.push_back(std::move(g.queue.front()));
          g.queue.pop_front();
        }
      }

      std::vector<Request> reqs;
      for (auto& e : fresh) {
        if (e.op == OpType::JOIN) {
          if (g.joined) {
            complete(e, Status::Error("axon: join already in progress"));
            continue;
          }
          g.joined = true;
          g.join_entry = std::move(e);
          g.join_device = g.join_entry.device;
          reqs.push_back(make_request(g.join_entry));
          continue;
        }
        if (g.pending.count(e.name)) {
          complete(e, Status::Error("axon: tensor name '" + e.name +
                                    "' already in flight — pass a unique name"));
          continue;
        }
        g.tl.begin(e.name, "NEGOTIATE");
        reqs.push_back(make_request(e));
        g.pending.emplace(e.name, std::move(e));
      }

      if (g.shutdown_requested.load() && !g.shutdown_barrier_sent) {
        TensorEntry e;
        e.op = OpType::BARRIER;
        e.name = kShutdownBarrier;
        reqs.push_back(make_request(e));
        g.pending.emplace(e.name, std::move(e));
        g.shutdown_barrier_sent = true;
        shutdown_since = std::chrono::steady_clock::now();
      }
      if (g.shutdown_barrier_sent && !shutdown &&
          std::chrono::duration<double>(std::chrono::steady_clock::now() -
                                        shutdown_since).count() > shutdown_timeout) {
        AXON_LOG(LogLevel::WARN,
                 "axon: shutdown barrier timed out after " +
                     std::to_string(shutdown_timeout) +
                     "s (peers still running?) — forcing local shutdown");
        break;
      }

      // Autotuner (rank 0): score by GPU-allreduced bytes/sec, ship knob
      // updates with the next broadcast (SURVEY.md §2.1 parameter manager).
      if (g.pm && g.pm->enabled()) {
        g.pm->record_steps(g.step_marks.exchange(0, std::memory_order_relaxed));
        int64_t thr = g.ctl->fusion_threshold();
        double cyc = g.cycle_ms;
        if (g.pm->maybe_update(&thr, &cyc)) {
          g.ctl->set_fusion_threshold(thr);
          g.ctl->announce_cycle_time(cyc);
        }
      }

      double new_cycle_ms = 0.0;
      auto responses = g.ctl->run_cycle(reqs, &shutdown, &new_cycle_ms);
      if (new_cycle_ms > 0) {
        g.cycle_ms = new_cycle_ms;
        cycle = std::chrono::duration<double, std::milli>(g.cycle_ms);
      }
      for (auto& r : responses) {
        // Score by allreduced bytes/sec on whichever data plane is in
        // use (the CPU path also makes the tuner testable without a GPU).
        if (g.pm && r.op == OpType::ALLREDUCE && r.error.empty()) {
          int64_t bytes = 0;
          size_t esz = dtype_size(id_dtype(r.wire_dtype ? r.wire_dtype : r.dtype));
          for (auto nn : r.numels) bytes += nn * static_cast<int64_t>(esz);
          g.pm->record_bytes(bytes);
        }
        execute_response(r);
      }
      g.tl.mark_cycle();
    }
  } catch (const std::exception& ex) {
    set_fatal(ex.what());
    AXON_LOG(LogLevel::ERROR, std::string("axon background loop failed: ") + ex.what());
  }

  if (g.gpu) g.gpu->finalize();
  abort_pending(g.fatal.empty() ? "axon shutting down" : g.fatal);
  if (g.cp) g.cp->close_all();
  g.running = false;
}

}  // namespace

void initialize(int rank_, int size_, int local_rank_, const std::string& path) {
  if (g.initialized.load())
    throw std::runtime_error("axon already initialized");
  if (size_ < 1 || size_ > 32)
    throw std::runtime_error(
        "axon: world size " + std::to_string(size_) +
        " unsupported — rank masks are 32-bit (single-node target; max 32)");
  g.rank = rank_;
  g.size = size_;
  g.local_rank = local_rank_;
  g.control_path = path;
  g.cycle_ms = env_double("AXON_CYCLE_TIME", 1.0);
  g.shutdown_requested = false;
  g.shutdown_barrier_sent = false;
  {
    std::lock_guard<std::mutex> lk(g.err_mu);
    g.fatal.clear();
  }
  {
    std::lock_guard<std::mutex> lk(g.ps_mu);
    g.psets.clear();
    ProcessSetInfo global;
    global.id = 0;
    for (int r = 0; r < size_; ++r) {
      global.ranks.push_back(r);
      global.mask |= (1u << r);
    }
    global.my_index = rank_;
    g.psets.push_back(std::move(global));
  }
  std::promise<void> ready;
  auto fut = ready.get_future();
  g.running = true;
  g.bg = std::thread(bg_loop, std::move(ready));
  try {
    fut.get();
  } catch (...) {
    g.bg.join();
    g.running = false;
    throw;
  }
  g.initialized = true;
}

void shutdown() {
  if (!g.initialized.load()) return;
  g.shutdown_requested = true;
  g.cv.notify_all();
  if (g.bg.joinable()) g.bg.join();
  g.initialized = false;
  g.gpu.reset();
  g.cpu.reset();
  g.pm.reset();
  g.ctl.reset();
  g.cp.reset();
}

bool is_initialized() { return g.initialized.load(); }
int rank() { return g.rank; }
int size() { return g.size; }
int local_rank() { return g.local_rank; }

std::string fatal_error() {
  std::lock_guard<std::mutex> lk(g.err_mu);
  return g.fatal;
}

int ps_register(const std::vector<int>& ranks) {
  std::vector<int> sorted = ranks;
  std::sort(sorted.begin(), sorted.end());
  sorted.erase(std::unique(sorted.begin(), sorted.end()), sorted.end());
  if (sorted.empty() || sorted.front() < 0 || sorted.back() >= g.size)
    throw std::runtime_error("axon: process-set ranks out of range");
  ProcessSetInfo psi;
  psi.ranks = std::move(sorted);
  for (int r : psi.ranks) psi.mask |= (1u << r);
  psi.my_index = psi.index_of(g.rank);
  std::lock_guard<std::mutex> lk(g.ps_mu);
  psi.id = static_cast<int>(g.psets.size());
  g.psets.push_back(psi);
  ++g.ps_version;
  return psi.id;
}

ProcessSetInfo ps_get(int id) {
  std::lock_guard<std::mutex> lk(g.ps_mu);
  if (id < 0 || id >= static_cast<int>(g.psets.size())) {
    ProcessSetInfo bad;
    bad.id = id;
    bad.valid = false;
    return bad;
  }
  return g.psets[id];
}

bool ps_known(int id) {
  std::lock_guard<std::mutex> lk(g.ps_mu);
  return id >= 0 && id < static_cast<int>(g.psets.size()) && g.psets[id].valid;
}

void ps_remove(int id) {
  std::lock_guard<std::mutex> lk(g.ps_mu);
  if (id <= 0 || id >= static_cast<int>(g.psets.size()))
    throw std::runtime_error("axon: cannot remove process set " + std::to_string(id));
  g.psets[id].valid = false;
  ++g.ps_version;
}

int ps_count() {
  std::lock_guard<std::mutex> lk(g.ps_mu);
  return static_cast<int>(g.psets.size());
}

void pm_step_mark() {
  g.step_marks.fetch_add(1, std::memory_order_relaxed);
}

void timeline_start(const std::string& path) { g.tl.start(path); }

void timeline_stop() { g.tl.stop(); }

int ps_version() {
  std::lock_guard<std::mutex> lk(g.ps_mu);
  return g.ps_version;
}

void enqueue(TensorEntry entry) {
  if (!g.initialized.load())
    throw std::runtime_error("axon not initialized — call axon.init() first");
  if (!g.running.load()) {
    std::string f = fatal_error();
    throw std::runtime_error(f.empty() ? "axon core stopped" : "axon core stopped: " + f);
  }
  {
    std::lock_guard<std::mutex> lk(g.mu);
    g.queue.push_back(std::move(entry));
  }
  g.cv.notify_all();
}

}  // namespace axon
