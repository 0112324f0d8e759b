// msbn C++ DDP reducer — bucketed, backward-overlapped gradient all-reduce.
//
// MI355X-native equivalent of the stock c10d::Reducer (SURVEY.md §2.2
// "reducer.hpp:45-581"), built directly on the c10d ProcessGroup C++ API
// (RCCL over xGMI on ROCm; gloo on CPU):
//
//  * autograd post-accumulate hooks registered in C++ on each parameter's
//    grad accumulator — they fire on the autograd engine's worker threads
//    (no GIL, no Python) during backward.
//  * gradients are copied into per-bucket flat buffers; when a bucket's
//    pending count hits zero, buckets launch IN ORDER onto the process
//    group, whose RCCL backend runs them on its dedicated HIP stream,
//    event-gated against the compute stream -> comm overlaps the rest of
//    backward (S7 in SURVEY.md §2.5).
//  * finalize (queued on the engine via queue_callback from the first hook)
//    waits the works and materializes averaged grads.
//  * first-iteration gradient-arrival order is recorded; rebuild_buckets()
//    re-bins parameters by that order (rank 0's order broadcast to all ranks)
//    so buckets fill in backward order — the overlap lever the stock reducer
//    calls _rebuild_buckets.
#pragma once

#include <torch/extension.h>

#include <torch/csrc/autograd/engine.h>
#include <torch/csrc/autograd/function.h>
#include <torch/csrc/autograd/variable.h>
#include <torch/csrc/autograd/utils/lambda_post_hook.h>
#include <torch/csrc/distributed/c10d/ProcessGroup.hpp>
#include <torch/csrc/distributed/c10d/Types.hpp>
#include <torch/csrc/distributed/c10d/Work.hpp>

#include <algorithm>
#include <map>
#include <memory>
#include <mutex>
#include <chrono>
#include <sstream>
#include <string>
#include <unordered_map>
#include <unordered_set>
#include <vector>

namespace msbn {

using torch::autograd::utils::LambdaPostHook;
using torch::autograd::variable_list;

// ---------------------------------------------------------------------------
// bucket assignment: greedy size-binning per (device, dtype) key, preserving
// the given order (callers pass params in reverse registration order to
// approximate backward order — stock behavior, distributed.py:1224-1230).
// ---------------------------------------------------------------------------
inline std::vector<std::vector<int64_t>> compute_bucket_assignment_by_size(
    const std::vector<at::Tensor>& tensors,
    const std::vector<size_t>& bucket_size_limits) {
  TORCH_CHECK(!bucket_size_limits.empty(), "need at least one bucket limit");
  struct Key {
    c10::DeviceType dev_type;
    c10::DeviceIndex dev_index;
    at::ScalarType dtype;
    bool operator<(const Key& o) const {
      return std::tie(dev_type, dev_index, dtype) <
             std::tie(o.dev_type, o.dev_index, o.dtype);
    }
  };
  struct Bin {
    std::vector<int64_t> indices;
    size_t bytes = 0;
    size_t limit_idx = 0;
  };
  std::map<Key, Bin> open;
  std::vector<std::pair<int64_t, std::vector<int64_t>>> done;  // (first idx, bin)

  for (int64_t i = 0; i < (int64_t)tensors.size(); ++i) {
    const auto& t = tensors[i];
    Key key{t.device().type(), t.device().index(), t.scalar_type()};
    auto& bin = open[key];
    const size_t bytes = (size_t)t.numel() * t.element_size();
    const size_t limit =
        bucket_size_limits[std::min(bin.limit_idx, bucket_size_limits.size() - 1)];
    if (!bin.indices.empty() && bin.bytes + bytes > limit) {
      done.emplace_back(bin.indices.front(), std::move(bin.indices));
      bin.indices.clear();
      bin.bytes = 0;
      bin.limit_idx++;
    }
    bin.indices.push_back(i);
    bin.bytes += bytes;
  }
  for (auto& kv : open) {
    if (!kv.second.indices.empty()) {
      done.emplace_back(kv.second.indices.front(),
                        std::move(kv.second.indices));
    }
  }
  // stable order: by first parameter index within the (reversed) list
  std::sort(done.begin(), done.end(),
            [](const auto& a, const auto& b) { return a.first < b.first; });
  std::vector<std::vector<int64_t>> out;
  out.reserve(done.size());
  for (auto& d : done) out.push_back(std::move(d.second));
  return out;
}

// ---------------------------------------------------------------------------
// coalesced broadcast: flatten tensors into <= buffer_bytes chunks, broadcast
// each from src, unflatten.  Used for init param sync (250 MiB) and the
// per-iteration buffer sync (S4/S10 in SURVEY.md §2.5).
// ---------------------------------------------------------------------------
inline void broadcast_coalesced(const c10::intrusive_ptr<c10d::ProcessGroup>& pg,
                         std::vector<at::Tensor> tensors, size_t buffer_bytes,
                         int64_t src_rank) {
  size_t i = 0;
  while (i < tensors.size()) {
    // take a run of tensors with same device+dtype up to buffer_bytes
    size_t j = i;
    size_t bytes = 0;
    const auto dev = tensors[i].device();
    const auto dt = tensors[i].scalar_type();
    while (j < tensors.size() && tensors[j].device() == dev &&
           tensors[j].scalar_type() == dt) {
      const size_t b = (size_t)tensors[j].numel() * tensors[j].element_size();
      if (j > i && bytes + b > buffer_bytes) break;
      bytes += b;
      ++j;
    }
    std::vector<at::Tensor> group(tensors.begin() + i, tensors.begin() + j);
    at::Tensor flat;
    if (group.size() == 1 && group[0].is_contiguous()) {
      flat = group[0].reshape({-1});
    } else {
      std::vector<at::Tensor> flats;
      flats.reserve(group.size());
      for (auto& t : group) flats.push_back(t.reshape({-1}));
      flat = at::cat(flats);
    }
    {
      std::vector<at::Tensor> v{flat};
      c10d::BroadcastOptions opts;
      opts.rootRank = src_rank;
      pg->broadcast(v, opts)->wait();
    }
    if (!(group.size() == 1 && group[0].is_contiguous())) {
      int64_t off = 0;
      for (auto& t : group) {
        t.copy_(flat.narrow(0, off, t.numel()).view_as(t));
        off += t.numel();
      }
    }
    i = j;
  }
}

// Broadcast rank-0 parameter metadata (ndim + sizes per param) and compare.
// Every rank learns about any mismatch COLLECTIVELY (min-reduced agreement
// flags) before the variable-size broadcast and before returning, so a
// mismatch raises a clean error on ALL ranks instead of hanging the ranks
// whose local check happened to pass (stock _verify_param_shape_across_processes, S3).
inline void verify_params_across_processes(
    const c10::intrusive_ptr<c10d::ProcessGroup>& pg,
    const std::vector<at::Tensor>& params) {
  std::vector<int64_t> meta;
  meta.push_back((int64_t)params.size());
  for (const auto& p : params) {
    meta.push_back(p.dim());
    for (auto s : p.sizes()) meta.push_back(s);
  }
  const auto dev =
      params.empty() ? at::Device(at::kCPU) : params[0].device();
  auto local = at::tensor(meta, at::TensorOptions().dtype(at::kLong)).to(dev);
  auto sz = at::tensor({(int64_t)meta.size()},
                       at::TensorOptions().dtype(at::kLong))
                .to(dev);
  {
    std::vector<at::Tensor> v{sz};
    c10d::BroadcastOptions bo;
    bo.rootRank = 0;
    pg->broadcast(v, bo)->wait();
  }
  const int64_t root_len = sz.cpu().item<int64_t>();
  const bool len_ok = (int64_t)meta.size() == root_len;
  // agreement round 1: does every rank have rank 0's metadata length?
  auto ok = at::tensor({len_ok ? (int64_t)1 : (int64_t)0},
                       at::TensorOptions().dtype(at::kLong))
                .to(dev);
  {
    std::vector<at::Tensor> v{ok};
    c10d::AllreduceOptions ao;
    ao.reduceOp = c10d::ReduceOp::MIN;
    pg->allreduce(v, ao)->wait();
  }
  TORCH_CHECK(ok.cpu().item<int64_t>() == 1,
              "msbn DDP: parameter metadata length differs from rank 0 ",
              len_ok ? std::string("on another rank")
                     : ("(" + std::to_string(meta.size()) + " vs " +
                        std::to_string(root_len) + ") locally"),
              " — all processes must hold identical models");
  auto root = local.clone();
  {
    std::vector<at::Tensor> v{root};
    c10d::BroadcastOptions bo;
    bo.rootRank = 0;
    pg->broadcast(v, bo)->wait();
  }
  const bool shapes_ok = root.cpu().equal(local.cpu());
  // agreement round 2: same-length but different shapes also raises everywhere
  auto ok2 = at::tensor({shapes_ok ? (int64_t)1 : (int64_t)0},
                        at::TensorOptions().dtype(at::kLong))
                 .to(dev);
  {
    std::vector<at::Tensor> v{ok2};
    c10d::AllreduceOptions ao;
    ao.reduceOp = c10d::ReduceOp::MIN;
    pg->allreduce(v, ao)->wait();
  }
  TORCH_CHECK(ok2.cpu().item<int64_t>() == 1,
              "msbn DDP: parameter shapes differ from rank 0 — all processes "
              "must hold identical models");
}

// ---------------------------------------------------------------------------
// Reducer
// ---------------------------------------------------------------------------
class Reducer : public std::enable_shared_from_this<Reducer> {
 public:
  Reducer(std::vector<at::Tensor> params,
          std::vector<std::vector<int64_t>> bucket_indices,
          c10::intrusive_ptr<c10d::ProcessGroup> pg, bool gradient_as_bucket_view,
          size_t first_bucket_bytes, size_t bucket_bytes)
      : params_(std::move(params)),
        pg_(std::move(pg)),
        gradient_as_bucket_view_(gradient_as_bucket_view),
        first_bucket_bytes_(first_bucket_bytes),
        bucket_bytes_(bucket_bytes),
        div_factor_((double)pg_->getSize()) {
    initialize_buckets(std::move(bucket_indices));
    attach_hooks();
  }

  ~Reducer() {
    *alive_ = false;
    // The last shared_ptr ref can drop on an autograd engine thread (the
    // queued finalize lambda holds one) — python comm-hook objects must be
    // released under the GIL.
    if (py_state_ || py_hook_ || py_bucket_cls_) {
      py::gil_scoped_acquire gil;
      py_state_ = py::object();
      py_hook_ = py::object();
      py_bucket_cls_ = py::object();
    }
  }

  void prepare_for_backward(const std::vector<int64_t>& unused_params) {
    std::lock_guard<std::mutex> lock(mutex_);
    expect_autograd_hooks_ = true;
    finalize_queued_ = false;
    next_bucket_ = 0;
    arrival_order_.clear();
    for (auto& b : buckets_) {
      b.pending = b.param_indices.size();
      b.ready = false;
      b.launched = false;
    }
    std::fill(param_ready_.begin(), param_ready_.end(), false);
    // Params known-unused this iteration contribute zeros (find_unused path).
    for (auto i : unused_params) {
      zero_view_locked(i);
      mark_ready_locked(i, /*from_hook=*/false);
    }
  }

  void set_grad_sync_enabled(bool enabled) {
    std::lock_guard<std::mutex> lock(mutex_);
    sync_enabled_ = enabled;
  }

  // Gradient-compression hook support (stock register_comm_hook's builtin
  // fp16/bf16 compress): buckets are cast to comm_dtype for the all-reduce
  // and cast back before the grad copy-out.
  void set_comm_dtype(c10::optional<at::ScalarType> dtype) {
    std::lock_guard<std::mutex> lock(mutex_);
    comm_dtype_ = dtype;
  }

  // Gradient averaging divisor; defaults to world size.  DDP.join() with
  // divide_by_initial_world_size=False sets this to the per-iteration count
  // of ranks still contributing real data (stock reducer div_factor_).
  void set_div_factor(double f) {
    std::lock_guard<std::mutex> lock(mutex_);
    div_factor_ = f;
  }

  // Arbitrary Python comm hook (stock register_comm_hook's general form).
  // The hook REPLACES the built-in allreduce+divide: it receives
  // (state, GradBucket) and returns a torch.futures.Future resolving to the
  // reduced flat tensor.  Python hooks run at FINALIZE, in bucket order, on
  // the engine-callback thread (documented caveat: no backward overlap —
  // the C++ fast path with the builtin hooks keeps the overlap).
  void set_python_comm_hook(py::object state, py::object hook,
                            py::object bucket_cls) {
    std::lock_guard<std::mutex> lock(mutex_);
    py_state_ = std::move(state);
    py_hook_ = std::move(hook);
    py_bucket_cls_ = std::move(bucket_cls);
    has_py_hook_ = !py_hook_.is_none();
  }

  double div_factor() const { return div_factor_; }

  // Walk the autograd graph backward from `outputs` on the C++ side (no
  // Python, no GIL churn) and return indices of parameters whose grad
  // accumulator is NOT reachable — the find_unused_parameters search the
  // stock reducer does in prepare_for_backward.
  std::vector<int64_t> find_unused(const std::vector<at::Tensor>& outputs) {
    std::unordered_map<const torch::autograd::Node*, int64_t> acc_to_idx;
    acc_to_idx.reserve(grad_accumulators_.size());
    for (size_t i = 0; i < grad_accumulators_.size(); ++i) {
      acc_to_idx.emplace(grad_accumulators_[i].get(), (int64_t)i);
    }
    std::unordered_set<const torch::autograd::Node*> seen;
    std::vector<torch::autograd::Node*> stack;
    for (const auto& out : outputs) {
      if (!out.defined() || !out.requires_grad()) continue;
      auto edge = torch::autograd::impl::gradient_edge(out);
      if (edge.function) stack.push_back(edge.function.get());
    }
    std::vector<bool> used(params_.size(), false);
    while (!stack.empty()) {
      auto* fn = stack.back();
      stack.pop_back();
      if (!seen.insert(fn).second) continue;
      auto it = acc_to_idx.find(fn);
      if (it != acc_to_idx.end()) used[it->second] = true;
      for (const auto& next : fn->next_edges()) {
        if (next.function) stack.push_back(next.function.get());
      }
    }
    std::vector<int64_t> unused;
    for (size_t i = 0; i < params_.size(); ++i)
      if (!used[i]) unused.push_back((int64_t)i);
    return unused;
  }

  // Debug guard (stock TORCH_NCCL_NAN_CHECK analog): raise before shipping a
  // bucket containing non-finite gradients.  Costs a device reduction +
  // host sync per bucket — debugging only.
  void set_nan_check(bool enabled) {
    std::lock_guard<std::mutex> lock(mutex_);
    nan_check_ = enabled;
  }

  // Called from the autograd post hook of parameter i.
  void autograd_hook(int64_t i) {
    std::lock_guard<std::mutex> lock(mutex_);
    if (!expect_autograd_hooks_ || !sync_enabled_) return;
    if (!param_ready_[i] && arrival_order_.size() < params_.size()) {
      arrival_order_.push_back(i);
      const int64_t now = std::chrono::duration_cast<std::chrono::nanoseconds>(
                              std::chrono::steady_clock::now().time_since_epoch())
                              .count();
      if (arrival_order_.size() == 1) first_ready_ns_ = now;
      last_ready_ns_[i] = now;
    }
    if (!finalize_queued_) {
      finalize_queued_ = true;
      auto self = shared_from_this();
      torch::autograd::Engine::get_default_engine().queue_callback(
          [self] { self->finalize_backward(); });
    }
    copy_grad_to_view_locked(i);
    mark_ready_locked(i, /*from_hook=*/true);
  }

  void finalize_backward() {
    {
      std::lock_guard<std::mutex> lock(mutex_);
      expect_autograd_hooks_ = false;
      // every bucket must have launched
      if (next_bucket_ != buckets_.size()) {
        std::ostringstream oss;
        oss << "msbn DDP: expected all gradient buckets to be ready at the "
               "end of backward ("
            << next_bucket_ << "/" << buckets_.size()
            << " launched). Some parameters did not receive gradients; unused "
               "parameter indices: [";
        for (size_t i = 0; i < params_.size(); ++i)
          if (!param_ready_[i]) oss << i << ", ";
        oss << "]. Run the DDP wrapper with find_unused_parameters=True.";
        finalize_queued_ = false;
        TORCH_CHECK(false, oss.str());
      }
    }
    // Python comm hooks run here, OUTSIDE the reducer mutex (GIL + mutex
    // inversion with other Python threads otherwise) — launch all, then
    // wait in bucket order.  No hooks can fire concurrently:
    // expect_autograd_hooks_ is already false.
    if (has_py_hook_) run_python_hooks();
    std::lock_guard<std::mutex> lock(mutex_);
    for (auto& b : buckets_) {
      if (b.work) {
        b.work->wait();
        b.work.reset();
      }
      if (b.wire.defined()) {
        b.flat.copy_(b.wire, /*non_blocking=*/true);
        b.wire = at::Tensor();
      }
      for (size_t k = 0; k < b.param_indices.size(); ++k) {
        auto& p = params_[b.param_indices[k]];
        auto view = b.views[k];
        auto grad = p.mutable_grad();
        if (gradient_as_bucket_view_) {
          if (!grad.defined() || grad.data_ptr() != view.data_ptr()) {
            p.mutable_grad() = view.view_as(p);
          }
        } else {
          if (grad.defined()) {
            grad.copy_(view.view_as(p), /*non_blocking=*/true);
          } else {
            p.mutable_grad() = view.view_as(p).clone();
          }
        }
      }
    }
    iterations_++;
    finalize_queued_ = false;
  }

  // Re-bin parameters by the recorded first-iteration arrival order; rank 0's
  // binning is broadcast so every rank allreduces identical buckets.
  // Returns true if buckets changed.  Call OUTSIDE backward.
  bool rebuild_buckets() {
    std::unique_lock<std::mutex> lock(mutex_);
    if (rebuilt_ || arrival_order_.size() != params_.size()) return false;
    std::vector<int64_t> order = arrival_order_;
    lock.unlock();

    // broadcast rank 0's order
    auto dev = params_[0].device();
    auto t = at::tensor(order, at::TensorOptions().dtype(at::kLong)).to(dev);
    {
      std::vector<at::Tensor> v{t};
      c10d::BroadcastOptions bo;
      bo.rootRank = 0;
      pg_->broadcast(v, bo)->wait();
    }
    auto tc = t.cpu();
    auto acc = tc.accessor<int64_t, 1>();
    std::vector<at::Tensor> ordered;
    std::vector<int64_t> order2((size_t)acc.size(0));
    ordered.reserve(params_.size());
    for (int64_t k = 0; k < acc.size(0); ++k) {
      order2[k] = acc[k];
      ordered.push_back(params_[acc[k]]);
    }
    auto bins = compute_bucket_assignment_by_size(
        ordered, {first_bucket_bytes_, bucket_bytes_});
    // translate back to original param indices
    for (auto& bin : bins)
      for (auto& idx : bin) idx = order2[idx];

    lock.lock();
    initialize_buckets(std::move(bins));
    rebuilt_ = true;
    return true;
  }

  std::vector<std::vector<int64_t>> get_bucket_indices() const {
    std::vector<std::vector<int64_t>> out;
    for (auto& b : buckets_) out.push_back(b.param_indices);
    return out;
  }

  // Per-parameter grad-ready timestamps (us since the first hook of the
  // iteration) from the LAST completed backward — the stock reducer's
  // backward_stats_, the data for bucket-order tuning.
  std::vector<double> get_backward_stats() const {
    std::vector<double> out(params_.size(), -1.0);
    for (size_t i = 0; i < params_.size(); ++i) {
      if (last_ready_ns_[i] >= 0)
        out[i] = (double)(last_ready_ns_[i] - first_ready_ns_) * 1e-3;
    }
    return out;
  }

  int64_t iterations() const { return iterations_; }
  bool rebuilt() const { return rebuilt_; }

 private:
  struct Bucket {
    at::Tensor flat;
    std::vector<at::Tensor> views;
    std::vector<int64_t> param_indices;
    size_t pending = 0;
    bool ready = false;
    bool launched = false;
    c10::intrusive_ptr<c10d::Work> work;
    at::Tensor wire;  // comm-dtype staging (gradient compression)
  };

  void initialize_buckets(std::vector<std::vector<int64_t>> bucket_indices) {
    buckets_.clear();
    param_to_bucket_.assign(params_.size(), {-1, -1});
    for (auto& idxs : bucket_indices) {
      Bucket b;
      b.param_indices = idxs;
      int64_t total = 0;
      for (auto i : idxs) total += params_[i].numel();
      b.flat = at::zeros({total}, params_[idxs[0]].options());
      int64_t off = 0;
      for (size_t k = 0; k < idxs.size(); ++k) {
        const auto n = params_[idxs[k]].numel();
        b.views.push_back(b.flat.narrow(0, off, n));
        param_to_bucket_[idxs[k]] = {(int64_t)buckets_.size(), (int64_t)k};
        off += n;
      }
      b.pending = idxs.size();
      buckets_.push_back(std::move(b));
    }
    param_ready_.assign(params_.size(), false);
    last_ready_ns_.assign(params_.size(), -1);
    for (size_t i = 0; i < params_.size(); ++i) {
      TORCH_CHECK(param_to_bucket_[i].first >= 0,
                  "msbn Reducer: parameter ", i, " missing from buckets");
    }
  }

  void attach_hooks() {
    grad_accumulators_.reserve(params_.size());
    for (size_t i = 0; i < params_.size(); ++i) {
      auto acc = torch::autograd::impl::grad_accumulator(params_[i]);
      TORCH_CHECK(acc, "msbn Reducer: parameter ", i,
                  " has no grad accumulator (requires_grad=False?)");
      auto alive = alive_;
      // `this` is safe while *alive: Reducer outlives hooks via DDP holder.
      Reducer* self = this;
      const int64_t idx = (int64_t)i;
      acc->add_post_hook(std::make_unique<LambdaPostHook>(
          [self, alive, idx](const variable_list& outputs,
                             const variable_list& /*inputs*/) -> variable_list {
            if (*alive) self->autograd_hook(idx);
            return outputs;
          }));
      grad_accumulators_.push_back(std::move(acc));
    }
  }

  void copy_grad_to_view_locked(int64_t i) {
    auto [bi, vi] = param_to_bucket_[i];
    auto& b = buckets_[bi];
    auto view = b.views[vi];
    auto& p = params_[i];
    auto grad = p.grad();
    if (!grad.defined()) {
      view.zero_();
      return;
    }
    if (grad.data_ptr() == view.data_ptr()) return;  // bucket-view grad
    view.copy_(grad.reshape({-1}), /*non_blocking=*/true);
  }

  void zero_view_locked(int64_t i) {
    auto [bi, vi] = param_to_bucket_[i];
    buckets_[bi].views[vi].zero_();
  }

  void mark_ready_locked(int64_t i, bool from_hook) {
    (void)from_hook;
    if (param_ready_[i]) return;  // e.g. shared params firing twice
    param_ready_[i] = true;
    auto [bi, vi] = param_to_bucket_[i];
    auto& b = buckets_[bi];
    TORCH_CHECK(b.pending > 0, "msbn Reducer: bucket ", bi,
                " marked ready too many times");
    if (--b.pending == 0) b.ready = true;
    // launch in fixed bucket order for cross-rank consistency
    while (next_bucket_ < buckets_.size() && buckets_[next_bucket_].ready &&
           !buckets_[next_bucket_].launched) {
      launch_bucket_locked(buckets_[next_bucket_]);
      next_bucket_++;
    }
  }

  void launch_bucket_locked(Bucket& b) {
    b.launched = true;
    if (nan_check_) {
      TORCH_CHECK(at::isfinite(b.flat).all().item<bool>(),
                  "msbn Reducer: non-finite gradient detected in bucket "
                  "(params ", b.param_indices, ") before all-reduce");
    }
    // a Python comm hook REPLACES the allreduce+divide; it runs at finalize
    if (has_py_hook_) return;
    if (div_factor_ != 1.0) b.flat.div_(div_factor_);
    c10d::AllreduceOptions opts;
    if (comm_dtype_.has_value() &&
        b.flat.scalar_type() != *comm_dtype_) {
      b.wire = b.flat.to(*comm_dtype_);
      std::vector<at::Tensor> v{b.wire};
      b.work = pg_->allreduce(v, opts);
    } else {
      b.wire = at::Tensor();
      std::vector<at::Tensor> v{b.flat};
      b.work = pg_->allreduce(v, opts);
    }
  }

  void run_python_hooks() {
    py::gil_scoped_acquire gil;
    std::vector<py::object> futs;
    futs.reserve(buckets_.size());
    for (size_t i = 0; i < buckets_.size(); ++i) {
      auto& b = buckets_[i];
      py::list views;
      for (auto& v : b.views) views.append(v);
      py::object bucket =
          py_bucket_cls_(b.flat, views, (int64_t)i,
                         i + 1 == buckets_.size());
      futs.push_back(py_hook_(py_state_, bucket));
    }
    for (size_t i = 0; i < buckets_.size(); ++i) {
      py::object res = futs[i].attr("wait")();
      auto out = res.cast<at::Tensor>();
      auto& flat = buckets_[i].flat;
      if (out.data_ptr() != flat.data_ptr()) {
        flat.copy_(out.reshape({-1}).to(flat.scalar_type()),
                   /*non_blocking=*/true);
      }
    }
  }

  std::vector<at::Tensor> params_;
  c10::intrusive_ptr<c10d::ProcessGroup> pg_;
  bool gradient_as_bucket_view_;
  size_t first_bucket_bytes_;
  size_t bucket_bytes_;
  double div_factor_;

  std::vector<Bucket> buckets_;
  std::vector<std::pair<int64_t, int64_t>> param_to_bucket_;
  std::vector<bool> param_ready_;
  std::vector<std::shared_ptr<torch::autograd::Node>> grad_accumulators_;
  std::shared_ptr<bool> alive_ = std::make_shared<bool>(true);

  c10::optional<at::ScalarType> comm_dtype_;
  bool nan_check_ = false;
  bool has_py_hook_ = false;
  py::object py_state_;
  py::object py_hook_;
  py::object py_bucket_cls_;

  std::vector<int64_t> last_ready_ns_;   // steady_clock ns per param
  int64_t first_ready_ns_ = 0;

  std::mutex mutex_;
  bool expect_autograd_hooks_ = false;
  bool sync_enabled_ = true;
  bool finalize_queued_ = false;
  bool rebuilt_ = false;
  size_t next_bucket_ = 0;
  std::vector<int64_t> arrival_order_;
  int64_t iterations_ = 0;
};

}  // namespace msbn
