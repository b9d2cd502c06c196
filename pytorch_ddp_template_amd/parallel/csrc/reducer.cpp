// Native bucketed gradient reducer — the MI355X framework's equivalent of
// the DDP reducer the reference template exercises through
// torch.nn.parallel.DistributedDataParallel (reference ddp.py:194-195).
//
// Design:
//  * Buckets are assigned in REVERSE parameter registration order (the order
//    gradients become ready during backward), capped at a byte budget chosen
//    for the 8-GPU xGMI topology: each MI355X has 7 point-to-point links at
//    ~153 GB/s, and RCCL ring all-reduce is per-link bound, so buckets must
//    be large enough to saturate a link when split across the ring while
//    staying numerous enough to overlap with the remaining backward.
//  * Gradients are VIEWS into one flat buffer per bucket
//    (gradient-as-bucket-view): backward accumulates directly into the
//    bucket, so bucket assembly costs zero extra kernels / zero extra HBM
//    traffic.
//  * When the last gradient of a bucket is accumulated, the all-reduce for
//    that flat buffer is launched asynchronously (RCCL on its comm stream on
//    ROCm; gloo for the CPU plumbing rung) and overlaps with the rest of
//    backward.  finalize() waits on all outstanding work and divides by the
//    world size.
//  * The per-parameter ready hooks are registered in C++ directly on the
//    AccumulateGrad autograd nodes (install_hooks): the entire
//    grad-ready → bucket-countdown → collective-launch path runs inside the
//    autograd engine thread with no Python re-entry and no GIL.
//
// The reference implements none of this in its own code — it relies on
// torch's C++ reducer; this file is the from-scratch native replacement.

#include <torch/extension.h>

#include <torch/csrc/autograd/function.h>
#include <torch/csrc/autograd/utils/lambda_post_hook.h>
#include <torch/csrc/autograd/variable.h>
#include <torch/csrc/distributed/c10d/ProcessGroup.hpp>
#include <torch/csrc/distributed/c10d/Work.hpp>

#include <algorithm>
#include <cstdint>
#include <map>
#include <memory>
#include <string>
#include <utility>
#include <vector>

namespace ddp_amd {

struct Bucket {
  at::Tensor flat;                 // flat buffer, param dtype/device
  std::vector<int64_t> param_indices;  // global param indices in this bucket
  std::vector<int64_t> offsets;        // element offset of each param in flat
  int64_t pending = 0;             // grads not yet accumulated this iteration
  bool ready = false;              // all grads accumulated, awaiting launch
  c10::intrusive_ptr<c10d::Work> work;  // in-flight all-reduce
};

class Reducer : public std::enable_shared_from_this<Reducer> {
 public:
  Reducer(std::vector<at::Tensor> params,
          c10::intrusive_ptr<c10d::ProcessGroup> pg,
          int64_t first_bucket_bytes,
          int64_t bucket_bytes)
      : params_(std::move(params)), pg_(std::move(pg)) {
    TORCH_CHECK(!params_.empty(), "Reducer needs at least one parameter");
    build_buckets(first_bucket_bytes, bucket_bytes);
    attach_grad_views();
    reset();
  }

  // ---- construction ----

  void build_buckets(int64_t first_cap, int64_t cap) {
    // Reverse registration order approximates gradient-ready order during
    // backward (same heuristic as the torch reducer the reference uses).
    // The first (i.e. last-registered) bucket is small so the first
    // all-reduce launches as early as possible.
    const int64_t n = static_cast<int64_t>(params_.size());
    int64_t cur_bytes = 0;
    int64_t cur_cap = first_cap;
    std::vector<int64_t> cur;
    auto flush = [&]() {
      if (cur.empty()) return;
      buckets_.emplace_back();
      Bucket& b = buckets_.back();
      b.param_indices = cur;
      int64_t total = 0;
      for (int64_t idx : cur) {
        b.offsets.push_back(total);
        total += params_[idx].numel();
      }
      const auto& p0 = params_[cur[0]];
      b.flat = at::empty({total}, p0.options());
      cur.clear();
      cur_bytes = 0;
      cur_cap = cap;
    };
    for (int64_t i = n - 1; i >= 0; --i) {
      const auto& p = params_[i];
      const int64_t bytes = p.numel() * p.element_size();
      // dtype/device boundaries split buckets (one flat buffer each)
      if (!cur.empty()) {
        const auto& prev = params_[cur[0]];
        const bool same = prev.scalar_type() == p.scalar_type() &&
                          prev.device() == p.device();
        if (!same || cur_bytes + bytes > cur_cap) flush();
      }
      cur.push_back(i);
      cur_bytes += bytes;
    }
    flush();
    // record bucket id per param
    bucket_of_.assign(n, -1);
    slot_of_.assign(n, -1);
    for (size_t bi = 0; bi < buckets_.size(); ++bi) {
      for (size_t s = 0; s < buckets_[bi].param_indices.size(); ++s) {
        bucket_of_[buckets_[bi].param_indices[s]] = static_cast<int64_t>(bi);
        slot_of_[buckets_[bi].param_indices[s]] = static_cast<int64_t>(s);
      }
    }
  }

  void attach_grad_views() {
    for (size_t bi = 0; bi < buckets_.size(); ++bi) {
      Bucket& b = buckets_[bi];
      b.flat.zero_();
      for (size_t s = 0; s < b.param_indices.size(); ++s) {
        auto& p = params_[b.param_indices[s]];
        auto view =
            b.flat.narrow(0, b.offsets[s], p.numel()).view(p.sizes());
        p.mutable_grad() = view;
      }
    }
  }

  // ---- per-iteration protocol ----

  void reset() {
    for (auto& b : buckets_) {
      b.pending = static_cast<int64_t>(b.param_indices.size());
      b.ready = false;
      b.work = nullptr;
    }
    next_expected_bucket_ = 0;
  }

  void set_sync(bool on) {
    const bool was = sync_;
    sync_ = on;
    // Re-entering sync mode (last micro-batch of an accumulation cycle):
    // restart the countdowns — hooks were no-ops while sync was off.
    if (on && !was) reset();
  }
  bool sync() const { return sync_; }

  // Register the per-parameter ready hooks directly on the AccumulateGrad
  // autograd nodes — pure C++ path from grad-ready to collective launch
  // (the round-1 version registered ~1 Python closure per parameter:
  // 60–150 interpreter hops per backward).
  void install_hooks() {
    TORCH_CHECK(grad_accumulators_.empty(), "hooks already installed");
    std::weak_ptr<Reducer> self = weak_from_this();
    for (size_t i = 0; i < params_.size(); ++i) {
      auto acc = torch::autograd::impl::grad_accumulator(params_[i]);
      TORCH_CHECK(acc, "parameter ", i, " has no grad accumulator");
      acc->add_post_hook(
          std::make_unique<torch::autograd::utils::LambdaPostHook>(
              [self, i](const torch::autograd::variable_list& outputs,
                        const torch::autograd::variable_list& /*inputs*/)
                  -> torch::autograd::variable_list {
                if (auto r = self.lock())
                  r->mark_ready(static_cast<int64_t>(i));
                return outputs;
              }));
      // keep the nodes alive for the hooks' lifetime (same ownership
      // pattern as torch's own reducer)
      grad_accumulators_.push_back(std::move(acc));
    }
  }

  // Called from the per-parameter post-accumulate-grad hook.
  void mark_ready(int64_t param_index) {
    if (!sync_) return;  // grad-accumulation micro-batch (no_sync)
    TORCH_CHECK(param_index >= 0 &&
                    param_index < static_cast<int64_t>(params_.size()),
                "bad param index");
    const int64_t bi = bucket_of_[param_index];
    Bucket& b = buckets_[bi];
    if (b.pending == 0) {
      // A second backward pass inside one sync window (shared-trunk /
      // multi-loss graph).  If the bucket's all-reduce has NOT launched yet
      // (strict-order launch can hold a ready bucket back), the extra
      // contribution already accumulated into the flat view before any
      // collective reads it — nothing to do.  Once the collective is in
      // flight a late gradient would race it, so that stays a hard error
      // with an actionable message.
      TORCH_CHECK(
          !b.work, "bucket ", bi, " received a gradient (param ", param_index,
          ") after its all-reduce launched. For multiple backward passes per "
          "optimizer step, run all but the last under "
          "DistributedModel.no_sync() so gradients accumulate locally first.");
      return;
    }
    if (--b.pending == 0 && sync_) {
      // RCCL/NCCL requires every rank to issue collectives in the SAME
      // order.  Buckets are built in reverse registration order (the usual
      // ready order), and launches are strictly in bucket-index order: a
      // bucket that becomes ready early waits for its turn, so dynamic
      // graphs or racy hook orderings can never produce a cross-rank
      // collective-order mismatch.
      b.ready = true;
      while (next_expected_bucket_ < static_cast<int64_t>(buckets_.size()) &&
             buckets_[next_expected_bucket_].ready) {
        launch(next_expected_bucket_);
        ++next_expected_bucket_;
      }
    }
  }

  // Launch any buckets whose grads never arrive through hooks (unused
  // params); engine calls this with the indices its graph walk found.
  void mark_unused(const std::vector<int64_t>& param_indices) {
    for (int64_t i : param_indices) mark_ready(i);
  }

  // Wait for every in-flight all-reduce, launch any straggler buckets
  // (all grads were unused), and average.
  void finalize() {
    if (!sync_) return;
    // Launch any bucket whose all-reduce has not gone out yet (possible when
    // some grads never arrived: unused params contribute zeros, which keeps
    // ranks aligned and the math exact), then drain all in-flight work.
    for (size_t bi = 0; bi < buckets_.size(); ++bi) {
      if (!buckets_[bi].work) launch(static_cast<int64_t>(bi));
    }
    next_expected_bucket_ = static_cast<int64_t>(buckets_.size());
    for (auto& b : buckets_) {
      if (b.work) {
        b.work->wait();
        b.work = nullptr;
      }
    }
    const double w = static_cast<double>(pg_->getSize());
    if (w > 1.0) {
      for (auto& b : buckets_) b.flat.div_(w);
    }
    reset();
  }

  // Zero gradient state for the next accumulation cycle.
  void zero_grads() {
    for (auto& b : buckets_) b.flat.zero_();
    reset();
  }

  // Broadcast all parameters (and any extra tensors, e.g. buffers) from
  // rank 0 — DDP-wrap-time semantics (reference ddp.py:194).  Tensors are
  // coalesced into ONE flat buffer per (dtype, device) group and the group
  // broadcasts are issued asynchronously, then unflattened — the round-1
  // version issued one BLOCKING broadcast per tensor (O(#params) latency at
  // wrap time on 8 ranks).
  void broadcast_state(std::vector<at::Tensor> extra) {
    torch::NoGradGuard no_grad;
    std::vector<at::Tensor> all;
    for (auto& t : params_) all.push_back(t.detach());
    for (auto& t : extra) all.push_back(t.detach());
    // group indices by (scalar type, device), preserving order
    std::map<std::pair<int, std::string>, std::vector<size_t>> groups;
    for (size_t i = 0; i < all.size(); ++i) {
      groups[{static_cast<int>(all[i].scalar_type()), all[i].device().str()}]
          .push_back(i);
    }
    struct InFlight {
      at::Tensor flat;
      std::vector<size_t> idx;
      c10::intrusive_ptr<c10d::Work> work;
    };
    std::vector<InFlight> inflight;
    for (auto& [key, idx] : groups) {
      std::vector<at::Tensor> views;
      views.reserve(idx.size());
      for (size_t i : idx) views.push_back(all[i].reshape({-1}));
      InFlight f;
      f.flat = at::cat(views);
      f.idx = idx;
      std::vector<at::Tensor> one{f.flat};
      c10d::BroadcastOptions o;
      o.rootRank = 0;
      f.work = pg_->broadcast(one, o);
      inflight.push_back(std::move(f));
    }
    for (auto& f : inflight) {
      f.work->wait();
      int64_t off = 0;
      for (size_t i : f.idx) {
        auto& dst = all[i];
        dst.copy_(f.flat.narrow(0, off, dst.numel()).view(dst.sizes()));
        off += dst.numel();
      }
    }
  }

  // ---- introspection (tests / tuning) ----

  std::vector<std::vector<int64_t>> bucket_assignment() const {
    std::vector<std::vector<int64_t>> out;
    for (const auto& b : buckets_) out.push_back(b.param_indices);
    return out;
  }

  std::vector<at::Tensor> bucket_flats() const {
    std::vector<at::Tensor> out;
    for (const auto& b : buckets_) out.push_back(b.flat);
    return out;
  }

  int64_t num_buckets() const { return static_cast<int64_t>(buckets_.size()); }

 private:
  void launch(int64_t bi) {
    // collectives fire from inside backward (autograd hooks); keep them out
    // of the autograd graph
    torch::NoGradGuard no_grad;
    Bucket& b = buckets_[bi];
    std::vector<at::Tensor> v{b.flat};
    c10d::AllreduceOptions o;  // SUM; averaged in finalize()
    b.work = pg_->allreduce(v, o);
  }

  std::vector<at::Tensor> params_;
  c10::intrusive_ptr<c10d::ProcessGroup> pg_;
  std::vector<std::shared_ptr<torch::autograd::Node>> grad_accumulators_;
  std::vector<Bucket> buckets_;
  std::vector<int64_t> bucket_of_;
  std::vector<int64_t> slot_of_;
  int64_t next_expected_bucket_ = 0;
  bool sync_ = true;
};

}  // namespace ddp_amd

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  py::class_<ddp_amd::Reducer, std::shared_ptr<ddp_amd::Reducer>>(m, "Reducer")
      .def(py::init<std::vector<at::Tensor>,
                    c10::intrusive_ptr<c10d::ProcessGroup>, int64_t, int64_t>(),
           py::arg("params"), py::arg("process_group"),
           py::arg("first_bucket_bytes") = 1 << 20,
           py::arg("bucket_bytes") = 50 << 20)
      .def("install_hooks", &ddp_amd::Reducer::install_hooks)
      .def("mark_ready", &ddp_amd::Reducer::mark_ready)
      .def("mark_unused", &ddp_amd::Reducer::mark_unused)
      .def("finalize", &ddp_amd::Reducer::finalize,
           py::call_guard<py::gil_scoped_release>())
      .def("zero_grads", &ddp_amd::Reducer::zero_grads)
      .def("reset", &ddp_amd::Reducer::reset)
      .def("set_sync", &ddp_amd::Reducer::set_sync)
      .def("sync", &ddp_amd::Reducer::sync)
      .def("broadcast_state", &ddp_amd::Reducer::broadcast_state,
           py::call_guard<py::gil_scoped_release>())
      .def("bucket_assignment", &ddp_amd::Reducer::bucket_assignment)
      .def("bucket_flats", &ddp_amd::Reducer::bucket_flats)
      .def("num_buckets", &ddp_amd::Reducer::num_buckets);
}
