// threads_ctx.hpp — single-process multi-GPU coordination: one host thread
// + one GpuEngine per device, sharing a DistCtx built on std::barrier and
// plain shared memory. Payloads never leave host DRAM (the per-GPU scan
// work itself runs on each thread's own device), so this is strictly
// lower-latency than any collective transport — the natural MI355X-native
// shape for a single node. The multi-process RCCL path
// (sboxgates_amd.parallel) covers torchrun-style deployment.
#pragma once

#include <barrier>
#include <climits>
#include <cstring>
#include <memory>
#include <vector>

#include "sbg/dist.hpp"

namespace sbg {

class ThreadGroup {
 public:
  explicit ThreadGroup(int world)
      : world_(world), barrier_(world), buf_(sizeof(WorkMsg)), vals_(world) {
    for (int r = 0; r < world; r++) {
      ctxs_.emplace_back(new Ctx(this, r));
    }
  }

  DistCtx* ctx(int rank) { return ctxs_[rank].get(); }
  int world() const { return world_; }

 private:
  class Ctx : public DistCtx {
   public:
    Ctx(ThreadGroup* g, int rank) : g_(g), rank_(rank) {}
    int rank() const override { return rank_; }
    int world() const override { return g_->world_; }

    void bcast(void* data, size_t n, int root) override {
      if (g_->world_ == 1) return;
      // buf_ is preallocated to the largest message (WorkMsg); concurrent
      // resizing would race.
      if (n > g_->buf_.size()) std::abort();
      if (rank_ == root) std::memcpy(g_->buf_.data(), data, n);
      g_->barrier_.arrive_and_wait();
      if (rank_ != root) std::memcpy(data, g_->buf_.data(), n);
      g_->barrier_.arrive_and_wait();
    }

    int allreduce_min(int v) override {
      if (g_->world_ == 1) return v;
      g_->vals_[rank_] = v;
      g_->barrier_.arrive_and_wait();
      int m = INT_MAX;
      for (int x : g_->vals_) m = std::min(m, x);
      g_->barrier_.arrive_and_wait();
      return m;
    }

    void barrier() override {
      if (g_->world_ > 1) g_->barrier_.arrive_and_wait();
    }

   private:
    ThreadGroup* g_;
    int rank_;
  };

  int world_;
  std::barrier<> barrier_;
  std::vector<u8> buf_;
  std::vector<int> vals_;
  std::vector<std::unique_ptr<Ctx>> ctxs_;
};

}  // namespace sbg
