// Intra-process rendezvous: named tensor hand-off between partitions and
// between client and executors (feeds/fetches), mirroring the reference's
// Rendezvous/IntraProcessRendezvous semantics (reference:
// core/framework/rendezvous.h, common_runtime/rendezvous_mgr.cc). Send is
// non-blocking; Recv is callback-async; either side may arrive first.
#pragma once

#include <condition_variable>
#include <deque>
#include <functional>
#include <map>
#include <mutex>

#include "core/base.h"
#include "core/tensor.h"

namespace stf {

class Rendezvous {
 public:
  using DoneCallback =
      std::function<void(const Status&, const Tensor&, bool is_dead)>;

  Status Send(const std::string& key, const Tensor& val, bool is_dead) {
    DoneCallback waiter;
    {
      std::lock_guard<std::mutex> l(mu_);
      if (!status_.ok()) return status_;
      auto& q = table_[key];
      if (!q.waiters.empty()) {
        waiter = std::move(q.waiters.front());
        q.waiters.pop_front();
      } else {
        q.items.push_back({val, is_dead});
      }
    }
    if (waiter) waiter(Status::OK(), val, is_dead);
    return Status::OK();
  }

  void RecvAsync(const std::string& key, DoneCallback done) {
    Item item;
    bool have = false;
    Status bad;
    {
      std::lock_guard<std::mutex> l(mu_);
      if (!status_.ok()) bad = status_;
    }
    if (!bad.ok()) {
      done(bad, Tensor(), false);
      return;
    }
    {
      std::lock_guard<std::mutex> l(mu_);
      auto& q = table_[key];
      if (!q.items.empty()) {
        item = q.items.front();
        q.items.pop_front();
        have = true;
      } else {
        q.waiters.push_back(std::move(done));
      }
    }
    if (have) done(Status::OK(), item.val, item.is_dead);
  }

  // Blocking recv (client-side fetch).
  Status Recv(const std::string& key, Tensor* out, bool* is_dead) {
    std::mutex mu;
    std::condition_variable cv;
    bool ready = false;
    Status status;
    RecvAsync(key, [&](const Status& s, const Tensor& t, bool dead) {
      std::lock_guard<std::mutex> l(mu);
      status = s;
      *out = t;
      if (is_dead) *is_dead = dead;
      ready = true;
      cv.notify_one();
    });
    std::unique_lock<std::mutex> l(mu);
    cv.wait(l, [&]() { return ready; });
    return status;
  }

  // Abort all pending recvs with `s` (used on step failure).
  void StartAbort(const Status& s) {
    std::map<std::string, Queue> table;
    {
      std::lock_guard<std::mutex> l(mu_);
      status_ = s;
      table.swap(table_);
    }
    for (auto& kv : table)
      for (auto& w : kv.second.waiters) w(s, Tensor(), false);
  }

 private:
  struct Item {
    Tensor val;
    bool is_dead = false;
  };
  struct Queue {
    std::deque<Item> items;
    std::deque<DoneCallback> waiters;
  };
  std::mutex mu_;
  std::map<std::string, Queue> table_;
  Status status_;
};

// Rendezvous key for a cross-partition edge or feed/fetch, following the
// shape of the reference's format (rendezvous.cc:51) without job/incarnation
// (single-process).
inline std::string RendezvousKey(const std::string& src_device,
                                 const std::string& dst_device,
                                 const std::string& tensor_name,
                                 const std::string& frame, int64_t iter) {
  return src_device + ";" + dst_device + ";" + tensor_name + ";" + frame +
         ":" + std::to_string(iter);
}

}  // namespace stf
