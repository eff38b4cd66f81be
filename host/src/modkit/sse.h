// Typed SSE broadcaster — the reference's modkit SseBroadcaster
// (libs/modkit/src/http/sse.rs:14-111): modules publish events, any number
// of HTTP subscribers receive them as `data:` frames with keep-alive
// comments; a dead client just drops its subscription.
#pragma once

#include <condition_variable>
#include <deque>
#include <memory>
#include <mutex>

#include "../http/http.h"

namespace hs {

class SseBroadcaster {
 public:
  void publish(const Json& event) {
    std::lock_guard<std::mutex> lk(mu_);
    for (auto& s : subs_) {
      std::lock_guard<std::mutex> sl(s->mu);
      s->q.push_back(event.dump());
      s->cv.notify_one();
    }
  }

  // Blocks serving this subscriber until the client disconnects.
  // Call from a request handler; headers/stream are managed here.
  void serve(ResponseWriter& w, int keepalive_ms = 15000) {
    auto sub = std::make_shared<Sub>();
    {
      std::lock_guard<std::mutex> lk(mu_);
      subs_.push_back(sub);
    }
    w.begin_stream(200, "text/event-stream");
    while (true) {
      std::string ev;
      {
        std::unique_lock<std::mutex> sl(sub->mu);
        sub->cv.wait_for(sl, std::chrono::milliseconds(keepalive_ms),
                         [&] { return !sub->q.empty(); });
        if (!sub->q.empty()) {
          ev = sub->q.front();
          sub->q.pop_front();
        }
      }
      const bool ok = ev.empty()
                          ? w.write_chunk(": keep-alive\n\n")
                          : w.write_chunk("data: " + ev + "\n\n");
      if (!ok) break;
    }
    w.end_stream();
    std::lock_guard<std::mutex> lk(mu_);
    for (size_t i = 0; i < subs_.size(); ++i)
      if (subs_[i] == sub) {
        subs_.erase(subs_.begin() + i);
        break;
      }
  }

  size_t subscribers() {
    std::lock_guard<std::mutex> lk(mu_);
    return subs_.size();
  }

 private:
  struct Sub {
    std::mutex mu;
    std::condition_variable cv;
    std::deque<std::string> q;
  };
  std::mutex mu_;
  std::vector<std::shared_ptr<Sub>> subs_;
};

}  // namespace hs
