// Unit tests: scoped ClientHub + single-flight GtsPluginSelector
// (reference client_hub.rs:155-195, plugins/mod.rs:44).
#include <atomic>
#include <cstdio>
#include <thread>

#include "../src/modkit/modkit.h"

using namespace hs;

static int failures = 0;
#define CHECK(x)                                                   \
  do {                                                             \
    if (!(x)) {                                                    \
      fprintf(stderr, "FAIL %s:%d: %s\n", __FILE__, __LINE__, #x); \
      failures++;                                                  \
    }                                                              \
  } while (0)

struct IFoo {
  virtual ~IFoo() = default;
  virtual int id() const = 0;
};
struct Foo : IFoo {
  explicit Foo(int i) : i_(i) {}
  int id() const override { return i_; }
  int i_;
};

static void test_scoped_hub() {
  ClientHub hub;
  hub.register_client<IFoo>("foo", std::make_shared<Foo>(1));
  hub.register_scoped<IFoo>("foo", "gts.x.plugin.a.v1~", 
                            std::make_shared<Foo>(2));
  CHECK(hub.get<IFoo>("foo")->id() == 1);
  CHECK(hub.get_scoped<IFoo>("foo", "gts.x.plugin.a.v1~")->id() == 2);
  // unknown scope falls back to the unscoped registration
  CHECK(hub.get_scoped<IFoo>("foo", "gts.x.plugin.b.v1~")->id() == 1);
  CHECK(hub.get<IFoo>("bar") == nullptr);
}

static void test_plugin_selector_single_flight() {
  std::atomic<int> calls{0};
  GtsPluginSelector sel([&](const std::string& t) {
    calls++;
    std::this_thread::sleep_for(std::chrono::milliseconds(30));
    return "inst-of-" + t;
  });
  std::vector<std::thread> ts;
  std::atomic<int> ok{0};
  for (int i = 0; i < 8; ++i)
    ts.emplace_back([&] {
      if (sel.select("typeA") == "inst-of-typeA") ok++;
    });
  for (auto& t : ts) t.join();
  CHECK(ok == 8);
  CHECK(calls == 1);             // single flight: resolver ran ONCE
  CHECK(sel.select("typeA") == "inst-of-typeA");
  CHECK(calls == 1);             // cached
  CHECK(sel.select("typeB") == "inst-of-typeB");
  CHECK(calls == 2);
  sel.invalidate("typeA");
  sel.select("typeA");
  CHECK(calls == 3);             // re-resolved after invalidation
}

static void test_selector_failure_not_cached() {
  std::atomic<int> calls{0};
  GtsPluginSelector sel([&](const std::string&) -> std::string {
    if (++calls == 1) throw std::runtime_error("registry down");
    return "ok";
  });
  bool threw = false;
  try { sel.select("t"); } catch (...) { threw = true; }
  CHECK(threw);
  CHECK(sel.select("t") == "ok");   // failure was NOT cached
}

static void test_with_lifecycle() {
  using WL = WithLifecycle;
  std::atomic<int> served{0};
  WL wl([&](std::atomic<bool>& cancel, std::function<void()> ready) {
    ready();
    while (!cancel) {
      served++;
      std::this_thread::sleep_for(std::chrono::milliseconds(2));
    }
  });
  CHECK(wl.state() == WL::State::Stopped);
  CHECK(wl.start());
  CHECK(wl.state() == WL::State::Running);
  CHECK(!wl.start());                      // double-start refused
  std::this_thread::sleep_for(std::chrono::milliseconds(20));
  CHECK(served > 0);
  CHECK(wl.stop());                        // clean unwind
  CHECK(wl.state() == WL::State::Stopped);
  CHECK(wl.start());                       // restartable
  CHECK(wl.stop());

  // a runnable that never signals ready fails start()
  WL late([&](std::atomic<bool>& cancel, std::function<void()>) {
    while (!cancel)
      std::this_thread::sleep_for(std::chrono::milliseconds(2));
  }, /*ready_timeout_ms=*/50);
  CHECK(!late.start());
  CHECK(late.state() == WL::State::Stopped);

  // a runnable that ignores cancel overruns its stop budget
  std::atomic<bool> hang{true};
  {
    WL stuck([&](std::atomic<bool>&, std::function<void()> ready) {
      ready();
      while (hang)
        std::this_thread::sleep_for(std::chrono::milliseconds(2));
    }, 1000, /*stop_timeout_ms=*/50);
    CHECK(stuck.start());
    CHECK(!stuck.stop());                  // reported as unclean
    hang = false;                          // let the detached thread die
    std::this_thread::sleep_for(std::chrono::milliseconds(20));
  }
}

int main() {
  test_with_lifecycle();
  test_scoped_hub();
  test_plugin_selector_single_flight();
  test_selector_failure_not_cached();
  if (failures) { fprintf(stderr, "%d failures\n", failures); return 1; }
  printf("ok: modkit hub/selector\n");
  return 0;
}
