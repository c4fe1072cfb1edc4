// BatchQueueCore: the native heart of the per-(epoch, rank) batch queue.
//
// Re-implements, in C++ with a mutex + condition variables, the semantics the
// reference implements as a single-threaded asyncio Ray actor
// (reference: ray_shuffling_data_loader/batch_queue.py:383-509 `_QueueActor`):
//
//   * a num_epochs x num_trainers grid of FIFO queues (maxsize 0 = unbounded)
//   * per-(epoch, rank) producer-done flag; producer_done() enqueues a None
//     sentinel through the normal (blocking) put path and sets the flag
//     (reference batch_queue.py:420-422)
//   * join accounting: every put increments an unfinished counter, task_done
//     decrements; join == wait for unfinished == 0 (the sentinel counts, as it
//     does for asyncio.Queue)
//   * new_epoch(epoch) enforces the max_concurrent_epochs window: when the
//     window is full the OLDEST in-flight epoch is evicted only after (a) all
//     its producer-done flags are set and (b) all its per-rank queues joined
//     (reference batch_queue.py:395-418)
//   * get_batch blocks for >= 1 item then drains greedily
//     (reference batch_queue.py:468-475)
//   * put_nowait_batch refuses a batch that would exceed maxsize, with the
//     reference's error message (reference batch_queue.py:480-488)
//
// The reference serializes all state mutation on one asyncio event loop; here
// the equivalent discipline is ONE mutex guarding the whole grid (queue ops
// are control-plane-rate: items are object handles / tensors, not bytes).
//
// GIL discipline: items are held as owned PyObject* raw pointers. Refcount
// changes happen only while holding the GIL and never while holding the queue
// mutex; all blocking waits release the GIL first, so a thread never holds
// the GIL while waiting on the mutex/CV and never takes the GIL while holding
// the mutex. This makes the lock order (GIL outside, mutex inside) acyclic.

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <chrono>
#include <condition_variable>
#include <deque>
#include <mutex>
#include <stdexcept>
#include <string>
#include <vector>

namespace py = pybind11;

namespace {

struct QueueEmpty : std::runtime_error {
  using std::runtime_error::runtime_error;
};
struct QueueFull : std::runtime_error {
  using std::runtime_error::runtime_error;
};
// Raised by every op (including ones blocked in a CV wait when close() fires)
// after the queue is closed — the analog of the reference's killed actor
// surfacing RayActorError to blocked clients
// (reference batch_queue.py:333-355).
struct QueueClosed : std::runtime_error {
  using std::runtime_error::runtime_error;
};

using Clock = std::chrono::steady_clock;

struct SubQueue {
  std::deque<PyObject*> items;  // owned references (incref'd under GIL)
  long long unfinished = 0;     // puts minus task_dones (join counter)
  bool producer_done = false;
};

class BatchQueueCore {
 public:
  BatchQueueCore(int max_concurrent_epochs, int num_epochs, int num_trainers,
                 long long maxsize)
      : max_epochs_(max_concurrent_epochs),
        num_epochs_(num_epochs),
        num_trainers_(num_trainers),
        maxsize_(maxsize),
        grid_(num_epochs) {
    if (num_epochs < 1 || num_trainers < 1 || max_concurrent_epochs < 1)
      throw std::invalid_argument(
          "num_epochs, num_trainers and max_concurrent_epochs must be >= 1");
    for (auto& row : grid_) row.resize(num_trainers_);
  }

  ~BatchQueueCore() {
    // Drop leftover items. Destructor runs with the GIL held (invoked from
    // Python object deallocation).
    for (auto& row : grid_)
      for (auto& q : row) {
        for (PyObject* it : q.items) Py_XDECREF(it);
        q.items.clear();
      }
  }

  BatchQueueCore(const BatchQueueCore&) = delete;
  BatchQueueCore& operator=(const BatchQueueCore&) = delete;

  // ----- epoch window ------------------------------------------------------

  void new_epoch(int epoch) {
    check_epoch(epoch);
    py::gil_scoped_release nogil;
    std::unique_lock<std::mutex> lk(mu_);
    throw_if_closed();
    if ((int)curr_epochs_.size() == max_epochs_) {
      int first = curr_epochs_.front();
      curr_epochs_.pop_front();
      // Producers for every rank of the evicted epoch must be done...
      cv_done_.wait(lk, [&] { return closed_ || all_producers_done(first); });
      // ...and every rank's queue must be fully joined.
      cv_done_.wait(lk, [&] { return closed_ || all_joined(first); });
      throw_if_closed();
    }
    curr_epochs_.push_back(epoch);
  }

  void wait_until_all_epochs_done() {
    py::gil_scoped_release nogil;
    std::unique_lock<std::mutex> lk(mu_);
    throw_if_closed();
    int last = num_epochs_ - 1;
    cv_done_.wait(lk, [&] { return closed_ || all_producers_done(last); });
    cv_done_.wait(lk, [&] { return closed_ || all_joined(last); });
    throw_if_closed();
  }

  // Close the queue: every thread blocked in a CV wait wakes and raises
  // Closed; every subsequent op raises Closed. Leftover items are dropped at
  // destruction (the grid keeps its references so close() itself needs no
  // GIL interaction).
  void close() {
    {
      py::gil_scoped_release nogil;
      std::lock_guard<std::mutex> lk(mu_);
      closed_ = true;
    }
    cv_items_.notify_all();
    cv_space_.notify_all();
    cv_done_.notify_all();
  }

  bool closed() {
    py::gil_scoped_release nogil;
    std::lock_guard<std::mutex> lk(mu_);
    return closed_;
  }

  // ----- producer side -----------------------------------------------------

  void put(int rank, int epoch, py::object item, bool block, double timeout) {
    check(rank, epoch);
    PyObject* raw = item.release().ptr();  // we own one reference now
    int res = put_raw(rank, epoch, raw, block, timeout);
    if (res != 0) {
      // Re-acquire ownership for proper decref (GIL is held again here).
      py::object steal = py::reinterpret_steal<py::object>(py::handle(raw));
      if (res == 2) throw QueueClosed("BatchQueue has been shut down");
      throw QueueFull("Full");
    }
  }

  void put_batch(int rank, int epoch, const std::vector<py::object>& items,
                 bool block, double timeout) {
    check(rank, epoch);
    // Reference put_batch puts item by item, each with its own timeout
    // (reference batch_queue.py:453-459).
    for (const auto& it : items) put(rank, epoch, it, block, timeout);
  }

  void put_nowait_batch(int rank, int epoch,
                        const std::vector<py::object>& items) {
    check(rank, epoch);
    std::vector<PyObject*> raws;
    raws.reserve(items.size());
    for (const auto& it : items) {
      Py_INCREF(it.ptr());
      raws.push_back(it.ptr());
    }
    bool ok = false;
    bool was_closed = false;
    long long cur = 0;
    {
      py::gil_scoped_release nogil;
      std::lock_guard<std::mutex> lk(mu_);
      SubQueue& q = sub(rank, epoch);
      cur = (long long)q.items.size();
      if (closed_) {
        was_closed = true;
      } else if (maxsize_ <= 0 || cur + (long long)raws.size() <= maxsize_) {
        for (PyObject* r : raws) {
          q.items.push_back(r);
          q.unfinished++;
        }
        ok = true;
      }
    }
    if (ok) {
      cv_items_.notify_all();
    } else {
      for (PyObject* r : raws) Py_DECREF(r);
      if (was_closed) throw QueueClosed("BatchQueue has been shut down");
      throw QueueFull("Cannot add " + std::to_string(items.size()) +
                      " items to queue of size " + std::to_string(cur) +
                      " and maxsize " + std::to_string(maxsize_) + ".");
    }
  }

  void producer_done(int rank, int epoch) {
    check(rank, epoch);
    // Sentinel goes through the blocking put path, exactly like the
    // reference's `await queue.put(None)` (batch_queue.py:420-422).
    Py_INCREF(Py_None);
    if (put_raw(rank, epoch, Py_None, /*block=*/true, /*timeout=*/-1.0) != 0) {
      Py_DECREF(Py_None);
      throw QueueClosed("BatchQueue has been shut down");
    }
    {
      py::gil_scoped_release nogil;
      std::lock_guard<std::mutex> lk(mu_);
      sub(rank, epoch).producer_done = true;
    }
    cv_done_.notify_all();
  }

  // ----- consumer side -----------------------------------------------------

  py::object get(int rank, int epoch, bool block, double timeout) {
    check(rank, epoch);
    PyObject* raw = nullptr;
    {
      py::gil_scoped_release nogil;
      std::unique_lock<std::mutex> lk(mu_);
      SubQueue& q = sub(rank, epoch);
      if (!wait_nonempty(lk, q, block, timeout)) {
        lk.unlock();
        // GIL re-acquired by scope exit before throw propagates.
        throw QueueEmpty("Empty");
      }
      raw = q.items.front();
      q.items.pop_front();
    }
    cv_space_.notify_all();
    return py::reinterpret_steal<py::object>(py::handle(raw));
  }

  py::list get_batch(int rank, int epoch) {
    check(rank, epoch);
    std::vector<PyObject*> raws;
    {
      py::gil_scoped_release nogil;
      std::unique_lock<std::mutex> lk(mu_);
      SubQueue& q = sub(rank, epoch);
      cv_items_.wait(lk, [&] { return closed_ || !q.items.empty(); });
      if (q.items.empty()) throw_if_closed();
      while (!q.items.empty()) {
        raws.push_back(q.items.front());
        q.items.pop_front();
      }
    }
    cv_space_.notify_all();
    py::list out;
    for (PyObject* r : raws)
      out.append(py::reinterpret_steal<py::object>(py::handle(r)));
    return out;
  }

  py::object get_nowait(int rank, int epoch) {
    return get(rank, epoch, /*block=*/false, -1.0);
  }

  py::list get_nowait_batch(int rank, int epoch, long long num_items) {
    check(rank, epoch);
    std::vector<PyObject*> raws;
    {
      py::gil_scoped_release nogil;
      std::lock_guard<std::mutex> lk(mu_);
      SubQueue& q = sub(rank, epoch);
      long long n = num_items < 0 ? (long long)q.items.size() : num_items;
      if (n > (long long)q.items.size()) {
        throw QueueEmpty("Cannot get " + std::to_string(n) +
                         " items from queue of size " +
                         std::to_string(q.items.size()) + ".");
      }
      for (long long i = 0; i < n; i++) {
        raws.push_back(q.items.front());
        q.items.pop_front();
      }
    }
    cv_space_.notify_all();
    py::list out;
    for (PyObject* r : raws)
      out.append(py::reinterpret_steal<py::object>(py::handle(r)));
    return out;
  }

  void task_done(int rank, int epoch, long long num_items) {
    check(rank, epoch);
    {
      py::gil_scoped_release nogil;
      std::lock_guard<std::mutex> lk(mu_);
      SubQueue& q = sub(rank, epoch);
      if (q.unfinished < num_items)
        throw std::invalid_argument("task_done() called too many times");
      q.unfinished -= num_items;
    }
    cv_done_.notify_all();
  }

  // ----- introspection -----------------------------------------------------

  long long size() {
    py::gil_scoped_release nogil;
    std::lock_guard<std::mutex> lk(mu_);
    long long total = 0;
    for (auto& row : grid_)
      for (auto& q : row) total += (long long)q.items.size();
    return total;
  }

  long long qsize(int rank, int epoch) {
    check(rank, epoch);
    py::gil_scoped_release nogil;
    std::lock_guard<std::mutex> lk(mu_);
    return (long long)sub(rank, epoch).items.size();
  }

  bool empty(int rank, int epoch) { return qsize(rank, epoch) == 0; }

  bool full(int rank, int epoch) {
    check(rank, epoch);
    if (maxsize_ <= 0) return false;
    py::gil_scoped_release nogil;
    std::lock_guard<std::mutex> lk(mu_);
    return (long long)sub(rank, epoch).items.size() >= maxsize_;
  }

  long long maxsize() const { return maxsize_; }
  int num_epochs() const { return num_epochs_; }
  int num_trainers() const { return num_trainers_; }

 private:
  SubQueue& sub(int rank, int epoch) { return grid_[epoch][rank]; }

  void check_epoch(int epoch) const {
    if (epoch < 0 || epoch >= num_epochs_)
      throw std::out_of_range("epoch " + std::to_string(epoch) +
                              " out of range [0, " +
                              std::to_string(num_epochs_) + ")");
  }
  void check(int rank, int epoch) const {
    check_epoch(epoch);
    if (rank < 0 || rank >= num_trainers_)
      throw std::out_of_range("rank " + std::to_string(rank) +
                              " out of range [0, " +
                              std::to_string(num_trainers_) + ")");
  }

  bool all_producers_done(int epoch) {
    for (auto& q : grid_[epoch])
      if (!q.producer_done) return false;
    return true;
  }
  bool all_joined(int epoch) {
    for (auto& q : grid_[epoch])
      if (q.unfinished != 0) return false;
    return true;
  }

  // Blocking insert of an owned raw reference. Returns 0 on success, 1 on
  // timeout / full (caller must decref), 2 on queue closed (caller must
  // decref and raise Closed). Called with the GIL HELD; releases it around
  // the wait.
  int put_raw(int rank, int epoch, PyObject* raw, bool block,
              double timeout) {
    int res = 1;
    {
      py::gil_scoped_release nogil;
      std::unique_lock<std::mutex> lk(mu_);
      SubQueue& q = sub(rank, epoch);
      auto has_space = [&] {
        return maxsize_ <= 0 || (long long)q.items.size() < maxsize_;
      };
      auto ready = [&] { return closed_ || has_space(); };
      if (!closed_ && !has_space() && block) {
        if (timeout >= 0.0) {
          cv_space_.wait_for(
              lk, std::chrono::duration<double>(timeout), ready);
        } else {
          cv_space_.wait(lk, ready);
        }
      }
      if (closed_) {
        res = 2;
      } else if (has_space()) {
        q.items.push_back(raw);
        q.unfinished++;
        res = 0;
      }
    }
    if (res == 0) cv_items_.notify_all();
    return res;
  }

  // Wait until q non-empty; caller holds lk. Returns false on timeout/empty;
  // throws Closed if the queue closes while waiting (or already was).
  bool wait_nonempty(std::unique_lock<std::mutex>& lk, SubQueue& q, bool block,
                     double timeout) {
    if (!q.items.empty()) return true;
    throw_if_closed();
    if (!block) return false;
    auto ready = [&] { return closed_ || !q.items.empty(); };
    if (timeout >= 0.0) {
      cv_items_.wait_for(lk, std::chrono::duration<double>(timeout), ready);
    } else {
      cv_items_.wait(lk, ready);
    }
    if (!q.items.empty()) return true;
    throw_if_closed();
    return false;
  }

  // Caller must hold mu_.
  void throw_if_closed() const {
    if (closed_) throw QueueClosed("BatchQueue has been shut down");
  }

  const int max_epochs_;
  const int num_epochs_;
  const int num_trainers_;
  const long long maxsize_;
  std::vector<std::vector<SubQueue>> grid_;
  std::deque<int> curr_epochs_;
  bool closed_ = false;
  std::mutex mu_;
  std::condition_variable cv_items_;
  std::condition_variable cv_space_;
  std::condition_variable cv_done_;
};

}  // namespace

PYBIND11_MODULE(_rsdl_cpp, m) {
  m.doc() =
      "MI355X-native batch-queue core (C++ mutex/CV re-implementation of the "
      "reference's asyncio queue actor semantics)";

  py::register_exception<QueueEmpty>(m, "Empty");
  py::register_exception<QueueFull>(m, "Full");
  py::register_exception<QueueClosed>(m, "Closed", PyExc_RuntimeError);

  py::class_<BatchQueueCore>(m, "BatchQueueCore")
      .def(py::init<int, int, int, long long>(),
           py::arg("max_concurrent_epochs"), py::arg("num_epochs"),
           py::arg("num_trainers"), py::arg("maxsize") = 0)
      .def("new_epoch", &BatchQueueCore::new_epoch, py::arg("epoch"))
      .def("wait_until_all_epochs_done",
           &BatchQueueCore::wait_until_all_epochs_done)
      .def("close", &BatchQueueCore::close)
      .def_property_readonly("is_closed", &BatchQueueCore::closed)
      .def("put", &BatchQueueCore::put, py::arg("rank"), py::arg("epoch"),
           py::arg("item"), py::arg("block") = true, py::arg("timeout") = -1.0)
      .def("put_batch", &BatchQueueCore::put_batch, py::arg("rank"),
           py::arg("epoch"), py::arg("items"), py::arg("block") = true,
           py::arg("timeout") = -1.0)
      .def("put_nowait_batch", &BatchQueueCore::put_nowait_batch,
           py::arg("rank"), py::arg("epoch"), py::arg("items"))
      .def("producer_done", &BatchQueueCore::producer_done, py::arg("rank"),
           py::arg("epoch"))
      .def("get", &BatchQueueCore::get, py::arg("rank"), py::arg("epoch"),
           py::arg("block") = true, py::arg("timeout") = -1.0)
      .def("get_batch", &BatchQueueCore::get_batch, py::arg("rank"),
           py::arg("epoch"))
      .def("get_nowait", &BatchQueueCore::get_nowait, py::arg("rank"),
           py::arg("epoch"))
      .def("get_nowait_batch", &BatchQueueCore::get_nowait_batch,
           py::arg("rank"), py::arg("epoch"), py::arg("num_items") = -1)
      .def("task_done", &BatchQueueCore::task_done, py::arg("rank"),
           py::arg("epoch"), py::arg("num_items") = 1)
      .def("size", &BatchQueueCore::size)
      .def("qsize", &BatchQueueCore::qsize, py::arg("rank"), py::arg("epoch"))
      .def("empty", &BatchQueueCore::empty, py::arg("rank"), py::arg("epoch"))
      .def("full", &BatchQueueCore::full, py::arg("rank"), py::arg("epoch"))
      .def_property_readonly("maxsize", &BatchQueueCore::maxsize)
      .def_property_readonly("num_epochs", &BatchQueueCore::num_epochs)
      .def_property_readonly("num_trainers", &BatchQueueCore::num_trainers);
}
