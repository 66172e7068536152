// Native inference batching runtime for the MI355X serving path.
//
// MI355X-native equivalent of the reference C++ inference runtime
// (reference torchrec/inference/include/torchrec/inference/BatchingQueue.h:107,
// GPUExecutor.h:38; impls inference_legacy/src/BatchingQueue.cpp:119
// `createBatch`, Batching.cpp:50 `combineFloat`/`combineSparse`,
// ResultSplit.cpp). Design, not a port:
//  * requests (dense features + per-feature sparse id lists) land in an MPMC
//    queue; a batching thread coalesces up to max_batch_size requests or
//    batching_interval_ms, combines them into one dense tensor + KJT-layout
//    (values + feature-major lengths), stages them in PINNED host memory so
//    the executor's hipMemcpyAsync H2D overlaps compute;
//  * executor threads pull combined batches, run the model callback (Python
//    callable under GIL or TorchScript), and split results back per request
//    via promised futures.

#include <torch/extension.h>

#include <atomic>
#include <chrono>
#include <condition_variable>
#include <deque>
#include <functional>
#include <future>
#include <memory>
#include <mutex>
#include <thread>
#include <vector>

namespace trec_amd_infer {

struct PredictionRequest {
  at::Tensor dense;                       // [b, num_dense]
  std::vector<at::Tensor> sparse_values;  // per feature: 1-D int64 ids
  std::vector<at::Tensor> sparse_lengths; // per feature: [b] int64
  std::shared_ptr<std::promise<at::Tensor>> promise;
  size_t batch_items() const { return dense.size(0); }
};

struct CombinedBatch {
  at::Tensor dense;    // [B, num_dense] pinned
  at::Tensor values;   // flat ids, feature-major, pinned
  at::Tensor lengths;  // [F * B] feature-major, pinned
  std::vector<std::shared_ptr<std::promise<at::Tensor>>> promises;
  std::vector<int64_t> sizes;  // items per request (ResultSplit boundaries)
};

class BatchingQueue {
 public:
  BatchingQueue(int num_features, int max_batch_size, int batching_interval_ms)
      : num_features_(num_features),
        max_batch_size_(max_batch_size),
        interval_ms_(batching_interval_ms),
        stop_(false) {
    worker_ = std::thread([this] { this->loop(); });
  }

  ~BatchingQueue() { shutdown(); }

  void shutdown() {
    {
      std::lock_guard<std::mutex> g(mu_);
      if (stop_) return;
      stop_ = true;
    }
    cv_.notify_all();
    out_cv_.notify_all();
    if (worker_.joinable()) worker_.join();
  }

  std::future<at::Tensor> add(at::Tensor dense, std::vector<at::Tensor> values,
                              std::vector<at::Tensor> lengths) {
    TORCH_CHECK((int)values.size() == num_features_, "feature count mismatch");
    PredictionRequest req;
    req.dense = std::move(dense);
    req.sparse_values = std::move(values);
    req.sparse_lengths = std::move(lengths);
    req.promise = std::make_shared<std::promise<at::Tensor>>();
    auto fut = req.promise->get_future();
    {
      std::lock_guard<std::mutex> g(mu_);
      pending_.push_back(std::move(req));
    }
    cv_.notify_one();
    return fut;
  }

  // Executor side: blocking pop of a combined batch. Returns false on stop.
  bool next_batch(CombinedBatch& out) {
    std::unique_lock<std::mutex> g(mu_);
    out_cv_.wait(g, [this] { return stop_ || !ready_.empty(); });
    if (ready_.empty()) return false;
    out = std::move(ready_.front());
    ready_.pop_front();
    return true;
  }

  int ready_size() {
    std::lock_guard<std::mutex> g(mu_);
    return (int)ready_.size();
  }

 private:
  void loop() {
    while (true) {
      std::vector<PredictionRequest> reqs;
      {
        std::unique_lock<std::mutex> g(mu_);
        cv_.wait_for(g, std::chrono::milliseconds(interval_ms_),
                     [this] { return stop_ || !pending_.empty(); });
        if (stop_ && pending_.empty()) return;
        int total = 0;
        while (!pending_.empty() && total < max_batch_size_) {
          total += (int)pending_.front().batch_items();
          reqs.push_back(std::move(pending_.front()));
          pending_.pop_front();
        }
      }
      if (reqs.empty()) continue;
      CombinedBatch batch = combine(reqs);
      {
        std::lock_guard<std::mutex> g(mu_);
        ready_.push_back(std::move(batch));
      }
      out_cv_.notify_one();
    }
  }

  static at::Tensor maybe_pin(at::Tensor t) {
    // pinning needs a GPU runtime; CPU-only test environments skip it
    try {
      return t.pin_memory();
    } catch (const c10::Error&) {
      return t;
    }
  }

  // combineFloat + combineSparse (reference Batching.cpp:50): concat request
  // tensors into one pinned batch; lengths re-laid out feature-major.
  CombinedBatch combine(std::vector<PredictionRequest>& reqs) {
    CombinedBatch out;
    int64_t B = 0;
    for (auto& r : reqs) B += r.batch_items();
    out.dense = at::empty({B, reqs[0].dense.size(1)}, reqs[0].dense.options());
    int64_t row = 0;
    for (auto& r : reqs) {
      out.dense.narrow(0, row, r.batch_items()).copy_(r.dense);
      row += r.batch_items();
    }
    // lengths feature-major [F, B]
    auto lengths = at::empty({num_features_, B}, at::kLong);
    row = 0;
    for (auto& r : reqs) {
      for (int f = 0; f < num_features_; ++f) {
        lengths[f].narrow(0, row, r.batch_items()).copy_(r.sparse_lengths[f]);
      }
      row += r.batch_items();
    }
    out.lengths = maybe_pin(lengths.reshape({-1}));
    // values feature-major: feature f = concat over requests
    std::vector<at::Tensor> vals;
    for (int f = 0; f < num_features_; ++f) {
      for (auto& r : reqs) vals.push_back(r.sparse_values[f]);
    }
    out.values = maybe_pin(at::cat(vals));
    out.dense = maybe_pin(out.dense);
    for (auto& r : reqs) {
      out.promises.push_back(r.promise);
      out.sizes.push_back(r.batch_items());
    }
    return out;
  }

  int num_features_;
  int max_batch_size_;
  int interval_ms_;
  bool stop_;
  std::mutex mu_;
  std::condition_variable cv_, out_cv_;
  std::deque<PredictionRequest> pending_;
  std::deque<CombinedBatch> ready_;
  std::thread worker_;
};

// GPUExecutor (reference GPUExecutor.h:38): N threads pull combined batches,
// invoke the model callback, split results back to request promises
// (ResultSplit.cpp splitResult).
class GPUExecutor {
 public:
  GPUExecutor(std::shared_ptr<BatchingQueue> queue,
              std::function<at::Tensor(at::Tensor, at::Tensor, at::Tensor)> model,
              int num_threads)
      : queue_(std::move(queue)), model_(std::move(model)) {
    for (int i = 0; i < num_threads; ++i) {
      threads_.emplace_back([this] { this->run(); });
    }
  }

  ~GPUExecutor() { join(); }

  void join() {
    queue_->shutdown();
    for (auto& t : threads_)
      if (t.joinable()) t.join();
    threads_.clear();
  }

 private:
  void run() {
    CombinedBatch batch;
    while (queue_->next_batch(batch)) {
      at::Tensor result;
      {
        pybind11::gil_scoped_acquire gil;  // model may be a Python callable
        result = model_(batch.dense, batch.values, batch.lengths);
      }
      int64_t row = 0;
      for (size_t i = 0; i < batch.promises.size(); ++i) {
        batch.promises[i]->set_value(result.narrow(0, row, batch.sizes[i]));
        row += batch.sizes[i];
      }
      batch = CombinedBatch();
    }
  }

  std::shared_ptr<BatchingQueue> queue_;
  std::function<at::Tensor(at::Tensor, at::Tensor, at::Tensor)> model_;
  std::vector<std::thread> threads_;
};

}  // namespace trec_amd_infer

namespace py = pybind11;

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  using namespace trec_amd_infer;
  py::class_<BatchingQueue, std::shared_ptr<BatchingQueue>>(m, "BatchingQueue")
      .def(py::init<int, int, int>(), py::arg("num_features"),
           py::arg("max_batch_size") = 1024, py::arg("batching_interval_ms") = 2)
      .def("add",
           [](BatchingQueue& q, at::Tensor dense, std::vector<at::Tensor> values,
              std::vector<at::Tensor> lengths) {
             auto fut = q.add(std::move(dense), std::move(values), std::move(lengths));
             return std::make_shared<std::shared_future<at::Tensor>>(fut.share());
           })
      .def("ready_size", &BatchingQueue::ready_size)
      .def("shutdown", &BatchingQueue::shutdown);
  py::class_<std::shared_future<at::Tensor>,
             std::shared_ptr<std::shared_future<at::Tensor>>>(m, "TensorFuture")
      .def("get",
           [](std::shared_future<at::Tensor>& f) {
             py::gil_scoped_release rel;
             return f.get();
           });
  py::class_<GPUExecutor, std::shared_ptr<GPUExecutor>>(m, "GPUExecutor")
      .def(py::init([](std::shared_ptr<BatchingQueue> q, py::function model,
                       int num_threads) {
             auto fn = [model](at::Tensor d, at::Tensor v, at::Tensor l) -> at::Tensor {
               // called under GIL (executor acquires before invoking)
               return model(d, v, l).cast<at::Tensor>();
             };
             return std::make_shared<GPUExecutor>(std::move(q), fn, num_threads);
           }),
           py::arg("queue"), py::arg("model"), py::arg("num_threads") = 1)
      .def("join", &GPUExecutor::join, py::call_guard<py::gil_scoped_release>());
}
