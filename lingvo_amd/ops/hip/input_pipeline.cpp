// C++ input pipeline: multi-threaded shuffling record yielder.
//
// MI355X-native equivalent of the reference's RecordYielder family
// (lingvo/core/ops/record_yielder.{h,cc}: BasicRecordYielder h:170,
// epoch-scoped shuffle buffer, sharded file reads) as a pybind11
// extension: N reader threads fill a bounded shuffle buffer; Yield()
// pops a uniformly-sampled record, releasing the GIL while waiting.
// Formats: 'text' (newline records), 'tfrecord' (TF's length-prefixed
// framing, parsed natively — no TensorFlow), 'bytes' (whole file).

#include <torch/extension.h>

#include "wpm_encoder.h"

#include <atomic>
#include <condition_variable>
#include <cstdint>
#include <cstring>
#include <fstream>
#include <mutex>
#include <random>
#include <string>
#include <deque>
#include <thread>
#include <vector>

namespace {

struct Record {
  std::string value;
  int32_t source_id;
};

class RecordYielder {
 public:
  RecordYielder(std::vector<std::string> files, std::string format,
                int64_t seed, int64_t buffer_size, int num_threads,
                bool repeat)
      : files_(std::move(files)),
        format_(std::move(format)),
        rng_(seed ? seed : 301),
        buffer_cap_(std::max<int64_t>(1, buffer_size)),
        repeat_(repeat),
        num_threads_(std::max(1, num_threads)) {
    TORCH_CHECK(!files_.empty(), "RecordYielder: no input files");
    Start();
  }

  ~RecordYielder() { Stop(); }

  void Start() {
    stop_ = false;
    epoch_ = 1;
    next_file_ = 0;
    exhausted_ = false;
    order_.resize(files_.size());
    for (size_t i = 0; i < order_.size(); ++i) order_[i] = i;
    std::shuffle(order_.begin(), order_.end(), rng_);
    for (int i = 0; i < num_threads_; ++i) {
      threads_.emplace_back([this] { ReaderLoop(); });
    }
  }

  void Stop() {
    {
      std::lock_guard<std::mutex> lk(mu_);
      stop_ = true;
    }
    cv_pop_.notify_all();
    cv_push_.notify_all();
    for (auto& t : threads_) {
      if (t.joinable()) t.join();
    }
    threads_.clear();
  }

  int64_t current_epoch() {
    std::lock_guard<std::mutex> lk(mu_);
    return epoch_;
  }

  // Pops one uniformly-sampled record (GIL released by the binding).
  std::pair<std::string, int32_t> Yield() {
    std::unique_lock<std::mutex> lk(mu_);
    cv_pop_.wait(lk, [this] {
      return stop_ || !buffer_.empty() || (exhausted_ && active_ == 0);
    });
    TORCH_CHECK(!stop_, "RecordYielder stopped");
    if (buffer_.empty()) {
      throw py::stop_iteration();
    }
    std::uniform_int_distribution<size_t> dist(0, buffer_.size() - 1);
    size_t idx = dist(rng_);
    std::swap(buffer_[idx], buffer_.back());
    Record rec = std::move(buffer_.back());
    buffer_.pop_back();
    cv_push_.notify_one();
    return {std::move(rec.value), rec.source_id};
  }

 private:
  bool NextFile(size_t* out) {
    std::unique_lock<std::mutex> lk(mu_);
    if (next_file_ >= order_.size()) {
      if (!repeat_) {
        exhausted_ = true;
        cv_pop_.notify_all();
        return false;
      }
      // New epoch: reshuffle file order.
      std::shuffle(order_.begin(), order_.end(), rng_);
      next_file_ = 0;
      ++epoch_;
    }
    *out = order_[next_file_++];
    return true;
  }

  void Push(Record rec) {
    std::unique_lock<std::mutex> lk(mu_);
    cv_push_.wait(lk, [this] {
      return stop_ || (int64_t)buffer_.size() < buffer_cap_;
    });
    if (stop_) return;
    buffer_.push_back(std::move(rec));
    cv_pop_.notify_one();
  }

  void ReadFile(const std::string& path, int32_t source_id) {
    std::ifstream f(path, std::ios::binary);
    if (!f) return;
    if (format_ == "text") {
      std::string line;
      while (!stop_ && std::getline(f, line)) {
        Push({line, source_id});
      }
    } else if (format_ == "tfrecord") {
      // TFRecord framing: u64 length, u32 crc(length), data, u32 crc.
      while (!stop_) {
        uint64_t len = 0;
        if (!f.read(reinterpret_cast<char*>(&len), 8)) break;
        f.seekg(4, std::ios::cur);  // length crc (not verified)
        std::string data(len, '\0');
        if (!f.read(data.data(), len)) break;
        f.seekg(4, std::ios::cur);  // data crc
        Push({std::move(data), source_id});
      }
    } else {  // bytes: whole file as one record
      std::string data((std::istreambuf_iterator<char>(f)),
                       std::istreambuf_iterator<char>());
      Push({std::move(data), source_id});
    }
  }

  void ReaderLoop() {
    ++active_;
    size_t fi;
    while (!stop_ && NextFile(&fi)) {
      ReadFile(files_[fi], (int32_t)fi);
    }
    --active_;
    cv_pop_.notify_all();
  }

  std::vector<std::string> files_;
  std::string format_;
  std::mt19937_64 rng_;
  int64_t buffer_cap_;
  bool repeat_;
  int num_threads_;

  std::mutex mu_;
  std::condition_variable cv_pop_, cv_push_;
  std::vector<Record> buffer_;
  std::vector<size_t> order_;
  size_t next_file_ = 0;
  int64_t epoch_ = 1;
  std::atomic<bool> stop_{false};
  std::atomic<int> active_{0};
  bool exhausted_ = false;
  std::vector<std::thread> threads_;
};


// Native text->LM-batch pipeline: reader threads feed worker threads
// that tokenize (WPM), bucket by token length, and emit padded int64
// batches — the C++ composition of RecordYielder + tokenizer +
// RecordBatcher (reference record_batcher.h:89), fully GIL-free.
class TextLmBatcher {
 public:
  TextLmBatcher(std::vector<std::string> files,
                std::vector<std::string> pieces, int64_t unk_id,
                int64_t sos_id, int64_t eos_id,
                std::vector<int64_t> bucket_bounds,
                std::vector<int64_t> bucket_limits, int64_t seed,
                int num_threads, int64_t buffer_size, bool repeat)
      : yielder_(std::move(files), "text", seed, buffer_size, 2, repeat),
        encoder_(std::move(pieces), unk_id),
        sos_id_(sos_id),
        eos_id_(eos_id),
        bounds_(std::move(bucket_bounds)),
        limits_(std::move(bucket_limits)) {
    TORCH_CHECK(bounds_.size() == limits_.size(),
                "bucket bounds/limits mismatch");
    buckets_.resize(bounds_.size());
    for (int i = 0; i < std::max(1, num_threads); ++i) {
      workers_.emplace_back([this] { WorkerLoop(); });
    }
  }

  ~TextLmBatcher() { Stop(); }

  void Stop() {
    {
      std::lock_guard<std::mutex> lk(mu_);
      stop_ = true;
    }
    cv_batch_.notify_all();
    yielder_.Stop();
    for (auto& t : workers_) {
      if (t.joinable()) t.join();
    }
    workers_.clear();
  }

  // Returns (ids [B, L+1] with SOS, labels [B, L+1] with EOS,
  // paddings [B, L+1]) for one ready bucket.
  std::vector<torch::Tensor> GetBatch() {
    std::unique_lock<std::mutex> lk(mu_);
    cv_batch_.wait(lk, [this] { return stop_ || !ready_.empty(); });
    TORCH_CHECK(!stop_, "TextLmBatcher stopped");
    auto batch = std::move(ready_.front());
    ready_.pop_front();
    lk.unlock();
    const int64_t bound = bounds_[batch.bucket] + 1;  // +1 for SOS/EOS
    const int64_t b = (int64_t)batch.examples.size();
    auto ids = torch::full({b, bound}, eos_id_, torch::kInt64);
    auto labels = torch::full({b, bound}, eos_id_, torch::kInt64);
    auto pad = torch::ones({b, bound}, torch::kFloat32);
    auto ids_a = ids.accessor<int64_t, 2>();
    auto lab_a = labels.accessor<int64_t, 2>();
    auto pad_a = pad.accessor<float, 2>();
    for (int64_t i = 0; i < b; ++i) {
      const auto& toks = batch.examples[i];
      int64_t n = (int64_t)toks.size();
      ids_a[i][0] = sos_id_;
      for (int64_t j = 0; j < n; ++j) {
        ids_a[i][j + 1] = toks[j];
        lab_a[i][j] = toks[j];
      }
      lab_a[i][n] = eos_id_;
      for (int64_t j = 0; j <= n; ++j) pad_a[i][j] = 0.0f;
    }
    return {ids, labels, pad};
  }

 private:
  struct PendingBatch {
    size_t bucket;
    std::vector<std::vector<int64_t>> examples;
  };

  void WorkerLoop() {
    while (!stop_) {
      std::pair<std::string, int32_t> rec;
      try {
        rec = yielder_.Yield();
      } catch (...) {
        break;  // stopped or exhausted
      }
      auto toks = encoder_.Encode(rec.first);
      int64_t len = (int64_t)toks.size();
      std::lock_guard<std::mutex> lk(mu_);
      for (size_t bi = 0; bi < bounds_.size(); ++bi) {
        if (len <= bounds_[bi]) {
          buckets_[bi].push_back(std::move(toks));
          if ((int64_t)buckets_[bi].size() >= limits_[bi]) {
            ready_.push_back({bi, std::move(buckets_[bi])});
            buckets_[bi].clear();
            cv_batch_.notify_one();
          }
          break;  // too-long records (> last bound) are dropped
        }
      }
    }
  }

  RecordYielder yielder_;
  lingvo_amd::WpmEncoder encoder_;
  int64_t sos_id_, eos_id_;
  std::vector<int64_t> bounds_, limits_;

  std::mutex mu_;
  std::condition_variable cv_batch_;
  std::vector<std::vector<std::vector<int64_t>>> buckets_;
  std::deque<PendingBatch> ready_;
  std::atomic<bool> stop_{false};
  std::vector<std::thread> workers_;
};

}  // namespace

void RegisterInputPipeline(py::module_& m) {
  py::class_<RecordYielder>(m, "RecordYielder")
      .def(py::init<std::vector<std::string>, std::string, int64_t,
                    int64_t, int, bool>(),
           py::arg("files"), py::arg("format") = "text",
           py::arg("seed") = 0, py::arg("buffer_size") = 10000,
           py::arg("num_threads") = 4, py::arg("repeat") = true)
      .def("yield_record",
           [](RecordYielder& self) {
             std::pair<std::string, int32_t> rec;
             {
               py::gil_scoped_release release;
               rec = self.Yield();
             }
             return py::make_tuple(py::bytes(rec.first), rec.second);
           })
      .def("current_epoch", &RecordYielder::current_epoch)
      .def("stop", &RecordYielder::Stop);
  py::class_<TextLmBatcher>(m, "TextLmBatcher")
      .def(py::init<std::vector<std::string>, std::vector<std::string>,
                    int64_t, int64_t, int64_t, std::vector<int64_t>,
                    std::vector<int64_t>, int64_t, int, int64_t, bool>(),
           py::arg("files"), py::arg("pieces"), py::arg("unk_id") = 0,
           py::arg("sos_id") = 1, py::arg("eos_id") = 2,
           py::arg("bucket_bounds") = std::vector<int64_t>{64},
           py::arg("bucket_limits") = std::vector<int64_t>{16},
           py::arg("seed") = 0, py::arg("num_threads") = 2,
           py::arg("buffer_size") = 10000, py::arg("repeat") = true)
      .def("get_batch",
           [](TextLmBatcher& self) {
             std::vector<torch::Tensor> out;
             {
               py::gil_scoped_release release;
               out = self.GetBatch();
             }
             return py::make_tuple(out[0], out[1], out[2]);
           })
      .def("stop", &TextLmBatcher::Stop);
}
