// Native RecordBatcher: RecordYielder -> native processor threadpool ->
// length-bucketed padded batches, fully GIL-free.
//
// MI355X-native equivalent of the reference's C++ batcher
// (lingvo/core/ops/record_batcher.h:89 Options bucket_upper_bound /
// bucket_batch_limit, processor threadpool, flush logic at
// record_batcher.cc:228): records are read+shuffled by native reader
// threads, processed (feature parse / tokenize) by native worker
// threads, bucketed by length, and emitted as padded torch CPU tensors
// ready for pinned-memory H2D copies. The Python-processor path stays in
// lingvo_amd/core/generic_input.py; this file is the fast path that
// keeps 8 GPUs fed without touching the GIL per record.

#include <torch/extension.h>

#include "wpm_encoder.h"

#include <atomic>
#include <condition_variable>
#include <cstdint>
#include <cstring>
#include <deque>
#include <fstream>
#include <functional>
#include <memory>
#include <mutex>
#include <string>
#include <thread>
#include <utility>
#include <vector>

namespace {

// ---------------------------------------------------------------------------
// Generic bucketing core (shared by the concrete batchers below).
//
// Semantics match reference record_batcher.cc: an example with length key
// k lands in the first bucket with bound >= k; k > last bound drops the
// example; a bucket flushes into the ready queue when it holds
// bucket_batch_limit[i] examples, or (partial) when Flush() is called /
// the upstream yielder is exhausted.
// ---------------------------------------------------------------------------
template <typename Example>
class BucketCore {
 public:
  BucketCore(std::vector<int64_t> bounds, std::vector<int64_t> limits,
             int64_t ready_cap)
      : bounds_(std::move(bounds)),
        limits_(std::move(limits)),
        ready_cap_(std::max<int64_t>(1, ready_cap)) {
    TORCH_CHECK(bounds_.size() == limits_.size() && !bounds_.empty(),
                "bucket_upper_bound/bucket_batch_limit size mismatch");
    for (size_t i = 1; i < bounds_.size(); ++i) {
      TORCH_CHECK(bounds_[i] > bounds_[i - 1],
                  "bucket_upper_bound must be increasing");
    }
    buckets_.resize(bounds_.size());
  }

  // Adds the example; if its bucket reached its limit, moves exactly
  // `limit` examples into *full (the caller collates them into tensors
  // OUTSIDE the lock — producer-side collation keeps the consumer
  // thread free). Returns the bucket index in *bi_out.
  // Dropped examples (key<0 or > last bound) return false with *full
  // empty.
  bool Add(Example ex, int64_t key, size_t* bi_out,
           std::vector<Example>* full) {
    full->clear();
    if (key < 0) return false;
    size_t bi = 0;
    while (bi < bounds_.size() && key > bounds_[bi]) ++bi;
    if (bi == bounds_.size()) return false;  // longer than last bound
    std::lock_guard<std::mutex> lk(mu_);
    buckets_[bi].push_back(std::move(ex));
    if ((int64_t)buckets_[bi].size() >= limits_[bi]) {
      auto& bkt = buckets_[bi];
      full->assign(std::make_move_iterator(bkt.begin()),
                   std::make_move_iterator(bkt.begin() + limits_[bi]));
      bkt.erase(bkt.begin(), bkt.begin() + limits_[bi]);
      *bi_out = bi;
    }
    return true;
  }

  // Takes all partial buckets (end of epoch / eval tail) for the caller
  // to collate.
  std::vector<std::pair<size_t, std::vector<Example>>> TakeTails() {
    std::lock_guard<std::mutex> lk(mu_);
    std::vector<std::pair<size_t, std::vector<Example>>> out;
    for (size_t bi = 0; bi < buckets_.size(); ++bi) {
      if (!buckets_[bi].empty()) {
        out.emplace_back(bi, std::move(buckets_[bi]));
        buckets_[bi].clear();
      }
    }
    return out;
  }

  // Queues a collated tensor batch; blocks on ready_cap backpressure.
  void PushReady(std::vector<torch::Tensor> batch) {
    std::unique_lock<std::mutex> lk(mu_);
    cv_space_.wait(lk, [this] {
      return stop_ || (int64_t)ready_.size() < ready_cap_;
    });
    if (stop_) return;
    ready_.push_back(std::move(batch));
    cv_ready_.notify_one();
  }

  // Marks the stream done: TryPop returns false once drained.
  void SetExhausted() {
    std::lock_guard<std::mutex> lk(mu_);
    exhausted_ = true;
    cv_ready_.notify_all();
  }

  // Returns false when the stream is exhausted and drained (caller
  // raises StopIteration with the GIL held).
  bool TryPop(std::vector<torch::Tensor>* out) {
    std::unique_lock<std::mutex> lk(mu_);
    cv_ready_.wait(lk, [this] {
      return stop_ || exhausted_ || !ready_.empty();
    });
    TORCH_CHECK(!stop_, "RecordBatcher stopped");
    if (ready_.empty()) return false;
    *out = std::move(ready_.front());
    ready_.pop_front();
    cv_space_.notify_one();
    return true;
  }

  void Stop() {
    std::lock_guard<std::mutex> lk(mu_);
    stop_ = true;
    cv_ready_.notify_all();
    cv_space_.notify_all();
  }

  int64_t bound(size_t bi) const { return bounds_[bi]; }

 private:
  std::vector<int64_t> bounds_, limits_;
  int64_t ready_cap_;
  std::mutex mu_;
  std::condition_variable cv_ready_, cv_space_;
  std::vector<std::vector<Example>> buckets_;
  std::deque<std::vector<torch::Tensor>> ready_;
  bool stop_ = false;
  bool exhausted_ = false;
};

// Pulls records from a per-thread yielder callable until it signals
// exhaustion; `process` turns one record into (example, key); `collate`
// turns a completed bucket into a tensor batch (runs on the worker
// thread — producer-side collation).
template <typename Example>
class WorkerPool {
 public:
  using Yield = std::function<bool(int, std::string*)>;
  using Process = std::function<int64_t(const std::string&, Example*)>;
  using Collate =
      std::function<void(size_t, std::vector<Example>&&)>;

  WorkerPool(BucketCore<Example>* core, Yield yield, Process process,
             Collate collate, int num_threads)
      : core_(core), yield_(std::move(yield)),
        process_(std::move(process)), collate_(std::move(collate)) {
    active_ = std::max(1, num_threads);
    for (int i = 0; i < std::max(1, num_threads); ++i) {
      threads_.emplace_back([this, i] { Loop(i); });
    }
  }

  void Join() {
    for (auto& t : threads_) {
      if (t.joinable()) t.join();
    }
    threads_.clear();
  }

 private:
  void Loop(int tid) {
    std::string rec;
    std::vector<Example> full;
    size_t bi = 0;
    while (true) {
      if (!yield_(tid, &rec)) break;  // exhausted or stopped
      Example ex;
      int64_t key = process_(rec, &ex);
      core_->Add(std::move(ex), key, &bi, &full);
      if (!full.empty()) collate_(bi, std::move(full));
    }
    if (--active_ == 0) {
      // Last worker out: flush + collate the partial tails.
      for (auto& [tbi, exs] : core_->TakeTails()) {
        collate_(tbi, std::move(exs));
      }
      core_->SetExhausted();
    }
  }

  BucketCore<Example>* core_;
  Yield yield_;
  Process process_;
  Collate collate_;
  std::atomic<int> active_{0};
  std::vector<std::thread> threads_;
};

// Minimal native reader feeding the worker pool (the shuffling
// RecordYielder in input_pipeline.cpp stays the Python-visible one).
// Shards are PARTITIONED across worker threads (thread t owns files
// t, t+N, t+2N, ...) so reads are lock-free per thread — the throughput
// path that keeps 8 GPUs fed. Mixing entropy comes from the partition
// interleaving + downstream bucketing.
class ShardedReader {
 public:
  ShardedReader(std::vector<std::string> files, bool repeat,
                bool binary_framed, int num_threads)
      : files_(std::move(files)), repeat_(repeat), framed_(binary_framed) {
    TORCH_CHECK(!files_.empty(), "no input files");
    states_.resize(std::max(1, num_threads));
  }

  // Lock-free per thread; returns false when this thread's shard set is
  // exhausted (non-repeat) or the reader is stopped.
  bool Next(int tid, std::string* out) {
    State& st = states_[tid];
    const size_t nthreads = states_.size();
    while (!stop_.load(std::memory_order_relaxed)) {
      if (!st.cur.is_open()) {
        size_t idx = tid + st.next * nthreads;
        if (idx >= files_.size()) {
          if (!repeat_ || files_.size() <= (size_t)tid) return false;
          st.next = 0;
          idx = tid;
          if (tid == 0) ++epoch_;
        }
        st.cur.open(files_[idx], std::ios::binary);
        ++st.next;
        if (!st.cur) {
          st.cur.close();
          continue;
        }
      }
      if (framed_) {
        uint32_t len = 0;
        if (st.cur.read(reinterpret_cast<char*>(&len), 4)) {
          out->resize(len);
          if (st.cur.read(out->data(), len)) return true;
        }
        st.cur.close();
        st.cur.clear();
      } else {
        if (std::getline(st.cur, *out)) return true;
        st.cur.close();
        st.cur.clear();
      }
    }
    return false;
  }

  void Stop() { stop_ = true; }

  int64_t epoch() const { return epoch_; }

 private:
  struct State {
    std::ifstream cur;
    size_t next = 0;  // next file index within this thread's partition
  };
  std::vector<std::string> files_;
  bool repeat_, framed_;
  std::vector<State> states_;
  std::atomic<int64_t> epoch_{1};
  std::atomic<bool> stop_{false};
};

// ---------------------------------------------------------------------------
// ASR frame batcher.
//
// Record layout (length-framed binary shards, written by
// tools/make_asr_shards.py): int32 T, int32 D, int32 L,
// float32 frames[T*D], int32 tokens[L]. Output batch matches AsrInput:
// src_frames [B,Tb,D], src_paddings [B,Tb], tgt_ids [B,Lm+1] (SOS-led),
// tgt_labels [B,Lm+1] (EOS-tailed), tgt_paddings [B,Lm+1].
// Bucket key = T (reference AsrInput buckets by frame count,
// lingvo/tasks/asr/input_generator.py).
// ---------------------------------------------------------------------------
struct AsrExample {
  int32_t t = 0, d = 0;
  std::vector<float> frames;
  std::vector<int32_t> tokens;
};

class AsrFrameBatcher {
 public:
  AsrFrameBatcher(std::vector<std::string> files,
                  std::vector<int64_t> bounds, std::vector<int64_t> limits,
                  int64_t sos_id, int64_t eos_id, int num_threads,
                  bool repeat, int64_t ready_cap)
      : core_(std::move(bounds), std::move(limits), ready_cap),
        reader_(std::move(files), repeat, /*binary_framed=*/true,
                num_threads),
        sos_id_(sos_id), eos_id_(eos_id) {
    pool_ = std::make_unique<WorkerPool<AsrExample>>(
        &core_,
        [this](int tid, std::string* r) { return reader_.Next(tid, r); },
        [](const std::string& rec, AsrExample* ex) -> int64_t {
          if (rec.size() < 12) return -1;
          const char* p = rec.data();
          int32_t t, d, l;
          std::memcpy(&t, p, 4);
          std::memcpy(&d, p + 4, 4);
          std::memcpy(&l, p + 8, 4);
          size_t need = 12 + (size_t)t * d * 4 + (size_t)l * 4;
          if (t <= 0 || d <= 0 || l < 0 || rec.size() < need) return -1;
          ex->t = t;
          ex->d = d;
          ex->frames.resize((size_t)t * d);
          std::memcpy(ex->frames.data(), p + 12, (size_t)t * d * 4);
          ex->tokens.resize(l);
          std::memcpy(ex->tokens.data(), p + 12 + (size_t)t * d * 4,
                      (size_t)l * 4);
          return t;
        },
        [this](size_t bi, std::vector<AsrExample>&& exs) {
          core_.PushReady(Collate(bi, exs));
        },
        num_threads);
  }

  ~AsrFrameBatcher() { Stop(); }

  void Stop() {
    reader_.Stop();
    core_.Stop();
    if (pool_) pool_->Join();
  }

  void Flush() {
    for (auto& [bi, exs] : core_.TakeTails()) {
      core_.PushReady(Collate(bi, exs));
    }
  }

  // (src_frames, src_paddings, tgt_ids, tgt_labels, tgt_paddings);
  // empty vector when exhausted.
  std::vector<torch::Tensor> GetBatch() {
    std::vector<torch::Tensor> out;
    if (!core_.TryPop(&out)) return {};
    return out;
  }

  std::vector<torch::Tensor> Collate(size_t bi,
                                     std::vector<AsrExample>& exs) {
    const int64_t b = (int64_t)exs.size();
    const int64_t tb = core_.bound(bi);
    const int64_t d = exs[0].d;
    int64_t lmax = 0;
    for (auto& ex : exs) lmax = std::max<int64_t>(lmax, ex.tokens.size());
    const int64_t lb = lmax + 1;  // room for SOS/EOS
    auto frames = torch::zeros({b, tb, d}, torch::kFloat32);
    auto fpad = torch::ones({b, tb}, torch::kFloat32);
    auto ids = torch::full({b, lb}, eos_id_, torch::kInt64);
    auto labels = torch::full({b, lb}, eos_id_, torch::kInt64);
    auto tpad = torch::ones({b, lb}, torch::kFloat32);
    auto fr = frames.accessor<float, 3>();
    auto fp = fpad.accessor<float, 2>();
    auto id = ids.accessor<int64_t, 2>();
    auto lb_a = labels.accessor<int64_t, 2>();
    auto tp = tpad.accessor<float, 2>();
    for (int64_t i = 0; i < b; ++i) {
      const auto& ex = exs[i];
      std::memcpy(fr[i].data(), ex.frames.data(),
                  (size_t)ex.t * d * sizeof(float));
      for (int64_t j = 0; j < ex.t; ++j) fp[i][j] = 0.f;
      id[i][0] = sos_id_;
      const int64_t l = (int64_t)ex.tokens.size();
      for (int64_t j = 0; j < l; ++j) {
        id[i][j + 1] = ex.tokens[j];
        lb_a[i][j] = ex.tokens[j];
      }
      lb_a[i][l] = eos_id_;
      for (int64_t j = 0; j <= l; ++j) tp[i][j] = 0.f;
    }
    return {frames, fpad, ids, labels, tpad};
  }

  int64_t epoch() const { return reader_.epoch(); }

 private:
  BucketCore<AsrExample> core_;
  ShardedReader reader_;
  int64_t sos_id_, eos_id_;
  std::unique_ptr<WorkerPool<AsrExample>> pool_;
};

// ---------------------------------------------------------------------------
// MT pair batcher: text records "src\ttgt", WPM-tokenized both sides
// (reference NmtInput wordpiece bucketing,
// lingvo/tasks/mt/input_generator.py). Bucket key = max(|src|, |tgt|+1).
// ---------------------------------------------------------------------------
struct MtExample {
  std::vector<int64_t> src, tgt;
};

class MtPairBatcher {
 public:
  MtPairBatcher(std::vector<std::string> files,
                std::vector<std::string> pieces, int64_t unk_id,
                int64_t sos_id, int64_t eos_id,
                std::vector<int64_t> bounds, std::vector<int64_t> limits,
                int num_threads, bool repeat, int64_t ready_cap)
      : core_(std::move(bounds), std::move(limits), ready_cap),
        reader_(std::move(files), repeat, /*binary_framed=*/false,
                num_threads),
        encoder_(std::move(pieces), unk_id),
        sos_id_(sos_id), eos_id_(eos_id) {
    pool_ = std::make_unique<WorkerPool<MtExample>>(
        &core_,
        [this](int tid, std::string* r) { return reader_.Next(tid, r); },
        [this](const std::string& rec, MtExample* ex) -> int64_t {
          auto tab = rec.find('\t');
          if (tab == std::string::npos) return -1;
          ex->src = encoder_.Encode(rec.substr(0, tab));
          ex->tgt = encoder_.Encode(rec.substr(tab + 1));
          if (ex->src.empty() || ex->tgt.empty()) return -1;
          return std::max<int64_t>(ex->src.size(), ex->tgt.size() + 1);
        },
        [this](size_t bi, std::vector<MtExample>&& exs) {
          core_.PushReady(Collate(bi, exs));
        },
        num_threads);
  }

  ~MtPairBatcher() { Stop(); }

  void Stop() {
    reader_.Stop();
    core_.Stop();
    if (pool_) pool_->Join();
  }

  void Flush() {
    for (auto& [bi, exs] : core_.TakeTails()) {
      core_.PushReady(Collate(bi, exs));
    }
  }

  // (src_ids, src_paddings, tgt_ids, tgt_labels, tgt_paddings);
  // empty vector when exhausted.
  std::vector<torch::Tensor> GetBatch() {
    std::vector<torch::Tensor> out;
    if (!core_.TryPop(&out)) return {};
    return out;
  }

  std::vector<torch::Tensor> Collate(size_t bi,
                                     std::vector<MtExample>& exs) {
    (void)bi;
    const int64_t b = (int64_t)exs.size();
    int64_t smax = 1, tmax = 1;
    for (auto& ex : exs) {
      smax = std::max<int64_t>(smax, ex.src.size());
      tmax = std::max<int64_t>(tmax, ex.tgt.size() + 1);
    }
    auto sids = torch::full({b, smax}, eos_id_, torch::kInt64);
    auto spad = torch::ones({b, smax}, torch::kFloat32);
    auto tids = torch::full({b, tmax}, eos_id_, torch::kInt64);
    auto tlab = torch::full({b, tmax}, eos_id_, torch::kInt64);
    auto tpad = torch::ones({b, tmax}, torch::kFloat32);
    auto si = sids.accessor<int64_t, 2>();
    auto sp = spad.accessor<float, 2>();
    auto ti = tids.accessor<int64_t, 2>();
    auto tl = tlab.accessor<int64_t, 2>();
    auto tp = tpad.accessor<float, 2>();
    for (int64_t i = 0; i < b; ++i) {
      const auto& ex = exs[i];
      for (size_t j = 0; j < ex.src.size(); ++j) {
        si[i][j] = ex.src[j];
        sp[i][j] = 0.f;
      }
      ti[i][0] = sos_id_;
      for (size_t j = 0; j < ex.tgt.size(); ++j) {
        ti[i][j + 1] = ex.tgt[j];
        tl[i][j] = ex.tgt[j];
      }
      tl[i][ex.tgt.size()] = eos_id_;
      for (size_t j = 0; j <= ex.tgt.size(); ++j) tp[i][j] = 0.f;
    }
    return {sids, spad, tids, tlab, tpad};
  }

  int64_t epoch() const { return reader_.epoch(); }

 private:
  BucketCore<MtExample> core_;
  ShardedReader reader_;
  lingvo_amd::WpmEncoder encoder_;
  int64_t sos_id_, eos_id_;
  std::unique_ptr<WorkerPool<MtExample>> pool_;
};

}  // namespace

void RegisterRecordBatcher(py::module_& m) {
  py::class_<AsrFrameBatcher>(m, "AsrFrameBatcher")
      .def(py::init<std::vector<std::string>, std::vector<int64_t>,
                    std::vector<int64_t>, int64_t, int64_t, int, bool,
                    int64_t>(),
           py::arg("files"), py::arg("bucket_upper_bound"),
           py::arg("bucket_batch_limit"), py::arg("sos_id") = 1,
           py::arg("eos_id") = 2, py::arg("num_threads") = 4,
           py::arg("repeat") = true, py::arg("ready_cap") = 8)
      .def("get_batch",
           [](AsrFrameBatcher& self) {
             std::vector<torch::Tensor> out;
             {
               py::gil_scoped_release rel;
               out = self.GetBatch();
             }
             if (out.empty()) throw py::stop_iteration();
             return py::make_tuple(out[0], out[1], out[2], out[3], out[4]);
           })
      .def("flush", &AsrFrameBatcher::Flush)
      .def("epoch", &AsrFrameBatcher::epoch)
      .def("stop", &AsrFrameBatcher::Stop,
           py::call_guard<py::gil_scoped_release>());
  py::class_<MtPairBatcher>(m, "MtPairBatcher")
      .def(py::init<std::vector<std::string>, std::vector<std::string>,
                    int64_t, int64_t, int64_t, std::vector<int64_t>,
                    std::vector<int64_t>, int, bool, int64_t>(),
           py::arg("files"), py::arg("pieces"), py::arg("unk_id") = 0,
           py::arg("sos_id") = 1, py::arg("eos_id") = 2,
           py::arg("bucket_upper_bound") = std::vector<int64_t>{64},
           py::arg("bucket_batch_limit") = std::vector<int64_t>{16},
           py::arg("num_threads") = 4, py::arg("repeat") = true,
           py::arg("ready_cap") = 8)
      .def("get_batch",
           [](MtPairBatcher& self) {
             std::vector<torch::Tensor> out;
             {
               py::gil_scoped_release rel;
               out = self.GetBatch();
             }
             if (out.empty()) throw py::stop_iteration();
             return py::make_tuple(out[0], out[1], out[2], out[3], out[4]);
           })
      .def("flush", &MtPairBatcher::Flush)
      .def("epoch", &MtPairBatcher::epoch)
      .def("stop", &MtPairBatcher::Stop,
           py::call_guard<py::gil_scoped_release>());
}
