// Native data pipeline: RecordIO parsing + threaded batch prefetch
// (reference: dmlc-core RecordIO + src/io/iter_image_recordio_2.cc:75-525 —
// the C++ pipeline stage structure re-built without the OpenCV/JPEG stage:
// records carry raw uint8 HWC payloads packed by tools/im2rec.py).
//
// On-disk format (dmlc RecordIO):
//   uint32 kMagic = 0xced7230a
//   uint32 lrec   = (cflag << 29) | length      (cflag 0 = whole record)
//   payload[length], padded to 4-byte alignment
// image records (reference image_recordio.h IRHeader):
//   uint32 flag; float label; uint64 id; uint64 id2;  then raw payload
#include <torch/extension.h>

#include <algorithm>
#include <atomic>
#include <random>
#include <condition_variable>
#include <cstring>
#include <deque>
#include <fstream>
#include <map>
#include <mutex>
#include <thread>
#include <vector>

namespace dtmx {

static constexpr uint32_t kRecMagic = 0xced7230a;

struct IRHeader {
  uint32_t flag;
  float label;
  uint64_t id;
  uint64_t id2;
};

class RecordIOReader {
 public:
  explicit RecordIOReader(const std::string& path) : path_(path) {
    std::ifstream f(path, std::ios::binary | std::ios::ate);
    TORCH_CHECK(f.good(), "cannot open ", path);
    size_t size = f.tellg();
    buf_.resize(size);
    f.seekg(0);
    f.read(buf_.data(), size);
    // index all records
    size_t pos = 0;
    while (pos + 8 <= buf_.size()) {
      uint32_t magic, lrec;
      std::memcpy(&magic, buf_.data() + pos, 4);
      std::memcpy(&lrec, buf_.data() + pos + 4, 4);
      TORCH_CHECK(magic == kRecMagic, "bad recordio magic at ", pos);
      uint32_t len = lrec & ((1u << 29) - 1);
      offsets_.emplace_back(pos + 8, len);
      pos += 8 + ((len + 3u) & ~3u);
    }
  }

  size_t size() const { return offsets_.size(); }

  std::pair<const char*, uint32_t> record(size_t i) const {
    auto [off, len] = offsets_.at(i);
    return {buf_.data() + off, len};
  }

  py::bytes read(size_t i) const {
    auto [p, len] = record(i);
    return py::bytes(p, len);
  }

 private:
  std::string path_;
  std::vector<char> buf_;
  std::vector<std::pair<size_t, uint32_t>> offsets_;
};

// Threaded batch loader: worker threads unpack raw-uint8 image records into
// pinned float batches; a bounded queue feeds the Python iterator
// (reference PrefetcherIter/BatchLoader pipeline, src/io/iter_prefetcher.h).
class RecordBatchLoader {
 public:
  RecordBatchLoader(std::shared_ptr<RecordIOReader> reader, int64_t batch_size,
                    std::vector<int64_t> data_shape, int64_t part_index,
                    int64_t num_parts, bool shuffle, int64_t num_threads,
                    int64_t queue_capacity, int64_t seed)
      : reader_(std::move(reader)),
        batch_(batch_size),
        shape_(std::move(data_shape)),
        shuffle_(shuffle),
        capacity_(std::max<int64_t>(1, queue_capacity)) {
    // shard records (reference part_index/num_parts sharding)
    size_t n = reader_->size();
    size_t per = n / num_parts;
    size_t start = part_index * per;
    size_t end = (part_index == num_parts - 1) ? n : start + per;
    for (size_t i = start; i < end; ++i) order_.push_back(i);
    rng_seed_ = seed;
    elem_ = 1;
    for (auto d : shape_) elem_ *= d;
    reset();
    for (int64_t t = 0; t < std::max<int64_t>(1, num_threads); ++t)
      workers_.emplace_back([this] { worker(); });
  }

  ~RecordBatchLoader() {
    stop_ = true;
    cv_space_.notify_all();
    cv_item_.notify_all();
    for (auto& w : workers_) w.join();
  }

  void reset() {
    std::lock_guard<std::mutex> lk(mu_);
    if (shuffle_) {
      std::mt19937_64 rng(rng_seed_ + epoch_);
      std::shuffle(order_.begin(), order_.end(), rng);
    }
    cursor_ = 0;
    next_emit_ = 0;
    epoch_++;
    ready_.clear();
    cv_space_.notify_all();
  }

  int64_t batches_per_epoch() const { return order_.size() / batch_; }

  // returns (data[B,*shape] float32, label[B] float32) or empty tensors at
  // epoch end
  // Batches are delivered IN ORDER (batch k of the epoch's shuffled order is
  // the k-th item) regardless of worker completion order — matching the
  // reference's ordered threadediter semantics (dmlc threadediter.h).
  std::vector<at::Tensor> next() {
    std::unique_lock<std::mutex> lk(mu_);
    cv_item_.wait(lk, [this] {
      return stop_ || ready_.count(next_emit_) ||
             next_emit_ >= batches_per_epoch();
    });
    auto it = ready_.find(next_emit_);
    if (it == ready_.end()) return {};
    auto out = std::move(it->second);
    ready_.erase(it);
    next_emit_++;
    cv_space_.notify_all();
    return out;
  }

 private:
  void worker() {
    while (!stop_) {
      int64_t b = -1, seq = 0;
      int64_t claim_epoch;
      {
        std::unique_lock<std::mutex> lk(mu_);
        // claim-gating bounds outstanding batches to the queue capacity so
        // the in-order reorder buffer can never deadlock on a slow worker
        cv_space_.wait(lk, [this] {
          return stop_ || cursor_ + batch_ > (int64_t)order_.size() ||
                 cursor_ / batch_ < next_emit_ + capacity_;
        });
        if (stop_) return;
        claim_epoch = epoch_;
        if (cursor_ + batch_ <= (int64_t)order_.size() &&
            cursor_ / batch_ < next_emit_ + capacity_) {
          b = cursor_;
          seq = b / batch_;
          cursor_ += batch_;
        }
      }
      if (b < 0) {
        std::this_thread::sleep_for(std::chrono::milliseconds(2));
        continue;
      }
      auto data = at::empty({batch_, shape_[0], shape_[1], shape_[2]},
                            at::kFloat);
      auto label = at::empty({batch_}, at::kFloat);
      float* dp = data.data_ptr<float>();
      float* lp = label.data_ptr<float>();
      for (int64_t i = 0; i < batch_; ++i) {
        auto [p, len] = reader_->record(order_[b + i]);
        TORCH_CHECK(len >= sizeof(IRHeader) + elem_, "record too short");
        IRHeader h;
        std::memcpy(&h, p, sizeof(h));
        lp[i] = h.label;
        const uint8_t* raw = (const uint8_t*)(p + sizeof(IRHeader));
        float* out = dp + i * elem_;
        for (int64_t e = 0; e < elem_; ++e) out[e] = raw[e] * (1.f / 255.f);
      }
      std::lock_guard<std::mutex> lk(mu_);
      if (epoch_ != claim_epoch) continue;  // reset() raced: drop stale batch
      ready_[seq] = {data, label};
      cv_item_.notify_all();
    }
  }

  std::shared_ptr<RecordIOReader> reader_;
  int64_t batch_;
  std::vector<int64_t> shape_;
  bool shuffle_;
  int64_t capacity_;
  int64_t elem_ = 0;
  std::vector<size_t> order_;
  int64_t cursor_ = 0;
  int64_t next_emit_ = 0;
  int64_t epoch_ = 0;
  uint64_t rng_seed_ = 0;
  std::map<int64_t, std::vector<at::Tensor>> ready_;
  std::mutex mu_;
  std::condition_variable cv_item_, cv_space_;
  std::atomic<bool> stop_{false};
  std::vector<std::thread> workers_;
};

void write_recordio(const std::string& path, const std::vector<py::bytes>& records) {
  std::ofstream f(path, std::ios::binary);
  TORCH_CHECK(f.good(), "cannot open ", path);
  for (const auto& r : records) {
    std::string s = r;
    uint32_t magic = kRecMagic;
    uint32_t lrec = (uint32_t)s.size();
    f.write((const char*)&magic, 4);
    f.write((const char*)&lrec, 4);
    f.write(s.data(), s.size());
    static const char pad[4] = {0, 0, 0, 0};
    size_t p = (4 - (s.size() & 3)) & 3;
    f.write(pad, p);
  }
}

void register_recordio(py::module_& m) {
  py::class_<RecordIOReader, std::shared_ptr<RecordIOReader>>(m, "RecordIOReader")
      .def(py::init<const std::string&>())
      .def("__len__", &RecordIOReader::size)
      .def("read", &RecordIOReader::read);
  py::class_<RecordBatchLoader>(m, "RecordBatchLoader")
      .def(py::init<std::shared_ptr<RecordIOReader>, int64_t,
                    std::vector<int64_t>, int64_t, int64_t, bool, int64_t,
                    int64_t, int64_t>())
      .def("reset", &RecordBatchLoader::reset)
      .def("batches_per_epoch", &RecordBatchLoader::batches_per_epoch)
      .def("next", &RecordBatchLoader::next,
           py::call_guard<py::gil_scoped_release>());
  m.def("write_recordio", &write_recordio);
}

}  // namespace dtmx
