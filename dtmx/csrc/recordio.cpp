// Native data pipeline: RecordIO parsing + JPEG decode + augmentation +
// threaded batch prefetch (reference: dmlc-core RecordIO +
// src/io/iter_image_recordio_2.cc:75-525 decode/augment threads +
// src/io/image_aug_default.cc resize/rand-crop/mirror augmenters — OpenCV
// replaced by libjpeg + a small bilinear resampler). Records carry either
// JPEG payloads (FFD8 magic; dims from the decoder) or raw uint8 HWC of
// exactly the target shape (the synthetic-bench fast path).
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
#include <csetjmp>
#include <cstring>
#include <deque>
#include <fstream>
#include <map>
#include <mutex>
#include <thread>
#include <vector>

#include <jpeglib.h>

namespace dtmx {

// ------------------------------------------------------------ image helpers

struct Image {
  std::vector<uint8_t> pix;  // HWC
  int h = 0, w = 0, c = 0;
};

struct JpegErr {
  jpeg_error_mgr pub;
  jmp_buf jb;
};

static void jpeg_err_exit(j_common_ptr cinfo) {
  longjmp(((JpegErr*)cinfo->err)->jb, 1);
}

static bool is_jpeg(const uint8_t* p, size_t len) {
  return len >= 3 && p[0] == 0xFF && p[1] == 0xD8 && p[2] == 0xFF;
}

static bool decode_jpeg(const uint8_t* buf, size_t len, int want_c, Image& im) {
  jpeg_decompress_struct cinfo;
  JpegErr err;
  cinfo.err = jpeg_std_error(&err.pub);
  err.pub.error_exit = jpeg_err_exit;
  if (setjmp(err.jb)) {
    jpeg_destroy_decompress(&cinfo);
    return false;
  }
  jpeg_create_decompress(&cinfo);
  jpeg_mem_src(&cinfo, const_cast<unsigned char*>(buf), len);
  jpeg_read_header(&cinfo, TRUE);
  cinfo.out_color_space = want_c == 1 ? JCS_GRAYSCALE : JCS_RGB;
  jpeg_start_decompress(&cinfo);
  im.h = cinfo.output_height;
  im.w = cinfo.output_width;
  im.c = cinfo.output_components;
  im.pix.resize((size_t)im.h * im.w * im.c);
  while (cinfo.output_scanline < cinfo.output_height) {
    uint8_t* row = im.pix.data() + (size_t)cinfo.output_scanline * im.w * im.c;
    jpeg_read_scanlines(&cinfo, &row, 1);
  }
  jpeg_finish_decompress(&cinfo);
  jpeg_destroy_decompress(&cinfo);
  return true;
}

static std::string encode_jpeg_impl(const uint8_t* pix, int h, int w, int c,
                                    int quality) {
  jpeg_compress_struct cinfo;
  JpegErr err;
  cinfo.err = jpeg_std_error(&err.pub);
  err.pub.error_exit = jpeg_err_exit;
  unsigned char* out = nullptr;
  unsigned long out_len = 0;
  if (setjmp(err.jb)) {
    jpeg_destroy_compress(&cinfo);
    if (out) free(out);
    return {};
  }
  jpeg_create_compress(&cinfo);
  jpeg_mem_dest(&cinfo, &out, &out_len);
  cinfo.image_width = w;
  cinfo.image_height = h;
  cinfo.input_components = c;
  cinfo.in_color_space = c == 1 ? JCS_GRAYSCALE : JCS_RGB;
  jpeg_set_defaults(&cinfo);
  jpeg_set_quality(&cinfo, quality, TRUE);
  jpeg_start_compress(&cinfo, TRUE);
  while (cinfo.next_scanline < cinfo.image_height) {
    JSAMPROW row = const_cast<uint8_t*>(pix + (size_t)cinfo.next_scanline * w * c);
    jpeg_write_scanlines(&cinfo, &row, 1);
  }
  jpeg_finish_compress(&cinfo);
  std::string s((const char*)out, out_len);
  jpeg_destroy_compress(&cinfo);
  free(out);
  return s;
}

// bilinear HWC uint8 resize (reference augmenter resize via cv::resize)
static Image resize_bilinear(const Image& src, int nh, int nw) {
  Image dst;
  dst.h = nh;
  dst.w = nw;
  dst.c = src.c;
  dst.pix.resize((size_t)nh * nw * src.c);
  const float sy = (float)src.h / nh, sx = (float)src.w / nw;
  for (int y = 0; y < nh; ++y) {
    float fy = (y + 0.5f) * sy - 0.5f;
    int y0 = std::max(0, (int)fy);
    int y1 = std::min(src.h - 1, y0 + 1);
    float wy = fy - y0;
    if (fy < 0) { y0 = y1 = 0; wy = 0.f; }
    for (int x = 0; x < nw; ++x) {
      float fx = (x + 0.5f) * sx - 0.5f;
      int x0 = std::max(0, (int)fx);
      int x1 = std::min(src.w - 1, x0 + 1);
      float wx = fx - x0;
      if (fx < 0) { x0 = x1 = 0; wx = 0.f; }
      for (int ch = 0; ch < src.c; ++ch) {
        float v00 = src.pix[((size_t)y0 * src.w + x0) * src.c + ch];
        float v01 = src.pix[((size_t)y0 * src.w + x1) * src.c + ch];
        float v10 = src.pix[((size_t)y1 * src.w + x0) * src.c + ch];
        float v11 = src.pix[((size_t)y1 * src.w + x1) * src.c + ch];
        float v = v00 * (1 - wy) * (1 - wx) + v01 * (1 - wy) * wx +
                  v10 * wy * (1 - wx) + v11 * wy * wx;
        dst.pix[((size_t)y * nw + x) * src.c + ch] = (uint8_t)(v + 0.5f);
      }
    }
  }
  return dst;
}

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
                    int64_t queue_capacity, int64_t seed,
                    bool rand_crop = false, bool rand_mirror = false,
                    int64_t resize = 0)
      : reader_(std::move(reader)),
        batch_(batch_size),
        shape_(std::move(data_shape)),
        shuffle_(shuffle),
        capacity_(std::max<int64_t>(1, queue_capacity)),
        rand_crop_(rand_crop),
        rand_mirror_(rand_mirror),
        resize_(resize) {
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
      const int TH = shape_[0], TW = shape_[1], TC = shape_[2];
      for (int64_t i = 0; i < batch_; ++i) {
        auto [p, len] = reader_->record(order_[b + i]);
        TORCH_CHECK(len > sizeof(IRHeader), "record too short");
        IRHeader h;
        std::memcpy(&h, p, sizeof(h));
        lp[i] = h.label;
        const uint8_t* raw = (const uint8_t*)(p + sizeof(IRHeader));
        size_t raw_len = len - sizeof(IRHeader);
        float* out = dp + i * elem_;
        // per-RECORD deterministic rng: augmentation is reproducible for a
        // given seed regardless of thread assignment (reference per-thread
        // kRandMagic seeds are scheduler-dependent — improved here)
        std::mt19937 rng((uint32_t)(rng_seed_ * 2654435761u +
                                    claim_epoch * 97 + order_[b + i]));
        if (is_jpeg(raw, raw_len) || rand_crop_ || rand_mirror_ || resize_) {
          Image im;
          if (is_jpeg(raw, raw_len)) {
            TORCH_CHECK(decode_jpeg(raw, raw_len, TC, im),
                        "JPEG decode failed for record ", order_[b + i]);
            TORCH_CHECK(im.c == TC, "record has ", im.c, " channels, want ", TC);
          } else {
            TORCH_CHECK((int64_t)raw_len >= elem_, "record too short");
            im.h = TH; im.w = TW; im.c = TC;
            im.pix.assign(raw, raw + elem_);
          }
          // resize shorter side (reference image_aug_default.cc resize aug)
          if (resize_ > 0) {
            int nh, nw;
            if (im.h < im.w) { nh = resize_; nw = (int)((int64_t)im.w * resize_ / im.h); }
            else { nw = resize_; nh = (int)((int64_t)im.h * resize_ / im.w); }
            if (nh != im.h || nw != im.w) im = resize_bilinear(im, nh, nw);
          }
          // crop to target (random or center); upscale if smaller
          if (im.h < TH || im.w < TW)
            im = resize_bilinear(im, std::max(im.h, TH), std::max(im.w, TW));
          int y0, x0;
          if (rand_crop_) {
            y0 = im.h == TH ? 0 : (int)(rng() % (im.h - TH + 1));
            x0 = im.w == TW ? 0 : (int)(rng() % (im.w - TW + 1));
          } else {
            y0 = (im.h - TH) / 2;
            x0 = (im.w - TW) / 2;
          }
          bool mirror = rand_mirror_ && (rng() & 1);
          for (int y = 0; y < TH; ++y)
            for (int x = 0; x < TW; ++x) {
              int sx = mirror ? (x0 + TW - 1 - x) : (x0 + x);
              const uint8_t* s = im.pix.data() +
                  (((size_t)(y0 + y)) * im.w + sx) * TC;
              float* o = out + ((size_t)y * TW + x) * TC;
              for (int ch = 0; ch < TC; ++ch) o[ch] = s[ch] * (1.f / 255.f);
            }
        } else {
          TORCH_CHECK((int64_t)raw_len >= elem_, "record too short");
          for (int64_t e = 0; e < elem_; ++e) out[e] = raw[e] * (1.f / 255.f);
        }
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
  bool rand_crop_ = false;
  bool rand_mirror_ = false;
  int64_t resize_ = 0;
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
                    int64_t, int64_t, bool, bool, int64_t>(),
           py::arg("reader"), py::arg("batch_size"), py::arg("data_shape"),
           py::arg("part_index"), py::arg("num_parts"), py::arg("shuffle"),
           py::arg("num_threads"), py::arg("queue_capacity"), py::arg("seed"),
           py::arg("rand_crop") = false, py::arg("rand_mirror") = false,
           py::arg("resize") = 0)
      .def("reset", &RecordBatchLoader::reset)
      .def("batches_per_epoch", &RecordBatchLoader::batches_per_epoch)
      .def("next", &RecordBatchLoader::next,
           py::call_guard<py::gil_scoped_release>());
  m.def("write_recordio", &write_recordio);
  m.def("encode_jpeg",
        [](py::bytes raw, int h, int w, int c, int quality) {
          std::string s = raw;
          TORCH_CHECK((int)s.size() >= h * w * c, "encode_jpeg: short buffer");
          auto out = encode_jpeg_impl((const uint8_t*)s.data(), h, w, c,
                                      quality);
          TORCH_CHECK(!out.empty(), "encode_jpeg failed");
          return py::bytes(out);
        },
        py::arg("raw"), py::arg("h"), py::arg("w"), py::arg("c"),
        py::arg("quality") = 95);
  m.def("decode_jpeg", [](py::bytes data, int want_c) {
    std::string s = data;
    Image im;
    TORCH_CHECK(decode_jpeg((const uint8_t*)s.data(), s.size(), want_c, im),
                "decode_jpeg failed");
    auto t = at::empty({im.h, im.w, im.c}, at::kByte);
    std::memcpy(t.data_ptr(), im.pix.data(), im.pix.size());
    return t;
  });
}

}  // namespace dtmx
