// Native prefetching token dataloader.
//
// Reference parity: hetu/graph/data/dataloader.h:18-160 — a C++ batch
// provider that keeps a prefetch ring ahead of the consumer.  MI355X-native
// shape: the corpus is a flat binary file of token ids (uint16 or int32,
// memory-mapped, zero-copy), a background thread assembles [B, S+1] batches
// into PINNED host buffers (so the H2D copy of input_ids/labels overlaps
// compute on a side stream), and epoch shuffling is a seeded permutation of
// sample windows.
#include <torch/extension.h>

#include <fcntl.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <unistd.h>

#include <atomic>
#include <condition_variable>
#include <cstring>
#include <mutex>
#include <numeric>
#include <queue>
#include <random>
#include <string>
#include <thread>
#include <vector>

namespace {

struct TokenBinLoader {
  TokenBinLoader(const std::string& path, int64_t batch, int64_t seq_len,
                 int64_t dtype_bytes, int64_t prefetch, int64_t seed,
                 bool pin, bool drop_last)
      : batch_(batch), seq_(seq_len), dtb_(dtype_bytes),
        depth_(std::max<int64_t>(prefetch, 1)), seed_(seed), pin_(pin),
        drop_last_(drop_last) {
    TORCH_CHECK(dtype_bytes == 2 || dtype_bytes == 4,
                "token dtype must be uint16 or int32");
    fd_ = open(path.c_str(), O_RDONLY);
    TORCH_CHECK(fd_ >= 0, "cannot open ", path);
    struct stat st;
    fstat(fd_, &st);
    bytes_ = st.st_size;
    n_tokens_ = bytes_ / dtb_;
    TORCH_CHECK(n_tokens_ > seq_ + 1, "file too small for seq_len");
    map_ = mmap(nullptr, bytes_, PROT_READ, MAP_PRIVATE, fd_, 0);
    TORCH_CHECK(map_ != MAP_FAILED, "mmap failed");
    madvise(map_, bytes_, MADV_SEQUENTIAL);
    // non-overlapping sample windows of seq_len+1 tokens
    n_samples_ = (n_tokens_ - 1) / seq_;
    n_batches_ = drop_last_ ? n_samples_ / batch_
                            : (n_samples_ + batch_ - 1) / batch_;
    start_epoch(0);
  }

  ~TokenBinLoader() { stop(); munmap(map_, bytes_); close(fd_); }

  void stop() {
    {
      std::lock_guard<std::mutex> lk(mu_);
      stopping_ = true;
    }
    cv_prod_.notify_all();
    cv_cons_.notify_all();
    if (worker_.joinable()) worker_.join();
  }

  void start_epoch(int64_t epoch) {
    stop();
    std::lock_guard<std::mutex> lk(mu_);
    stopping_ = false;
    epoch_ = epoch;
    perm_.resize(n_samples_);
    std::iota(perm_.begin(), perm_.end(), (int64_t)0);
    std::mt19937_64 rng(seed_ + epoch);
    std::shuffle(perm_.begin(), perm_.end(), rng);
    next_batch_ = 0;
    while (!ready_.empty()) ready_.pop();
    worker_ = std::thread([this] { run(); });
  }

  int64_t token_at(int64_t i) const {
    if (dtb_ == 2)
      return ((const uint16_t*)map_)[i];
    return ((const int32_t*)map_)[i];
  }

  void fill(int64_t b, torch::Tensor& out) {
    int64_t* p = out.data_ptr<int64_t>();
    int64_t rows = std::min(batch_, n_samples_ - b * batch_);
    for (int64_t r = 0; r < rows; ++r) {
      int64_t s = perm_[b * batch_ + r];
      int64_t base = s * seq_;
      for (int64_t j = 0; j <= seq_; ++j)
        p[r * (seq_ + 1) + j] = token_at(base + j);
    }
    for (int64_t r = rows; r < batch_; ++r)  // pad tail batch by wrap
      std::memcpy(p + r * (seq_ + 1), p, (seq_ + 1) * sizeof(int64_t));
  }

  void run() {
    for (;;) {
      int64_t b;
      {
        std::unique_lock<std::mutex> lk(mu_);
        cv_prod_.wait(lk, [this] {
          return stopping_ || (int64_t)ready_.size() < depth_;
        });
        if (stopping_) return;
        b = next_batch_;
        if (b >= n_batches_) return;  // epoch done
        ++next_batch_;
        ++in_flight_;
      }
      auto opts = torch::TensorOptions().dtype(torch::kInt64);
      auto t = pin_ && torch::cuda::is_available()
                   ? torch::empty({batch_, seq_ + 1},
                                  opts.pinned_memory(true))
                   : torch::empty({batch_, seq_ + 1}, opts);
      fill(b, t);
      {
        std::lock_guard<std::mutex> lk(mu_);
        ready_.push(std::move(t));
        --in_flight_;
      }
      cv_cons_.notify_one();
    }
  }

  // next() -> [B, S+1] int64 batch; throws StopIteration semantics via
  // an undefined tensor when the epoch is exhausted.
  torch::Tensor next() {
    std::unique_lock<std::mutex> lk(mu_);
    cv_cons_.wait(lk, [this] {
      return stopping_ || !ready_.empty() ||
             (next_batch_ >= n_batches_ && in_flight_ == 0);
    });
    if (ready_.empty())
      return torch::empty({0}, torch::TensorOptions().dtype(torch::kInt64));
    auto t = std::move(ready_.front());
    ready_.pop();
    cv_prod_.notify_one();
    return t;
  }

  int64_t num_batches() const { return n_batches_; }
  int64_t num_samples() const { return n_samples_; }

  int fd_;
  void* map_;
  int64_t bytes_, n_tokens_, n_samples_, n_batches_;
  int64_t batch_, seq_, dtb_, depth_, seed_;
  bool pin_, drop_last_;
  int64_t epoch_ = 0, next_batch_ = 0, in_flight_ = 0;
  std::vector<int64_t> perm_;
  std::queue<torch::Tensor> ready_;
  std::mutex mu_;
  std::condition_variable cv_prod_, cv_cons_;
  std::thread worker_;
  bool stopping_ = false;
};

}  // namespace

void register_dataloader(pybind11::module& m) {
  pybind11::class_<TokenBinLoader>(m, "TokenBinLoader")
      .def(pybind11::init<const std::string&, int64_t, int64_t, int64_t,
                          int64_t, int64_t, bool, bool>(),
           pybind11::arg("path"), pybind11::arg("batch"),
           pybind11::arg("seq_len"), pybind11::arg("dtype_bytes") = 2,
           pybind11::arg("prefetch") = 4, pybind11::arg("seed") = 0,
           pybind11::arg("pin") = true, pybind11::arg("drop_last") = true)
      .def("next", &TokenBinLoader::next,
           pybind11::call_guard<pybind11::gil_scoped_release>())
      .def("start_epoch", &TokenBinLoader::start_epoch)
      .def("num_batches", &TokenBinLoader::num_batches)
      .def("num_samples", &TokenBinLoader::num_samples);
}
