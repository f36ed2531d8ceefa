// Async tensor <-> file I/O engine (reference: csrc/aio — libaio io_submit
// batching + pthread pool, deepspeed_aio_thread.cpp).
//
// This image has no libaio, so the engine is a std::thread pool issuing
// positional pread/pwrite in block_size chunks — the same role (overlap
// NVMe/page-cache traffic with GPU compute; saturate the device with
// queue_depth concurrent requests) with zero external dependencies. Each
// job is split into chunks that the pool's threads service concurrently,
// which is what achieves queue depth on NVMe.

#include <fcntl.h>
#include <sys/stat.h>
#include <unistd.h>

#include <atomic>
#include <condition_variable>
#include <cstdint>
#include <cstdlib>
#include <cstring>
#include <functional>
#include <mutex>
#include <queue>
#include <stdexcept>
#include <string>
#include <thread>
#include <vector>

namespace dsaio {

struct Job {
  std::function<void()> fn;
};

class ThreadPool {
 public:
  explicit ThreadPool(int n) : stop_(false), pending_(0) {
    for (int i = 0; i < n; ++i)
      workers_.emplace_back([this] { this->run(); });
  }
  ~ThreadPool() {
    {
      std::lock_guard<std::mutex> lk(mu_);
      stop_ = true;
    }
    cv_.notify_all();
    for (auto& t : workers_) t.join();
  }
  void submit(std::function<void()> fn) {
    {
      std::lock_guard<std::mutex> lk(mu_);
      ++pending_;
      q_.push(Job{std::move(fn)});
    }
    cv_.notify_one();
  }
  void wait_all() {
    std::unique_lock<std::mutex> lk(mu_);
    done_cv_.wait(lk, [this] { return pending_ == 0; });
  }

 private:
  void run() {
    for (;;) {
      Job job;
      {
        std::unique_lock<std::mutex> lk(mu_);
        cv_.wait(lk, [this] { return stop_ || !q_.empty(); });
        if (stop_ && q_.empty()) return;
        job = std::move(q_.front());
        q_.pop();
      }
      job.fn();
      {
        std::lock_guard<std::mutex> lk(mu_);
        if (--pending_ == 0) done_cv_.notify_all();
      }
    }
  }
  std::vector<std::thread> workers_;
  std::queue<Job> q_;
  std::mutex mu_;
  std::condition_variable cv_, done_cv_;
  bool stop_;
  int pending_;
};

class AioEngine {
 public:
  AioEngine(int64_t block_size, int n_threads, bool o_direct = false)
      : block_size_(block_size), pool_(n_threads), errors_(0),
        o_direct_(o_direct && block_size % 4096 == 0) {}

  static constexpr int64_t kAlign = 4096;

  // thread-local 4K-aligned bounce buffer for O_DIRECT transfers
  char* bounce(int64_t len) {
    thread_local char* buf = nullptr;
    thread_local int64_t cap = 0;
    if (cap < len) {
      if (buf) ::free(buf);
      cap = (len + kAlign - 1) / kAlign * kAlign;
      buf = static_cast<char*>(::aligned_alloc(kAlign, cap));
    }
    return buf;
  }

  int open_write(const std::string& path) {
    if (o_direct_) {
      int fd = ::open(path.c_str(), O_WRONLY | O_CREAT | O_TRUNC | O_DIRECT,
                      0644);
      if (fd >= 0) return fd;  // else: fs without O_DIRECT -> fall back
    }
    return ::open(path.c_str(), O_WRONLY | O_CREAT | O_TRUNC, 0644);
  }

  int open_read(const std::string& path) {
    if (o_direct_) {
      int fd = ::open(path.c_str(), O_RDONLY | O_DIRECT);
      if (fd >= 0) return fd;
    }
    return ::open(path.c_str(), O_RDONLY);
  }

  void pwrite(const void* data, int64_t nbytes, const std::string& path) {
    int fd = open_write(path);
    if (fd < 0) throw std::runtime_error("aio: cannot open " + path);
    const int fl = ::fcntl(fd, F_GETFL);
    const bool direct = (fl & O_DIRECT) != 0;
    if (!direct && ::ftruncate(fd, nbytes) != 0) {
      ::close(fd);
      throw std::runtime_error("aio: ftruncate failed for " + path);
    }
    const char* p = static_cast<const char*>(data);
    for (int64_t off = 0; off < nbytes; off += block_size_) {
      const int64_t len = std::min(block_size_, nbytes - off);
      pool_.submit([this, fd, p, off, len, direct] {
        if (direct) {
          // O_DIRECT: aligned bounce buffer, length rounded up to 4K
          const int64_t wlen = (len + kAlign - 1) / kAlign * kAlign;
          char* buf = bounce(wlen);
          std::memcpy(buf, p + off, len);
          if (wlen > len) std::memset(buf + len, 0, wlen - len);
          int64_t done = 0;
          while (done < wlen) {
            ssize_t w = ::pwrite(fd, buf + done, wlen - done, off + done);
            if (w <= 0) { ++errors_; return; }
            done += w;
          }
          return;
        }
        int64_t done = 0;
        while (done < len) {
          ssize_t w = ::pwrite(fd, p + off + done, len - done, off + done);
          if (w <= 0) {
            ++errors_;
            return;
          }
          done += w;
        }
      });
    }
    fds_.push_back(fd);
    if (direct) trunc_targets_.emplace_back(fd, nbytes);
  }

  void pread(void* data, int64_t nbytes, const std::string& path) {
    int fd = open_read(path);
    if (fd < 0) throw std::runtime_error("aio: cannot open " + path);
    const int fl = ::fcntl(fd, F_GETFL);
    const bool direct = (fl & O_DIRECT) != 0;
    char* p = static_cast<char*>(data);
    for (int64_t off = 0; off < nbytes; off += block_size_) {
      const int64_t len = std::min(block_size_, nbytes - off);
      pool_.submit([this, fd, p, off, len, direct] {
        if (direct) {
          const int64_t rlen = (len + kAlign - 1) / kAlign * kAlign;
          char* buf = bounce(rlen);
          int64_t done = 0;
          while (done < rlen) {
            ssize_t r = ::pread(fd, buf + done, rlen - done, off + done);
            if (r < 0) { ++errors_; return; }
            if (r == 0) break;  // EOF inside the rounded-up tail
            done += r;
          }
          if (done < len) { ++errors_; return; }
          std::memcpy(p + off, buf, len);
          return;
        }
        int64_t done = 0;
        while (done < len) {
          ssize_t r = ::pread(fd, p + off + done, len - done, off + done);
          if (r <= 0) {
            ++errors_;
            return;
          }
          done += r;
        }
      });
    }
    fds_.push_back(fd);
  }

  int wait() {
    pool_.wait_all();
    // O_DIRECT writes rounded the tail up; restore exact file sizes
    for (auto& t : trunc_targets_) ::ftruncate(t.first, t.second);
    trunc_targets_.clear();
    for (int fd : fds_) ::close(fd);
    fds_.clear();
    int e = errors_.exchange(0);
    return e;
  }

 private:
  int64_t block_size_;
  bool o_direct_;
  std::vector<std::pair<int, int64_t>> trunc_targets_;
  ThreadPool pool_;
  std::vector<int> fds_;
  std::atomic<int> errors_;
};

}  // namespace dsaio

extern "C" {
void* ds_aio_create(long long block_size, int n_threads) {
  // DS_AIO_O_DIRECT=1 bypasses the page cache (real-NVMe measurement
  // mode, reference csrc/aio O_DIRECT path); transparent fallback when
  // the filesystem refuses O_DIRECT or block_size is not 4K-aligned.
  const char* od = ::getenv("DS_AIO_O_DIRECT");
  if (od != nullptr && od[0] == '1')
    return new dsaio::AioEngine(block_size, n_threads, true);
  return new dsaio::AioEngine(block_size, n_threads);
}
void ds_aio_destroy(void* h) { delete static_cast<dsaio::AioEngine*>(h); }
int ds_aio_pwrite(void* h, const void* data, long long nbytes,
                  const char* path) {
  try {
    static_cast<dsaio::AioEngine*>(h)->pwrite(data, nbytes, path);
    return 0;
  } catch (...) {
    return -1;
  }
}
int ds_aio_pread(void* h, void* data, long long nbytes, const char* path) {
  try {
    static_cast<dsaio::AioEngine*>(h)->pread(data, nbytes, path);
    return 0;
  } catch (...) {
    return -1;
  }
}
int ds_aio_wait(void* h) { return static_cast<dsaio::AioEngine*>(h)->wait(); }
}
