// Async file I/O engine for the NVMe offload tier (ZeRO-Infinity role).
//
// Role parity: reference csrc/aio/ (libaio-based aio_handle,
// py_ds_aio.cpp:23). This image ships no libaio/liburing, so the MI355X
// design uses a C++ thread pool issuing O_DIRECT pread/pwrite at
// block_size granularity — same interface (pread/pwrite/async_*/wait),
// same alignment rules, pinned (page-aligned) host buffers.
#include <torch/extension.h>

#include <fcntl.h>
#include <unistd.h>

#include <atomic>
#include <condition_variable>
#include <functional>
#include <mutex>
#include <queue>
#include <string>
#include <thread>
#include <vector>

namespace {

constexpr size_t kAlign = 4096;

struct IoTask {
  std::function<void()> fn;
};

class ThreadPool {
 public:
  explicit ThreadPool(int n) : stop_(false), inflight_(0) {
    for (int i = 0; i < n; ++i)
      workers_.emplace_back([this] { this->run(); });
  }
  ~ThreadPool() {
    {
      std::unique_lock<std::mutex> lk(mu_);
      stop_ = true;
    }
    cv_.notify_all();
    for (auto& w : workers_) w.join();
  }
  void submit(std::function<void()> fn) {
    {
      std::unique_lock<std::mutex> lk(mu_);
      ++inflight_;
      q_.push({std::move(fn)});
    }
    cv_.notify_one();
  }
  void wait_all() {
    std::unique_lock<std::mutex> lk(mu_);
    done_cv_.wait(lk, [this] { return inflight_ == 0; });
  }

 private:
  void run() {
    for (;;) {
      IoTask t;
      {
        std::unique_lock<std::mutex> lk(mu_);
        cv_.wait(lk, [this] { return stop_ || !q_.empty(); });
        if (stop_ && q_.empty()) return;
        t = std::move(q_.front());
        q_.pop();
      }
      t.fn();
      {
        std::unique_lock<std::mutex> lk(mu_);
        if (--inflight_ == 0) done_cv_.notify_all();
      }
    }
  }
  std::vector<std::thread> workers_;
  std::queue<IoTask> q_;
  std::mutex mu_;
  std::condition_variable cv_, done_cv_;
  bool stop_;
  int inflight_;
};

bool aligned_for_direct(const void* p, size_t bytes) {
  return ((uintptr_t)p % kAlign == 0) && (bytes % kAlign == 0);
}

int open_file(const std::string& fname, bool write, bool try_direct) {
  int flags = write ? (O_WRONLY | O_CREAT) : O_RDONLY;
  if (try_direct) {
    int fd = ::open(fname.c_str(), flags | O_DIRECT, 0644);
    if (fd >= 0) return fd;
  }
  return ::open(fname.c_str(), flags, 0644);
}

}  // namespace

class AioHandle {
 public:
  AioHandle(long block_size, long queue_depth, bool single_submit,
            bool overlap_events, long intra_op_parallelism)
      : block_size_(block_size < (long)kAlign ? (long)kAlign : block_size),
        pool_(std::max<long>(1, intra_op_parallelism)),
        error_(0) {}

  long get_block_size() const { return block_size_; }

  void _rw(at::Tensor buffer, const std::string& fname, bool write,
           long file_offset = 0, bool truncate = true) {
    TORCH_CHECK(buffer.is_contiguous() && buffer.device().is_cpu(),
                "aio buffers must be contiguous host tensors");
    char* data = reinterpret_cast<char*>(buffer.data_ptr());
    size_t nbytes = buffer.numel() * buffer.element_size();
    bool direct = aligned_for_direct(data, nbytes) &&
                  (file_offset % kAlign == 0);
    int fd = open_file(fname, write, direct);
    TORCH_CHECK(fd >= 0, "aio: cannot open ", fname);
    if (write && truncate) {
      int rc = ftruncate(fd, file_offset + nbytes);
      (void)rc;
    }
    size_t nblocks = (nbytes + block_size_ - 1) / block_size_;
    std::atomic<int>* err = &error_;
    for (size_t b = 0; b < nblocks; ++b) {
      size_t off = b * block_size_;
      size_t len = std::min((size_t)block_size_, nbytes - off);
      pool_.submit([fd, data, off, len, write, err, file_offset] {
        ssize_t done = 0;
        while (done < (ssize_t)len) {
          ssize_t r = write
                          ? ::pwrite(fd, data + off + done, len - done,
                                     file_offset + off + done)
                          : ::pread(fd, data + off + done, len - done,
                                    file_offset + off + done);
          if (r <= 0) {
            err->store(errno ? errno : -1);
            return;
          }
          done += r;
        }
      });
    }
    pool_.wait_all();
    ::close(fd);
    TORCH_CHECK(error_.load() == 0, "aio IO error: ", error_.load());
  }

  long sync_pread(at::Tensor buffer, const std::string& fname) {
    _rw(buffer, fname, false);
    return buffer.numel() * buffer.element_size();
  }

  long sync_pwrite(at::Tensor buffer, const std::string& fname) {
    _rw(buffer, fname, true);
    return buffer.numel() * buffer.element_size();
  }

  // async API: one outstanding batch at a time (wait() joins)
  long async_pread(at::Tensor buffer, const std::string& fname) {
    pending_.emplace_back(std::thread(
        [this, buffer, fname] { _rw(buffer, fname, false); }));
    return 0;
  }

  long async_pwrite(at::Tensor buffer, const std::string& fname) {
    pending_.emplace_back(std::thread(
        [this, buffer, fname] { _rw(buffer, fname, true); }));
    return 0;
  }

  long wait() {
    for (auto& t : pending_) t.join();
    long n = pending_.size();
    pending_.clear();
    return n;
  }

  // offset variants (streaming fast-file-writer): write `buffer` at
  // `file_offset` without truncating — the writer truncates at close.
  long sync_pwrite_at(at::Tensor buffer, const std::string& fname,
                      long file_offset) {
    _rw(buffer, fname, true, file_offset, /*truncate=*/false);
    return buffer.numel() * buffer.element_size();
  }

  long sync_pread_at(at::Tensor buffer, const std::string& fname,
                     long file_offset) {
    _rw(buffer, fname, false, file_offset, false);
    return buffer.numel() * buffer.element_size();
  }

  long async_pwrite_at(at::Tensor buffer, const std::string& fname,
                       long file_offset) {
    pending_.emplace_back(std::thread([this, buffer, fname, file_offset] {
      _rw(buffer, fname, true, file_offset, false);
    }));
    return 0;
  }

 private:
  long block_size_;
  ThreadPool pool_;
  std::vector<std::thread> pending_;
  std::atomic<int> error_;
};

void bind_aio(py::module_& m) {
  py::class_<AioHandle>(m, "aio_handle")
      .def(py::init<long, long, bool, bool, long>(),
           py::arg("block_size") = 1 << 20, py::arg("queue_depth") = 8,
           py::arg("single_submit") = false,
           py::arg("overlap_events") = false,
           py::arg("intra_op_parallelism") = 4)
      .def("get_block_size", &AioHandle::get_block_size)
      .def("sync_pread", &AioHandle::sync_pread,
           py::call_guard<py::gil_scoped_release>())
      .def("sync_pwrite", &AioHandle::sync_pwrite,
           py::call_guard<py::gil_scoped_release>())
      .def("async_pread", &AioHandle::async_pread)
      .def("async_pwrite", &AioHandle::async_pwrite)
      .def("sync_pwrite_at", &AioHandle::sync_pwrite_at,
           py::call_guard<py::gil_scoped_release>())
      .def("sync_pread_at", &AioHandle::sync_pread_at,
           py::call_guard<py::gil_scoped_release>())
      .def("async_pwrite_at", &AioHandle::async_pwrite_at)
      .def("wait", &AioHandle::wait,
           py::call_guard<py::gil_scoped_release>());
}
