// kfio.cpp — multi-threaded flat-tensor file I/O for checkpointing.
//
// The reference's state layer (kube-apiserver/etcd, kubelet) is native code;
// this is the rebuild's native IO path: checkpoint save/load of the flat
// parameter/optimizer buffers (runtime/checkpoint.py). torch.save funnels
// everything through single-threaded pickle; an 8B-model checkpoint moves
// 16 GB of bf16 weights + 96 GB of fp32 optimizer state per node, so
// threaded pwrite/pread against page cache / NVMe is the difference between
// seconds and minutes of stall per save.
//
// Plain C ABI (ctypes from kubeflow_amd/utils/fastio.py). CPU-only: callers
// stage GPU tensors through pinned/pageable host memory first.

#include <cstdint>
#include <cstring>
#include <fcntl.h>
#include <sys/stat.h>
#include <thread>
#include <unistd.h>
#include <vector>

#define KF_EXPORT extern "C" __attribute__((visibility("default")))

namespace {

int run_chunks(int fd, char* buf, int64_t n, int nthreads, bool write_mode) {
  if (nthreads < 1) nthreads = 1;
  if (nthreads > 32) nthreads = 32;
  const int64_t min_chunk = 8 << 20;  // don't spawn threads for small files
  if (n < min_chunk * 2) nthreads = 1;
  std::vector<std::thread> threads;
  std::vector<int> errs(nthreads, 0);
  const int64_t chunk = (n + nthreads - 1) / nthreads;
  for (int t = 0; t < nthreads; ++t) {
    const int64_t lo = t * chunk;
    const int64_t hi = lo + chunk < n ? lo + chunk : n;
    if (lo >= hi) break;
    threads.emplace_back([=, &errs] {
      int64_t off = lo;
      while (off < hi) {
        ssize_t r = write_mode ? pwrite(fd, buf + off, hi - off, off)
                               : pread(fd, buf + off, hi - off, off);
        if (r <= 0) {
          errs[t] = errno ? errno : -1;
          return;
        }
        off += r;
      }
    });
  }
  for (auto& th : threads) th.join();
  for (int e : errs)
    if (e) return e;
  return 0;
}

}  // namespace

// Write n bytes at buf to path (created/truncated). Returns 0 or errno.
KF_EXPORT int kf_write_file(const char* path, const void* buf, int64_t n,
                            int nthreads) {
  int fd = open(path, O_WRONLY | O_CREAT | O_TRUNC, 0644);
  if (fd < 0) return errno;
  if (ftruncate(fd, n) != 0) {
    int e = errno;
    close(fd);
    return e;
  }
  int rc = run_chunks(fd, (char*)const_cast<void*>(buf), n, nthreads, true);
  if (rc == 0 && fsync(fd) != 0) rc = errno;
  close(fd);
  return rc;
}

// Read exactly n bytes from path into buf. Returns 0, errno, or -2 if the
// file is smaller than n.
KF_EXPORT int kf_read_file(const char* path, void* buf, int64_t n,
                           int nthreads) {
  int fd = open(path, O_RDONLY);
  if (fd < 0) return errno;
  struct stat st;
  if (fstat(fd, &st) != 0 || st.st_size < n) {
    close(fd);
    return -2;
  }
  int rc = run_chunks(fd, (char*)buf, n, nthreads, false);
  close(fd);
  return rc;
}

KF_EXPORT int64_t kf_file_size(const char* path) {
  struct stat st;
  if (stat(path, &st) != 0) return -1;
  return (int64_t)st.st_size;
}
