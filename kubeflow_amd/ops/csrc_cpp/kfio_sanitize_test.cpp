// kfio_sanitize_test.cpp — TSAN/ASAN harness for the threaded IO engine.
//
// SURVEY.md §5 maps the reference's implicit concurrency discipline (Go
// race detector + controller-runtime's single-reconciler-per-key) to
// sanitizer builds of the native layer. This harness compiles TOGETHER
// with kfio.cpp under -fsanitize=thread or -fsanitize=address (driven by
// tests/test_sanitizers.py) and exercises the concurrent chunked
// pwrite/pread path plus concurrent independent files — any data race /
// heap error makes the sanitizer abort with a nonzero exit.
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <string>
#include <thread>
#include <vector>

extern "C" {
int kf_write_file(const char* path, const void* buf, int64_t n,
                  int nthreads);
int kf_read_file(const char* path, void* buf, int64_t n, int nthreads);
int64_t kf_file_size(const char* path);
}

int main(int argc, char** argv) {
  const std::string dir = argc > 1 ? argv[1] : "/tmp";
  const int64_t n = 8 << 20;  // 8 MiB -> multiple chunks per thread
  std::vector<char> src(n), dst(n);
  for (int64_t i = 0; i < n; ++i) src[i] = (char)(i * 1315423911u >> 17);

  // threaded write + read of one file
  const std::string p = dir + "/kfio_san.bin";
  if (kf_write_file(p.c_str(), src.data(), n, 8) != 0) return 2;
  if (kf_file_size(p.c_str()) != n) return 3;
  if (kf_read_file(p.c_str(), dst.data(), n, 8) != 0) return 4;
  if (std::memcmp(src.data(), dst.data(), n) != 0) return 5;

  // concurrent independent files (checkpoint shards pattern: every rank
  // writes its own shard at once)
  std::vector<std::thread> ts;
  std::vector<int> rcs(4, 0);
  for (int t = 0; t < 4; ++t) {
    ts.emplace_back([&, t] {
      std::string sp = dir + "/kfio_san_shard" + std::to_string(t) + ".bin";
      std::vector<char> lsrc(n / 4), ldst(n / 4);
      for (size_t i = 0; i < lsrc.size(); ++i) lsrc[i] = (char)(i + t);
      if (kf_write_file(sp.c_str(), lsrc.data(), lsrc.size(), 4) != 0) {
        rcs[t] = 6;
        return;
      }
      if (kf_read_file(sp.c_str(), ldst.data(), ldst.size(), 4) != 0) {
        rcs[t] = 7;
        return;
      }
      if (std::memcmp(lsrc.data(), ldst.data(), lsrc.size()) != 0) rcs[t] = 8;
      std::remove(sp.c_str());
    });
  }
  for (auto& t : ts) t.join();
  std::remove(p.c_str());
  for (int rc : rcs)
    if (rc) return rc;
  std::puts("kfio sanitize: OK");
  return 0;
}
