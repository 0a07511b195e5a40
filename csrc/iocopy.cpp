// io_uring bulk copy engine (raw syscalls — liburing is not present).
//
// Replaces the reference's two data movers (SURVEY.md §2.4 row "Data
// migration"): the shell tar pipe for the container writable layer
// (/root/reference/utils/copy.go:17-27) and the throwaway-ubuntu-container
// `mv` for volumes (utils/copy.go:74-128). One engine, host-side:
//
//   * pipelined io_uring READ/WRITE chains, queue depth 16, 1 MiB chunks —
//     reads complete out of order and immediately requeue as writes;
//   * sparse-aware: data extents via SEEK_DATA/SEEK_HOLE, holes are never
//     read or written (dst is ftruncated to full size);
//   * preserves mode/uid/gid/mtime, symlinks, hardlinks (within one call),
//     device nodes (overlayfs whiteouts are 0:0 char devices) and xattrs
//     (overlayfs opaque-dir markers etc.);
//   * graceful fallback: if io_uring is unavailable (EPERM in seccomp
//     sandboxes, old kernels), per-file copy_file_range, then read/write.
//
// Exposed to Python as gpu_docker_api_amd.ops._iocopy (pybind11).
#include <pybind11/pybind11.h>

#include <atomic>
#include <cerrno>
#include <climits>
#include <cstring>
#include <map>
#include <stdexcept>
#include <string>
#include <vector>

#include <dirent.h>
#include <fcntl.h>
#include <linux/io_uring.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <sys/syscall.h>
#include <sys/types.h>
#include <sys/xattr.h>
#include <unistd.h>
#include <utime.h>

namespace py = pybind11;

namespace {

[[noreturn]] void die(const std::string& what) {
  throw std::runtime_error(what + ": " + std::strerror(errno));
}

// ---------------------------------------------------------------------------
// Minimal io_uring wrapper (setup/enter via raw syscalls)
// ---------------------------------------------------------------------------

int sys_io_uring_setup(unsigned entries, struct io_uring_params* p) {
  return (int)syscall(__NR_io_uring_setup, entries, p);
}
int sys_io_uring_enter(int fd, unsigned to_submit, unsigned min_complete,
                       unsigned flags) {
  return (int)syscall(__NR_io_uring_enter, fd, to_submit, min_complete, flags,
                      nullptr, 0);
}

class Ring {
 public:
  static constexpr unsigned kEntries = 32;

  bool init() {
    std::memset(&params_, 0, sizeof(params_));
    fd_ = sys_io_uring_setup(kEntries, &params_);
    if (fd_ < 0) return false;

    size_t sring_sz = params_.sq_off.array + params_.sq_entries * sizeof(unsigned);
    size_t cring_sz =
        params_.cq_off.cqes + params_.cq_entries * sizeof(io_uring_cqe);
    bool single_mmap = params_.features & IORING_FEAT_SINGLE_MMAP;
    if (single_mmap && cring_sz > sring_sz) sring_sz = cring_sz;

    sq_ptr_ = mmap(nullptr, sring_sz, PROT_READ | PROT_WRITE,
                   MAP_SHARED | MAP_POPULATE, fd_, IORING_OFF_SQ_RING);
    if (sq_ptr_ == MAP_FAILED) return false;
    cq_ptr_ = single_mmap
                  ? sq_ptr_
                  : mmap(nullptr, cring_sz, PROT_READ | PROT_WRITE,
                         MAP_SHARED | MAP_POPULATE, fd_, IORING_OFF_CQ_RING);
    if (cq_ptr_ == MAP_FAILED) return false;
    sqes_ = (io_uring_sqe*)mmap(nullptr, params_.sq_entries * sizeof(io_uring_sqe),
                                PROT_READ | PROT_WRITE, MAP_SHARED | MAP_POPULATE,
                                fd_, IORING_OFF_SQES);
    if (sqes_ == MAP_FAILED) return false;

    auto* sq = (char*)sq_ptr_;
    sq_tail_ = (std::atomic<unsigned>*)(sq + params_.sq_off.tail);
    sq_mask_ = *(unsigned*)(sq + params_.sq_off.ring_mask);
    sq_array_ = (unsigned*)(sq + params_.sq_off.array);
    auto* cq = (char*)cq_ptr_;
    cq_head_ = (std::atomic<unsigned>*)(cq + params_.cq_off.head);
    cq_tail_ = (std::atomic<unsigned>*)(cq + params_.cq_off.tail);
    cq_mask_ = *(unsigned*)(cq + params_.cq_off.ring_mask);
    cqes_ = (io_uring_cqe*)(cq + params_.cq_off.cqes);
    ok_ = true;
    return true;
  }

  bool ok() const { return ok_; }

  // Queue one SQE (rw: IORING_OP_READ / IORING_OP_WRITE).
  void queue(unsigned op, int fd, void* buf, unsigned len, off_t off,
             unsigned long long user_data) {
    unsigned tail = sq_tail_->load(std::memory_order_relaxed);
    unsigned idx = tail & sq_mask_;
    io_uring_sqe* sqe = &sqes_[idx];
    std::memset(sqe, 0, sizeof(*sqe));
    sqe->opcode = op;
    sqe->fd = fd;
    sqe->addr = (unsigned long long)buf;
    sqe->len = len;
    sqe->off = (unsigned long long)off;
    sqe->user_data = user_data;
    sq_array_[idx] = idx;
    sq_tail_->store(tail + 1, std::memory_order_release);
    ++pending_submit_;
  }

  // Submit queued SQEs; optionally wait for at least `wait` completions.
  void submit(unsigned wait) {
    int ret = sys_io_uring_enter(fd_, pending_submit_, wait,
                                 wait ? IORING_ENTER_GETEVENTS : 0);
    if (ret < 0) die("io_uring_enter");
    pending_submit_ = 0;
  }

  // Pop one completion if available. Returns false when the CQ is empty.
  bool pop(unsigned long long* user_data, int* res) {
    unsigned head = cq_head_->load(std::memory_order_relaxed);
    if (head == cq_tail_->load(std::memory_order_acquire)) return false;
    io_uring_cqe* cqe = &cqes_[head & cq_mask_];
    *user_data = cqe->user_data;
    *res = cqe->res;
    cq_head_->store(head + 1, std::memory_order_release);
    return true;
  }

  void wait_one() { submit(1); }

  ~Ring() {
    if (fd_ >= 0) close(fd_);
  }

 private:
  int fd_ = -1;
  bool ok_ = false;
  io_uring_params params_{};
  void* sq_ptr_ = nullptr;
  void* cq_ptr_ = nullptr;
  io_uring_sqe* sqes_ = nullptr;
  std::atomic<unsigned>* sq_tail_ = nullptr;
  unsigned sq_mask_ = 0;
  unsigned* sq_array_ = nullptr;
  std::atomic<unsigned>* cq_head_ = nullptr;
  std::atomic<unsigned>* cq_tail_ = nullptr;
  unsigned cq_mask_ = 0;
  io_uring_cqe* cqes_ = nullptr;
  unsigned pending_submit_ = 0;
};

// ---------------------------------------------------------------------------
// File data copy
// ---------------------------------------------------------------------------

constexpr size_t kChunk = 1 << 20;  // 1 MiB
constexpr unsigned kDepth = 16;     // in-flight chunks

struct Chunk {
  std::vector<char> buf;
  off_t off = 0;
  unsigned len = 0;
  bool reading = false;
};

// Data extents of a (possibly sparse) file.
std::vector<std::pair<off_t, off_t>> data_extents(int fd, off_t size) {
  std::vector<std::pair<off_t, off_t>> out;
  off_t pos = 0;
  while (pos < size) {
    off_t data = lseek(fd, pos, SEEK_DATA);
    if (data < 0) {
      if (errno == ENXIO) break;          // trailing hole
      out.push_back({pos, size});         // SEEK_DATA unsupported: whole file
      break;
    }
    off_t hole = lseek(fd, data, SEEK_HOLE);
    if (hole < 0) hole = size;
    out.push_back({data, hole});
    pos = hole;
  }
  return out;
}

// Pipelined io_uring copy of one file's data. Returns false if the ring is
// unusable (caller falls back).
bool copy_data_uring(Ring& ring, int in_fd, int out_fd, off_t size) {
  if (!ring.ok()) return false;
  auto extents = data_extents(in_fd, size);
  if (ftruncate(out_fd, size) != 0) die("ftruncate");

  std::vector<Chunk> chunks(kDepth);
  for (auto& c : chunks) c.buf.resize(kChunk);
  std::vector<unsigned> free_idx;
  for (unsigned i = 0; i < kDepth; ++i) free_idx.push_back(i);
  unsigned in_flight = 0;

  size_t ext_i = 0;
  off_t cur = extents.empty() ? 0 : extents[0].first;

  auto queue_next_read = [&]() -> bool {
    while (ext_i < extents.size() && cur >= extents[ext_i].second) {
      ++ext_i;
      if (ext_i < extents.size()) cur = extents[ext_i].first;
    }
    if (ext_i >= extents.size() || free_idx.empty()) return false;
    unsigned idx = free_idx.back();
    free_idx.pop_back();
    Chunk& c = chunks[idx];
    c.off = cur;
    c.len = (unsigned)std::min<off_t>((off_t)kChunk, extents[ext_i].second - cur);
    c.reading = true;
    cur += c.len;
    ring.queue(IORING_OP_READ, in_fd, c.buf.data(), c.len, c.off, idx);
    ++in_flight;
    return true;
  };

  while (queue_next_read()) {
  }
  ring.submit(0);

  while (in_flight > 0) {
    unsigned long long ud;
    int res;
    if (!ring.pop(&ud, &res)) {
      ring.wait_one();
      continue;
    }
    --in_flight;
    Chunk& c = chunks[ud];
    if (res < 0) {
      errno = -res;
      die(c.reading ? "io_uring read" : "io_uring write");
    }
    if (c.reading) {
      if ((unsigned)res != c.len) c.len = (unsigned)res;  // short read near EOF
      c.reading = false;
      ring.queue(IORING_OP_WRITE, out_fd, c.buf.data(), c.len, c.off, ud);
      ++in_flight;
      ring.submit(0);
    } else {
      free_idx.push_back((unsigned)ud);
      bool queued = false;
      while (queue_next_read()) queued = true;
      if (queued) ring.submit(0);
    }
  }
  return true;
}

void copy_data_fallback(int in_fd, int out_fd, off_t size) {
  auto extents = data_extents(in_fd, size);
  if (ftruncate(out_fd, size) != 0) die("ftruncate");
  std::vector<char> buf(kChunk);
  for (auto [start, end] : extents) {
    off_t off = start;
    while (off < end) {
      size_t want = (size_t)std::min<off_t>((off_t)kChunk, end - off);
      // copy_file_range first (in-kernel, reflink-capable)
      off_t off_out = off;
      ssize_t n = copy_file_range(in_fd, &off, out_fd, &off_out, want, 0);
      if (n > 0) continue;  // both offsets advanced by the kernel
      // plain pread/pwrite
      ssize_t r = pread(in_fd, buf.data(), want, off);
      if (r < 0) die("pread");
      if (r == 0) break;
      ssize_t w = 0;
      while (w < r) {
        ssize_t k = pwrite(out_fd, buf.data() + w, r - w, off + w);
        if (k < 0) die("pwrite");
        w += k;
      }
      off += r;
    }
  }
}

// ---------------------------------------------------------------------------
// Metadata
// ---------------------------------------------------------------------------

void copy_xattrs(const std::string& src, const std::string& dst) {
  ssize_t list_sz = llistxattr(src.c_str(), nullptr, 0);
  if (list_sz <= 0) return;
  std::vector<char> names(list_sz);
  list_sz = llistxattr(src.c_str(), names.data(), names.size());
  if (list_sz <= 0) return;
  std::vector<char> value;
  for (char* p = names.data(); p < names.data() + list_sz;) {
    std::string name(p);
    p += name.size() + 1;
    ssize_t vs = lgetxattr(src.c_str(), name.c_str(), nullptr, 0);
    if (vs < 0) continue;
    value.resize(vs);
    vs = lgetxattr(src.c_str(), name.c_str(), value.data(), value.size());
    if (vs < 0) continue;
    // best-effort: security.* may need privileges we lack
    (void)lsetxattr(dst.c_str(), name.c_str(), value.data(), vs, 0);
  }
}

void copy_meta(const std::string& dst, const struct stat& st) {
  (void)lchown(dst.c_str(), st.st_uid, st.st_gid);
  if (!S_ISLNK(st.st_mode)) (void)chmod(dst.c_str(), st.st_mode & 07777);
  struct timespec times[2] = {st.st_atim, st.st_mtim};
  (void)utimensat(AT_FDCWD, dst.c_str(), times, AT_SYMLINK_NOFOLLOW);
}

// ---------------------------------------------------------------------------
// Tree walk
// ---------------------------------------------------------------------------

struct CopyStats {
  unsigned long long files = 0, dirs = 0, symlinks = 0, specials = 0;
  unsigned long long bytes = 0;
  bool used_uring = false;
};

void copy_tree_rec(Ring& ring, const std::string& src, const std::string& dst,
                   std::map<std::pair<dev_t, ino_t>, std::string>& hardlinks,
                   CopyStats& stats) {
  struct stat st;
  if (lstat(src.c_str(), &st) != 0) die("lstat " + src);

  if (S_ISDIR(st.st_mode)) {
    if (mkdir(dst.c_str(), st.st_mode & 07777) != 0 && errno != EEXIST)
      die("mkdir " + dst);
    ++stats.dirs;
    DIR* d = opendir(src.c_str());
    if (!d) die("opendir " + src);
    while (dirent* e = readdir(d)) {
      std::string n(e->d_name);
      if (n == "." || n == "..") continue;
      copy_tree_rec(ring, src + "/" + n, dst + "/" + n, hardlinks, stats);
    }
    closedir(d);
    copy_xattrs(src, dst);
    copy_meta(dst, st);
    return;
  }

  if (S_ISLNK(st.st_mode)) {
    std::vector<char> target(st.st_size ? st.st_size + 1 : PATH_MAX);
    ssize_t n = readlink(src.c_str(), target.data(), target.size() - 1);
    if (n < 0) die("readlink " + src);
    target[n] = 0;
    (void)unlink(dst.c_str());
    if (symlink(target.data(), dst.c_str()) != 0) die("symlink " + dst);
    copy_meta(dst, st);
    ++stats.symlinks;
    return;
  }

  if (S_ISCHR(st.st_mode) || S_ISBLK(st.st_mode) || S_ISFIFO(st.st_mode) ||
      S_ISSOCK(st.st_mode)) {
    // overlayfs whiteouts are 0:0 char devices — must be replicated
    (void)unlink(dst.c_str());
    if (mknod(dst.c_str(), st.st_mode, st.st_rdev) != 0) {
      // unprivileged environments cannot mknod arbitrary devices; skip
      ++stats.specials;
      return;
    }
    copy_xattrs(src, dst);
    copy_meta(dst, st);
    ++stats.specials;
    return;
  }

  // regular file
  auto key = std::make_pair(st.st_dev, st.st_ino);
  if (st.st_nlink > 1) {
    auto it = hardlinks.find(key);
    if (it != hardlinks.end()) {
      (void)unlink(dst.c_str());
      if (link(it->second.c_str(), dst.c_str()) == 0) {
        ++stats.files;
        return;
      }
    }
  }
  int in_fd = open(src.c_str(), O_RDONLY | O_CLOEXEC);
  if (in_fd < 0) die("open " + src);
  int out_fd =
      open(dst.c_str(), O_WRONLY | O_CREAT | O_TRUNC | O_CLOEXEC, 0600);
  if (out_fd < 0) {
    close(in_fd);
    die("open " + dst);
  }
  bool did = copy_data_uring(ring, in_fd, out_fd, st.st_size);
  if (!did) copy_data_fallback(in_fd, out_fd, st.st_size);
  stats.used_uring |= did;
  close(in_fd);
  close(out_fd);
  copy_xattrs(src, dst);
  copy_meta(dst, st);
  if (st.st_nlink > 1) hardlinks[key] = dst;
  ++stats.files;
  stats.bytes += (unsigned long long)st.st_size;
}

CopyStats copy_tree_impl(const std::string& src, const std::string& dst) {
  Ring ring;
  (void)ring.init();  // failure => per-file fallback path
  std::map<std::pair<dev_t, ino_t>, std::string> hardlinks;
  CopyStats stats;

  struct stat st;
  if (lstat(src.c_str(), &st) != 0) die("lstat " + src);
  if (!S_ISDIR(st.st_mode)) throw std::runtime_error(src + " is not a directory");
  if (mkdir(dst.c_str(), st.st_mode & 07777) != 0 && errno != EEXIST)
    die("mkdir " + dst);
  DIR* d = opendir(src.c_str());
  if (!d) die("opendir " + src);
  while (dirent* e = readdir(d)) {
    std::string n(e->d_name);
    if (n == "." || n == "..") continue;
    copy_tree_rec(ring, src + "/" + n, dst + "/" + n, hardlinks, stats);
  }
  closedir(d);
  copy_xattrs(src, dst);
  return stats;
}

py::dict copy_tree(const std::string& src, const std::string& dst) {
  CopyStats stats;
  {
    py::gil_scoped_release rel;  // the walk does blocking IO
    stats = copy_tree_impl(src, dst);
  }
  py::dict out;
  out["files"] = stats.files;
  out["dirs"] = stats.dirs;
  out["symlinks"] = stats.symlinks;
  out["specials"] = stats.specials;
  out["bytes"] = stats.bytes;
  out["io_uring"] = stats.used_uring;
  return out;
}

bool uring_available() {
  Ring r;
  return r.init();
}

}  // namespace

PYBIND11_MODULE(_iocopy, m) {
  m.doc() = "io_uring bulk copy engine (sparse-aware, xattr/whiteout-preserving)";
  m.def("copy_tree", &copy_tree, py::arg("src"), py::arg("dst"));
  m.def("uring_available", &uring_available);
}
