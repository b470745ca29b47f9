// Work pools (deques of search nodes).
//
// Semantics mirror the reference's pools:
//   - Pool<T>       == SinglePool      (reference `lib/commons/Pool.chpl:12-74`)
//   - ParPool<T>    == SinglePool_par  (reference `lib/commons/Pool_par.chpl:12-192`)
//
// A pool is a contiguous deque: DFS pops from the back, BFS pops from the front,
// chunked GPU offload bulk-pops from the back.  popBackBulk returns 0 when
// size < m, otherwise pops min(size, M) nodes (Pool.chpl:50-60).
// ParPool adds a CAS spin lock plus "Free" (caller-holds-lock) variants, a
// steal-half-from-front bulk pop used by intra-node work stealing
// (Pool_par.chpl:180-191) and a steal-half-from-back variant (Pool_par.chpl:153).
//
// Implementation is our own: std::vector-backed with front compaction on growth
// (the reference leaks front space and just doubles; behavior is equivalent).
#pragma once
#include <atomic>
#include <cstring>
#include <thread>
#include <vector>

namespace gats {

constexpr size_t POOL_INITIAL_CAPACITY = 1024;

template <class T>
class Pool {
 public:
  Pool() : buf_(POOL_INITIAL_CAPACITY) {}

  size_t size() const { return sz_; }
  bool empty() const { return sz_ == 0; }

  void pushBack(const T& v) {
    ensure_back_space(1);
    buf_[front_ + sz_] = v;
    sz_ += 1;
  }

  void pushBackBulk(const T* v, size_t n) {
    ensure_back_space(n);
    std::memcpy(buf_.data() + front_ + sz_, v, n * sizeof(T));
    sz_ += n;
  }

  bool popBack(T& out) {
    if (sz_ == 0) return false;
    sz_ -= 1;
    out = buf_[front_ + sz_];
    return true;
  }

  bool popFront(T& out) {
    if (sz_ == 0) return false;
    out = buf_[front_];
    front_ += 1;
    sz_ -= 1;
    return true;
  }

  // Bulk removal from the back; the m/M offload window (Pool.chpl:50-60).
  size_t popBackBulk(size_t m, size_t M, T* out) {
    if (sz_ < m) return 0;
    size_t n = sz_ < M ? sz_ : M;
    sz_ -= n;
    std::memcpy(out, buf_.data() + front_ + sz_, n * sizeof(T));
    return n;
  }

  // Bulk removal from the front (used to split a BFS frontier).
  size_t popFrontBulk(size_t n, T* out) {
    if (n > sz_) n = sz_;
    std::memcpy(out, buf_.data() + front_, n * sizeof(T));
    front_ += n;
    sz_ -= n;
    return n;
  }

  const T* data() const { return buf_.data() + front_; }
  T* data() { return buf_.data() + front_; }

  void clear() {
    front_ = 0;
    sz_ = 0;
  }

 private:
  void ensure_back_space(size_t n) {
    if (front_ + sz_ + n <= buf_.size()) return;
    if (front_ > 0 && sz_ + n <= buf_.size() / 2) {
      // plenty of dead space at the front: compact instead of growing
      std::memmove(buf_.data(), buf_.data() + front_, sz_ * sizeof(T));
      front_ = 0;
      return;
    }
    size_t need = front_ + sz_ + n;
    size_t cap = buf_.size();
    while (cap < need) cap *= 2;
    buf_.resize(cap);
  }

  std::vector<T> buf_;
  size_t front_ = 0;
  size_t sz_ = 0;
};

// Parallel-safe pool guarded by a CAS spin lock (Pool_par.chpl:28-41).
// The *Free methods assume the caller holds the lock (or exclusively owns the
// pool), matching the reference's naming.
template <class T>
class ParPool {
 public:
  void acquireLock() {
    bool expected = false;
    while (!lock_.compare_exchange_weak(expected, true, std::memory_order_acquire)) {
      expected = false;
      std::this_thread::yield();
    }
  }
  bool tryLock() {
    bool expected = false;
    return lock_.compare_exchange_strong(expected, true, std::memory_order_acquire);
  }
  void releaseLock() { lock_.store(false, std::memory_order_release); }

  size_t sizeApprox() const { return inner_.size(); }

  void pushBack(const T& v) {
    acquireLock();
    inner_.pushBack(v);
    releaseLock();
  }
  void pushBackFree(const T& v) { inner_.pushBack(v); }
  void pushBackBulk(const T* v, size_t n) {
    acquireLock();
    inner_.pushBackBulk(v, n);
    releaseLock();
  }
  void pushBackBulkFree(const T* v, size_t n) { inner_.pushBackBulk(v, n); }

  bool popBack(T& out) {
    acquireLock();
    bool ok = inner_.popBack(out);
    releaseLock();
    return ok;
  }
  bool popBackFree(T& out) { return inner_.popBack(out); }

  size_t popBackBulk(size_t m, size_t M, T* out) {
    acquireLock();
    size_t n = inner_.popBackBulk(m, M, out);
    releaseLock();
    return n;
  }
  size_t popBackBulkFree(size_t m, size_t M, T* out) { return inner_.popBackBulk(m, M, out); }

  // Steal a FRACTION of the victim's nodes from the FRONT (oldest = shallowest
  // = biggest subtrees), only when the victim holds >= 2m nodes — the
  // reference's configurable steal percentage (popFrontBulkFree perc,
  // pfsp_multigpu_cuda.c:539 / Pool_ext.c:138-147; Chapel fixes perc=0.5,
  // Pool_par.chpl:180-191). Caller must hold the lock.
  size_t popFrontFracFree(size_t m, double perc, std::vector<T>& out) {
    size_t sz = inner_.size();
    if (sz < 2 * m) return 0;
    size_t n = static_cast<size_t>(sz * perc);
    if (n == 0) return 0;
    out.resize(n);
    return inner_.popFrontBulk(n, out.data());
  }

  size_t popFrontHalfFree(size_t m, std::vector<T>& out) {
    return popFrontFracFree(m, 0.5, out);
  }

  // Steal half from the BACK (Pool_par.chpl:153-165 popBackBulkFree(half)).
  size_t popBackHalfFree(size_t m, std::vector<T>& out) {
    size_t sz = inner_.size();
    if (sz < 2 * m) return 0;
    size_t n = sz / 2;
    out.resize(n);
    return inner_.popBackBulk(0, n, out.data());
  }

  Pool<T>& inner() { return inner_; }

 private:
  Pool<T> inner_;
  std::atomic<bool> lock_{false};
};

}  // namespace gats
