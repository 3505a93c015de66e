// queues.h — native flow-control queue primitives.
//
// Capability parity: reference SafeQueue plugins `listqueue` (FIFO
// doubly-linked list) and `maxminheap` (priority heap) behind a
// capability-based factory (`pkg/epp/flowcontrol/framework/plugins/queue/`),
// plus the atomic len/bytes stats decorator (`managedqueue.go:57`).
// Re-designed: items are referenced by opaque uint64 handles (the Python
// layer owns request objects); both queues support O(log n)/O(1) removal of
// arbitrary items (needed by the eviction subsystem) and expose head
// selection by an ordering key computed by the caller (fcfs = enqueue time,
// edf/slo-deadline = deadline), so ordering policies stay pluggable.
#pragma once
#include <cstdint>
#include <vector>
#include <unordered_map>
#include <stdexcept>

namespace ldsr {

struct QItem {
  uint64_t id;
  double key;     // ordering key (lower = dispatch first)
  int64_t bytes;
};

// FIFO doubly-linked list over a slab (listqueue).
class ListQueue {
 public:
  int64_t len() const { return len_; }
  int64_t bytes() const { return bytes_; }

  void push(uint64_t id, double key, int64_t nbytes) {
    (void)key;
    int32_t slot;
    if (!free_.empty()) { slot = free_.back(); free_.pop_back(); }
    else { slot = (int32_t)pool_.size(); pool_.emplace_back(); }
    Node& nd = pool_[slot];
    nd.item = {id, 0.0, nbytes};
    nd.prev = tail_; nd.next = -1; nd.live = true;
    if (tail_ >= 0) pool_[tail_].next = slot; else head_ = slot;
    tail_ = slot;
    pos_[id] = slot;
    len_++; bytes_ += nbytes;
  }

  bool peek(uint64_t* id) const {
    if (head_ < 0) return false;
    *id = pool_[head_].item.id;
    return true;
  }

  bool pop(uint64_t* id, int64_t* nbytes) {
    if (head_ < 0) return false;
    int32_t slot = head_;
    *id = pool_[slot].item.id;
    *nbytes = pool_[slot].item.bytes;
    unlink(slot);
    return true;
  }

  bool remove(uint64_t id, int64_t* nbytes) {
    auto it = pos_.find(id);
    if (it == pos_.end()) return false;
    *nbytes = pool_[it->second].item.bytes;
    unlink(it->second);
    return true;
  }

  // tail = newest (eviction picks newest-first victims by default)
  bool peek_tail(uint64_t* id) const {
    if (tail_ < 0) return false;
    *id = pool_[tail_].item.id;
    return true;
  }

 private:
  struct Node { QItem item; int32_t prev = -1, next = -1; bool live = false; };
  void unlink(int32_t slot) {
    Node& nd = pool_[slot];
    if (nd.prev >= 0) pool_[nd.prev].next = nd.next; else head_ = nd.next;
    if (nd.next >= 0) pool_[nd.next].prev = nd.prev; else tail_ = nd.prev;
    nd.live = false;
    pos_.erase(nd.item.id);
    free_.push_back(slot);
    len_--; bytes_ -= nd.item.bytes;
  }
  std::vector<Node> pool_;
  std::vector<int32_t> free_;
  std::unordered_map<uint64_t, int32_t> pos_;
  int32_t head_ = -1, tail_ = -1;
  int64_t len_ = 0, bytes_ = 0;
};

// Indexed min-heap by key with max tracking + arbitrary removal (maxminheap).
class MaxMinHeap {
 public:
  int64_t len() const { return (int64_t)heap_.size(); }
  int64_t bytes() const { return bytes_; }

  void push(uint64_t id, double key, int64_t nbytes) {
    int32_t idx = (int32_t)heap_.size();
    heap_.push_back({id, key, nbytes});
    pos_[id] = idx;
    bytes_ += nbytes;
    sift_up(idx);
  }

  bool peek(uint64_t* id) const {  // min-key head (dispatch order)
    if (heap_.empty()) return false;
    *id = heap_[0].id;
    return true;
  }

  bool pop(uint64_t* id, int64_t* nbytes) {
    if (heap_.empty()) return false;
    *id = heap_[0].id;
    *nbytes = heap_[0].bytes;
    remove_at(0);
    return true;
  }

  // max-key item = lowest dispatch priority = default eviction victim
  bool peek_max(uint64_t* id) const {
    if (heap_.empty()) return false;
    int32_t best = 0;
    // max of a min-heap lives in the leaves; linear scan of lower half
    for (int32_t i = (int32_t)heap_.size() / 2; i < (int32_t)heap_.size(); ++i)
      if (heap_[i].key > heap_[best].key) best = i;
    *id = heap_[best].id;
    return true;
  }

  bool remove(uint64_t id, int64_t* nbytes) {
    auto it = pos_.find(id);
    if (it == pos_.end()) return false;
    *nbytes = heap_[it->second].bytes;
    remove_at(it->second);
    return true;
  }

 private:
  void sift_up(int32_t i) {
    while (i > 0) {
      int32_t p = (i - 1) / 2;
      if (heap_[p].key <= heap_[i].key) break;
      swap_(i, p);
      i = p;
    }
  }
  void sift_down(int32_t i) {
    int32_t n = (int32_t)heap_.size();
    for (;;) {
      int32_t l = 2 * i + 1, r = 2 * i + 2, best = i;
      if (l < n && heap_[l].key < heap_[best].key) best = l;
      if (r < n && heap_[r].key < heap_[best].key) best = r;
      if (best == i) break;
      swap_(i, best);
      i = best;
    }
  }
  void swap_(int32_t a, int32_t b) {
    std::swap(heap_[a], heap_[b]);
    pos_[heap_[a].id] = a;
    pos_[heap_[b].id] = b;
  }
  void remove_at(int32_t i) {
    bytes_ -= heap_[i].bytes;
    pos_.erase(heap_[i].id);
    int32_t last = (int32_t)heap_.size() - 1;
    if (i != last) {
      heap_[i] = heap_[last];
      pos_[heap_[i].id] = i;
      heap_.pop_back();
      sift_down(i);
      sift_up(i);
    } else {
      heap_.pop_back();
    }
  }
  std::vector<QItem> heap_;
  std::unordered_map<uint64_t, int32_t> pos_;
  int64_t bytes_ = 0;
};

}  // namespace ldsr
