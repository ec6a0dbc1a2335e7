// shm_ring.cpp — lock-free shared-memory SPSC transition rings.
//
// Native replacement for the reference's Redis transition plane (SURVEY
// §2.7 C1/C2: per-step pickled rpush + 10ms lrange/ltrim polling): each
// player process owns one single-producer/single-consumer ring in POSIX
// shared memory; the learner drains all rings and stages blocks straight
// into its pinned H2D path.  The learner can hipHostRegister the mapped
// region (pin()) so copies out of the ring are DMA-able.
//
// Slot layout: [u32 task | u32 n | f32 payload...] with payload =
// states[n*Ds] actions[n*Da] rewards[n] next_states[n*Ds] dones[n].
// head/tail are release/acquire atomics in the mapped header — correct for
// one writer + one reader without locks.

#include <atomic>
#include <cstdint>
#include <cstring>
#include <string>
#include <stdexcept>

#include <fcntl.h>
#include <sys/mman.h>
#include <unistd.h>

#include <hip/hip_runtime.h>
#include <torch/extension.h>

namespace {

struct RingHeader {
  std::atomic<uint64_t> head;   // next slot to write (producer)
  std::atomic<uint64_t> tail;   // next slot to read (consumer)
  std::atomic<uint64_t> drops;  // full-ring push failures (any producer) —
                                // lives in the header so the CONSUMER's
                                // ring object reports them too
  uint32_t n_slots;
  uint32_t slot_floats;         // payload capacity per slot
  uint32_t state_dim;
  uint32_t action_dim;
  uint32_t magic;
};

constexpr uint32_t MAGIC = 0xD5AC0002;

struct Slot {
  uint32_t task;
  uint32_t n;
  // float payload[] follows
};

class ShmRing {
 public:
  ShmRing(const std::string& name, int64_t n_slots, int64_t slot_floats,
          int64_t state_dim, int64_t action_dim, bool create)
      : name_(name), owner_(create) {
    const size_t slot_bytes = sizeof(Slot) + (size_t)slot_floats * 4;
    size_ = sizeof(RingHeader) + slot_bytes * (size_t)n_slots;
    int fd = create
        ? shm_open(name.c_str(), O_CREAT | O_RDWR, 0600)
        : shm_open(name.c_str(), O_RDWR, 0600);
    if (fd < 0) throw std::runtime_error("shm_open failed for " + name);
    if (create && ftruncate(fd, (off_t)size_) != 0) {
      close(fd);
      throw std::runtime_error("ftruncate failed");
    }
    if (!create) {
      // size from existing header
      void* probe = mmap(nullptr, sizeof(RingHeader), PROT_READ, MAP_SHARED,
                         fd, 0);
      if (probe == MAP_FAILED) { close(fd); throw std::runtime_error("mmap probe"); }
      auto* h = (RingHeader*)probe;
      if (h->magic != MAGIC) { munmap(probe, sizeof(RingHeader)); close(fd);
        throw std::runtime_error("ring not initialised: " + name); }
      n_slots = h->n_slots;
      slot_floats = h->slot_floats;
      state_dim = h->state_dim;
      action_dim = h->action_dim;
      munmap(probe, sizeof(RingHeader));
      size_ = sizeof(RingHeader)
              + (sizeof(Slot) + (size_t)slot_floats * 4) * (size_t)n_slots;
    }
    base_ = mmap(nullptr, size_, PROT_READ | PROT_WRITE, MAP_SHARED, fd, 0);
    close(fd);
    if (base_ == MAP_FAILED) throw std::runtime_error("mmap failed");
    hdr_ = (RingHeader*)base_;
    slots_ = (char*)base_ + sizeof(RingHeader);
    slot_bytes_ = sizeof(Slot) + (size_t)slot_floats * 4;
    if (create) {
      hdr_->head.store(0);
      hdr_->tail.store(0);
      hdr_->drops.store(0);
      hdr_->n_slots = (uint32_t)n_slots;
      hdr_->slot_floats = (uint32_t)slot_floats;
      hdr_->state_dim = (uint32_t)state_dim;
      hdr_->action_dim = (uint32_t)action_dim;
      hdr_->magic = MAGIC;
    }
  }

  ~ShmRing() {
    if (pinned_) hipHostUnregister(base_);
    munmap(base_, size_);
    if (owner_) shm_unlink(name_.c_str());
  }

  void pin() {
    if (!pinned_ && hipHostRegister(base_, size_, hipHostRegisterDefault)
                        == hipSuccess) {
      pinned_ = true;
    }
  }

  int64_t capacity() const { return hdr_->n_slots; }
  int64_t pending() const {
    return (int64_t)(hdr_->head.load(std::memory_order_acquire)
                     - hdr_->tail.load(std::memory_order_relaxed));
  }
  int64_t state_dim() const { return hdr_->state_dim; }
  int64_t action_dim() const { return hdr_->action_dim; }
  int64_t dropped() const {
    return (int64_t)hdr_->drops.load(std::memory_order_relaxed);
  }

  // producer: one transition block -> one slot. Returns false (and counts
  // a drop) when the ring is full — the producer never blocks.
  bool push(int64_t task, torch::Tensor states, torch::Tensor actions,
            torch::Tensor rewards, torch::Tensor next_states,
            torch::Tensor dones) {
    const auto Ds = hdr_->state_dim, Da = hdr_->action_dim;
    auto s = states.contiguous(), a = actions.contiguous();
    auto r = rewards.contiguous(), ns = next_states.contiguous();
    auto d = dones.contiguous();
    const int64_t n = s.size(0);
    const size_t need = (size_t)n * (2 * Ds + Da + 2);
    TORCH_CHECK(need <= hdr_->slot_floats, "block too large for ring slot");
    const uint64_t head = hdr_->head.load(std::memory_order_relaxed);
    const uint64_t tail = hdr_->tail.load(std::memory_order_acquire);
    if (head - tail >= hdr_->n_slots) {
      hdr_->drops.fetch_add(1, std::memory_order_relaxed);
      return false;
    }
    auto* slot = (Slot*)(slots_ + slot_bytes_ * (head % hdr_->n_slots));
    slot->task = (uint32_t)task;
    slot->n = (uint32_t)n;
    float* p = (float*)(slot + 1);
    auto cp = [&](const torch::Tensor& t, size_t cnt) {
      std::memcpy(p, t.data_ptr<float>(), cnt * 4);
      p += cnt;
    };
    cp(s, n * Ds);
    cp(a, n * Da);
    cp(r, n);
    cp(ns, n * Ds);
    cp(d, n);
    hdr_->head.store(head + 1, std::memory_order_release);
    return true;
  }

  // consumer: pops one block as CPU tensors (empty list when drained).
  // pinned=true allocates page-locked tensors so the follow-up H2D copies
  // are true async DMA (pageable copies measured ~0.5-2 ms each on the
  // learner hot loop).
  std::vector<torch::Tensor> pop(bool pinned = false) {
    const uint64_t tail = hdr_->tail.load(std::memory_order_relaxed);
    const uint64_t head = hdr_->head.load(std::memory_order_acquire);
    if (tail == head) return {};
    const auto Ds = hdr_->state_dim, Da = hdr_->action_dim;
    auto* slot = (Slot*)(slots_ + slot_bytes_ * (tail % hdr_->n_slots));
    const int64_t n = slot->n;
    const int64_t task = slot->task;
    auto opts = torch::TensorOptions().dtype(torch::kFloat32)
        .pinned_memory(pinned);
    auto s = torch::empty({n, (int64_t)Ds}, opts);
    auto a = torch::empty({n, (int64_t)Da}, opts);
    auto r = torch::empty({n, 1}, opts);
    auto ns = torch::empty({n, (int64_t)Ds}, opts);
    auto d = torch::empty({n, 1}, opts);
    const float* p = (const float*)(slot + 1);
    auto cp = [&](torch::Tensor& t, size_t cnt) {
      std::memcpy(t.data_ptr<float>(), p, cnt * 4);
      p += cnt;
    };
    cp(s, n * Ds);
    cp(a, n * Da);
    cp(r, n);
    cp(ns, n * Ds);
    cp(d, n);
    hdr_->tail.store(tail + 1, std::memory_order_release);
    auto t = torch::full({1}, (double)task, opts);
    return {t, s, a, r, ns, d};
  }

 private:
  std::string name_;
  bool owner_;
  bool pinned_ = false;
  size_t size_ = 0;
  size_t slot_bytes_ = 0;
  void* base_ = nullptr;
  RingHeader* hdr_ = nullptr;
  char* slots_ = nullptr;
};

}  // namespace

void register_shm_ring(pybind11::module_& m) {
  namespace py = pybind11;
  py::class_<ShmRing>(m, "ShmRing")
      .def(py::init([](const std::string& name, int64_t n_slots,
                       int64_t slot_floats, int64_t state_dim,
                       int64_t action_dim) {
             return new ShmRing(name, n_slots, slot_floats, state_dim,
                                action_dim, /*create=*/true);
           }),
           py::arg("name"), py::arg("n_slots"), py::arg("slot_floats"),
           py::arg("state_dim"), py::arg("action_dim"))
      .def_static("open", [](const std::string& name) {
        return new ShmRing(name, 0, 0, 0, 0, /*create=*/false);
      })
      .def("push", &ShmRing::push)
      .def("pop", &ShmRing::pop, pybind11::arg("pinned") = false)
      .def("pin", &ShmRing::pin)
      .def("pending", &ShmRing::pending)
      .def("capacity", &ShmRing::capacity)
      .def("dropped", &ShmRing::dropped)
      .def("state_dim", &ShmRing::state_dim)
      .def("action_dim", &ShmRing::action_dim);
}
