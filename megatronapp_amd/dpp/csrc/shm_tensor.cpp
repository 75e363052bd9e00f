// MegaDPP transport: POSIX shared-memory tensor channel for pipeline
// parallelism (MI355X-native redesign of the reference's
// shm_tensor_new_rdma{,_pre_alloc}.cpp — SURVEY.md §2.1).
//
// Design differences from the reference, by intent:
//  * slots are TAGGED ((chunk, microbatch) headers) instead of strict
//    FIFO, so the SENDER may reorder pushes freely (the MegaDPP dynamic
//    schedule) while receivers block on the exact tensor they need;
//  * the greedy ordering policy lives in Python worker threads
//    (dpp/transport.py) — the C++ layer is a dumb, fast mailbox;
//  * device staging uses hipMemcpy through the slot buffer (the
//    reference's cudaMemcpy D2H/H2D staging); on a CPU build the same
//    code path memcpys host tensors, which is how the channel is unit
//    tested off-GPU.  Cross-node RDMA is out of scope for this round —
//    single-node PP rides this channel, multi-node PP rides RCCL.
//
// Channel layout (one per (src_rank, dst_rank, direction)):
//   shm segment  /mgapp_dpp_{dir}_{src}_{dst}:
//     ChannelHeader { nslots, slot_bytes }
//     Slot[nslots]  { state (0 free / 1 full), chunk, microbatch, nbytes,
//                     sem_ready, sem_free, payload[] }

#include <torch/extension.h>

#include <fcntl.h>
#include <semaphore.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <unistd.h>

#include <cstring>
#include <map>
#include <stdexcept>
#include <string>

#ifdef __HIP_PLATFORM_AMD__
#include <hip/hip_runtime.h>
#endif

namespace {

struct SlotHeader {
  int state;       // 0 free, 1 full (guarded by sems)
  int chunk;
  int microbatch;
  long nbytes;
  sem_t sem_ready; // posted by producer when slot filled
  sem_t sem_free;  // posted by consumer when slot drained
};

struct ChannelHeader {
  int nslots;
  long slot_bytes;  // payload bytes per slot
  // device fast path: the creator hipMalloc's nslots*slot_bytes of GPU
  // memory and publishes its hipIpcMemHandle here; the opener maps it
  // and GPU tensors move device-to-device (xGMI peer / same-GPU copy)
  // while the control plane (tags + semaphores) stays in this shm.
  int device_ok;
  unsigned char ipc_handle[64];  // sizeof(hipIpcMemHandle_t)
};

struct Channel {
  std::string name;
  ChannelHeader* hdr{};
  char* base{};
  size_t total_bytes{};
  bool creator{};
  char* dev_base{};   // mapped device payload area (or null)

  SlotHeader* slot(int i) {
    return (SlotHeader*)(base + sizeof(ChannelHeader) +
                         (size_t)i * (sizeof(SlotHeader) + hdr->slot_bytes));
  }
  char* payload(int i) { return (char*)slot(i) + sizeof(SlotHeader); }
  char* dev_payload(int i) {
    return dev_base ? dev_base + (size_t)i * hdr->slot_bytes : nullptr;
  }
};

std::map<std::string, Channel>& channels() {
  static std::map<std::string, Channel> c;
  return c;
}

std::string chan_name(const std::string& dir, int src, int dst) {
  return "/mgapp_dpp_" + dir + "_" + std::to_string(src) + "_" +
         std::to_string(dst);
}

Channel& get_channel(const std::string& dir, int src, int dst) {
  auto it = channels().find(chan_name(dir, src, dst));
  TORCH_CHECK(it != channels().end(), "DPP channel not initialized: ",
              chan_name(dir, src, dst));
  return it->second;
}

void copy_in(char* dst, char* dev_dst, const torch::Tensor& t) {
  const long n = t.numel() * t.element_size();
#ifdef __HIP_PLATFORM_AMD__
  if (t.is_cuda()) {
    if (dev_dst) {
      auto st = hipMemcpy(dev_dst, t.data_ptr(), n, hipMemcpyDeviceToDevice);
      TORCH_CHECK(st == hipSuccess, "hipMemcpy D2D failed");
      return;
    }
    auto st = hipMemcpy(dst, t.data_ptr(), n, hipMemcpyDeviceToHost);
    TORCH_CHECK(st == hipSuccess, "hipMemcpy D2H failed");
    return;
  }
#endif
  std::memcpy(dst, t.data_ptr(), n);
}

void copy_out(torch::Tensor& t, const char* src, const char* dev_src,
              long n) {
#ifdef __HIP_PLATFORM_AMD__
  if (t.is_cuda()) {
    if (dev_src) {
      auto st = hipMemcpy(t.data_ptr(), dev_src, n,
                          hipMemcpyDeviceToDevice);
      TORCH_CHECK(st == hipSuccess, "hipMemcpy D2D failed");
      return;
    }
    auto st = hipMemcpy(t.data_ptr(), src, n, hipMemcpyHostToDevice);
    TORCH_CHECK(st == hipSuccess, "hipMemcpy H2D failed");
    return;
  }
#endif
  std::memcpy(t.data_ptr(), src, n);
}

}  // namespace

// Create (or open) a channel.  Exactly one side passes create=true and the
// pair must agree on nslots/slot_bytes.
void init_channel(const std::string& dir, int src, int dst, long slot_bytes,
                  int nslots, bool create, bool use_device) {
  const std::string name = chan_name(dir, src, dst);
  if (channels().count(name)) return;
  const size_t total = sizeof(ChannelHeader) +
                       (size_t)nslots * (sizeof(SlotHeader) + slot_bytes);
  int fd;
  if (create) {
    shm_unlink(name.c_str());
    fd = shm_open(name.c_str(), O_CREAT | O_EXCL | O_RDWR, 0600);
    TORCH_CHECK(fd >= 0, "shm_open create failed for ", name);
    TORCH_CHECK(ftruncate(fd, total) == 0, "ftruncate failed");
  } else {
    // wait for the creator
    for (int tries = 0; tries < 6000; ++tries) {
      fd = shm_open(name.c_str(), O_RDWR, 0600);
      if (fd >= 0) break;
      usleep(10000);
    }
    TORCH_CHECK(fd >= 0, "shm_open open failed for ", name);
    // wait until sized
    struct stat st {};
    for (int tries = 0; tries < 6000; ++tries) {
      fstat(fd, &st);
      if ((size_t)st.st_size >= total) break;
      usleep(10000);
    }
  }
  void* base = mmap(nullptr, total, PROT_READ | PROT_WRITE, MAP_SHARED, fd, 0);
  close(fd);
  TORCH_CHECK(base != MAP_FAILED, "mmap failed for ", name);

  Channel ch;
  ch.name = name;
  ch.base = (char*)base;
  ch.hdr = (ChannelHeader*)base;
  ch.total_bytes = total;
  ch.creator = create;
  if (create) {
    ch.hdr->slot_bytes = slot_bytes;
    ch.hdr->device_ok = 0;
#ifdef __HIP_PLATFORM_AMD__
    if (use_device) {
      void* dptr = nullptr;
      if (hipMalloc(&dptr, (size_t)nslots * slot_bytes) == hipSuccess) {
        hipIpcMemHandle_t h;
        if (hipIpcGetMemHandle(&h, dptr) == hipSuccess) {
          static_assert(sizeof(h) <= 64, "ipc handle too large");
          std::memcpy(ch.hdr->ipc_handle, &h, sizeof(h));
          ch.dev_base = (char*)dptr;
          ch.hdr->device_ok = 1;
        } else {
          hipFree(dptr);
        }
      }
    }
#endif
    for (int i = 0; i < nslots; ++i) {
      SlotHeader* s = (SlotHeader*)(ch.base + sizeof(ChannelHeader) +
                                    (size_t)i * (sizeof(SlotHeader) + slot_bytes));
      s->state = 0;
      s->chunk = -1;
      s->microbatch = -1;
      s->nbytes = 0;
      sem_init(&s->sem_ready, 1, 0);
      sem_init(&s->sem_free, 1, 1);
    }
    __sync_synchronize();
    ch.hdr->nslots = nslots;  // publish last
  } else {
    // wait for publication
    for (int tries = 0; tries < 6000 && ch.hdr->nslots == 0; ++tries)
      usleep(10000);
    TORCH_CHECK(ch.hdr->nslots == nslots, "channel slot mismatch");
#ifdef __HIP_PLATFORM_AMD__
    if (use_device && ch.hdr->device_ok) {
      hipIpcMemHandle_t h;
      std::memcpy(&h, ch.hdr->ipc_handle, sizeof(h));
      void* dptr = nullptr;
      if (hipIpcOpenMemHandle(&dptr, h, hipIpcMemLazyEnablePeerAccess) ==
          hipSuccess)
        ch.dev_base = (char*)dptr;
    }
#endif
  }
  channels().emplace(name, ch);
}

// Blocking put: claims a free slot, copies the tensor (D2H when on GPU),
// tags it and posts ready.
void put_tensor(const std::string& dir, int src, int dst, int chunk,
                int microbatch, torch::Tensor t) {
  Channel& ch = get_channel(dir, src, dst);
  const long n = t.numel() * t.element_size();
  TORCH_CHECK(n <= ch.hdr->slot_bytes, "tensor larger than DPP slot: ", n,
              " > ", ch.hdr->slot_bytes);
  TORCH_CHECK(t.is_contiguous());
  // claim any free slot (spin over slots with trywait; block on slot 0's
  // free sem as a backstop to avoid busy-wait)
  for (;;) {
    for (int i = 0; i < ch.hdr->nslots; ++i) {
      SlotHeader* s = ch.slot(i);
      if (sem_trywait(&s->sem_free) == 0) {
        copy_in(ch.payload(i), ch.dev_payload(i), t);
        s->chunk = chunk;
        s->microbatch = microbatch;
        s->nbytes = n;
        __sync_synchronize();
        s->state = 1;
        sem_post(&s->sem_ready);
        return;
      }
    }
    usleep(50);
  }
}

// Blocking get of the tensor tagged (chunk, microbatch) into `out`.
void get_tensor(const std::string& dir, int src, int dst, int chunk,
                int microbatch, torch::Tensor out) {
  Channel& ch = get_channel(dir, src, dst);
  for (;;) {
    for (int i = 0; i < ch.hdr->nslots; ++i) {
      SlotHeader* s = ch.slot(i);
      if (s->state == 1 && s->chunk == chunk && s->microbatch == microbatch) {
        // claim it: ready sem must be consumable
        if (sem_trywait(&s->sem_ready) == 0) {
          if (s->chunk == chunk && s->microbatch == microbatch) {
            const long n = s->nbytes;
            TORCH_CHECK(out.numel() * out.element_size() == n,
                        "DPP get size mismatch");
            copy_out(out, ch.payload(i), ch.dev_payload(i), n);
            s->state = 0;
            s->chunk = -1;
            s->microbatch = -1;
            __sync_synchronize();
            sem_post(&s->sem_free);
            return;
          }
          sem_post(&s->sem_ready);  // raced; put back
        }
      }
    }
    usleep(50);
  }
}

// Non-blocking probe: is (chunk, microbatch) available?
bool probe_tensor(const std::string& dir, int src, int dst, int chunk,
                  int microbatch) {
  Channel& ch = get_channel(dir, src, dst);
  for (int i = 0; i < ch.hdr->nslots; ++i) {
    SlotHeader* s = ch.slot(i);
    if (s->state == 1 && s->chunk == chunk && s->microbatch == microbatch)
      return true;
  }
  return false;
}

void clean_channels() {
  for (auto& [name, ch] : channels()) {
#ifdef __HIP_PLATFORM_AMD__
    if (ch.dev_base) {
      if (ch.creator) hipFree(ch.dev_base);
      else hipIpcCloseMemHandle(ch.dev_base);
    }
#endif
    munmap(ch.base, ch.total_bytes);
    if (ch.creator) shm_unlink(name.c_str());
  }
  channels().clear();
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("init_channel", &init_channel, "create/open a DPP shm channel",
        pybind11::arg("dir"), pybind11::arg("src"), pybind11::arg("dst"),
        pybind11::arg("slot_bytes"), pybind11::arg("nslots"),
        pybind11::arg("create"), pybind11::arg("use_device") = true);
  // GIL released during blocking copies/spins so Python worker threads
  // (the DPP policy scheduler) keep running
  m.def("put_tensor", &put_tensor,
        pybind11::call_guard<pybind11::gil_scoped_release>());
  m.def("get_tensor", &get_tensor,
        pybind11::call_guard<pybind11::gil_scoped_release>());
  m.def("probe_tensor", &probe_tensor,
        pybind11::call_guard<pybind11::gil_scoped_release>());
  m.def("clean_channels", &clean_channels);
}
