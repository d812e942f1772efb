// Native RCCL comm core (SURVEY.md §1b "Comm core (C++)"): the framework's
// own communicator over RCCL/xGMI, bootstrapped out-of-band (the unique id
// travels over the torch.distributed rendezvous store from Python — RCCL
// itself never needs a second rendezvous).
//
// Collectives are enqueued on the CURRENT torch stream (stream-ordered
// semantics, hipGraph-capturable); one communicator per process, one process
// per GPU.
#include <rccl/rccl.h>

#include <stdexcept>
#include <string>

namespace comm {

inline void RCCL_CHECK_FN(ncclResult_t r) {
  if (r != ncclSuccess)
    throw std::runtime_error(std::string("RCCL error: ") + ncclGetErrorString(r));
}

static ncclComm_t g_comm = nullptr;
static int g_rank = -1, g_world = 0;

}  // namespace comm
