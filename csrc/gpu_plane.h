// The MI355X data plane: same-host metadata over lock-free shm rings,
// payloads moved GPU-to-GPU over xGMI (hipIpc-mapped pools +
// hipMemcpyAsync / HIP kernels on per-peer side streams), completions
// via a hipEvent poller thread.
//
// Reference parity: replaces ps-lite's RDMA data path — rendezvous +
// one-sided RDMA_WRITE (src/rdma_van.h, rdma_transport.h) and the POSIX
// shm IPCTransport (rdma_transport.h:591-617) — per SURVEY.md §5.8.
#pragma once

#include <memory>

#include "van.h"

namespace xps {

class Postoffice;

// Returns the plane for this van, or nullptr when no GPU is attached.
std::shared_ptr<DataPlane> CreateGpuPlane(Postoffice* po, int device);

}  // namespace xps
