// Placeholder plane: fills the Node's pool/device fields so ADD_NODE
// carries the hipIpc handle; the fast send path lands in the next phase
// (shm rings + xGMI copies). TCP staging keeps device payloads correct
// meanwhile.
#include "gpu_plane.h"

#include <cstring>

#include "hip_pool.h"
#include "postoffice.h"

namespace xps {

class StubPlane : public DataPlane {
 public:
  StubPlane(Postoffice* po, int device) : po_(po), device_(device) {}
  bool CanSend(const Message&, const Node&) override { return false; }
  int64_t Send(Message&, const Node&) override { return -1; }
  void FillSelf(Node* self) override {
    self->dev_id = device_;
    auto* pool = HbmPool::Get();
    if (pool->initialized()) {
      self->pool_capacity = pool->capacity();
      memcpy(self->pool_handle, pool->ipc_handle(), kIpcHandleBytes);
    }
  }

 private:
  Postoffice* po_;
  int device_;
};

std::shared_ptr<DataPlane> CreateGpuPlane(Postoffice* po, int device) {
  if (device < 0) return nullptr;
  return std::make_shared<StubPlane>(po, device);
}

}  // namespace xps
