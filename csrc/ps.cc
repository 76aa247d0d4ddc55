#include "ps.h"

#include <thread>

#include "gpu_plane.h"
#include "hip_pool.h"
#include "hip_util.h"
#include "van.h"

namespace xps {

Postoffice* GetPO(const std::string& role, int idx) {
  if (role == "scheduler") return Postoffice::GetScheduler();
  if (role == "server") return Postoffice::GetServer(idx);
  if (role == "worker") return Postoffice::GetWorker(idx);
  XPS_LOG(Fatal) << "unknown role " << role;
  return nullptr;
}

static int ResolveDevice(int gpu_device) {
  if (gpu_device != -2) return gpu_device;
  int dev = Environment::Get()->GetInt("XPS_DEV_ID", -1);
  if (dev >= 0 && !gpu::Available()) {
    XPS_LOG(Warning) << "XPS_DEV_ID=" << dev << " but no GPU visible; running CPU-only";
    return -1;
  }
  return dev;
}

static void StartOne(const std::string& role, int customer_id, bool do_barrier, int device) {
  Postoffice* po = GetPO(role);
  po->EnsureVan();
  if (device >= 0 && role != "scheduler") {
    HbmPool::Get()->Init(device);
    auto plane = CreateGpuPlane(po, device);
    if (plane) po->van()->SetDataPlane(plane);
  }
  po->Start(customer_id, do_barrier);
}

void Start(int customer_id, const std::string& role, int rank, bool do_barrier, int gpu_device) {
  if (rank >= 0) Environment::Get()->Set("DMLC_RANK", std::to_string(rank));
  int device = ResolveDevice(gpu_device);
  if (role == "joint") {
    // server + worker instances must register concurrently
    std::thread ts([&] { StartOne("server", customer_id, do_barrier, device); });
    StartOne("worker", customer_id, do_barrier, device);
    ts.join();
  } else {
    StartOne(role, customer_id, do_barrier, device);
  }
}

void Finalize(int customer_id, const std::string& role, bool do_barrier) {
  if (role == "joint") {
    std::thread ts([&] { Postoffice::GetServer()->Finalize(customer_id, do_barrier); });
    Postoffice::GetWorker()->Finalize(customer_id, do_barrier);
    ts.join();
  } else {
    GetPO(role)->Finalize(customer_id, do_barrier);
  }
}

int NumWorkers() { return Environment::Get()->GetInt("DMLC_NUM_WORKER", 0); }
int NumServers() { return Environment::Get()->GetInt("DMLC_NUM_SERVER", 0); }

}  // namespace xps
