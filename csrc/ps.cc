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

static void StartOne(const std::string& role, int customer_id, bool do_barrier, int device,
                     int idx) {
  Postoffice* po = GetPO(role, idx);
  int base = Environment::Get()->GetInt("DMLC_RANK", -1);
  if (base >= 0 && role != "scheduler") {
    po->set_preferred_rank(base * std::max(1, Environment::Get()->GetInt("DMLC_GROUP_SIZE", 1)) +
                           idx);
  }
  po->EnsureVan();
  if (role != "scheduler") {
    if (device >= 0) HbmPool::Get()->Init(device);
    // device < 0 still gets the plane: same-host HOST payloads ride the
    // shm rings + host pool zero-copy (XPS_HOST_PLANE=0 forces pure TCP)
    if (device >= 0 || Environment::Get()->GetInt("XPS_HOST_PLANE", 1)) {
      auto plane = CreateGpuPlane(po, device);
      if (plane) po->van()->SetDataPlane(plane);
    }
  }
  po->Start(customer_id, do_barrier);
}

// DMLC_GROUP_SIZE instance groups (ps-lite ps.h:110-138 parity): one
// logical rank expands into group_size PS instances per role in this
// process, each with its own Van. All instances of this process share
// one GPU (the per-GPU sharding is the joint-process-per-GPU layout).
static int GroupSize() { return std::max(1, Environment::Get()->GetInt("DMLC_GROUP_SIZE", 1)); }

void Start(int customer_id, const std::string& role, int rank, bool do_barrier, int gpu_device) {
  if (rank >= 0) Environment::Get()->Set("DMLC_RANK", std::to_string(rank));
  int device = ResolveDevice(gpu_device);
  int group = role == "scheduler" ? 1 : GroupSize();
  std::vector<std::thread> threads;
  for (int g = 0; g < group; ++g) {
    if (role == "joint" || role == "server") {
      threads.emplace_back([=] { StartOne("server", customer_id, do_barrier, device, g); });
    }
    if (role == "joint" || role == "worker") {
      threads.emplace_back([=] { StartOne("worker", customer_id, do_barrier, device, g); });
    }
  }
  if (role == "scheduler") StartOne("scheduler", customer_id, do_barrier, device, 0);
  for (auto& t : threads) t.join();
}

void Finalize(int customer_id, const std::string& role, bool do_barrier) {
  int group = role == "scheduler" ? 1 : GroupSize();
  std::vector<std::thread> threads;
  for (int g = 0; g < group; ++g) {
    if (role == "joint" || role == "server") {
      threads.emplace_back(
          [=] { Postoffice::GetServer(g)->Finalize(customer_id, do_barrier); });
    }
    if (role == "joint" || role == "worker") {
      threads.emplace_back(
          [=] { Postoffice::GetWorker(g)->Finalize(customer_id, do_barrier); });
    }
  }
  if (role == "scheduler") Postoffice::GetScheduler()->Finalize(customer_id, do_barrier);
  for (auto& t : threads) t.join();
}

int NumWorkers() { return Environment::Get()->GetInt("DMLC_NUM_WORKER", 0); }
int NumServers() { return Environment::Get()->GetInt("DMLC_NUM_SERVER", 0); }

}  // namespace xps
