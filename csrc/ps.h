// System bring-up / teardown.
//
// Reference parity: ps-lite include/ps/ps.h (StartPS :110, Finalize :183,
// JOINT mode :66-75). Re-designed: roles of any mix can coexist in one
// process; "joint" starts one server + one worker instance (the BytePS
// per-GPU layout); a GPU device attaches the HBM pool + xGMI data plane.
#pragma once

#include <string>

#include "postoffice.h"

namespace xps {

// role: "scheduler" | "server" | "worker" | "joint"
// gpu_device: -2 = auto (XPS_DEV_ID env or CPU), -1 = CPU, >=0 = HIP ordinal
void Start(int customer_id, const std::string& role, int rank = -1, bool do_barrier = true,
           int gpu_device = -2);
void Finalize(int customer_id, const std::string& role, bool do_barrier = true);

Postoffice* GetPO(const std::string& role, int idx = 0);

int NumWorkers();
int NumServers();

}  // namespace xps
