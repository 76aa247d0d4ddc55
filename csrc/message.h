// Message / Meta / Node / Control — the in-memory message format.
//
// Reference parity: ps-lite include/ps/internal/message.h (Meta :177,
// Node :66, Control :139, Message :262) and src/meta.h (packed wire
// structs). Re-designed: the wire format is a flat length-prefixed byte
// serialization (wire.h), and Node carries the MI355X-specific fields —
// host hash for same-host detection, GPU ordinal, and the process HBM
// pool hipIpc handle that replaces per-buffer RDMA rendezvous.
#pragma once

#include <array>
#include <sstream>
#include <string>
#include <vector>

#include "sarray.h"

namespace xps {

// opaque copy of hipIpcMemHandle_t (64 bytes on ROCm)
static const int kIpcHandleBytes = 64;

struct Node {
  enum Role : int { SCHEDULER = 0, SERVER = 1, WORKER = 2, JOINT = 3 };
  static const char* RoleStr(int r) {
    static const char* s[] = {"scheduler", "server", "worker", "joint"};
    return (r >= 0 && r < 4) ? s[r] : "?";
  }
  int role = -1;
  int id = kEmptyNodeID;       // assigned node id (scheduler=1, 8+2r / 9+2r)
  std::string hostname;        // ip or hostname used for TCP
  int port = -1;               // TCP control/data port
  int aux_id = -1;             // preferred rank (DMLC_RANK), -1 = any
  int dev_id = kCPU;           // HIP device ordinal owned by this node (-1 = cpu only)
  uint64_t host_hash = 0;      // same value <=> same physical host (for shm/ipc fast path)
  int is_recovery = 0;
  // the exported HBM pool: uniform <= 1 GiB slabs (hipIpcOpenMemHandle
  // deadlocks at >= 2 GiB), one ipc handle per slab; addresses on the
  // wire are global offsets over the concatenated slabs
  uint64_t pool_capacity = 0;  // total bytes (0 = no pool)
  uint64_t pool_slab_bytes = 0;
  std::vector<std::array<char, kIpcHandleBytes>> pool_handles;  // per slab
  uint64_t shm_uid = 0;        // uid for naming this node's shm data-plane segment
  // host-shm arena (zero-copy HOST payloads on the same host)
  uint64_t host_pool_uid = 0;  // names the segment (per process, shared by its nodes)
  uint64_t host_pool_capacity = 0;

  std::string DebugString() const {
    std::ostringstream os;
    os << "role=" << RoleStr(role) << " id=" << id << " addr=" << hostname << ":" << port
       << " dev=" << dev_id << " aux=" << aux_id << (is_recovery ? " recovery" : "");
    return os.str();
  }
};

struct Control {
  enum Command : int { EMPTY = 0, ADD_NODE, BARRIER, HEARTBEAT, TERMINATE, ACK, HANDSHAKE };
  int cmd = EMPTY;
  std::vector<Node> node;
  int barrier_group = 0;
  int barrier_token = 0;       // distinguishes consecutive barriers
  int instance_barrier = 0;    // 1 = count every instance (ps-lite INSTANCE_BARRIER)
  uint64_t msg_sig = 0;        // resender signature being ACKed
  bool empty() const { return cmd == EMPTY; }
  std::string DebugString() const {
    std::ostringstream os;
    static const char* names[] = {"EMPTY",     "ADD_NODE", "BARRIER",  "HEARTBEAT",
                                  "TERMINATE", "ACK",      "HANDSHAKE"};
    os << "cmd=" << names[cmd];
    if (cmd == BARRIER) os << " group=" << barrier_group << " tok=" << barrier_token;
    for (auto& n : node) os << " {" << n.DebugString() << "}";
    return os.str();
  }
};

// meta.option bits (shared by kv_app and the data plane)
static const int kOptInPlace = 1;   // pull response already written into dst buffer
static const int kOptValsByRef = 2; // vals travel as {pool offset, len}, not bytes
static const int kOptPullAddr = 4;  // meta.addr holds a valid pull-destination offset
static const int kOptHostAddr = 8;  // meta.addr is a HOST-shm-pool offset (else HBM pool)
// the response's handler work ran on the peer's PULL stream lane: the
// plane must enqueue the in-place write / completion event on that lane
// (handlers that stay on lane 0, e.g. sparse, leave this unset)
static const int kOptPullLane = 16;
// push ACK: meta.addr/val_len advertise the server's store-entry pool
// offset for this key — the worker may write subsequent assign pushes
// there ONE-SIDED (the rdma_van push_addr_ steady state, :486-508)
static const int kOptEntryAddr = 32;
// push REQUEST: the sender has a cached entry offset in meta.addr and
// asks the plane to write vals there itself + deliver the meta only
static const int kOptEntryPush = 64;

enum DataType : int { kChar = 0, kInt32, kInt64, kUint64, kFloat32, kFloat64, kUint8 };
inline size_t DataTypeSize(int t) {
  switch (t) {
    case kChar: case kUint8: return 1;
    case kInt32: case kFloat32: return 4;
    default: return 8;
  }
}

struct Meta {
  static const int kEmpty = -1;
  int app_id = kEmpty;
  int customer_id = kEmpty;
  int timestamp = kEmpty;
  int sender = kEmptyNodeID;
  int recver = kEmptyNodeID;
  bool request = false;
  bool push = false;
  bool pull = false;
  bool simple_app = false;
  int head = kEmpty;           // user-defined command
  std::string body;            // user-defined payload (SimpleApp)
  // fast-path fields (single-key messages; mirrors ps-lite meta.key/addr/val_len/option)
  Key key = 0;
  uint64_t addr = 0;           // pull: requester's destination pool offset
  int64_t val_len = 0;         // vals bytes
  int option = 0;
  uint64_t msg_sig = 0;        // resender signature (0 = none)
  // per-(sender -> recver) data-message sequence number (1-based; 0 =
  // unsequenced). A flow's messages may split between the shm/xGMI
  // plane and the TCP fallback by payload kind; the receiver re-orders
  // on this so per-peer FIFO holds across transports (the reference's
  // UCX sid-reordering guarantee, ucx_van.h:1217-1257).
  uint64_t seq = 0;
  // per-data-blob types (parallel to Message::data)
  std::vector<int> data_type;
  // device ordinal the vals blob lives on at the SENDER (-1 host);
  // dst_dev = device the receiver should place it on
  int src_dev = kCPU;
  int dst_dev = kCPU;
  Control control;

  std::string DebugString() const {
    std::ostringstream os;
    os << "meta[app=" << app_id << " cust=" << customer_id << " ts=" << timestamp
       << " " << sender << "->" << recver
       << (request ? " req" : " res") << (push ? " push" : "") << (pull ? " pull" : "")
       << " key=" << key << " val_len=" << val_len;
    if (!control.empty()) os << " ctrl{" << control.DebugString() << "}";
    os << "]";
    return os.str();
  }
};

struct Message {
  Meta meta;
  std::vector<SArray<char>> data;  // [keys, vals, lens]

  template <typename V>
  void AddData(const SArray<V>& v) {
    XPS_CHECK_LT(data.size(), 4u);
    meta.data_type.push_back(DataTypeOf<V>());
    data.push_back(SArray<char>::View(v));
  }

  template <typename V>
  static int DataTypeOf();

  std::string DebugString() const {
    std::ostringstream os;
    os << meta.DebugString();
    if (!data.empty()) {
      os << " data:[";
      for (auto& d : data) os << d.size() << (d.on_device() ? "d" : "h") << ",";
      os << "]";
    }
    return os.str();
  }
};

template <> inline int Message::DataTypeOf<char>() { return kChar; }
template <> inline int Message::DataTypeOf<uint8_t>() { return kUint8; }
template <> inline int Message::DataTypeOf<int>() { return kInt32; }
template <> inline int Message::DataTypeOf<int64_t>() { return kInt64; }
template <> inline int Message::DataTypeOf<uint64_t>() { return kUint64; }
template <> inline int Message::DataTypeOf<float>() { return kFloat32; }
template <> inline int Message::DataTypeOf<double>() { return kFloat64; }

}  // namespace xps
