#include "wire.h"

namespace xps {

static const uint8_t kWireVersion = 2;

static void PackNode(const Node& n, ByteWriter* w) {
  w->I32(n.role);
  w->I32(n.id);
  w->Str(n.hostname);
  w->I32(n.port);
  w->I32(n.aux_id);
  w->I32(n.dev_id);
  w->U64(n.host_hash);
  w->I32(n.is_recovery);
  w->U64(n.pool_capacity);
  w->U64(n.pool_slab_bytes);
  w->I32(static_cast<int32_t>(n.pool_handles.size()));
  for (auto& h : n.pool_handles) w->Raw(h.data(), kIpcHandleBytes);
  w->U64(n.shm_uid);
  w->U64(n.host_pool_uid);
  w->U64(n.host_pool_capacity);
}

static void UnpackNode(ByteReader* r, Node* n) {
  n->role = r->I32();
  n->id = r->I32();
  n->hostname = r->Str();
  n->port = r->I32();
  n->aux_id = r->I32();
  n->dev_id = r->I32();
  n->host_hash = r->U64();
  n->is_recovery = r->I32();
  n->pool_capacity = r->U64();
  n->pool_slab_bytes = r->U64();
  int nh = r->I32();
  XPS_CHECK(nh >= 0 && nh <= 4096) << "wire: bad pool handle count " << nh;
  n->pool_handles.resize(nh);
  for (int i = 0; i < nh; ++i) r->Raw(n->pool_handles[i].data(), kIpcHandleBytes);
  n->shm_uid = r->U64();
  n->host_pool_uid = r->U64();
  n->host_pool_capacity = r->U64();
}

void PackMeta(const Meta& m, std::string* out) {
  out->clear();
  ByteWriter w(out);
  w.U8(kWireVersion);
  w.I32(m.app_id);
  w.I32(m.customer_id);
  w.I32(m.timestamp);
  w.I32(m.sender);
  w.I32(m.recver);
  uint8_t flags = (m.request ? 1 : 0) | (m.push ? 2 : 0) | (m.pull ? 4 : 0) |
                  (m.simple_app ? 8 : 0);
  w.U8(flags);
  w.I32(m.head);
  w.Str(m.body);
  w.U64(m.key);
  w.U64(m.addr);
  w.I64(m.val_len);
  w.I32(m.option);
  w.U64(m.msg_sig);
  w.U64(m.seq);
  w.I32(m.src_dev);
  w.I32(m.dst_dev);
  w.U8(static_cast<uint8_t>(m.data_type.size()));
  for (int t : m.data_type) w.U8(static_cast<uint8_t>(t));
  // control
  w.U8(static_cast<uint8_t>(m.control.cmd));
  if (!m.control.empty()) {
    w.I32(m.control.barrier_group);
    w.I32(m.control.barrier_token);
    w.U8(static_cast<uint8_t>(m.control.instance_barrier));
    w.U64(m.control.msg_sig);
    w.I32(static_cast<int32_t>(m.control.node.size()));
    for (auto& n : m.control.node) PackNode(n, &w);
  }
}

void UnpackMeta(const char* buf, size_t len, Meta* m) {
  ByteReader r(buf, len);
  uint8_t ver = r.U8();
  XPS_CHECK_EQ(static_cast<int>(ver), static_cast<int>(kWireVersion)) << "wire version mismatch";
  m->app_id = r.I32();
  m->customer_id = r.I32();
  m->timestamp = r.I32();
  m->sender = r.I32();
  m->recver = r.I32();
  uint8_t flags = r.U8();
  m->request = flags & 1;
  m->push = flags & 2;
  m->pull = flags & 4;
  m->simple_app = flags & 8;
  m->head = r.I32();
  m->body = r.Str();
  m->key = r.U64();
  m->addr = r.U64();
  m->val_len = r.I64();
  m->option = r.I32();
  m->msg_sig = r.U64();
  m->seq = r.U64();
  m->src_dev = r.I32();
  m->dst_dev = r.I32();
  int nt = r.U8();
  m->data_type.clear();
  for (int i = 0; i < nt; ++i) m->data_type.push_back(r.U8());
  m->control.cmd = r.U8();
  if (!m->control.empty()) {
    m->control.barrier_group = r.I32();
    m->control.barrier_token = r.I32();
    m->control.instance_barrier = r.U8();
    m->control.msg_sig = r.U64();
    int nn = r.I32();
    XPS_CHECK(nn >= 0 && nn <= 65536) << "wire: bad node count " << nn;
    m->control.node.resize(nn);
    for (int i = 0; i < nn; ++i) UnpackNode(&r, &m->control.node[i]);
  }
}

}  // namespace xps
