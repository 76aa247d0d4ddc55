// Python bindings for the xps core (package ps_lite_amd).
#include <pybind11/functional.h>
#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <atomic>
#include <thread>

#include "gpu_plane.h"
#include "hip_pool.h"
#include "hip_util.h"
#include "host_pool.h"
#include "kernels.h"
#include "kv_app.h"
#include "kv_utils.h"
#include "ps.h"
#include "server_handlers.h"
#include "shm_ring.h"
#include "simple_app.h"
#include "van.h"
#include "wire.h"

namespace py = pybind11;
using namespace xps;

namespace {

struct PoolBuffer {
  SArray<char> arr;
  uintptr_t ptr() const { return reinterpret_cast<uintptr_t>(arr.data()); }
  size_t nbytes() const { return arr.size(); }
  int device() const { return arr.device(); }
  void copy_from(py::array src) {
    py::buffer_info info = src.request();
    size_t n = static_cast<size_t>(info.size) * info.itemsize;
    XPS_CHECK_LE(n, arr.size());
    if (arr.on_device()) {
      gpu::CopyHostToDevice(arr.data(), info.ptr, n, arr.device());
    } else {
      memcpy(arr.data(), info.ptr, n);
    }
  }
  py::array_t<float> to_numpy_f32() const {
    size_t n = arr.size() / sizeof(float);
    py::array_t<float> out(n);
    if (arr.on_device()) {
      gpu::CopyDeviceToHost(out.mutable_data(), arr.data(), n * sizeof(float));
    } else {
      memcpy(out.mutable_data(), arr.data(), n * sizeof(float));
    }
    return out;
  }
};

class PyKVWorker {
 public:
  PyKVWorker(int app_id, int customer_id, int instance_idx)
      : w_(app_id, customer_id, instance_idx) {}

  int Push(py::array_t<uint64_t> keys, py::array_t<float> vals, py::array_t<int> lens) {
    // copying variant (safe w.r.t. numpy lifetimes)
    SArray<Key> k;
    k.CopyFrom(keys.data(), keys.size());
    SArray<float> v;
    v.CopyFrom(vals.data(), vals.size());
    SArray<int> l;
    if (lens.size()) l.CopyFrom(lens.data(), lens.size());
    py::gil_scoped_release rel;
    return w_.ZPush(k, v, l);
  }

  py::array_t<float> PullBlocking(py::array_t<uint64_t> keys) {
    std::vector<Key> k(keys.data(), keys.data() + keys.size());
    std::vector<float> vals;
    std::vector<int> lens;
    {
      py::gil_scoped_release rel;
      int ts = w_.Pull(k, &vals, &lens);
      w_.Wait(ts);
    }
    py::array_t<float> out(vals.size());
    std::copy(vals.begin(), vals.end(), out.mutable_data());
    return out;
  }

  int ZPushPtr(py::array_t<uint64_t> keys, uintptr_t vals_ptr, size_t vals_bytes, int device,
               py::array_t<int> lens, int cmd, uintptr_t keys_dev_ptr) {
    SArray<Key> k;
    k.CopyFrom(keys.data(), keys.size());
    SArray<float> v(reinterpret_cast<float*>(vals_ptr), vals_bytes / sizeof(float), device);
    SArray<int> l;
    if (lens.size()) l.CopyFrom(lens.data(), lens.size());
    SArray<Key> kd;
    if (keys_dev_ptr) {
      kd = SArray<Key>(reinterpret_cast<Key*>(keys_dev_ptr), keys.size(), device);
    }
    py::gil_scoped_release rel;
    return w_.ZPush(k, v, l, cmd, nullptr, kd);
  }

  // fused push+pull (ZPushPull): push vals_ptr, receive the post-push
  // values into out_ptr — one round trip (sparse lookup+update)
  int ZPushPullPtr(py::array_t<uint64_t> keys, uintptr_t vals_ptr, uintptr_t out_ptr,
                   size_t vals_bytes, int device, py::array_t<int> lens, int cmd,
                   uintptr_t keys_dev_ptr) {
    SArray<Key> k;
    k.CopyFrom(keys.data(), keys.size());
    SArray<float> v(reinterpret_cast<float*>(vals_ptr), vals_bytes / sizeof(float), device);
    auto* out = new SArray<float>(reinterpret_cast<float*>(out_ptr), vals_bytes / sizeof(float),
                                  device);
    SArray<int> l;
    if (lens.size()) l.CopyFrom(lens.data(), lens.size());
    SArray<Key> kd;
    if (keys_dev_ptr) {
      kd = SArray<Key>(reinterpret_cast<Key*>(keys_dev_ptr), keys.size(), device);
    }
    py::gil_scoped_release rel;
    return w_.ZPushPull(k, v, out, l, cmd, [out]() { delete out; }, kd);
  }

  int ZPullPtr(py::array_t<uint64_t> keys, uintptr_t vals_ptr, size_t vals_bytes, int device,
               py::array_t<int> lens, int cmd, uintptr_t keys_dev_ptr) {
    SArray<Key> k;
    k.CopyFrom(keys.data(), keys.size());
    auto* v = new SArray<float>(reinterpret_cast<float*>(vals_ptr), vals_bytes / sizeof(float),
                                device);
    SArray<int>* l = nullptr;
    if (lens.size()) {
      l = new SArray<int>();
      l->CopyFrom(lens.data(), lens.size());
    }
    SArray<Key> kd;
    if (keys_dev_ptr) {
      kd = SArray<Key>(reinterpret_cast<Key*>(keys_dev_ptr), keys.size(), device);
    }
    py::gil_scoped_release rel;
    // delete the temporaries when the pull completes
    return w_.ZPull(k, v, l, cmd, [v, l]() {
      delete v;
      delete l;
    }, kd);
  }

  void Wait(int ts) {
    py::gil_scoped_release rel;
    w_.Wait(ts);
  }

  // rn50/mixed-size fast path: per key, push (and optionally the pull
  // right behind it — per-peer ordering makes that safe), then wait all
  void RoundMixed(py::array_t<uint64_t> keys, const std::vector<uintptr_t>& push_ptrs,
                  const std::vector<uintptr_t>& pull_ptrs, const std::vector<size_t>& sizes,
                  int device, int cmd, bool overlap) {
    size_t nk = static_cast<size_t>(keys.size());
    XPS_CHECK_EQ(push_ptrs.size(), nk);
    std::vector<SArray<Key>> karrs(nk);
    for (size_t i = 0; i < nk; ++i) karrs[i] = SArray<Key>({keys.data()[i]});
    py::gil_scoped_release rel;
    std::vector<int> tss;
    tss.reserve(2 * nk);
    std::vector<std::unique_ptr<SArray<float>>> dsts;
    std::vector<int> pull_tss;
    auto issue_pull = [&](size_t i) {
      size_t n = sizes[i] / sizeof(float);
      dsts.emplace_back(new SArray<float>(reinterpret_cast<float*>(pull_ptrs[i]), n, device));
      return w_.ZPull(karrs[i], dsts.back().get(), nullptr, cmd);
    };
    for (size_t i = 0; i < nk; ++i) {
      size_t n = sizes[i] / sizeof(float);
      SArray<int> lens(1);
      lens[0] = static_cast<int>(n);
      SArray<float> v(reinterpret_cast<float*>(push_ptrs[i]), n, device);
      tss.push_back(w_.ZPush(karrs[i], v, lens, cmd));
      if (overlap && !pull_ptrs.empty()) tss.push_back(issue_pull(i));
    }
    if (!overlap && !pull_ptrs.empty()) {
      for (int ts : tss) w_.Wait(ts);
      tss.clear();
      for (size_t i = 0; i < nk; ++i) tss.push_back(issue_pull(i));
    }
    for (int ts : tss) w_.Wait(ts);
  }

  // benchmark fast path: issue one single-key message per key (push or
  // pull) and wait for all — the whole round runs in C++ (the reference
  // benchmark is C++; this keeps the comparison honest at small sizes)
  void Round(py::array_t<uint64_t> keys, const std::vector<uintptr_t>& ptrs, size_t nbytes,
             int device, int cmd, bool pull) {
    size_t nk = static_cast<size_t>(keys.size());
    XPS_CHECK_EQ(ptrs.size(), nk);
    size_t n = nbytes / sizeof(float);
    std::vector<SArray<Key>> karrs(nk);
    for (size_t i = 0; i < nk; ++i) karrs[i] = SArray<Key>({keys.data()[i]});
    SArray<int> lens(1);
    lens[0] = static_cast<int>(n);
    py::gil_scoped_release rel;
    std::vector<int> tss(nk);
    std::vector<std::unique_ptr<SArray<float>>> dsts;
    for (size_t i = 0; i < nk; ++i) {
      if (pull) {
        dsts.emplace_back(new SArray<float>(reinterpret_cast<float*>(ptrs[i]), n, device));
        tss[i] = w_.ZPull(karrs[i], dsts.back().get(), nullptr, cmd);
      } else {
        SArray<float> v(reinterpret_cast<float*>(ptrs[i]), n, device);
        tss[i] = w_.ZPush(karrs[i], v, lens, cmd);
      }
    }
    for (int ts : tss) w_.Wait(ts);
  }

 private:
  KVWorker<float> w_;
};

class PyKVServer {
 public:
  explicit PyKVServer(int app_id, int instance_idx) : s_(app_id, instance_idx) {}

  void SetDefaultHandle() {
    auto h = std::make_shared<KVServerDefaultHandle<float>>();
    default_ = h;
    s_.set_request_handle([h](const KVMeta& m, const KVPairs<float>& kvs, KVServer<float>* srv) {
      (*h)(m, kvs, srv);
    });
  }

  // CPU BytePS reduce rounds (same protocol as the GPU reduce handler)
  void SetReduceHandle(int num_workers) {
    auto h = std::make_shared<KVServerReduceHandle<float>>(num_workers);
    reduce_ = h;
    s_.set_request_handle([h](const KVMeta& m, const KVPairs<float>& kvs, KVServer<float>* srv) {
      (*h)(m, kvs, srv);
    });
  }

  // RegisterRecvBufferWithRank parity: the app's own device buffer
  // becomes the dense store entry for `key`
  void RegisterEntry(uint64_t key, uintptr_t ptr, size_t nbytes, int device) {
    XPS_CHECK(dense_) << "register_entry needs set_gpu_dense_handle first";
    dense_->RegisterEntry(key, reinterpret_cast<void*>(ptr), nbytes, device);
  }

  // checkpoint/resume of the installed handler's server state
  void SaveCheckpoint(const std::string& path) {
    py::gil_scoped_release rel;
    if (dense_) {
      dense_->Save(path);
    } else if (sparse_) {
      sparse_->Save(path);
    } else if (default_ || reduce_) {
      FILE* f = fopen(path.c_str(), "wb");
      XPS_CHECK(f) << "cannot open checkpoint " << path;
      uint64_t n = default_ ? default_->store.size() : reduce_->store.size();
      fwrite(&n, 8, 1, f);
      auto write_one = [f](uint64_t key, const float* data, uint64_t len) {
        fwrite(&key, 8, 1, f);
        fwrite(&len, 8, 1, f);
        fwrite(data, 1, len, f);
      };
      if (default_) {
        for (auto& kv : default_->store) {
          write_one(kv.first, kv.second.data(), kv.second.size() * sizeof(float));
        }
      } else {
        for (auto& kv : reduce_->store) {
          write_one(kv.first, kv.second.buf.data(), kv.second.buf.size() * sizeof(float));
        }
      }
      fclose(f);
    } else {
      XPS_LOG(Fatal) << "no checkpointable handler installed";
    }
  }

  void LoadCheckpoint(const std::string& path) {
    py::gil_scoped_release rel;
    if (dense_) {
      dense_->Load(path);
    } else if (sparse_) {
      sparse_->Load(path);
    } else if (default_ || reduce_) {
      FILE* f = fopen(path.c_str(), "rb");
      XPS_CHECK(f) << "cannot open checkpoint " << path;
      uint64_t n = 0;
      XPS_CHECK_EQ(fread(&n, 8, 1, f), 1u);
      for (uint64_t i = 0; i < n; ++i) {
        uint64_t key, len;
        XPS_CHECK_EQ(fread(&key, 8, 1, f), 1u);
        XPS_CHECK_EQ(fread(&len, 8, 1, f), 1u);
        auto& v = default_ ? default_->store[key]
                           : reduce_->store[key].buf;
        v.resize(len / sizeof(float));
        XPS_CHECK_EQ(fread(v.data(), 1, len, f), len);
      }
      fclose(f);
    } else {
      XPS_LOG(Fatal) << "no checkpointable handler installed";
    }
  }

  void SetGpuDenseHandle(const std::string& mode, const std::string& dtype) {
    DenseMode m = mode == "sum" ? DenseMode::kSum
                  : mode == "reduce" ? DenseMode::kReduce
                                     : DenseMode::kAssign;
    DenseDtype d = dtype == "bf16" ? DenseDtype::kBf16 : DenseDtype::kF32;
    auto h = std::make_shared<GpuDenseHandler>(s_.postoffice(), m, d);
    dense_ = h;
    s_.set_request_handle([h](const KVMeta& m2, const KVPairs<float>& kvs, KVServer<float>* srv) {
      (*h)(m2, kvs, srv);
    });
  }

  void SetGpuSparseHandle(size_t rows, size_t row_len, bool accumulate, int key_shift) {
    auto h = std::make_shared<GpuSparseHandler>(s_.postoffice(), rows, row_len, accumulate,
                                                key_shift);
    sparse_ = h;
    s_.set_request_handle([h](const KVMeta& m, const KVPairs<float>& kvs, KVServer<float>* srv) {
      (*h)(m, kvs, srv);
    });
  }

  uintptr_t SparseTablePtr() const { return sparse_ ? sparse_->table_ptr() : 0; }

  // fn(meta: dict, keys: ndarray[u64], vals: ndarray[f32]) -> ndarray[f32] | None
  void SetPythonHandle(py::function fn) {
    s_.set_request_handle([fn](const KVMeta& m, const KVPairs<float>& kvs, KVServer<float>* srv) {
      KVPairs<float> res;
      {
        py::gil_scoped_acquire gil;
        py::dict meta;
        meta["cmd"] = m.cmd;
        meta["push"] = m.push;
        meta["pull"] = m.pull;
        meta["sender"] = m.sender;
        meta["timestamp"] = m.timestamp;
        meta["key"] = m.key;
        meta["val_len"] = m.val_len;
        py::array_t<uint64_t> keys(kvs.keys.size());
        std::copy(kvs.keys.begin(), kvs.keys.end(), keys.mutable_data());
        py::array_t<float> vals(kvs.vals.size());
        if (kvs.vals.size()) {
          XPS_CHECK(!kvs.vals.on_device()) << "python handle got device vals";
          std::copy(kvs.vals.begin(), kvs.vals.end(), vals.mutable_data());
        }
        py::object out = fn(meta, keys, vals);
        if (m.pull) {
          XPS_CHECK(!out.is_none()) << "pull handler must return vals";
          auto arr = py::cast<py::array_t<float>>(out);
          res.keys = kvs.keys;
          res.vals.CopyFrom(arr.data(), arr.size());
          size_t n = kvs.keys.size();
          SArray<int> lens(n);
          for (size_t i = 0; i < n; ++i) lens[i] = static_cast<int>(arr.size() / n);
          res.lens = lens;
        }
      }
      srv->Response(m, res);
    });
  }

 private:
  KVServer<float> s_;
  std::shared_ptr<GpuSparseHandler> sparse_;
  std::shared_ptr<GpuDenseHandler> dense_;
  std::shared_ptr<KVServerDefaultHandle<float>> default_;
  std::shared_ptr<KVServerReduceHandle<float>> reduce_;
};

class PySimpleApp {
 public:
  // server-side customers register under app_id (the Van routes requests
  // to (app_id, app_id) on non-worker nodes — same convention as KVServer)
  PySimpleApp(const std::string& role, int app_id, int customer_id)
      : app_(app_id, role == "worker" ? customer_id : app_id, GetPO(role)) {}

  void SetRequestHandle(py::function fn) {
    app_.set_request_handle([fn](const SimpleData& d, SimpleApp* app) {
      std::string reply;
      {
        py::gil_scoped_acquire gil;
        py::object out = fn(d.head, py::bytes(d.body));
        if (!out.is_none()) reply = py::cast<std::string>(out);
      }
      app->Response(d, reply);
    });
  }

  void SetResponseHandle(py::function fn) {
    app_.set_response_handle([fn](const SimpleData& d, SimpleApp*) {
      py::gil_scoped_acquire gil;
      fn(d.head, py::bytes(d.body));
    });
  }

  int Request(int head, const std::string& body, int recver) {
    py::gil_scoped_release rel;
    return app_.Request(head, body, recver);
  }

  void Wait(int ts) {
    py::gil_scoped_release rel;
    app_.Wait(ts);
  }

 private:
  SimpleApp app_;
};

}  // namespace

PYBIND11_MODULE(_core, m) {
  m.doc() = "xps: MI355X-native parameter-server core";

  m.def("init_env", [](const std::map<std::string, std::string>& kv) {
    Environment::Get()->Init(kv);
  });
  m.def("env", [](const std::string& k) { return Environment::Get()->GetStr(k, ""); });
  m.def("start",
        [](int customer_id, const std::string& role, int rank, bool barrier, int device) {
          py::gil_scoped_release rel;
          Start(customer_id, role, rank, barrier, device);
        },
        py::arg("customer_id") = 0, py::arg("role") = "worker", py::arg("rank") = -1,
        py::arg("barrier") = true, py::arg("device") = -2);
  m.def("finalize",
        [](int customer_id, const std::string& role, bool barrier) {
          py::gil_scoped_release rel;
          Finalize(customer_id, role, barrier);
        },
        py::arg("customer_id") = 0, py::arg("role") = "worker", py::arg("barrier") = true);
  m.def("clear_registry", []() { Postoffice::ClearRegistry(); });
  m.def("barrier", [](const std::string& role, int group, int idx) {
    py::gil_scoped_release rel;
    GetPO(role, idx)->Barrier(0, group);
  }, py::arg("role"), py::arg("group"), py::arg("idx") = 0);
  m.def("num_workers", &NumWorkers);
  m.def("num_servers", &NumServers);
  m.def("my_rank", [](const std::string& role, int idx) { return GetPO(role, idx)->my_rank(); },
        py::arg("role"), py::arg("idx") = 0);
  m.def("node_id", [](const std::string& role, int idx) { return GetPO(role, idx)->node_id(); },
        py::arg("role"), py::arg("idx") = 0);
  m.def("send_bytes", [](const std::string& role) {
    auto* van = GetPO(role)->van();
    return van ? van->send_bytes_.load() : int64_t(0);
  });
  m.def("recv_bytes", [](const std::string& role) {
    auto* van = GetPO(role)->van();
    return van ? van->recv_bytes_.load() : int64_t(0);
  });
  m.def("gpu_count", []() { return gpu::DeviceCount(); });
  m.def("device_sync", [](int dev) {
    py::gil_scoped_release rel;
    gpu::DeviceSync(dev);
  }, py::arg("dev") = -1);

  m.attr("SCHEDULER_GROUP") = kScheduler;
  m.attr("SERVER_GROUP") = kServerGroup;
  m.attr("WORKER_GROUP") = kWorkerGroup;

  m.def("pool_init", [](int device, size_t capacity) { HbmPool::Get()->Init(device, capacity); },
        py::arg("device"), py::arg("capacity") = 0);
  m.def("pool_ipc_handle", [](size_t slab) {
    return py::bytes(HbmPool::Get()->slab_handle(slab), kIpcHandleBytes);
  }, py::arg("slab") = 0);
  m.def("make_stream_events", []() {
    // mimic the plane's device state: nonblocking streams + events
    py::gil_scoped_release rel;
    hipStream_t s;
    XPS_CHECK(hipStreamCreateWithFlags(&s, hipStreamNonBlocking) == hipSuccess);
    hipEvent_t ev;
    XPS_CHECK(hipEventCreateWithFlags(&ev, hipEventDisableTiming) == hipSuccess);
    XPS_CHECK(hipEventRecord(ev, s) == hipSuccess);
    while (hipEventQuery(ev) != hipSuccess) {}
    return reinterpret_cast<uintptr_t>(s);
  });
  m.def("kernel_on_stream", [](uintptr_t stream, uintptr_t dst, uintptr_t src, size_t n) {
    py::gil_scoped_release rel;
    kern::DenseSumF32(reinterpret_cast<float*>(dst), reinterpret_cast<float*>(src), n,
                      reinterpret_cast<hipStream_t>(stream));
    hipEvent_t ev;
    XPS_CHECK(hipEventCreateWithFlags(&ev, hipEventDisableTiming) == hipSuccess);
    XPS_CHECK(hipEventRecord(ev, reinterpret_cast<hipStream_t>(stream)) == hipSuccess);
    while (hipEventQuery(ev) != hipSuccess) {}
    (void)hipEventDestroy(ev);
  });
  m.def("ipc_open", [](py::bytes handle) {
    std::string h = handle;
    py::gil_scoped_release rel;
    hipIpcMemHandle_t hh;
    memcpy(&hh, h.data(), sizeof(hh));
    void* ptr = nullptr;
    hipError_t e = hipIpcOpenMemHandle(&ptr, hh, hipIpcMemLazyEnablePeerAccess);
    XPS_CHECK(e == hipSuccess) << "hipIpcOpenMemHandle: " << hipGetErrorString(e);
    return reinterpret_cast<uintptr_t>(ptr);
  });
  m.def("pool_in_use", []() { return HbmPool::Get()->bytes_in_use(); });
  // is this device pointer inside the zero-copy (bootstrap-exported)
  // pool window? (torch-allocator tensors usually are)
  m.def("pool_contains", [](uintptr_t p) {
    uint64_t off = 0;
    return HbmPool::Get()->OffsetOf(reinterpret_cast<const void*>(p), &off);
  });
  py::class_<PoolBuffer>(m, "PoolBuffer")
      .def_property_readonly("ptr", &PoolBuffer::ptr)
      .def_property_readonly("nbytes", &PoolBuffer::nbytes)
      .def_property_readonly("device", &PoolBuffer::device)
      .def("copy_from", &PoolBuffer::copy_from)
      .def("to_numpy_f32", &PoolBuffer::to_numpy_f32);
  m.def("pool_alloc", [](size_t nbytes) {
    PoolBuffer b;
    b.arr = HbmPool::Get()->AllocArray(nbytes);
    return b;
  });
  // host-shm pool buffer: same-host zero-copy for CPU payloads
  m.def("host_alloc", [](size_t nbytes) {
    PoolBuffer b;
    b.arr = HostShmPool::Get()->AllocArray(nbytes);
    return b;
  });

  py::class_<PyKVWorker>(m, "KVWorker")
      .def(py::init<int, int, int>(), py::arg("app_id") = 0, py::arg("customer_id") = 0,
           py::arg("instance_idx") = 0)
      .def("push", &PyKVWorker::Push, py::arg("keys"), py::arg("vals"),
           py::arg("lens") = py::array_t<int>())
      .def("pull", &PyKVWorker::PullBlocking, py::arg("keys"))
      .def("zpush_ptr", &PyKVWorker::ZPushPtr, py::arg("keys"), py::arg("vals_ptr"),
           py::arg("vals_bytes"), py::arg("device"), py::arg("lens") = py::array_t<int>(),
           py::arg("cmd") = 0, py::arg("keys_dev_ptr") = 0)
      .def("zpushpull_ptr", &PyKVWorker::ZPushPullPtr, py::arg("keys"), py::arg("vals_ptr"),
           py::arg("out_ptr"), py::arg("nbytes"), py::arg("device") = -1,
           py::arg("lens") = py::array_t<int>(), py::arg("cmd") = 0,
           py::arg("keys_dev_ptr") = 0)
      .def("zpull_ptr", &PyKVWorker::ZPullPtr, py::arg("keys"), py::arg("vals_ptr"),
           py::arg("vals_bytes"), py::arg("device"), py::arg("lens") = py::array_t<int>(),
           py::arg("cmd") = 0, py::arg("keys_dev_ptr") = 0)
      .def("wait", &PyKVWorker::Wait)
      .def("round", &PyKVWorker::Round, py::arg("keys"), py::arg("ptrs"), py::arg("nbytes"),
           py::arg("device"), py::arg("cmd") = 0, py::arg("pull") = false)
      .def("round_mixed", &PyKVWorker::RoundMixed, py::arg("keys"), py::arg("push_ptrs"),
           py::arg("pull_ptrs"), py::arg("sizes"), py::arg("device"), py::arg("cmd") = 0,
           py::arg("overlap") = true);

  py::class_<PyKVServer>(m, "KVServer")
      .def(py::init<int, int>(), py::arg("app_id") = 0, py::arg("instance_idx") = 0)
      .def("set_default_handle", &PyKVServer::SetDefaultHandle)
      .def("set_reduce_handle", &PyKVServer::SetReduceHandle, py::arg("num_workers"))
      .def("set_gpu_dense_handle", &PyKVServer::SetGpuDenseHandle, py::arg("mode") = "assign",
           py::arg("dtype") = "f32")
      .def("set_gpu_sparse_handle", &PyKVServer::SetGpuSparseHandle, py::arg("rows"),
           py::arg("row_len"), py::arg("accumulate") = true, py::arg("key_shift") = 0)
      .def("register_entry", &PyKVServer::RegisterEntry, py::arg("key"), py::arg("ptr"),
           py::arg("nbytes"), py::arg("device") = 0)
      .def("sparse_table_ptr", &PyKVServer::SparseTablePtr)
      .def("save_checkpoint", &PyKVServer::SaveCheckpoint)
      .def("load_checkpoint", &PyKVServer::LoadCheckpoint)
      .def("set_python_handle", &PyKVServer::SetPythonHandle);

  // wire-format roundtrip (unit-test hook)
  // transport-level zero-copy assertion (reference parity:
  // test_benchmark.cc:169-181 registered-buffer pointer equality)
  m.def("zero_copy_recv_count", []() { return g_zero_copy_recv.load(); });

  // per-peer plane traffic: {node_id: (tx_bytes, rx_bytes)} — each peer
  // pair rides its own xGMI link, so this is the per-link utilization
  // report of SURVEY §5.8 (bench.py emits it for the scaling runs)
  m.def("plane_peer_bytes", [](const std::string& role, int idx) {
    py::dict out;
    auto* po = GetPO(role, idx);
    auto* plane = po->van() ? dynamic_cast<GpuPlane*>(po->van()->plane()) : nullptr;
    if (plane) {
      for (auto& t : plane->PeerBytes()) {
        out[py::int_(std::get<0>(t))] = py::make_tuple(std::get<1>(t), std::get<2>(t));
      }
    }
    return out;
  }, py::arg("role") = "worker", py::arg("idx") = 0);

  // rank-ordering policy hook (tested in test_utils.py): takes
  // (role, host, port) tuples, returns them in rank-assignment order
  m.def("_order_nodes", [](std::vector<std::tuple<std::string, std::string, int>> in) {
    std::vector<Node> nodes;
    for (auto& t : in) {
      Node n;
      n.role = std::get<0>(t) == "worker" ? Node::WORKER : Node::SERVER;
      n.hostname = std::get<1>(t);
      n.port = std::get<2>(t);
      nodes.push_back(n);
    }
    auto less = NodeRankOrder(nodes);
    std::sort(nodes.begin(), nodes.end(), less);
    std::vector<std::tuple<std::string, std::string, int>> out;
    for (auto& n : nodes) {
      out.emplace_back(n.role == Node::WORKER ? "worker" : "server", n.hostname, n.port);
    }
    return out;
  });

  // stale-shm GC hooks (tested in test_utils.py)
  m.def("_host_pool_init_for_test", [](uint64_t uid, size_t bytes) {
    HostShmPool::Get()->Init(uid, bytes);
  });
  m.def("_gc_stale_shm", []() { HostShmPool::GcStaleSegments(); });

  // shm-ring stress hook: P producer threads push `per` tagged messages
  // each through one ring while a consumer drains; returns (received,
  // payload checksum ok). Deliberately overruns kSlots so the full-ring
  // CAS backoff path runs (the single consumer must never wedge).
  m.def("_ring_stress", [](int producers, int per, int payload) {
    ShmRing ring;
    XPS_CHECK(ring.Create(0xabcdef12345ull + getpid()));
    std::atomic<long> received{0}, sum{0}, pushed{0};
    std::atomic<bool> done{false};
    std::thread consumer([&]() {
      std::vector<char> buf(ShmRing::MaxPayload());
      while (true) {
        uint32_t n = ring.Pop(buf.data());
        if (n == 0) {
          if (done.load() && received.load() >= pushed.load()) break;
          std::this_thread::yield();
          continue;
        }
        long v = 0;
        for (uint32_t i = 0; i < n; ++i) v += static_cast<unsigned char>(buf[i]);
        sum.fetch_add(v);
        received.fetch_add(1);
      }
    });
    std::vector<std::thread> ps;
    std::atomic<long> expect{0};
    for (int p = 0; p < producers; ++p) {
      ps.emplace_back([&, p]() {
        std::vector<char> msg(payload);
        for (int i = 0; i < per; ++i) {
          for (int b = 0; b < payload; ++b) msg[b] = static_cast<char>((p + i + b) & 0x7f);
          long v = 0;
          for (int b = 0; b < payload; ++b) v += static_cast<unsigned char>(msg[b]);
          if (ring.Push(msg.data(), payload)) {
            expect.fetch_add(v);
            pushed.fetch_add(1);
          }
        }
      });
    }
    for (auto& t : ps) t.join();
    done.store(true);
    consumer.join();
    ring.CloseAndUnlink();
    return py::make_tuple(static_cast<long>(received.load()),
                          static_cast<long>(pushed.load()),
                          sum.load() == expect.load());
  });

  m.def("_test_meta_roundtrip", []() {
    Meta m;
    m.app_id = 7;
    m.customer_id = 3;
    m.timestamp = 42;
    m.sender = 9;
    m.recver = 8;
    m.request = true;
    m.push = true;
    m.head = 2;
    m.body = "hello\x00world";
    m.key = 0xDEADBEEFCAFEull;
    m.addr = 1234567;
    m.val_len = 1 << 20;
    m.option = kOptPullAddr | kOptHostAddr;
    m.seq = 987654321;
    m.data_type = {kUint64, kFloat32, kInt32};
    m.control.cmd = Control::ADD_NODE;
    Node n;
    n.role = Node::WORKER;
    n.id = 11;
    n.hostname = "10.0.0.5";
    n.port = 1234;
    n.shm_uid = 0x1122334455667788ull;
    n.host_pool_uid = 77;
    n.host_pool_capacity = 1 << 30;
    n.pool_capacity = 2ull << 30;
    n.pool_slab_bytes = 1ull << 30;
    n.pool_handles.resize(2);
    n.pool_handles[0][0] = 'x';
    n.pool_handles[1][63] = 'y';
    m.control.node.push_back(n);
    std::string buf;
    PackMeta(m, &buf);
    Meta out;
    UnpackMeta(buf.data(), buf.size(), &out);
    bool ok = out.app_id == m.app_id && out.customer_id == m.customer_id &&
              out.timestamp == m.timestamp && out.sender == m.sender &&
              out.recver == m.recver && out.request == m.request && out.push == m.push &&
              out.head == m.head && out.body == m.body && out.key == m.key &&
              out.addr == m.addr && out.val_len == m.val_len && out.option == m.option &&
              out.seq == m.seq &&
              out.data_type == m.data_type && out.control.cmd == m.control.cmd &&
              out.control.node.size() == 1;
    auto& on = out.control.node[0];
    ok = ok && on.role == n.role && on.id == n.id && on.hostname == n.hostname &&
         on.port == n.port && on.shm_uid == n.shm_uid && on.host_pool_uid == n.host_pool_uid &&
         on.host_pool_capacity == n.host_pool_capacity && on.pool_capacity == n.pool_capacity &&
         on.pool_slab_bytes == n.pool_slab_bytes && on.pool_handles.size() == 2 &&
         on.pool_handles[0][0] == 'x' && on.pool_handles[1][63] == 'y';
    return ok;
  });

  // raw wire access for robustness tests: pack a representative meta,
  // and unpack arbitrary bytes (XPS_CHECK-aborts on malformed input, so
  // callers exercise it from a subprocess)
  m.def("_pack_meta_sample", []() {
    Meta m;
    m.app_id = 1;
    m.body = "abc";
    m.control.cmd = Control::ADD_NODE;
    Node n;
    n.hostname = "127.0.0.1";
    n.pool_handles.resize(1);
    m.control.node.push_back(n);
    std::string buf;
    PackMeta(m, &buf);
    return py::bytes(buf);
  });
  // property-test hook: build a Meta from fields, round-trip through the
  // wire, compare (hypothesis drives the field values)
  m.def("_meta_roundtrip_fields",
        [](int app_id, int customer_id, int timestamp, int sender, int recver, bool request,
           bool push, bool pull, bool simple_app, int head, const std::string& body, uint64_t key,
           uint64_t addr, int64_t val_len, int option, uint64_t msg_sig, uint64_t seq,
           const std::string& hostname, int port, uint64_t shm_uid, int nhandles) {
          Meta m;
          m.app_id = app_id;
          m.customer_id = customer_id;
          m.timestamp = timestamp;
          m.sender = sender;
          m.recver = recver;
          m.request = request;
          m.push = push;
          m.pull = pull;
          m.simple_app = simple_app;
          m.head = head;
          m.body = body;
          m.key = key;
          m.addr = addr;
          m.val_len = val_len;
          m.option = option;
          m.msg_sig = msg_sig;
          m.seq = seq;
          m.control.cmd = Control::ADD_NODE;
          Node n;
          n.hostname = hostname;
          n.port = port;
          n.shm_uid = shm_uid;
          n.pool_handles.resize(nhandles);
          m.control.node.push_back(n);
          std::string buf;
          PackMeta(m, &buf);
          Meta o;
          UnpackMeta(buf.data(), buf.size(), &o);
          auto& on = o.control.node.at(0);
          return o.app_id == m.app_id && o.customer_id == m.customer_id &&
                 o.timestamp == m.timestamp && o.sender == m.sender && o.recver == m.recver &&
                 o.request == m.request && o.push == m.push && o.pull == m.pull &&
                 o.simple_app == m.simple_app && o.head == m.head && o.body == m.body &&
                 o.key == m.key && o.addr == m.addr && o.val_len == m.val_len &&
                 o.option == m.option && o.msg_sig == m.msg_sig && o.seq == m.seq &&
                 on.hostname == n.hostname && on.port == n.port && on.shm_uid == n.shm_uid &&
                 on.pool_handles.size() == static_cast<size_t>(nhandles);
        });

  m.def("_unpack_meta_raw", [](py::bytes b) {
    std::string s = b;
    Meta out;
    UnpackMeta(s.data(), s.size(), &out);
    return true;
  });

  // utility parity (ps-lite parallel_kv_match.h / parallel_sort.h)
  m.def("parallel_ordered_match",
        [](py::array_t<uint64_t> src_keys, py::array_t<float> src_vals,
           py::array_t<uint64_t> dst_keys, size_t k, bool accumulate, int nthreads) {
          SArray<Key> sk;
          sk.CopyFrom(src_keys.data(), src_keys.size());
          SArray<float> sv;
          sv.CopyFrom(src_vals.data(), src_vals.size());
          SArray<Key> dk;
          dk.CopyFrom(dst_keys.data(), dst_keys.size());
          SArray<float> dv;
          size_t n = ParallelOrderedMatch(sk, sv, dk, &dv, k,
                                          accumulate ? AssignOp::kPlus : AssignOp::kAssign,
                                          nthreads);
          py::array_t<float> out(dv.size());
          std::copy(dv.begin(), dv.end(), out.mutable_data());
          return py::make_tuple(n, out);
        },
        py::arg("src_keys"), py::arg("src_vals"), py::arg("dst_keys"), py::arg("k") = 1,
        py::arg("accumulate") = false, py::arg("nthreads") = 4);
  m.def("parallel_sort", [](py::array_t<uint64_t> keys, int nthreads) {
    SArray<Key> k;
    k.CopyFrom(keys.data(), keys.size());
    ParallelSort(&k, nthreads);
    py::array_t<uint64_t> out(k.size());
    std::copy(k.begin(), k.end(), out.mutable_data());
    return out;
  }, py::arg("keys"), py::arg("nthreads") = 4);

  // raw kernel entry points (synchronous; for numerics tests)
  m.def("k_dense_assign", [](uintptr_t dst, uintptr_t src, size_t nbytes) {
    kern::DenseAssign(reinterpret_cast<void*>(dst), reinterpret_cast<void*>(src), nbytes, nullptr);
    gpu::DeviceSync(-1);
  });
  m.def("k_dense_sum_f32", [](uintptr_t dst, uintptr_t src, size_t n) {
    kern::DenseSumF32(reinterpret_cast<float*>(dst), reinterpret_cast<float*>(src), n, nullptr);
    gpu::DeviceSync(-1);
  });
  m.def("k_dense_sum_bf16", [](uintptr_t dst, uintptr_t src, size_t n) {
    kern::DenseSumBf16(reinterpret_cast<uint16_t*>(dst), reinterpret_cast<uint16_t*>(src), n,
                       nullptr);
    gpu::DeviceSync(-1);
  });
  m.def("k_sparse_gather_f32",
        [](uintptr_t table, uintptr_t rows, size_t nrows, size_t row_len, uintptr_t out) {
          kern::SparseGatherF32(reinterpret_cast<float*>(table),
                                reinterpret_cast<uint64_t*>(rows), nrows, row_len,
                                reinterpret_cast<float*>(out), nullptr);
          gpu::DeviceSync(-1);
        });
  m.def("k_sparse_scatter_add_f32",
        [](uintptr_t table, uintptr_t rows, size_t nrows, size_t row_len, uintptr_t src,
           bool atomic) {
          kern::SparseScatterAddF32(reinterpret_cast<float*>(table),
                                    reinterpret_cast<uint64_t*>(rows), nrows, row_len,
                                    reinterpret_cast<float*>(src), atomic, nullptr);
          gpu::DeviceSync(-1);
        });

  py::class_<PySimpleApp>(m, "SimpleApp")
      .def(py::init<const std::string&, int, int>(), py::arg("role"), py::arg("app_id") = 10,
           py::arg("customer_id") = 0)
      .def("set_request_handle", &PySimpleApp::SetRequestHandle)
      .def("set_response_handle", &PySimpleApp::SetResponseHandle)
      .def("request", &PySimpleApp::Request)
      .def("wait", &PySimpleApp::Wait);
}
