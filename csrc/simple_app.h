// SimpleApp: (head:int, body:string) request/response RPC over the stack.
//
// Reference parity: ps-lite include/ps/simple_app.h (Request :133,
// Response :153, Process :170).
#pragma once

#include <functional>
#include <memory>
#include <string>

#include "customer.h"
#include "postoffice.h"
#include "van.h"

namespace xps {

struct SimpleData {
  int head = 0;
  std::string body;
  int timestamp = -1;
  int sender = kEmptyNodeID;
  int customer_id = 0;
  bool request = false;
};

class SimpleApp {
 public:
  using Handle = std::function<void(const SimpleData&, SimpleApp*)>;

  // app_id must match the peer SimpleApp's id
  SimpleApp(int app_id, int customer_id, Postoffice* po) : po_(po) {
    request_handle_ = [](const SimpleData& d, SimpleApp* app) { app->Response(d); };
    response_handle_ = [](const SimpleData&, SimpleApp*) {};
    obj_.reset(new Customer(app_id, customer_id,
                            [this](const Message& m) { Process(m); }, po));
  }
  virtual ~SimpleApp() = default;

  void set_request_handle(Handle h) { request_handle_ = std::move(h); }
  void set_response_handle(Handle h) { response_handle_ = std::move(h); }

  // send a request to node id or group mask; returns timestamp
  int Request(int head, const std::string& body, int recver) {
    int ts = obj_->NewRequest(recver);
    Message msg;
    msg.meta.app_id = obj_->app_id();
    msg.meta.customer_id = obj_->customer_id();
    msg.meta.request = true;
    msg.meta.simple_app = true;
    msg.meta.timestamp = ts;
    msg.meta.head = head;
    msg.meta.body = body;
    msg.meta.recver = recver;
    po_->van()->Send(msg);
    return ts;
  }

  void Response(const SimpleData& req, const std::string& body = "") {
    Message msg;
    msg.meta.app_id = obj_->app_id();
    msg.meta.customer_id = req.customer_id;
    msg.meta.request = false;
    msg.meta.simple_app = true;
    msg.meta.timestamp = req.timestamp;
    msg.meta.head = req.head;
    msg.meta.body = body;
    msg.meta.recver = req.sender;
    po_->van()->Send(msg);
  }

  void Wait(int timestamp) { obj_->WaitRequest(timestamp); }

  Postoffice* postoffice() const { return po_; }

 protected:
  // subclass (KV apps) constructor: defer customer creation to subclass
  struct DeferCustomer {};
  SimpleApp(Postoffice* po, DeferCustomer) : po_(po) {}

  virtual void Process(const Message& msg) {
    SimpleData d;
    d.head = msg.meta.head;
    d.body = msg.meta.body;
    d.timestamp = msg.meta.timestamp;
    d.sender = msg.meta.sender;
    d.customer_id = msg.meta.customer_id;
    d.request = msg.meta.request;
    if (msg.meta.request) {
      request_handle_(d, this);
    } else {
      response_handle_(d, this);
    }
  }

  Postoffice* po_;
  std::unique_ptr<Customer> obj_;
  Handle request_handle_;
  Handle response_handle_;
};

}  // namespace xps
