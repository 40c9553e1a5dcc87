// gpushare_amd._devlist — pre-encoded ListAndWatchResponse payloads.
//
// On an 8×MI355X node at GiB granularity the plugin advertises
// 8 × 288 = 2,304 fake devices; every ListAndWatch send (initial, and on any
// health flip) re-transmits the whole list (kubelet contract — reference
// streams the full list too, server.go:172-184).  The reference re-marshals
// the Go slice every send; here the per-device protobuf segments are encoded
// once at startup (both health variants) and a send is assembled with plain
// memcpy, so the steady-state all-healthy payload is a single cached buffer.
//
// Wire format encoded (kubelet device-plugin v1beta1, api.proto:72-88):
//   ListAndWatchResponse{ repeated Device devices = 1 }
//   Device{ string ID = 1; string health = 2; TopologyInfo topology = 3 }
//   TopologyInfo{ repeated NUMANode nodes = 1 };  NUMANode{ int64 ID = 1 }
// (topology is the modern upstream field — NUMA hints for the kubelet
// TopologyManager; emitted only when a per-device numa list is given,
// and skipped as an unknown field by pre-1.17 kubelets)

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <cstdint>
#include <cstring>
#include <stdexcept>
#include <string>
#include <vector>

namespace py = pybind11;

namespace {

void put_varint(std::string &out, uint64_t v) {
  while (v >= 0x80) {
    out.push_back(static_cast<char>((v & 0x7f) | 0x80));
    v >>= 7;
  }
  out.push_back(static_cast<char>(v));
}

// Encode one Device submessage wrapped as field 1 of ListAndWatchResponse.
// numa < 0 omits the topology field entirely.
std::string encode_device(const std::string &id, const std::string &health,
                          int64_t numa) {
  std::string dev;
  dev.push_back('\x0a');  // Device.ID, wire type 2
  put_varint(dev, id.size());
  dev += id;
  dev.push_back('\x12');  // Device.health, wire type 2
  put_varint(dev, health.size());
  dev += health;
  if (numa >= 0) {
    std::string numanode;
    numanode.push_back('\x08');  // NUMANode.ID, varint
    put_varint(numanode, static_cast<uint64_t>(numa));
    std::string topo;
    topo.push_back('\x0a');  // TopologyInfo.nodes, wire type 2
    put_varint(topo, numanode.size());
    topo += numanode;
    dev.push_back('\x1a');  // Device.topology, wire type 2
    put_varint(dev, topo.size());
    dev += topo;
  }

  std::string out;
  out.push_back('\x0a');  // ListAndWatchResponse.devices, wire type 2
  put_varint(out, dev.size());
  out += dev;
  return out;
}

class DeviceListCodec {
 public:
  explicit DeviceListCodec(const std::vector<std::string> &ids,
                           const std::vector<int64_t> &numa = {}) {
    if (!numa.empty() && numa.size() != ids.size())
      throw std::invalid_argument("numa list length != ids length");
    healthy_.reserve(ids.size());
    unhealthy_.reserve(ids.size());
    size_t total = 0;
    for (size_t i = 0; i < ids.size(); ++i) {
      const auto &id = ids[i];
      if (id.size() > 63)
        throw std::invalid_argument("Device.ID exceeds 63 chars: " + id);
      int64_t n = numa.empty() ? -1 : numa[i];
      healthy_.push_back(encode_device(id, "Healthy", n));
      unhealthy_.push_back(encode_device(id, "Unhealthy", n));
      total += healthy_.back().size();
    }
    all_healthy_.reserve(total);
    for (const auto &seg : healthy_) all_healthy_ += seg;
  }

  size_t size() const { return healthy_.size(); }

  // unhealthy: sorted-or-not list of fake-device indices currently Unhealthy.
  py::bytes encode(const std::vector<size_t> &unhealthy) const {
    if (unhealthy.empty())
      return py::bytes(all_healthy_);  // steady state: cached buffer
    std::vector<bool> bad(healthy_.size(), false);
    for (size_t i : unhealthy) {
      if (i >= healthy_.size())
        throw std::out_of_range("device index out of range");
      bad[i] = true;
    }
    std::string out;
    out.reserve(all_healthy_.size() + unhealthy.size() * 2);
    for (size_t i = 0; i < healthy_.size(); ++i)
      out += bad[i] ? unhealthy_[i] : healthy_[i];
    return py::bytes(out);
  }

 private:
  std::vector<std::string> healthy_, unhealthy_;
  std::string all_healthy_;
};

}  // namespace

PYBIND11_MODULE(_devlist, m) {
  m.doc() = "ListAndWatchResponse wire-format pre-encoder";
  py::class_<DeviceListCodec>(m, "DeviceListCodec")
      .def(py::init<const std::vector<std::string> &,
                    const std::vector<int64_t> &>(),
           py::arg("ids"), py::arg("numa") = std::vector<int64_t>{})
      .def("__len__", &DeviceListCodec::size)
      .def("encode", &DeviceListCodec::encode,
           py::arg("unhealthy") = std::vector<size_t>{});
}
