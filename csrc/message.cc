#include "message.h"

#include <cstring>
#include <stdexcept>

namespace hvd {

// --- primitive encoders ----------------------------------------------------
namespace {

template <typename T>
void put(std::string& out, T v) {
  out.append(reinterpret_cast<const char*>(&v), sizeof(T));
}

template <typename T>
T get(const char*& p, const char* end) {
  if (p + sizeof(T) > end) throw std::runtime_error("message underflow");
  T v;
  std::memcpy(&v, p, sizeof(T));
  p += sizeof(T);
  return v;
}

void put_str(std::string& out, const std::string& s) {
  put<uint32_t>(out, (uint32_t)s.size());
  out.append(s);
}

std::string get_str(const char*& p, const char* end) {
  uint32_t n = get<uint32_t>(p, end);
  if (p + n > end) throw std::runtime_error("message underflow (str)");
  std::string s(p, p + n);
  p += n;
  return s;
}

void put_i64vec(std::string& out, const std::vector<int64_t>& v) {
  put<uint32_t>(out, (uint32_t)v.size());
  for (auto x : v) put<int64_t>(out, x);
}

std::vector<int64_t> get_i64vec(const char*& p, const char* end) {
  uint32_t n = get<uint32_t>(p, end);
  std::vector<int64_t> v(n);
  for (uint32_t i = 0; i < n; ++i) v[i] = get<int64_t>(p, end);
  return v;
}

}  // namespace

// --- dtype helpers ---------------------------------------------------------

DataType DataTypeFromTorch(at::ScalarType t) {
  switch (t) {
    case at::kByte: return DataType::HVD_UINT8;
    case at::kChar: return DataType::HVD_INT8;
    case at::kInt: return DataType::HVD_INT32;
    case at::kLong: return DataType::HVD_INT64;
    case at::kHalf: return DataType::HVD_FLOAT16;
    case at::kFloat: return DataType::HVD_FLOAT32;
    case at::kDouble: return DataType::HVD_FLOAT64;
    case at::kBool: return DataType::HVD_BOOL;
    case at::kBFloat16: return DataType::HVD_BFLOAT16;
    case at::kUInt16: return DataType::HVD_UINT16;
    case at::kShort: return DataType::HVD_INT16;
    default:
      throw std::runtime_error(std::string("horovod_amd: unsupported tensor dtype ") +
                               std::string(at::toString(t)));
  }
}

at::ScalarType DataTypeToTorch(DataType t) {
  switch (t) {
    case DataType::HVD_UINT8: return at::kByte;
    case DataType::HVD_INT8: return at::kChar;
    case DataType::HVD_INT32: return at::kInt;
    case DataType::HVD_INT64: return at::kLong;
    case DataType::HVD_FLOAT16: return at::kHalf;
    case DataType::HVD_FLOAT32: return at::kFloat;
    case DataType::HVD_FLOAT64: return at::kDouble;
    case DataType::HVD_BOOL: return at::kBool;
    case DataType::HVD_BFLOAT16: return at::kBFloat16;
    case DataType::HVD_UINT16: return at::kUInt16;
    case DataType::HVD_INT16: return at::kShort;
  }
  throw std::runtime_error("horovod_amd: bad DataType");
}

size_t DataTypeSize(DataType t) {
  switch (t) {
    case DataType::HVD_UINT8:
    case DataType::HVD_INT8:
    case DataType::HVD_BOOL: return 1;
    case DataType::HVD_FLOAT16:
    case DataType::HVD_BFLOAT16:
    case DataType::HVD_UINT16:
    case DataType::HVD_INT16: return 2;
    case DataType::HVD_INT32:
    case DataType::HVD_FLOAT32: return 4;
    case DataType::HVD_INT64:
    case DataType::HVD_FLOAT64: return 8;
  }
  return 0;
}

const char* DataTypeName(DataType t) {
  switch (t) {
    case DataType::HVD_UINT8: return "uint8";
    case DataType::HVD_INT8: return "int8";
    case DataType::HVD_INT32: return "int32";
    case DataType::HVD_INT64: return "int64";
    case DataType::HVD_FLOAT16: return "float16";
    case DataType::HVD_FLOAT32: return "float32";
    case DataType::HVD_FLOAT64: return "float64";
    case DataType::HVD_BOOL: return "bool";
    case DataType::HVD_BFLOAT16: return "bfloat16";
    case DataType::HVD_UINT16: return "uint16";
    case DataType::HVD_INT16: return "int16";
  }
  return "?";
}

// --- Request ---------------------------------------------------------------

void Request::Serialize(std::string& out) const {
  put<uint8_t>(out, (uint8_t)type);
  put<int32_t>(out, rank);
  put_str(out, name);
  put<uint8_t>(out, (uint8_t)dtype);
  put_i64vec(out, shape);
  put<int32_t>(out, root_rank);
  put<uint8_t>(out, (uint8_t)reduce_op);
  put<double>(out, prescale);
  put<double>(out, postscale);
  put<int32_t>(out, process_set_id);
  put<int32_t>(out, device);
  put_i64vec(out, splits);
  put_str(out, group_key);
  put<int32_t>(out, group_size);
}

Request Request::Deserialize(const char*& p, const char* end) {
  Request r;
  r.type = (RequestType)get<uint8_t>(p, end);
  r.rank = get<int32_t>(p, end);
  r.name = get_str(p, end);
  r.dtype = (DataType)get<uint8_t>(p, end);
  r.shape = get_i64vec(p, end);
  r.root_rank = get<int32_t>(p, end);
  r.reduce_op = (ReduceOp)get<uint8_t>(p, end);
  r.prescale = get<double>(p, end);
  r.postscale = get<double>(p, end);
  r.process_set_id = get<int32_t>(p, end);
  r.device = get<int32_t>(p, end);
  r.splits = get_i64vec(p, end);
  r.group_key = get_str(p, end);
  r.group_size = get<int32_t>(p, end);
  return r;
}

void RequestList::Serialize(std::string& out) const {
  put<uint8_t>(out, shutdown ? 1 : 0);
  put<uint32_t>(out, (uint32_t)requests.size());
  for (auto& r : requests) r.Serialize(out);
}

RequestList RequestList::Deserialize(const char* p, const char* end) {
  RequestList l;
  l.shutdown = get<uint8_t>(p, end) != 0;
  uint32_t n = get<uint32_t>(p, end);
  l.requests.reserve(n);
  for (uint32_t i = 0; i < n; ++i) l.requests.push_back(Request::Deserialize(p, end));
  return l;
}

// --- Response --------------------------------------------------------------

void Response::Serialize(std::string& out) const {
  put<uint8_t>(out, (uint8_t)type);
  put<uint32_t>(out, (uint32_t)names.size());
  for (auto& n : names) put_str(out, n);
  put<uint8_t>(out, (uint8_t)dtype);
  put<uint8_t>(out, (uint8_t)reduce_op);
  put<int32_t>(out, process_set_id);
  put<int32_t>(out, device);
  put_str(out, error_msg);
  put_i64vec(out, tensor_sizes);
  put<int32_t>(out, root_rank);
  put<int32_t>(out, last_joined_rank);
  put_i64vec(out, tensor_shapes);
  put_str(out, group_key);
  put<int32_t>(out, group_size);
}

Response Response::Deserialize(const char*& p, const char* end) {
  Response r;
  r.type = (ResponseType)get<uint8_t>(p, end);
  uint32_t n = get<uint32_t>(p, end);
  r.names.reserve(n);
  for (uint32_t i = 0; i < n; ++i) r.names.push_back(get_str(p, end));
  r.dtype = (DataType)get<uint8_t>(p, end);
  r.reduce_op = (ReduceOp)get<uint8_t>(p, end);
  r.process_set_id = get<int32_t>(p, end);
  r.device = get<int32_t>(p, end);
  r.error_msg = get_str(p, end);
  r.tensor_sizes = get_i64vec(p, end);
  r.root_rank = get<int32_t>(p, end);
  r.last_joined_rank = get<int32_t>(p, end);
  r.tensor_shapes = get_i64vec(p, end);
  r.group_key = get_str(p, end);
  r.group_size = get<int32_t>(p, end);
  return r;
}

void ResponseList::Serialize(std::string& out) const {
  put<uint8_t>(out, shutdown ? 1 : 0);
  put<uint32_t>(out, (uint32_t)responses.size());
  for (auto& r : responses) r.Serialize(out);
}

ResponseList ResponseList::Deserialize(const char* p, const char* end) {
  ResponseList l;
  l.shutdown = get<uint8_t>(p, end) != 0;
  uint32_t n = get<uint32_t>(p, end);
  l.responses.reserve(n);
  for (uint32_t i = 0; i < n; ++i) l.responses.push_back(Response::Deserialize(p, end));
  return l;
}

}  // namespace hvd
