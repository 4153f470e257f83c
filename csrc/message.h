// Wire messages for the negotiation protocol.
//
// Re-design of the reference's Request/Response/flatbuffers layer
// (horovod/common/message.{h,cc}, common/wire/message.fbs).  We use a simple
// hand-rolled binary encoding instead of flatbuffers: the messages are small,
// interned per-cycle, and only ever cross a localhost/ethernet TCP socket.
#pragma once

#include <cstdint>
#include <string>
#include <vector>

#include "common.h"

namespace hvd {

enum class RequestType : uint8_t {
  ALLREDUCE = 0,
  ALLGATHER = 1,
  BROADCAST = 2,
  JOIN = 3,
  ADASUM = 4,
  ALLTOALL = 5,
  BARRIER = 6,
  REDUCESCATTER = 7,
};

enum class ResponseType : uint8_t {
  ALLREDUCE = 0,
  ALLGATHER = 1,
  BROADCAST = 2,
  JOIN = 3,
  ADASUM = 4,
  ALLTOALL = 5,
  BARRIER = 6,
  REDUCESCATTER = 7,
  ERROR = 8,
  TUNE = 9,  // autotuner parameter broadcast (tensor_sizes = [fusion_bytes,
             // cycle_time_us])
};

// A single collective announcement from one rank.
struct Request {
  RequestType type = RequestType::ALLREDUCE;
  int32_t rank = 0;
  std::string name;
  DataType dtype = DataType::HVD_FLOAT32;
  std::vector<int64_t> shape;
  int32_t root_rank = -1;
  ReduceOp reduce_op = ReduceOp::SUM;
  double prescale = 1.0;
  double postscale = 1.0;
  int32_t process_set_id = 0;
  int32_t device = CPU_DEVICE_ID;
  std::vector<int64_t> splits;  // alltoall send splits (set-local order)
  // grouped ops (reference: GroupTable): all tensors sharing group_key must
  // negotiate and execute as ONE fused unit of group_size tensors.
  std::string group_key;
  int32_t group_size = 0;

  void Serialize(std::string& out) const;
  static Request Deserialize(const char*& p, const char* end);
};

struct RequestList {
  std::vector<Request> requests;
  bool shutdown = false;

  void Serialize(std::string& out) const;
  static RequestList Deserialize(const char* p, const char* end);
};

// The coordinator's instruction: execute this (possibly fused) op now.
struct Response {
  ResponseType type = ResponseType::ALLREDUCE;
  std::vector<std::string> names;
  DataType dtype = DataType::HVD_FLOAT32;
  ReduceOp reduce_op = ReduceOp::SUM;
  int32_t process_set_id = 0;
  int32_t device = CPU_DEVICE_ID;
  std::string error_msg;
  // ALLGATHER / ALLTOALL / REDUCESCATTER: first-dimension size contributed by
  // each rank (per fused tensor, rank-major: names.size() * set_size entries).
  std::vector<int64_t> tensor_sizes;
  // BROADCAST/JOIN bookkeeping
  int32_t root_rank = -1;
  // For JOIN: the last rank to join (returned to the user).
  int32_t last_joined_rank = -1;
  // Shapes of the tensors (flattened [ndims..., -1 terminator] per name) so a
  // joined rank can allocate zero substitutes.  Populated for ALLREDUCE.
  std::vector<int64_t> tensor_shapes;
  // grouped-op identity (see Request)
  std::string group_key;
  int32_t group_size = 0;

  void Serialize(std::string& out) const;
  static Response Deserialize(const char*& p, const char* end);
};

struct ResponseList {
  std::vector<Response> responses;
  bool shutdown = false;

  void Serialize(std::string& out) const;
  static ResponseList Deserialize(const char* p, const char* end);
};

}  // namespace hvd
