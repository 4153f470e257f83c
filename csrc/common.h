// horovod_amd core — common types.
//
// MI355X-native re-design of the framework-agnostic abstractions in the
// reference (horovod/common/common.h:195-397).  Unlike the reference we are
// PyTorch-only, so the tensor abstraction is at::Tensor directly instead of
// a virtual Tensor/OpContext hierarchy.
#pragma once

#include <cstdint>
#include <functional>
#include <memory>
#include <string>
#include <vector>

#include <ATen/ATen.h>

namespace hvd {

constexpr int CPU_DEVICE_ID = -1;

// ---------------------------------------------------------------------------
// Status (reference: common/common.h Status {OK, UNKNOWN_ERROR, ...})
// ---------------------------------------------------------------------------
enum class StatusType : uint8_t {
  OK = 0,
  UNKNOWN_ERROR = 1,
  PRECONDITION_ERROR = 2,
  ABORTED = 3,
  INVALID_ARGUMENT = 4,
  IN_PROGRESS = 5,
};

struct Status {
  StatusType type = StatusType::OK;
  std::string reason;

  Status() = default;
  Status(StatusType t, std::string r) : type(t), reason(std::move(r)) {}

  static Status OK() { return Status(); }
  static Status UnknownError(std::string msg) {
    return Status(StatusType::UNKNOWN_ERROR, std::move(msg));
  }
  static Status PreconditionError(std::string msg) {
    return Status(StatusType::PRECONDITION_ERROR, std::move(msg));
  }
  static Status Aborted(std::string msg) {
    return Status(StatusType::ABORTED, std::move(msg));
  }
  static Status InvalidArgument(std::string msg) {
    return Status(StatusType::INVALID_ARGUMENT, std::move(msg));
  }
  static Status InProgress() { return Status(StatusType::IN_PROGRESS, ""); }

  bool ok() const { return type == StatusType::OK; }
  bool in_progress() const { return type == StatusType::IN_PROGRESS; }
};

// ---------------------------------------------------------------------------
// Data types on the wire (reference: common/message.h DataType)
// ---------------------------------------------------------------------------
enum class DataType : uint8_t {
  HVD_UINT8 = 0,
  HVD_INT8 = 1,
  HVD_INT32 = 2,
  HVD_INT64 = 3,
  HVD_FLOAT16 = 4,
  HVD_FLOAT32 = 5,
  HVD_FLOAT64 = 6,
  HVD_BOOL = 7,
  HVD_BFLOAT16 = 8,
  HVD_UINT16 = 9,
  HVD_INT16 = 10,
};

DataType DataTypeFromTorch(at::ScalarType t);
at::ScalarType DataTypeToTorch(DataType t);
size_t DataTypeSize(DataType t);
const char* DataTypeName(DataType t);

// Reduction op (reference: common/message.h ReduceOp)
enum class ReduceOp : uint8_t {
  AVERAGE = 0,
  SUM = 1,
  ADASUM = 2,
  MIN = 3,
  MAX = 4,
  PRODUCT = 5,
};

// ---------------------------------------------------------------------------
// TensorTableEntry: one pending op on this rank.
// (reference: common/common.h:320-370 TensorTableEntry)
// ---------------------------------------------------------------------------
struct TensorTableEntry;
// Completion callback: receives the final entry so exec-time-allocated
// outputs (allgather/alltoall) reach the caller.
using StatusCallback = std::function<void(const Status&, TensorTableEntry&)>;

struct TensorTableEntry {
  std::string name;
  at::Tensor tensor;       // input (undefined for joined-rank zero substitution)
  at::Tensor output;       // output (allocated by the op if undefined)
  int device = CPU_DEVICE_ID;
  int root_rank = -1;      // broadcast
  ReduceOp reduce_op = ReduceOp::SUM;
  double prescale = 1.0;
  double postscale = 1.0;
  int32_t process_set_id = 0;
  std::vector<int64_t> splits;         // alltoall send splits
  at::Tensor received_splits;          // alltoall recv splits output (int64 cpu)
  int32_t join_result = -1;            // JOIN: last rank to join
  StatusCallback callback;
  // event handle recorded on the submitting torch stream; the hvd stream
  // waits on it before touching tensor memory (GPU only).
  uintptr_t ready_event = 0;
};

// Round a fusion-buffer offset up to a 64-element boundary so every fused
// tensor's slice starts 16B-aligned for any dtype (reference:
// common/common.h FUSION_BUFFER_ATOMIC_UNIT=64).
constexpr int64_t kFusionAlignElems = 64;
inline int64_t AlignedElems(int64_t n) {
  return ((n + kFusionAlignElems - 1) / kFusionAlignElems) * kFusionAlignElems;
}

}  // namespace hvd
