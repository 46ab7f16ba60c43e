#include "blackbird/common/error.h"

namespace blackbird {

std::string_view to_string(ErrorCode c) {
  switch (c) {
    case ErrorCode::OK: return "OK";
    case ErrorCode::INTERNAL_ERROR: return "INTERNAL_ERROR";
    case ErrorCode::NOT_IMPLEMENTED: return "NOT_IMPLEMENTED";
    case ErrorCode::TIMEOUT: return "TIMEOUT";
    case ErrorCode::SHUTTING_DOWN: return "SHUTTING_DOWN";
    case ErrorCode::INVALID_STATE: return "INVALID_STATE";
    case ErrorCode::NO_SPACE: return "NO_SPACE";
    case ErrorCode::POOL_NOT_FOUND: return "POOL_NOT_FOUND";
    case ErrorCode::INVALID_OFFSET: return "INVALID_OFFSET";
    case ErrorCode::RESERVATION_NOT_FOUND: return "RESERVATION_NOT_FOUND";
    case ErrorCode::RESERVATION_EXPIRED: return "RESERVATION_EXPIRED";
    case ErrorCode::SHARD_NOT_FOUND: return "SHARD_NOT_FOUND";
    case ErrorCode::BACKEND_INIT_FAILED: return "BACKEND_INIT_FAILED";
    case ErrorCode::SIZE_MISMATCH: return "SIZE_MISMATCH";
    case ErrorCode::ALLOCATION_FAILED: return "ALLOCATION_FAILED";
    case ErrorCode::CONNECT_FAILED: return "CONNECT_FAILED";
    case ErrorCode::SEND_FAILED: return "SEND_FAILED";
    case ErrorCode::RECV_FAILED: return "RECV_FAILED";
    case ErrorCode::PROTOCOL_ERROR: return "PROTOCOL_ERROR";
    case ErrorCode::ENDPOINT_INVALID: return "ENDPOINT_INVALID";
    case ErrorCode::RPC_FAILED: return "RPC_FAILED";
    case ErrorCode::CONNECTION_CLOSED: return "CONNECTION_CLOSED";
    case ErrorCode::COORD_UNAVAILABLE: return "COORD_UNAVAILABLE";
    case ErrorCode::KEY_NOT_FOUND: return "KEY_NOT_FOUND";
    case ErrorCode::LEASE_EXPIRED: return "LEASE_EXPIRED";
    case ErrorCode::WATCH_FAILED: return "WATCH_FAILED";
    case ErrorCode::NOT_LEADER: return "NOT_LEADER";
    case ErrorCode::CAS_FAILED: return "CAS_FAILED";
    case ErrorCode::OBJECT_NOT_FOUND: return "OBJECT_NOT_FOUND";
    case ErrorCode::OBJECT_EXISTS: return "OBJECT_EXISTS";
    case ErrorCode::OBJECT_EXPIRED: return "OBJECT_EXPIRED";
    case ErrorCode::CHECKSUM_MISMATCH: return "CHECKSUM_MISMATCH";
    case ErrorCode::OBJECT_NOT_COMMITTED: return "OBJECT_NOT_COMMITTED";
    case ErrorCode::NO_PLACEMENT: return "NO_PLACEMENT";
    case ErrorCode::SESSION_STALE: return "SESSION_STALE";
    case ErrorCode::INVALID_ARGUMENT: return "INVALID_ARGUMENT";
    case ErrorCode::TRANSFER_FAILED: return "TRANSFER_FAILED";
    case ErrorCode::NOT_CONNECTED: return "NOT_CONNECTED";
    case ErrorCode::CONFIG_PARSE_ERROR: return "CONFIG_PARSE_ERROR";
    case ErrorCode::CONFIG_INVALID: return "CONFIG_INVALID";
    case ErrorCode::HIP_ERROR: return "HIP_ERROR";
    case ErrorCode::NO_GPU: return "NO_GPU";
    case ErrorCode::IPC_OPEN_FAILED: return "IPC_OPEN_FAILED";
    case ErrorCode::KERNEL_FAILED: return "KERNEL_FAILED";
    case ErrorCode::RCCL_ERROR: return "RCCL_ERROR";
  }
  return "UNKNOWN";
}

std::string_view to_string(ErrorDomain d) {
  switch (d) {
    case ErrorDomain::NONE: return "NONE";
    case ErrorDomain::SYSTEM: return "SYSTEM";
    case ErrorDomain::STORAGE: return "STORAGE";
    case ErrorDomain::NETWORK: return "NETWORK";
    case ErrorDomain::COORDINATION: return "COORDINATION";
    case ErrorDomain::DATA: return "DATA";
    case ErrorDomain::CLIENT: return "CLIENT";
    case ErrorDomain::CONFIG: return "CONFIG";
    case ErrorDomain::GPU: return "GPU";
  }
  return "UNKNOWN";
}

}  // namespace blackbird
