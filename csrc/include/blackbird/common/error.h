// Error domains + codes for the whole framework.
// Capability parity with the reference's domain-partitioned error system
// (reference: include/blackbird/common/error/error_codes.h:15-79), designed
// fresh: adds a GPU domain for the HIP/HBM tier and keeps codes the data
// model actually uses (the reference's CXL backend referenced codes that do
// not exist — we do not reproduce that defect).
#pragma once

#include <cstdint>
#include <string_view>

namespace blackbird {

enum class ErrorCode : int32_t {
  OK = 0,

  // SYSTEM (1000)
  INTERNAL_ERROR = 1000,
  NOT_IMPLEMENTED = 1001,
  TIMEOUT = 1002,
  SHUTTING_DOWN = 1003,
  INVALID_STATE = 1004,

  // STORAGE (2000)
  NO_SPACE = 2000,
  POOL_NOT_FOUND = 2001,
  INVALID_OFFSET = 2002,
  RESERVATION_NOT_FOUND = 2003,
  RESERVATION_EXPIRED = 2004,
  SHARD_NOT_FOUND = 2005,
  BACKEND_INIT_FAILED = 2006,
  SIZE_MISMATCH = 2007,
  ALLOCATION_FAILED = 2008,

  // NETWORK (3000)
  CONNECT_FAILED = 3000,
  SEND_FAILED = 3001,
  RECV_FAILED = 3002,
  PROTOCOL_ERROR = 3003,
  ENDPOINT_INVALID = 3004,
  RPC_FAILED = 3005,
  CONNECTION_CLOSED = 3006,

  // COORDINATION (4000)
  COORD_UNAVAILABLE = 4000,
  KEY_NOT_FOUND = 4001,
  LEASE_EXPIRED = 4002,
  WATCH_FAILED = 4003,
  NOT_LEADER = 4004,
  CAS_FAILED = 4005,

  // DATA (5000)
  OBJECT_NOT_FOUND = 5000,
  OBJECT_EXISTS = 5001,
  OBJECT_EXPIRED = 5002,
  CHECKSUM_MISMATCH = 5003,
  OBJECT_NOT_COMMITTED = 5004,
  NO_PLACEMENT = 5005,
  SESSION_STALE = 5006,  // put-session token no longer matches server state

  // CLIENT (6000)
  INVALID_ARGUMENT = 6000,
  TRANSFER_FAILED = 6001,
  NOT_CONNECTED = 6002,

  // CONFIG (7000)
  CONFIG_PARSE_ERROR = 7000,
  CONFIG_INVALID = 7001,

  // GPU (8000) — MI355X tier, no analog in the reference
  HIP_ERROR = 8000,
  NO_GPU = 8001,
  IPC_OPEN_FAILED = 8002,
  KERNEL_FAILED = 8003,
  RCCL_ERROR = 8004,
};

enum class ErrorDomain : int32_t {
  NONE = 0,
  SYSTEM = 1,
  STORAGE = 2,
  NETWORK = 3,
  COORDINATION = 4,
  DATA = 5,
  CLIENT = 6,
  CONFIG = 7,
  GPU = 8,
};

constexpr ErrorDomain domain_of(ErrorCode c) {
  auto v = static_cast<int32_t>(c);
  if (v == 0) return ErrorDomain::NONE;
  return static_cast<ErrorDomain>(v / 1000);
}

std::string_view to_string(ErrorCode c);
std::string_view to_string(ErrorDomain d);

}  // namespace blackbird
