// RPC method ids shared by servers and clients.
#pragma once

#include <cstdint>

namespace blackbird::rpc::methods {

// keystone control plane (parity with the reference's 14-handler surface,
// rpc_service.cpp:369-382, + batch ops)
constexpr uint16_t PING = 1;
constexpr uint16_t OBJECT_EXISTS = 2;
constexpr uint16_t GET_WORKERS = 3;
constexpr uint16_t PUT_START = 4;
constexpr uint16_t PUT_COMPLETE = 5;
constexpr uint16_t PUT_CANCEL = 6;
constexpr uint16_t REMOVE_OBJECT = 7;
constexpr uint16_t REMOVE_ALL_OBJECTS = 8;
constexpr uint16_t GET_WORKERS_INFO = 9;
constexpr uint16_t GET_MEMORY_POOLS = 10;
constexpr uint16_t REMOVE_WORKER = 11;
constexpr uint16_t GET_CLUSTER_STATS = 12;
constexpr uint16_t GET_VIEW_VERSION = 13;
constexpr uint16_t BATCH_PUT_START = 14;
constexpr uint16_t BATCH_PUT_COMPLETE = 15;
constexpr uint16_t BATCH_PUT_CANCEL = 16;
constexpr uint16_t BATCH_GET_WORKERS = 17;
constexpr uint16_t BATCH_OBJECT_EXISTS = 18;
constexpr uint16_t BATCH_REMOVE = 19;
// compact indexed batch protocol (v2): pool table + fixed-width placements,
// single-shard copies only (max_workers_per_copy == 1)
constexpr uint16_t BATCH_PUT_START2 = 20;
constexpr uint16_t BATCH_GET_WORKERS2 = 21;
constexpr uint16_t LIST_OBJECTS = 22;  // prefix scan (operator tooling)
// admin maintenance triggers (operator tooling: bbctl scrub/repair/compact)
constexpr uint16_t ADMIN_SCRUB = 23;
constexpr uint16_t ADMIN_REPAIR = 24;
constexpr uint16_t ADMIN_COMPACT = 25;

// sessionful upsert protocol: a BATCH_PUT_START2 response can carry a
// session token; steady-state re-puts of the same batch then cost two tiny
// RPCs (start: 8-byte token; commit: token + per-item digests) instead of
// re-sending every key both ways
constexpr uint16_t BATCH_UPSERT_START = 26;
constexpr uint16_t BATCH_COMMIT_TOKEN = 27;

// worker data plane (TCP fallback path; SHM/HIP-IPC paths bypass RPC)
constexpr uint16_t DATA_WRITE = 200;
constexpr uint16_t DATA_READ = 201;
constexpr uint16_t DATA_BATCH_WRITE = 202;
constexpr uint16_t DATA_BATCH_READ = 203;
constexpr uint16_t DATA_CHECKSUM = 204;
constexpr uint16_t DATA_STATS = 205;
constexpr uint16_t DATA_PULL = 206;  // tier migration: pull ranges into a local pool

}  // namespace blackbird::rpc::methods
