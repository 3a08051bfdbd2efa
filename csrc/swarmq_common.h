// swarmq_common.h — device data layout shared between the HIP kernels,
// the DeviceQueue host runtime, and (by convention) the Python side
// (swarmdb_amd/runtime/engine.py REC_DTYPE and status/type codes).
//
// MI355X-native replacement for the reference's Kafka tier (reference
// "swarmdb/ main.py":192-207, 334-345, 466-484 — librdkafka produce /
// consumer-group poll / partitioned log). The log lives in HBM3E as a
// slot ring; per-agent inbox rings + read cursors replace consumer
// groups; delivery acks are enqueue-kernel completion.
#pragma once

#include <cstdint>

namespace swarmq {

using u8 = uint8_t;
using u16 = uint16_t;
using u32 = uint32_t;
using u64 = uint64_t;
using i64 = int64_t;

// ---- codes (must match swarmdb_amd/runtime/engine.py) ----
constexpr u32 BROADCAST = 0xFFFFFFFFu;
constexpr u32 NO_BITMAP = 0xFFFFFFFFu;
constexpr u32 EPOCH_NONE = 0xFFFFFFFFu; // bitmap-epoch slot "never allocated"

constexpr u32 ST_PENDING = 0;
constexpr u32 ST_DELIVERED = 1;
constexpr u32 ST_READ = 2;
constexpr u32 ST_PROCESSED = 3;
constexpr u32 ST_FAILED = 4;
constexpr u32 ST_DELETED = 5;
constexpr int N_STATUS = 6;
constexpr int N_TYPES = 7;

constexpr u8 VIS_ALL = 0;
constexpr u8 VIS_BITMAP = 1; // filtered at dequeue (broadcast semantics)
constexpr u8 VIS_GROUP = 2;  // filtered at fan-out (group semantics: only
                             // member inboxes get the entry)

// ---- message record: 48 B, == numpy REC_DTYPE bit-for-bit ----
struct __attribute__((aligned(16))) Rec {
  u32 sender;
  u32 receiver;     // BROADCAST for broadcast
  u8 type;          // MessageType code
  u8 priority;      // 0..3
  u8 vis_mode;      // VIS_ALL / VIS_BITMAP
  u8 flags;
  u32 token_count;
  double timestamp; // epoch seconds
  u64 payload_off;  // staging offset on enqueue; slot offset device-side
  u32 payload_len;
  u32 bitmap;       // visibility bitmap index or NO_BITMAP
  u32 content_len;  // content prefix of the payload (search window)
  u32 bitmap_epoch; // allocation counter of the referenced bitmap; the
                    // pool is a ring, so dequeue/fan-out verify
                    // bitmap_epochs[bitmap] == bitmap_epoch and treat a
                    // recycled entry as NOT visible (never misdelivered)
};
static_assert(sizeof(Rec) == 48, "Rec layout must match REC_DTYPE");

// ---- device queue geometry (immutable after construction) ----
struct QueueGeom {
  u32 num_slots;      // slot ring capacity (power of two NOT required)
  u32 slot_bytes;     // payload bytes per slot
  u32 max_agents;
  u32 inbox_capacity; // entries per agent inbox ring
  u32 num_bitmaps;    // visibility bitmap pool depth
  u32 bitmap_words;   // max_agents / 64
  u32 num_backends;
  u32 recv_window;    // max entries examined per dequeue (LDS sort cap)
};

// dequeue sort window (LDS u64 keys): 4096 * 8 B = 32 KiB of the
// 160 KiB/CU LDS
constexpr u32 RECV_WINDOW = 4096;

// priority dequeue key: (3 - priority) in the top bits so higher
// priority sorts first, seq in the low 48 bits preserves FIFO within a
// priority level
__host__ __device__ inline u64 prio_key(u32 priority, u64 seq) {
  return ((u64)(3u - priority) << 48) | (seq & 0xFFFFFFFFFFFFull);
}
constexpr u64 KEY_INVALID = ~0ull;

} // namespace swarmq
