// adapm_amd core types.
//
// MI355X-native adaptive parameter manager: key->dense-vector store with
// intent-driven relocation/replication (capability parity with the reference
// AdaPM: /root/reference/include/ps/coloc_kv_worker.h, coloc_kv_server.h,
// coloc_kv_server_handle.h — re-designed for one-process-per-GPU with an
// HBM-resident slab and RCCL-over-xGMI sync rounds instead of ZeroMQ).
#pragma once

#include <cstdint>
#include <vector>

namespace adapm {

using Key = int64_t;
using Clock = int64_t;

// per-key metadata flags (host-side; cf. reference Parameter struct,
// coloc_kv_server_handle.h:121-152). Packed with the slab offset into
// ONE atomic int64 per key (see Server::meta_), so the worker metadata
// pass costs one cache miss per key, not two.
enum KeyFlags : uint8_t {
  F_PRESENT = 1,   // value allocated in local slab
  F_OWNER = 2,     // we hold the main copy
  F_STUB = 4,      // replica placeholder, no data received yet (version "-1")
  F_UPDATED = 8,   // replica has local (unsynced) updates
  F_HASREP = 32,   // owned key with >=1 granted replica: version bumps are
                   // observable (unreplicated keys skip the version_ touch)
};

// management techniques (reference postoffice.h management_techniques)
enum Techniques : int {
  TECH_ALL = 0,
  TECH_REPLICATION_ONLY = 1,
  TECH_RELOCATION_ONLY = 2,
};

// sync-message record codes. Every record is REC_I64 int64 fields
// (code, key, f0, f1, f2) followed (in the payload tensor, in record
// order) by `payload_len(code)` floats.
enum MsgCode : int64_t {
  // phase A (requests / deltas)
  M_DELTA = 0,     // replica->owner. f0=version, f1=delta flags, f2=origin rank. payload: len floats if (f1&D_HAS_PAYLOAD)
  M_PUSH_REQ = 1,  // f0=origin, f1=req_id, f2=hops. payload: len floats
  M_PULL_REQ = 2,  // f0=origin, f1=req_id, f2=out_index<<8|hops. no payload
  M_SET_REQ = 3,   // like push, overwrite semantics
  // phase B (responses) — may be sent by any rank to any rank
  M_REFRESH = 10,    // owner->replica holder. f0=version, f1=refresh flags. payload: len floats
  M_PULL_RESP = 11,  // f0=req_id, f1=out_index. payload: len floats
  M_PUSH_ACK = 12,   // f0=req_id, f1=count acked
  M_RESIDENCE = 13,  // ->manager. f0=new_owner, f1=relocation counter
  // Bulk records (uniform-length stores): the 5-word header is followed
  // IN THE META STREAM by extra int64 words (keys, and out-indices for
  // pulls). They collapse thousands of per-key records into one record +
  // one batched kernel on each side — the multi-GPU hot path.
  M_PULL_REQ_BULK = 4,   // hdr(code, nkeys, origin, req_id, hops) + keys + out_idx
  M_PUSH_REQ_BULK = 5,   // hdr(code, nkeys, origin, req_id, set<<32|hops) + keys. payload: nkeys rows
  M_PULL_RESP_BULK = 14, // hdr(code, nkeys, req_id, 0, 0) + keys + out_idx. payload: nkeys rows
  M_NACK = 15,           // hop-limit give-up: f0=req_id, f1=count. The origin's
                         // ticket fails loudly instead of hanging/acking a drop.
  // Bulk forms of the replica-sync flow (uniform-length stores): the
  // per-key M_DELTA/M_REFRESH/M_RESIDENCE records dominate round host
  // cost at relocation churn (record parse + response-object per key).
  M_DELTA_BULK = 16,     // hdr(code, nkeys, origin, has_payload, 0) + keys
                         //  + info[nk] ((version<<8)|dflags). payload: nk rows iff has_payload
  M_REFRESH_BULK = 17,   // hdr(code, nkeys, 0, 0, 0) + keys + vers[nk]
                         //  + fc[nk] ((reloc_ctr<<8)|rflags). payload: nk rows
  M_RESIDENCE_BULK = 18, // hdr(code, nkeys, 0, 0, 0) + keys + oc[nk] ((ctr<<8)|owner)
};

enum DeltaFlags : int64_t {
  D_HAS_PAYLOAD = 1,  // replica accumulated updates since last sync
  D_DROPPING = 2,     // replica holder is dropping the replica (intent expired)
  D_WANT_REFRESH = 4, // active intent: owner should send refreshed value
  D_NEW = 8,          // first announcement: requests replica setup / relocation
};

enum RefreshFlags : int64_t {
  R_RELOCATE = 1,  // receiver becomes the owner
  R_DROP_ACK = 2,  // owner acknowledges replica drop (no payload)
  R_NOOP = 4,      // value unchanged since reported version (no payload)
};

constexpr int REC_I64 = 5;  // int64 fields per record

constexpr int N_STRIPES = 16384;  // lock striping (reference handle: PS_BACKEND_NUM_LOCKS)

}  // namespace adapm
