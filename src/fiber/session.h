// brpc_amd: fiber session — versioned 64-bit handle + internal lock +
// error channel. Capability parity with reference bthread/id.h
// (bthread_id_*): one session per RPC call ("correlation id"); concurrent
// events (response / timeout / socket failure) serialize through the lock,
// retries use a version range so stale events are recognizable.
#pragma once

#include <stdint.h>

namespace bam {

typedef uint64_t SessionId;  // 0 = invalid

// Called when session_error() delivers an error. Runs with the session
// LOCKED; the handler must eventually session_unlock() or
// session_unlock_and_destroy() the id. Return value is ignored.
typedef int (*SessionOnError)(SessionId id, void* data, int error_code);

// Creates a session with `range` versions (>=1); retries bump the active
// version within the range (all versions join/destroy together).
int session_create(SessionId* id, void* data, SessionOnError on_error, int range = 1);

// Diagnostics for the /ids builtin page (≙ reference builtin/ids_service):
// process-lifetime created/destroyed counts (active = difference).
void session_stats(int64_t* created, int64_t* destroyed);

// Locks the session. Returns 0 and fills *data; EINVAL if destroyed/stale.
int session_lock(SessionId id, void** data);
int session_unlock(SessionId id);
// Unlocks, invalidates the id and wakes joiners.
int session_unlock_and_destroy(SessionId id);

// Blocks until the session is destroyed (0 if already gone).
int session_join(SessionId id);

// Delivers an error event: locks and invokes on_error (which must unlock
// or destroy). EINVAL if the session is gone.
int session_error(SessionId id, int error_code);

// --- version-range helpers (retry / backup-request bookkeeping) ---
// Must hold the lock. Returns the currently-active version slot [0,range).
int session_active_slot(SessionId id);
// Must hold the lock. Advances the active slot (new retry attempt).
int session_bump_slot(SessionId id);
// True if `id`'s version equals the session's active version (a response
// for an old retry is stale). Must hold the lock on any version of the id.
bool session_is_current(SessionId id);
// The id with the active version (to stamp into a retried request).
SessionId session_current_id(SessionId id);

bool session_exists(SessionId id);

}  // namespace bam
