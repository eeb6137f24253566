#include "fiber/session.h"

#include <errno.h>

#include <atomic>
#include <mutex>
#include <vector>

#include "base/logging.h"
#include "base/resource_pool.h"
#include "fiber/butex.h"

namespace bam {

namespace {

// Lock model (parity with reference bthread/id.cpp): a session has an
// internal lock serializing events. session_error() while the session is
// locked ENQUEUES the error; the unlocker re-dispatches pending errors
// before truly releasing — so an error handler that triggers another error
// (e.g. a retry whose Write fails synchronously) never deadlocks.
struct SessionMeta {
  std::mutex mu;  // guards the fields below (short critical sections only)
  bool locked = false;
  std::vector<std::pair<SessionId, int>> pending_errors;
  std::atomic<int>* lock_butex = nullptr;  // wait word for contended lock
  std::atomic<int>* join_word = nullptr;   // holds (int)first_ver; bumped at destroy
  std::atomic<uint32_t> first_ver{1};      // monotonic across reuse
  uint32_t range = 1;
  uint32_t active_slot = 0;
  void* data = nullptr;
  SessionOnError on_error = nullptr;
};

inline ResourceId rid_of(SessionId id) { return (uint32_t)(id & 0xffffffffu) - 1; }
inline uint32_t ver_of(SessionId id) { return (uint32_t)(id >> 32); }

SessionMeta* meta_of(SessionId id) {
  if (id == 0) return nullptr;
  return address_resource<SessionMeta>(rid_of(id));
}

inline bool valid_ver_locked(const SessionMeta* m, uint32_t ver) {
  uint32_t fv = m->first_ver.load(std::memory_order_relaxed);
  return ver >= fv && ver < fv + m->range;
}

}  // namespace

static std::atomic<int64_t> g_sessions_created{0};
static std::atomic<int64_t> g_sessions_destroyed{0};

void session_stats(int64_t* created, int64_t* destroyed) {
  *created = g_sessions_created.load(std::memory_order_relaxed);
  *destroyed = g_sessions_destroyed.load(std::memory_order_relaxed);
}

int session_create(SessionId* id, void* data, SessionOnError on_error, int range) {
  g_sessions_created.fetch_add(1, std::memory_order_relaxed);
  if (range < 1) range = 1;
  ResourceId rid;
  SessionMeta* m = get_resource<SessionMeta>(&rid);
  if (m == nullptr) return ENOMEM;
  if (m->lock_butex == nullptr) {
    m->lock_butex = butex_create();
    m->lock_butex->store(0, std::memory_order_relaxed);
    m->join_word = butex_create();
  }
  {
    std::lock_guard<std::mutex> lk(m->mu);
    m->range = (uint32_t)range;
    m->active_slot = 0;
    m->data = data;
    m->on_error = on_error;
    m->locked = false;
    m->pending_errors.clear();
    uint32_t fv = m->first_ver.load(std::memory_order_relaxed);
    m->join_word->store((int)fv, std::memory_order_release);
    *id = ((uint64_t)fv << 32) | (rid + 1);
  }
  return 0;
}

int session_lock(SessionId id, void** data) {
  SessionMeta* m = meta_of(id);
  if (m == nullptr) return EINVAL;
  for (;;) {
    int wait_val;
    {
      std::lock_guard<std::mutex> lk(m->mu);
      if (!valid_ver_locked(m, ver_of(id))) return EINVAL;
      if (!m->locked) {
        m->locked = true;
        if (data != nullptr) *data = m->data;
        return 0;
      }
      wait_val = m->lock_butex->load(std::memory_order_relaxed);
    }
    butex_wait(m->lock_butex, wait_val, nullptr);
  }
}

namespace {

// Releases the lock, draining pending errors first. destroy_after: bump
// version + wake joiners + recycle.
int unlock_impl(SessionId id, bool destroy) {
  SessionMeta* m = meta_of(id);
  if (m == nullptr) return EINVAL;
  for (;;) {
    SessionId err_id = 0;
    int err_code = 0;
    SessionOnError handler = nullptr;
    void* data = nullptr;
    {
      std::lock_guard<std::mutex> lk(m->mu);
      if (!destroy && !m->pending_errors.empty() && valid_ver_locked(m, ver_of(id))) {
        err_id = m->pending_errors.front().first;
        err_code = m->pending_errors.front().second;
        m->pending_errors.erase(m->pending_errors.begin());
        handler = m->on_error;
        data = m->data;
        // stay locked; dispatch below
      } else {
        if (destroy) {
          uint32_t fv = m->first_ver.load(std::memory_order_relaxed);
          m->first_ver.store(fv + m->range, std::memory_order_release);
          m->pending_errors.clear();
          m->join_word->store((int)(fv + m->range), std::memory_order_release);
        }
        m->locked = false;
        m->lock_butex->fetch_add(1, std::memory_order_release);
      }
    }
    if (handler == nullptr && err_id == 0) {
      butex_wake_all(m->lock_butex);
      if (destroy) {
        butex_wake_all(m->join_word);
        return_resource<SessionMeta>(rid_of(id));
      }
      return 0;
    }
    // Dispatch a pending error while still holding the logical lock. The
    // handler must unlock or destroy; it may re-enter unlock_impl, which
    // will drain further pending errors — so we return here.
    if (handler != nullptr) {
      handler(err_id, data, err_code);
      return 0;
    }
    session_unlock_and_destroy(err_id);  // no handler: default = destroy
    return 0;
  }
}

}  // namespace

int session_unlock(SessionId id) { return unlock_impl(id, false); }

int session_unlock_and_destroy(SessionId id) {
  int rc = unlock_impl(id, true);
  if (rc == 0) g_sessions_destroyed.fetch_add(1, std::memory_order_relaxed);
  return rc;
}

int session_join(SessionId id) {
  SessionMeta* m = meta_of(id);
  if (m == nullptr) return 0;
  const uint32_t ver = ver_of(id);
  for (;;) {
    uint32_t fv = m->first_ver.load(std::memory_order_acquire);
    if (!(ver >= fv && ver < fv + m->range)) return 0;
    butex_wait(m->join_word, (int)fv, nullptr);
  }
}

int session_error(SessionId id, int error_code) {
  SessionMeta* m = meta_of(id);
  if (m == nullptr) return EINVAL;
  SessionOnError handler = nullptr;
  void* data = nullptr;
  {
    std::lock_guard<std::mutex> lk(m->mu);
    if (!valid_ver_locked(m, ver_of(id))) return EINVAL;
    if (m->locked) {
      m->pending_errors.emplace_back(id, error_code);
      return 0;
    }
    m->locked = true;
    handler = m->on_error;
    data = m->data;
  }
  if (handler != nullptr) {
    handler(id, data, error_code);  // must unlock or destroy
    return 0;
  }
  return session_unlock_and_destroy(id);
}

int session_active_slot(SessionId id) {
  SessionMeta* m = meta_of(id);
  return m != nullptr ? (int)m->active_slot : -1;
}

int session_bump_slot(SessionId id) {
  SessionMeta* m = meta_of(id);
  if (m == nullptr) return EINVAL;
  std::lock_guard<std::mutex> lk(m->mu);
  if (m->active_slot + 1 >= m->range) return ERANGE;
  m->active_slot += 1;
  return 0;
}

bool session_is_current(SessionId id) {
  SessionMeta* m = meta_of(id);
  if (m == nullptr) return false;
  uint32_t fv = m->first_ver.load(std::memory_order_acquire);
  return ver_of(id) == fv + m->active_slot;
}

SessionId session_current_id(SessionId id) {
  SessionMeta* m = meta_of(id);
  if (m == nullptr) return 0;
  uint32_t fv = m->first_ver.load(std::memory_order_acquire);
  return ((uint64_t)(fv + m->active_slot) << 32) | (id & 0xffffffffu);
}

bool session_exists(SessionId id) {
  SessionMeta* m = meta_of(id);
  if (m == nullptr) return false;
  std::lock_guard<std::mutex> lk(m->mu);
  return valid_ver_locked(m, ver_of(id));
}

}  // namespace bam
