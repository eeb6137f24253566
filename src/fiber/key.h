// brpc_amd: fiber-local storage (≙ reference bthread_key / keytable).
// Keys are process-global versioned slots; each fiber lazily allocates a
// KeyTable destroyed (with destructors) at fiber exit. Non-fiber threads
// fall back to plain thread-local storage.
#pragma once

#include <stdint.h>

namespace bam {

struct fiber_key_t {
  uint32_t index = 0;
  uint32_t version = 0;
};

int fiber_key_create(fiber_key_t* key, void (*dtor)(void*));
// Deleting a key invalidates it everywhere; destructors no longer run for
// values stored under it.
int fiber_key_delete(fiber_key_t key);
int fiber_setspecific(fiber_key_t key, void* data);
void* fiber_getspecific(fiber_key_t key);

// internal: run destructors + free the current context's keytable (called
// at fiber exit).
void destroy_current_keytable();

}  // namespace bam
