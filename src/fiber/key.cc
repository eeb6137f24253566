#include "fiber/key.h"

#include <mutex>
#include <vector>

#include "fiber/scheduler.h"

namespace bam {

namespace {

const uint32_t kMaxKeys = 4096;

struct KeyInfo {
  uint32_t version = 0;  // odd = in use
  void (*dtor)(void*) = nullptr;
};

struct KeyRegistry {
  std::mutex mu;
  std::vector<KeyInfo> keys;
  std::vector<uint32_t> free_list;
};

KeyRegistry& registry() {
  static KeyRegistry* r = new KeyRegistry;
  return *r;
}

struct Slot {
  uint32_t version = 0;
  void* data = nullptr;
};

struct KeyTable {
  std::vector<Slot> slots;
};

// Non-fiber threads use a plain TLS table.
thread_local KeyTable tls_fallback_table;

KeyTable* current_table(bool create) {
  TaskGroup* g = current_task_group();
  if (g == nullptr || g->cur() == nullptr || g->cur()->is_main) {
    return &tls_fallback_table;
  }
  FiberMeta* m = g->cur();
  if (m->keytable == nullptr && create) m->keytable = new KeyTable;
  return (KeyTable*)m->keytable;
}

}  // namespace

int fiber_key_create(fiber_key_t* key, void (*dtor)(void*)) {
  KeyRegistry& r = registry();
  std::lock_guard<std::mutex> lk(r.mu);
  uint32_t idx;
  if (!r.free_list.empty()) {
    idx = r.free_list.back();
    r.free_list.pop_back();
  } else {
    if (r.keys.size() >= kMaxKeys) return -1;
    idx = (uint32_t)r.keys.size();
    r.keys.push_back(KeyInfo());
  }
  r.keys[idx].version += 1;  // becomes odd = live
  r.keys[idx].dtor = dtor;
  key->index = idx;
  key->version = r.keys[idx].version;
  return 0;
}

int fiber_key_delete(fiber_key_t key) {
  KeyRegistry& r = registry();
  std::lock_guard<std::mutex> lk(r.mu);
  if (key.index >= r.keys.size() || r.keys[key.index].version != key.version) return -1;
  r.keys[key.index].version += 1;  // even = free
  r.keys[key.index].dtor = nullptr;
  r.free_list.push_back(key.index);
  return 0;
}

int fiber_setspecific(fiber_key_t key, void* data) {
  KeyTable* t = current_table(true);
  if (t == nullptr) return -1;
  if (t->slots.size() <= key.index) t->slots.resize(key.index + 1);
  t->slots[key.index].version = key.version;
  t->slots[key.index].data = data;
  return 0;
}

void* fiber_getspecific(fiber_key_t key) {
  KeyTable* t = current_table(false);
  if (t == nullptr || t->slots.size() <= key.index) return nullptr;
  const Slot& s = t->slots[key.index];
  return s.version == key.version ? s.data : nullptr;
}

void destroy_current_keytable() {
  TaskGroup* g = current_task_group();
  if (g == nullptr || g->cur() == nullptr) return;
  FiberMeta* m = g->cur();
  KeyTable* t = (KeyTable*)m->keytable;
  if (t == nullptr) return;
  m->keytable = nullptr;
  KeyRegistry& r = registry();
  for (uint32_t i = 0; i < t->slots.size(); ++i) {
    Slot& s = t->slots[i];
    if (s.data == nullptr) continue;
    void (*dtor)(void*) = nullptr;
    {
      std::lock_guard<std::mutex> lk(r.mu);
      if (i < r.keys.size() && r.keys[i].version == s.version) dtor = r.keys[i].dtor;
    }
    if (dtor != nullptr) dtor(s.data);
  }
  delete t;
}

}  // namespace bam
