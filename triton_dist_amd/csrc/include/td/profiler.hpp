// Intra-kernel profiler: device-side wallclock stamps into a ring buffer,
// exported host-side as a chrome/perfetto trace.
//
// Capability parity with Triton-distributed tools/profiler/{language.py:
// 38-145, context.py:50-76} (device Profiler.create/record + ProfilerBuffer
// + trace viewer — behavior only). gfx950: s_memrealtime is a ~100 MHz
// constant-rate counter, coherent across CUs.
#pragma once
#include "td/device.hpp"

namespace td {

struct KProf {
  unsigned long long *buf;  // [capacity][4]: block, tag, t0, t1
  unsigned *cursor;         // device atomic slot counter
  unsigned capacity;
};

// Record one interval from thread 0 of the block (call uniformly).
TD_DEV void kprof_record(const KProf &p, unsigned tag,
                         unsigned long long t0, unsigned long long t1) {
  if (!p.buf) return;
  if (threadIdx.x == 0) {
    unsigned slot = atomic_add<Scope::Gpu>(p.cursor, 1u);
    if (slot < p.capacity) {
      unsigned long long *rec = p.buf + (size_t)slot * 4;
      rec[0] = blockIdx.x + ((unsigned long long)blockIdx.y << 32);
      rec[1] = tag;
      rec[2] = t0;
      rec[3] = t1;
    }
  }
}

}  // namespace td
