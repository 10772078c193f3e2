// Argument block for the persistent decode engine (csrc/engine.hip),
// shared with the torch-extension binding (csrc/ext.hip).  All pointers
// are device pointers; granule buffers are u64 {tag,payload} arrays that
// the host zeroes (hipMemsetAsync / tensor.zero_) before EVERY launch.
#pragma once
#include <hip/hip_runtime.h>

struct EngineArgs {
  const char* stream;                // per-CU packed weight streams, concat
  unsigned long long stream_per_cu;  // bytes (identical per CU)
  int n_slots_total;                 // ceil(stream_per_cu / 16384)
  int nslot_ring;                    // LDS ring depth in 16 KiB slots
  int n_layers, hidden, inter, nqkv, hq, hkv;
  int log2g;                         // log2(group_size)
  float rms_eps, attn_scale;
  const float* inv_freq;             // [d/2] fp32
  const unsigned short *kc, *vc;     // [L][hkv][tmax][d] bf16
  unsigned long long kv_layer_stride;  // u16 elements per layer
  int tmax, nch_max;                 // nch_max = ceil(tmax/256)
  const int* T_dev;                  // seq len AFTER this token (state[1])
  const long long* tok;              // [1] input token id
  const unsigned short* embed;       // [vocab, hidden] bf16
  const unsigned short* final_w;     // [hidden] final-norm weight
  unsigned short* xout;              // [hidden] normalized final hidden
  void *g_qkv, *g_attn, *g_o, *g_h, *g_down, *g_scr;
  void* tmo;                         // u32 give-up code (0 = healthy)
  unsigned long long* prof;          // optional [n_layers*8] s_memtime, CU0
  unsigned debug;                    // race-bisect flags (LFMQ_ENGINE_DEBUG):
                                     // 1 = conservative loader (vmcnt(0)
                                     //     publish per slot), 2 = arrival
                                     //     barrier around every phase,
                                     // 4 = agent-acquire fence after sweeps
};

// record byte size in the weight stream v2 (padded to 16 B); must match
// lfmq/models/engine.py::_rec_bytes: lane-major nibbles + fp32 (s,sz) pairs
inline int lfmq_engine_rec_bytes(int k, int G) {
  return (k / 2 + G * 8 + 15) & ~15;
}

extern "C" hipError_t lfmq_engine_step(const EngineArgs* args, int d,
                                       hipStream_t stream);
