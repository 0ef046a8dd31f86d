// Shared C ABI for the gfx950 health probe library (libcroprobe.so).
// Consumers: probe.hip (implementation), agent/croagent.cpp (native CLI),
// nodeops/probe.py (ctypes — keep CroProbeResult field order in sync there).
#pragma once

#ifdef __cplusplus
extern "C" {
#endif

struct CroProbeResult {
  int ok;
  int mfma_f32_exact;  // 1 = bitwise match vs host fmaf chain
  double hbm_gbps;     // achieved copy bandwidth (read+write bytes)
  double bf16_tflops;  // dense bf16 MFMA issue rate
  long long vram_total;
  long long vram_free;
  double t_setup_ms;  // host wall per probe section (attach-latency budget)
  double t_mfma_ms;
  double t_bw_ms;
  double t_bf16_ms;
  char gcn_arch[64];
  char msg[256];
};

int cro_probe_run(int device, struct CroProbeResult* out);
int cro_probe_mfma_f32(int device, const float* A, const float* B, float* D, int K);
void* cro_probe_alloc(int device, long long bytes);
void cro_probe_free(void* p);
int cro_probe_device_count(void);
int cro_probe_pci_bus_id(int device, char* buf, int len);

#ifdef __cplusplus
}
#endif
