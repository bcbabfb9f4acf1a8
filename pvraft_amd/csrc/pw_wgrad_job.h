// Job descriptor shared between the pw_wgrad batched kernel (device) and
// the bindings (host builds the array).  Plain POD, 8-byte aligned.
#pragma once

struct PwWgradJob {
  const void *dy;  // (B, Co, S) bf16
  const void *x;   // (B, Ci, S) bf16
  float *dw;       // (Co, Ci) fp32, ACCUMULATED into
  float *dbias;    // (Co) fp32 or null
  int Co, Ci, B, schunks;
  long S;
};
