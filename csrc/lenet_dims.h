// Compile-time geometry of the LeNet-5-class network shared by the CPU
// reference ops and the gfx950 HIP kernels.
//
// Network (behavioral parity with the reference survey, SURVEY.md §0.1;
// reference topology defined at Sequential/Main.cpp:17-20):
//   input  28x28x1
//   conv1  6 filters 5x5 valid stride 1 -> 6x24x24, sigmoid
//   pool   ONE shared trainable 4x4 kernel stride 4 (+1 scalar bias)
//          -> 6x6x6, sigmoid   (a weighted-sum downsample, NOT max pool)
//   fc     216 -> 10, sigmoid
#pragma once

namespace pcnn {

constexpr int IN_H = 28;
constexpr int IN_W = 28;
constexpr int IN_PIX = IN_H * IN_W;  // 784

constexpr int C1_CH = 6;   // conv1 output channels
constexpr int C1_K = 5;    // conv1 kernel size
constexpr int C1_H = 24;   // conv1 output height (valid conv)
constexpr int C1_W = 24;
constexpr int C1_PIX = C1_H * C1_W;            // 576
constexpr int C1_OUT = C1_CH * C1_PIX;         // 3456
constexpr int C1_WSZ = C1_CH * C1_K * C1_K;    // 150

constexpr int S1_K = 4;    // pool kernel size == stride (non-overlapping)
constexpr int S1_H = 6;
constexpr int S1_W = 6;
constexpr int S1_PIX = S1_H * S1_W;            // 36
constexpr int S1_OUT = C1_CH * S1_PIX;         // 216
constexpr int S1_WSZ = S1_K * S1_K;            // 16 (shared across channels)

constexpr int FC_IN = S1_OUT;                  // 216
constexpr int FC_OUT = 10;
constexpr int FC_WSZ = FC_OUT * FC_IN;         // 2160

// Flat parameter layout == the checkpoint format: per layer weight then bias,
// in Layer-ctor order (c1, s1, f).  SURVEY.md §5.4.
constexpr int OFF_C1W = 0;
constexpr int OFF_C1B = OFF_C1W + C1_WSZ;              // 150
constexpr int OFF_S1W = OFF_C1B + C1_CH;               // 156
constexpr int OFF_S1B = OFF_S1W + S1_WSZ;              // 172
constexpr int OFF_FW = OFF_S1B + 1;                    // 173
constexpr int OFF_FB = OFF_FW + FC_WSZ;                // 2333
constexpr int N_PARAMS = OFF_FB + FC_OUT;              // 2343

// Reference hyperparameters (Sequential/layer.h:12-13).
constexpr float REF_DT = 1.0e-1f;
constexpr float REF_THRESHOLD = 1.0e-2f;

}  // namespace pcnn
