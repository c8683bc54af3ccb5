"""Network geometry and flat-parameter layout.

Must mirror csrc/lenet_dims.h exactly (the tests assert the extension's
exported offsets match these).  Layout == the checkpoint format: per layer
weight then bias, in Layer-ctor order (c1, s1, f) — SURVEY.md §5.4.
"""
IN_H = 28
IN_W = 28
IN_PIX = IN_H * IN_W            # 784

C1_CH = 6
C1_K = 5
C1_H = 24
C1_W = 24
C1_PIX = C1_H * C1_W            # 576
C1_OUT = C1_CH * C1_PIX         # 3456
C1_WSZ = C1_CH * C1_K * C1_K    # 150

S1_K = 4
S1_H = 6
S1_W = 6
S1_PIX = S1_H * S1_W            # 36
S1_OUT = C1_CH * S1_PIX         # 216
S1_WSZ = S1_K * S1_K            # 16

FC_IN = S1_OUT                  # 216
FC_OUT = 10
FC_WSZ = FC_OUT * FC_IN         # 2160

OFF_C1W = 0
OFF_C1B = OFF_C1W + C1_WSZ      # 150
OFF_S1W = OFF_C1B + C1_CH       # 156
OFF_S1B = OFF_S1W + S1_WSZ      # 172
OFF_FW = OFF_S1B + 1            # 173
OFF_FB = OFF_FW + FC_WSZ        # 2333
N_PARAMS = OFF_FB + FC_OUT      # 2343

REF_DT = 0.1                    # Sequential/layer.h:12
REF_THRESHOLD = 1.0e-2          # Sequential/layer.h:13
