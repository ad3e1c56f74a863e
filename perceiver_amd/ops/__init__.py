from perceiver_amd.ops.attention import eager_attention, scaled_dot_attention
