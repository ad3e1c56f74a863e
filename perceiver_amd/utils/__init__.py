from perceiver_amd.utils.profiling import StepTimer, kernel_table, kernel_trace
