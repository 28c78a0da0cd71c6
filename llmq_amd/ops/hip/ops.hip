// Umbrella translation unit: all kernels + torch bindings compiled together
// (template instantiation happens at the launch sites in bindings.cpp).
#include "elementwise.hip"
#include "kvcache.hip"
#include "attention.hip"
#include "skinny_gemm.hip"
#include "sampling.hip"
#include "bindings.cpp"
