// Single translation unit for the midgpt_amd HIP extension: kernel sources
// + bindings. One TU keeps template instantiation simple and lets hipcc see
// every kernel definition from the binding code.
#include "norms.hip"
#include "ce.hip"
#include "adamw.hip"
#include "attention.hip"
#include "embedding.hip"
#include "gelu.hip"
#include "probe.hip"
#include "bindings.cpp"
