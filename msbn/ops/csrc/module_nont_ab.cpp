// A/B build variant translation unit (see bn_kernels_nont_ab.hip).
#include "module.cpp"
