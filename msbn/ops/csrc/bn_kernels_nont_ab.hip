// A/B build variant: identical kernels with nontemporal accesses disabled
// (plain cached loads/stores).  Own translation unit so the object file does
// not collide with the product build's bn_kernels.hip object.
#define MSBN_DISABLE_NT 1
#include "bn_kernels.hip"
